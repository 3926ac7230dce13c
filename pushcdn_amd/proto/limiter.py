"""Flow control / memory accounting.

Host-side mirror of the reference limiter (``cdn-proto/src/connection/limiter/``):
a global byte budget (one permit == one byte, reference ``pool.rs:28-68``) plus
optional bounded per-connection message channels.  Receivers block until the
global pool has room, which backpressures the socket (``protocols/mod.rs:328``).

On the GPU data plane the same semantics are provided by the HBM message pool
(``pushcdn_amd.broker.hbm_pool.HbmMessagePool``): ingest staging comes out of
one pre-sized HBM arena with bounded total bytes, refcounted release, and
allocation backpressure.
"""

from __future__ import annotations

import asyncio
import time
from typing import Optional

from ..utils.metrics import LATENCY


class MemoryPool:
    """An async byte-budget semaphore: ``alloc(n)`` acquires n byte-permits."""

    def __init__(self, size: int) -> None:
        self._size = size
        self._available = size
        self._cond = asyncio.Condition()

    @property
    def size(self) -> int:
        return self._size

    async def alloc(self, n: int) -> "AllocationPermit":
        if n > self._size:
            raise ValueError(f"allocation of {n} exceeds pool size {self._size}")
        async with self._cond:
            while self._available < n:
                await self._cond.wait()
            self._available -= n
        return AllocationPermit(self, n)

    def _release(self, n: int) -> None:
        self._available += n
        # Wake waiters from whatever loop context we're in.
        async def _notify() -> None:
            async with self._cond:
                self._cond.notify_all()
        try:
            loop = asyncio.get_running_loop()
            loop.create_task(_notify())
        except RuntimeError:
            pass  # no running loop (teardown): nothing is waiting


class AllocationPermit:
    """Releases its bytes exactly once; logs allocation lifetime to LATENCY
    (the reference's in-broker residency histogram, ``pool.rs:44-52``)."""

    __slots__ = ("_pool", "_n", "_born", "_released")

    def __init__(self, pool: MemoryPool, n: int) -> None:
        self._pool = pool
        self._n = n
        self._born = time.monotonic()
        self._released = False

    def release(self) -> None:
        if not self._released:
            self._released = True
            LATENCY.observe(time.monotonic() - self._born)
            self._pool._release(self._n)

    def __del__(self) -> None:  # safety net; explicit release() preferred
        if not self._released:
            self.release()


class Bytes:
    """Refcounted message bytes + optional pool permit (reference ``Allocation<T>``,
    ``pool.rs:85-111``).  Shared zero-copy across fan-out recipients; the permit
    is released when the last holder drops."""

    __slots__ = ("data", "_permit", "_refs")

    def __init__(self, data: bytes, permit: Optional[AllocationPermit] = None) -> None:
        self.data = data
        self._permit = permit
        self._refs = 1

    def clone(self) -> "Bytes":
        self._refs += 1
        return self

    def drop(self) -> None:
        self._refs -= 1
        if self._refs == 0 and self._permit is not None:
            self._permit.release()
            self._permit = None

    def __len__(self) -> int:
        return len(self.data)


class Limiter:
    """Bundles the global pool and the per-connection channel bound
    (reference ``limiter/mod.rs:29-68``)."""

    def __init__(
        self,
        global_memory_pool_size: Optional[int] = None,
        connection_message_pool_size: Optional[int] = None,
    ) -> None:
        self.global_pool = (
            MemoryPool(global_memory_pool_size) if global_memory_pool_size else None
        )
        self.connection_message_pool_size = connection_message_pool_size

    async def allocate_message_bytes(self, n: int) -> Optional[AllocationPermit]:
        if self.global_pool is None:
            return None
        return await self.global_pool.alloc(n)
