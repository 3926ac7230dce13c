"""HBM message pool — the device-side analog of the reference limiter
(``cdn-proto/src/connection/limiter/pool.rs:28-111``): ingest staging is
carved out of ONE pre-sized HBM arena with a byte budget, refcounted
release, and allocation backpressure, instead of ad-hoc per-tick tensor
allocations.

Semantics matched to the reference:
  - bounded total bytes: an alloc beyond the budget WAITS (``pool.rs:60-68``
    acquires n semaphore permits == n bytes) — the backpressure that stalls
    the socket reader when consumers fall behind;
  - refcounted release: :class:`PoolBytes` mirrors ``Allocation<T>``
    (``pool.rs:85-111``): clones share the bytes, the LAST drop releases
    them and records the allocation lifetime into the ``latency``
    histogram (``pool.rs:44-52``);
  - no persistence: the arena is scratch, like every broker structure
    (SURVEY §5.4).

Layout: a ring (bump pointer + FIFO reclamation).  Tick staging is
allocated and released in arrival order, so ring reclamation never
fragments; an allocation that cannot fit in the tail gap wraps to offset 0
(the gap is skipped, accounted, and reclaimed with its predecessor).
Sizing: 288 GB HBM3E per MI355X — the default 1 GiB matches the
reference's CLI default (broker.rs:71-73); production brokers can budget
hundreds of GiB.
"""

from __future__ import annotations

import asyncio
import time
from collections import deque
from typing import Optional

import torch

from ..utils.metrics import LATENCY


class HbmPoolError(Exception):
    pass


class PoolBytes:
    """A refcounted slice of the arena (reference ``Allocation<T>``)."""

    __slots__ = ("pool", "offset", "length", "span", "tensor", "_refs", "_t0")

    def __init__(self, pool: "HbmMessagePool", offset: int, length: int,
                 span: int) -> None:
        self.pool = pool
        self.offset = offset
        self.length = length
        self.span = span  # bytes reclaimed on release (incl. wrap gap)
        self.tensor = pool.arena[offset:offset + length]
        self._refs = 1
        self._t0 = time.perf_counter()

    def clone(self) -> "PoolBytes":
        self._refs += 1
        return self

    def drop(self) -> None:
        if self._refs <= 0:
            return
        self._refs -= 1
        if self._refs == 0:
            LATENCY.observe(time.perf_counter() - self._t0)
            self.pool._release(self)


class HbmMessagePool:
    """Byte-budgeted device arena with FIFO (ring) reclamation."""

    def __init__(self, capacity: int, device: str = "cuda:0") -> None:
        assert capacity > 0
        self.capacity = capacity
        self.device = torch.device(device)
        self.arena = torch.empty(capacity, dtype=torch.uint8, device=self.device)
        self._head = 0          # next alloc offset
        self._used = 0          # live bytes (incl. wrap gaps)
        self._live: "deque[PoolBytes]" = deque()  # FIFO of outstanding allocs
        self._cond = asyncio.Condition()
        self._loop = None  # owning loop, learned on first loop-side release

    @property
    def used_bytes(self) -> int:
        return self._used

    @property
    def free_bytes(self) -> int:
        return self.capacity - self._used

    def try_alloc(self, n: int) -> Optional[PoolBytes]:
        """Non-blocking allocation; None when the budget is exhausted (the
        sync-context path — the engine tick thread cannot await)."""
        if n > self.capacity:
            raise HbmPoolError(f"allocation of {n} exceeds pool capacity {self.capacity}")
        span = n
        offset = self._head
        if offset + n > self.capacity:
            # wrap: the tail gap is dead until this allocation releases
            span = n + (self.capacity - offset)
            offset = 0
        if self._used + span > self.capacity:
            return None
        self._used += span
        self._head = offset + n
        b = PoolBytes(self, offset, n, span)
        self._live.append(b)
        return b

    async def alloc(self, n: int) -> PoolBytes:
        """Blocking allocation: waits for releases when the pool is full —
        the reference's semaphore acquire (``pool.rs:60-68``)."""
        self._loop = asyncio.get_running_loop()
        async with self._cond:
            while True:
                b = self.try_alloc(n)
                if b is not None:
                    return b
                await self._cond.wait()

    def _release(self, b: PoolBytes) -> None:
        b.length = -1  # mark dead; reclaimed when it reaches the FIFO head
        freed = False
        while self._live and self._live[0].length == -1:
            head = self._live.popleft()
            self._used -= head.span
            freed = True
        if freed:
            # wake blocked allocators; safe from any thread (e.g. a drop on
            # the mesh executor thread) via the owning loop
            try:
                loop = asyncio.get_running_loop()
            except RuntimeError:
                loop = self._loop
                if loop is not None and not loop.is_closed():
                    loop.call_soon_threadsafe(
                        lambda: loop.create_task(self._notify()))
                return
            self._loop = loop
            loop.create_task(self._notify())

    async def _notify(self) -> None:
        async with self._cond:
            self._cond.notify_all()
