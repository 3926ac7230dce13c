"""Native TCP transport: frames are read/written by the C++ epoll pump
(csrc/net/pump.h) on its own thread; asyncio only sees WHOLE messages in
batches through an eventfd it watches with ``loop.add_reader``.  This is
the MI355X-native analog of the reference's tokio connection core
(``cdn-proto/src/connection/protocols/mod.rs:139-217``): per-connection
reader/writer actors, 4-byte-BE length framing (``:311-394``), max-size
guard, soft-close flush.

Accept/connect (the rare path) stay on asyncio's non-blocking socket
helpers; the connected fd is then handed to the pump, which owns it from
that point on.  Backpressure is byte-based: senders await while the
connection's unflushed outbox exceeds the high-water mark.
"""

from __future__ import annotations

import asyncio
import socket
from typing import Dict

from ..errors import ConnectionError_
from ..limiter import Bytes, Limiter
from ...utils.metrics import BYTES_RECV, BYTES_SENT
from .base import Connection, Listener, Protocol, UnfinalizedConnection
from .tcp import parse_endpoint

SEND_HWM_BYTES = 64 << 20  # per-connection unflushed-outbox high-water mark
RECV_BATCH = 1024


def _get_core():
    from ...ops.build import build_core

    return build_core()


class _PumpManager:
    """C++ pump shard(s) + one eventfd watcher each, per event loop.

    PUSHCDN_PUMP_SHARDS (default 1) sets how many epoll threads serve this
    process's connections, round-robin assigned: a broker pushing multiple
    GB/s of egress spreads the send() work across shards; client processes
    keep the single-thread default."""

    _by_loop: "Dict[int, _PumpManager]" = {}

    def __init__(self, loop: asyncio.AbstractEventLoop) -> None:
        import os

        n = max(1, int(os.environ.get("PUSHCDN_PUMP_SHARDS", "1")))
        core = _get_core()
        self.pumps = [core.Pump() for _ in range(n)]
        self.loop = loop
        # conns keyed per shard: connection ids are per-pump
        self.conns: "list[Dict[int, PumpConnection]]" = [{} for _ in range(n)]
        self._next_shard = 0
        for i, p in enumerate(self.pumps):
            loop.add_reader(p.notify_fd(), self._on_notify, i)

    # single-shard compatibility accessor
    @property
    def pump(self):
        return self.pumps[0]

    @classmethod
    def current(cls) -> "_PumpManager":
        loop = asyncio.get_running_loop()
        key = id(loop)
        mgr = cls._by_loop.get(key)
        if mgr is None or mgr.loop.is_closed():
            # reap pumps whose loops are gone (tests create many loops)
            for k, old in list(cls._by_loop.items()):
                if old.loop.is_closed():
                    for p in old.pumps:
                        p.stop()
                    del cls._by_loop[k]
            mgr = cls(loop)
            cls._by_loop[key] = mgr
        return mgr

    def _on_notify(self, shard: int) -> None:
        for cid in self.pumps[shard].poll_dirty():
            conn = self.conns[shard].get(cid)
            if conn is not None:
                conn._pump_dirty()

    def attach(self, sock: socket.socket):
        """-> (shard index, conn id); the pump owns the fd now."""
        fd = sock.detach()
        shard = self._next_shard
        self._next_shard = (shard + 1) % len(self.pumps)
        return shard, self.pumps[shard].add(fd)


class PumpConnection(Connection):
    """Connection whose data path lives in the C++ pump."""

    def __init__(self, mgr: _PumpManager, shard: int, cid: int,
                 limiter: Limiter) -> None:
        self._mgr = mgr
        self._pump = mgr.pumps[shard]
        self._shard = shard
        self._cid = cid
        self._limiter = limiter
        self._closed = False
        self._dead = False
        self._ingest = False
        self._count_mode = False
        self._recv_q: "asyncio.Queue[Bytes]" = asyncio.Queue()
        self._wakeup = asyncio.Event()
        mgr.conns[shard][cid] = self

    def _poll_inbox(self) -> bool:
        """Drain the C++ inbox into the asyncio queue (no signaling)."""
        frames, closed = self._pump.recv_batch(self._cid, RECV_BATCH)
        total = 0
        for f in frames:
            total += len(f)
            self._recv_q.put_nowait(Bytes(f))
        if total:
            BYTES_RECV.inc(total)
        if closed:
            self._dead = True
        return bool(frames) or closed

    # called from the manager's eventfd callback
    def _pump_dirty(self) -> None:
        if self._ingest or self._count_mode:
            self._wakeup.set()
            return
        if self._poll_inbox():
            self._wakeup.set()

    def enable_count_mode(self) -> None:
        """Counting-subscriber path (benchmarks/relays): frames stay in
        C++; recv_drain returns (count, bytes, last_frame) per call."""
        self._count_mode = True

    async def recv_drain(self):
        """(count, bytes, last_frame) — awaits until >=1 frame arrived;
        raises when the peer is gone and drained."""
        while True:
            n, nbytes, last, closed = self._pump.recv_drain(self._cid)
            if n:
                return n, nbytes, last
            if closed or self._dead or self._closed:
                self._dead = True
                raise ConnectionError_("connection reader closed")
            self._wakeup.clear()
            n, nbytes, last, closed = self._pump.recv_drain(self._cid)
            if n:
                return n, nbytes, last
            if closed or self._dead or self._closed:
                self._dead = True
                raise ConnectionError_("connection reader closed")
            await self._wakeup.wait()

    def enable_ingest(self) -> None:
        """Switch this connection to the C++ ingest path: the pump
        accumulates classified frames in one contiguous buffer and Python
        pulls a whole tick's worth per call (recv_ingest_batch) — the
        per-message interpreter round-trip disappears from the broker's
        user plane."""
        self._ingest = True
        self._pump.set_ingest(self._cid)

    async def recv_ingest_batch(self):
        """(blob, end_offsets, discs, topics_off, topics_cnt, recip_off,
        recip_len) — awaits until at least one frame arrived; raises when
        the peer is gone and everything is drained."""
        while True:
            blob, offs, disc, toff, tcnt, roff, rlen, closed = \
                self._pump.recv_ingest(self._cid)
            if blob:
                return blob, offs, disc, toff, tcnt, roff, rlen
            if closed or self._dead or self._closed:
                self._dead = True
                raise ConnectionError_("connection reader closed")
            self._wakeup.clear()
            # re-check: a frame may have raced the notify
            blob, offs, disc, toff, tcnt, roff, rlen, closed = \
                self._pump.recv_ingest(self._cid)
            if blob:
                return blob, offs, disc, toff, tcnt, roff, rlen
            if closed or self._dead or self._closed:
                self._dead = True
                raise ConnectionError_("connection reader closed")
            await self._wakeup.wait()

    async def send_message_raw(self, raw: Bytes) -> None:
        size = len(raw.data)
        try:
            ok = self._pump.send(self._cid, raw.data)
        finally:
            raw.drop()
        if not ok:
            raise ConnectionError_("connection writer closed")
        BYTES_SENT.inc(size)
        while self._pump.send_backlog(self._cid) > SEND_HWM_BYTES:
            await asyncio.sleep(0.001)

    async def recv_message_raw(self) -> Bytes:
        while True:
            if not self._recv_q.empty():
                return self._recv_q.get_nowait()
            if self._dead:
                raise ConnectionError_("connection reader closed")
            self._wakeup.clear()
            # drain anything that raced the notify callback (e.g. frames
            # that landed before this connection registered)
            self._poll_inbox()
            if self._recv_q.empty() and not self._dead:
                await self._wakeup.wait()

    def pump_handle(self):
        """(pump, conn id) for the batched tick drain (Pump.send_rings_batch):
        all connections of one event loop share one pump, so a whole tick's
        egress goes out in ONE C++ call."""
        if self._closed or self._dead:
            raise ConnectionError_("connection writer closed")
        return self._pump, self._cid

    def send_ring_records(self, ring: bytes, wpos: int) -> int:
        """Egress fast path for the GPU broker drain: hand a drained ring
        (16 B {len,seq} headers + wire payloads, 16-aligned records) to the
        pump, which parses and enqueues every frame in C++ — one Python
        call per (user, tick) instead of one per delivery."""
        if self._closed or self._dead:
            raise ConnectionError_("connection writer closed")
        n, payload_bytes = self._pump.send_ring(self._cid, ring, wpos)
        if payload_bytes:
            BYTES_SENT.inc(payload_bytes)
        return n

    async def soft_close(self) -> None:
        if self._closed:
            return
        self._closed = True
        self._pump.soft_close(self._cid)
        self._release()

    def close(self) -> None:
        if not self._closed:
            self._closed = True
            self._pump.hard_close(self._cid)
        self._release()

    def _release(self) -> None:
        self._mgr.conns[self._shard].pop(self._cid, None)
        self._pump.forget(self._cid)
        self._dead = True
        self._wakeup.set()


class TcpNativeUnfinalized(UnfinalizedConnection):
    def __init__(self, sock: socket.socket) -> None:
        self._sock = sock

    async def finalize(self, limiter: Limiter) -> Connection:
        mgr = _PumpManager.current()
        shard, cid = mgr.attach(self._sock)
        return PumpConnection(mgr, shard, cid, limiter)


class TcpNativeListener(Listener):
    def __init__(self, sock: socket.socket) -> None:
        self._sock = sock

    async def accept(self) -> TcpNativeUnfinalized:
        loop = asyncio.get_running_loop()
        conn, _addr = await loop.sock_accept(self._sock)
        return TcpNativeUnfinalized(conn)

    async def close(self) -> None:
        self._sock.close()

    @property
    def port(self) -> int:
        return self._sock.getsockname()[1]


class TcpNative(Protocol):
    @classmethod
    async def connect(cls, endpoint: str, use_local_authority: bool,
                      limiter: Limiter) -> Connection:
        host, port = parse_endpoint(endpoint)
        loop = asyncio.get_running_loop()
        sock = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
        sock.setblocking(False)
        try:
            await loop.sock_connect(sock, (host or "127.0.0.1", port))
        except OSError as e:
            sock.close()
            raise ConnectionError_(f"failed to connect to {endpoint}: {e}") from e
        mgr = _PumpManager.current()
        shard, cid = mgr.attach(sock)
        return PumpConnection(mgr, shard, cid, limiter)

    @classmethod
    async def bind(cls, endpoint: str, certificate=None, key=None) -> TcpNativeListener:
        host, port = parse_endpoint(endpoint)
        sock = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
        sock.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        sock.setblocking(False)
        sock.bind((host or "0.0.0.0", port))
        sock.listen(1024)
        return TcpNativeListener(sock)
