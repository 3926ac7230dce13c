"""In-process memory transport for deterministic tests
(reference ``cdn-proto/src/connection/protocols/memory.rs``): a global
endpoint registry plays the network; connect/accept hand each side a duplex
pair of asyncio streams.
"""

from __future__ import annotations

import asyncio
from typing import Dict, Tuple

from ..errors import ConnectionError_
from ..limiter import Bytes, Limiter
from ..message import Message, deserialize, serialize
from .base import Connection, Listener, Protocol, UnfinalizedConnection

# endpoint -> queue of (client_side_pair, server_side_pair)
_LISTENERS: Dict[str, "asyncio.Queue"] = {}


class _DuplexEnd:
    """One end of an in-memory duplex: a recv queue and a peer's recv queue."""

    def __init__(self, inbox: "asyncio.Queue", outbox: "asyncio.Queue") -> None:
        self.inbox = inbox
        self.outbox = outbox
        self.closed = False


class MemoryConnection(Connection):
    """Queue-backed Connection: skips the socket framing entirely but keeps
    identical send/recv semantics (the reference Memory protocol also skips
    real framing by moving whole messages over a duplex stream)."""

    def __init__(self, end: _DuplexEnd, limiter: Limiter) -> None:  # noqa: super
        self._end = end
        self._limiter = limiter
        self._closed = False

    async def send_message_raw(self, raw: Bytes) -> None:
        if self._end.closed:
            raw.drop()
            raise ConnectionError_("memory connection closed")
        await self._end.outbox.put(raw.data)
        raw.drop()

    async def send_message(self, message: Message) -> None:
        await self.send_message_raw(Bytes(serialize(message)))

    async def recv_message_raw(self) -> Bytes:
        if self._end.closed:
            raise ConnectionError_("memory connection closed")
        item = await self._end.inbox.get()
        if item is None:
            self._end.closed = True
            raise ConnectionError_("memory connection closed by peer")
        permit = await self._limiter.allocate_message_bytes(len(item))
        return Bytes(item, permit)

    async def recv_message(self) -> Message:
        raw = await self.recv_message_raw()
        try:
            return deserialize(raw.data)
        finally:
            raw.drop()

    async def soft_close(self) -> None:
        await self._end.outbox.put(None)
        self._end.closed = True

    def close(self) -> None:
        self._end.closed = True
        try:
            self._end.outbox.put_nowait(None)
        except asyncio.QueueFull:
            pass


class MemoryUnfinalized(UnfinalizedConnection):
    def __init__(self, end: _DuplexEnd) -> None:
        self._end = end

    async def finalize(self, limiter: Limiter) -> Connection:
        return MemoryConnection(self._end, limiter)


class MemoryListener(Listener):
    def __init__(self, endpoint: str, queue: "asyncio.Queue") -> None:
        self._endpoint = endpoint
        self._queue = queue

    async def accept(self) -> MemoryUnfinalized:
        end = await self._queue.get()
        return MemoryUnfinalized(end)

    async def close(self) -> None:
        _LISTENERS.pop(self._endpoint, None)


class Memory(Protocol):
    @classmethod
    async def connect(cls, endpoint: str, use_local_authority: bool, limiter: Limiter) -> Connection:
        q = _LISTENERS.get(endpoint)
        if q is None:
            raise ConnectionError_(f"no memory listener at {endpoint!r}")
        a_to_b: "asyncio.Queue" = asyncio.Queue()
        b_to_a: "asyncio.Queue" = asyncio.Queue()
        client = _DuplexEnd(inbox=b_to_a, outbox=a_to_b)
        server = _DuplexEnd(inbox=a_to_b, outbox=b_to_a)
        await q.put(server)
        return MemoryConnection(client, limiter)

    @classmethod
    async def bind(cls, endpoint: str, certificate=None, key=None) -> MemoryListener:
        q: "asyncio.Queue" = asyncio.Queue()
        _LISTENERS[endpoint] = q
        return MemoryListener(endpoint, q)


def gen_testing_connection_pair(limiter: Limiter) -> Tuple[MemoryConnection, MemoryConnection]:
    """Directly create a connected pair (reference memory.rs:189-200)."""
    a_to_b: "asyncio.Queue" = asyncio.Queue()
    b_to_a: "asyncio.Queue" = asyncio.Queue()
    a = MemoryConnection(_DuplexEnd(inbox=b_to_a, outbox=a_to_b), limiter)
    b = MemoryConnection(_DuplexEnd(inbox=a_to_b, outbox=b_to_a), limiter)
    return a, b
