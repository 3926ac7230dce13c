"""Transport abstraction: ``Protocol`` / ``Listener`` / ``Connection``.

Mirrors the reference trait surface (``cdn-proto/src/connection/protocols/mod.rs:40-306``):

- ``Protocol.connect(endpoint, use_local_authority, limiter) -> Connection``
- ``Protocol.bind(endpoint, cert, key) -> Listener``
- ``Listener.accept() -> UnfinalizedConnection`` (split from ``finalize`` so a
  slow TLS handshake can't block the accept loop, mod.rs:76-81)
- ``Connection.send_message / send_message_raw / recv_message /
  recv_message_raw / soft_close``

Framing: 4-byte big-endian length prefix + body (mod.rs:311-394), size capped
at ``MAX_MESSAGE_SIZE``; 5 s I/O timeouts on body-read and writes.  Each
connection runs one writer task and one reader task bridged to callers by
asyncio queues (bounded iff the limiter sets a per-connection pool size,
mod.rs:139-217).
"""

from __future__ import annotations

import asyncio
import struct
from abc import ABC, abstractmethod
from typing import Optional

from ..errors import ConnectionError_
from ..limiter import Bytes, Limiter
from ..message import MAX_MESSAGE_SIZE, Message, deserialize, serialize
from ...utils.metrics import BYTES_RECV, BYTES_SENT

IO_TIMEOUT_S = 5.0


async def read_length_delimited(reader: asyncio.StreamReader, limiter: Limiter) -> Bytes:
    """Read one framed message; blocks on the global byte pool (backpressure)."""
    try:
        header = await reader.readexactly(4)
    except (asyncio.IncompleteReadError, ConnectionResetError, OSError) as e:
        raise ConnectionError_(f"failed to read message size: {e}") from e
    (size,) = struct.unpack(">I", header)
    if size > MAX_MESSAGE_SIZE:
        raise ConnectionError_("message was too large")
    permit = await limiter.allocate_message_bytes(size)
    try:
        body = await asyncio.wait_for(reader.readexactly(size), IO_TIMEOUT_S)
    except (asyncio.TimeoutError, asyncio.IncompleteReadError, ConnectionResetError, OSError) as e:
        if permit:
            permit.release()
        raise ConnectionError_(f"failed to read message body: {e}") from e
    BYTES_RECV.inc(size)
    return Bytes(body, permit)


async def write_length_delimited(writer: asyncio.StreamWriter, message: Bytes) -> None:
    await write_frames(writer, [message])


async def write_frames(writer: asyncio.StreamWriter, messages) -> None:
    """Write a batch of frames with ONE flush — the writer task coalesces
    everything queued behind a connection (a burst of fan-out deliveries)
    into a single syscall-ish drain, like the reference's buffered tokio
    writer (protocols/mod.rs write half)."""
    total = 0
    try:
        bufs = []
        pending = 0
        for message in messages:
            data = message.data
            bufs.append(struct.pack(">I", len(data)))
            bufs.append(data)
            total += len(data)
            pending += len(data)
            # the reference's 5 s timeout is PER MESSAGE (mod.rs:368) —
            # applying it to a whole coalesced burst would kill connections
            # that are making healthy progress through a big backlog, so
            # drain (with the per-message bound) whenever the accumulated
            # chunk gets large
            if pending >= (4 << 20):
                writer.writelines(bufs)
                await asyncio.wait_for(writer.drain(), IO_TIMEOUT_S)
                bufs = []
                pending = 0
        if bufs:
            writer.writelines(bufs)
        await asyncio.wait_for(writer.drain(), IO_TIMEOUT_S)
    except (asyncio.TimeoutError, ConnectionResetError, OSError) as e:
        raise ConnectionError_(f"failed to send message: {e}") from e
    BYTES_SENT.inc(total)


class Connection:
    """A live connection with dedicated reader/writer tasks.

    ``from_streams`` spawns the two tasks; send/recv go through queues so a
    slow peer never blocks the caller beyond the queue bound
    (reference mod.rs:139-217).
    """

    def __init__(
        self,
        send_q: "asyncio.Queue[Optional[Bytes]]",
        recv_q: "asyncio.Queue[Bytes]",
        writer_task: asyncio.Task,
        reader_task: asyncio.Task,
        writer: asyncio.StreamWriter,
    ) -> None:
        self._send_q = send_q
        self._recv_q = recv_q
        self._writer_task = writer_task
        self._reader_task = reader_task
        self._writer = writer
        self._closed = False

    @classmethod
    def from_streams(
        cls,
        reader: asyncio.StreamReader,
        writer: asyncio.StreamWriter,
        limiter: Limiter,
    ) -> "Connection":
        qsize = limiter.connection_message_pool_size or 0
        send_q: "asyncio.Queue[Optional[Bytes]]" = asyncio.Queue(qsize)
        recv_q: "asyncio.Queue[Bytes]" = asyncio.Queue(qsize)

        async def writer_loop() -> None:
            try:
                while True:
                    item = await send_q.get()
                    closing = False
                    batch = []
                    while True:  # coalesce everything already queued
                        if item is None:  # soft close: flush then stop
                            closing = True
                            break
                        batch.append(item)
                        try:
                            item = send_q.get_nowait()
                        except asyncio.QueueEmpty:
                            break
                    if batch:
                        try:
                            await write_frames(writer, batch)
                        finally:
                            for b in batch:
                                b.drop()
                    if closing:
                        try:
                            await writer.drain()
                        except Exception:
                            pass
                        return
            except ConnectionError_:
                pass

        async def reader_loop() -> None:
            try:
                while True:
                    msg = await read_length_delimited(reader, limiter)
                    await recv_q.put(msg)
            except ConnectionError_:
                pass

        wt = asyncio.get_running_loop().create_task(writer_loop())
        rt = asyncio.get_running_loop().create_task(reader_loop())
        return cls(send_q, recv_q, wt, rt, writer)

    async def send_message_raw(self, raw: Bytes) -> None:
        if self._writer_task.done():
            raw.drop()
            raise ConnectionError_("connection writer closed")
        await self._send_q.put(raw)

    async def send_message(self, message: Message) -> None:
        await self.send_message_raw(Bytes(serialize(message)))

    async def recv_message_raw(self) -> Bytes:
        getter = asyncio.ensure_future(self._recv_q.get())
        done, _ = await asyncio.wait(
            {getter, self._reader_task}, return_when=asyncio.FIRST_COMPLETED
        )
        if getter in done:
            return getter.result()
        # reader died; drain anything already queued, else fail
        if not self._recv_q.empty():
            getter.cancel()
            return self._recv_q.get_nowait()
        getter.cancel()
        raise ConnectionError_("connection reader closed")

    async def recv_message(self) -> Message:
        raw = await self.recv_message_raw()
        try:
            return deserialize(raw.data)
        finally:
            raw.drop()

    async def soft_close(self) -> None:
        """Flush pending writes, then close (reference mod.rs:283-306)."""
        if self._closed:
            return
        self._closed = True
        try:
            await asyncio.wait_for(self._send_q.put(None), IO_TIMEOUT_S)
            await asyncio.wait_for(self._writer_task, IO_TIMEOUT_S)
        except asyncio.TimeoutError:
            pass
        self.close()

    def close(self) -> None:
        self._closed = True
        self._writer_task.cancel()
        self._reader_task.cancel()
        try:
            self._writer.close()
        except Exception:
            pass


class UnfinalizedConnection(ABC):
    @abstractmethod
    async def finalize(self, limiter: Limiter) -> Connection: ...


class Listener(ABC):
    @abstractmethod
    async def accept(self) -> UnfinalizedConnection: ...

    @abstractmethod
    async def close(self) -> None: ...


class Protocol(ABC):
    """Class-method factory, parameterized per ConnectionDef (reference def.rs:62-66)."""

    @classmethod
    @abstractmethod
    async def connect(
        cls, endpoint: str, use_local_authority: bool, limiter: Limiter
    ) -> Connection: ...

    @classmethod
    @abstractmethod
    async def bind(
        cls, endpoint: str, certificate: Optional[object], key: Optional[object]
    ) -> Listener: ...
