"""Plain TCP transport (reference ``cdn-proto/src/connection/protocols/tcp.rs``).

Sets TCP_NODELAY on both sides (tcp.rs:84,161).  Endpoint format "host:port".
"""

from __future__ import annotations

import asyncio
import socket

from ..errors import ConnectionError_, ParseError
from ..limiter import Limiter
from .base import Connection, Listener, Protocol, UnfinalizedConnection


def parse_endpoint(endpoint: str):
    host, sep, port = endpoint.rpartition(":")
    if not sep:
        raise ParseError(f"endpoint {endpoint!r} missing port")
    try:
        return host, int(port)
    except ValueError as e:
        raise ParseError(f"bad port in endpoint {endpoint!r}") from e


def _set_nodelay(writer: asyncio.StreamWriter) -> None:
    sock = writer.get_extra_info("socket")
    if sock is not None:
        try:
            sock.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
        except OSError:
            pass


class TcpUnfinalized(UnfinalizedConnection):
    def __init__(self, reader: asyncio.StreamReader, writer: asyncio.StreamWriter) -> None:
        self._reader, self._writer = reader, writer

    async def finalize(self, limiter: Limiter) -> Connection:
        _set_nodelay(self._writer)
        return Connection.from_streams(self._reader, self._writer, limiter)


class TcpListener(Listener):
    def __init__(self, server: asyncio.AbstractServer, queue: "asyncio.Queue[TcpUnfinalized]") -> None:
        self._server = server
        self._queue = queue

    async def accept(self) -> TcpUnfinalized:
        return await self._queue.get()

    async def close(self) -> None:
        self._server.close()
        await self._server.wait_closed()

    @property
    def port(self) -> int:
        return self._server.sockets[0].getsockname()[1]


class Tcp(Protocol):
    @classmethod
    async def connect(cls, endpoint: str, use_local_authority: bool, limiter: Limiter) -> Connection:
        host, port = parse_endpoint(endpoint)
        try:
            reader, writer = await asyncio.open_connection(host, port)
        except OSError as e:
            raise ConnectionError_(f"failed to connect to {endpoint}: {e}") from e
        _set_nodelay(writer)
        return Connection.from_streams(reader, writer, limiter)

    @classmethod
    async def bind(cls, endpoint: str, certificate=None, key=None) -> TcpListener:
        host, port = parse_endpoint(endpoint)
        queue: "asyncio.Queue[TcpUnfinalized]" = asyncio.Queue()

        async def on_conn(reader: asyncio.StreamReader, writer: asyncio.StreamWriter) -> None:
            await queue.put(TcpUnfinalized(reader, writer))

        server = await asyncio.start_server(on_conn, host or "0.0.0.0", port)
        return TcpListener(server, queue)
