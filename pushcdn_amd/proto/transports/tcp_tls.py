"""TCP+TLS transport (reference ``cdn-proto/src/connection/protocols/tcp_tls.rs``):
TCP with a TLS handshake against the fixed SNI name "espresso"; the server
presents a per-boot leaf cert signed by the CA (local testing CA by default).
The handshake happens in ``finalize`` so a slow handshake can't block the
accept loop (reference protocols/mod.rs:76-81).
"""

from __future__ import annotations

import asyncio
import ssl
from typing import Optional

from ...crypto import tls as tlslib
from ..errors import ConnectionError_
from ..limiter import Limiter
from .base import Connection, Listener, Protocol, UnfinalizedConnection
from .tcp import _set_nodelay, parse_endpoint


class TcpTlsUnfinalized(UnfinalizedConnection):
    """The TLS handshake already ran inside asyncio's per-connection task
    (start_server(ssl=...)), so the accept loop was never blocked by a slow
    handshake — the same guarantee the reference gets from its accept/
    finalize split (protocols/mod.rs:76-81); finalize just wires the actor
    tasks."""

    def __init__(self, reader, writer) -> None:
        self._reader, self._writer = reader, writer

    async def finalize(self, limiter: Limiter) -> Connection:
        _set_nodelay(self._writer)
        return Connection.from_streams(self._reader, self._writer, limiter)


class TcpTlsListener(Listener):
    def __init__(self, server: asyncio.AbstractServer, queue: "asyncio.Queue") -> None:
        self._server = server
        self._queue = queue

    async def accept(self) -> TcpTlsUnfinalized:
        return await self._queue.get()

    async def close(self) -> None:
        self._server.close()
        await self._server.wait_closed()

    @property
    def port(self) -> int:
        return self._server.sockets[0].getsockname()[1]


class TcpTls(Protocol):
    """Configure CA paths via class attributes (RunDef-style wiring) or rely
    on the process-local testing CA."""

    ca_cert_path: Optional[str] = None
    ca_key_path: Optional[str] = None

    @classmethod
    async def connect(cls, endpoint: str, use_local_authority: bool, limiter: Limiter) -> Connection:
        host, port = parse_endpoint(endpoint)
        ctx = tlslib.client_context(use_local_authority, cls.ca_cert_path)
        try:
            reader, writer = await asyncio.open_connection(
                host, port, ssl=ctx, server_hostname=tlslib.CERT_NAME
            )
        except (OSError, ssl.SSLError) as e:
            raise ConnectionError_(f"failed to connect to {endpoint}: {e}") from e
        return Connection.from_streams(reader, writer, limiter)

    @classmethod
    async def bind(cls, endpoint: str, certificate=None, key=None) -> TcpTlsListener:
        host, port = parse_endpoint(endpoint)
        ctx = tlslib.server_context(cls.ca_cert_path, cls.ca_key_path)
        queue: "asyncio.Queue" = asyncio.Queue()

        async def on_conn(reader, writer) -> None:
            await queue.put(TcpTlsUnfinalized(reader, writer))

        server = await asyncio.start_server(on_conn, host or "0.0.0.0", port, ssl=ctx)
        return TcpTlsListener(server, queue)
