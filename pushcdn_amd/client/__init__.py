"""The user-facing elastic client (reference ``cdn-client/src/lib.rs``).

Auto-reconnecting: marshal-auth -> broker-auth -> subscription replay.  Only
one reconnect runs at a time (the reference's 1-permit connecting-guard
semaphore, lib.rs:204-258); any send/recv failure drops the connection so
the next call reconnects (disconnect_on_error!, lib.rs:149-165); subscribe/
unsubscribe maintain the topic set under a lock so re-auth replays exactly
the current subscriptions (lib.rs:383-444).
"""

from __future__ import annotations

import asyncio
from dataclasses import dataclass, field
from typing import List, Optional, Sequence

from ..auth.user import UserAuth
from ..crypto import bls
from ..proto import message as m
from ..proto.errors import ConnectionError_
from ..proto.limiter import Limiter
from ..proto.transports.base import Connection

RECONNECT_ATTEMPT_TIMEOUT_S = 10.0
RECONNECT_BACKOFF_S = 2.0


@dataclass
class ClientConfig:
    endpoint: str                      # marshal endpoint
    keypair: bls.KeyPair
    subscribed_topics: List[int] = field(default_factory=list)
    use_local_authority: bool = True
    protocol: Optional[type] = None
    global_memory_pool_size: Optional[int] = None


class Client:
    def __init__(self, config: ClientConfig) -> None:
        from ..proto.transports.tcp import Tcp

        self.config = config
        self.protocol = config.protocol or Tcp
        self.limiter = Limiter(config.global_memory_pool_size)
        self._connection: Optional[Connection] = None
        self._topics = set(config.subscribed_topics)
        self._topics_lock = asyncio.Lock()
        self._connecting = asyncio.Semaphore(1)
        self._closed = False

    # ------------------------------ connection ------------------------------

    async def _connect_once(self) -> Connection:
        """marshal auth -> broker auth -> subscription replay
        (reference ClientRef::connect, lib.rs:79-126)."""
        marshal_conn = await self.protocol.connect(
            self.config.endpoint, self.config.use_local_authority, self.limiter
        )
        try:
            broker_endpoint, permit = await UserAuth.authenticate_with_marshal(
                marshal_conn, self.config.keypair
            )
        finally:
            await marshal_conn.soft_close()
        broker_conn = await self.protocol.connect(
            broker_endpoint, self.config.use_local_authority, self.limiter
        )
        try:
            async with self._topics_lock:
                topics = sorted(self._topics)
            await UserAuth.authenticate_with_broker(broker_conn, permit, topics)
        except BaseException:
            # auth failure / attempt-timeout cancellation after the broker
            # socket opened: a reconnecting client must not leak one
            # connection (and, for QUIC, one endpoint thread) per retry
            broker_conn.close()
            raise
        return broker_conn

    async def _get_connection(self) -> Connection:
        if self._closed:
            raise ConnectionError_("client is closed")
        if self._connection is not None:
            return self._connection
        async with self._connecting:
            if self._connection is not None:
                return self._connection
            while True:
                try:
                    conn = await asyncio.wait_for(
                        self._connect_once(), RECONNECT_ATTEMPT_TIMEOUT_S
                    )
                    self._connection = conn
                    return conn
                except asyncio.CancelledError:
                    raise
                except Exception:
                    if self._closed:
                        raise ConnectionError_("client is closed")
                    await asyncio.sleep(RECONNECT_BACKOFF_S)

    def _disconnect_on_error(self) -> None:
        if self._connection is not None:
            self._connection.close()
            self._connection = None

    async def ensure_initialized(self) -> None:
        """Eagerly connect (reference lib.rs:321-338)."""
        await self._get_connection()

    # ------------------------------ public API ------------------------------

    async def send_broadcast_message(self, topics: Sequence[int], message: bytes) -> None:
        conn = await self._get_connection()
        try:
            await conn.send_message(m.Broadcast(list(topics), message))
        except Exception as e:
            self._disconnect_on_error()
            raise ConnectionError_(str(e)) from e

    async def send_direct_message(self, recipient: bytes, message: bytes) -> None:
        conn = await self._get_connection()
        try:
            await conn.send_message(m.Direct(recipient, message))
        except Exception as e:
            self._disconnect_on_error()
            raise ConnectionError_(str(e)) from e

    async def receive_message(self) -> m.Message:
        conn = await self._get_connection()
        try:
            return await conn.recv_message()
        except asyncio.CancelledError:
            raise
        except Exception as e:
            self._disconnect_on_error()
            raise ConnectionError_(str(e)) from e

    async def receive_messages(self, max_n: int = 1024) -> "List[m.Message]":
        """Batched receive (extension over the reference API): await the
        first message, then drain everything already queued on the
        connection, up to max_n.  High-rate subscribers pay one await per
        BATCH instead of one per message."""
        conn = await self._get_connection()
        try:
            out = [await conn.recv_message()]
            q = getattr(conn, "_recv_q", None)
            while q is not None and not q.empty() and len(out) < max_n:
                raw = q.get_nowait()
                try:
                    out.append(m.deserialize(raw.data))
                finally:
                    raw.drop()
            return out
        except asyncio.CancelledError:
            raise
        except Exception as e:
            self._disconnect_on_error()
            raise ConnectionError_(str(e)) from e

    async def receive_raw_batch(self, max_n: int = 4096) -> "List[bytes]":
        """Batched receive of RAW wire frames (no deserialization): the
        high-rate benchmark/relay path — callers that only count, forward,
        or peek fixed offsets skip the per-message parse entirely.  The
        reference forwards raw bytes verbatim the same way
        (user/handler.rs:109 keeps the raw alongside the parsed form)."""
        conn = await self._get_connection()
        try:
            raw = await conn.recv_message_raw()
            out = [raw.data]
            raw.drop()
            q = getattr(conn, "_recv_q", None)
            while q is not None and not q.empty() and len(out) < max_n:
                r = q.get_nowait()
                out.append(r.data)
                r.drop()
            return out
        except asyncio.CancelledError:
            raise
        except Exception as e:
            self._disconnect_on_error()
            raise ConnectionError_(str(e)) from e

    async def subscribe(self, topics: Sequence[int]) -> None:
        """Update the replay set first, then best-effort send
        (reference lib.rs:383-414)."""
        async with self._topics_lock:
            new = [t for t in topics if t not in self._topics]
            self._topics.update(new)
        if not new:
            return
        if self._connection is not None:
            try:
                await self._connection.send_message(m.Subscribe(new))
            except Exception:
                self._disconnect_on_error()

    async def unsubscribe(self, topics: Sequence[int]) -> None:
        async with self._topics_lock:
            gone = [t for t in topics if t in self._topics]
            self._topics.difference_update(gone)
        if not gone:
            return
        if self._connection is not None:
            try:
                await self._connection.send_message(m.Unsubscribe(gone))
            except Exception:
                self._disconnect_on_error()

    @property
    def public_key(self) -> bytes:
        return self.config.keypair.public_key

    def close(self) -> None:
        self._closed = True
        self._disconnect_on_error()

    @property
    def is_closed(self) -> bool:
        return self._closed
