"""Deterministic single-broker test harness (reference
``cdn-broker/src/tests/mod.rs`` — exported, not cfg(test)-gated, because the
benches reuse it; same here).

Builds a real ``Broker`` over the Memory transport + temp SQLite discovery,
then *injects* fake users and brokers directly: creates connection pairs,
spawns the broker's own receive loops on the server half, and inserts into
``Connections`` — bypassing auth (reference inject_users :258-300 /
inject_brokers :308-389).  Fake brokers then push hand-built ``TopicSync`` /
``UserSync`` maps so the broker "knows" their users.
"""

from __future__ import annotations

import asyncio
import tempfile
import uuid
from dataclasses import dataclass, field
from typing import List, Sequence

from ..crypto import bls
from ..discovery import BrokerIdentifier
from ..proto import message as m
from ..proto.transports.base import Connection
from ..proto.transports.memory import Memory, gen_testing_connection_pair
from .service import Broker, BrokerConfig, BrokerHandle, UserHandle
from .versioned_map import Versioned, serialize_delta


def at_index(i: int) -> bytes:
    """Index-derived user identity (reference at_index!, tests/mod.rs:110-115)."""
    return f"user-{i}".encode()


@dataclass
class TestUser:
    __test__ = False  # not a pytest class
    topics: List[int]


@dataclass
class TestBroker:
    __test__ = False  # not a pytest class
    connected_users: List[int]  # indices of users this remote broker owns
    topics: List[int] = field(default_factory=list)


@dataclass
class TestRun:
    __test__ = False  # not a pytest class
    broker: Broker
    users: List[Connection]     # client half of each injected user
    brokers: List[Connection]   # client half of each injected broker

    async def close(self) -> None:
        await self.broker.close()


@dataclass
class TestDefinition:
    """reference tests/mod.rs:154-157."""

    __test__ = False  # not a pytest class

    connected_users: List[TestUser] = field(default_factory=list)
    connected_brokers: List[TestBroker] = field(default_factory=list)
    topic_space: object = None  # TopicSpace override (default ALL_TOPICS)
    user_message_hook: object = None
    broker_message_hook: object = None

    async def into_run(self) -> TestRun:
        n = uuid.uuid4().hex[:8]
        db = tempfile.mktemp(prefix=f"pushcdn-harness-{n}", suffix=".db")
        cfg = BrokerConfig(
            public_bind_endpoint=f"harness-pub-{n}",
            public_advertise_endpoint=f"harness-pub-{n}",
            private_bind_endpoint=f"harness-priv-{n}",
            private_advertise_endpoint=f"harness-priv-{n}",
            discovery_endpoint=db,
            keypair=bls.KeyPair.from_seed(0),
            user_protocol=Memory,
            broker_protocol=Memory,
            heartbeat_interval_s=3600,  # harness drives everything manually
            sync_interval_s=3600,
            whitelist_interval_s=3600,
        )
        if self.topic_space is not None:
            cfg.topic_space = self.topic_space
        cfg.user_message_hook = self.user_message_hook
        cfg.broker_message_hook = self.broker_message_hook
        broker = Broker(cfg)
        await broker.start()

        users: List[Connection] = []
        for i, tu in enumerate(self.connected_users):
            client_half = await self._inject_user(broker, at_index(i), tu.topics)
            users.append(client_half)

        brokers: List[Connection] = []
        for j, tb in enumerate(self.connected_brokers):
            client_half = await self._inject_broker(broker, j, tb)
            brokers.append(client_half)
        # let the injected sync messages apply
        await asyncio.sleep(0.05)
        return TestRun(broker=broker, users=users, brokers=brokers)

    @staticmethod
    async def _inject_user(broker: Broker, pubkey: bytes, topics: Sequence[int]) -> Connection:
        client_half, server_half = gen_testing_connection_pair(broker.limiter)
        handle = UserHandle(connection=server_half)
        old = broker.connections.add_user(pubkey, handle, topics)
        assert old is None
        handle.task = asyncio.get_running_loop().create_task(
            broker._user_receive_loop(pubkey, handle)
        )
        return client_half

    @staticmethod
    async def _inject_broker(broker: Broker, index: int, tb: TestBroker) -> Connection:
        ident = BrokerIdentifier(f"fake-broker-{index}-pub", f"fake-broker-{index}-priv")
        client_half, server_half = gen_testing_connection_pair(broker.limiter)
        handle = BrokerHandle(connection=server_half)
        broker.connections.add_broker(ident, handle)
        handle.task = asyncio.get_running_loop().create_task(
            broker._broker_receive_loop(ident, server_half)
        )
        # hand-built TopicSync: the fake broker subscribes to its topics
        topic_delta = {
            t: Versioned(True, 1, str(ident)) for t in tb.topics
        }
        await client_half.send_message(
            m.TopicSync(serialize_delta(topic_delta, lambda t: bytes([t]), lambda v: b"\x01"))
        )
        # hand-built UserSync: the fake broker owns these users
        user_delta = {
            at_index(i): Versioned(str(ident), 10_000, str(ident))
            for i in tb.connected_users
        }
        await client_half.send_message(
            m.UserSync(serialize_delta(user_delta, bytes, lambda v: v.encode()))
        )
        return client_half


async def assert_received(conn: Connection, expected: m.Message, timeout: float = 0.1) -> None:
    """reference assert_received! (yes variant)."""
    msg = await asyncio.wait_for(conn.recv_message(), timeout)
    assert msg == expected, f"expected {expected!r}, got {msg!r}"


async def assert_not_received(conn: Connection, timeout: float = 0.05) -> None:
    """reference assert_received! (no variant)."""
    try:
        msg = await asyncio.wait_for(conn.recv_message(), timeout)
    except asyncio.TimeoutError:
        return
    raise AssertionError(f"unexpectedly received {msg!r}")
