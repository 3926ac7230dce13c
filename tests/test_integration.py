"""End-to-end integration tests: real marshal + broker(s) + clients in one
process over the Memory transport + shared temp-SQLite discovery — the
reference's tier-4 strategy (tests/src/tests/mod.rs:62-143).

Scenarios mirror the reference:
  - basic connect + direct-to-self (basic_connect.rs:15-56)
  - subscribe/unsubscribe delivery semantics (subscribe.rs:19-186)
  - duplicate-key connect kicks the old session (double_connect.rs:17-141)
  - whitelist rejection (whitelist.rs:15-77)
  - broadcast across two brokers over the mesh
"""

import asyncio
import uuid

import pytest

from pushcdn_amd.broker.service import Broker, BrokerConfig
from pushcdn_amd.client import Client, ClientConfig
from pushcdn_amd.crypto import bls
from pushcdn_amd.marshal import Marshal, MarshalConfig
from pushcdn_amd.proto import message as m
from pushcdn_amd.proto.errors import ConnectionError_
from pushcdn_amd.proto.transports.memory import Memory


def run(coro):
    return asyncio.run(asyncio.wait_for(coro, timeout=60))


def new_db(tmp_path):
    return str(tmp_path / f"disc-{uuid.uuid4().hex}.db")


_ENDPOINT_N = [0]


def make_broker(db, *, keypair=None, tag=None, fast=True):
    _ENDPOINT_N[0] += 1
    n = tag or f"{uuid.uuid4().hex[:6]}-{_ENDPOINT_N[0]}"
    cfg = BrokerConfig(
        public_bind_endpoint=f"bpub-{n}",
        public_advertise_endpoint=f"bpub-{n}",
        private_bind_endpoint=f"bpriv-{n}",
        private_advertise_endpoint=f"bpriv-{n}",
        discovery_endpoint=db,
        keypair=keypair or bls.KeyPair.from_seed(1000),
        user_protocol=Memory,
        broker_protocol=Memory,
        heartbeat_interval_s=0.2,
        sync_interval_s=0.2,
        whitelist_interval_s=0.3,
    )
    return Broker(cfg)


def make_marshal(db, *, tag=None):
    n = tag or uuid.uuid4().hex[:6]
    cfg = MarshalConfig(bind_endpoint=f"marshal-{n}", discovery_endpoint=db, protocol=Memory)
    return Marshal(cfg), f"marshal-{n}"


def make_client(marshal_endpoint, seed, topics):
    return Client(
        ClientConfig(
            endpoint=marshal_endpoint,
            keypair=bls.KeyPair.from_seed(seed),
            subscribed_topics=list(topics),
            protocol=Memory,
        )
    )


async def start_stack(tmp_path, n_brokers=1):
    db = new_db(tmp_path)
    brokers = [make_broker(db) for _ in range(n_brokers)]
    for b in brokers:
        await b.start()
    # let heartbeats register the brokers before the marshal picks one
    for b in brokers:
        await b.discovery.perform_heartbeat(0, 60)
    marshal, endpoint = make_marshal(db)
    await marshal.start()
    # give brokers a moment to dial each other
    if n_brokers > 1:
        await asyncio.sleep(0.6)
    return brokers, marshal, endpoint


async def stop_stack(brokers, marshal, *clients):
    for c in clients:
        c.close()
    await marshal.close()
    for b in brokers:
        await b.close()


def test_basic_connect_and_direct_to_self(tmp_path):
    async def go():
        brokers, marshal, endpoint = await start_stack(tmp_path)
        client = make_client(endpoint, seed=1, topics=[0])
        await client.ensure_initialized()
        await client.send_direct_message(client.public_key, b"echo-me")
        msg = await client.receive_message()
        assert isinstance(msg, m.Direct)
        assert msg.message == b"echo-me"
        await stop_stack(brokers, marshal, client)

    run(go())


def test_broadcast_subscribe_semantics(tmp_path):
    async def go():
        brokers, marshal, endpoint = await start_stack(tmp_path)
        alice = make_client(endpoint, seed=1, topics=[0])
        bob = make_client(endpoint, seed=2, topics=[1])
        await alice.ensure_initialized()
        await bob.ensure_initialized()
        await asyncio.sleep(0.1)
        # alice broadcasts on topic 1 -> bob receives, alice does not
        await alice.send_broadcast_message([1], b"topic-1-msg")
        msg = await bob.receive_message()
        assert isinstance(msg, m.Broadcast) and msg.message == b"topic-1-msg"
        # bob subscribes to 0 as well, then gets topic-0 traffic
        await bob.subscribe([0])
        await asyncio.sleep(0.1)
        await alice.send_broadcast_message([0], b"topic-0-msg")
        msg = await bob.receive_message()
        assert msg.message == b"topic-0-msg"
        # alice (subscribed to 0) also received her own broadcast
        msg = await alice.receive_message()
        assert msg.message == b"topic-0-msg"
        # bob unsubscribes from 1: no more topic-1 traffic (next recv times out)
        await bob.unsubscribe([1])
        await asyncio.sleep(0.1)
        await alice.send_broadcast_message([1], b"gone")
        with pytest.raises(asyncio.TimeoutError):
            await asyncio.wait_for(bob.receive_message(), timeout=0.5)
        await stop_stack(brokers, marshal, alice, bob)

    run(go())


def test_double_connect_kicks_old_session(tmp_path):
    async def go():
        brokers, marshal, endpoint = await start_stack(tmp_path)
        first = make_client(endpoint, seed=7, topics=[0])
        await first.ensure_initialized()
        second = make_client(endpoint, seed=7, topics=[0])  # same keypair
        await second.ensure_initialized()
        await asyncio.sleep(0.2)
        # the broker now has exactly one user with that key
        assert len(brokers[0].connections.users) == 1
        # the new session works
        await second.send_direct_message(second.public_key, b"to-new")
        msg = await second.receive_message()
        assert msg.message == b"to-new"
        await stop_stack(brokers, marshal, first, second)

    run(go())


def test_whitelist_rejection(tmp_path):
    async def go():
        db = new_db(tmp_path)
        broker = make_broker(db)
        await broker.start()
        await broker.discovery.perform_heartbeat(0, 60)
        marshal, endpoint = make_marshal(db)
        await marshal.start()
        allowed = bls.KeyPair.from_seed(100)
        await marshal.discovery.set_whitelist([allowed.public_key])
        good = Client(ClientConfig(endpoint=endpoint, keypair=allowed,
                                   subscribed_topics=[0], protocol=Memory))
        await good.ensure_initialized()
        bad = Client(ClientConfig(endpoint=endpoint, keypair=bls.KeyPair.from_seed(101),
                                  subscribed_topics=[0], protocol=Memory))
        with pytest.raises((ConnectionError_, asyncio.TimeoutError)):
            await asyncio.wait_for(bad.ensure_initialized(), timeout=3)
        await stop_stack([broker], marshal, good, bad)

    run(go())


def test_two_broker_mesh_broadcast_and_direct(tmp_path):
    async def go():
        db = new_db(tmp_path)
        kp = bls.KeyPair.from_seed(1000)  # shared cluster keypair
        b1 = make_broker(db, keypair=kp, tag="one")
        b2 = make_broker(db, keypair=kp, tag="two")
        await b1.start()
        await b2.start()
        await b1.discovery.perform_heartbeat(0, 60)
        await b2.discovery.perform_heartbeat(0, 60)
        await asyncio.sleep(0.8)  # let the mesh form
        assert len(b1.connections.brokers) == 1
        assert len(b2.connections.brokers) == 1

        marshal, endpoint = make_marshal(db)
        await marshal.start()

        # steer alice to b1, bob to b2 via artificial load reports
        await b1.discovery.perform_heartbeat(0, 60)
        await b2.discovery.perform_heartbeat(10, 60)
        alice = make_client(endpoint, seed=11, topics=[3])
        await alice.ensure_initialized()
        await b1.discovery.perform_heartbeat(10, 60)
        await b2.discovery.perform_heartbeat(0, 60)
        bob = make_client(endpoint, seed=12, topics=[3])
        await bob.ensure_initialized()
        assert len(b1.connections.users) == 1 and len(b2.connections.users) == 1

        await asyncio.sleep(0.6)  # allow topic/user sync

        # cross-broker broadcast
        await alice.send_broadcast_message([3], b"cross-broker")
        msg = await asyncio.wait_for(bob.receive_message(), timeout=5)
        assert msg.message == b"cross-broker"

        # cross-broker direct
        await alice.send_direct_message(bob.public_key, b"direct-cross")
        msg = await asyncio.wait_for(bob.receive_message(), timeout=5)
        assert isinstance(msg, m.Direct) and msg.message == b"direct-cross"

        await stop_stack([b1, b2], marshal, alice, bob)

    run(go())


def test_double_connect_across_brokers_kicks_old(tmp_path):
    """Same keypair connects to broker 1, then (marshal steered by load
    reports) to broker 2: the UserSync CRDT moves ownership and broker 1
    kicks its stale session (reference double_connect.rs:76-141)."""

    async def go():
        db = new_db(tmp_path)
        kp = bls.KeyPair.from_seed(1000)
        b1 = make_broker(db, keypair=kp, tag="dc-one")
        b2 = make_broker(db, keypair=kp, tag="dc-two")
        await b1.start()
        await b2.start()
        await b1.discovery.perform_heartbeat(0, 60)
        await b2.discovery.perform_heartbeat(0, 60)
        await asyncio.sleep(0.8)  # mesh forms
        marshal, endpoint = make_marshal(db)
        await marshal.start()

        # steer the first session to b1
        await b1.discovery.perform_heartbeat(0, 60)
        await b2.discovery.perform_heartbeat(10, 60)
        first = make_client(endpoint, seed=77, topics=[0])
        await first.ensure_initialized()
        assert len(b1.connections.users) == 1 and len(b2.connections.users) == 0

        # steer the second session (same key!) to b2
        await b1.discovery.perform_heartbeat(10, 60)
        await b2.discovery.perform_heartbeat(0, 60)
        second = make_client(endpoint, seed=77, topics=[0])
        await second.ensure_initialized()
        assert len(b2.connections.users) == 1

        # b1 learns via UserSync that b2 owns the key now and kicks its
        # stale session (sync_interval_s=0.2 in make_broker)
        for _ in range(40):
            if len(b1.connections.users) == 0:
                break
            await asyncio.sleep(0.1)
        assert len(b1.connections.users) == 0

        # and the surviving session is fully functional
        await second.send_direct_message(second.public_key, b"to-survivor")
        msg = await asyncio.wait_for(second.receive_message(), timeout=5)
        assert isinstance(msg, m.Direct) and msg.message == b"to-survivor"

        await stop_stack([b1, b2], marshal, first, second)

    run(go())


def test_cli_parser_smoke():
    """CLI surface exists with the reference's flag set (no daemon start)."""
    from pushcdn_amd import cli

    p_err = None
    try:
        cli.main(["broker", "--help"])
    except SystemExit as e:
        p_err = e.code
    assert p_err == 0


def test_large_message_flood(tmp_path):
    """9 MB direct + broadcast to self in a loop — the reference's
    bad-sender load shape (bad-sender.rs:24-105) — through the memory pool
    and framing limits."""

    async def go():
        brokers, marshal, endpoint = await start_stack(tmp_path)
        client = make_client(endpoint, seed=9, topics=[0])
        await client.ensure_initialized()
        payload = bytes(9 * 1024 * 1024)
        for i in range(3):
            await client.send_direct_message(client.public_key, payload)
            msg = await asyncio.wait_for(client.receive_message(), timeout=30)
            assert isinstance(msg, m.Direct) and len(msg.message) == len(payload)
            await client.send_broadcast_message([0], payload)
            msg = await asyncio.wait_for(client.receive_message(), timeout=30)
            assert isinstance(msg, m.Broadcast) and len(msg.message) == len(payload)
        await stop_stack(brokers, marshal, client)

    run(go())


def test_multiple_marshals_share_discovery(tmp_path):
    """Several marshals over one discovery namespace: permits issued by any
    marshal validate at the broker (the reference runs marshal fleets
    behind a load balancer; permits live in discovery, not the marshal)."""

    async def go():
        brokers, marshal1, ep1 = await start_stack(tmp_path)
        marshal2, ep2 = make_marshal(brokers[0].config.discovery_endpoint)
        await marshal2.start()

        a = make_client(ep1, seed=31, topics=[0])
        b = make_client(ep2, seed=32, topics=[0])  # different marshal
        await a.ensure_initialized()
        await b.ensure_initialized()
        await asyncio.sleep(0.2)
        await a.send_broadcast_message([0], b"via-marshal-1")
        msg = await asyncio.wait_for(b.receive_message(), timeout=10)
        assert msg.message == b"via-marshal-1"
        a.close(); b.close()
        await marshal2.close()
        await stop_stack(brokers, marshal1)

    run(go())


def test_marshal_rejects_garbage_and_wrong_type(tmp_path):
    """The marshal's one-shot handler: a non-auth first frame or undecodable
    bytes ends the attempt cleanly (reference handlers.rs:21-37 bails), and
    the marshal keeps serving afterwards."""
    from pushcdn_amd.proto.limiter import Bytes, Limiter
    from pushcdn_amd.proto.transports.memory import Memory

    async def go():
        brokers, marshal, endpoint = await start_stack(tmp_path)
        limiter = Limiter(global_memory_pool_size=1 << 20)

        # wrong first message type
        conn = await Memory.connect(endpoint, True, limiter)
        await conn.send_message(m.Subscribe([1]))
        try:
            reply = await asyncio.wait_for(conn.recv_message(), timeout=5)
            assert getattr(reply, "permit", 0) == 0  # failure response
        except Exception:
            pass  # or the marshal just dropped us — also fine
        conn.close()

        # undecodable bytes
        conn = await Memory.connect(endpoint, True, limiter)
        await conn.send_message_raw(Bytes(b"\x00\x01\x02\x03garbage"))
        await asyncio.sleep(0.2)
        conn.close()

        # the marshal still serves real clients
        client = make_client(endpoint, seed=44, topics=[0])
        await client.ensure_initialized()
        await client.send_direct_message(client.public_key, b"still-up")
        assert (await asyncio.wait_for(client.receive_message(), timeout=10)).message == b"still-up"
        await stop_stack(brokers, marshal, client)

    run(go())
