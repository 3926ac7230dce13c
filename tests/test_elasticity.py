"""Elastic-recovery tests (reference SURVEY §5.3): client auto-reconnect +
subscription replay after its broker dies; broker mesh self-healing after a
peer kill/rejoin; eviction on send failure."""

import asyncio


from tests.test_integration import (
    make_broker,
    make_client,
    make_marshal,
    new_db,
    stop_stack,
)
from pushcdn_amd.crypto import bls
from pushcdn_amd.proto import message as m


def run(coro):
    return asyncio.run(asyncio.wait_for(coro, timeout=90))


def test_client_reconnects_after_broker_restart(tmp_path):
    async def go():
        db = new_db(tmp_path)
        broker = make_broker(db, tag="restarting")
        await broker.start()
        await broker.discovery.perform_heartbeat(0, 60)
        marshal, endpoint = make_marshal(db)
        await marshal.start()

        client = make_client(endpoint, seed=5, topics=[2])
        await client.ensure_initialized()
        await client.send_direct_message(client.public_key, b"before")
        assert (await client.receive_message()).message == b"before"

        # kill the broker; its discovery record expires via TTL, a new broker
        # (same endpoints) comes back — rejoin-from-network, no persistence
        await broker.close()
        broker2 = make_broker(db, tag="restarting")  # same endpoints
        await broker2.start()
        await broker2.discovery.perform_heartbeat(0, 60)

        # the client's next operations trigger reconnect + subscription replay
        ok = False
        for _ in range(10):
            try:
                await client.send_broadcast_message([2], b"after")
                msg = await asyncio.wait_for(client.receive_message(), timeout=2)
                if isinstance(msg, m.Broadcast) and msg.message == b"after":
                    ok = True
                    break
            except Exception:
                await asyncio.sleep(0.3)
        assert ok, "client did not recover after broker restart"
        await stop_stack([broker2], marshal, client)

    run(go())


def test_mesh_reheals_after_peer_kill(tmp_path):
    async def go():
        db = new_db(tmp_path)
        kp = bls.KeyPair.from_seed(1000)
        b1 = make_broker(db, keypair=kp, tag="stay")
        b2 = make_broker(db, keypair=kp, tag="die")
        await b1.start()
        await b2.start()
        await b1.discovery.perform_heartbeat(0, 60)
        await b2.discovery.perform_heartbeat(0, 60)
        await asyncio.sleep(0.8)
        assert len(b1.connections.brokers) == 1

        await b2.close()
        # b1 notices on next send/recv failure; eventually a replacement joins
        b3 = make_broker(db, keypair=kp, tag="reborn")
        await b3.start()
        await b3.discovery.perform_heartbeat(0, 60)
        deadline = asyncio.get_event_loop().time() + 10
        while asyncio.get_event_loop().time() < deadline:
            if any(str(b.public_advertise_endpoint).startswith("bpub-reborn")
                   for b in b1.connections.brokers):
                break
            await asyncio.sleep(0.2)
        assert any(str(b.public_advertise_endpoint).startswith("bpub-reborn")
                   for b in b1.connections.brokers), "mesh did not re-heal"
        await b1.close()
        await b3.close()

    run(go())


def test_user_evicted_on_dead_connection(tmp_path):
    async def go():
        db = new_db(tmp_path)
        broker = make_broker(db)
        await broker.start()
        await broker.discovery.perform_heartbeat(0, 60)
        marshal, endpoint = make_marshal(db)
        await marshal.start()
        client = make_client(endpoint, seed=9, topics=[0])
        await client.ensure_initialized()
        await asyncio.sleep(0.1)
        assert len(broker.connections.users) == 1
        # hard-kill the client's connection; broker's receive loop errors and
        # evicts the user (reference sender.rs/user handler eviction)
        client._connection.close()
        deadline = asyncio.get_event_loop().time() + 5
        while asyncio.get_event_loop().time() < deadline:
            if len(broker.connections.users) == 0:
                break
            await asyncio.sleep(0.1)
        assert len(broker.connections.users) == 0
        await stop_stack([broker], marshal, client)

    run(go())


def test_concurrent_sends_single_reconnect(tmp_path):
    """Concurrent sends during a dead connection trigger exactly ONE
    reconnect (the connecting-guard semaphore, reference lib.rs:204-258)."""
    async def go():
        db = new_db(tmp_path)
        broker = make_broker(db)
        await broker.start()
        await broker.discovery.perform_heartbeat(0, 60)
        marshal, endpoint = make_marshal(db)
        await marshal.start()
        client = make_client(endpoint, seed=41, topics=[0])
        await client.ensure_initialized()

        connects = 0
        orig = client._connect_once

        async def counting_connect():
            nonlocal connects
            connects += 1
            return await orig()

        client._connect_once = counting_connect
        client._disconnect_on_error()  # force a dead connection
        await asyncio.gather(*(
            client.send_broadcast_message([0], f"c{i}".encode()) for i in range(8)
        ))
        assert connects == 1, f"expected one reconnect, got {connects}"
        assert len(broker.connections.users) == 1
        await stop_stack([broker], marshal, client)

    run(go())


def test_receive_messages_batched(tmp_path):
    """receive_messages drains queued frames in order, one await per batch."""
    import asyncio

    from tests.test_integration import make_client, make_marshal, new_db, run, start_stack

    async def go():
        brokers, marshal, endpoint = await start_stack(tmp_path)
        a = make_client(endpoint, seed=3, topics=[0])
        b = make_client(endpoint, seed=4, topics=[0])
        await a.ensure_initialized()
        await b.ensure_initialized()
        for i in range(30):
            await a.send_broadcast_message([0], f"b-{i}".encode())
        got = []
        while len(got) < 30:
            got.extend(msg.message for msg in await b.receive_messages())
        assert got == [f"b-{i}".encode() for i in range(30)]
        a.close(); b.close()
        await marshal.close()
        for br in brokers:
            await br.close()

    run(go())


def test_dynamic_subscriptions_replay_after_restart(tmp_path):
    """Subscriptions changed at runtime (subscribe/unsubscribe AFTER
    connect) are what gets replayed on reconnect (reference lib.rs:383-414
    updates the replay set first, then best-effort sends)."""

    async def go():
        db = new_db(tmp_path)
        broker = make_broker(db, tag="replay")
        await broker.start()
        await broker.discovery.perform_heartbeat(0, 60)
        marshal, endpoint = make_marshal(db)
        await marshal.start()

        client = make_client(endpoint, seed=6, topics=[2])
        await client.ensure_initialized()
        await client.subscribe([7])      # added at runtime
        await client.unsubscribe([2])    # removed at runtime
        await asyncio.sleep(0.2)

        await broker.close()
        broker2 = make_broker(db, tag="replay")
        await broker2.start()
        await broker2.discovery.perform_heartbeat(0, 60)

        # after reconnect, topic 7 must deliver (replayed) ...
        ok = False
        for _ in range(10):
            try:
                await client.send_broadcast_message([7], b"on-seven")
                msg = await asyncio.wait_for(client.receive_message(), timeout=2)
                if isinstance(msg, m.Broadcast) and msg.message == b"on-seven":
                    ok = True
                    break
            except Exception:
                await asyncio.sleep(0.3)
        assert ok
        # ... and topic 2 must NOT (unsubscribed before the restart)
        await client.send_broadcast_message([2], b"on-two")
        try:
            msg = await asyncio.wait_for(client.receive_message(), timeout=1)
            raise AssertionError(f"unsubscribed topic delivered: {msg}")
        except asyncio.TimeoutError:
            pass
        await stop_stack([broker2], marshal, client)

    run(go())


def test_client_explicit_close_is_final(tmp_path):
    """Explicit Client.close() is FINAL: reconnect only heals FAILURES,
    never a user-initiated shutdown — post-close operations raise instead
    of silently reviving the session."""
    from pushcdn_amd.proto.errors import ConnectionError_
    from tests.test_integration import make_broker, make_client, make_marshal, new_db, run

    async def go():
        db = new_db(tmp_path)
        broker = make_broker(db, tag="reopen")
        await broker.start()
        await broker.discovery.perform_heartbeat(0, 60)
        marshal, endpoint = make_marshal(db)
        await marshal.start()
        client = make_client(endpoint, seed=8, topics=[1])
        await client.ensure_initialized()
        await client.ensure_initialized()  # idempotent
        client.close()
        assert client.is_closed
        try:
            await client.send_broadcast_message([1], b"after-close")
            raise AssertionError("closed client accepted a send")
        except ConnectionError_:
            pass
        await marshal.close()
        await broker.close()

    run(go())
