"""GPU-data-plane broker service test: a real broker service with
data_plane="gpu" routes live client traffic through the CDNA4 kernel
pipeline (ingest batch -> parse -> topic match -> fanout -> ring drain ->
socket write-back)."""

import asyncio

import pytest

from tests.test_integration import make_client, make_marshal, new_db, stop_stack
from pushcdn_amd.broker.service import Broker, BrokerConfig
from pushcdn_amd.crypto import bls
from pushcdn_amd.proto import message as m
from pushcdn_amd.proto.transports.memory import Memory

pytestmark = pytest.mark.gpu


def run(coro):
    return asyncio.run(asyncio.wait_for(coro, timeout=120))


def test_gpu_broker_service_end_to_end(tmp_path):
    async def go():
        db = new_db(tmp_path)
        cfg = BrokerConfig(
            public_bind_endpoint="gpub-svc",
            public_advertise_endpoint="gpub-svc",
            private_bind_endpoint="gpriv-svc",
            private_advertise_endpoint="gpriv-svc",
            discovery_endpoint=db,
            keypair=bls.KeyPair.from_seed(1000),
            user_protocol=Memory,
            broker_protocol=Memory,
            heartbeat_interval_s=0.2,
            sync_interval_s=0.2,
            data_plane="gpu",
            gpu_device="cuda:0",
            gpu_max_users=256,
            gpu_ring_bytes=1 << 16,
        )
        broker = Broker(cfg)
        await broker.start()
        await broker.discovery.perform_heartbeat(0, 60)
        marshal, endpoint = make_marshal(db)
        await marshal.start()

        alice = make_client(endpoint, seed=21, topics=[1])
        bob = make_client(endpoint, seed=22, topics=[1, 2])
        await alice.ensure_initialized()
        await bob.ensure_initialized()
        await asyncio.sleep(0.2)

        # broadcast through the GPU pipeline
        await alice.send_broadcast_message([1], b"gpu-routed-broadcast")
        msg = await asyncio.wait_for(bob.receive_message(), timeout=10)
        assert isinstance(msg, m.Broadcast) and msg.message == b"gpu-routed-broadcast"
        msg = await asyncio.wait_for(alice.receive_message(), timeout=10)
        assert msg.message == b"gpu-routed-broadcast"  # sender subscribed too

        # direct through the GPU pipeline (K5 DirectMap)
        await alice.send_direct_message(bob.public_key, b"gpu-routed-direct")
        msg = await asyncio.wait_for(bob.receive_message(), timeout=10)
        assert isinstance(msg, m.Direct) and msg.message == b"gpu-routed-direct"

        # several messages in one tick, FIFO preserved
        for i in range(10):
            await alice.send_broadcast_message([2], f"burst-{i}".encode())
        got = []
        for _ in range(10):
            msg = await asyncio.wait_for(bob.receive_message(), timeout=10)
            got.append(msg.message)
        assert got == [f"burst-{i}".encode() for i in range(10)]

        await stop_stack([broker], marshal, alice, bob)

    run(go())
