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


def test_marshal_gpu_batch_verify(tmp_path):
    """Auth storm through the K1 batched verifier: many clients authenticate
    concurrently against a gpu_verify marshal; a bad signature still fails."""
    import time as _time

    from pushcdn_amd.marshal import Marshal, MarshalConfig

    async def go():
        db = new_db(tmp_path)
        cfg = BrokerConfig(
            public_bind_endpoint="gvb-pub", public_advertise_endpoint="gvb-pub",
            private_bind_endpoint="gvb-priv", private_advertise_endpoint="gvb-priv",
            discovery_endpoint=db, keypair=bls.KeyPair.from_seed(1000),
            user_protocol=Memory, broker_protocol=Memory,
            heartbeat_interval_s=0.2, sync_interval_s=0.2,
        )
        broker = Broker(cfg)
        await broker.start()
        await broker.discovery.perform_heartbeat(0, 60)
        mcfg = MarshalConfig(bind_endpoint="gvb-marshal", discovery_endpoint=db,
                             protocol=Memory, gpu_verify=True)
        marshal = Marshal(mcfg)
        await marshal.start()
        assert marshal._verifier is not None

        clients = [make_client("gvb-marshal", seed=500 + i, topics=[0]) for i in range(32)]
        await asyncio.gather(*(c.ensure_initialized() for c in clients))
        assert len(broker.connections.users) == 32

        # direct GPU-verifier rejection check (bit-flipped signature)
        kp = bls.KeyPair.from_seed(999)
        sig = bytearray(bls.sign_timestamp(kp.private_key, bls.USER_MARSHAL_NAMESPACE,
                                           int(_time.time())))
        sig[3] ^= 1
        ok = await marshal._verifier.verify(
            kp.public_key, bls.USER_MARSHAL_NAMESPACE,
            int(_time.time()).to_bytes(8, "little"), bytes(sig))
        assert not ok

        for c in clients:
            c.close()
        await marshal.close()
        await broker.close()

    run(go())


def test_mesh_broker_gpu_single_rank(tmp_path):
    """MeshBroker (the RCCL-plane broker) with the REAL CDNA4 engine,
    world_size 1: the collective degenerates to self-exchange but the whole
    mesh tick (pack -> H2D -> kernel pipeline -> drain) runs on the GPU."""
    import os

    from pushcdn_amd.broker.mesh_service import MeshBroker

    async def go():
        os.environ.pop("WORLD_SIZE", None)  # single-rank mesh
        db = new_db(tmp_path)
        cfg = BrokerConfig(
            public_bind_endpoint="mgpu-pub",
            public_advertise_endpoint="mgpu-pub",
            private_bind_endpoint="mgpu-priv",
            private_advertise_endpoint="mgpu-priv",
            discovery_endpoint=db,
            keypair=bls.KeyPair.from_seed(1000),
            user_protocol=Memory,
            broker_protocol=Memory,
            heartbeat_interval_s=0.2,
            data_plane="gpu",
            gpu_device="cuda:0",
            gpu_max_users=64,
            gpu_ring_bytes=1 << 16,
            gpu_tick_interval_s=0.01,
        )
        broker = MeshBroker(cfg, batch_capacity=1 << 16)
        await broker.start()
        await broker.discovery.perform_heartbeat(0, 60)
        marshal, endpoint = make_marshal(db)
        await marshal.start()

        alice = make_client(endpoint, seed=31, topics=[4])
        bob = make_client(endpoint, seed=32, topics=[4])
        await alice.ensure_initialized()
        await bob.ensure_initialized()
        await asyncio.sleep(0.3)

        await alice.send_broadcast_message([4], b"mesh-gpu-broadcast")
        msg = await asyncio.wait_for(bob.receive_message(), timeout=15)
        assert isinstance(msg, m.Broadcast) and msg.message == b"mesh-gpu-broadcast"

        await bob.send_direct_message(alice.public_key, b"mesh-gpu-direct")
        msg = await asyncio.wait_for(alice.receive_message(), timeout=15)
        # alice also got her own broadcast echo first (subscribed to 4)
        if isinstance(msg, m.Broadcast):
            msg = await asyncio.wait_for(alice.receive_message(), timeout=15)
        assert isinstance(msg, m.Direct) and msg.message == b"mesh-gpu-direct"

        await stop_stack([broker], marshal, alice, bob)

    run(go())


def test_gpu_broker_native_tcp_full_path(tmp_path):
    """The full production path on real hardware: native C++ TCP pump on
    the user plane + CDNA4 kernel routing + C++ send_ring egress drain."""
    import uuid as _uuid
    from pushcdn_amd.broker.service import Broker, BrokerConfig
    from pushcdn_amd.client import Client, ClientConfig
    from pushcdn_amd.discovery import BrokerIdentifier
    from pushcdn_amd.marshal import Marshal, MarshalConfig
    from pushcdn_amd.proto.transports.tcp_native import TcpNative

    async def go():
        db = str(tmp_path / f"natcuda-{_uuid.uuid4().hex}.db")
        broker = Broker(BrokerConfig(
            public_bind_endpoint="127.0.0.1:0",
            public_advertise_endpoint="127.0.0.1:0",
            private_bind_endpoint="127.0.0.1:0",
            private_advertise_endpoint="127.0.0.1:0",
            discovery_endpoint=db,
            keypair=bls.KeyPair.from_seed(1000),
            user_protocol=TcpNative,
            broker_protocol=TcpNative,
            data_plane="gpu",
            gpu_device="cuda:0",
            gpu_max_users=64,
            gpu_ring_bytes=1 << 16,
            gpu_tick_interval_s=0.005,
        ))
        await broker.start()
        pub = f"127.0.0.1:{broker._user_listener.port}"
        priv = f"127.0.0.1:{broker._broker_listener.port}"
        broker.config.public_advertise_endpoint = pub
        broker.config.private_advertise_endpoint = priv
        broker.identity = BrokerIdentifier(pub, priv)
        broker.discovery.identity = broker.identity
        broker.connections.identity = broker.identity
        await broker.discovery.perform_heartbeat(0, 600)
        marshal = Marshal(MarshalConfig(bind_endpoint="127.0.0.1:0",
                                        discovery_endpoint=db, protocol=TcpNative))
        await marshal.start()
        ep = f"127.0.0.1:{marshal._listener.port}"

        alice = Client(ClientConfig(endpoint=ep, keypair=bls.KeyPair.from_seed(41),
                                    subscribed_topics=[7], protocol=TcpNative))
        bob = Client(ClientConfig(endpoint=ep, keypair=bls.KeyPair.from_seed(42),
                                  subscribed_topics=[7], protocol=TcpNative))
        await alice.ensure_initialized()
        await bob.ensure_initialized()
        await asyncio.sleep(0.3)

        for i in range(50):
            await alice.send_broadcast_message([7], f"cuda-ring-{i}".encode())
        for i in range(50):
            msg = await asyncio.wait_for(bob.receive_message(), timeout=15)
            assert msg.message == f"cuda-ring-{i}".encode()
        await alice.send_direct_message(bob.public_key, b"cuda-direct")
        msg = await asyncio.wait_for(bob.receive_message(), timeout=15)
        assert msg.message == b"cuda-direct"

        alice.close()
        bob.close()
        await marshal.close()
        await broker.close()

    run(go())


def test_direct_burst_order_preserved(tmp_path):
    """A burst of DIRECT messages from one sender in one tick must arrive
    in send order: K5b's atomic ring claims interleave, and the drain's
    seq sort restores arrival order (reference per-connection FIFO)."""
    async def go():
        db = new_db(tmp_path)
        cfg = BrokerConfig(
            public_bind_endpoint="dord-pub",
            public_advertise_endpoint="dord-pub",
            private_bind_endpoint="dord-priv",
            private_advertise_endpoint="dord-priv",
            discovery_endpoint=db,
            keypair=bls.KeyPair.from_seed(1000),
            user_protocol=Memory,
            broker_protocol=Memory,
            data_plane="gpu",
            gpu_device="cuda:0",
            gpu_max_users=16,
            gpu_ring_bytes=1 << 18,
            gpu_tick_interval_s=0.05,  # long tick so the burst lands in ONE batch
        )
        broker = Broker(cfg)
        await broker.start()
        await broker.discovery.perform_heartbeat(0, 60)
        marshal, endpoint = make_marshal(db)
        await marshal.start()
        alice = make_client(endpoint, seed=61, topics=[])
        bob = make_client(endpoint, seed=62, topics=[])
        await alice.ensure_initialized()
        await bob.ensure_initialized()
        await asyncio.sleep(0.3)

        for round_ in range(3):
            for i in range(64):
                await alice.send_direct_message(bob.public_key,
                                                f"d-{round_}-{i}".encode())
            got = [
                (await asyncio.wait_for(bob.receive_message(), timeout=15)).message
                for _ in range(64)
            ]
            assert got == [f"d-{round_}-{i}".encode() for i in range(64)], got[:8]

        await stop_stack([broker], marshal, alice, bob)

    run(go())
