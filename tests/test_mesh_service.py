"""MeshBroker end-to-end over gloo (world_size 2, CPU engine): two brokers
whose broker-plane is the RcclMesh collective exchange; cross-rank broadcast
and direct delivery through the full pipeline (marshal auth -> user plane ->
mesh tick -> egress drain -> client)."""

import os
import subprocess
import sys
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent

WORKER = r'''
import asyncio, os, sys
sys.path.insert(0, os.environ["PUSHCDN_REPO"])

from pushcdn_amd.broker.mesh_service import MeshBroker
from pushcdn_amd.broker.service import BrokerConfig
from pushcdn_amd.client import Client, ClientConfig
from pushcdn_amd.crypto import bls
from pushcdn_amd.marshal import Marshal, MarshalConfig
from pushcdn_amd.proto import message as m
from pushcdn_amd.proto.transports.memory import Memory

RANK = int(os.environ["RANK"])


async def main():
    db = f"/tmp/mesh-svc-{os.environ['MASTER_PORT']}-{RANK}.db"
    cfg = BrokerConfig(
        public_bind_endpoint=f"mesh-pub-{RANK}",
        public_advertise_endpoint=f"mesh-pub-{RANK}",
        private_bind_endpoint=f"mesh-priv-{RANK}",
        private_advertise_endpoint=f"mesh-priv-{RANK}",
        discovery_endpoint=db,
        keypair=bls.KeyPair.from_seed(1000),
        user_protocol=Memory,
        broker_protocol=Memory,
        heartbeat_interval_s=0.2,
        data_plane="gpu",
        gpu_device="cpu",        # CPU engine under gloo for the test
        gpu_max_users=32,
        gpu_ring_bytes=1 << 14,
        gpu_tick_interval_s=0.01,
    )
    broker = MeshBroker(cfg, batch_capacity=1 << 14)
    await broker.start()
    await broker.discovery.perform_heartbeat(0, 60)
    marshal = Marshal(MarshalConfig(bind_endpoint=f"mesh-marshal-{RANK}",
                                    discovery_endpoint=db, protocol=Memory))
    await marshal.start()

    client = Client(ClientConfig(endpoint=f"mesh-marshal-{RANK}",
                                 keypair=bls.KeyPair.from_seed(100 + RANK),
                                 subscribed_topics=[0], protocol=Memory))
    await client.ensure_initialized()
    await asyncio.sleep(0.5)  # both ranks up, mesh ticking

    if RANK == 0:
        await client.send_broadcast_message([0], b"hello-mesh")
        # our own echo (subscribed to topic 0)
        msg = await asyncio.wait_for(client.receive_message(), timeout=20)
        assert isinstance(msg, m.Broadcast) and msg.message == b"hello-mesh", msg
        # expect rank 1's direct reply
        msg = await asyncio.wait_for(client.receive_message(), timeout=20)
        assert isinstance(msg, m.Direct) and msg.message == b"direct-back", msg
    else:
        msg = await asyncio.wait_for(client.receive_message(), timeout=20)
        assert isinstance(msg, m.Broadcast) and msg.message == b"hello-mesh", msg
        peer_pk = bls.KeyPair.from_seed(100).public_key
        await client.send_direct_message(peer_pk, b"direct-back")
        await asyncio.sleep(1.0)  # let the mesh deliver before teardown

    print(f"rank {RANK} mesh-service OK", flush=True)
    # keep ticking so the peer's collectives never stall, then hard-exit
    await asyncio.sleep(3.0)
    os._exit(0)


asyncio.run(main())
'''


def test_mesh_broker_service_gloo(tmp_path):
    script = tmp_path / "mesh_worker.py"
    script.write_text(WORKER)
    env = dict(os.environ)
    env["PUSHCDN_REPO"] = str(REPO)
    out = subprocess.run(
        [
            sys.executable, "-m", "torch.distributed.run",
            "--nnodes=1", "--nproc-per-node=2",
            "--master-addr", "127.0.0.1", "--master-port", "29523",
            str(script),
        ],
        capture_output=True, text=True, timeout=300, env=env,
    )
    assert out.returncode == 0, (out.stdout[-2000:], out.stderr[-3000:])
    assert "rank 0 mesh-service OK" in out.stdout
    assert "rank 1 mesh-service OK" in out.stdout


def test_mesh_broker_interest_routed_gloo(tmp_path):
    """Same end-to-end scenario, but with interest-routed grouped P2P
    instead of the all-gather."""
    script = tmp_path / "mesh_worker_p2p.py"
    script.write_text(WORKER.replace(
        "broker = MeshBroker(cfg, batch_capacity=1 << 14)",
        "broker = MeshBroker(cfg, batch_capacity=1 << 14, interest_routed=True)"))
    env = dict(os.environ)
    env["PUSHCDN_REPO"] = str(REPO)
    out = subprocess.run(
        [
            sys.executable, "-m", "torch.distributed.run",
            "--nnodes=1", "--nproc-per-node=2",
            "--master-addr", "127.0.0.1", "--master-port", "29527",
            str(script),
        ],
        capture_output=True, text=True, timeout=300, env=env,
    )
    assert out.returncode == 0, (out.stdout[-2000:], out.stderr[-3000:])
    assert "rank 0 mesh-service OK" in out.stdout
    assert "rank 1 mesh-service OK" in out.stdout


def test_mesh_broker_blob_ingest_single_rank(tmp_path):
    """MeshBroker consumes C++ ingest blobs: native-TCP users on a
    single-rank mesh (the collective degenerates to self-exchange), with
    interest digests computed from the pump's frame classification."""
    import asyncio
    import uuid

    from pushcdn_amd.broker.mesh_service import MeshBroker
    from pushcdn_amd.broker.service import BrokerConfig
    from pushcdn_amd.client import Client, ClientConfig
    from pushcdn_amd.crypto import bls
    from pushcdn_amd.discovery import BrokerIdentifier
    from pushcdn_amd.marshal import Marshal, MarshalConfig
    from pushcdn_amd.proto import message as m
    from pushcdn_amd.proto.transports.tcp_native import TcpNative

    async def go():
        os.environ.pop("WORLD_SIZE", None)
        db = str(tmp_path / f"meshblob-{uuid.uuid4().hex}.db")
        broker = MeshBroker(BrokerConfig(
            public_bind_endpoint="127.0.0.1:0",
            public_advertise_endpoint="127.0.0.1:0",
            private_bind_endpoint="127.0.0.1:0",
            private_advertise_endpoint="127.0.0.1:0",
            discovery_endpoint=db,
            keypair=bls.KeyPair.from_seed(1000),
            user_protocol=TcpNative,
            broker_protocol=TcpNative,
            data_plane="gpu",
            gpu_device="cpu",
            gpu_max_users=16,
            gpu_ring_bytes=1 << 14,
            gpu_tick_interval_s=0.01,
        ), batch_capacity=1 << 16, interest_routed=True)
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

        a = Client(ClientConfig(endpoint=ep, keypair=bls.KeyPair.from_seed(71),
                                subscribed_topics=[6], protocol=TcpNative))
        b = Client(ClientConfig(endpoint=ep, keypair=bls.KeyPair.from_seed(72),
                                subscribed_topics=[6], protocol=TcpNative))
        await a.ensure_initialized()
        await b.ensure_initialized()
        await asyncio.sleep(0.3)

        for i in range(30):
            await a.send_broadcast_message([6], f"mb-{i}".encode())
        got = [(await asyncio.wait_for(b.receive_message(), timeout=15)).message
               for _ in range(30)]
        assert got == [f"mb-{i}".encode() for i in range(30)]
        await a.send_direct_message(b.public_key, b"mb-direct")
        msg = await asyncio.wait_for(b.receive_message(), timeout=15)
        assert msg.message == b"mb-direct"

        a.close()
        b.close()
        await marshal.close()
        await broker.close()

    asyncio.run(asyncio.wait_for(go(), timeout=90))
