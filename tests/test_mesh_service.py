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
