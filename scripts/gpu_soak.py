#!/usr/bin/env python3
"""GPU service soak: a MeshBroker (world_size 1) + marshal + pumping clients
for --seconds of sustained traffic; asserts deliveries keep flowing and HBM
use stays flat (no allocator growth tick-over-tick)."""

import argparse
import asyncio
import sys
import tempfile
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch

from pushcdn_amd.broker.mesh_service import MeshBroker
from pushcdn_amd.broker.service import BrokerConfig
from pushcdn_amd.client import Client, ClientConfig
from pushcdn_amd.crypto import bls
from pushcdn_amd.marshal import Marshal, MarshalConfig
from pushcdn_amd.proto.transports.memory import Memory


async def main(seconds: int) -> None:
    db = tempfile.mktemp(suffix=".db")
    cfg = BrokerConfig(
        public_bind_endpoint="soak-pub", public_advertise_endpoint="soak-pub",
        private_bind_endpoint="soak-priv", private_advertise_endpoint="soak-priv",
        discovery_endpoint=db, keypair=bls.KeyPair.from_seed(1000),
        user_protocol=Memory, broker_protocol=Memory,
        heartbeat_interval_s=1.0, data_plane="gpu", gpu_device="cuda:0",
        gpu_max_users=64, gpu_ring_bytes=1 << 20, gpu_tick_interval_s=0.002,
    )
    broker = MeshBroker(cfg, batch_capacity=1 << 20, interest_routed=False)
    await broker.start()
    await broker.discovery.perform_heartbeat(0, 600)
    marshal = Marshal(MarshalConfig(bind_endpoint="soak-marshal",
                                    discovery_endpoint=db, protocol=Memory))
    await marshal.start()

    n_clients = 8
    clients = [Client(ClientConfig(endpoint="soak-marshal",
                                   keypair=bls.KeyPair.from_seed(3000 + i),
                                   subscribed_topics=[0], protocol=Memory))
               for i in range(n_clients)]
    for c in clients:
        await c.ensure_initialized()
    await asyncio.sleep(0.5)

    received = [0] * n_clients
    stop = asyncio.Event()

    async def rx(i):
        while not stop.is_set():
            try:
                await asyncio.wait_for(clients[i].receive_message(), timeout=0.5)
                received[i] += 1
            except asyncio.TimeoutError:
                pass

    async def tx(i):
        seq = 0
        payload = bytes(1024)
        while not stop.is_set():
            try:
                await clients[i].send_broadcast_message([0], payload)
                seq += 1
            except Exception:
                pass
            await asyncio.sleep(0.005)

    tasks = [asyncio.ensure_future(rx(i)) for i in range(n_clients)]
    tasks += [asyncio.ensure_future(tx(i)) for i in range(2)]  # 2 publishers, 400 msg/s

    mem0 = None
    t0 = time.time()
    while time.time() - t0 < seconds:
        await asyncio.sleep(2)
        mem = torch.cuda.memory_allocated()
        if mem0 is None:
            mem0 = mem
        print(f"t={time.time()-t0:5.1f}s received={sum(received)} "
              f"hbm_alloc={mem/2**20:.1f} MiB (delta {(mem-mem0)/2**20:+.2f})", flush=True)
    stop.set()
    for t in tasks:
        t.cancel()
    total = sum(received)
    # stability bound, not a throughput bound: the soak's Python clients are
    # event-loop-limited (the engine delivers Gdeliveries/s; these asyncio
    # clients consume a few hundred/s)
    expected_min = 20 * seconds
    assert total > expected_min, f"too few deliveries: {total} < {expected_min}"
    mem_end = torch.cuda.memory_allocated()
    assert mem_end - mem0 < 32 * 2**20, f"HBM allocator grew {mem_end - mem0} bytes"
    print(f"SOAK OK: {total} deliveries over {seconds}s, allocator flat")
    for c in clients:
        c.close()
    await marshal.close()
    await broker.close()


if __name__ == "__main__":
    ap = argparse.ArgumentParser()
    ap.add_argument("--seconds", type=int, default=30)
    args = ap.parse_args()
    asyncio.run(main(args.seconds))
