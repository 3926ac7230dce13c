#!/usr/bin/env python3
"""End-to-end SERVICE throughput: marshal + one broker + N subscriber
clients over real loopback TCP, broadcast storm from one sender, measured
at the subscribers.  Exercises the full serving path: auth/permits, the
selected transport, the broker data plane (host or GPU kernels + C++
send_ring drain), and client receive.

Usage: python scripts/bench_service_e2e.py [--native] [--gpu] [--clients N]
       [--msgs M] [--payload B]
"""

import argparse
import asyncio
import json
import sys
import tempfile
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

from pushcdn_amd.broker.service import Broker, BrokerConfig
from pushcdn_amd.client import Client, ClientConfig
from pushcdn_amd.crypto import bls
from pushcdn_amd.discovery import BrokerIdentifier
from pushcdn_amd.marshal import Marshal, MarshalConfig


async def main(args) -> None:
    if args.native:
        from pushcdn_amd.proto.transports.tcp_native import TcpNative as Proto
    else:
        from pushcdn_amd.proto.transports.tcp import Tcp as Proto

    db = tempfile.mktemp(suffix=".db")
    broker = Broker(BrokerConfig(
        public_bind_endpoint="127.0.0.1:0",
        public_advertise_endpoint="127.0.0.1:0",
        private_bind_endpoint="127.0.0.1:0",
        private_advertise_endpoint="127.0.0.1:0",
        discovery_endpoint=db,
        keypair=bls.KeyPair.from_seed(1000),
        user_protocol=Proto,
        broker_protocol=Proto,
        data_plane="gpu" if args.gpu else "host",
        gpu_device=("cuda:0" if __import__("torch").cuda.is_available() else "cpu"),
        gpu_max_users=max(64, args.clients + 8),
        gpu_ring_bytes=1 << 22,
        gpu_tick_interval_s=0.002,
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
                                    discovery_endpoint=db, protocol=Proto))
    await marshal.start()
    ep = f"127.0.0.1:{marshal._listener.port}"

    sender = Client(ClientConfig(endpoint=ep, keypair=bls.KeyPair.from_seed(1),
                                 subscribed_topics=[], protocol=Proto))
    await sender.ensure_initialized()
    subs = []
    for i in range(args.clients):
        c = Client(ClientConfig(endpoint=ep, keypair=bls.KeyPair.from_seed(100 + i),
                                subscribed_topics=[0], protocol=Proto))
        await c.ensure_initialized()
        subs.append(c)
    await asyncio.sleep(0.5)

    payload = bytes(args.payload)

    async def drain(c, n):
        got = 0
        while got < n:
            got += len(await c.receive_messages(n - got))

    # warmup
    for _ in range(20):
        await sender.send_broadcast_message([0], payload)
    await asyncio.gather(*[drain(c, 20) for c in subs])

    if args.seconds:
        # soak mode: storm rounds until the clock runs out; every message
        # must reach every subscriber each round (loss check built in)
        import resource

        t0 = time.perf_counter()
        rounds = 0
        sent = 0
        rss0 = resource.getrusage(resource.RUSAGE_SELF).ru_maxrss
        while time.perf_counter() - t0 < args.seconds:
            for _ in range(args.msgs):
                await sender.send_broadcast_message([0], payload)
            await asyncio.gather(*[drain(c, args.msgs) for c in subs])
            sent += args.msgs
            rounds += 1
        dt = time.perf_counter() - t0
        rss1 = resource.getrusage(resource.RUSAGE_SELF).ru_maxrss
        print(json.dumps({
            "config": ("native-tcp" if args.native else "asyncio-tcp")
                      + ("+gpu-engine" if args.gpu else "+host-plane")
                      + f" SOAK {args.seconds}s: {args.clients} subs",
            "rounds": rounds,
            "deliveries_per_sec": round(sent * args.clients / dt, 1),
            "rss_growth_kb": rss1 - rss0,
            "elapsed_s": round(dt, 1),
        }))
        sender.close()
        for c in subs:
            c.close()
        await marshal.close()
        await broker.close()
        return

    t0 = time.perf_counter()

    async def blast():
        for _ in range(args.msgs):
            await sender.send_broadcast_message([0], payload)

    task = asyncio.get_running_loop().create_task(blast())
    await asyncio.gather(*[drain(c, args.msgs) for c in subs])
    await task
    dt = time.perf_counter() - t0

    deliveries = args.msgs * args.clients
    print(json.dumps({
        "config": ("native-tcp" if args.native else "asyncio-tcp")
                  + ("+gpu-engine" if args.gpu else "+host-plane")
                  + ("" if not args.gpu or __import__("torch").cuda.is_available() else "(cpu-ref)")
                  + f" e2e: {args.clients} subscribers, {args.payload}B",
        "msgs_per_sec": round(args.msgs / dt, 1),
        "deliveries_per_sec": round(deliveries / dt, 1),
        "elapsed_s": round(dt, 3),
        "msgs": args.msgs,
    }))
    sender.close()
    for c in subs:
        c.close()
    await marshal.close()
    await broker.close()


if __name__ == "__main__":
    p = argparse.ArgumentParser()
    p.add_argument("--native", action="store_true")
    p.add_argument("--gpu", action="store_true")
    p.add_argument("--clients", type=int, default=20)
    p.add_argument("--msgs", type=int, default=500)
    p.add_argument("--payload", type=int, default=1024)
    p.add_argument("--seconds", type=int, default=0, help="soak mode: run for N seconds")
    args = p.parse_args()
    asyncio.run(main(args))
