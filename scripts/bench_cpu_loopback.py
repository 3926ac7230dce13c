#!/usr/bin/env python3
"""BASELINE config 1: single broker + marshal + 2 clients over loopback TCP,
direct-message echo (the reference's process-compose path, CPU only).

Measures msgs/s and p50 round-trip latency of the full host control-plane
path: client -> TCP -> broker (auth'd session) -> DirectMap -> TCP -> client.
"""

import asyncio
import json
import statistics
import sys
import tempfile
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

from pushcdn_amd.broker.service import Broker, BrokerConfig
from pushcdn_amd.client import Client, ClientConfig
from pushcdn_amd.crypto import bls
from pushcdn_amd.marshal import Marshal, MarshalConfig
from pushcdn_amd.proto.transports.tcp import Tcp


async def main(n_msgs: int = 2000, payload: int = 1024, native: bool = False) -> None:
    if native:
        from pushcdn_amd.proto.transports.tcp_native import TcpNative as Proto
    else:
        Proto = Tcp
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
    ))
    await broker.start()
    # fix up advertise endpoints with the real bound ports
    pub_port = broker._user_listener.port
    priv_port = broker._broker_listener.port
    broker.config.public_advertise_endpoint = f"127.0.0.1:{pub_port}"
    broker.config.private_advertise_endpoint = f"127.0.0.1:{priv_port}"
    from pushcdn_amd.discovery import BrokerIdentifier

    broker.identity = BrokerIdentifier(f"127.0.0.1:{pub_port}", f"127.0.0.1:{priv_port}")
    broker.discovery.identity = broker.identity
    broker.connections.identity = broker.identity
    await broker.discovery.perform_heartbeat(0, 600)

    marshal = Marshal(MarshalConfig(bind_endpoint="127.0.0.1:0", discovery_endpoint=db, protocol=Proto))
    await marshal.start()
    marshal_ep = f"127.0.0.1:{marshal._listener.port}"

    alice = Client(ClientConfig(endpoint=marshal_ep, keypair=bls.KeyPair.from_seed(1),
                                subscribed_topics=[0], protocol=Proto))
    bob = Client(ClientConfig(endpoint=marshal_ep, keypair=bls.KeyPair.from_seed(2),
                              subscribed_topics=[0], protocol=Proto))
    await alice.ensure_initialized()
    await bob.ensure_initialized()
    await asyncio.sleep(0.2)

    payload_bytes = bytes(payload)
    # warmup
    for _ in range(50):
        await alice.send_direct_message(bob.public_key, payload_bytes)
        await bob.receive_message()

    lat = []
    t0 = time.perf_counter()
    for _ in range(n_msgs):
        s = time.perf_counter()
        await alice.send_direct_message(bob.public_key, payload_bytes)
        await bob.receive_message()
        lat.append(time.perf_counter() - s)
    dt = time.perf_counter() - t0

    # burst (pipelined) phase: all messages in flight at once — exercises
    # the writer task's frame coalescing under queue depth
    async def blast():
        for _ in range(n_msgs):
            await alice.send_direct_message(bob.public_key, payload_bytes)

    tb = time.perf_counter()
    sender = asyncio.get_running_loop().create_task(blast())
    for _ in range(n_msgs):
        await bob.receive_message()
    await sender
    burst_dt = time.perf_counter() - tb

    print(json.dumps({
        "config": ("cpu-loopback native-pump" if native else "cpu-loopback") + ": marshal + 1 broker + 2 clients, TCP, direct echo",
        "burst_msgs_per_sec": n_msgs / burst_dt,
        "msgs_per_sec": n_msgs / dt,
        "p50_latency_ms": statistics.median(lat) * 1000,
        "p99_latency_ms": sorted(lat)[int(len(lat) * 0.99)] * 1000,
        "payload_bytes": payload,
        "n_msgs": n_msgs,
    }))
    alice.close()
    bob.close()
    await marshal.close()
    await broker.close()


if __name__ == "__main__":
    import sys as _sys
    _native = "--native" in _sys.argv
    asyncio.run(main(native=_native))
