#!/usr/bin/env python3
"""Raw transport throughput by message size — the reference's
cdn-proto/benches/protocols.rs:103-152 harness (TCP/QUIC at
{100 B, 1 KiB, 100 KiB, 10 MiB, 100 MiB}), here for the asyncio TCP
transport and the native C++ pump over loopback.

Usage: python scripts/bench_transport.py [--native]
"""

import argparse
import asyncio
import json
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

from pushcdn_amd.proto import message as m
from pushcdn_amd.proto.limiter import Limiter

SIZES = [(100, 2000), (1 << 10, 2000), (100 << 10, 400), (10 << 20, 24), (100 << 20, 4)]


async def run_one(proto, payload_bytes: int, n_msgs: int) -> float:
    limiter = Limiter(global_memory_pool_size=1 << 31)
    listener = await proto.bind("127.0.0.1:0", None, None)
    endpoint = f"127.0.0.1:{listener.port}"
    payload = bytes(payload_bytes)

    async def server():
        conn = await (await listener.accept()).finalize(limiter)
        for _ in range(n_msgs):
            await conn.recv_message()
        await conn.send_message(m.Direct(b"s", b"done"))
        await conn.soft_close()

    async def client() -> float:
        conn = await proto.connect(endpoint, True, limiter)
        t0 = time.perf_counter()
        for _ in range(n_msgs):
            await conn.send_message(m.Direct(b"c", payload))
        await conn.recv_message()  # server saw everything
        dt = time.perf_counter() - t0
        await conn.soft_close()
        return dt

    _, dt = await asyncio.gather(server(), client())
    await listener.close()
    return n_msgs * payload_bytes / dt


async def main(native: bool, proto_name: str = "") -> None:
    if proto_name == "quic":
        from pushcdn_amd.proto.transports.quic import Quic as proto
    elif proto_name == "quic-native":
        from pushcdn_amd.proto.transports.quic import QuicNative as proto
    elif proto_name == "tls":
        from pushcdn_amd.proto.transports.tcp_tls import TcpTls as proto
    elif native or proto_name == "native":
        from pushcdn_amd.proto.transports.tcp_native import TcpNative as proto
    else:
        from pushcdn_amd.proto.transports.tcp import Tcp as proto
    out = {}
    for size, n in SIZES:
        bps = await run_one(proto, size, n)
        key = f"{size}B" if size < 1024 else (
            f"{size >> 10}KiB" if size < (1 << 20) else f"{size >> 20}MiB")
        out[key] = round(bps / 1e9, 3)
    name = proto_name or ("tcp-native" if native else "tcp-asyncio")
    print(json.dumps({
        "config": name + " raw transfer, loopback (GB/s by message size)",
        "gbps": out,
    }))


if __name__ == "__main__":
    p = argparse.ArgumentParser()
    p.add_argument("--native", action="store_true")
    p.add_argument("--proto", default="",
                   choices=["", "tcp", "native", "tls", "quic", "quic-native"])
    a = p.parse_args()
    asyncio.run(main(a.native, a.proto))
