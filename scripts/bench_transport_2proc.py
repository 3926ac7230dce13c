#!/usr/bin/env python3
"""Two-process transport throughput: server in a child process, client in
this one — the deployment shape (a broker and a remote client), unlike
bench_transport.py's single-loop harness where the pure-Python QUIC
endpoint's "network" is an in-loop function call and cross-thread wakeup
latency counts against the native pump only.

Usage: python scripts/bench_transport_2proc.py --proto quic-native
       python scripts/bench_transport_2proc.py --proto quic --sizes 1048576:100
"""

import argparse
import asyncio
import json
import os
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

from pushcdn_amd.proto import message as m
from pushcdn_amd.proto.limiter import Limiter

SIZES = [(100, 4000), (1 << 10, 4000), (100 << 10, 800), (1 << 20, 200),
         (10 << 20, 24), (100 << 20, 4)]


def _share_ca(proto, name: str) -> None:
    # cross-process TLS: both ends must use ONE CA (each process otherwise
    # generates its own per-boot local CA and the client rejects the server)
    ca_cert, ca_key = os.environ.get("BENCH_CA_CERT"), os.environ.get("BENCH_CA_KEY")
    if ca_cert:
        # make the process-local CA BE the shared one (use_local_authority
        # clients trust local_ca(); subprocesses would otherwise mint their own)
        from pushcdn_amd.crypto import tls as tlslib
        tlslib._LOCAL_CA = (ca_cert, ca_key)
    if ca_cert and hasattr(proto, "ca_cert_path"):
        proto.ca_cert_path = ca_cert
        proto.ca_key_path = ca_key
    if ca_cert and name == "tls":
        proto.ca_cert_path = ca_cert
        proto.ca_key_path = ca_key


def get_proto(name: str):
    if name == "quic":
        from pushcdn_amd.proto.transports.quic import Quic as proto
    elif name == "quic-native":
        from pushcdn_amd.proto.transports.quic import QuicNative as proto
    elif name == "tls":
        from pushcdn_amd.proto.transports.tcp_tls import TcpTls as proto
    elif name == "native":
        from pushcdn_amd.proto.transports.tcp_native import TcpNative as proto
    else:
        from pushcdn_amd.proto.transports.tcp import Tcp as proto
    return proto


async def server_main(proto_name: str, sizes) -> None:
    proto = get_proto(proto_name)
    _share_ca(proto, proto_name)
    limiter = Limiter(global_memory_pool_size=1 << 31)
    listener = await proto.bind("127.0.0.1:0", None, None)
    print(json.dumps({"port": listener.port}), flush=True)
    for _, n_msgs in sizes:
        conn = await (await listener.accept()).finalize(limiter)
        for _ in range(n_msgs):
            await conn.recv_message()
        await conn.send_message(m.Direct(b"s", b"done"))
        await conn.soft_close()
    await listener.close()


async def client_main(proto_name: str, sizes, port: int) -> None:
    proto = get_proto(proto_name)
    _share_ca(proto, proto_name)
    limiter = Limiter(global_memory_pool_size=1 << 31)
    out = {}
    for size, n_msgs in sizes:
        conn = await proto.connect(f"127.0.0.1:{port}", True, limiter)
        payload = bytes(size)
        t0 = time.perf_counter()
        for _ in range(n_msgs):
            await conn.send_message(m.Direct(b"c", payload))
        await conn.recv_message()
        dt = time.perf_counter() - t0
        await conn.soft_close()
        key = f"{size}B" if size < 1024 else (
            f"{size >> 10}KiB" if size < (1 << 20) else f"{size >> 20}MiB")
        out[key] = round(n_msgs * size / dt / 1e9, 3)
        print(f"  {key}: {out[key]} GB/s", file=sys.stderr, flush=True)
    print(json.dumps({
        "config": proto_name + " raw transfer, two processes, loopback (GB/s)",
        "gbps": out,
    }))


async def server_fan(proto_name: str, k: int, size: int, n_msgs: int) -> None:
    """Fan-in: K concurrent clients blast one server — the broker ingest
    shape.  The win condition for the native pump: per-datagram work for
    ALL connections stays off the (busy) server loop."""
    proto = get_proto(proto_name)
    _share_ca(proto, proto_name)
    limiter = Limiter(global_memory_pool_size=1 << 32)
    listener = await proto.bind("127.0.0.1:0", None, None)
    print(json.dumps({"port": listener.port}), flush=True)

    busy_us = int(os.environ.get("BENCH_BUSY_US", "0"))

    async def one(unf):
        conn = await unf.finalize(limiter)
        for _ in range(n_msgs):
            await conn.recv_message()
            if busy_us:   # simulated broker work (routing tick share)
                t_end = time.perf_counter() + busy_us / 1e6
                while time.perf_counter() < t_end:
                    pass
        await conn.send_message(m.Direct(b"s", b"done"))
        await conn.soft_close()

    unfs = [await listener.accept() for _ in range(k)]
    t0 = time.perf_counter()
    await asyncio.gather(*(one(u) for u in unfs))
    dt = time.perf_counter() - t0
    print(json.dumps({"elapsed": dt,
                      "agg_gbps": round(k * n_msgs * size / dt / 1e9, 3)}),
          flush=True)
    await listener.close()


async def client_fan(proto_name: str, size: int, n_msgs: int, port: int) -> None:
    proto = get_proto(proto_name)
    _share_ca(proto, proto_name)
    limiter = Limiter(global_memory_pool_size=1 << 31)
    conn = await proto.connect(f"127.0.0.1:{port}", True, limiter)
    payload = bytes(size)
    for _ in range(n_msgs):
        await conn.send_message(m.Direct(b"c", payload))
    await conn.recv_message()
    await conn.soft_close()


def main() -> None:
    p = argparse.ArgumentParser()
    p.add_argument("--proto", default="quic-native",
                   choices=["tcp", "native", "tls", "quic", "quic-native"])
    p.add_argument("--sizes", default="",
                   help="size:count,size:count override")
    p.add_argument("--server", action="store_true", help=argparse.SUPPRESS)
    p.add_argument("--port", type=int, default=0, help=argparse.SUPPRESS)
    p.add_argument("--fan", type=int, default=0,
                   help="K concurrent client processes (fan-in mode)")
    p.add_argument("--fan-size", type=int, default=100 << 10)
    p.add_argument("--fan-msgs", type=int, default=400)
    p.add_argument("--fan-client", action="store_true", help=argparse.SUPPRESS)
    a = p.parse_args()
    sizes = SIZES
    if a.sizes:
        sizes = [tuple(int(x) for x in part.split(":"))
                 for part in a.sizes.split(",")]
    if a.fan_client:
        asyncio.run(client_fan(a.proto, a.fan_size, a.fan_msgs, a.port))
        return
    if a.server and a.fan:
        if os.environ.get("BENCH_PROFILE"):
            import cProfile, pstats, io
            pr = cProfile.Profile()
            pr.enable()
            asyncio.run(server_fan(a.proto, a.fan, a.fan_size, a.fan_msgs))
            pr.disable()
            sio = io.StringIO()
            pstats.Stats(pr, stream=sio).sort_stats("tottime").print_stats(22)
            print(sio.getvalue(), file=sys.stderr)
        else:
            asyncio.run(server_fan(a.proto, a.fan, a.fan_size, a.fan_msgs))
        return
    if a.server:
        asyncio.run(server_main(a.proto, sizes))
        return
    import subprocess
    from pushcdn_amd.crypto import tls as tlslib
    ca_cert, ca_key = tlslib.local_ca()
    os.environ["BENCH_CA_CERT"] = ca_cert
    os.environ["BENCH_CA_KEY"] = ca_key
    root = str(Path(__file__).resolve().parent.parent)
    if a.fan:
        srv = subprocess.Popen(
            [sys.executable, __file__, "--proto", a.proto, "--server",
             "--fan", str(a.fan), "--fan-size", str(a.fan_size),
             "--fan-msgs", str(a.fan_msgs)],
            stdout=subprocess.PIPE, stderr=sys.stderr, text=True, cwd=root)
        clients = []
        try:
            port = json.loads(srv.stdout.readline())["port"]
            for _ in range(a.fan):
                clients.append(subprocess.Popen(
                    [sys.executable, __file__, "--proto", a.proto,
                     "--fan-client", "--fan-size", str(a.fan_size),
                     "--fan-msgs", str(a.fan_msgs), "--port", str(port)],
                    stderr=sys.stderr, cwd=root))
            res = json.loads(srv.stdout.readline())
            print(json.dumps({
                "config": (f"{a.proto} fan-in x{a.fan} clients, "
                           f"{a.fan_size}B msgs, two+ processes"),
                **res}))
            for c in clients:
                c.wait(timeout=60)
            srv.wait(timeout=30)
        finally:
            for c in clients:
                if c.poll() is None:
                    c.kill()
            if srv.poll() is None:
                srv.kill()
        return
    srv = subprocess.Popen(
        [sys.executable, __file__, "--proto", a.proto, "--server"]
        + (["--sizes", a.sizes] if a.sizes else []),
        stdout=subprocess.PIPE, stderr=sys.stderr, text=True, cwd=root)
    try:
        port = json.loads(srv.stdout.readline())["port"]
        asyncio.run(client_main(a.proto, sizes, port))
        srv.wait(timeout=30)
    finally:
        if srv.poll() is None:
            srv.kill()


if __name__ == "__main__":
    main()
