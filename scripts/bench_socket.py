#!/usr/bin/env python3
"""Socket-level end-to-end benchmark with MULTI-PROCESS clients.

Round 1's e2e number (212k deliveries/s) was CLIENT-bound: 50 asyncio
clients shared one interpreter.  Here subscribers and senders run in their
own processes, so the measured number is the BROKER's: marshal auth + real
loopback TCP + GPU kernel routing + K7-compacted one-copy egress drain +
the C++ pump's coalesced writes.  Latency is CLIENT-OBSERVED: senders stamp
a wall-clock timestamp into each payload and subscribers diff it on
receipt (same host, same clock).

Roles (one script, spawned by the coordinator):
  coordinator  broker + marshal in this process; spawns workers; aggregates
  sub          N subscriber clients; counts frames in the measurement
               window, samples latency from the payload timestamps
  send         paced broadcast sender (token-bucket at --rate/senders)

Usage (GPU box):
  python scripts/bench_socket.py --subs 50 --sub-procs 10 --senders 2 \
      --rate 30000 --seconds 10 --payload 1024
"""

import argparse
import asyncio
import json
import struct
import sys
import tempfile
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

TOPIC = 0


def _seed_shared_ca() -> None:
    # TLS-bearing transports (quic profiles): every process must trust ONE
    # CA; the coordinator exports its local CA and workers adopt it
    import os

    cert, key = os.environ.get("SOCKBENCH_CA_CERT"), os.environ.get("SOCKBENCH_CA_KEY")
    if cert and key:
        from pushcdn_amd.crypto import tls as tlslib

        tlslib._LOCAL_CA = (cert, key)


def _proto(name: str):
    if name == "quic":
        from pushcdn_amd.proto.transports.quic import Quic
        return Quic
    if name == "quic-native":
        from pushcdn_amd.proto.transports.quic import QuicNative
        return QuicNative
    from pushcdn_amd.proto.transports.tcp_native import TcpNative
    return TcpNative


def _client(endpoint, seed, topics, transport="tcp-native"):
    from pushcdn_amd.client import Client, ClientConfig
    from pushcdn_amd.crypto import bls

    _seed_shared_ca()

    return Client(ClientConfig(endpoint=endpoint, keypair=bls.KeyPair.from_seed(seed),
                               subscribed_topics=list(topics),
                               protocol=_proto(transport)))


async def run_subscriber(args) -> None:
    import os

    from pushcdn_amd.proto import message as m

    clients = [_client(args.endpoint, args.seed + i, [TOPIC], args.transport)
               for i in range(args.clients)]
    for c in clients:
        await c.ensure_initialized()
    print("READY", flush=True)
    loop = asyncio.get_running_loop()
    line = await loop.run_in_executor(None, sys.stdin.readline)
    t0, t1 = json.loads(line)

    count = 0
    lats = []

    count_mode = bool(int(os.environ.get("SOCKBENCH_COUNT_MODE", "0")))

    async def drain(c):
        nonlocal count
        payload_off = None
        conn = await c._get_connection()
        if count_mode and hasattr(conn, "enable_count_mode"):
            conn.enable_count_mode()
        while time.time() < t1 + 0.5:
            try:
                if count_mode:
                    n, _nb, f = await asyncio.wait_for(conn.recv_drain(), timeout=0.5)
                else:
                    frames = await asyncio.wait_for(c.receive_raw_batch(), timeout=0.5)
                    n, f = len(frames), frames[0]
            except asyncio.TimeoutError:
                continue
            except Exception:
                return
            now = time.time()
            if now < t0 or now >= t1:
                continue
            count += n
            if payload_off is None:
                try:
                    payload_off = m.parse_offsets(f)["payload_off"]
                except Exception:
                    continue
            ts = struct.unpack_from("<d", f, payload_off)[0]
            lats.append(now - ts)

    await asyncio.gather(*(drain(c) for c in clients))
    for c in clients:
        c.close()
    lats.sort()
    step = max(1, len(lats) // 1000)
    sampled = [round(x, 6) for x in lats[::step]][:1200]
    print(json.dumps({"count": count, "lats": sampled}), flush=True)


async def run_sender(args) -> None:
    from pushcdn_amd.proto import message as m

    c = _client(args.endpoint, args.seed, [], args.transport)
    await c.ensure_initialized()
    # pre-serialize one FRAMED Broadcast; each burst patches timestamps into
    # a repeated template blob and hands the whole burst to the C++ pump in
    # ONE call (send_raw) — the serializer AND the per-message Python call
    # both leave the sender's hot loop
    payload = struct.pack("<d", 0.0) + b"\x00" * max(0, args.payload - 8)
    wire = m.serialize(m.Broadcast(topics=[TOPIC], message=payload))
    payload_off = m.parse_offsets(wire)["payload_off"]
    framed = struct.pack(">I", len(wire)) + wire
    conn = await c._get_connection()
    pump = cid = None
    if hasattr(conn, "pump_handle"):   # tcp-native fast path (send_raw bursts)
        pump, cid = conn.pump_handle()
    now = time.time()
    if args.t0 > now:
        await asyncio.sleep(args.t0 - now)
    sent = 0
    burst = max(1, int(args.rate / 500))  # ~500 pacing wakeups/s
    blob = bytearray(framed * burst)
    stride = len(framed)
    while time.time() < args.t1:
        deadline = time.time() + burst / args.rate
        for j in range(burst):
            struct.pack_into("<d", blob, j * stride + 4 + payload_off, time.time())
        if pump is not None:
            if not pump.send_raw(cid, bytes(blob)):
                break
            sent += burst
            while pump.send_backlog(cid) > (64 << 20):
                await asyncio.sleep(0.001)
        else:
            # transport-generic path (QUIC profiles): client API sends
            try:
                for j in range(burst):
                    pl = struct.pack("<d", time.time()) + b"\x00" * max(0, args.payload - 8)
                    await c.send_broadcast_message([TOPIC], pl)
            except Exception:
                break
            sent += burst
        dt = deadline - time.time()
        if dt > 0:
            await asyncio.sleep(dt)
    c.close()
    print(json.dumps({"sent": sent}), flush=True)


async def run_coordinator(args) -> None:
    import os

    os.environ.setdefault("PUSHCDN_PUMP_SHARDS", str(args.pump_shards))
    if args.count_mode:
        os.environ["SOCKBENCH_COUNT_MODE"] = "1"
    import torch

    from pushcdn_amd.broker.service import Broker, BrokerConfig
    from pushcdn_amd.crypto import bls
    from pushcdn_amd.discovery import BrokerIdentifier
    from pushcdn_amd.marshal import Marshal, MarshalConfig
    from pushcdn_amd.proto.transports.tcp_native import TcpNative

    user_proto = _proto(args.transport)
    device = args.device or ("cuda:0" if torch.cuda.is_available() else "cpu")
    db = tempfile.mktemp(suffix=".db")
    broker = Broker(BrokerConfig(
        public_bind_endpoint="127.0.0.1:0",
        public_advertise_endpoint="127.0.0.1:0",
        private_bind_endpoint="127.0.0.1:0",
        private_advertise_endpoint="127.0.0.1:0",
        discovery_endpoint=db,
        keypair=bls.KeyPair.from_seed(1000),
        user_protocol=user_proto,
        broker_protocol=TcpNative,
        data_plane="gpu",
        gpu_device=device,
        gpu_max_users=max(64, args.subs + args.senders + 8),
        gpu_ring_bytes=args.ring_bytes,
        gpu_tick_interval_s=args.tick_s,
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
                                    discovery_endpoint=db, protocol=user_proto))
    await marshal.start()
    if args.transport != "tcp-native":
        import os as _os

        from pushcdn_amd.crypto import tls as tlslib

        _ca_cert, _ca_key = tlslib.local_ca()
        _os.environ["SOCKBENCH_CA_CERT"] = _ca_cert
        _os.environ["SOCKBENCH_CA_KEY"] = _ca_key
    ep = f"127.0.0.1:{marshal._listener.port}"

    # subscriber workers
    per = args.subs // args.sub_procs
    extra = args.subs - per * args.sub_procs
    sub_procs = []
    seed = 100
    for w in range(args.sub_procs):
        n = per + (1 if w < extra else 0)
        if n == 0:
            continue
        p = await asyncio.create_subprocess_exec(
            sys.executable, __file__, "--role", "sub", "--endpoint", ep,
            "--clients", str(n), "--seed", str(seed),
            "--transport", args.transport,
            stdin=asyncio.subprocess.PIPE, stdout=asyncio.subprocess.PIPE,
            limit=32 << 20)
        sub_procs.append(p)
        seed += n

    for p in sub_procs:
        line = await p.stdout.readline()
        assert line.strip() == b"READY", line

    t0 = time.time() + 1.0
    t1 = t0 + args.seconds
    window = json.dumps([t0, t1]).encode() + b"\n"
    for p in sub_procs:
        p.stdin.write(window)
        await p.stdin.drain()

    send_procs = []
    for s in range(args.senders):
        p = await asyncio.create_subprocess_exec(
            sys.executable, __file__, "--role", "send", "--endpoint", ep,
            "--seed", str(5000 + s), "--rate", str(args.rate / args.senders),
            "--payload", str(args.payload), "--t0", str(t0), "--t1", str(t1),
            "--transport", args.transport,
            stdout=asyncio.subprocess.PIPE)
        send_procs.append(p)

    sent = 0
    for p in send_procs:
        out = await p.stdout.readline()
        sent += json.loads(out)["sent"]
    total = 0
    lats = []
    for p in sub_procs:
        out = await p.stdout.readline()
        r = json.loads(out)
        total += r["count"]
        lats.extend(r["lats"])
    await asyncio.gather(*(p.wait() for p in sub_procs + send_procs))

    lats.sort()
    dt = t1 - t0

    def pct(q):
        return round(lats[min(len(lats) - 1, int(q * len(lats)))] * 1e3, 3) if lats else None

    result = {
        "config": f"socket-e2e mp: {args.subs} subs/{args.sub_procs} procs, "
                  f"{args.senders} senders @ {args.rate} msgs/s, {args.payload}B, "
                  f"{device}",
        "deliveries_per_sec": round(total / dt, 1),
        "msgs_sent": sent,
        "deliveries_counted": total,
        "expected_deliveries_approx": sent * args.subs,
        "p50_ms": pct(0.50),
        "p99_ms": pct(0.99),
        "lat_samples": len(lats),
        "seconds": args.seconds,
    }
    ts = getattr(broker, "tick_stats", None)
    if ts and ts["ticks"]:
        result["broker_tick_stats"] = {
            k: (round(v / ts["ticks"] * 1e3, 3) if isinstance(v, float) else v)
            for k, v in ts.items()
        }
    print(json.dumps(result), flush=True)
    out = Path("gpurun_out")
    out.mkdir(exist_ok=True)
    name = f"bench_socket_{args.tag}.json" if args.tag else "bench_socket.json"
    (out / name).write_text(json.dumps(result, indent=1))
    await marshal.close()
    await broker.close()


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--role", default="coordinator", choices=["coordinator", "sub", "send"])
    p.add_argument("--endpoint")
    p.add_argument("--clients", type=int, default=5)
    p.add_argument("--seed", type=int, default=100)
    p.add_argument("--subs", type=int, default=50)
    p.add_argument("--sub-procs", type=int, default=10)
    p.add_argument("--senders", type=int, default=2)
    p.add_argument("--rate", type=float, default=30000)
    p.add_argument("--seconds", type=float, default=10)
    p.add_argument("--payload", type=int, default=1024)
    p.add_argument("--ring-bytes", type=int, default=1 << 23)
    p.add_argument("--tick-s", type=float, default=0.002)
    p.add_argument("--device", default=None)
    p.add_argument("--t0", type=float, default=0)
    p.add_argument("--t1", type=float, default=0)
    p.add_argument("--tag", default="")
    p.add_argument("--pump-shards", type=int, default=4)
    p.add_argument("--count-mode", action="store_true")
    p.add_argument("--transport", default="tcp-native",
                   choices=["tcp-native", "quic", "quic-native"],
                   help="user-plane transport (broker mesh stays tcp-native)")
    args = p.parse_args()
    if args.role == "sub":
        asyncio.run(run_subscriber(args))
    elif args.role == "send":
        asyncio.run(run_sender(args))
    else:
        asyncio.run(run_coordinator(args))


if __name__ == "__main__":
    main()
