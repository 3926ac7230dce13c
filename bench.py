#!/usr/bin/env python3
"""Flagship benchmark: broadcast msgs/sec through the GPU broker node.

Measures the BASELINE.json north-star metric — broadcast messages per second
(whole node) + p50 end-to-end latency, 1 KiB payloads — on N GPU-brokers
(one process per GPU, RCCL over xGMI via torch.distributed).

Per timed step (one routing tick per broker):
  1. H2D ingest of a batch of M serialized Cap'n Proto Broadcast messages
     (fresh copy every step — the socket-read analog)
  2. K4 parse_batch (on-device capnp parse)
  3. K2a topic_mask against the HBM subscription bitmap
  4. N>1: all-gather of the message batches over RCCL/xGMI
     (the broker->broker mesh fan-out; 1-hop, to_users_only semantics)
  5. K2b assign_emit + K3 fanout: payload copied into every local
     subscriber's egress ring in HBM (the reference's per-connection
     channel push, sender.rs:16-33)
  6. ring cursors D2H + reset (the drain/notify analog)

The subscriber population (default 10,000) is split evenly across brokers,
mirroring the marshal's least-connections placement; every subscriber is
subscribed to every benched topic, so every message is delivered to every
local subscriber of every broker (config 2/3 of BASELINE.json).

Work is synthetic: random 1 KiB payloads, serialized with the real wire
format. Nothing in the timed region is cached or skipped: every step
re-copies the wire batch H2D, re-parses, re-routes, re-copies every payload
to every subscriber ring, and re-drains the cursors.
"""

from __future__ import annotations

import argparse
import json
import os
import statistics
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent))

import torch

from pushcdn_amd.proto import message as msglib


def build_mixed_batch(n_msgs, payload_bytes, n_topics, n_users, seed):
    """Alternating Direct/Broadcast batch (config 4). BOTH message types are
    padded to one common wire size (Direct carries a recipient key, so it is
    a few words larger than Broadcast), which makes the whole batch uniform:
    identical offsets on every rank AND the flat ~100%-lane-utilization K3
    path instead of wave-per-pair. Trailing zero padding is inert to the
    capnp parser (it reads only the declared segment)."""
    import random

    rng = random.Random(seed)
    probe_b = msglib.serialize(msglib.Broadcast([0], bytes(payload_bytes)))
    probe_d = msglib.serialize(msglib.Direct(b"user-00000000", bytes(payload_bytes)))
    wire_len = (max(len(probe_b), len(probe_d)) + 15) & ~15
    buf = bytearray()
    offsets = [0]
    for i in range(n_msgs):
        payload = rng.randbytes(payload_bytes)
        if i % 2 == 0:
            msg = msglib.Broadcast([rng.randrange(n_topics)], payload)
        else:
            msg = msglib.Direct(f"user-{rng.randrange(n_users):08d}".encode(), payload)
        raw = msglib.serialize(msg)
        assert len(raw) <= wire_len
        buf += raw + b"\x00" * (wire_len - len(raw))
        offsets.append(len(buf))
    return bytes(buf), offsets, wire_len


def build_batch(n_msgs: int, payload_bytes: int, n_topics: int, seed: int):
    """Serialize one ingest batch; returns (buf, offsets, wire_len).  Every
    message start is 16-byte aligned (so the flat K3 kernel's 16 B units stay
    vector-aligned) and every message has the same wire length."""
    import random

    rng = random.Random(seed)
    buf = bytearray()
    offsets = [0]
    wire_len = None
    for i in range(n_msgs):
        payload = rng.randbytes(payload_bytes)
        msg = msglib.Broadcast([i % n_topics], payload)
        raw = msglib.serialize(msg)
        padded = (len(raw) + 15) & ~15
        if wire_len is None:
            wire_len = padded
        assert padded == wire_len, "non-uniform wire sizes in batch"
        buf += raw + b"\x00" * (padded - len(raw))
        offsets.append(len(buf))
    return bytes(buf), offsets, wire_len


def main() -> None:
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=30)
    p.add_argument("--warmup", type=int, default=10)
    p.add_argument("--batch", type=int, default=256, help="messages ingested per broker per step")
    p.add_argument("--payload", type=int, default=1024, help="payload bytes per message")
    p.add_argument("--subscribers", type=int, default=10_000, help="total subscriber population")
    p.add_argument("--topics", type=int, default=8)
    p.add_argument("--device", default=None, help="cpu to force the CPU reference path")
    p.add_argument("--mode", choices=["broadcast", "mixed"], default="broadcast",
                   help="broadcast: every subscriber gets every message (configs 2-3); "
                        "mixed: 50%% direct + 50%% broadcast, users spread over topics "
                        "(config 4: 64 KiB payloads at -100k clients)")
    p.add_argument("--ring-kb", type=int, default=0, help="override per-user ring size (KiB)")
    p.add_argument("--population", choices=["all", "modulo"], default="all",
                   help="all: every subscriber on every benched topic (dense fan-out); "
                        "modulo: user u subscribes only to topic u%%topics (sparse)")
    p.add_argument("--no-overlap", action="store_true",
                   help="disable the double-buffered ingest overlap (A/B)")
    p.add_argument("--graph", action="store_true",
                   help="hipGraph-capture the tick (measured ~2%% slower than "
                        "eager at this kernel count; kept for A/B)")
    args = p.parse_args()
    if args.mode == "mixed" and args.payload == 1024:
        args.payload = 65536  # config-4 default

    world_size = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))

    use_cpu = args.device == "cpu" or not torch.cuda.is_available()
    if use_cpu and args.device != "cpu":
        print("ERROR: no GPU available and --device cpu not requested", file=sys.stderr)
        sys.exit(1)

    if use_cpu:
        device = "cpu"
    else:
        torch.cuda.set_device(local_rank)
        device = f"cuda:{local_rank}"

    from pushcdn_amd.broker.gpu_engine import GpuBrokerEngine, ring_rec

    n_local_users = (args.subscribers + world_size - 1) // world_size
    ring_bytes = 1 << 9
    # Ring must hold one step's worth of WIRE messages (payload + ~64 B capnp
    # envelope), each as a 64-aligned ring record (see gpu_engine.ring_rec).
    wire_est = ring_rec((args.payload + 64 + 15) & ~15)
    need = world_size * args.batch * wire_est
    while ring_bytes < need * 2:
        ring_bytes <<= 1

    if args.mode == "mixed":
        # expected per-user bytes/tick is small (users spread over topics);
        # ring sized for bursts
        ring_bytes = (args.ring_kb or 1024) << 10
        pair_cap = 1 << 22
    else:
        pair_cap = max(1 << 20, world_size * args.batch * n_local_users)
        if args.ring_kb:
            ring_bytes = args.ring_kb << 10
    eng = GpuBrokerEngine(
        device=device,
        n_users=n_local_users,
        ring_bytes=ring_bytes,
        use_gpu_ops=not use_cpu,
        fanout_wire=True,       # forward raw wire bytes verbatim (reference semantics)
        direct_enabled=(args.mode == "mixed"),
        pair_capacity=pair_cap,
        direct_table_size=1 << max(10, (n_local_users * 2).bit_length()),
    )
    if args.mode == "mixed":
        eng.subscribe_modulo(args.topics)
        eng.register_direct_bulk(
            (f"user-{u:08d}".encode(), u) for u in range(n_local_users)
        )
    elif args.population == "modulo":
        eng.subscribe_modulo(args.topics)
    else:
        eng.subscribe_all(list(range(args.topics)))

    # Pre-serialize a few distinct wire batches (client-side work in the real
    # system); the H2D copy + full GPU pipeline still runs fresh every step.
    n_variants = 4
    builder = (
        (lambda **kw: build_mixed_batch(args.batch, args.payload, args.topics,
                                        n_local_users, kw["seed"]))
        if args.mode == "mixed"
        else (lambda **kw: build_batch(args.batch, args.payload, args.topics, kw["seed"]))
    )
    host_batches = [builder(seed=rank * 1000 + v) for v in range(n_variants)]
    wire_len = host_batches[0][2]
    # capacity EXACTLY the (uniform) batch size: the all-gathered buffer is
    # then gap-free, so one combined offsets table covers every rank's
    # messages and the whole step routes in ONE kernel-pipeline pass
    cap = len(host_batches[0][0])
    assert all(len(b) == cap for b, _, _ in host_batches)
    offsets_t = torch.tensor(host_batches[0][1], dtype=torch.int64)
    pinned = []
    for b, off, _ in host_batches:
        t = torch.zeros(cap, dtype=torch.uint8)
        t[: len(b)] = torch.frombuffer(bytearray(b), dtype=torch.uint8)
        if not use_cpu:
            t = t.pin_memory()
        pinned.append(t)

    dev_offsets = offsets_t.to(device)
    from pushcdn_amd.parallel.mesh import RcclMesh

    mesh = RcclMesh(torch.device(device), batch_capacity=cap)
    # combined offsets across the gathered [world_size * cap] buffer:
    # every rank's batch is identical in shape, so the combined table is
    # just the local table tiled with a +cap stride
    if world_size > 1:
        combined = torch.cat(
            [offsets_t[:-1] + r * cap for r in range(world_size)]
            + [torch.tensor([world_size * cap], dtype=torch.int64)]
        )
        dev_comb_offsets = combined.to(device)
    use_graph = (not use_cpu) and args.graph and args.mode == "broadcast"

    # Double-buffered ingest: the NEXT step's H2D copy runs on a side stream
    # while this step's kernels execute (the socket-read/compute overlap a
    # real broker gets from its per-connection reader tasks).
    if not use_cpu:
        dev_bufs = [torch.zeros(cap, dtype=torch.uint8, device=device) for _ in range(2)]
        copy_stream = torch.cuda.Stream(device=device)
        copy_done = [torch.cuda.Event(), torch.cuda.Event()]
        tick_done = [torch.cuda.Event(), torch.cuda.Event()]
        with torch.cuda.stream(copy_stream):
            dev_bufs[0].copy_(pinned[0], non_blocking=True)
            copy_done[0].record(copy_stream)

    def step(i: int) -> None:
        v = i % n_variants
        if use_cpu:
            buf = pinned[v]
        else:
            cur, nxt = i % 2, (i + 1) % 2
            if args.no_overlap:
                dev_bufs[cur].copy_(pinned[v], non_blocking=True)
            else:
                torch.cuda.current_stream().wait_event(copy_done[cur])
                with torch.cuda.stream(copy_stream):
                    copy_stream.wait_event(tick_done[nxt])  # buffer free?
                    dev_bufs[nxt].copy_(pinned[(i + 1) % n_variants], non_blocking=True)
                    copy_done[nxt].record(copy_stream)
            buf = dev_bufs[cur]
        # broker->broker mesh: all-gather this tick's batches over xGMI,
        # then route ALL ranks' messages in one pipeline pass
        if world_size > 1:
            gathered = mesh.exchange_flat(buf)
            if use_cpu:
                hb = bytes(gathered.numpy().tobytes())
                eng.tick(gathered, dev_comb_offsets, host_batch=hb,
                         host_offsets=[int(x) for x in combined],
                         uniform_wire_len=wire_len)
            elif use_graph:
                eng.tick_graphed(gathered, dev_comb_offsets, wire_len)
            else:
                eng.tick(gathered, dev_comb_offsets, uniform_wire_len=wire_len)
        elif use_graph:
            eng.tick_graphed(buf, dev_offsets, wire_len)
        else:
            eng.tick(
                buf,
                dev_offsets,
                host_batch=None if not use_cpu else host_batches[v][0],
                host_offsets=None if not use_cpu else host_batches[v][1],
                uniform_wire_len=wire_len,
            )
        if not use_cpu:
            tick_done[i % 2].record(torch.cuda.current_stream())
        eng.drain_cursors()

    def barrier_sync() -> None:
        mesh.barrier()
        if not use_cpu:
            torch.cuda.synchronize()

    for i in range(args.warmup):
        step(i)
    barrier_sync()

    step_times = []
    t0 = time.perf_counter()
    for i in range(args.steps):
        s = time.perf_counter()
        step(args.warmup + i)
        if not use_cpu:
            torch.cuda.synchronize()
        step_times.append(time.perf_counter() - s)
    barrier_sync()
    t1 = time.perf_counter()
    elapsed = t1 - t0

    # MAX elapsed over ranks (the slowest rank defines the job)
    elapsed = mesh.max_over_ranks(elapsed)

    total_msgs = world_size * args.batch * args.steps
    msgs_per_sec = total_msgs / elapsed
    ms_per_step = elapsed / args.steps * 1000
    p50_ms = statistics.median(step_times) * 1000
    if args.mode == "mixed":
        # ~half the messages are broadcasts to n_local/topics subscribers,
        # half are directs to one user
        deliveries_per_step = world_size * (
            (args.batch // 2) * (n_local_users // max(1, args.topics)) + args.batch // 2
        )
    elif args.population == "modulo":
        deliveries_per_step = world_size * args.batch * (n_local_users // max(1, args.topics))
    else:
        deliveries_per_step = world_size * args.batch * n_local_users

    if rank == 0:
        result = {
            "metric": "broadcast_msgs_per_sec" if args.mode == "broadcast"
                      else "mixed_direct_broadcast_msgs_per_sec",
            "value": msgs_per_sec,
            "unit": "msgs/s",
            "n_gpus": world_size,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "uint8",
            "data": "synthetic",
            "config": {
                "model": "gpu-broker-mesh",
                "payload_bytes": args.payload,
                "batch_msgs_per_broker": args.batch,
                "subscribers_total": args.subscribers,
                "subscribers_per_broker": n_local_users,
                "topics": args.topics,
                "deliveries_per_step_node": deliveries_per_step,
                "deliveries_per_sec_node": deliveries_per_step * args.steps / elapsed,
                "p50_e2e_latency_ms": p50_ms,
                "mode": args.mode,
                "population": args.population,
                "drops": int(eng._drops.cpu()[0]) if not use_cpu else 0,
                "egress_hbm_gb": round(n_local_users * ring_bytes / 2**30, 1),
                "parallelism": f"mesh{world_size} (RCCL all-gather over xGMI)" if world_size > 1 else "single-broker",
                "global_batch": world_size * args.batch,
                "seq_len": args.payload,
            },
        }
        print(json.dumps(result))

    if mesh.enabled:
        mesh.dist.destroy_process_group()


if __name__ == "__main__":
    main()
