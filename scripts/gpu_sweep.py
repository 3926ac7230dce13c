#!/usr/bin/env python3
"""Grid + batch sweeps for the broadcast tick (within-process A/B)."""

import random
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch

from pushcdn_amd.broker.gpu_engine import GpuBrokerEngine
from pushcdn_amd.proto import message as msglib


def build(n_msgs):
    rng = random.Random(0)
    buf = bytearray()
    offsets = [0]
    wl = None
    for i in range(n_msgs):
        raw = msglib.serialize(msglib.Broadcast([i % 8], rng.randbytes(1024)))
        padded = (len(raw) + 15) & ~15
        wl = padded
        buf += raw + b"\x00" * (padded - len(raw))
        offsets.append(len(buf))
    return bytes(buf), offsets, wl


def bench_tick(batch, steps=30, grid=0):
    eng = GpuBrokerEngine(device="cuda:0", n_users=10000, ring_bytes=1 << 22,
                          fanout_wire=True, direct_enabled=False, pair_capacity=8 << 20)
    eng.subscribe_all(list(range(8)))
    buf, offsets, wl = build(batch)
    dbuf, doff = eng.ingest(buf, offsets)
    # patch the flat2 grid for the sweep
    ops = eng._ops
    orig = ops.fanout_flat2
    if grid:
        def patched(*a):
            a = list(a)
            a[-1] = grid
            return orig(*a)
        eng._ops = type("O", (), {k: getattr(ops, k) for k in dir(ops) if not k.startswith("__")})()
        eng._ops.fanout_flat2 = patched
    for _ in range(5):
        eng.tick(dbuf, doff, uniform_wire_len=wl)
        eng.drain_cursors()
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(steps):
        eng.tick(dbuf, doff, uniform_wire_len=wl)
        eng.drain_cursors()
    torch.cuda.synchronize()
    dt = (time.time() - t0) / steps
    return dt * 1000, batch / dt


for rnd in range(2):
    for grid in (4096, 8192, 16384):
        ms, rate = bench_tick(256, grid=grid)
        print(f"round {rnd} grid={grid:6d} batch=256: {ms:.3f} ms/tick  {rate/1e3:.0f}k msgs/s")
for batch in (128, 256, 512, 1024):
    ms, rate = bench_tick(batch)
    print(f"batch={batch:5d}: {ms:.3f} ms/tick  {rate/1e3:.0f}k msgs/s  p50~{ms:.2f} ms")
