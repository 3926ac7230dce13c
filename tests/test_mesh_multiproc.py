"""Multi-process mesh path over gloo (world_size 2, CPU) — validates the
exact all-gather exchange bench.py uses for the RCCL/xGMI broker mesh, and
that every broker delivers every rank's messages to its local subscribers."""

import json
import os
import subprocess
import sys
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent


def test_bench_world_size_2_gloo():
    env = dict(os.environ)
    env["MASTER_ADDR"] = "127.0.0.1"
    env["MASTER_PORT"] = "29511"
    out = subprocess.run(
        [
            sys.executable, "-m", "torch.distributed.run",
            "--nnodes=1", "--nproc-per-node=2",
            "--master-addr", "127.0.0.1", "--master-port", "29511",
            str(REPO / "bench.py"),
            "--gpus", "2", "--steps", "2", "--warmup", "1",
            "--batch", "4", "--payload", "64", "--subscribers", "32",
            "--device", "cpu",
        ],
        capture_output=True, text=True, timeout=300, env=env, cwd=str(REPO),
    )
    assert out.returncode == 0, out.stderr[-3000:]
    line = [l for l in out.stdout.splitlines() if l.startswith("{")][-1]
    result = json.loads(line)
    assert result["n_gpus"] == 2
    assert result["value"] > 0
    assert result["config"]["subscribers_per_broker"] == 16
    assert result["config"]["drops"] == 0
    # whole-node deliveries: both ranks' 4-msg batches to 16 local users each
    assert result["config"]["deliveries_per_step_node"] == 2 * 4 * 16


def test_mesh_exchange_semantics():
    """Two engines exchanging batches = every local user of each broker gets
    both brokers' messages (1-hop mesh, no re-forwarding)."""
    import torch

    from pushcdn_amd.broker.gpu_engine import GpuBrokerEngine, parse_ring_records
    from pushcdn_amd.proto import message as m

    engines = [
        GpuBrokerEngine(device="cpu", n_users=4, ring_bytes=1 << 12,
                        direct_table_size=64, use_gpu_ops=False)
        for _ in range(2)
    ]
    for eng in engines:
        eng.subscribe_all([0])

    batches = []
    for r in range(2):
        msg = m.Broadcast([0], f"from-broker-{r}".encode())
        raw = m.serialize(msg)
        batches.append((raw, [0, len(raw)]))

    # mesh all-gather: each broker processes every broker's batch in rank order
    for eng in engines:
        for raw, offsets in batches:
            buf, off = eng.ingest(raw, offsets)
            eng.tick(buf, off, host_batch=raw, host_offsets=offsets)

    for eng in engines:
        wpos = eng.drain_cursors()
        for u in range(4):
            recs = parse_ring_records(eng.read_ring(u), int(wpos[u]))
            assert [p for _, p in recs] == [b"from-broker-0", b"from-broker-1"]
