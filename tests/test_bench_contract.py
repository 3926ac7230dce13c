"""bench.py's output contract: the harness that runs this repo's benchmarks
parses ONE JSON line with specific fields — pin them so refactors cannot
silently break the measurement pipeline."""

import json
import subprocess
import sys
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent


def run_bench(*extra):
    out = subprocess.run(
        [sys.executable, str(REPO / "bench.py"), "--device", "cpu",
         "--steps", "2", "--warmup", "1", "--subscribers", "64",
         "--batch", "8", *extra],
        capture_output=True, text=True, timeout=240, cwd=REPO,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    return json.loads(out.stdout.strip().splitlines()[-1])


def test_bench_json_contract():
    d = run_bench()
    assert d["metric"] == "broadcast_msgs_per_sec"
    assert d["unit"] == "msgs/s"
    assert d["n_gpus"] == 1 and d["steps"] == 2 and d["warmup"] == 1
    assert d["value"] > 0 and d["ms_per_step"] > 0
    assert d["higher_is_better"] is True
    assert d["scaling"] == "weak"
    assert d["vs_baseline"] is None
    assert d["data"] == "synthetic"
    cfg = d["config"]
    assert cfg["payload_bytes"] == 1024
    assert cfg["global_batch"] == 8
    assert cfg["drops"] == 0
    assert "deliveries_per_sec_node" in cfg and "p50_e2e_latency_ms" in cfg
    # whole-job semantics: value x payload implies the node delivery rate
    assert abs(cfg["deliveries_per_sec_node"] / d["value"] - 64) < 1e-6


def test_bench_mixed_contract():
    d = run_bench("--mode", "mixed", "--payload", "2048", "--topics", "4")
    assert d["metric"] == "broadcast_msgs_per_sec" or "msgs" in d["metric"]
    assert d["config"]["mode"] == "mixed"
    assert d["value"] > 0
