#!/bin/bash
# Round validation on the GPU box: gpu tests, bench, K1 microbench, rocprof.
set -x
mkdir -p gpurun_out
export TMPDIR=/tmp

# 1. full gpu test suite
timeout 900 python -m pytest tests/ -x -q -m gpu > gpurun_out/pytest_gpu.log 2>&1
echo "pytest_gpu exit: $?" >> gpurun_out/pytest_gpu.log

# 2. bench: default config (256 msgs/tick, 10k subscribers, 1 KiB)
timeout 600 python bench.py --steps 30 --warmup 10 > gpurun_out/bench1.json 2> gpurun_out/bench1.err

# 3. K1 batch-verify timing (10k auths — BASELINE config 2)
timeout 900 python - > gpurun_out/k1_timing.log 2>&1 <<'PYEOF'
import time, torch
from pushcdn_amd.crypto import bls
from pushcdn_amd.ops import get_gpu_ops
ops = get_gpu_ops()
ns = bls.USER_MARSHAL_NAMESPACE
N = 10000
kp = [bls.KeyPair.from_seed(i % 256) for i in range(N)]  # 256 distinct keys reused
vks, sigs, msgs, offsets = [], [], bytearray(), [0]
sig_cache = {}
for i in range(N):
    k = kp[i]
    msg = f"ts-{i%256}".encode()
    key = (k.private_key, msg)
    if key not in sig_cache:
        sig_cache[key] = bls.sign(k.private_key, ns, msg)
    vks.append(k.public_key); sigs.append(sig_cache[key])
    msgs += ns.encode() + msg + b"\x00"; offsets.append(len(msgs))
vks_t = torch.frombuffer(bytearray(b"".join(vks)), dtype=torch.uint8).to("cuda")
sigs_t = torch.frombuffer(bytearray(b"".join(sigs)), dtype=torch.uint8).to("cuda")
msgs_t = torch.frombuffer(bytearray(msgs), dtype=torch.uint8).to("cuda")
moff_t = torch.tensor(offsets, dtype=torch.int64, device="cuda")
# warmup small
ok = ops.bls_verify_batch(vks_t[:128*64], sigs_t[:64*64], msgs_t, moff_t[:65]); torch.cuda.synchronize()
t0 = time.time()
ok = ops.bls_verify_batch(vks_t, sigs_t, msgs_t, moff_t)
torch.cuda.synchronize()
dt = time.time() - t0
good = int(ok.sum())
print(f"K1: verified {N} BLS sigs in {dt*1000:.1f} ms -> {N/dt:.0f} verifies/s (all_valid={good==N}, good={good})")
# host comparison (sequential, same workload shape, 256 verifies scaled)
from pushcdn_amd.ops.build import build_core
core = build_core()
t0 = time.time()
for i in range(64):
    core.verify(vks[i], ns, f"ts-{i%256}".encode(), sigs[i])
host_dt = (time.time() - t0) / 64
print(f"host: {host_dt*1000:.2f} ms/verify -> {1/host_dt:.0f} verifies/s single-core")
print(f"speedup vs 1 host core: {(N/dt)/(1/host_dt):.1f}x")
PYEOF

# 4. rocprof kernel stats for the bench (short run)
cd /tmp && export TMPDIR=/tmp && cd /root/repo
timeout 600 rocprofv3 --kernel-trace --stats -d gpurun_out/prof -o bench_prof -- \
  python bench.py --steps 10 --warmup 3 > gpurun_out/bench_prof.log 2>&1
echo "rocprof exit: $?" >> gpurun_out/bench_prof.log
ls -la gpurun_out/prof >> gpurun_out/bench_prof.log 2>&1
find gpurun_out/prof -name "*stats*" -exec head -30 {} \; >> gpurun_out/bench_prof.log 2>&1

tail -5 gpurun_out/pytest_gpu.log gpurun_out/bench1.json gpurun_out/k1_timing.log gpurun_out/bench_prof.log
