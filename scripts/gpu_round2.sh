#!/bin/bash
# Round validation: full gpu tests (incl. GPU service e2e), mixed bench, K1 rocprof.
set -x
mkdir -p gpurun_out
export TMPDIR=/tmp

timeout 900 python -m pytest tests/ -x -q -m gpu > gpurun_out/pytest_gpu.log 2>&1
echo "pytest_gpu exit: $?" >> gpurun_out/pytest_gpu.log

# config 2/3 headline
timeout 600 python bench.py --steps 30 --warmup 10 > gpurun_out/bench_bcast.json 2>&1
# config 4: 64 KiB mixed at 100k clients (HBM pool sizing)
timeout 900 python bench.py --mode mixed --steps 20 --warmup 5 --subscribers 100000 --topics 64 --batch 256 > gpurun_out/bench_mixed.json 2>&1

# K1 kernel-trace profile (for profiles/)
cd /tmp && export TMPDIR=/tmp && cd /root/repo
timeout 900 rocprofv3 --kernel-trace --stats -d gpurun_out/prof_k1 -o k1_prof -- python - > gpurun_out/k1_prof.log 2>&1 <<'PYEOF'
import torch
from pushcdn_amd.crypto import bls
from pushcdn_amd.ops import get_gpu_ops
ops = get_gpu_ops()
ns = bls.USER_MARSHAL_NAMESPACE
N = 4096
vks, sigs, msgs, offsets = [], [], bytearray(), [0]
cache = {}
for i in range(N):
    s = i % 128
    if s not in cache:
        kp = bls.KeyPair.from_seed(s)
        msg = f"ts-{s}".encode()
        cache[s] = (kp.public_key, bls.sign(kp.private_key, ns, msg), msg)
    vk, sg, msg = cache[s]
    vks.append(vk); sigs.append(sg)
    msgs += ns.encode() + msg + b"\x00"; offsets.append(len(msgs))
vks_t = torch.frombuffer(bytearray(b"".join(vks)), dtype=torch.uint8).to("cuda")
sigs_t = torch.frombuffer(bytearray(b"".join(sigs)), dtype=torch.uint8).to("cuda")
msgs_t = torch.frombuffer(bytearray(msgs), dtype=torch.uint8).to("cuda")
moff_t = torch.tensor(offsets, dtype=torch.int64, device="cuda")
ok = ops.bls_verify_batch(vks_t, sigs_t, msgs_t, moff_t)
torch.cuda.synchronize()
print("verified:", int(ok.sum()), "/", N)
PYEOF

tail -n 6 gpurun_out/pytest_gpu.log
tail -n 1 gpurun_out/bench_bcast.json
tail -n 1 gpurun_out/bench_mixed.json
tail -n 3 gpurun_out/k1_prof.log
