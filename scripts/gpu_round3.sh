#!/bin/bash
set -x
mkdir -p gpurun_out
export TMPDIR=/tmp

timeout 900 python -m pytest tests/ -x -q -m gpu > gpurun_out/pytest_gpu.log 2>&1
echo "pytest_gpu exit: $?" >> gpurun_out/pytest_gpu.log

# headline bench after mesh refactor
timeout 600 python bench.py --steps 30 --warmup 10 > gpurun_out/bench_bcast.json 2>&1

# K1 with the x-chain final exponentiation
timeout 900 python - > gpurun_out/k1_timing.log 2>&1 <<'PYEOF'
import time, torch
from pushcdn_amd.crypto import bls
from pushcdn_amd.ops import get_gpu_ops
ops = get_gpu_ops()
ns = bls.USER_MARSHAL_NAMESPACE
for N in (4096, 10000, 65536):
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
    torch.cuda.synchronize(); t0 = time.time()
    ok = ops.bls_verify_batch(vks_t, sigs_t, msgs_t, moff_t)
    torch.cuda.synchronize(); dt = time.time() - t0
    print(f"K1 N={N}: {dt*1000:.1f} ms -> {N/dt:.0f} verifies/s (good={int(ok.sum())})")
PYEOF

tail -n 4 gpurun_out/pytest_gpu.log
tail -n 1 gpurun_out/bench_bcast.json
tail -n 4 gpurun_out/k1_timing.log
