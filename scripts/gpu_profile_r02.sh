#!/bin/bash
# Round-2 profiling pass: kernel stats for the headline bench and the K1 v2
# verifier, plus a socket soak and a 300k-offered point.
set -x
mkdir -p /root/repo/gpurun_out
export TMPDIR=/tmp

# headline bench kernel profile
timeout 400 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/prof_bench -o b -- \
  python bench.py --steps 10 --warmup 3 > /root/repo/gpurun_out/prof_bench.log 2>&1
python scripts/prof_extract.py /root/repo/gpurun_out/prof_bench \
  /root/repo/gpurun_out/bench_kernel_stats_r02.txt >> /root/repo/gpurun_out/prof_bench.log 2>&1

# K1 v2 kernel profile (2-lane Fp2 decomposition + fixed-g2 lines)
timeout 600 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/prof_k1v2 -o k -- \
  python - > /root/repo/gpurun_out/prof_k1v2.log 2>&1 <<'PYEOF'
import torch
from pushcdn_amd.crypto import bls
from pushcdn_amd.ops import get_gpu_ops
ops = get_gpu_ops()
ns = bls.USER_MARSHAL_NAMESPACE
N = 10240
vks, sigs, msgs, offsets = [], [], bytearray(), [0]
cache = {}
for i in range(N):
    s = i % 64
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
probe = torch.zeros(1, dtype=torch.uint8, device="cuda")
lines = ops.precompute_g2_lines(probe)
ok = ops.bls_verify_batch2(vks_t, sigs_t, msgs_t, moff_t, lines)
torch.cuda.synchronize()
print("v2 verified:", int(ok.sum()), "/", N)
ok1 = ops.bls_verify_batch(vks_t, sigs_t, msgs_t, moff_t)  # v1 in same trace for A/B
torch.cuda.synchronize()
print("v1 verified:", int(ok1.sum()), "/", N)
PYEOF
python scripts/prof_extract.py /root/repo/gpurun_out/prof_k1v2 \
  /root/repo/gpurun_out/k1v2_kernel_stats_r02.txt >> /root/repo/gpurun_out/prof_k1v2.log 2>&1

# socket: 300k offered (ceiling probe) + 60 s soak at the 120k clean point
timeout 200 python scripts/bench_socket.py --subs 50 --sub-procs 24 --senders 10 \
  --rate 300000 --seconds 10 --tag v4r300 2>&1 | tail -1
timeout 300 python scripts/bench_socket.py --subs 50 --sub-procs 16 --senders 6 \
  --rate 120000 --seconds 60 --tag soak120 2>&1 | tail -1

tail -3 /root/repo/gpurun_out/prof_bench.log
tail -4 /root/repo/gpurun_out/prof_k1v2.log
head -8 /root/repo/gpurun_out/bench_kernel_stats_r02.txt
head -6 /root/repo/gpurun_out/k1v2_kernel_stats_r02.txt
