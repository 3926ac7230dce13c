#!/bin/bash
# Round validation on the GPU box: gpu tests, bench (nt A/B), rocprof.
set -x
mkdir -p gpurun_out
export TMPDIR=/tmp

timeout 900 python -m pytest tests/ -x -q -m gpu > gpurun_out/pytest_gpu.log 2>&1
echo "pytest_gpu exit: $?" >> gpurun_out/pytest_gpu.log

# bench A/B: nt vs plain fanout stores
timeout 600 python bench.py --steps 30 --warmup 10 > gpurun_out/bench_nt.json 2>&1
timeout 600 python - > gpurun_out/fanout_ab.log 2>&1 <<'PYEOF'
# within-process A/B of the K3 variants at bench shape
import time, torch, random
from pushcdn_amd.broker.gpu_engine import GpuBrokerEngine
from pushcdn_amd.proto import message as msglib

def bench_engine(nt, flat, steps=20):
    eng = GpuBrokerEngine(device="cuda:0", n_users=10000, ring_bytes=1<<21,
                          fanout_wire=True, direct_enabled=False, nt_fanout=nt,
                          pair_capacity=4<<20)
    eng.subscribe_all(list(range(8)))
    rng = random.Random(0)
    buf = bytearray(); offsets=[0]; wl=None
    for i in range(256):
        raw = msglib.serialize(msglib.Broadcast([i % 8], bytes(rng.randrange(256) for _ in range(1024))))
        padded = (len(raw)+15)&~15; wl = padded
        buf += raw + b"\x00"*(padded-len(raw)); offsets.append(len(buf))
    dbuf, doff = eng.ingest(bytes(buf), offsets)
    uw = wl if flat else None
    for _ in range(5):
        eng.tick(dbuf, doff, uniform_wire_len=uw); eng.drain_cursors()
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(steps):
        eng.tick(dbuf, doff, uniform_wire_len=uw); eng.drain_cursors()
    torch.cuda.synchronize()
    return (time.time()-t0)/steps*1000

for rnd in range(3):
    for nt in (0,1):
        for flat in (0,1):
            print(f"round {rnd} nt={nt} flat={flat}: {bench_engine(bool(nt), bool(flat)):.3f} ms/tick")
PYEOF

cd /tmp && export TMPDIR=/tmp && cd /root/repo
timeout 600 rocprofv3 --kernel-trace --stats -d gpurun_out/prof -o bench_prof -- \
  python bench.py --steps 10 --warmup 3 > gpurun_out/bench_prof.log 2>&1
echo "rocprof exit: $?" >> gpurun_out/bench_prof.log

tail -n 6 gpurun_out/pytest_gpu.log; tail -n 2 gpurun_out/bench_nt.json; tail -n 8 gpurun_out/fanout_ab.log
