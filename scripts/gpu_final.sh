#!/bin/bash
set -x
mkdir -p gpurun_out
export TMPDIR=/tmp

timeout 900 python -m pytest tests/ -x -q -m gpu > gpurun_out/pytest_gpu.log 2>&1
echo "exit=$?" >> gpurun_out/pytest_gpu.log

timeout 300 python -c "import __graft_entry__; __graft_entry__.smoke()" > gpurun_out/smoke.log 2>&1

# payload-size curve (broadcast, 10k subscribers)
for p in 256 1024 4096 16384; do
  timeout 200 python bench.py --steps 30 --warmup 8 --payload $p 2>/dev/null | \
    grep -o "\"value\": [0-9.]*\|\"ms_per_step\": [0-9.]*" | tr "\n" " "; echo " payload=$p"
done > gpurun_out/payload_curve.log 2>&1

# headline + mixed re-validation on THIS binary
timeout 300 python bench.py --steps 50 --warmup 10 > gpurun_out/bench_final.json 2>&1
timeout 600 python bench.py --mode mixed --steps 20 --warmup 5 --subscribers 100000 --topics 64 > gpurun_out/bench_mixed_final.json 2>&1

# fresh kernel profile of the final binary
timeout 400 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/prof_final -o p -- \
  python bench.py --steps 10 --warmup 3 > gpurun_out/prof_final.log 2>&1

tail -n 3 gpurun_out/pytest_gpu.log
tail -n 1 gpurun_out/smoke.log
cat gpurun_out/payload_curve.log
tail -c 260 gpurun_out/bench_final.json; echo
tail -c 260 gpurun_out/bench_mixed_final.json
