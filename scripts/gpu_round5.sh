#!/bin/bash
set -x
mkdir -p gpurun_out
export TMPDIR=/tmp

timeout 900 python -m pytest tests/ -x -q -m gpu > gpurun_out/pytest_gpu.log 2>&1
echo "pytest_gpu exit: $?" >> gpurun_out/pytest_gpu.log

timeout 600 python bench.py --steps 40 --warmup 10 > gpurun_out/bench_graph.json 2>&1
timeout 600 python bench.py --steps 40 --warmup 10 --no-graph > gpurun_out/bench_nograph.json 2>&1

cd /tmp && export TMPDIR=/tmp && cd /root/repo
timeout 600 rocprofv3 --kernel-trace --stats -d gpurun_out/prof -o bench_prof -- \
  python bench.py --steps 10 --warmup 3 > gpurun_out/bench_prof.log 2>&1

tail -n 4 gpurun_out/pytest_gpu.log
tail -n 1 gpurun_out/bench_graph.json
tail -n 1 gpurun_out/bench_nograph.json
