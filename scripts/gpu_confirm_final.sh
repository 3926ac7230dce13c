#!/bin/bash
# Final-binary confirmation: the round-2 socket headline config re-run on
# the exact bits the driver will test (after the QuicNative additions).
set -x
cd /root/repo
export PYTHONUNBUFFERED=1
timeout 150 python -m pytest tests -m gpu -x -q 2>&1 | tail -2
timeout 200 python scripts/bench_socket.py --subs 50 --sub-procs 24 --senders 8 \
  --rate 150000 --pump-shards 8 --seconds 30 --tag final8 2>&1 | tail -2
cp gpurun_out/bench_socket_final8*.json gpurun_out/ 2>/dev/null || true
