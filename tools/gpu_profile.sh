#!/bin/bash
# Run on the GPU box via gpurun: collects rocprofv3 kernel stats for the
# flagship bench and writes CSV summaries under gpurun_out/prof/.
set -x
cd /tmp && export TMPDIR=/tmp
cd "$GRAFT_REPO_ROOT" || cd /root/repo
mkdir -p gpurun_out/prof
rocprofv3 --kernel-trace --stats --output-format csv -d gpurun_out/prof -o bench_stats -- \
  timeout 300 python bench.py --gpus 1 --steps 10 --warmup 5 --no-graph \
  > gpurun_out/prof/bench_nograph.log 2>&1
tail -2 gpurun_out/prof/bench_nograph.log
ls gpurun_out/prof/
