#!/bin/bash
# Kernel-level check of the two-level BN reduce + bench variance.
cd /root/repo
export TMPDIR=/tmp
mkdir -p gpurun_out/prof_bnfuse
echo "=== kernel stats (3 steps) ==="
timeout 300 rocprofv3 --kernel-trace --stats -d gpurun_out/prof_bnfuse -o bnfuse \
  -- python bench.py --steps 3 --warmup 4 > gpurun_out/prof_bnfuse/run.log 2>&1
echo "rc=$?"
echo "=== bench x2 ==="
timeout 240 python bench.py --steps 20 --warmup 6 2>/dev/null | tail -1
timeout 240 python bench.py --steps 20 --warmup 6 2>/dev/null | tail -1
echo "=== ALLDONE ==="
