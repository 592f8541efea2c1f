#!/bin/bash
cd /root/repo
export TMPDIR=/tmp
mkdir -p gpurun_out/prof_r20
timeout 260 rocprofv3 --kernel-trace --stats -d gpurun_out/prof_r20 -o r20 \
  -- python bench.py --model resnet20 --steps 5 --warmup 5 > gpurun_out/prof_r20/run.log 2>&1
echo "rc=$?"; tail -1 gpurun_out/prof_r20/run.log
echo "=== ALLDONE ==="
