#!/bin/bash
# Evidence refresh at the bs768 default: steady-state kernel stats and
# the WRITE_SIZE counter for the BN kernels.
cd /root/repo
export TMPDIR=/tmp
mkdir -p gpurun_out/pmc gpurun_out/prof768
echo "=== kernel stats bs768 ==="
timeout 360 rocprofv3 --kernel-trace --stats -d gpurun_out/prof768 -o ks768 \
  -- python bench.py --steps 5 --warmup 6 > gpurun_out/prof768/run.log 2>&1
echo "rc=$?"; tail -2 gpurun_out/prof768/run.log
ls gpurun_out/prof768/ | head
echo "=== pmc WRITE_SIZE bench (bn kernels) ==="
timeout 360 rocprofv3 --pmc WRITE_SIZE -d gpurun_out/pmc -o bench_WRITE \
  --kernel-include-regex "bn_" \
  -- python bench.py --steps 3 --warmup 4 > gpurun_out/pmc/bench_WRITE.log 2>&1
echo "rc=$?"; tail -2 gpurun_out/pmc/bench_WRITE.log
echo "=== ALLDONE ==="
