#!/bin/bash
# Validate the fused reduce+finalize BN kernel: numerics + throughput.
cd /root/repo
export TMPDIR=/tmp
mkdir -p gpurun_out
echo "=== pytest gpu (bn + kernels) ==="
timeout 420 python -m pytest tests/test_fusedbn_gpu.py tests/test_kernels_gpu.py -q -x > gpurun_out/pytest_bnfuse.log 2>&1
echo "rc=$?"; tail -3 gpurun_out/pytest_bnfuse.log
echo "=== bench default bs768 ==="
timeout 300 python bench.py --steps 20 --warmup 8 > gpurun_out/bench_bnfuse.log 2>&1
echo "rc=$?"; tail -1 gpurun_out/bench_bnfuse.log
echo "=== ALLDONE ==="
