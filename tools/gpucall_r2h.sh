#!/bin/bash
# Full final validation: GPU suite + smoke + bench x2 + kernel stats.
cd /root/repo
export TMPDIR=/tmp
mkdir -p gpurun_out/prof_bnfuse2
echo "=== pytest -m gpu ==="
timeout 480 python -m pytest tests/ -q -m gpu -x > gpurun_out/pytest_gpu_final2.log 2>&1
echo "rc=$?"; tail -2 gpurun_out/pytest_gpu_final2.log
echo "=== smoke ==="
timeout 240 python -c "import __graft_entry__ as g; g.smoke(); print('__SMOKE_OK__')" 2>&1 | tail -1
echo "=== bench x2 ==="
timeout 240 python bench.py --steps 20 --warmup 6 2>/dev/null | tail -1
timeout 240 python bench.py --steps 20 --warmup 6 2>/dev/null | tail -1
echo "=== kernel stats ==="
timeout 240 rocprofv3 --kernel-trace --stats -d gpurun_out/prof_bnfuse2 -o bnfuse2 \
  -- python bench.py --steps 3 --warmup 4 > gpurun_out/prof_bnfuse2/run.log 2>&1
echo "rc=$?"
echo "=== ALLDONE ==="
