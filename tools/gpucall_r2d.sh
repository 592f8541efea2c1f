#!/bin/bash
# Final round-2 validation on the end-of-round tree: GPU tests, smoke,
# bench default (bs768), and the mode triple at bs768.
cd /root/repo
export TMPDIR=/tmp
mkdir -p gpurun_out
echo "=== pytest -m gpu ==="
timeout 480 python -m pytest tests/ -q -m gpu -x > gpurun_out/pytest_gpu_final.log 2>&1
echo "pytest rc=$?"
tail -3 gpurun_out/pytest_gpu_final.log
echo "=== smoke ==="
timeout 240 python -c "import __graft_entry__ as g; g.smoke(); print('__SMOKE_OK__')" 2>&1 | tail -2
echo "=== bench default bs768 gtopk ==="
timeout 300 python bench.py --steps 20 --warmup 8 > gpurun_out/bench_final_gtopk.log 2>&1
echo "rc=$?"; tail -1 gpurun_out/bench_final_gtopk.log
echo "=== bench bs768 dense ==="
timeout 240 python bench.py --mode dense --steps 20 --warmup 6 > gpurun_out/bench_final_dense.log 2>&1
echo "rc=$?"; tail -1 gpurun_out/bench_final_dense.log
echo "=== bench bs768 topk ==="
timeout 240 python bench.py --mode topk --steps 20 --warmup 6 > gpurun_out/bench_final_topk.log 2>&1
echo "rc=$?"; tail -1 gpurun_out/bench_final_topk.log
echo "=== ALLDONE ==="
