#!/bin/bash
# Round-2 consolidated GPU validation: full GPU test suite, headline
# re-check, lstman4 bench, PMC counter profile of the HIP kernels.
cd /root/repo
export TMPDIR=/tmp
mkdir -p gpurun_out/pmc
echo "=== pytest -m gpu ==="
timeout 480 python -m pytest tests/ -q -m gpu -x > gpurun_out/pytest_gpu.log 2>&1
echo "pytest rc=$?"
tail -4 gpurun_out/pytest_gpu.log
echo "=== headline bench ==="
timeout 240 python bench.py --steps 20 --warmup 8 > gpurun_out/bench_headline.log 2>&1
echo "bench rc=$?"
tail -2 gpurun_out/bench_headline.log
echo "=== lstman4 bench ==="
timeout 300 python bench.py --model lstman4 --steps 20 --warmup 5 > gpurun_out/bench_lstman4.log 2>&1
echo "lstman4 rc=$?"
tail -2 gpurun_out/bench_lstman4.log
echo "=== pmc profile ==="
timeout 300 rocprofv3 --pmc FETCH_SIZE WRITE_SIZE -d gpurun_out/pmc -o pmc \
  --kernel-include-regex "bn_|ef_accum|refine_hist|compact|scatter_add|sgd_kernel|gather_zero|scan_threshold" \
  -- python bench.py --steps 5 --warmup 5 > gpurun_out/pmc/run.log 2>&1
rc=$?
echo "pmc rc=$rc"
find gpurun_out/pmc -type f | head
if [ $rc -ne 0 ]; then
  tail -15 gpurun_out/pmc/run.log
fi
echo "=== ALLDONE ==="
