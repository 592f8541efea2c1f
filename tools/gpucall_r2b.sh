#!/bin/bash
# Round-2 PMC counter evidence: one counter per run (FETCH_SIZE +
# WRITE_SIZE together exceed gfx950's one-pass counter capacity).
cd /root/repo
export TMPDIR=/tmp
mkdir -p gpurun_out/pmc
for C in FETCH_SIZE WRITE_SIZE; do
  echo "=== pmc $C selfbench ==="
  timeout 240 rocprofv3 --pmc $C -d gpurun_out/pmc -o sel_$C \
    --kernel-include-regex "ef_accum|refine_hist|compact|scatter_add|sgd_kernel|gather_zero|scan_threshold|hist" \
    -- python -m gtopkssgd_amd.parallel.allreducer --mode gtopk \
       --iters 20 --warmup 5 --simulate-rounds 3 \
       > gpurun_out/pmc/sel_$C.log 2>&1
  echo "rc=$?"; tail -2 gpurun_out/pmc/sel_$C.log
done
echo "=== pmc FETCH_SIZE bench (bn kernels) ==="
timeout 420 rocprofv3 --pmc FETCH_SIZE -d gpurun_out/pmc -o bench_FETCH \
  --kernel-include-regex "bn_" \
  -- python bench.py --steps 3 --warmup 4 > gpurun_out/pmc/bench_FETCH.log 2>&1
echo "rc=$?"; tail -2 gpurun_out/pmc/bench_FETCH.log
find gpurun_out/pmc -name "*counter_collection*" | head
echo "=== ALLDONE ==="
