#!/bin/bash
# Validate the bs768 headline default on a fresh box: smoke, bench,
# quick GPU-test sanity.
cd /root/repo
export TMPDIR=/tmp
mkdir -p gpurun_out
echo "=== smoke ==="
timeout 240 python -c "import __graft_entry__ as g; g.smoke(); print('__SMOKE_OK__')" 2>&1 | tail -3
echo "=== bench default (bs768) ==="
timeout 300 python bench.py --steps 20 --warmup 8 > gpurun_out/bench_bs768_default.log 2>&1
echo "rc=$?"
tail -2 gpurun_out/bench_bs768_default.log
echo "=== ALLDONE ==="
