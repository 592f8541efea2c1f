#!/bin/bash
cd /root/repo
export TMPDIR=/tmp
timeout 240 python -m pytest tests/ -q -m gpu -x 2>&1 | tail -2
timeout 120 python -c "import __graft_entry__ as g; g.smoke()" 2>&1 | tail -1
timeout 200 python bench.py --steps 20 --warmup 6 2>/dev/null | tail -1
echo "=== ALLDONE ==="
