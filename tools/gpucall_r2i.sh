#!/bin/bash
# Re-measure the small-model configs with the fused BN reduce.
cd /root/repo
export TMPDIR=/tmp
for M in resnet20 vgg16 alexnet; do
  echo "=== $M ==="
  timeout 300 python bench.py --model $M --steps 20 --warmup 6 2>/dev/null | tail -1
done
echo "=== ALLDONE ==="
