#!/bin/bash
export TMPDIR=/tmp
cd /root/repo
timeout 200 python -m pytest tests/test_gpu_parity.py::TestScanAggParity -x -q 2>&1 | tail -1
for M in walk random; do
  OUT=$(timeout 200 python bench.py --query mean --mode $M --steps 15 --warmup 4 --skip-cpu-baseline 2>/dev/null | tail -1)
  echo "$OUT" | python3 -c "
import json,sys
d=json.load(sys.stdin)
print('$M: %.1f Gpts/s %.3f ms decode=%.0f GB/s' % (d['value']/1e9, d['ms_per_step'], d['roofline']['achieved']))"
done
