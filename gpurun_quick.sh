#!/bin/bash
export TMPDIR=/tmp
cd /root/repo
for Q in rate downsample mean; do
  OUT=$(timeout 200 python bench.py --query $Q --steps 15 --warmup 4 --skip-cpu-baseline 2>/dev/null | tail -1)
  echo "$OUT" | python3 -c "
import json,sys
d=json.load(sys.stdin)
print('$Q: %.1f Gpts/s %.3f ms' % (d['value']/1e9, d['ms_per_step']))"
done
