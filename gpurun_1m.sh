#!/bin/bash
export TMPDIR=/tmp
cd /root/repo
for Q in mean tags; do
  OUT=$(timeout 400 python bench.py --query $Q --series 1000000 --pts 100 --steps 8 --warmup 3 --skip-cpu-baseline 2>/tmp/e1m.log | tail -1)
  if [ -z "$OUT" ]; then echo "$Q: FAIL"; tail -3 /tmp/e1m.log; else
  echo "$OUT" | python3 -c "
import json,sys
d=json.load(sys.stdin)
print('$Q 1M-series: %.1f Gpts/s %.3f ms decode=%.0f GB/s' % (d['value']/1e9, d['ms_per_step'], d['roofline']['achieved']))"
  fi
done
