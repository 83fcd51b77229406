#!/bin/bash
export TMPDIR=/tmp
cd /root/repo
echo "== config #4 per-GPU shape: 12.5k series x 100k pts (1.25B/8)"
timeout 600 python bench.py --query downsample --series 12500 --pts 100000 --steps 5 --warmup 2 --skip-cpu-baseline 2>/tmp/c4.log | tail -1 | python3 -c "
import json,sys
d=json.load(sys.stdin)
print('downsample: %.1f Gpts/s %.2f ms decode=%.0f GB/s' % (d['value']/1e9, d['ms_per_step'], d['roofline']['achieved']))" || tail -3 /tmp/c4.log
echo "== config #5-ish per-GPU shape: 10k series x 86400 pts (24h @ 1s)"
timeout 600 python bench.py --query rate --series 10000 --pts 86400 --steps 5 --warmup 2 --skip-cpu-baseline 2>/tmp/c5.log | tail -1 | python3 -c "
import json,sys
d=json.load(sys.stdin)
print('rate: %.1f Gpts/s %.2f ms decode=%.0f GB/s' % (d['value']/1e9, d['ms_per_step'], d['roofline']['achieved']))" || tail -3 /tmp/c5.log
echo "== config #1 shape: 1 series x 10M pts"
timeout 300 python bench.py --query mean --series 1 --pts 10000000 --steps 5 --warmup 2 --skip-cpu-baseline 2>/tmp/c1.log | tail -1 | python3 -c "
import json,sys
d=json.load(sys.stdin)
print('mean 1x10M: %.1f Gpts/s %.2f ms decode=%.0f GB/s' % (d['value']/1e9, d['ms_per_step'], d['roofline']['achieved']))" || tail -3 /tmp/c1.log
