#!/bin/bash
export TMPDIR=/tmp
cd /root/repo
timeout 200 python -m pytest tests/test_gpu_parity.py::TestScanAggParity tests/test_gpu_parity.py::TestSingleBigSeries -q 2>&1 | tail -1
timeout 300 python bench.py --query mean --series 1 --pts 10000000 --steps 5 --warmup 2 --skip-cpu-baseline 2>/dev/null | tail -1 | python3 -c "
import json,sys
d=json.load(sys.stdin)
print('mean 1x10M: %.1f Gpts/s %.2f ms' % (d['value']/1e9, d['ms_per_step']))"
timeout 200 python bench.py --query mean --steps 10 --warmup 3 --skip-cpu-baseline 2>/dev/null | tail -1 | python3 -c "
import json,sys
d=json.load(sys.stdin)
print('mean 100kx1k: %.1f Gpts/s %.3f ms' % (d['value']/1e9, d['ms_per_step']))"
