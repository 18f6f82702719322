#!/bin/bash
# v4 ring-scatter bring-up: parity (6M two-phase tests) then bench timing at
# three worker/flusher splits, each leg under its own timeout.
set -x
cd /root/repo
mkdir -p gpurun_out
export AURON_AGG2_V4=1
timeout 240 python -m pytest tests/test_gpu_twophase.py -x -q 2>&1 | tail -4
for WW in 12 14 10; do
  AURON_AGG2_V4_WW=$WW timeout 240 python bench.py --steps 3 --warmup 1 \
    --skip-cpu-baseline 2>/tmp/b$WW.err | python -c "
import json,sys
for l in sys.stdin:
    l=l.strip()
    if l.startswith('{'):
        d=json.loads(l); print('WW=$WW', d['ms_per_step'], round(d['value']/1e9,2),'G rows/s'); break
" || tail -3 /tmp/b$WW.err
done
