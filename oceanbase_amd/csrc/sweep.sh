#!/bin/bash
# on-box variant sweep: rebuild libobx.so with -D overrides and bench
set -e
cd "$(dirname "$0")"
for cfg in "1 4" "0 4" "0 2" "1 2" "1 1"; do
  set -- $cfg
  PIPE=$1; STR=$2
  hipcc --offload-arch=gfx950 -O3 -std=c++17 -fPIC -shared \
      -DOBX_PIPELINE=$PIPE -DOBX_STRIPES=$STR \
      obx_engine.cpp obx_kernels.hip -o ../libobx.so 2>/dev/null
  cd ../..
  for w in q1 filter-int64; do
    python bench.py --workload $w --rows 50000000 --steps 5 --warmup 2 --no-cpu-baseline 2>/dev/null | tail -1 | python -c "
import json,sys
d=json.load(sys.stdin)
print('pipe=$PIPE stripes=$STR $w:', round(d['config']['kernel_ms_avg'],3), 'ms')"
  done
  cd oceanbase_amd/csrc
done
