#!/bin/bash
set -e
cd "$(dirname "$0")"
for FL in 4 2; do
  hipcc --offload-arch=gfx950 -O3 -std=c++17 -fPIC -shared \
      -DOBX_FAST_LEAVES_OVR=$FL obx_engine.cpp obx_kernels.hip -o ../libobx.so 2>/dev/null || echo "build failed FL=$FL"
  cd ../..
  for w in q1 q6; do
    python bench.py --workload $w --rows 50000000 --steps 5 --warmup 2 --no-cpu-baseline 2>/dev/null | tail -1 | python -c "
import json,sys; d=json.load(sys.stdin); print('FL=$FL $w:', round(d['config']['kernel_ms_avg'],3),'ms')"
  done
  cd oceanbase_amd/csrc
done
hipcc --offload-arch=gfx950 -O3 -std=c++17 -fPIC -shared obx_engine.cpp obx_kernels.hip -o ../libobx.so 2>/dev/null
