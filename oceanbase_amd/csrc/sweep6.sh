#!/bin/bash
set -e
cd "$(dirname "$0")"
for OPT in "-O3" "-O2" "-O3 -mllvm -amdgpu-early-inline-all=true" "-O3 -ffast-math"; do
  hipcc --offload-arch=gfx950 $OPT -std=c++17 -fPIC -shared obx_engine.cpp obx_kernels.hip -o ../libobx.so 2>/dev/null || { echo "build failed: $OPT"; continue; }
  cd ../..
  python bench.py --workload q1 --rows 50000000 --steps 5 --warmup 2 --no-cpu-baseline 2>/dev/null | tail -1 | python -c "
import json,sys; d=json.load(sys.stdin); print('$OPT q1:', round(d['config']['kernel_ms_avg'],3),'ms')"
  cd oceanbase_amd/csrc
done
hipcc --offload-arch=gfx950 -O3 -std=c++17 -fPIC -shared obx_engine.cpp obx_kernels.hip -o ../libobx.so 2>/dev/null
