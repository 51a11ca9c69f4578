#!/bin/bash
set -e
cd "$(dirname "$0")"
for W in 256 512 1024; do
  hipcc --offload-arch=gfx950 -O3 -std=c++17 -fPIC -shared \
      -DWG=$W -DOBX_WG_HOST=$W obx_engine.cpp obx_kernels.hip -o ../libobx.so 2>/dev/null
  cd ../..
  python -m pytest tests/test_gpu_parity.py::test_q1_agg_parity -m gpu -q 2>&1 | tail -1
  for w in q1 q6 filter-int64; do
    python bench.py --workload $w --rows 50000000 --steps 5 --warmup 2 --no-cpu-baseline 2>/dev/null | tail -1 | python -c "
import json,sys; d=json.load(sys.stdin); print('WG=$W $w:', round(d['config']['kernel_ms_avg'],3),'ms')"
  done
  cd oceanbase_amd/csrc
done
hipcc --offload-arch=gfx950 -O3 -std=c++17 -fPIC -shared obx_engine.cpp obx_kernels.hip -o ../libobx.so 2>/dev/null
