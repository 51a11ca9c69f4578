#!/bin/bash
set -e
cd "$(dirname "$0")"
python - << 'PYEOF'
# grid-size knob via env in engine? Not present; bench only. Use default.
PYEOF
cd ../..
python -m pytest tests/test_gpu_parity.py -m gpu -q 2>&1 | tail -1
for w in q1 q6 filter-int64 decode-filter; do
  python bench.py --workload $w --rows 50000000 --steps 5 --warmup 2 --no-cpu-baseline 2>/dev/null | tail -1 | python -c "
import json,sys
d=json.load(sys.stdin)
print('$w:', round(d['value']/1e9,2),'Grows/s', round(d['roofline']['achieved'],1),'GB/s', round(d['config']['kernel_ms_avg'],3),'ms')"
done
for n in 0 1 2 4 6; do
  OBX_Q1_AGGS=$n python bench.py --workload q1 --rows 50000000 --steps 5 --warmup 2 --no-cpu-baseline 2>/dev/null | tail -1 | python -c "
import json,sys; d=json.load(sys.stdin); print('q1 aggs=$n:', round(d['config']['kernel_ms_avg'],3),'ms')"
done
