#!/bin/bash
# Builds the MI355X product engine (libobx.so) in-tree for gfx950.
# The .so is git-ignored but ships to the GPU box via the gpurun snapshot.
set -e
cd "$(dirname "$0")"
HIPCC=${HIPCC:-hipcc}
# embed the device headers for the hipRTC JIT (obx_jit.inc)
python3 - << 'PYEOF'
for src, out in (("obx_dev.h", "obx_embed_dev_h.inc"),
                 ("obx_dev_common.h", "obx_embed_common_h.inc")):
    txt = open(src).read()
    assert ')OBXRAW"' not in txt
    open(out, "w").write('R"OBXRAW(' + txt + ')OBXRAW"\n')
PYEOF
$HIPCC --offload-arch=gfx950 -O3 -std=c++17 -fPIC -shared \
    obx_engine.cpp obx_kernels.hip obx_cs_load.cpp obx_cs_kernels.hip \
    -L/opt/rocm/lib -lhiprtc \
    -o ../libobx.so
echo "built oceanbase_amd/libobx.so"
