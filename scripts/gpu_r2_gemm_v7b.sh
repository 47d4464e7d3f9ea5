#!/usr/bin/env bash
# v7 schedule-style grid (same-box interleaved) + PMC on the winner shape.
set -x
cd /root/repo
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0 PYTHONPATH=/root/repo

timeout 900 python - > gpurun_out/gemm_ab_v7b.log 2>&1 <<'EOF'
from gpud_amd.diag import _diag
_diag.set_device(0)
import json

for size in (512, 1024):
    for style in (0, 1, 2):
        r = _diag.gemm_stress_bf16_v7_style(size=size, iters=3, style=style)
        assert r["verified"], (size, style, r)
print(json.dumps({"race_screen": "ok"}), flush=True)

variants = {
    "v2":      lambda s: _diag.gemm_stress_bf16_v2(size=s, iters=5),
    "v7base":  lambda s: _diag.gemm_stress_bf16_v7_style(size=s, iters=5, style=0),
    "v7sp":    lambda s: _diag.gemm_stress_bf16_v7_style(size=s, iters=5, style=1),
    "v7late":  lambda s: _diag.gemm_stress_bf16_v7_style(size=s, iters=5, style=2),
}
for size in (4096, 8192):
    for rep in range(3):
        for name, fn in variants.items():
            r = fn(size)
            print(json.dumps({"size": size, "rep": rep, "variant": name,
                              "tflops": round(r["tflops"], 1),
                              "verified": r["verified"]}), flush=True)
EOF
echo "ab rc=$?" >> gpurun_out/gemm_ab_v7b.log

cd /tmp && export TMPDIR=/tmp
timeout 420 rocprofv3 --pmc SQ_VALU_MFMA_BUSY_CYCLES,GRBM_GUI_ACTIVE,SQ_WAVE_CYCLES,SQ_WAIT_ANY \
  -d /root/repo/gpurun_out/pmc_v7b -- \
  python -c "
from gpud_amd.diag import _diag
_diag.set_device(0)
print('v7@8192', _diag.gemm_stress_bf16_v7(size=8192, iters=2))
print('v7@4096', _diag.gemm_stress_bf16_v7(size=4096, iters=2))
" > /root/repo/gpurun_out/pmc_v7b.log 2>&1
echo "pmc rc=$?" >> /root/repo/gpurun_out/pmc_v7b.log

tail -30 /root/repo/gpurun_out/gemm_ab_v7b.log
