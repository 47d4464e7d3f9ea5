#!/usr/bin/env bash
# v7P vs v7P2 (glds-burst-first head) + MfmaUtil PMC on both @8192.
set -x
cd /root/repo
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0 PYTHONPATH=/root/repo

timeout 700 python - > gpurun_out/gemm_ab_v7p2.log 2>&1 <<'EOF'
from gpud_amd.diag import _diag
_diag.set_device(0)
import json

for size in (512, 1024):
    r = _diag.gemm_stress_bf16_v7_style(size=size, iters=3, style=9)
    assert r["verified"], (size, r)
print(json.dumps({"race_screen": "ok"}), flush=True)

variants = {
    "v7P":  lambda s: _diag.gemm_stress_bf16_v7_style(size=s, iters=5, style=8),
    "v7P2": lambda s: _diag.gemm_stress_bf16_v7_style(size=s, iters=5, style=9),
}
for size in (4096, 8192):
    for rep in range(3):
        for name, fn in variants.items():
            r = fn(size)
            print(json.dumps({"size": size, "rep": rep, "variant": name,
                              "tflops": round(r["tflops"], 1),
                              "verified": r["verified"]}), flush=True)
EOF
echo "ab rc=$?" >> gpurun_out/gemm_ab_v7p2.log

cd /tmp && export TMPDIR=/tmp
timeout 300 rocprofv3 --pmc SQ_VALU_MFMA_BUSY_CYCLES,GRBM_GUI_ACTIVE,SQ_WAVE_CYCLES,SQ_WAIT_ANY \
  -d /root/repo/gpurun_out/pmc_v7p -- python -c "
from gpud_amd.diag import _diag
_diag.set_device(0)
for st in (8, 9):
    print(st, _diag.gemm_stress_bf16_v7_style(size=8192, iters=2, style=st)['tflops'])
    print(st, _diag.gemm_stress_bf16_v7_style(size=4096, iters=2, style=st)['tflops'])
" > /root/repo/gpurun_out/pmc_v7p.log 2>&1
echo "pmc rc=$?" >> /root/repo/gpurun_out/pmc_v7p.log
tail -30 /root/repo/gpurun_out/gemm_ab_v7p2.log
