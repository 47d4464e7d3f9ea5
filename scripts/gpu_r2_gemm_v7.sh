#!/usr/bin/env bash
# v7 (hand-scheduled asm K-loop) race-screen + same-box interleaved A/B vs
# the shipped v2, then PMC MfmaUtil on v7@8192.
set -x
cd /root/repo
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0 PYTHONPATH=/root/repo

timeout 900 python - > gpurun_out/gemm_ab_v7.log 2>&1 <<'EOF'
from gpud_amd.diag import _diag
_diag.set_device(0)
import json

# race screen: sync-structure-identical to v2, but asm is new — multi-run
# exact verification at small sizes first (methodology: two-lane discipline)
for size in (512, 1024, 2048):
    for rep in range(4):
        r = _diag.gemm_stress_bf16_v7(size=size, iters=3)
        assert r["verified"], (size, rep, r)
print(json.dumps({"race_screen": "ok"}), flush=True)

variants = {
    "v2": lambda s: _diag.gemm_stress_bf16_v2(size=s, iters=5),
    "v7": lambda s: _diag.gemm_stress_bf16_v7(size=s, iters=5),
    "v7_nosp": lambda s: _diag.gemm_stress_bf16_v7_nosp(size=s, iters=5),
}
for size in (4096, 8192):
    for rep in range(3):
        for name, fn in variants.items():
            r = fn(size)
            print(json.dumps({"size": size, "rep": rep, "variant": name,
                              "tflops": round(r["tflops"], 1),
                              "verified": r["verified"]}), flush=True)
EOF
echo "ab rc=$?" >> gpurun_out/gemm_ab_v7.log

cd /tmp && export TMPDIR=/tmp
timeout 420 rocprofv3 --pmc SQ_VALU_MFMA_BUSY_CYCLES,GRBM_GUI_ACTIVE,SQ_WAVE_CYCLES,SQ_WAIT_ANY \
  -d /root/repo/gpurun_out/pmc_v7 -- \
  python -c "
from gpud_amd.diag import _diag
_diag.set_device(0)
print('v7@8192', _diag.gemm_stress_bf16_v7(size=8192, iters=2))
" > /root/repo/gpurun_out/pmc_v7.log 2>&1
echo "pmc rc=$?" >> /root/repo/gpurun_out/pmc_v7.log

tail -25 /root/repo/gpurun_out/gemm_ab_v7.log
