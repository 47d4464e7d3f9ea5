#!/usr/bin/env bash
# Wave-park attribution: v7X (full) vs N-skeleton (no glds, barrier kept)
# vs Q-skeleton (no glds, no barrier) + SQ_WAIT_INST_LDS PMC on all three.
set -x
cd /root/repo
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0 PYTHONPATH=/root/repo

timeout 600 python - > gpurun_out/gemm_skel.log 2>&1 <<'EOF'
from gpud_amd.diag import _diag
_diag.set_device(0)
import json
variants = {
    "v7X":   lambda s: _diag.gemm_stress_bf16_v7_style(size=s, iters=5, style=5),
    "skelN": lambda s: _diag.gemm_stress_bf16_v7_style(size=s, iters=5, style=6),
    "skelQ": lambda s: _diag.gemm_stress_bf16_v7_style(size=s, iters=5, style=7),
}
for size in (4096, 8192):
    for rep in range(3):
        for name, fn in variants.items():
            r = fn(size)
            print(json.dumps({"size": size, "rep": rep, "variant": name,
                              "tflops": round(r["tflops"], 1)}), flush=True)
EOF
echo "skel rc=$?" >> gpurun_out/gemm_skel.log

cd /tmp && export TMPDIR=/tmp
timeout 300 rocprofv3 --pmc SQ_WAIT_INST_LDS,SQ_WAIT_ANY,SQ_WAVE_CYCLES,GRBM_GUI_ACTIVE \
  -d /root/repo/gpurun_out/pmc_skel -- python -c "
from gpud_amd.diag import _diag
_diag.set_device(0)
for st in (5, 6, 7):
    print(st, _diag.gemm_stress_bf16_v7_style(size=8192, iters=2, style=st)['tflops'])
" > /root/repo/gpurun_out/pmc_skel.log 2>&1
echo "pmc rc=$?" >> /root/repo/gpurun_out/pmc_skel.log
tail -25 /root/repo/gpurun_out/gemm_skel.log
