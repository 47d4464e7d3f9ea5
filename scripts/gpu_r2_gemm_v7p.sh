#!/usr/bin/env bash
# v7P (cross-barrier-pipelined phase 3) vs v7X: race screen + same-box reps.
set -x
cd /root/repo
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0 PYTHONPATH=/root/repo

timeout 900 python - > gpurun_out/gemm_ab_v7p.log 2>&1 <<'EOF'
from gpud_amd.diag import _diag
_diag.set_device(0)
import json

for size in (512, 768, 1024, 1536):
    r = _diag.gemm_stress_bf16_v7_style(size=size, iters=3, style=8)
    assert r["verified"], (size, r)
print(json.dumps({"race_screen": "ok"}), flush=True)

variants = {
    "v7X": lambda s: _diag.gemm_stress_bf16_v7_style(size=s, iters=5, style=5),
    "v7P": lambda s: _diag.gemm_stress_bf16_v7_style(size=s, iters=5, style=8),
}
for size in (4096, 8192):
    for rep in range(3):
        for name, fn in variants.items():
            r = fn(size)
            print(json.dumps({"size": size, "rep": rep, "variant": name,
                              "tflops": round(r["tflops"], 1),
                              "verified": r["verified"]}), flush=True)
EOF
echo "ab rc=$?" >> gpurun_out/gemm_ab_v7p.log
tail -30 gpurun_out/gemm_ab_v7p.log
