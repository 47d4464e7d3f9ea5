#!/usr/bin/env bash
# v7 schedule-style grid round C: base vs E (reordered p0 head) vs
# F (all-8-glds-at-p0) vs X (E+F), same-box interleaved reps.
set -x
cd /root/repo
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0 PYTHONPATH=/root/repo

timeout 900 python - > gpurun_out/gemm_ab_v7c.log 2>&1 <<'EOF'
from gpud_amd.diag import _diag
_diag.set_device(0)
import json

for size in (512, 1024):
    for style in (0, 3, 4, 5):
        r = _diag.gemm_stress_bf16_v7_style(size=size, iters=3, style=style)
        assert r["verified"], (size, style, r)
print(json.dumps({"race_screen": "ok"}), flush=True)

variants = {
    "v7base": lambda s: _diag.gemm_stress_bf16_v7_style(size=s, iters=5, style=0),
    "v7E":    lambda s: _diag.gemm_stress_bf16_v7_style(size=s, iters=5, style=3),
    "v7F":    lambda s: _diag.gemm_stress_bf16_v7_style(size=s, iters=5, style=4),
    "v7X":    lambda s: _diag.gemm_stress_bf16_v7_style(size=s, iters=5, style=5),
}
for size in (4096, 8192):
    for rep in range(3):
        for name, fn in variants.items():
            r = fn(size)
            print(json.dumps({"size": size, "rep": rep, "variant": name,
                              "tflops": round(r["tflops"], 1),
                              "verified": r["verified"]}), flush=True)
EOF
echo "ab rc=$?" >> gpurun_out/gemm_ab_v7c.log
tail -40 gpurun_out/gemm_ab_v7c.log
