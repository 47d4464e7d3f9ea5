#!/usr/bin/env bash
# v9 (2 blocks/CU) vs v7P: race screen + same-box interleaved reps.
set -x
cd /root/repo
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0 PYTHONPATH=/root/repo
timeout 700 python - > gpurun_out/gemm_ab_v9.log 2>&1 <<'EOF'
from gpud_amd.diag import _diag
_diag.set_device(0)
import json
for size in (512, 768, 1024, 1536):
    r = _diag.gemm_stress_bf16_v9(size=size, iters=3)
    assert r["verified"], (size, r)
print(json.dumps({"race_screen": "ok"}), flush=True)
variants = {
    "v7P": lambda s: _diag.gemm_stress_bf16_v7_style(size=s, iters=5, style=8),
    "v9":  lambda s: _diag.gemm_stress_bf16_v9(size=s, iters=5),
}
for size in (8192, 4096):
    for rep in range(3):
        for name, fn in variants.items():
            r = fn(size)
            print(json.dumps({"size": size, "rep": rep, "variant": name,
                              "tflops": round(r["tflops"], 1),
                              "verified": r["verified"]}), flush=True)
EOF
echo "ab rc=$?" >> gpurun_out/gemm_ab_v9.log
tail -30 gpurun_out/gemm_ab_v9.log
