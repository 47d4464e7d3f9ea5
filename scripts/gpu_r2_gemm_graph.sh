#!/usr/bin/env bash
# v7P event-timed loop vs hipGraph-timed replay, same box.
set -x
cd /root/repo
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0 PYTHONPATH=/root/repo
timeout 600 python - > gpurun_out/gemm_ab_graph.log 2>&1 <<'EOF'
from gpud_amd.diag import _diag
_diag.set_device(0)
import json
r = _diag.gemm_stress_bf16_v7_graph(size=1024, iters=3)
assert r["verified"], r
print(json.dumps({"race_screen": "ok"}), flush=True)
variants = {
    "v7P_loop":  lambda s: _diag.gemm_stress_bf16_v7_style(size=s, iters=8, style=8),
    "v7P_graph": lambda s: _diag.gemm_stress_bf16_v7_graph(size=s, iters=8),
}
for size in (8192, 4096):
    for rep in range(3):
        for name, fn in variants.items():
            r = fn(size)
            print(json.dumps({"size": size, "rep": rep, "variant": name,
                              "tflops": round(r["tflops"], 1),
                              "verified": r["verified"]}), flush=True)
EOF
echo "ab rc=$?" >> gpurun_out/gemm_ab_graph.log
tail -20 gpurun_out/gemm_ab_graph.log
