#!/usr/bin/env bash
# Post-sweep final validation: GPU suite, smoke, bench, graph-form verify.
set -x
cd /root/repo
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0 PYTHONPATH=/root/repo
timeout 900 python -m pytest tests -m gpu -q > gpurun_out/pytest_gpu_final2.log 2>&1
echo "pytest rc=$?" >> gpurun_out/pytest_gpu_final2.log
tail -3 gpurun_out/pytest_gpu_final2.log
timeout 300 python -c "import __graft_entry__; __graft_entry__.smoke()" \
  > gpurun_out/smoke_final.log 2>&1
echo "smoke rc=$?" >> gpurun_out/smoke_final.log
tail -5 gpurun_out/smoke_final.log
timeout 240 python bench.py --steps 300 --warmup 30 > gpurun_out/bench_final2.json 2>/dev/null
timeout 300 python - >> gpurun_out/bench_final2.json 2>&1 <<'EOF'
from gpud_amd.diag import _diag
_diag.set_device(0)
import json
r = _diag.gemm_stress_bf16_v7_graph(size=8192, iters=16)
print(json.dumps({"gemm_v7p_graph_iters16_tflops": round(r["tflops"],1), "verified": r["verified"]}))
EOF
cat gpurun_out/bench_final2.json
