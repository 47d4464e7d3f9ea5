#!/usr/bin/env bash
# Final round-2 validation on the shipped state: full GPU test suite,
# bench in both shapes, the diag component exercised through the
# component layer (v7P default + floors), and a diagnose report.
set -x
cd /root/repo
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0 PYTHONPATH=/root/repo

timeout 900 python -m pytest tests -m gpu -q > gpurun_out/pytest_gpu_final.log 2>&1
echo "pytest rc=$?" >> gpurun_out/pytest_gpu_final.log
tail -3 gpurun_out/pytest_gpu_final.log

timeout 240 python bench.py --steps 300 --warmup 30 > gpurun_out/bench_final.json 2> gpurun_out/bench_final.err
echo "bench rc=$?" >> gpurun_out/bench_final.err
timeout 240 python bench.py --single-process --gpus 8 --steps 300 --warmup 30 \
  > gpurun_out/bench_final_sp.json 2>> gpurun_out/bench_final.err

timeout 300 python - > gpurun_out/diag_component_final.log 2>&1 <<'PYEOF'
import sys
sys.path.insert(0, "/root/repo")
from gpud_amd.components.base import GPUdInstance
from gpud_amd.components.accelerator.diag import MFMADiagComponent, BandwidthDiagComponent
inst = GPUdInstance()
c = MFMADiagComponent(inst)
r = c.check()
print("mfma-diag:", r.health, r.reason)
for k, v in sorted((r.extra_info or {}).items()):
    print("  ", k, "=", v)
b = BandwidthDiagComponent(inst)
rb = b.check()
print("bandwidth-diag:", rb.health, rb.reason)
assert r.health == "Healthy", (r.health, r.reason)
PYEOF
echo "diag rc=$?" >> gpurun_out/diag_component_final.log

cat gpurun_out/bench_final.json gpurun_out/bench_final_sp.json
tail -12 gpurun_out/diag_component_final.log
