#!/usr/bin/env bash
# Round-2 validation pass: full GPU test suite (incl. the new kmsg-seam
# e2e fallback), bench in both shapes (rank-per-GPU contract + the new
# single-process daemon shape), live sanity of the new telemetry (per-XCC
# busy, CPER sections), then the ASan concurrency hunt (which rebuilds
# _amdsmi instrumented — keep LAST, the box copy is disposable).
set -x
cd /root/repo
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0 PYTHONPATH=/root/repo

timeout 600 python -m pytest tests -m gpu -q > gpurun_out/pytest_gpu_r2.log 2>&1
echo "pytest rc=$?" >> gpurun_out/pytest_gpu_r2.log
tail -5 gpurun_out/pytest_gpu_r2.log

timeout 240 python bench.py --steps 300 --warmup 30 > gpurun_out/bench_r2.json 2> gpurun_out/bench_r2.err
echo "bench rc=$?" >> gpurun_out/bench_r2.err
timeout 240 python bench.py --single-process --gpus 8 --steps 300 --warmup 30 \
  > gpurun_out/bench_r2_singleproc.json 2>> gpurun_out/bench_r2.err
echo "bench sp rc=$?" >> gpurun_out/bench_r2.err

timeout 120 python - > gpurun_out/live_telemetry_r2.log 2>&1 <<'PYEOF'
import json, sys
sys.path.insert(0, "/root/repo")
from gpud_amd.smi import _amdsmi
_amdsmi.init()
snap = _amdsmi.metrics_snapshot(0)
act = snap.get("activity") or {}
print("xcc_busy_pct:", act.get("xcc_busy_pct"))
cper = _amdsmi.cper_entries(0, 0xFFFFFFFF, 0, 4)
print("cper supported:", cper.get("supported"), "entries:", len(cper.get("entries", [])))
for e in list(cper.get("entries", []))[:3]:
    print("cper entry:", json.dumps({k: v for k, v in e.items()}, default=str)[:400])
PYEOF

bash scripts/gpu_asan_stress.sh

cat gpurun_out/bench_r2.json gpurun_out/bench_r2_singleproc.json
