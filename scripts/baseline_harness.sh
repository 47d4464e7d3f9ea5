#!/usr/bin/env bash
# Baseline harness: the measured numbers for the
# BASELINE.json configs on one MI355X —
#   config 2: single-GPU amdsmi telemetry poll (bench.py)
#   config 4: active diag MFMA/HBM/LDS/GEMM (+ rocprof evidence)
#   config 5: RCCL fabric check + fault-injector replay load
# Writes gpurun_out/baseline_summary.json for commit into profiles/.
set -x
cd /root/repo
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0
export PYTHONPATH=/root/repo

timeout 420 python -m pytest tests -m gpu -q > gpurun_out/pytest_gpu3.log 2>&1
echo "pytest rc=$?" >> gpurun_out/pytest_gpu3.log

timeout 300 python bench.py --steps 2000 --warmup 100 > gpurun_out/bench_poll.json 2>/dev/null
timeout 200 python bench.py --steps 200 --warmup 20 --fault-replay > gpurun_out/bench_faults.json 2>/dev/null

timeout 300 python - > gpurun_out/diag_summary.json 2> gpurun_out/diag_summary.err <<'EOF'
import json, subprocess, os
from gpud_amd.diag import _diag
_diag.set_device(0)
out = {}
out["mfma_bf16"] = _diag.mfma_stress_bf16(iters=2048, workgroups=1024)
out["mfma_fp8"] = _diag.mfma_stress_fp8(iters=2048, workgroups=1024)
out["mfma_mxfp8"] = _diag.mfma_stress_mxfp8(iters=2048, workgroups=1024)
out["mfma_mxfp4"] = _diag.mfma_stress_mxfp4(iters=2048, workgroups=1024)
out["gemm_bf16_v7p_4096"] = _diag.gemm_stress_bf16_v7(size=4096, iters=5)
out["gemm_bf16_v7p_8192"] = _diag.gemm_stress_bf16_v7(size=8192, iters=5)
out["gemm_mxfp8_8192"] = _diag.gemm_stress_mxfp8(size=8192, iters=5)
out["hbm"] = _diag.hbm_bandwidth(buffer_gb=4.0, iters=8)
out["lds"] = _diag.lds_bandwidth(iters=20000, workgroups=512)
env = dict(os.environ); env.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
fc = subprocess.run(["gpud_amd/diag/gpud-fabric-check", "--max-bytes", str(64<<20), "--iters", "2"],
                    capture_output=True, text=True, timeout=240, env=env)
out["fabric_check"] = json.loads(fc.stdout.strip().splitlines()[-1]) if fc.stdout.strip() else {"rc": fc.returncode}
print(json.dumps(out, indent=1, default=str))
EOF

cd /tmp && export TMPDIR=/tmp
timeout 240 rocprofv3 --pmc SQ_VALU_MFMA_BUSY_CYCLES,GRBM_GUI_ACTIVE,SQ_WAVE_CYCLES \
  -d /root/repo/gpurun_out/pmc_gemm -- \
  python -c "
from gpud_amd.diag import _diag
_diag.set_device(0)
print('gemm', _diag.gemm_stress_bf16(size=8192, iters=1))
" > /root/repo/gpurun_out/pmc_gemm.log 2>&1

cd /root/repo
python - > gpurun_out/baseline_summary.json <<'EOF'
import json
s = {}
for name, path in [("poll", "gpurun_out/bench_poll.json"),
                   ("fault_replay", "gpurun_out/bench_faults.json"),
                   ("diag", "gpurun_out/diag_summary.json")]:
    try:
        s[name] = json.load(open(path))
    except Exception as e:
        s[name] = {"error": str(e)}
print(json.dumps(s, indent=1, default=str))
EOF
tail -2 gpurun_out/pytest_gpu3.log
cat gpurun_out/bench_poll.json
