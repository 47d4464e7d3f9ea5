#!/usr/bin/env bash
# Round-2 GEMM A/B: shipped v2 (glds 8-phase, boundary barriers) vs the new
# v5 (register-staged, no LDS, no barriers) — interleaved same-box reps per
# methodology rule 24, then PMC MfmaUtil on the winner's shape.
set -x
cd /root/repo
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0
export PYTHONPATH=/root/repo

timeout 900 python - > gpurun_out/gemm_ab_v6.log 2>&1 <<'EOF'
from gpud_amd.diag import _diag
_diag.set_device(0)
import json
variants = {
    "v2":      lambda s: _diag.gemm_stress_bf16_v2(size=s, iters=5),
    "v6":      lambda s: _diag.gemm_stress_bf16_v6(size=s, iters=5),
    "v6_nosp": lambda s: _diag.gemm_stress_bf16_v6_nosp(size=s, iters=5),
}
for size in (4096, 8192):
    for rep in range(3):
        for name, fn in variants.items():
            r = fn(size)
            print(json.dumps({"size": size, "rep": rep, "variant": name,
                              "tflops": round(r["tflops"], 1),
                              "verified": r["verified"]}), flush=True)
EOF
echo "ab rc=$?" >> gpurun_out/gemm_ab_v6.log

timeout 420 bash scripts/gpu_asan_stress.sh > gpurun_out/asan_run.log 2>&1
echo "asan rc=$?" >> gpurun_out/asan_run.log
# the asan script replaced the box-local _amdsmi with the instrumented
# build; the diag module is untouched, so the PMC pass below still runs
cd /tmp && export TMPDIR=/tmp
timeout 420 rocprofv3 --pmc SQ_VALU_MFMA_BUSY_CYCLES,GRBM_GUI_ACTIVE,SQ_WAVE_CYCLES,SQ_WAIT_ANY \
  -d /root/repo/gpurun_out/pmc_v6 -- \
  python -c "
from gpud_amd.diag import _diag
_diag.set_device(0)
print('v6@8192', _diag.gemm_stress_bf16_v6(size=8192, iters=2))
" > /root/repo/gpurun_out/pmc_v6.log 2>&1
echo "pmc rc=$?" >> /root/repo/gpurun_out/pmc_v6.log
# per-dispatch counter summary (counter_collection.csv lives under -d)
python3 - >> /root/repo/gpurun_out/pmc_v6.log 2>&1 <<'EOF'
import glob, csv, collections
for f in glob.glob('/root/repo/gpurun_out/pmc_v6/**/*counter_collection.csv', recursive=True):
    agg = collections.defaultdict(lambda: collections.defaultdict(float))
    rows = list(csv.DictReader(open(f)))
    for r in rows:
        k = r.get('Kernel_Name', '')[:60]
        agg[k][r['Counter_Name']] += float(r['Counter_Value'])
    for k, c in agg.items():
        if 'v5' in k or 'mfma' in k.lower() or 'gemm' in k.lower():
            mfma = c.get('SQ_VALU_MFMA_BUSY_CYCLES', 0.0)
            gui = c.get('GRBM_GUI_ACTIVE', 0.0)
            util = mfma / (1024 * gui) * 100 if gui else 0.0
            print(f, k, dict(c), 'MfmaUtil%%=%.1f' % util)
EOF

tail -30 /root/repo/gpurun_out/gemm_ab_v6.log
