#!/usr/bin/env bash
# Second GPU pass: full gpu test suite, longer bench, PMC counters on the
# MFMA + LDS kernels (counters in their own rocprofv3 run, per pool rules).
set -x
cd /root/repo
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0
export PYTHONPATH=/root/repo

timeout 420 python -m pytest tests -m gpu -q > gpurun_out/pytest_gpu2.log 2>&1
echo "pytest rc=$?" >> gpurun_out/pytest_gpu2.log

timeout 300 python bench.py --steps 500 --warmup 50 > gpurun_out/bench2.json 2> gpurun_out/bench2.err
echo "bench rc=$?" >> gpurun_out/bench2.err

cd /tmp && export TMPDIR=/tmp
timeout 300 rocprofv3 --pmc SQ_VALU_MFMA_BUSY_CYCLES,GRBM_GUI_ACTIVE,SQ_WAVE_CYCLES \
  -d /root/repo/gpurun_out/pmc_mfma -- \
  python -c "
from gpud_amd.diag import _diag
_diag.set_device(0)
print('bf16', _diag.mfma_stress_bf16(iters=1024, workgroups=1024))
" > /root/repo/gpurun_out/pmc_mfma.log 2>&1
echo "pmc mfma rc=$?" >> /root/repo/gpurun_out/pmc_mfma.log

timeout 300 rocprofv3 --pmc SQ_LDS_BANK_CONFLICT,SQ_LDS_IDX_ACTIVE \
  -d /root/repo/gpurun_out/pmc_lds -- \
  python -c "
from gpud_amd.diag import _diag
_diag.set_device(0)
print('lds', _diag.lds_bandwidth(iters=20000, workgroups=512))
" > /root/repo/gpurun_out/pmc_lds.log 2>&1
echo "pmc lds rc=$?" >> /root/repo/gpurun_out/pmc_lds.log

tail -3 /root/repo/gpurun_out/pytest_gpu2.log
cat /root/repo/gpurun_out/bench2.json
