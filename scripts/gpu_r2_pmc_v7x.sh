#!/usr/bin/env bash
# Wait-source decomposition for v7X: list available counters, then PMC
# passes — (1) LDS activity, (2) any SQ_WAIT_* decomposition counters the
# image exposes, (3) the MfmaUtil reference set.
set -x
cd /tmp && export TMPDIR=/tmp
export HSA_ENABLE_IPC_MODE_LEGACY=0 PYTHONPATH=/root/repo
mkdir -p /root/repo/gpurun_out

rocprofv3 --list-avail > /root/repo/gpurun_out/counters_avail.txt 2>&1 || true
grep -oE "SQ_[A-Z0-9_]+" /root/repo/gpurun_out/counters_avail.txt | sort -u \
  > /root/repo/gpurun_out/sq_counters.txt
wc -l /root/repo/gpurun_out/sq_counters.txt

RUN='
from gpud_amd.diag import _diag
_diag.set_device(0)
print(_diag.gemm_stress_bf16_v7_style(size=8192, iters=2, style=5))
print(_diag.gemm_stress_bf16_v7_style(size=4096, iters=2, style=5))
'

timeout 300 rocprofv3 --pmc SQ_LDS_IDX_ACTIVE,SQ_LDS_BANK_CONFLICT,GRBM_GUI_ACTIVE,SQ_VALU_MFMA_BUSY_CYCLES \
  -d /root/repo/gpurun_out/pmc_v7x_lds -- python -c "$RUN" \
  > /root/repo/gpurun_out/pmc_v7x_lds.log 2>&1
echo "lds rc=$?" >> /root/repo/gpurun_out/pmc_v7x_lds.log

# wait decomposition: pick up to 4 SQ_WAIT*-ish counters that exist
WAITS=$(grep -E "SQ_(WAIT|ACTIVE)_INST|SQ_WAIT_" /root/repo/gpurun_out/sq_counters.txt | head -4 | paste -sd,)
echo "wait counters chosen: $WAITS"
if [ -n "$WAITS" ]; then
  timeout 300 rocprofv3 --pmc "$WAITS" \
    -d /root/repo/gpurun_out/pmc_v7x_wait -- python -c "$RUN" \
    > /root/repo/gpurun_out/pmc_v7x_wait.log 2>&1
  echo "wait rc=$?" >> /root/repo/gpurun_out/pmc_v7x_wait.log
fi

grep -E "SQ_(WAIT|ACTIVE)" /root/repo/gpurun_out/sq_counters.txt | head -40
tail -6 /root/repo/gpurun_out/pmc_v7x_lds.log
