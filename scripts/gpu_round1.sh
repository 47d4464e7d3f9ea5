#!/usr/bin/env bash
# First GPU validation pass: gpu tests, bench, rocprof of diag kernels.
set -x
cd /root/repo
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0
export PYTHONPATH=/root/repo

rocm-smi --showproductname > gpurun_out/rocm_smi.log 2>&1 || true

timeout 420 python -m pytest tests -m gpu -x -q > gpurun_out/pytest_gpu.log 2>&1
echo "pytest rc=$?" >> gpurun_out/pytest_gpu.log

timeout 300 python bench.py --steps 200 --warmup 20 > gpurun_out/bench1.json 2> gpurun_out/bench1.err
echo "bench rc=$?" >> gpurun_out/bench1.err

cd /tmp && export TMPDIR=/tmp
timeout 300 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/prof -- \
  python -c "
from gpud_amd.diag import _diag
_diag.set_device(0)
print('mfma_bf16', _diag.mfma_stress_bf16(iters=1024, workgroups=1024))
print('mfma_fp8 ', _diag.mfma_stress_fp8(iters=1024, workgroups=1024))
print('hbm      ', _diag.hbm_bandwidth(buffer_gb=2.0, iters=4))
print('lds      ', _diag.lds_bandwidth(iters=20000, workgroups=512))
" > /root/repo/gpurun_out/rocprof.log 2>&1
echo "rocprof rc=$?" >> /root/repo/gpurun_out/rocprof.log
tail -5 /root/repo/gpurun_out/pytest_gpu.log
cat /root/repo/gpurun_out/bench1.json
