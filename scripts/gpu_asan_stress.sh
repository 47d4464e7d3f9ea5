#!/usr/bin/env bash
# ASan hunt for the round-1 soak heap corruption (docs/ROADMAP.md
# "Observed-but-unresolved"): rebuild _amdsmi with AddressSanitizer ON THE
# GPU BOX (the snapshot copy is disposable), then run a targeted
# concurrency stress twice — first with the call mutex held (the shipped
# mitigation), then with GPUD_AMDSMI_NO_CALL_MUTEX=1 so concurrent amdsmi
# entry actually happens and a race in the binding or libamd_smi trips
# ASan. Cores enabled throughout.
set -x
cd /root/repo
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0 PYTHONPATH=/root/repo
ulimit -c unlimited

PYEXT=$(python3-config --extension-suffix)
ROCM=${ROCM_PATH:-/opt/rocm}

# ASan build over the in-tree module (box-local copy only)
g++ -O1 -g -fno-omit-frame-pointer -fsanitize=address -shared -fPIC \
    -std=c++17 $(python3 -m pybind11 --includes) \
    -I${ROCM}/include csrc/smi/amdsmi_ext.cpp \
    -L${ROCM}/lib -lamd_smi -Wl,-rpath,${ROCM}/lib \
    -o gpud_amd/smi/_amdsmi${PYEXT} 2> gpurun_out/asan_build.log
echo "asan build rc=$?" >> gpurun_out/asan_build.log

ASAN_LIB=$(g++ -print-file-name=libasan.so)
# libstdc++ must be preloaded right after libasan: libamd_smi throws (and
# catches) std::ios_failure during KFD discovery, and without libstdc++ in
# the preload chain ASan's lazy __cxa_throw interceptor cannot resolve the
# real symbol ("real___cxa_throw != 0" CHECK abort at init)
STDCXX=$(ldconfig -p | awk '/libstdc\+\+\.so\.6 \(libc6,x86-64\)/{print $NF; exit}')
export ASAN_OPTIONS="detect_leaks=0:abort_on_error=1:disable_coredump=0:unmap_shadow_on_exit=1:verify_asan_link_order=0"

run_stress() {
  local label="$1"
  timeout 600 env LD_PRELOAD="${ASAN_LIB} ${STDCXX}" python - > "gpurun_out/asan_${label}.log" 2>&1 <<'PYEOF'
import concurrent.futures as cf
import os, random, sys, time
sys.path.insert(0, "/root/repo")
from gpud_amd.smi import _amdsmi

_amdsmi.init()
n = _amdsmi.device_count()
print("devices:", n, "mutex_disabled:", os.environ.get("GPUD_AMDSMI_NO_CALL_MUTEX"))

OPS = [
    lambda i: _amdsmi.metrics_snapshot(i),
    lambda i: _amdsmi.asic_info(i),
    lambda i: _amdsmi.vram_usage(i),
    lambda i: _amdsmi.activity(i),
    lambda i: _amdsmi.power_info(i),
    lambda i: _amdsmi.clock_info(i),
    lambda i: _amdsmi.violation_status(i),
    lambda i: _amdsmi.link_metrics(i),
    lambda i: _amdsmi.partition_info(i),
    lambda i: _amdsmi.pcie_info(i),
    lambda i: _amdsmi.xgmi_error_status(i),
    lambda i: _amdsmi.ecc_count_total(i),
    lambda i: _amdsmi.process_list(i),
    lambda i: _amdsmi.bad_page_info(i),
    lambda i: _amdsmi.xgmi_link_status(i),
    lambda i: _amdsmi.cper_entries(i, 0xFFFFFFFF, 0, 4),
    lambda i: _amdsmi.board_info(i),
    lambda i: _amdsmi.driver_info(),
]

def worker(seed):
    rnd = random.Random(seed)
    t_end = time.time() + 150
    calls = 0
    while time.time() < t_end:
        op = rnd.choice(OPS)
        try:
            op(rnd.randrange(max(n, 1)))
        except Exception:
            pass  # NOT_SUPPORTED etc. — we only hunt memory errors
        calls += 1
    return calls

with cf.ThreadPoolExecutor(max_workers=12) as ex:
    totals = list(ex.map(worker, range(12)))
print("stress ok, calls:", sum(totals))
PYEOF
  echo "stress ${label} rc=$?" >> "gpurun_out/asan_${label}.log"
  ls core* /tmp/core* 2>/dev/null >> "gpurun_out/asan_${label}.log" || true
}

# arm 1: shipped shape (mutex held) — validates the mitigation under ASan
run_stress "mutex_on"
# arm 2: mutex disabled — concurrent library entry, the actual race hunt
GPUD_AMDSMI_NO_CALL_MUTEX=1 run_stress "mutex_off"

tail -n 5 gpurun_out/asan_mutex_on.log; tail -n 5 gpurun_out/asan_mutex_off.log
