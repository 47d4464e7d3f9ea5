#!/usr/bin/env bash
# Build every native extension IN-TREE (the built .so files travel to the GPU
# box with the repo snapshot; a JIT cache would not).
#   csrc/smi/amdsmi_ext.cpp -> gpud_amd/smi/_amdsmi.*.so      (C++, libamd_smi)
#   csrc/diag/*.hip         -> gpud_amd/diag/_diag.*.so       (HIP, gfx950)
#   csrc/diag/fabric_rccl.cpp -> gpud_amd/diag/gpud-fabric-check (C++, librccl)
set -euo pipefail
cd "$(dirname "$0")/.."

PYEXT=$(python3-config --extension-suffix)
PYINC=$(python3 -m pybind11 --includes)
ROCM=${ROCM_PATH:-/opt/rocm}

echo "[build] smi extension"
g++ -O2 -shared -fPIC -std=c++17 ${PYINC} \
    -I${ROCM}/include csrc/smi/amdsmi_ext.cpp \
    -L${ROCM}/lib -lamd_smi -Wl,-rpath,${ROCM}/lib \
    -o gpud_amd/smi/_amdsmi${PYEXT}

if [ -f csrc/diag/diag_ext.hip ]; then
  echo "[build] diag HIP extension (gfx950)"
  ${ROCM}/bin/hipcc --offload-arch=gfx950 -O3 -std=c++17 -shared -fPIC ${PYINC} \
      csrc/diag/diag_ext.hip \
      -o gpud_amd/diag/_diag${PYEXT}
fi

if [ -f csrc/diag/fabric_rccl.cpp ]; then
  echo "[build] rccl fabric check binary"
  ${ROCM}/bin/hipcc --offload-arch=gfx950 -O2 -std=c++17 \
      -I${ROCM}/include csrc/diag/fabric_rccl.cpp \
      -L${ROCM}/lib -lrccl -Wl,-rpath,${ROCM}/lib \
      -o gpud_amd/diag/gpud-fabric-check
fi

echo "[build] done"
