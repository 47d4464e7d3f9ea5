"""gpud_amd — an MI355X-native GPU health-monitoring daemon.

A from-scratch rebuild of the capabilities of leptonai/gpud (reference:
/root/reference, pure Go + go-nvml) designed AMD-first:

- telemetry through a native pybind11 C++ binding of ROCm's libamd_smi
  (``csrc/smi``), one shared handle for all components;
- an amdgpu RAS / dmesg event watcher with a curated kernel-message catalog
  (the equivalent of gpud's NVRM Xid catalog,
  reference: components/accelerator/nvidia/xid/xid.go);
- xGMI link-health readers (the NVLink/fabric-manager/infiniband trio on
  single-node MI355X, reference: components/accelerator/nvidia/nvlink);
- active diagnostics as hand-written HIP/CDNA4 kernels (MFMA bf16/fp8 GEMM
  stress, LDS/HBM bandwidth) plus an RCCL all-reduce xGMI fabric check
  (``csrc/diag``), rocprof-verified.

The daemon framework (component registry, per-minute pollers, SQLite
event/metric stores, HTTPS REST API, CLI, control-plane session) mirrors the
reference's architecture (reference: pkg/server/server.go:117,
components/registry.go) while staying idiomatic Python; the hot paths
(SMI polling, diag kernels) are native C++/HIP.
"""

__version__ = "0.3.0"
