"""Active diagnostic components (manual run mode).

The DCGM-diag analog the reference lacks (its "diagnostic" is
nvidia-bug-report.sh collection — reference: pkg/session/diagnostic.go:48;
BASELINE.json names these as first-class new components):

  * accelerator-amd-diag-mfma       — CDNA4 MFMA bf16/fp8 stress (csrc/diag)
  * accelerator-amd-diag-bandwidth  — HBM3E triad + LDS sweep
  * accelerator-amd-diag-fabric     — RCCL all-reduce / pairwise xGMI check

All are RunMode=manual (reference custom-plugin RunModeTypeManual —
api/v1 types): they never run on the ticker; the server's trigger endpoint
or the session's ``diagnostic`` method runs them on demand. Results are
thresholded against per-board floors and the MFMA kernels carry built-in
numeric verification (all-ones inputs ⇒ exact expected accumulators).

On a host with an AMD GPU (/dev/kfd present) a missing native extension is
an UNHEALTHY result, never a silent skip — the HIP path must be the one
that runs.
"""

from __future__ import annotations

import json
import os
import subprocess
from typing import Any, Callable, Dict, List

from ...apiv1.types import HealthStateType, RepairActionType, RunModeType, SuggestedActions
from ..base import CheckResult, GPUdInstance, TickerComponent

NAME_MFMA = "accelerator-amd-diag-mfma"
NAME_BANDWIDTH = "accelerator-amd-diag-bandwidth"
NAME_FABRIC = "accelerator-amd-diag-fabric"

# Per-board floors for a healthy MI355X (measured ceilings: bf16 MFMA
# ~2382 TF register-resident, HBM ~6.3 TB/s, LDS ~150 TB/s chip-wide —
# /opt/skills/guides/MI355X_MICROARCH.md). Floors sit well below so only a
# genuinely sick board trips them.
DEFAULT_BF16_TFLOPS_FLOOR = 1600.0
DEFAULT_FP8_TFLOPS_FLOOR = 1600.0  # non-scaled fp8 runs at the bf16 rate
DEFAULT_MXFP8_TFLOPS_FLOOR = 3400.0  # MX-scaled path, measured ~4790 TF
DEFAULT_MXFP4_TFLOPS_FLOOR = 6500.0  # fp4 MX path, measured ~8730 TF
DEFAULT_GEMM_TFLOPS_FLOOR = 1000.0  # v7P asm-K-loop bf16 GEMM, ~1390-1490
DEFAULT_GEMM_FP8_TFLOPS_FLOOR = 1400.0  # 8-phase MX-fp8 GEMM, measured ~1940
DEFAULT_HBM_GBPS_FLOOR = 4500.0
DEFAULT_LDS_TBPS_FLOOR = 60.0
DEFAULT_XGMI_PAIR_GBPS_FLOOR = 30.0  # per direction, pairwise sendrecv


def gpu_present() -> bool:
    return os.path.exists("/dev/kfd")


def _load_diag():
    from ...diag import _diag  # type: ignore[attr-defined]

    return _diag


class _ManualDiagComponent(TickerComponent):
    run_mode = RunModeType.MANUAL

    def __init__(self, inst: GPUdInstance):
        super().__init__()
        self._inst = inst

    def tags(self) -> List[str]:
        return ["accelerator", "amd", "gpu", "diag", self.name]

    def is_supported(self) -> bool:
        return gpu_present()

    def _ext_missing_result(self, err: Exception) -> CheckResult:
        if gpu_present():
            return CheckResult(
                self.name,
                health=HealthStateType.UNHEALTHY,
                reason="GPU present but the native diag extension is not "
                "built/loadable — run csrc/build.sh",
                error=str(err),
                run_mode=self.run_mode,
            )
        return CheckResult(
            self.name,
            reason="no AMD GPU on this host; diagnostics not applicable",
            run_mode=self.run_mode,
        )


class MFMADiagComponent(_ManualDiagComponent):
    def __init__(self, inst: GPUdInstance):
        super().__init__(inst)
        self.bf16_floor = DEFAULT_BF16_TFLOPS_FLOOR
        self.fp8_floor = DEFAULT_FP8_TFLOPS_FLOOR
        self.mxfp8_floor = DEFAULT_MXFP8_TFLOPS_FLOOR
        self.mxfp4_floor = DEFAULT_MXFP4_TFLOPS_FLOOR
        self.gemm_floor = DEFAULT_GEMM_TFLOPS_FLOOR
        self.gemm_fp8_floor = DEFAULT_GEMM_FP8_TFLOPS_FLOOR
        self.iters = 2048
        self.workgroups = 1024
        self.gemm_size = 4096
        self.gemm_iters = 3

    @property
    def name(self) -> str:
        return NAME_MFMA

    def check(self) -> CheckResult:
        try:
            diag = _load_diag()
        except Exception as e:
            return self._ext_missing_result(e)
        try:
            ndev = int(diag.device_info()["device_count"])
        except Exception as e:
            return self._ext_missing_result(e)
        if ndev == 0:
            return CheckResult(
                self.name, reason="no HIP devices", run_mode=self.run_mode
            )
        failures, extra = [], {}
        for dev in range(ndev):
            diag.set_device(dev)
            bf16 = diag.mfma_stress_bf16(iters=self.iters, workgroups=self.workgroups)
            fp8 = diag.mfma_stress_fp8(iters=self.iters, workgroups=self.workgroups)
            mxfp8 = diag.mfma_stress_mxfp8(
                iters=self.iters, workgroups=self.workgroups
            )
            mxfp4 = diag.mfma_stress_mxfp4(
                iters=self.iters, workgroups=self.workgroups
            )
            gemm = diag.gemm_stress_bf16_v7(
                size=self.gemm_size, iters=self.gemm_iters
            )
            gemm_fp8 = diag.gemm_stress_mxfp8(
                size=self.gemm_size, iters=self.gemm_iters
            )
            extra[f"gpu{dev}.bf16_tflops"] = f"{bf16['tflops']:.0f}"
            extra[f"gpu{dev}.fp8_tflops"] = f"{fp8['tflops']:.0f}"
            extra[f"gpu{dev}.mxfp8_tflops"] = f"{mxfp8['tflops']:.0f}"
            extra[f"gpu{dev}.mxfp4_tflops"] = f"{mxfp4['tflops']:.0f}"
            extra[f"gpu{dev}.gemm_bf16_tflops"] = f"{gemm['tflops']:.0f}"
            extra[f"gpu{dev}.gemm_mxfp8_tflops"] = f"{gemm_fp8['tflops']:.0f}"
            for name, res in (
                ("bf16", bf16),
                ("fp8", fp8),
                ("mxfp8", mxfp8),
                ("mxfp4", mxfp4),
                ("gemm", gemm),
                ("gemm_fp8", gemm_fp8),
            ):
                if not res["verified"]:
                    failures.append(
                        f"gpu{dev}: {name} MFMA numeric verification FAILED "
                        f"(bad={res['verify_failures']})"
                    )
            for name, res, floor in (
                ("bf16", bf16, self.bf16_floor),
                ("fp8", fp8, self.fp8_floor),
                ("mxfp8", mxfp8, self.mxfp8_floor),
                ("mxfp4", mxfp4, self.mxfp4_floor),
                ("gemm bf16", gemm, self.gemm_floor),
                ("gemm mx-fp8", gemm_fp8, self.gemm_fp8_floor),
            ):
                if res["tflops"] < floor:
                    failures.append(
                        f"gpu{dev}: {name} {res['tflops']:.0f} TF below floor "
                        f"{floor:.0f}"
                    )
        if failures:
            return CheckResult(
                self.name,
                health=HealthStateType.UNHEALTHY,
                reason="; ".join(failures),
                extra_info=extra,
                run_mode=self.run_mode,
                suggested_actions=SuggestedActions(
                    description="MFMA compute diagnostic failed",
                    repair_actions=[RepairActionType.HARDWARE_INSPECTION],
                ),
            )
        return CheckResult(
            self.name,
            reason=f"MFMA stress passed on {ndev} GPU(s)",
            extra_info=extra,
            run_mode=self.run_mode,
        )


class BandwidthDiagComponent(_ManualDiagComponent):
    def __init__(self, inst: GPUdInstance):
        super().__init__(inst)
        self.hbm_floor_gbps = DEFAULT_HBM_GBPS_FLOOR
        self.lds_floor_tbps = DEFAULT_LDS_TBPS_FLOOR
        self.buffer_gb = 4.0
        self.iters = 8

    @property
    def name(self) -> str:
        return NAME_BANDWIDTH

    def check(self) -> CheckResult:
        try:
            diag = _load_diag()
            ndev = int(diag.device_info()["device_count"])
        except Exception as e:
            return self._ext_missing_result(e)
        if ndev == 0:
            return CheckResult(
                self.name, reason="no HIP devices", run_mode=self.run_mode
            )
        failures, extra = [], {}
        for dev in range(ndev):
            diag.set_device(dev)
            hbm = diag.hbm_bandwidth(buffer_gb=self.buffer_gb, iters=self.iters)
            lds = diag.lds_bandwidth(iters=20000, workgroups=512)
            extra[f"gpu{dev}.hbm_triad_gbps"] = f"{hbm['triad_gbps']:.0f}"
            extra[f"gpu{dev}.hbm_read_gbps"] = f"{hbm['read_gbps']:.0f}"
            extra[f"gpu{dev}.lds_tbps"] = f"{lds['lds_tbps']:.1f}"
            if hbm["triad_gbps"] < self.hbm_floor_gbps:
                failures.append(
                    f"gpu{dev}: HBM triad {hbm['triad_gbps']:.0f} GB/s below "
                    f"floor {self.hbm_floor_gbps:.0f}"
                )
            if lds["lds_tbps"] < self.lds_floor_tbps:
                failures.append(
                    f"gpu{dev}: LDS {lds['lds_tbps']:.1f} TB/s below floor "
                    f"{self.lds_floor_tbps:.0f}"
                )
        if failures:
            return CheckResult(
                self.name,
                health=HealthStateType.UNHEALTHY,
                reason="; ".join(failures),
                extra_info=extra,
                run_mode=self.run_mode,
                suggested_actions=SuggestedActions(
                    description="memory-bandwidth diagnostic failed",
                    repair_actions=[RepairActionType.HARDWARE_INSPECTION],
                ),
            )
        return CheckResult(
            self.name,
            reason=f"HBM/LDS bandwidth within expectations on {ndev} GPU(s)",
            extra_info=extra,
            run_mode=self.run_mode,
        )


class FabricDiagComponent(_ManualDiagComponent):
    def __init__(self, inst: GPUdInstance):
        super().__init__(inst)
        self.pair_floor_gbps = DEFAULT_XGMI_PAIR_GBPS_FLOOR
        self.timeout_seconds = 300.0
        self.binary = os.path.join(os.path.dirname(__file__), "..", "..", "diag", "gpud-fabric-check")
        self.max_bytes = 256 << 20
        self.run_binary: Callable = self._run_binary

    @property
    def name(self) -> str:
        return NAME_FABRIC

    def _run_binary(self) -> Dict[str, Any]:
        path = os.path.abspath(self.binary)
        if not os.path.exists(path):
            raise FileNotFoundError(f"fabric-check binary missing: {path}")
        env = dict(os.environ)
        env.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
        out = subprocess.run(
            [path, "--max-bytes", str(self.max_bytes), "--iters", "3"],
            capture_output=True,
            text=True,
            timeout=self.timeout_seconds,
            env=env,
        )
        if out.returncode not in (0, 4):
            raise RuntimeError(
                f"fabric-check exited {out.returncode}: {out.stderr[-400:]}"
            )
        return json.loads(out.stdout.strip().splitlines()[-1])

    def check(self) -> CheckResult:
        if not gpu_present():
            return CheckResult(
                self.name,
                reason="no AMD GPU on this host; fabric check not applicable",
                run_mode=self.run_mode,
            )
        try:
            res = self.run_binary()
        except Exception as e:
            return CheckResult(
                self.name,
                health=HealthStateType.UNHEALTHY,
                reason="RCCL fabric check could not run",
                error=str(e),
                run_mode=self.run_mode,
            )
        if not res.get("ok"):
            return CheckResult(
                self.name,
                health=HealthStateType.UNHEALTHY,
                reason=f"fabric check failed: {res.get('error', 'unknown')}",
                run_mode=self.run_mode,
            )
        ndev = int(res.get("ndev", 0))
        extra = {"ndev": str(ndev)}
        ar = res.get("allreduce") or []
        if ar:
            best = max(a.get("busbw_gbps", 0) for a in ar)
            extra["allreduce_busbw_gbps"] = f"{best:.1f}"
        slow_pairs = []
        for p in res.get("pairwise") or []:
            gbps = float(p.get("bidir_gbps_per_dir", 0))
            extra[f"pair_{p['a']}_{p['b']}_gbps"] = f"{gbps:.1f}"
            if gbps < self.pair_floor_gbps:
                slow_pairs.append(f"{p['a']}<->{p['b']} ({gbps:.1f} GB/s)")
        if not res.get("verified", False):
            return CheckResult(
                self.name,
                health=HealthStateType.UNHEALTHY,
                reason="RCCL all-reduce returned WRONG data — fabric corruption",
                extra_info=extra,
                run_mode=self.run_mode,
                suggested_actions=SuggestedActions(
                    description="data corruption over the xGMI fabric",
                    repair_actions=[RepairActionType.HARDWARE_INSPECTION],
                ),
            )
        if slow_pairs and ndev > 1:
            return CheckResult(
                self.name,
                health=HealthStateType.UNHEALTHY,
                reason="slow GPU pair(s) below "
                f"{self.pair_floor_gbps:.0f} GB/s: " + ", ".join(slow_pairs),
                extra_info=extra,
                run_mode=self.run_mode,
                suggested_actions=SuggestedActions(
                    description="degraded xGMI link bandwidth",
                    repair_actions=[
                        RepairActionType.REBOOT_SYSTEM,
                        RepairActionType.HARDWARE_INSPECTION,
                    ],
                ),
            )
        return CheckResult(
            self.name,
            reason=f"RCCL fabric verified across {ndev} GPU(s)",
            extra_info=extra,
            run_mode=self.run_mode,
        )


def init_funcs():
    return [
        lambda inst: MFMADiagComponent(inst),
        lambda inst: BandwidthDiagComponent(inst),
        lambda inst: FabricDiagComponent(inst),
    ]
