"""SMI layer: shared amdsmi session + device objects + mock.

The AMD-native equivalent of the reference's NVML instance layer
(reference: pkg/nvidia/nvml/instance.go:43-97 Instance,
device/device.go:14 Device, lib/lib.go mock swap): one shared native
session for all components, cached identity, per-device snapshot getters,
and a whole-library mock selected by ``GPUD_AMDSMI_MOCK=1`` (the analog of
``GPUD_NVML_MOCK_ALL_SUCCESS`` — lib/default.go:15-26) so the real daemon
exercises GPU code paths on GPU-less CI.

Failure injection wraps the real/mock backend at this layer
(reference: nvml.NewWithFailureInjector — instance.go:115).
"""

from __future__ import annotations

import os
import threading
from typing import Any, Dict, List, Optional

from ..pkg.fault_injector import SMIFailureInjector
from ..pkg.log import logger

from . import mock as mock_backend

MOCK_ENV = "GPUD_AMDSMI_MOCK"

_EXT_IMPORT_ERROR: Optional[Exception] = None
try:
    from . import _amdsmi  # type: ignore[attr-defined]
except Exception as e:  # pragma: no cover - build missing
    _amdsmi = None
    _EXT_IMPORT_ERROR = e


def mock_enabled() -> bool:
    return os.environ.get(MOCK_ENV, "") in ("1", "true", "TRUE", "yes")


class Device:
    """Cached identity + live getters for one GPU."""

    def __init__(self, backend: Any, index: int):
        self._b = backend
        self.index = index
        self.uuid: str = backend.device_uuid(index)
        self.bdf: str = backend.device_bdf(index)
        asic = backend.asic_info(index)
        self.product_name: str = asic.get("market_name", "")
        self.num_compute_units: int = int(asic.get("num_compute_units", 0))
        self.asic_serial: str = asic.get("asic_serial", "")
        self.oam_id: int = int(asic.get("oam_id", 0))
        try:
            board = backend.board_info(index)
            self.board_product: str = board.get("product_name", "")
            self.board_serial: str = board.get("product_serial", "")
        except Exception:
            self.board_product = ""
            self.board_serial = ""

    # live telemetry — one native call for the whole sweep
    def snapshot(self) -> Dict[str, Any]:
        return self._b.metrics_snapshot(self.index)

    def temp_metric(self, sensor: int, metric: int) -> int:
        return self._b.temp_metric(self.index, sensor, metric)

    def power_info(self) -> Dict[str, Any]:
        return self._b.power_info(self.index)

    def clock_info(self, clk_type: int) -> Dict[str, Any]:
        return self._b.clock_info(self.index, clk_type)

    def activity(self) -> Dict[str, Any]:
        return self._b.activity(self.index)

    def vram_usage(self) -> Dict[str, Any]:
        return self._b.vram_usage(self.index)

    def ecc_count_total(self) -> Dict[str, Any]:
        return self._b.ecc_count_total(self.index)

    def ecc_count_block(self, block: int) -> Dict[str, Any]:
        return self._b.ecc_count_block(self.index, block)

    def bad_page_info(self) -> Dict[str, Any]:
        return self._b.bad_page_info(self.index)

    def process_list(self) -> List[Dict[str, Any]]:
        return list(self._b.process_list(self.index))

    def power_management_enabled(self) -> bool:
        return bool(self._b.power_management_enabled(self.index))

    def violation_status(self) -> Dict[str, Any]:
        return self._b.violation_status(self.index)

    def xgmi_link_status(self) -> Dict[str, Any]:
        return self._b.xgmi_link_status(self.index)

    def xgmi_error_status(self) -> int:
        return self._b.xgmi_error_status(self.index)

    def link_metrics(self) -> Dict[str, Any]:
        return self._b.link_metrics(self.index)

    def vram_info(self) -> Dict[str, Any]:
        return self._b.vram_info(self.index)

    def vbios_info(self) -> Dict[str, Any]:
        return self._b.vbios_info(self.index)

    def partition_info(self) -> Dict[str, Any]:
        return self._b.partition_info(self.index)

    def pcie_info(self) -> Dict[str, Any]:
        return self._b.pcie_info(self.index)

    def cper_entries(
        self, severity_mask: int = 0xFFFFFFFF, cursor: int = 0
    ) -> Dict[str, Any]:
        return self._b.cper_entries(
            self.index, severity_mask=severity_mask, cursor=cursor
        )


def rocm_version() -> str:
    try:
        with open(os.path.join(os.environ.get("ROCM_PATH", "/opt/rocm"), ".info", "version")) as f:
            return f.read().strip()
    except OSError:
        return ""


class Instance:
    """Shared SMI session (reference nvml.Instance shape).

    Construct via :func:`new`. ``exists`` is False when no AMD GPU driver is
    loaded — components then report healthy no-op like the reference's
    NVML-missing path (reference: instance.go:101 New no-op on missing lib).
    """

    def __init__(
        self,
        backend: Any = None,
        init_error: str = "",
        failure_injector: Optional[SMIFailureInjector] = None,
    ):
        self._b = backend
        self._init_error = init_error
        self.failure_injector = failure_injector
        self._lock = threading.Lock()
        self._devices: Dict[str, Device] = {}
        self._device_order: List[str] = []
        if backend is not None and not init_error:
            try:
                for i in range(backend.device_count()):
                    d = Device(backend, i)
                    self._devices[d.uuid] = d
                    self._device_order.append(d.uuid)
            except Exception as e:
                self._init_error = f"device enumeration failed: {e}"

    # -- lifecycle ----------------------------------------------------------

    @property
    def exists(self) -> bool:
        return self._b is not None and not self._init_error

    def init_error(self) -> str:
        return self._init_error

    def shutdown(self) -> None:
        if self._b is not None:
            try:
                self._b.shutdown()
            except Exception:
                pass

    # -- identity -----------------------------------------------------------

    def devices(self) -> Dict[str, Device]:
        if self.failure_injector and self.failure_injector.device_enumeration_error:
            raise RuntimeError(self.failure_injector.device_enumeration_error)
        fi = self.failure_injector
        if fi and fi.gpu_lost_uuids:
            return {
                u: d for u, d in self._devices.items() if u not in fi.gpu_lost_uuids
            }
        return dict(self._devices)

    def device_uuids(self) -> List[str]:
        return list(self._device_order)

    def device_count(self) -> int:
        return len(self.devices())

    @property
    def product_name(self) -> str:
        fi = self.failure_injector
        if fi and fi.product_name_override:
            return fi.product_name_override
        for u in self._device_order:
            return self._devices[u].product_name
        return ""

    @property
    def driver_version(self) -> str:
        for u in self._device_order:
            try:
                return self._devices[u]._b.driver_info(self._devices[u].index).get(
                    "driver_version", ""
                )
            except Exception:
                return ""
        return ""

    @property
    def rocm_version(self) -> str:
        return rocm_version()

    # -- the poll hot path --------------------------------------------------

    def snapshot_all(self) -> Dict[str, Dict[str, Any]]:
        """Telemetry snapshots keyed by uuid, one native sweep, with
        failure-injection overlays applied (reference FailureInjector
        semantics, components/registry.go:82-109)."""
        if not self.exists:
            return {}
        if self.failure_injector and self.failure_injector.device_enumeration_error:
            raise RuntimeError(self.failure_injector.device_enumeration_error)
        raw = self._b.metrics_snapshot_all()
        out: Dict[str, Dict[str, Any]] = {}
        fi = self.failure_injector
        for i, uuid in enumerate(self._device_order):
            if i >= len(raw):
                break
            if fi and uuid in fi.gpu_lost_uuids:
                continue
            snap = raw[i]
            if fi is not None:
                snap = self._apply_injection(uuid, snap, fi)
            out[uuid] = snap
        return out

    @staticmethod
    def _apply_injection(
        uuid: str, snap: Dict[str, Any], fi: SMIFailureInjector
    ) -> Dict[str, Any]:
        if not fi.any_active():
            return snap
        snap = dict(snap)
        if uuid in fi.ecc_uncorrectable_uuids:
            ecc = dict(snap.get("ecc") or {})
            ecc["uncorrectable"] = int(ecc.get("uncorrectable", 0)) + 4
            snap["ecc"] = ecc
        if uuid in fi.throttle_uuids or uuid in fi.thermal_throttle_uuids:
            v = dict(snap.get("violation") or {})
            if uuid in fi.throttle_uuids:
                v["active_ppt_pwr"] = 1
                v["acc_ppt_pwr"] = int(v.get("acc_ppt_pwr", 0)) + 1
            if uuid in fi.thermal_throttle_uuids:
                v["active_socket_thrm"] = 1
                v["acc_socket_thrm"] = int(v.get("acc_socket_thrm", 0)) + 1
            snap["violation"] = v
        if uuid in fi.xgmi_unhealthy_uuids:
            x = dict(snap.get("xgmi_link_status") or {"total_links": 7, "states": [1] * 7})
            states = list(x.get("states") or [])
            if states:
                states[0] = 0  # first link down
            x["states"] = states
            snap["xgmi_link_status"] = x
            snap["xgmi_error_status"] = 1
        if uuid in fi.bad_page_pending_uuids or uuid in fi.bad_page_threshold_uuids:
            bp = dict(snap.get("bad_pages") or {})
            bp["total"] = int(bp.get("total", 0)) + (1 if uuid in fi.bad_page_pending_uuids else 0)
            if uuid in fi.bad_page_pending_uuids:
                bp["pending"] = int(bp.get("pending", 0)) + 1
            if uuid in fi.bad_page_threshold_uuids:
                bp["threshold"] = 1
                bp["total"] = max(int(bp.get("total", 0)), 2)
            snap["bad_pages"] = bp
        return snap


def new(
    failure_injector: Optional[SMIFailureInjector] = None,
) -> Instance:
    """Create the shared Instance: mock if GPUD_AMDSMI_MOCK=1, else real
    amdsmi; missing driver/hardware yields a non-exists Instance (no-op)."""
    if mock_enabled():
        backend = mock_backend.MockBackend.from_env()
        return Instance(backend=backend, failure_injector=failure_injector)
    if _amdsmi is None:
        return Instance(
            backend=None,
            init_error=f"_amdsmi extension unavailable: {_EXT_IMPORT_ERROR}",
            failure_injector=failure_injector,
        )
    try:
        _amdsmi.init()
    except Exception as e:
        logger.info("amdsmi init failed (no AMD GPU?): %s", e)
        return Instance(backend=None, init_error=str(e), failure_injector=failure_injector)
    return Instance(backend=_amdsmi, failure_injector=failure_injector)
