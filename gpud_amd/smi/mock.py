"""In-memory mock SMI backend — a full 8×MI355X node without hardware.

The analog of the reference's whole-library NVML mock
(reference: pkg/nvidia/nvml/lib/default.go:15-26 GPUD_NVML_MOCK_ALL_SUCCESS
+ lib/mock_fixtures.go): enabled via ``GPUD_AMDSMI_MOCK=1``
(``GPUD_AMDSMI_MOCK_GPUS`` sets the device count, default 8), it exposes the
same callable surface as the native ``_amdsmi`` module so the real daemon
runs its GPU code paths on GPU-less CI. Values model a healthy MI355X:
288 GB HBM3E, 256 CUs, 7 xGMI links up, 1400 W board power limit.
"""

from __future__ import annotations

import os
import threading
from typing import Any, Dict, List

MI355X_NAME = "AMD Instinct MI355X"
MI355X_VRAM_MB = 294_912  # 288 GB
MI355X_CUS = 256
MI355X_XGMI_LINKS = 7
MI355X_POWER_LIMIT_W = 1400


class MockBackend:
    """Object with the `_amdsmi` module's function surface."""

    def __init__(self, num_gpus: int = 8):
        self.num_gpus = num_gpus
        self._lock = threading.Lock()
        # mutable per-device state tests may tweak directly
        self.state: List[Dict[str, Any]] = [
            self._default_state(i) for i in range(num_gpus)
        ]

    @staticmethod
    def from_env() -> "MockBackend":
        n = int(os.environ.get("GPUD_AMDSMI_MOCK_GPUS", "8"))
        return MockBackend(num_gpus=n)

    @staticmethod
    def _default_state(i: int) -> Dict[str, Any]:
        return {
            "temp_edge": 42 + i % 3,
            "temp_hotspot": 55 + i % 3,
            "temp_vram": 48 + i % 3,
            "temp_edge_limit": 100,
            "temp_hotspot_limit": 110,
            "temp_vram_limit": 105,
            "temp_hotspot_shutdown": 115,
            "power_w": 620 + 5 * i,
            "power_limit_w": MI355X_POWER_LIMIT_W,
            "gfx_mhz": 2200,
            "gfx_max_mhz": 2400,
            "mem_mhz": 1350,
            "mem_max_mhz": 1400,
            "gfx_activity": 37,
            "umc_activity": 22,
            "mm_activity": 0,
            "vram_used_mb": 2048,
            "ecc_correctable": 0,
            "ecc_uncorrectable": 0,
            "ecc_deferred": 0,
            "bad_pages_total": 0,
            "bad_pages_pending": 0,
            "bad_page_threshold": 256,
            "xgmi_states": [1] * MI355X_XGMI_LINKS,
            "xgmi_error_status": 0,
            "throttle": {},
            "compute_partition": "SPX",
            "memory_partition": "NPS1",
            # list of CPER header dicts; tests append to inject RAS records
            "cper": [],
            "processes": [
                {
                    "name": "python3",
                    "pid": 4242 + i,
                    "mem_bytes": 1 << 30,
                    "vram_mem_bytes": 1 << 30,
                    "gtt_mem_bytes": 0,
                    "gfx_usage": 50,
                    "cu_occupancy": 128,
                }
            ],
        }

    # -- lifecycle ----------------------------------------------------------

    def init(self) -> None:
        pass

    def shutdown(self) -> None:
        pass

    def device_count(self) -> int:
        return self.num_gpus

    # -- identity -----------------------------------------------------------

    def device_uuid(self, i: int) -> str:
        self._check(i)
        return f"GPU-mi355x-mock-{i:02d}"

    def device_bdf(self, i: int) -> str:
        self._check(i)
        return f"0000:{0x0a + i:02x}:00.0"

    def asic_info(self, i: int) -> Dict[str, Any]:
        self._check(i)
        return {
            "market_name": MI355X_NAME,
            "vendor_id": 0x1002,
            "device_id": 0x75A0,
            "rev_id": 0,
            "asic_serial": f"0xMOCKSERIAL{i:02d}",
            "oam_id": i,
            "num_compute_units": MI355X_CUS,
            "target_graphics_version": 90500,
        }

    def board_info(self, i: int) -> Dict[str, Any]:
        self._check(i)
        return {
            "model_number": "102-G30212",
            "product_serial": f"MOCKBOARD{i:04d}",
            "fru_id": "",
            "product_name": "Instinct MI355X OAM",
            "manufacturer_name": "AMD",
        }

    def driver_info(self, i: int) -> Dict[str, Any]:
        self._check(i)
        return {
            "driver_version": "6.14.14",
            "driver_date": "2026/01/01 00:00",
            "driver_name": "amdgpu",
        }

    def vbios_info(self, i: int) -> Dict[str, Any]:
        self._check(i)
        return {
            "name": "MI355X VBIOS",
            "version": "022.040.003.042",
            "part_number": "113-MI355X-XL",
            "build_date": "2026/01/01",
        }

    def vram_info(self, i: int) -> Dict[str, Any]:
        self._check(i)
        return {
            "vram_type": 10,
            "vram_vendor": "hynix",
            "vram_size_bytes": MI355X_VRAM_MB * 1024 * 1024,
            "vram_bit_width": 8192,
            "vram_max_bandwidth": 8000,
        }

    # -- telemetry ----------------------------------------------------------

    def _check(self, i: int) -> Dict[str, Any]:
        if i < 0 or i >= self.num_gpus:
            raise IndexError("gpu index out of range")
        return self.state[i]

    def temp_metric(self, i: int, sensor: int, metric: int) -> int:
        s = self._check(i)
        key = {0: "edge", 1: "hotspot", 2: "vram"}.get(sensor, "edge")
        if metric == 0:
            return int(s[f"temp_{key}"])
        if metric == 5:  # critical
            return int(s[f"temp_{key}_limit"])
        if metric == 15:  # shutdown
            return int(s.get("temp_hotspot_shutdown", 115))
        return int(s[f"temp_{key}"])

    def power_info(self, i: int) -> Dict[str, Any]:
        s = self._check(i)
        return {
            "socket_power_w": s["power_w"],
            "current_socket_power_w": s["power_w"],
            "average_socket_power_w": s["power_w"],
            "gfx_voltage_mv": 750,
            "power_limit_w": s["power_limit_w"],
            "power_cap_uw": s["power_limit_w"] * 1_000_000,
        }

    def clock_info(self, i: int, clk_type: int) -> Dict[str, Any]:
        s = self._check(i)
        if clk_type == 4:  # MEM
            return {
                "clk_mhz": s["mem_mhz"],
                "min_clk_mhz": 900,
                "max_clk_mhz": s["mem_max_mhz"],
                "clk_locked": 0,
                "clk_deep_sleep": 0,
            }
        return {
            "clk_mhz": s["gfx_mhz"],
            "min_clk_mhz": 500,
            "max_clk_mhz": s["gfx_max_mhz"],
            "clk_locked": 0,
            "clk_deep_sleep": 0,
        }

    def activity(self, i: int) -> Dict[str, Any]:
        s = self._check(i)
        return {
            "gfx_activity_pct": s["gfx_activity"],
            "umc_activity_pct": s["umc_activity"],
            "mm_activity_pct": s["mm_activity"],
            # per-XCC (per-XCD) busy from xcp_stats: 8 XCDs on MI355X
            "xcc_busy_pct": [s["gfx_activity"]] * 8,
        }

    def vram_usage(self, i: int) -> Dict[str, Any]:
        s = self._check(i)
        return {"vram_total_mb": MI355X_VRAM_MB, "vram_used_mb": s["vram_used_mb"]}

    def ecc_count_total(self, i: int) -> Dict[str, Any]:
        s = self._check(i)
        return {
            "correctable": s["ecc_correctable"],
            "uncorrectable": s["ecc_uncorrectable"],
            "deferred": s["ecc_deferred"],
        }

    def ecc_count_block(self, i: int, block: int) -> Dict[str, Any]:
        return self.ecc_count_total(i)

    def bad_page_info(self, i: int) -> Dict[str, Any]:
        s = self._check(i)
        return {
            "total": s["bad_pages_total"],
            "reserved": max(0, s["bad_pages_total"] - s["bad_pages_pending"]),
            "pending": s["bad_pages_pending"],
            "unreservable": 0,
            "threshold": s["bad_page_threshold"],
        }

    def process_list(self, i: int) -> List[Dict[str, Any]]:
        return list(self._check(i)["processes"])

    def power_management_enabled(self, i: int) -> bool:
        self._check(i)
        return True

    def violation_status(self, i: int) -> Dict[str, Any]:
        s = self._check(i)
        base = {
            "acc_counter": 1000,
            "acc_prochot_thrm": 0,
            "acc_ppt_pwr": 0,
            "acc_socket_thrm": 0,
            "acc_vr_thrm": 0,
            "acc_hbm_thrm": 0,
            "per_prochot_thrm": 0,
            "per_ppt_pwr": 0,
            "per_socket_thrm": 0,
            "per_vr_thrm": 0,
            "per_hbm_thrm": 0,
            "active_prochot_thrm": 0,
            "active_ppt_pwr": 0,
            "active_socket_thrm": 0,
            "active_vr_thrm": 0,
            "active_hbm_thrm": 0,
        }
        base.update(s.get("throttle", {}))
        return base

    def xgmi_link_status(self, i: int) -> Dict[str, Any]:
        s = self._check(i)
        return {"total_links": len(s["xgmi_states"]), "states": list(s["xgmi_states"])}

    def xgmi_error_status(self, i: int) -> int:
        return int(self._check(i)["xgmi_error_status"])

    def xgmi_info(self, i: int) -> Dict[str, Any]:
        self._check(i)
        return {"xgmi_lanes": 16, "xgmi_hive_id": 1, "xgmi_node_id": i, "index": i}

    def link_metrics(self, i: int) -> Dict[str, Any]:
        s = self._check(i)
        links = []
        for li, st in enumerate(s["xgmi_states"]):
            links.append(
                {
                    "bit_rate": 32,
                    "max_bandwidth": 153,
                    "link_type": 2,  # XGMI
                    "read_kb": 1024 * (li + 1) if st == 1 else 0,
                    "write_kb": 1024 * (li + 1) if st == 1 else 0,
                    "bdf": f"0000:{0x0a + ((i + li + 1) % self.num_gpus):02x}:00.0",
                }
            )
        return {"num_links": len(links), "links": links}

    def pcie_info(self, i: int) -> Dict[str, Any]:
        s = self._check(i)
        return {
            "max_width": 16,
            "max_speed_gts": 32,
            "interface_version": 5,
            "width": s.get("pcie_width", 16),
            "speed_mts": s.get("pcie_speed_mts", 32000),
            "bandwidth_mbps": 512000,
            "replay_count": s.get("pcie_replay_count", 0),
            "l0_to_recovery_count": s.get("pcie_l0_to_recovery_count", 0),
            "replay_rollover_count": 0,
            "nak_sent_count": s.get("pcie_nak_sent_count", 0),
            "nak_received_count": s.get("pcie_nak_received_count", 0),
        }

    def partition_info(self, i: int) -> Dict[str, Any]:
        s = self._check(i)
        return {
            "compute_partition": s.get("compute_partition", "SPX"),
            "memory_partition": s.get("memory_partition", "NPS1"),
            "accelerator_profile_type": s.get("compute_partition", "SPX"),
            "num_partitions": 1,
            "partition_id": 0,
        }

    def cper_entries(
        self, i: int, severity_mask: int = 0xFFFFFFFF, cursor: int = 0, **_kw
    ) -> Dict[str, Any]:
        s = self._check(i)
        entries = [dict(e) for e in s.get("cper", [])[int(cursor):]]
        return {
            "supported": True,
            "entries": entries,
            "cursor": int(cursor) + len(entries),
        }

    def energy_count(self, i: int) -> Dict[str, Any]:
        self._check(i)
        return {
            "energy_accumulator": 123456789,
            "counter_resolution_uj": 15.259,
            "timestamp": 1_000_000,
        }

    def metrics_snapshot(self, i: int) -> Dict[str, Any]:
        s = self._check(i)
        viol = self.violation_status(i)
        return {
            "temperature": {
                "edge_c": s["temp_edge"],
                "hotspot_c": s["temp_hotspot"],
                "vram_c": s["temp_vram"],
                "edge_limit_c": s["temp_edge_limit"],
                "hotspot_limit_c": s["temp_hotspot_limit"],
                "vram_limit_c": s["temp_vram_limit"],
                "hotspot_shutdown_c": s["temp_hotspot_shutdown"],
            },
            "power": self.power_info(i),
            "clock": {
                "gfx_mhz": s["gfx_mhz"],
                "gfx_max_mhz": s["gfx_max_mhz"],
                "gfx_deep_sleep": 0,
                "mem_mhz": s["mem_mhz"],
                "mem_max_mhz": s["mem_max_mhz"],
            },
            "activity": self.activity(i),
            "vram": self.vram_usage(i),
            "ecc": self.ecc_count_total(i),
            "violation": viol,
            "xgmi_link_status": self.xgmi_link_status(i),
            "xgmi_error_status": self.xgmi_error_status(i),
            "bad_pages": self.bad_page_info(i),
            "gpu_metrics": {
                "throttle_status": 0,
                "indep_throttle_status": 0,
                "current_gfxclk_mhz": s["gfx_mhz"],
                "current_uclk_mhz": s["mem_mhz"],
                "average_socket_power_w": s["power_w"],
            },
        }

    def metrics_snapshot_all(self) -> List[Dict[str, Any]]:
        return [self.metrics_snapshot(i) for i in range(self.num_gpus)]

    # enum constants mirrored from _amdsmi
    TEMP_EDGE = 0
    TEMP_HOTSPOT = 1
    TEMP_VRAM = 2
    TEMP_CURRENT = 0
    TEMP_CRITICAL = 5
    TEMP_SHUTDOWN = 15
    CLK_GFX = 0
    CLK_MEM = 4
    XGMI_LINK_DOWN = 0
    XGMI_LINK_UP = 1
    XGMI_LINK_DISABLE = 2
    XGMI_STATUS_NO_ERRORS = 0
