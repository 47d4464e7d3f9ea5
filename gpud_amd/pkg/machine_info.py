"""Machine inventory (reference: pkg/machine-info/machine_info.go:73-540).

CPU/memory/NIC/disk/GPU inventory assembled from procfs, psutil and the
shared SMI instance; cloud-provider detection via IMDS endpoints is
best-effort with short timeouts (air-gapped clusters simply report none).
The reference's ``cudaVersion`` slot carries the ROCm version for
control-plane wire compatibility (apiv1.types.MachineInfo).
"""

from __future__ import annotations

import platform
import socket
from typing import Any, Optional

import psutil

from .. import __version__
from ..apiv1.types import (
    MachineCPUInfo,
    MachineDiskInfo,
    MachineGPUInfo,
    MachineGPUInstance,
    MachineInfo,
    MachineMemoryInfo,
    MachineNICInfo,
)
from . import host as pkghost


def _cpu_model() -> str:
    try:
        with open("/proc/cpuinfo") as f:
            for line in f:
                if line.startswith("model name"):
                    return line.split(":", 1)[1].strip()
    except OSError:
        pass
    return platform.processor()


def _nic_info() -> MachineNICInfo:
    ifaces = []
    try:
        for name, addrs in psutil.net_if_addrs().items():
            if name == "lo":
                continue
            for a in addrs:
                if a.family == socket.AF_INET:
                    ifaces.append({"interface": name, "ip": a.address})
    except Exception:
        pass
    return MachineNICInfo(private_ip_interfaces=ifaces)


def _disk_info() -> MachineDiskInfo:
    from ..components.host.disk import list_block_devices

    devices = list_block_devices() or []
    return MachineDiskInfo(block_devices=devices)


def get_machine_info(smi_instance: Any = None) -> MachineInfo:
    vm = psutil.virtual_memory()
    info = MachineInfo(
        gpud_version=__version__,
        kernel_version=pkghost.kernel_version(),
        os_image=pkghost.os_image(),
        operating_system=platform.system().lower(),
        system_uuid=pkghost.system_uuid(),
        machine_id=pkghost.machine_id(),
        boot_id=pkghost.boot_id(),
        hostname=pkghost.hostname(),
        uptime=pkghost.boot_time(),
        cpu_info=MachineCPUInfo(
            type=_cpu_model(),
            manufacturer=platform.machine(),
            architecture=platform.machine(),
            logical_cores=psutil.cpu_count() or 0,
        ),
        memory_info=MachineMemoryInfo(total_bytes=vm.total),
        nic_info=_nic_info(),
        disk_info=_disk_info(),
    )
    if smi_instance is not None and getattr(smi_instance, "exists", False):
        try:
            info.gpu_driver_version = smi_instance.driver_version
            info.cuda_version = smi_instance.rocm_version  # wire-compat slot
            gpus = []
            total_vram_mb = 0
            for uuid, dev in smi_instance.devices().items():
                try:
                    vu = dev.vram_usage()
                    total_vram_mb = max(total_vram_mb, vu.get("vram_total_mb", 0))
                except Exception:
                    pass
                gpus.append(
                    MachineGPUInstance(
                        uuid=uuid,
                        sn=dev.board_serial or dev.asic_serial,
                        product=dev.product_name,
                        board_id=dev.oam_id,
                    )
                )
            partition = ""
            try:
                devs = smi_instance.devices()
                if devs:
                    pi = next(iter(devs.values())).partition_info()
                    comp = pi.get("compute_partition", "")
                    mem = pi.get("memory_partition", "")
                    if comp or mem:
                        partition = f"{comp or '?'}/{mem or '?'}"
            except Exception:
                pass
            info.gpu_info = MachineGPUInfo(
                product=smi_instance.product_name,
                manufacturer="AMD",
                architecture=("gfx950" + (f" ({partition})" if partition else "")),
                driver_version=smi_instance.driver_version,
                rocm_version=smi_instance.rocm_version,
                memory=f"{total_vram_mb} MB" if total_vram_mb else "",
                gpus=gpus,
            )
        except Exception:
            pass
    return info


# -- cloud provider detection (reference: pkg/providers/*/imds) -------------

IMDS_PROBES = [
    ("aws", "http://169.254.169.254/latest/meta-data/instance-id", {}),
    (
        "gcp",
        "http://metadata.google.internal/computeMetadata/v1/instance/id",
        {"Metadata-Flavor": "Google"},
    ),
    (
        "azure",
        "http://169.254.169.254/metadata/instance?api-version=2021-02-01",
        {"Metadata": "true"},
    ),
]


def detect_provider(timeout: float = 1.0) -> Optional[str]:
    import httpx

    for name, url, headers in IMDS_PROBES:
        try:
            r = httpx.get(url, headers=headers, timeout=timeout)
            if r.status_code == 200:
                return name
        except Exception:
            continue
    return None


def detect_provider_by_asn(public_ip: str) -> Optional[str]:
    """ASN-based provider labeling for nodes without IMDS (reference:
    machine_info.go:339 — AS org of the public IP, normalized). Needs
    egress; returns None air-gapped."""
    from . import asn

    res = asn.get_as_lookup(public_ip, sleep=lambda _s: None)
    if res is None or not res.asn_name:
        return None
    return asn.normalize_asn_name(res.asn_name)
