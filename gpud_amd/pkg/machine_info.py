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


def _cpuinfo_field(key: str) -> str:
    try:
        with open("/proc/cpuinfo") as f:
            for line in f:
                if line.startswith(key):
                    return line.split(":", 1)[1].strip()
    except OSError:
        pass
    return ""


def _cpu_model() -> str:
    return _cpuinfo_field("model name") or platform.processor()


def _cpu_vendor() -> str:
    """CPU vendor id (reference: pkghost.CPUVendorID — e.g. AuthenticAMD)."""
    return _cpuinfo_field("vendor_id") or platform.machine()


# virtual/overlay interfaces excluded from the private-IP inventory
# (reference: machine_info.go:250-266 — lo/cali/cni/docker/flannel/
# nodelocaldns/tailscale/tunl/veth/vxlan/ib prefixes, .calico suffix)
NIC_PREFIXES_TO_SKIP = (
    "lo", "cali", "cni", "docker", "flannel", "nodelocaldns", "tailscale",
    "tunl", "veth", "vxlan", "ib",
)
NIC_SUFFIXES_TO_SKIP = (".calico",)

_PRIVATE_NETS = (
    ("10.", ""),
    ("192.168.", ""),
) + tuple((f"172.{i}.", "") for i in range(16, 32))


def _is_private_ipv4(addr: str) -> bool:
    return any(addr.startswith(p) for p, _ in _PRIVATE_NETS)


def _nic_info() -> MachineNICInfo:
    """Private-IP interface inventory with MACs, virtual interfaces
    filtered, sorted by IP (reference: GetMachineNICInfo)."""
    ifaces = []
    try:
        for name, addrs in psutil.net_if_addrs().items():
            if any(name.startswith(p) for p in NIC_PREFIXES_TO_SKIP):
                continue
            if any(name.endswith(s) for s in NIC_SUFFIXES_TO_SKIP):
                continue
            mac = ""
            for a in addrs:
                if a.family == psutil.AF_LINK:
                    mac = a.address
            for a in addrs:
                if a.family == socket.AF_INET and _is_private_ipv4(a.address):
                    ifaces.append(
                        {"interface": name, "mac": mac, "ip": a.address}
                    )
    except Exception:
        pass
    ifaces.sort(key=lambda i: i["ip"])
    return MachineNICInfo(private_ip_interfaces=ifaces)


def _disk_info() -> MachineDiskInfo:
    """Mounted block devices with identity columns (reference:
    GetMachineDiskInfo — lsblk flatten, mounted devices only, name/type/
    size/used/rota/serial/wwn/vendor/model/rev/mountpoint/fstype/pkname)."""
    from ..components.host.disk import list_block_devices

    devices = list_block_devices() or []
    mounted = []
    for d in devices:
        if not d.get("mountpoint"):
            continue
        mounted.append(
            {
                "name": d.get("name", ""),
                "type": d.get("type", ""),
                "size": int(d.get("size") or 0),
                "used": int(d.get("fsused") or 0),
                "rota": bool(d.get("rota")),
                "serial": d.get("serial") or "",
                "wwn": d.get("wwn") or "",
                "vendor": (d.get("vendor") or "").strip(),
                "model": (d.get("model") or "").strip(),
                "rev": (d.get("rev") or "").strip(),
                "mount_point": d.get("mountpoint", ""),
                "fstype": d.get("fstype") or "",
                "parent_device": d.get("pkname") or "",
            }
        )
    return MachineDiskInfo(block_devices=mounted)


def _container_runtime_version() -> str:
    """containerd CRI version as 'containerd://<version>' (reference:
    machine_info.go:118-127)."""
    from ..components.host.containerd import DEFAULT_SOCKET, cri_version

    try:
        cri = cri_version(DEFAULT_SOCKET, timeout=5.0)
    except Exception:
        cri = None
    if cri and cri.get("runtime_name"):
        return f"{cri['runtime_name']}://{cri['runtime_version']}"
    return ""


def _tailscale_version() -> str:
    import subprocess

    try:
        out = subprocess.run(
            ["tailscale", "version"], capture_output=True, text=True, timeout=5
        )
        if out.returncode == 0 and out.stdout.strip():
            return out.stdout.strip().splitlines()[0]
    except (OSError, subprocess.TimeoutExpired):
        pass
    return ""


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
            manufacturer=_cpu_vendor(),
            architecture=platform.machine(),
            logical_cores=psutil.cpu_count() or 0,
        ),
        memory_info=MachineMemoryInfo(total_bytes=vm.total),
        nic_info=_nic_info(),
        disk_info=_disk_info(),
        container_runtime_version=_container_runtime_version(),
        tailscale_version=_tailscale_version(),
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


def detect_provider_info(detectors=None):
    """Full provider Info (provider/public_ip/private_ip/region/
    vm_environment/instance_id) via the per-cloud IMDS detectors
    (pkg/providers.py — reference: pkg/providers/detect.go)."""
    from . import providers as pkgproviders

    return pkgproviders.detect(detectors)


def detect_provider(timeout: float = 1.0) -> Optional[str]:
    from . import providers as pkgproviders

    get = lambda url, headers=None, t=timeout: pkgproviders._httpx_get(  # noqa: E731
        url, headers, t
    )
    put = lambda url, headers=None, t=timeout: pkgproviders._httpx_put(  # noqa: E731
        url, headers, t
    )
    info = pkgproviders.detect(pkgproviders.default_detectors(get, put))
    return info.provider if info is not None else None


def detect_provider_by_asn(public_ip: str) -> Optional[str]:
    """ASN-based provider labeling for nodes without IMDS (reference:
    machine_info.go:339 — AS org of the public IP, normalized). Needs
    egress; returns None air-gapped."""
    from . import asn

    res = asn.get_as_lookup(public_ip, sleep=lambda _s: None)
    if res is None or not res.asn_name:
        return None
    return asn.normalize_asn_name(res.asn_name)
