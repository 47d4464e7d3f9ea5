"""accelerator-amd-peer-mem — GPUDirect-RDMA-style peer memory capability.

Reference: components/accelerator/nvidia/peermem (nvidia_peermem module
check + kmsg matcher — peermem/component.go:25). The ROCm equivalents for
RDMA-into-GPU-memory are (a) the legacy PeerDirect registration
(/sys/kernel/mm/memory_peers/<name>/version, provided by amdp2p or the
ib_peer_mem patches) or (b) the modern in-kernel DMABUF path, which needs
no extra module and is in use whenever the amdgpu driver and an RDMA stack
coexist on a recent kernel. The check is only meaningful (and only
degrades) when RDMA NICs are present.
"""

from __future__ import annotations

import os
from typing import Callable

from ...apiv1.types import HealthStateType
from ..base import CheckResult, Component, GPUdInstance, TickerComponent
from .shared import SmiComponentMixin

NAME = "accelerator-amd-peer-mem"

MEMORY_PEERS_DIR = "/sys/kernel/mm/memory_peers"
INFINIBAND_DIR = "/sys/class/infiniband"


def peer_providers(root: str = MEMORY_PEERS_DIR) -> list:
    try:
        return sorted(os.listdir(root))
    except OSError:
        return []


def kernel_has_dmabuf_rdma() -> bool:
    """The modern peer path: kernel >= 5.12 ships dma-buf RDMA support;
    presence of the dma_heap/dmabuf sysfs knobs is the practical signal."""
    return os.path.isdir("/sys/kernel/dmabuf") or os.path.exists(
        "/sys/kernel/debug/dma_buf"
    ) or _kernel_at_least(5, 12)


def _kernel_at_least(major: int, minor: int) -> bool:
    try:
        rel = os.uname().release.split("-")[0].split(".")
        return (int(rel[0]), int(rel[1])) >= (major, minor)
    except (ValueError, IndexError):
        return False


class PeerMemComponent(TickerComponent, SmiComponentMixin):
    def __init__(self, inst: GPUdInstance):
        super().__init__()
        self._smi = inst.smi
        self.get_providers: Callable = peer_providers
        self.has_rdma_nics: Callable = lambda: os.path.isdir(INFINIBAND_DIR)
        self.has_dmabuf: Callable = kernel_has_dmabuf_rdma

    @property
    def name(self) -> str:
        return NAME

    def tags(self) -> list:
        return ["accelerator", "amd", "gpu", NAME]

    def is_supported(self) -> bool:
        # meaningful only where GPU + RDMA NICs coexist
        return (
            self._smi is not None
            and self._smi.exists
            and self.has_rdma_nics()
        )

    def check(self) -> CheckResult:
        guard = self.smi_guard()
        if guard is not None:
            return guard
        if not self.has_rdma_nics():
            return CheckResult(
                NAME, reason="no RDMA NICs present; peer memory not applicable"
            )
        providers = self.get_providers()
        if providers:
            return CheckResult(
                NAME,
                reason="PeerDirect providers registered: " + ", ".join(providers),
                extra_info={"providers": ",".join(providers)},
            )
        if self.has_dmabuf():
            return CheckResult(
                NAME,
                reason="no PeerDirect provider; kernel DMABUF RDMA path available",
            )
        return CheckResult(
            NAME,
            health=HealthStateType.DEGRADED,
            reason="RDMA NICs present but neither PeerDirect providers nor "
            "a DMABUF-capable kernel found — GPUDirect-style transfers will "
            "fall back to host bounce buffers",
        )


def new(inst: GPUdInstance) -> Component:
    return PeerMemComponent(inst)
