"""Ordered registration list of all built-in components.

Reference: components/all/all.go:52-85 — the canonical component order:
accelerator components first, then host components. Diag components
(manual run mode) are appended when their native extensions are present.
"""

from __future__ import annotations

from typing import Callable, List

from .base import Component, GPUdInstance

from .accelerator import (
    bad_envs,
    bad_pages,
    clock_speed,
    cper,
    ecc,
    error_ras,
    gpm,
    gpu_counts,
    gpu_memory,
    partition,
    pcie,
    peer_mem,
    power,
    power_management,
    processes,
    rccl,
    temperature,
    throttle,
    utilization,
    xgmi,
)
from .host import (
    containerd,
    infiniband,
    cpu,
    disk,
    docker,
    fuse,
    kernel_module,
    library,
    memory,
    network_latency,
    nfs,
    os_component,
    pci,
    tailscale,
)

InitFunc = Callable[[GPUdInstance], Component]


def all_init_funcs() -> List[InitFunc]:
    funcs: List[InitFunc] = [
        # accelerator (reference order: all.go:53-71)
        clock_speed.new,
        ecc.new,
        gpm.new,
        gpu_counts.new,
        throttle.new,
        gpu_memory.new,
        rccl.new,
        xgmi.new,
        peer_mem.new,
        power.new,
        power_management.new,
        processes.new,
        bad_pages.new,
        bad_envs.new,
        partition.new,
        pcie.new,
        cper.new,
        temperature.new,
        utilization.new,
        error_ras.new,
        # host (reference order: all.go:72-84)
        containerd.new,
        cpu.new,
        disk.new,
        docker.new,
        fuse.new,
        infiniband.new,
        kernel_module.new,
        library.new,
        memory.new,
        network_latency.new,
        nfs.new,
        os_component.new,
        pci.new,
        tailscale.new,
    ]
    # active diagnostics (manual run mode) — present once csrc/diag is built
    try:
        from .accelerator import diag  # noqa: WPS433

        funcs.extend(diag.init_funcs())
    except ImportError:
        pass
    return funcs
