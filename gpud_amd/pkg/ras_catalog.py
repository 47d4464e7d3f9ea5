"""Curated amdgpu / RAS kernel-message catalog for MI355X.

The AMD-native equivalent of the reference's NVRM Xid table
(reference: components/accelerator/nvidia/xid/xid.go Detail catalog and
xid/kmsg.go regex parsing): a set of dmesg signatures emitted by the amdgpu
driver, the KFD, and the kernel RAS machinery, each mapped to an event name,
a severity, and suggested repair actions. There is no 1:1 numeric code table
on AMD — these are message-shape signatures curated from the amdgpu kernel
driver (drivers/gpu/drm/amd/) message set.

Also carries the injectable-message catalog used by the fault injector
(reference: xid/kmsg.go:270 GetMessageToInject).
"""

from __future__ import annotations

import re
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Pattern, Tuple

from ..apiv1.types import EventType, RepairActionType, SuggestedActions


@dataclass
class Detail:
    """One catalog entry: a recognised amdgpu/RAS kernel message class."""

    name: str
    pattern: Pattern
    description: str
    event_type: str
    repair_actions: List[str] = field(default_factory=list)
    # whether an occurrence marks the GPU unhealthy (drives the health state
    # machine, reference: xid/health_state.go)
    critical: bool = False

    def suggested_actions(self) -> Optional[SuggestedActions]:
        if not self.repair_actions:
            return None
        return SuggestedActions(
            description=self.description, repair_actions=list(self.repair_actions)
        )


def _d(
    name: str,
    regex: str,
    description: str,
    event_type: str,
    repair: Optional[List[str]] = None,
    critical: bool = False,
) -> Detail:
    return Detail(
        name=name,
        pattern=re.compile(regex),
        description=description,
        event_type=event_type,
        repair_actions=repair or [],
        critical=critical,
    )


_REBOOT = [RepairActionType.REBOOT_SYSTEM]
_HW = [RepairActionType.HARDWARE_INSPECTION]
_APP = [RepairActionType.CHECK_USER_APP_AND_GPU]
_IGNORE = [RepairActionType.IGNORE_NO_ACTION_REQUIRED]

# Order matters: first match wins; put the most specific signatures first.
CATALOG: List[Detail] = [
    # ---- GPU reset lifecycle ---------------------------------------------
    _d(
        "amdgpu_gpu_reset_failed",
        r"amdgpu.*GPU reset\(\d+\) failed",
        "amdgpu GPU reset failed — the device could not be recovered by the "
        "driver; the node needs a reboot and, if it recurs, hardware service",
        EventType.FATAL,
        _REBOOT + _HW,
        critical=True,
    ),
    _d(
        "amdgpu_gpu_reset_begin",
        r"amdgpu.*(GPU reset begin|GPU recovery (?:begin|disabled))",
        "amdgpu initiated a GPU reset — a preceding fault (hang, RAS "
        "uncorrectable error, page-fault storm) forced device recovery; "
        "running workloads on this GPU were lost",
        EventType.CRITICAL,
        _APP,
        critical=True,
    ),
    _d(
        "amdgpu_gpu_reset_succeeded",
        r"amdgpu.*GPU reset(\(\d+\))? succeeded",
        "amdgpu GPU reset completed successfully; workloads must be restarted",
        EventType.WARNING,
        _IGNORE,
    ),
    _d(
        "amdgpu_mode2_reset",
        r"amdgpu.*mode2 reset",
        "amdgpu performed a MODE2 (per-engine) reset",
        EventType.WARNING,
        _IGNORE,
    ),
    # ---- hangs / timeouts -------------------------------------------------
    _d(
        "amdgpu_ring_timeout",
        # drm-log form: "[drm:amdgpu_job_timedout [amdgpu]] *ERROR* ring X
        # timeout"; newer kernels use dev_err: "amdgpu 0000:..: ring X
        # timeout, signaled seq=.." — accept both
        r"amdgpu.*?(?:\*ERROR\* )?ring (?P<ring>\S+) timeout",
        "amdgpu command-ring timeout — a GPU job exceeded the scheduler "
        "timeout; usually an application-level hang (infinite kernel, "
        "deadlocked wave), occasionally a hardware fault if recurring",
        EventType.CRITICAL,
        _APP,
        critical=True,
    ),
    _d(
        "amdgpu_job_timeout",
        r"amdgpu_job_timedout",
        "amdgpu job scheduler timeout — GPU work did not complete in time",
        EventType.CRITICAL,
        _APP,
        critical=True,
    ),
    _d(
        "amdgpu_soft_recovery",
        r"amdgpu.*soft recovery succeeded",
        "amdgpu soft recovery (wave kill) succeeded without a full reset",
        EventType.WARNING,
        _IGNORE,
    ),
    # ---- page faults / memory violations ---------------------------------
    _d(
        "amdgpu_page_fault",
        r"amdgpu.*\[(?P<hub>gfxhub|mmhub)[^\]]*\].*(page fault|no-retry page fault)"
        r".*(?:pasid[: ]+(?P<pasid>\d+))?",
        "GPU VM page fault — a shader accessed an unmapped or protected "
        "address; almost always an application bug (out-of-bounds access, "
        "use-after-free of GPU memory)",
        EventType.CRITICAL,
        _APP,
        critical=True,
    ),
    _d(
        "amdgpu_vm_fault",
        r"amdgpu.*VM_L2_PROTECTION_FAULT_STATUS",
        "GPU VM L2 protection fault detail record",
        EventType.WARNING,
        _APP,
    ),
    # ---- RAS: ECC / poison ------------------------------------------------
    _d(
        "amdgpu_ras_uncorrectable",
        r"amdgpu.*(uncorrectable hardware error|ERREVENT_ATHUB_INTERRUPT"
        r"|[Uu]ncorrectable error detected)",
        "Uncorrectable hardware (RAS) error — the GPU detected a UE; the "
        "driver will typically reset the device; recurring UEs indicate "
        "failing HBM or logic and need hardware inspection",
        EventType.FATAL,
        _REBOOT + _HW,
        critical=True,
    ),
    _d(
        "amdgpu_ras_poison_consumption",
        r"amdgpu.*poison (?:consumption|is consumed)",
        "RAS poison consumed — a compute unit read poisoned (UE-marked) "
        "memory; the consuming process is killed; check HBM health",
        EventType.CRITICAL,
        _APP + _HW,
        critical=True,
    ),
    _d(
        "amdgpu_ras_poison_creation",
        r"amdgpu.*poison (?:creation|is created)",
        "RAS poison created — an uncorrectable memory error was contained "
        "by poisoning the affected page",
        EventType.WARNING,
        _IGNORE,
    ),
    _d(
        "amdgpu_ras_corrected_error",
        r"amdgpu.*(correctable hardware error|corrected error|"
        r"\d+ correctable (?:hardware )?errors? detected)",
        "Correctable hardware (RAS) error — corrected by ECC; elevated "
        "rates forecast uncorrectable errors",
        EventType.WARNING,
        _IGNORE,
    ),
    _d(
        "amdgpu_ras_bad_page",
        r"amdgpu.*(bad page|reserve memory for bad page|umc bad page)",
        "RAS retired (bad) HBM page recorded — the page is removed from "
        "the usable pool; many retirements indicate degrading HBM",
        EventType.WARNING,
        _HW,
    ),
    _d(
        "amdgpu_ras_bad_page_threshold",
        r"amdgpu.*(bad page threshold|exceed(?:s|ed)? threshold)",
        "RAS retired-page count reached the saved-page threshold — the GPU "
        "may refuse initialization; hardware service required",
        EventType.FATAL,
        _HW,
        critical=True,
    ),
    _d(
        "amdgpu_ras_eeprom",
        r"amdgpu.*RAS EEPROM",
        "RAS EEPROM (bad-page table) access message",
        EventType.WARNING,
        _IGNORE,
    ),
    _d(
        "amdgpu_ras_event",
        r"amdgpu.*RAS.*(error|event).*detected",
        "Generic RAS error event detected by amdgpu",
        EventType.WARNING,
        _IGNORE,
    ),
    # ---- SMU / power management ------------------------------------------
    _d(
        "amdgpu_smu_error",
        r"amdgpu.*SMU.*(?:failed|error|timed? ?out|not done with your previous command)",
        "SMU (System Management Unit) command failure — firmware power "
        "management did not respond; thermal/power telemetry may be stale",
        EventType.CRITICAL,
        _REBOOT,
        critical=True,
    ),
    _d(
        "amdgpu_thermal_shutdown",
        r"amdgpu.*(thermal.*shutdown|emergency.*thermal)",
        "GPU thermal emergency — the device shut down or throttled hard to "
        "protect itself; check cooling",
        EventType.FATAL,
        _HW,
        critical=True,
    ),
    # ---- KFD / compute stack ---------------------------------------------
    _d(
        "kfd_evict_failed",
        r"kfd.*Failed to evict process queues",
        "KFD could not evict process queues — compute preemption failure; "
        "often precedes a GPU reset",
        EventType.CRITICAL,
        _APP,
        critical=True,
    ),
    _d(
        "kfd_hws_hang",
        r"kfd.*(HWS hang|hqd slot|CP hang|unmap queue failed)",
        "KFD hardware scheduler problem (CP/HWS hang)",
        EventType.CRITICAL,
        _REBOOT,
        critical=True,
    ),
    _d(
        "kfd_queue_preemption_failed",
        r"kfd.*queue preemption failed",
        "KFD queue preemption failure",
        EventType.CRITICAL,
        _APP,
        critical=True,
    ),
    # ---- xGMI -------------------------------------------------------------
    _d(
        "amdgpu_xgmi_error",
        r"amdgpu.*[Xx]GMI.*(error|failed|down)",
        "xGMI inter-GPU link error — peer-to-peer fabric degraded; RCCL "
        "collectives will slow down or fail",
        EventType.CRITICAL,
        _REBOOT + _HW,
        critical=True,
    ),
    # ---- PCIe / bus -------------------------------------------------------
    _d(
        "pcie_aer_fatal",
        r"(?:AER|pcieport).*(?:Uncorrected|Fatal) \(?(?:Fatal|Uncorrected)?\)?.*error",
        "PCIe fatal/uncorrected AER error on the GPU link — the device may "
        "drop off the bus; typically needs a reboot and slot inspection",
        EventType.FATAL,
        _REBOOT + _HW,
        critical=True,
    ),
    _d(
        "amdgpu_fallen_off_bus",
        r"amdgpu.*(?:GPU (?:has )?fallen off the bus|failed to read from the bus|"
        r"device (?:is )?(?:gone|lost))",
        "GPU no longer responds on the PCIe bus — hardware or power fault",
        EventType.FATAL,
        _REBOOT + _HW,
        critical=True,
    ),
    # ---- init failures ----------------------------------------------------
    _d(
        "amdgpu_init_failed",
        r"amdgpu.*(Fatal error during GPU init|amdgpu_device_ip_init failed|"
        r"amdgpu_init failed)",
        "amdgpu driver failed to initialize the GPU",
        EventType.FATAL,
        _REBOOT + _HW,
        critical=True,
    ),
    # ---- RCCL crash signature (reference: nccl/kmsg_matcher.go:12) --------
    _d(
        "amd_rccl_segfault_in_librccl",
        r"segfault at .* in librccl\.so",
        "A process crashed inside librccl (RCCL collective library) — "
        "usually an application/library issue, occasionally fabric trouble",
        EventType.WARNING,
        _APP,
    ),
    # ---- GPU self-test failures ------------------------------------------
    _d(
        "amdgpu_ib_test_failed",
        r"amdgpu.*\*ERROR\* (?:IB test failed|ring test failed) on (?P<ring>\S+)",
        "amdgpu indirect-buffer/ring self-test failed — the engine did not "
        "execute a trivial command buffer; driver or hardware fault",
        EventType.CRITICAL,
        _REBOOT + _HW,
        critical=True,
    ),
    _d(
        "amdgpu_firmware_load_failed",
        r"amdgpu.*(?:failed to load firmware|Failed to load gpu firmware|"
        r"Direct firmware load for amdgpu.* failed)",
        "amdgpu firmware load failure — ROCm/driver installation problem",
        EventType.CRITICAL,
        _REBOOT,
        critical=True,
    ),
    # ---- host memory / machine-check (reference: memory component kmsg) ---
    _d(
        "memory_edac_uncorrectable",
        r"EDAC .*\bUE\b|EDAC MC\d+: \d+ UE",
        "Uncorrectable host (DIMM) memory error reported by EDAC",
        EventType.FATAL,
        _HW,
        critical=True,
    ),
    _d(
        "memory_edac_correctable",
        r"EDAC .*\bCE\b|EDAC MC\d+: \d+ CE",
        "Corrected host (DIMM) memory error reported by EDAC",
        EventType.WARNING,
        _IGNORE,
    ),
    _d(
        "host_mce",
        r"mce: \[Hardware Error\]",
        "Host machine-check (MCE) hardware error logged",
        EventType.CRITICAL,
        _HW,
    ),
    # ---- OOM (host memory pressure killing GPU jobs) -----------------------
    _d(
        "memory_oom_kill",
        r"Out of memory: Killed process (?P<pid>\d+)",
        "The kernel OOM-killer terminated a process",
        EventType.WARNING,
        _APP,
    ),
    _d(
        "memory_oom_cgroup",
        r"Memory cgroup out of memory: Killed process (?P<pid>\d+)",
        "A cgroup memory limit killed a process",
        EventType.WARNING,
        _APP,
    ),
    # ---- additional amdgpu / platform signatures -------------------------
    _d(
        "amdgpu_vram_lost",
        r"amdgpu.*VRAM is lost due to GPU reset",
        "VRAM contents were lost across a GPU reset — every context running "
        "on the device lost its data; jobs must be restarted",
        EventType.CRITICAL,
        _APP,
        critical=True,
    ),
    _d(
        "amdgpu_aca_error",
        r"amdgpu.*(?:ACA|aca).*(?:error|bank)|Accelerator Check Architecture",
        "ACA (Accelerator Check Architecture) error bank report — the "
        "MI3xx-generation RAS telemetry block logged a hardware error",
        EventType.WARNING,
        _HW,
    ),
    _d(
        "amdgpu_kiq_timeout",
        r"amdgpu.*(?:KIQ|kiq).*(?:timeout|failed)",
        "KIQ (kernel interface queue) register access timeout — the command "
        "processor stopped servicing privileged requests; usually precedes "
        "a ring timeout and GPU reset",
        EventType.CRITICAL,
        _REBOOT,
        critical=True,
    ),
    _d(
        "amdgpu_psp_cmd_failed",
        r"amdgpu.*(?:PSP|psp).*(?:command|cmd|load).*fail",
        "PSP (platform security processor) command failure — firmware "
        "loading or security setup failed",
        EventType.CRITICAL,
        _REBOOT + _HW,
        critical=True,
    ),
    _d(
        "amdgpu_mes_error",
        r"amdgpu.*MES.*(?:failed|hang|timeout)",
        "MES (micro-engine scheduler) failure",
        EventType.CRITICAL,
        _REBOOT,
        critical=True,
    ),
    _d(
        "amdgpu_fence_fallback",
        r"amdgpu.*[Ff]ence fallback timer expired",
        "Fence interrupt was lost and the fallback timer fired — benign "
        "when rare; frequent occurrences indicate interrupt delivery "
        "problems",
        EventType.WARNING,
        _IGNORE,
    ),
    _d(
        "pcie_aer_corrected",
        r"AER:.*[Cc]orrected error (?:received|message)",
        "PCIe AER corrected error — recovered by hardware; track the rate",
        EventType.INFO,
        _IGNORE,
    ),
    _d(
        "pcie_bandwidth_limited",
        r"available PCIe bandwidth, limited by",
        "Device trained at lower PCIe speed/width than the platform "
        "supports — check slot seating and BIOS lane configuration",
        EventType.WARNING,
        _HW,
    ),
    _d(
        "host_hung_task",
        r"INFO: task .* blocked for more than \d+ seconds",
        "Kernel hung-task watchdog: a task sat in uninterruptible sleep — "
        "often storage or driver stalls; correlates with the os component's "
        "D-state tracker",
        EventType.WARNING,
        _APP,
    ),
    _d(
        "amdgpu_ip_resume_failed",
        r"amdgpu.*resume of IP block <[^>]+> failed",
        "An IP block failed to resume after reset/suspend — the device is "
        "in a partial state; reboot to reinitialize",
        EventType.CRITICAL,
        _REBOOT,
        critical=True,
    ),
    _d(
        "amdgpu_ip_suspend_failed",
        r"amdgpu.*suspend of IP block <[^>]+> failed",
        "An IP block failed to suspend cleanly — the following resume is "
        "suspect",
        EventType.WARNING,
        _REBOOT,
    ),
    _d(
        "kfd_process_vm_failed",
        r"(?:kfd|amdgpu).*Failed to create process VM",
        "KFD could not create a process VM — compute process start "
        "failure; often follows earlier VM faults or exhausted resources",
        EventType.WARNING,
        _APP,
    ),
    _d(
        "host_swiotlb_full",
        r"swiotlb buffer is full",
        "SWIOTLB bounce-buffer exhaustion — DMA is being bounced (IOMMU/"
        "mapping config); expect severe transfer slowdowns",
        EventType.WARNING,
        _APP,
    ),
    _d(
        "host_soft_lockup",
        r"BUG: soft lockup - CPU#\d+ stuck",
        "CPU soft lockup — a kernel thread monopolized a CPU; node health "
        "is suspect",
        EventType.CRITICAL,
        _REBOOT,
        critical=True,
    ),
]

_CATALOG_BY_NAME: Dict[str, Detail] = {d.name: d for d in CATALOG}
# first-match-wins order preserved within the filtered view
_NON_AMDGPU_CATALOG: List[Detail] = [
    d for d in CATALOG if not d.name.startswith("amdgpu_")
]


def lookup(name: str) -> Optional[Detail]:
    return _CATALOG_BY_NAME.get(name)


def match(line: str) -> Optional[Tuple[Detail, Dict[str, str]]]:
    """Match one kernel-message line; returns (detail, captured groups).

    The ``amdgpu_*`` signatures all anchor on the literal driver name, so
    lines without it walk a pre-split sub-catalog that skips those ~30
    regexes entirely — full-ring replay on boot stays cheap for the
    non-amdgpu traffic that dominates a mixed dmesg."""
    cat = CATALOG if "amdgpu" in line else _NON_AMDGPU_CATALOG
    for d in cat:
        m = d.pattern.search(line)
        if m:
            groups = {k: v for k, v in m.groupdict().items() if v}
            return d, groups
    return None


# ---------------------------------------------------------------------------
# Injectable messages (fault injection; reference: xid/kmsg.go:270)
# ---------------------------------------------------------------------------

INJECTABLE: Dict[str, str] = {
    "amdgpu_vram_lost": (
        "amdgpu 0000:0a:00.0: amdgpu: VRAM is lost due to GPU reset!"
    ),
    "amdgpu_kiq_timeout": (
        "amdgpu 0000:0a:00.0: amdgpu: KIQ reg write timeout (0x1f2c)"
    ),
    "host_soft_lockup": (
        "BUG: soft lockup - CPU#12 stuck for 23s! [kworker/12:1:12345]"
    ),
    "amdgpu_ring_timeout": (
        "[drm:amdgpu_job_timedout [amdgpu]] *ERROR* ring gfx_0.0.0 timeout, "
        "signaled seq=1234, emitted seq=1236"
    ),
    "amdgpu_page_fault": (
        "amdgpu 0000:0a:00.0: amdgpu: [gfxhub] no-retry page fault "
        "(src_id:0 ring:24 vmid:3 pasid:32770)"
    ),
    "amdgpu_ras_uncorrectable": (
        "amdgpu 0000:0a:00.0: amdgpu: uncorrectable hardware error"
        "(ERREVENT_ATHUB_INTERRUPT) detected!"
    ),
    "amdgpu_ras_corrected_error": (
        "amdgpu 0000:0a:00.0: amdgpu: 1 correctable hardware errors detected "
        "in UMC block"
    ),
    "amdgpu_ras_bad_page": (
        "amdgpu 0000:0a:00.0: amdgpu: umc bad page: retired page 0x1f000 "
        "reserved"
    ),
    "amdgpu_gpu_reset_begin": "amdgpu 0000:0a:00.0: amdgpu: GPU reset begin!",
    "amdgpu_gpu_reset_failed": "amdgpu 0000:0a:00.0: amdgpu: GPU reset(3) failed",
    "amdgpu_xgmi_error": (
        "amdgpu 0000:0a:00.0: amdgpu: XGMI: link 3 error detected, fabric degraded"
    ),
    "amdgpu_smu_error": (
        "amdgpu 0000:0a:00.0: amdgpu: SMU: I'm not done with your previous "
        "command: SMN_C2PMSG_66:0x0000000E"
    ),
    "amd_rccl_segfault_in_librccl": (
        "python[12345]: segfault at 7f0000000000 ip 00007f1234567890 sp "
        "00007ffc12345678 error 4 in librccl.so.1.0[7f1234000000+1000000]"
    ),
    "amdgpu_fallen_off_bus": (
        "amdgpu 0000:0a:00.0: amdgpu: GPU has fallen off the bus"
    ),
}


def get_message_to_inject(name: str) -> Optional[str]:
    return INJECTABLE.get(name)
