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
    # ---- round-2 expansion: amdgpu/KFD lifecycle depth ---------------------
    _d(
        "amdgpu_mode1_reset",
        r"amdgpu.*GPU mode1 reset",
        "Whole-ASIC mode-1 reset — the heaviest reset path (full chip, all "
        "XCDs); every context on the device was lost",
        EventType.CRITICAL,
        _APP,
        critical=True,
    ),
    _d(
        "amdgpu_atombios_hang",
        r"amdgpu.*atombios stuck in loop",
        "AtomBIOS command-table execution hung — VBIOS-level wedge; the "
        "device usually needs a reset and the board firmware is suspect",
        EventType.FATAL,
        _REBOOT + _HW,
        critical=True,
    ),
    _d(
        "amdgpu_flr_notification",
        r"amdgpu.*FLR notification",
        "SR-IOV function-level-reset notification from the host — the "
        "hypervisor reset this virtual function; running work was lost",
        EventType.CRITICAL,
        _APP,
    ),
    _d(
        "amdgpu_bo_va_update_failed",
        r"amdgpu.*Couldn't update BO_VA",
        "GPU VM mapping update failed (usually -ENOMEM under memory "
        "pressure) — the submitting process sees submission errors",
        EventType.WARNING,
        _APP,
    ),
    _d(
        "amdgpu_evict_resources_failed",
        r"amdgpu.*evicting device resources failed",
        "Device resource eviction failed during suspend/reset preparation — "
        "the following reset/resume is suspect",
        EventType.WARNING,
        _REBOOT,
    ),
    _d(
        "amdgpu_deferred_error",
        r"amdgpu.*deferred (?:hardware )?error",
        "RAS deferred (latent) hardware error logged — not yet consumed; "
        "pages will be poisoned on touch. Watch for poison-consumption "
        "events and bad-page growth",
        EventType.CRITICAL,
        _HW,
    ),
    _d(
        "kfd_migrate_failed",
        r"(?:kfd|amdgpu).*fail(?:ed)? to migrate",
        "SVM range migration between VRAM and host memory failed — the "
        "compute process may stall or abort",
        EventType.WARNING,
        _APP,
    ),
    _d(
        "kfd_restore_queues_failed",
        r"(?:kfd|amdgpu).*[Ff]ailed to restore queues",
        "KFD could not restore a process's queues after eviction — the "
        "process's GPU work is stopped",
        EventType.CRITICAL,
        _APP,
        critical=True,
    ),
    _d(
        "kfd_reset_wavefronts",
        r"(?:kfd|amdgpu).*[Rr]esetting wave fronts",
        "KFD reset in-flight wavefronts (hang recovery on a compute queue) "
        "— the owning process's kernels were killed",
        EventType.CRITICAL,
        _APP,
        critical=True,
    ),
    _d(
        "kfd_unmap_queue_failed",
        r"(?:kfd|amdgpu).*[Ff]ailed to unmap (?:legacy )?queue",
        "KFD queue unmap failed — frequently the precursor of a HWS hang "
        "and a following GPU reset",
        EventType.CRITICAL,
        _APP,
        critical=True,
    ),
    # ---- host kernel-integrity signatures (Xid-table host analogs) ---------
    _d(
        "host_kernel_panic",
        r"Kernel panic - not syncing",
        "Kernel panic on the current boot ring (pstore carries prior-boot "
        "panics) — node integrity lost",
        EventType.FATAL,
        _REBOOT + _HW,
        critical=True,
    ),
    _d(
        "host_kernel_bug",
        r"BUG: (?:unable to handle|kernel NULL pointer)",
        "Kernel page-fault BUG (NULL dereference / unhandled fault) — a "
        "kernel code path crashed; node stability is compromised",
        EventType.FATAL,
        _REBOOT,
        critical=True,
    ),
    _d(
        "host_kernel_oops",
        r"Oops: [0-9a-f]{4}",
        "Kernel oops logged — a kernel thread died; taint follows and later "
        "failures are suspect until reboot",
        EventType.CRITICAL,
        _REBOOT,
    ),
    _d(
        "host_rcu_stall",
        r"rcu.*detected (?:stalls|expedited stalls)",
        "RCU stall — a CPU stopped responding to the RCU state machine for "
        "seconds; driver wedges and hard IRQ storms look like this",
        EventType.CRITICAL,
        _REBOOT,
    ),
    _d(
        "host_hard_lockup",
        r"Watchdog detected hard LOCKUP",
        "NMI watchdog hard lockup — a CPU stopped servicing interrupts; "
        "usually firmware/hardware level trouble",
        EventType.FATAL,
        _REBOOT + _HW,
        critical=True,
    ),
    _d(
        "host_apei_hardware_error",
        r"\{\d+\}\[Hardware Error\]",
        "APEI/GHES firmware-reported hardware error (the kernel-side print "
        "of a CPER record) — correlate with the cper component's records",
        EventType.CRITICAL,
        _HW,
    ),
    _d(
        "host_memory_failure",
        r"Memory failure: 0x",
        "Kernel memory-failure handling (hwpoison) ran on a page — ECC "
        "uncorrectable host memory; track recurrence per DIMM via EDAC",
        EventType.CRITICAL,
        _HW,
    ),
    _d(
        "host_list_corruption",
        r"list_(?:del|add) corruption",
        "Kernel list corruption detected — memory corruption inside the "
        "kernel; frequently a driver bug or failing DIMM",
        EventType.FATAL,
        _REBOOT + _HW,
        critical=True,
    ),
    _d(
        "host_irq_nobody_cared",
        r"irq \d+: nobody cared",
        "An IRQ line fired with no handler claiming it — the kernel "
        "disabled the line; devices behind it stop interrupting",
        EventType.WARNING,
        _REBOOT,
    ),
    _d(
        "host_page_alloc_failure",
        r"page allocation failure",
        "Kernel page allocation failure (fragmentation/pressure) — DMA "
        "buffer allocations may be failing",
        EventType.WARNING,
        _APP,
    ),
    _d(
        "host_cpu_thermal_throttle",
        r"[Cc]ore temperature above threshold",
        "Host CPU thermal throttling engaged — check chassis cooling; GPU "
        "thermals on the same node are suspect",
        EventType.WARNING,
        _HW,
    ),
    # ---- storage / filesystem / network health -----------------------------
    _d(
        "host_io_error",
        r"I/O error, dev",
        "Block-layer I/O error — failing disk or transport; check the "
        "device's SMART state",
        EventType.CRITICAL,
        _HW,
    ),
    _d(
        "host_filesystem_error",
        r"(?:EXT4-fs error|XFS \([^)]*\): (?:Internal error|Corruption)|"
        r"BTRFS error)",
        "Filesystem-level error/corruption — data integrity at risk; fsck "
        "and underlying device inspection needed",
        EventType.CRITICAL,
        _HW,
        critical=True,
    ),
    _d(
        "host_filesystem_readonly",
        r"Remounting filesystem read-only",
        "A filesystem remounted itself read-only after errors — writes are "
        "failing node-wide on that mount",
        EventType.FATAL,
        _HW,
        critical=True,
    ),
    _d(
        "nvme_io_timeout",
        r"nvme nvme\d+: (?:I/O .*timeout|controller is down|Device not ready)",
        "NVMe command timeout / controller failure — local storage is "
        "degraded or lost",
        EventType.CRITICAL,
        _HW,
    ),
    _d(
        "host_nfs_not_responding",
        r"nfs: server .* not responding",
        "NFS server stopped responding — mounts hang and D-state process "
        "counts rise (see the os component's tracker)",
        EventType.WARNING,
        _APP,
    ),
    _d(
        "host_netdev_watchdog",
        r"NETDEV WATCHDOG",
        "Network transmit queue timeout — NIC or driver wedge on a host "
        "interface",
        EventType.CRITICAL,
        _HW,
    ),
    # ---- IOMMU / PCIe depth -------------------------------------------------
    _d(
        "iommu_io_page_fault",
        r"AMD-Vi.*IO_PAGE_FAULT",
        "IOMMU DMA page fault — a device (often a GPU after a bad mapping "
        "or reset) performed DMA to an unmapped address",
        EventType.CRITICAL,
        _APP,
    ),
    _d(
        "pcie_dpc_containment",
        r"DPC: containment event",
        "PCIe Downstream Port Containment fired — the link was cut to "
        "contain an uncorrected error; devices below dropped off",
        EventType.FATAL,
        _HW,
        critical=True,
    ),
    _d(
        "pcie_link_down",
        r"pciehp.*Link Down",
        "PCIe hotplug reported Link Down — a device (possibly a GPU) left "
        "the bus",
        EventType.CRITICAL,
        _HW,
        critical=True,
    ),
    # ---- round-2b expansion: engine tests, firmware, fabric NICs, disks ----
    # (appended AFTER the families above so first-match-wins keeps every
    # earlier, more specific signature; each entry matcher-tested)
    _d(
        "amdgpu_ring_test_failed",
        r"amdgpu.*ring (?P<ring>\S+) test failed",
        "A ring's start-of-day test failed (gfx/sdma/vcn/jpeg engine did "
        "not answer) — the engine is hung or the IP block failed init",
        EventType.CRITICAL,
        _REBOOT,
        critical=True,
    ),
    _d(
        "amdgpu_ras_init_failed",
        r"amdgpu.*RAS.*init.*fail",
        "RAS subsystem failed to initialise — error telemetry is blind "
        "until the driver reloads",
        EventType.WARNING,
        _REBOOT,
    ),
    _d(
        "amdgpu_ras_ta_missing",
        r"amdgpu.*ras ta ucode is not available",
        "RAS trusted application firmware missing — RAS features degraded; "
        "usually a ROCm/firmware packaging problem",
        EventType.WARNING,
    ),
    _d(
        "amdgpu_vbios_invalid",
        r"amdgpu.*(?:Invalid VBIOS|BIOS signature incorrect)",
        "Video BIOS image invalid or unsigned — board firmware corruption",
        EventType.CRITICAL,
        _HW,
        critical=True,
    ),
    _d(
        "amdgpu_gfxoff_failed",
        r"amdgpu.*[Ff]ailed to (?:disable|enable) gfxoff",
        "GFXOFF state transition failed — power-management degraded; "
        "recurrence alongside SMU errors implicates the SMU firmware",
        EventType.WARNING,
    ),
    _d(
        "amdgpu_reg_write_failed",
        r"amdgpu.*failed to write reg \S+ wait reg",
        "Register write-and-confirm failed — the GFX core is not "
        "responding (typically accompanies a hang/reset sequence)",
        EventType.CRITICAL,
        _REBOOT,
    ),
    _d(
        "mlx5_device_error",
        r"mlx5_core.*(?:health compromised|firmware internal error|"
        r"assert_var|health buffer)",
        "Fabric NIC (mlx5) device health error — firmware assert or "
        "internal error; RDMA traffic on this port is suspect",
        EventType.CRITICAL,
        _HW,
        critical=True,
    ),
    _d(
        "mlx5_port_module_error",
        r"mlx5_core.*Port module event\[error\]",
        "Fabric NIC port module (cable/transceiver) error — check the "
        "cable, transceiver seating and power budget",
        EventType.CRITICAL,
        _HW,
    ),
    _d(
        "pcie_card_removed",
        r"pciehp.*(?:Card not present|Surprise removal)",
        "PCIe hotplug controller reports the card gone — a device "
        "(possibly a GPU) dropped off the slot",
        EventType.CRITICAL,
        _HW,
        critical=True,
    ),
    _d(
        "host_thermal_critical_shutdown",
        r"critical temperature reached.*shutting down",
        "A thermal zone hit its critical trip point and the kernel is "
        "shutting the machine down — cooling failure",
        EventType.FATAL,
        _HW,
        critical=True,
    ),
    _d(
        "host_disk_medium_error",
        r"critical medium error, dev (?P<dev>\S+)",
        "Disk medium error (unreadable sector) — the drive is failing; "
        "check SMART and plan replacement",
        EventType.CRITICAL,
        _HW,
    ),
    _d(
        "host_md_disk_failure",
        r"md/raid.*Disk failure on (?P<dev>\S+)",
        "Software-RAID member failed and was kicked from the array — "
        "redundancy reduced; replace the member",
        EventType.CRITICAL,
        _HW,
    ),
    _d(
        "host_jbd2_io_error",
        r"JBD2: Detected IO errors",
        "Journal layer saw I/O errors flushing data — filesystem may "
        "degrade to read-only next; inspect the underlying device",
        EventType.CRITICAL,
        _HW,
    ),
    _d(
        "host_acpi_error",
        r"ACPI (?:BIOS )?Error[ :]",
        "ACPI/BIOS error — firmware table or method problem; benign "
        "recurrences are common but new onset after a BIOS update is not",
        EventType.WARNING,
    ),
    _d(
        "host_firmware_bug",
        r"\[Firmware Bug\]:",
        "Kernel flagged a platform firmware bug — record for correlation; "
        "persistent new entries warrant a BIOS/BMC update",
        EventType.WARNING,
    ),
    _d(
        "host_clocksource_unstable",
        r"clocksource.*Marking clocksource .* as unstable",
        "TSC/clocksource marked unstable — timekeeping degraded, timers "
        "and profiling results are suspect on this node",
        EventType.WARNING,
    ),
    _d(
        "host_vfs_file_max_reached",
        r"VFS: file-max limit \d+ reached",
        "System-wide open-file-handle limit reached — new opens fail "
        "everywhere until handles are released or fs.file-max is raised "
        "(reference: os/kmsg_matcher.go VFS file-max event)",
        EventType.CRITICAL,
        _APP,
    ),
    _d(
        "host_tcp_oom",
        r"TCP: out of memory",
        "TCP stack out of memory — socket buffers exhausted under load; "
        "tune tcp_mem or find the flood source",
        EventType.WARNING,
    ),
    _d(
        "host_conntrack_full",
        r"nf_conntrack: .*table full, dropping packet",
        "Connection-tracking table full — new connections are being "
        "dropped; raise nf_conntrack_max or reduce connection churn",
        EventType.WARNING,
    ),
    # ---- ROCm user-space crash signatures ----------------------------------
    _d(
        "amd_hip_segfault_in_libamdhip",
        r"segfault at .* in libamdhip64",
        "Process crashed inside the HIP runtime — correlate with preceding "
        "amdgpu/KFD events; repeated crashes across processes implicate the "
        "node, a single app implicates the app",
        EventType.CRITICAL,
        _APP,
    ),
    _d(
        "amd_rocm_lib_segfault",
        r"segfault at .* in (?:librocblas|libhipblaslt|libMIOpen|"
        r"librocsolver|librocfft)",
        "Process crashed inside a ROCm math library — usually an "
        "application/library-version issue; node-wide recurrence points at "
        "the GPU or driver",
        EventType.WARNING,
        _APP,
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
    "amdgpu_deferred_error": (
        "amdgpu 0000:0a:00.0: amdgpu: 3 deferred hardware errors detected "
        "in UMC block"
    ),
    "iommu_io_page_fault": (
        "AMD-Vi: Event logged [IO_PAGE_FAULT domain=0x0035 "
        "address=0x7f8100000000 flags=0x0070]"
    ),
    "pcie_dpc_containment": (
        "pcieport 0000:00:01.1: DPC: containment event, status:0x1f01 "
        "source:0x0000"
    ),
    "host_rcu_stall": (
        "rcu: INFO: rcu_sched detected stalls on CPUs/tasks: { 12-.... } "
        "(detected by 0, t=60002 jiffies)"
    ),
    "kfd_reset_wavefronts": (
        "kfd kfd: amdgpu: Resetting wave fronts on dev 0xb3c"
    ),
}


def get_message_to_inject(name: str) -> Optional[str]:
    return INJECTABLE.get(name)
