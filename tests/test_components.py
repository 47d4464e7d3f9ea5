"""Component tests against the mock SMI backend (no GPU).

Pattern follows the reference test strategy: injected getter
functions per component + the whole-library mock via GPUD_AMDSMI_MOCK.
"""

import datetime
import os

import pytest

from gpud_amd.apiv1.types import HealthStateType, RepairActionType, utcnow
from gpud_amd.bootstrap import build_core
from gpud_amd.components.base import GPUdInstance
from gpud_amd.pkg.config import Config
from gpud_amd.pkg.fault_injector import SMIFailureInjector
from gpud_amd.smi import Instance
from gpud_amd.smi.mock import MockBackend


@pytest.fixture()
def mock_core(monkeypatch):
    monkeypatch.setenv("GPUD_AMDSMI_MOCK", "1")
    monkeypatch.setenv("GPUD_AMDSMI_MOCK_GPUS", "4")
    core = build_core(
        in_memory_db=True, kmsg_writable=False, record_reboot=False
    )
    yield core
    core.close()


def _check(core, name):
    c = core.registry.get(name)
    assert c is not None, f"component {name} not registered"
    return c.trigger_check()


def test_all_components_registered(mock_core):
    names = mock_core.registry.names()
    for expected in [
        "accelerator-amd-temperature",
        "accelerator-amd-power",
        "accelerator-amd-clock-speed",
        "accelerator-amd-utilization",
        "accelerator-amd-memory",
        "accelerator-amd-processes",
        "accelerator-amd-gpu-counts",
        "accelerator-amd-ecc",
        "accelerator-amd-bad-pages",
        "accelerator-amd-throttle",
        "accelerator-amd-xgmi",
        "accelerator-amd-gpm",
        "accelerator-amd-rccl",
        "accelerator-amd-error-ras",
        "cpu",
        "memory",
        "disk",
        "os",
        "fuse",
        "kernel-module",
        "library",
        "nfs",
        "network-latency",
        "pci",
    ]:
        assert expected in names, f"{expected} missing from registry"


def test_healthy_mock_scan(mock_core):
    # every accelerator component healthy on the 4-GPU mock
    for name in [
        "accelerator-amd-temperature",
        "accelerator-amd-power",
        "accelerator-amd-clock-speed",
        "accelerator-amd-utilization",
        "accelerator-amd-memory",
        "accelerator-amd-ecc",
        "accelerator-amd-bad-pages",
        "accelerator-amd-xgmi",
        "accelerator-amd-gpm",
    ]:
        cr = _check(mock_core, name)
        assert cr.health == HealthStateType.HEALTHY, f"{name}: {cr.reason} {cr.error}"


def test_temperature_margin_degraded(mock_core):
    backend = mock_core.smi_instance._b
    backend.state[0]["temp_hotspot"] = 104  # limit 110, margin 6 < 10
    mock_core.shared_snapshots.refresh()
    cr = _check(mock_core, "accelerator-amd-temperature")
    assert cr.health == HealthStateType.DEGRADED
    backend.state[0]["temp_hotspot"] = 111  # over limit
    mock_core.shared_snapshots.refresh()
    cr = _check(mock_core, "accelerator-amd-temperature")
    assert cr.health == HealthStateType.UNHEALTHY


def test_ecc_uncorrectable_unhealthy(mock_core):
    backend = mock_core.smi_instance._b
    backend.state[1]["ecc_uncorrectable"] = 3
    mock_core.shared_snapshots.refresh()
    cr = _check(mock_core, "accelerator-amd-ecc")
    assert cr.health == HealthStateType.UNHEALTHY
    assert RepairActionType.HARDWARE_INSPECTION in cr.suggested_actions.repair_actions


def test_bad_pages_pending_suggests_reboot(mock_core):
    backend = mock_core.smi_instance._b
    backend.state[2]["bad_pages_total"] = 2
    backend.state[2]["bad_pages_pending"] = 1
    mock_core.shared_snapshots.refresh()
    cr = _check(mock_core, "accelerator-amd-bad-pages")
    assert cr.health == HealthStateType.UNHEALTHY
    assert RepairActionType.REBOOT_SYSTEM in cr.suggested_actions.repair_actions


def test_xgmi_link_down_unhealthy(mock_core):
    backend = mock_core.smi_instance._b
    cr = _check(mock_core, "accelerator-amd-xgmi")  # prime last_states
    backend.state[0]["xgmi_states"][3] = 0
    mock_core.shared_snapshots.refresh()
    cr = _check(mock_core, "accelerator-amd-xgmi")
    assert cr.health == HealthStateType.UNHEALTHY
    assert "links down" in cr.reason
    # flap event recorded
    comp = mock_core.registry.get("accelerator-amd-xgmi")
    evs = comp.events(utcnow() - datetime.timedelta(minutes=5))
    assert any(e.name == "amd_xgmi_link_down" for e in evs)


def test_throttle_active_degraded(mock_core):
    backend = mock_core.smi_instance._b
    backend.state[0]["throttle"] = {"active_ppt_pwr": 1, "acc_ppt_pwr": 7}
    mock_core.shared_snapshots.refresh()
    cr = _check(mock_core, "accelerator-amd-throttle")
    assert cr.health in (HealthStateType.DEGRADED, HealthStateType.UNHEALTHY)
    assert "power" in cr.reason


def test_smi_failure_injector_gpu_lost(mock_core):
    fi = mock_core.smi_failure_injector
    uuids = mock_core.smi_instance.device_uuids()
    fi.gpu_lost_uuids.add(uuids[0])
    mock_core.shared_snapshots.refresh()
    snaps = mock_core.shared_snapshots.get()
    assert uuids[0] not in snaps
    assert len(snaps) == 3


def test_smi_failure_injector_ecc(mock_core):
    fi = mock_core.smi_failure_injector
    uuids = mock_core.smi_instance.device_uuids()
    fi.ecc_uncorrectable_uuids.add(uuids[1])
    mock_core.shared_snapshots.refresh()
    cr = _check(mock_core, "accelerator-amd-ecc")
    assert cr.health == HealthStateType.UNHEALTHY


def test_gpu_counts_with_expected(mock_core):
    comp = mock_core.registry.get("accelerator-amd-gpu-counts")
    comp.count_lspci = lambda: 4
    comp.expected = 4
    cr = comp.trigger_check()
    assert cr.health == HealthStateType.HEALTHY
    comp.expected = 8
    cr = comp.trigger_check()
    assert cr.health == HealthStateType.UNHEALTHY
    assert "expected 8" in cr.reason
    comp.expected = 4
    comp.count_lspci = lambda: 3  # bus shows fewer than the driver: hard fail
    cr = comp.trigger_check()
    assert cr.health == HealthStateType.UNHEALTHY
    comp.count_lspci = lambda: 8  # bus shows more: container visibility case
    cr = comp.trigger_check()
    assert cr.health == HealthStateType.DEGRADED


def test_error_ras_state_machine(mock_core):
    from gpud_amd.apiv1.types import Event, EventType

    comp = mock_core.registry.get("accelerator-amd-error-ras")
    bucket = mock_core.event_store.bucket("accelerator-amd-error-ras")
    now = utcnow()
    # healthy with no events
    cr = comp.trigger_check()
    assert cr.health == HealthStateType.HEALTHY
    # a critical catalog event makes it unhealthy
    bucket.insert(
        Event(
            time=now,
            component=comp.name,
            name="amdgpu_ring_timeout",
            type=EventType.CRITICAL,
            message="ring gfx_0.0.0 timeout",
        )
    )
    cr = comp.trigger_check()
    assert cr.health == HealthStateType.UNHEALTHY
    assert "amdgpu_ring_timeout" in cr.reason
    assert RepairActionType.CHECK_USER_APP_AND_GPU in cr.suggested_actions.repair_actions
    # SetHealthy clears it
    assert comp.can_set_healthy()
    comp.set_healthy()
    cr = comp.last_check_result()
    assert cr.health == HealthStateType.HEALTHY


def test_error_ras_reboot_escalation(mock_core):
    from gpud_amd.apiv1.types import Event, EventType
    from gpud_amd.pkg.host import EVENT_NAME_REBOOT, REBOOT_BUCKET

    comp = mock_core.registry.get("accelerator-amd-error-ras")
    bucket = mock_core.event_store.bucket("accelerator-amd-error-ras")
    os_bucket = mock_core.event_store.bucket(REBOOT_BUCKET)
    now = utcnow()
    first = now - datetime.timedelta(hours=10)
    # event appeared, we rebooted twice, event came back
    bucket.insert(Event(time=first, component=comp.name, name="amdgpu_ras_uncorrectable", type=EventType.FATAL, message="ue"))
    os_bucket.insert(Event(time=now - datetime.timedelta(hours=8), component="os", name=EVENT_NAME_REBOOT, type=EventType.WARNING, message="reboot"))
    os_bucket.insert(Event(time=now - datetime.timedelta(hours=4), component="os", name=EVENT_NAME_REBOOT, type=EventType.WARNING, message="reboot"))
    bucket.insert(Event(time=now, component=comp.name, name="amdgpu_ras_uncorrectable", type=EventType.FATAL, message="ue again"))
    cr = comp.trigger_check()
    assert cr.health == HealthStateType.UNHEALTHY
    assert cr.suggested_actions.repair_actions == [RepairActionType.HARDWARE_INSPECTION]


def test_host_components_run(mock_core):
    for name in ["cpu", "memory", "disk", "os"]:
        cr = _check(mock_core, name)
        assert cr.health in (
            HealthStateType.HEALTHY,
            HealthStateType.DEGRADED,
        ), f"{name}: {cr.reason}"


def test_kernel_module_component(mock_core):
    comp = mock_core.registry.get("kernel-module")
    comp.modules_to_check = ["amdgpu", "bogus_mod"]
    comp.get_loaded = lambda: {"amdgpu", "ext4"}
    cr = comp.trigger_check()
    assert cr.health == HealthStateType.UNHEALTHY
    assert "bogus_mod" in cr.reason
    comp.modules_to_check = ["amdgpu"]
    cr = comp.trigger_check()
    assert cr.health == HealthStateType.HEALTHY


def test_nfs_component(tmp_path, mock_core):
    from gpud_amd.components.host.nfs import GroupConfig

    comp = mock_core.registry.get("nfs")
    comp.set_configs([GroupConfig(volume_path=str(tmp_path))])
    cr = comp.trigger_check()
    assert cr.health == HealthStateType.HEALTHY, cr.reason
    # a regular file as the "volume" makes mkdir fail even as root
    blocker = tmp_path / "blocker"
    blocker.write_text("x")
    comp.set_configs([GroupConfig(volume_path=str(blocker))])
    cr = comp.trigger_check()
    assert cr.health == HealthStateType.UNHEALTHY


def test_check_duration_histogram_populated(mock_core):
    _check(mock_core, "cpu")
    scraped = mock_core.metrics_scraper.scrape()
    names = {m.name for m in scraped}
    assert "gpud_component_check_duration_seconds_count" in names


def test_gpm_per_xcc_busy(mock_core):
    """gpm exports the per-XCC (per-XCD) busy breakdown and its spread —
    the CDNA per-pipe utilization analog (amdsmi has no
    MFMA-pipe counter; xcp_stats per-XCD busy is the real signal)."""
    comp = mock_core.registry.get("accelerator-amd-gpm")
    cr = comp.trigger_check()
    assert cr.health == "Healthy"
    xcc_keys = [k for k in (cr.extra_info or {}) if k.endswith(".xcc_busy")]
    assert xcc_keys, cr.extra_info
    # the mock reports 8 XCDs per GPU
    assert len((cr.extra_info[xcc_keys[0]]).split(",")) == 8
    from prometheus_client import generate_latest

    text = generate_latest(mock_core.metrics_registry).decode()
    assert "accelerator_amd_gpm_xcc_busy_percent" in text
    assert "accelerator_amd_gpm_xcc_busy_spread_percent" in text
    assert "accelerator_amd_gpm_mm_activity_percent" in text


def test_power_at_limit_degraded(mock_core):
    """Sustained draw at >=98% of the enforced limit flips power to
    Degraded (reference: power component's usage-vs-limit gauges —
    components/accelerator/nvidia/power/component.go)."""
    backend = mock_core.smi_instance._b
    backend.state[0]["power_w"] = 1390  # limit 1400 -> 99.3%
    mock_core.shared_snapshots.refresh()
    cr = _check(mock_core, "accelerator-amd-power")
    assert cr.health == HealthStateType.DEGRADED
    assert "power limit" in cr.reason
    backend.state[0]["power_w"] = 620
    mock_core.shared_snapshots.refresh()
    cr = _check(mock_core, "accelerator-amd-power")
    assert cr.health == HealthStateType.HEALTHY


def test_power_management_disabled_degraded(mock_core, monkeypatch):
    """power-management mirrors the reference persistence-mode check:
    informational Degraded when the mode is off (persistence-mode
    component.go analog)."""
    backend = mock_core.smi_instance._b
    monkeypatch.setattr(
        type(backend), "power_management_enabled", lambda self, i: i != 1
    )
    cr = _check(mock_core, "accelerator-amd-power-management")
    assert cr.health == HealthStateType.DEGRADED
    assert "disabled" in cr.reason


def test_network_latency_unreachable_unhealthy(mock_core):
    """All targets unreachable -> Unhealthy; one in-threshold target ->
    Healthy (reference: network-latency global thresholds,
    components/network/latency/component.go)."""
    from gpud_amd.components.host.network_latency import NetworkLatencyComponent

    comp = NetworkLatencyComponent(mock_core.gpud_instance)
    comp.targets = [("198.51.100.1", 443), ("198.51.100.2", 443)]
    comp.probe = lambda host, port: -1.0
    cr = comp.check()
    assert cr.health == HealthStateType.UNHEALTHY
    assert cr.extra_info["198.51.100.1:443"] == "unreachable"
    comp.probe = lambda host, port: 3.5 if host.endswith(".1") else -1.0
    cr = comp.check()
    assert cr.health == HealthStateType.HEALTHY
    assert "1/2 targets" in cr.reason


def test_peer_mem_degraded_without_providers_or_dmabuf(mock_core, monkeypatch):
    """RDMA NICs present but no PeerDirect provider and no DMABUF kernel
    support -> Degraded (reference: peermem ko/lsmod check,
    components/accelerator/nvidia/peermem/component.go)."""
    comp = mock_core.registry.get("accelerator-amd-peer-mem")
    monkeypatch.setattr(comp, "has_rdma_nics", lambda: True)
    monkeypatch.setattr(comp, "get_providers", lambda: [])
    monkeypatch.setattr(comp, "has_dmabuf", lambda: False)
    cr = comp.trigger_check()
    assert cr.health == HealthStateType.DEGRADED
    assert "bounce buffers" in cr.reason
    monkeypatch.setattr(comp, "get_providers", lambda: ["amdgpu_peerdirect"])
    cr = comp.trigger_check()
    assert cr.health == HealthStateType.HEALTHY


def test_gpu_counts_render_node_triangulation(mock_core, monkeypatch):
    """/dev/dri renderD* is the third count source (reference:
    pkg/nvidia/dev/device_count.go): fewer render nodes than amdsmi
    devices means device files are missing."""
    import gpud_amd.smi as smi_pkg

    comp = mock_core.registry.get("accelerator-amd-gpu-counts")
    monkeypatch.setattr(smi_pkg, "mock_enabled", lambda: False)
    comp.count_lspci = lambda: 4
    comp.count_render = lambda: 2  # 2 nodes for 4 GPUs
    cr = comp.trigger_check()
    assert cr.health == HealthStateType.UNHEALTHY
    assert "render node" in cr.reason
    comp.count_render = lambda: 4
    assert comp.trigger_check().health == HealthStateType.HEALTHY
    comp.count_render = lambda: None  # /dev/dri unreadable: no claim
    assert comp.trigger_check().health == HealthStateType.HEALTHY
