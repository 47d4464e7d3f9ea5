"""Edge-case component tests: throttle window, config gating, diagnostics
over the session, gzip/trigger-tag endpoints, power normalization."""

import datetime
import gzip as gzip_mod

import pytest

from gpud_amd.apiv1.types import HealthStateType, utcnow


@pytest.fixture()
def mock_core(monkeypatch, tmp_path):
    monkeypatch.setenv("GPUD_AMDSMI_MOCK", "1")
    monkeypatch.setenv("GPUD_AMDSMI_MOCK_GPUS", "2")
    from gpud_amd.bootstrap import build_core
    from gpud_amd.pkg.config import Config

    cfg = Config(data_dir=str(tmp_path))
    core = build_core(cfg, in_memory_db=True, kmsg_writable=False, record_reboot=False)
    yield core
    core.close()


def test_throttle_windowed_unhealthy(mock_core):
    """Sustained throttle events over the 10-min window ⇒ Unhealthy
    (reference hw-slowdown rule: ≥0.6 events/min)."""
    from gpud_amd.apiv1.types import Event, EventType
    from gpud_amd.components.accelerator.throttle import EVENT_NAME, WINDOW

    comp = mock_core.registry.get("accelerator-amd-throttle")
    bucket = mock_core.event_store.bucket("accelerator-amd-throttle")
    now = utcnow()
    # 8 events in the last 10 minutes = 0.8/min ≥ 0.6
    for i in range(8):
        bucket.insert(
            Event(
                time=now - datetime.timedelta(minutes=i),
                component=comp.name,
                name=EVENT_NAME,
                type=EventType.WARNING,
                message=f"synthetic throttle {i}",
            )
        )
    comp.get_now = lambda: now
    cr = comp.trigger_check()
    assert cr.health == HealthStateType.UNHEALTHY
    assert "sustained throttling" in cr.reason


def test_throttle_residency_delta_detection(mock_core):
    """Rising residency accumulators between polls flag active throttling."""
    backend = mock_core.smi_instance._b
    comp = mock_core.registry.get("accelerator-amd-throttle")
    comp.trigger_check()  # prime the accumulator cache
    backend.state[0]["throttle"] = {"acc_hbm_thrm": 50}
    mock_core.shared_snapshots.refresh()
    cr = comp.trigger_check()
    assert cr.health in (HealthStateType.DEGRADED, HealthStateType.UNHEALTHY)
    assert "HBM" in cr.reason


def test_component_enable_disable(monkeypatch, tmp_path):
    monkeypatch.setenv("GPUD_AMDSMI_MOCK", "1")
    from gpud_amd.bootstrap import build_core
    from gpud_amd.pkg.config import Config

    cfg = Config(data_dir=str(tmp_path))
    cfg.disabled_components = ["docker", "tailscale"]
    core = build_core(cfg, in_memory_db=True, kmsg_writable=False, record_reboot=False)
    try:
        names = core.registry.names()
        assert "docker" not in names and "tailscale" not in names
        assert "cpu" in names
    finally:
        core.close()

    cfg2 = Config(data_dir=str(tmp_path / "b"))
    cfg2.enabled_components = ["cpu", "memory"]
    core2 = build_core(cfg2, in_memory_db=True, kmsg_writable=False, record_reboot=False)
    try:
        assert sorted(core2.registry.names()) == ["cpu", "memory"]
    finally:
        core2.close()


def test_session_diagnostic_method(mock_core):
    from gpud_amd.session import Session

    s = Session(
        mock_core,
        endpoint="unused",
        open_reader=lambda: iter(()),
        send_response=lambda f: None,
    )
    resp = s.process_request({"req_id": "d", "method": "diagnostic", "data": {}})
    diags = resp["data"]["diagnostics"]
    assert "accelerator-amd-diag-mfma" in diags
    # no /dev/kfd in CI: diag components report healthy "not applicable"
    # (they would be UNHEALTHY on a GPU host without the extension)
    assert diags["accelerator-amd-diag-mfma"]["health"] in ("Healthy", "Unhealthy")


def test_power_management_component(mock_core):
    cr = mock_core.registry.get("accelerator-amd-power-management").trigger_check()
    assert cr.health == HealthStateType.HEALTHY
    assert "enabled" in cr.reason


def test_config_yaml_roundtrip(tmp_path):
    from gpud_amd.pkg.config import Config

    cfg = Config(data_dir="/tmp/x", expected_gpu_count=8)
    cfg.disabled_components = ["docker"]
    path = str(tmp_path / "cfg.yaml")
    cfg.save(path)
    back = Config.load(path)
    assert back.expected_gpu_count == 8
    assert back.disabled_components == ["docker"]
    assert back.state_path == "/tmp/x/gpud.state"


def test_update_version_file_trigger(tmp_path):
    from gpud_amd.pkg.config import Config
    from gpud_amd.pkg.update import check_version_file, write_target_version

    cfg = Config(data_dir=str(tmp_path))
    assert check_version_file(cfg) is None
    write_target_version(cfg, "2.0.0")
    assert check_version_file(cfg) == "2.0.0"
    # same-version target is not a pending update
    from gpud_amd import __version__

    write_target_version(cfg, __version__)
    assert check_version_file(cfg) is None


def test_tar_path_traversal_refused(tmp_path):
    import io
    import tarfile

    from gpud_amd.pkg.config import Config
    from gpud_amd.pkg.update import update_to_version

    evil = tmp_path / "gpud-amd_6.6.6.tar.gz"
    buf = io.BytesIO()
    with tarfile.open(fileobj=buf, mode="w:gz") as tf:
        data = b"evil"
        info = tarfile.TarInfo(name="../../../etc/evil")
        info.size = len(data)
        tf.addfile(info, io.BytesIO(data))
    evil.write_bytes(buf.getvalue())
    cfg = Config(data_dir=str(tmp_path / "data"))
    err = update_to_version(
        cfg, "6.6.6", base_url=f"file://{tmp_path}", install_dir=str(tmp_path / "out")
    )
    assert err is not None and "unsafe path" in err


def test_xgmi_flap_auto_clear(mock_core):
    """A recovered link keeps the component Degraded inside the auto-clear
    window (reference: infiniband flap store semantics)."""
    backend = mock_core.smi_instance._b
    comp = mock_core.registry.get("accelerator-amd-xgmi")
    comp.trigger_check()  # prime last_states
    backend.state[0]["xgmi_states"][2] = 0  # link drops
    mock_core.shared_snapshots.refresh()
    cr = comp.trigger_check()
    assert cr.health == HealthStateType.UNHEALTHY
    backend.state[0]["xgmi_states"][2] = 1  # link recovers
    mock_core.shared_snapshots.refresh()
    cr = comp.trigger_check()
    assert cr.health == HealthStateType.DEGRADED  # still in the window
    assert "flapped" in cr.reason


def test_infiniband_component_with_fixture(tmp_path, mock_core):
    """sysfs fixture tree (reference pattern: class parser root param)."""
    root = tmp_path / "infiniband"
    for dev, port, state, rate, downed in [
        ("mlx5_0", "1", "4: ACTIVE", "400 Gb/sec (4X NDR)", 0),
        ("mlx5_1", "1", "1: DOWN", "400 Gb/sec (4X NDR)", 2),
    ]:
        pdir = root / dev / "ports" / port
        (pdir / "counters").mkdir(parents=True)
        (pdir / "state").write_text(state + "\n")
        (pdir / "phys_state").write_text("5: LinkUp\n")
        (pdir / "rate").write_text(rate + "\n")
        (pdir / "counters" / "link_downed").write_text(f"{downed}\n")
        (pdir / "counters" / "symbol_error").write_text("0\n")

    comp = mock_core.registry.get("infiniband")
    assert comp is not None
    comp.sysfs_root = str(root)
    cr = comp.trigger_check()
    assert cr.health == HealthStateType.UNHEALTHY
    assert "mlx5_1/1" in cr.reason

    # port comes back up, but a rising link_downed counter records a flap
    (root / "mlx5_1" / "ports" / "1" / "state").write_text("4: ACTIVE\n")
    (root / "mlx5_1" / "ports" / "1" / "counters" / "link_downed").write_text("3\n")
    cr = comp.trigger_check()
    assert cr.health == HealthStateType.DEGRADED
    assert "flapped" in cr.reason
    evs = comp.events(utcnow() - datetime.timedelta(minutes=5))
    assert any(e.name == "ib_port_flap" for e in evs)


def test_infiniband_expected_ports(mock_core, tmp_path):
    comp = mock_core.registry.get("infiniband")
    comp.sysfs_root = str(tmp_path / "none")
    comp.expected_ports = 2
    cr = comp.trigger_check()
    assert cr.health == HealthStateType.UNHEALTHY
    comp.expected_ports = 0
    cr = comp.trigger_check()
    assert cr.health == HealthStateType.HEALTHY


def test_peer_mem_component(mock_core):
    comp = mock_core.registry.get("accelerator-amd-peer-mem")
    assert comp is not None
    comp.has_rdma_nics = lambda: True
    comp.get_providers = lambda: ["amdgpu"]
    cr = comp.trigger_check()
    assert cr.health == HealthStateType.HEALTHY
    assert "amdgpu" in cr.reason
    comp.get_providers = lambda: []
    comp.has_dmabuf = lambda: False
    cr = comp.trigger_check()
    assert cr.health == HealthStateType.DEGRADED
    comp.has_dmabuf = lambda: True
    cr = comp.trigger_check()
    assert cr.health == HealthStateType.HEALTHY


def test_power_at_limit_degraded(mock_core):
    backend = mock_core.smi_instance._b
    backend.state[0]["power_w"] = 1390  # ≥98% of 1400
    mock_core.shared_snapshots.refresh()
    cr = mock_core.registry.get("accelerator-amd-power").trigger_check()
    assert cr.health == HealthStateType.DEGRADED
    assert "power limit" in cr.reason


def test_gpu_memory_component_values(mock_core):
    from gpud_amd.pkg.metrics import Scraper

    mock_core.registry.get("accelerator-amd-memory").trigger_check()
    scraped = Scraper(mock_core.metrics_registry).scrape()
    by_name = {}
    for m in scraped:
        if m.name == "accelerator_amd_memory_total_bytes":
            by_name[m.labels.get("uuid")] = m.value
    assert by_name  # per-uuid totals present
    assert all(v == 294_912 * 1024 * 1024 for v in by_name.values())  # 288 GB


def test_processes_component_lists(mock_core):
    cr = mock_core.registry.get("accelerator-amd-processes").trigger_check()
    assert cr.health == HealthStateType.HEALTHY
    assert "process" in cr.reason
    # pids surfaced in extra info
    assert any(k.endswith(".pids") for k in (cr.extra_info or {}))


def test_error_ras_noncritical_events_stay_healthy(mock_core):
    """Warning-class catalog events (e.g. reset succeeded) never flip the
    state machine."""
    from gpud_amd.apiv1.types import Event, EventType

    comp = mock_core.registry.get("accelerator-amd-error-ras")
    bucket = mock_core.event_store.bucket("accelerator-amd-error-ras")
    bucket.insert(
        Event(
            time=utcnow(),
            component=comp.name,
            name="amdgpu_gpu_reset_succeeded",
            type=EventType.WARNING,
            message="GPU reset(1) succeeded",
        )
    )
    cr = comp.trigger_check()
    assert cr.health == HealthStateType.HEALTHY


def test_error_ras_reboot_clears_state(mock_core):
    """A reboot AFTER the critical event clears the unhealthy state
    (reference: reboot events reset error states — pkg/host/event.go)."""
    from gpud_amd.apiv1.types import Event, EventType
    from gpud_amd.pkg.host import EVENT_NAME_REBOOT, REBOOT_BUCKET

    comp = mock_core.registry.get("accelerator-amd-error-ras")
    bucket = mock_core.event_store.bucket("accelerator-amd-error-ras")
    os_bucket = mock_core.event_store.bucket(REBOOT_BUCKET)
    now = utcnow()
    bucket.insert(
        Event(
            time=now - datetime.timedelta(hours=2),
            component=comp.name,
            name="amdgpu_ring_timeout",
            type=EventType.CRITICAL,
            message="timeout",
        )
    )
    cr = comp.trigger_check()
    assert cr.health == HealthStateType.UNHEALTHY
    os_bucket.insert(
        Event(
            time=now - datetime.timedelta(hours=1),
            component="os",
            name=EVENT_NAME_REBOOT,
            type=EventType.WARNING,
            message="reboot",
        )
    )
    cr = comp.trigger_check()
    assert cr.health == HealthStateType.HEALTHY


def test_scraper_histogram_series(mock_core):
    """The check-duration histogram surfaces bucket/sum/count series."""
    from gpud_amd.pkg.metrics import Scraper

    mock_core.registry.get("cpu").trigger_check()
    names = {m.name for m in Scraper(mock_core.metrics_registry).scrape()}
    assert "gpud_component_check_duration_seconds_bucket" in names
    assert "gpud_component_check_duration_seconds_sum" in names


# -- partition + CPER (MI355X-specific coverage) ----------------------------


def test_partition_informational_healthy(mock_core):
    comp = mock_core.registry.get("accelerator-amd-partition")
    cr = comp.trigger_check()
    assert cr.health == HealthStateType.HEALTHY
    assert "SPX/NPS1" in cr.reason


def test_partition_expected_mismatch_unhealthy(monkeypatch, tmp_path):
    monkeypatch.setenv("GPUD_AMDSMI_MOCK", "1")
    monkeypatch.setenv("GPUD_AMDSMI_MOCK_GPUS", "2")
    from gpud_amd.bootstrap import build_core
    from gpud_amd.pkg.config import Config

    cfg = Config(data_dir=str(tmp_path), expected_compute_partition="CPX")
    core = build_core(cfg, in_memory_db=True, kmsg_writable=False, record_reboot=False)
    try:
        comp = core.registry.get("accelerator-amd-partition")
        cr = comp.trigger_check()
        assert cr.health == HealthStateType.UNHEALTHY
        assert "expected CPX" in cr.reason
    finally:
        core.close()


def test_cper_no_records_healthy(mock_core):
    comp = mock_core.registry.get("accelerator-amd-cper")
    cr = comp.trigger_check()
    assert cr.health == HealthStateType.HEALTHY
    assert "no CPER records" in cr.reason


def test_cper_fatal_record_unhealthy_then_set_healthy(mock_core):
    from gpud_amd.apiv1.types import RepairActionType

    backend = mock_core.smi_instance._b
    comp = mock_core.registry.get("accelerator-amd-cper")
    comp.trigger_check()  # establish cursors
    backend.state[0]["cper"].append(
        {
            "severity": 1,
            "severity_name": "fatal",
            "record_id": "aa01",
            "notify_type": "MCE",
            "section_count": 2,
        }
    )
    cr = comp.trigger_check()
    assert cr.health == HealthStateType.UNHEALTHY
    assert "fatal CPER" in cr.reason
    assert cr.suggested_actions.repair_actions == [
        RepairActionType.HARDWARE_INSPECTION
    ]
    # dedup: same record must not re-alert counts
    before = comp._counts[mock_core.smi_instance.device_uuids()[0]][1]
    comp.trigger_check()
    assert comp._counts[mock_core.smi_instance.device_uuids()[0]][1] == before
    # events surfaced through the component API
    evs = comp.events(utcnow() - datetime.timedelta(minutes=5))
    assert any("record_id=" in e.message for e in evs)
    # operator clears
    comp.set_healthy()
    cr = comp.trigger_check()
    assert cr.health == HealthStateType.HEALTHY


def test_cper_uncorrected_degraded(mock_core):
    from gpud_amd.apiv1.types import RepairActionType

    backend = mock_core.smi_instance._b
    comp = mock_core.registry.get("accelerator-amd-cper")
    comp.trigger_check()
    backend.state[1]["cper"].append(
        {
            "severity": 0,
            "severity_name": "non_fatal_uncorrected",
            "record_id": "bb02",
            "notify_type": "CMC",
            "section_count": 1,
        }
    )
    cr = comp.trigger_check()
    assert cr.health == HealthStateType.DEGRADED
    assert cr.suggested_actions.repair_actions == [RepairActionType.REBOOT_SYSTEM]


def test_cper_restart_dedup_via_event_store(mock_core):
    """A record already in the durable event store must not re-alert after
    a simulated daemon restart (start() re-learns seen record ids)."""
    backend = mock_core.smi_instance._b
    comp = mock_core.registry.get("accelerator-amd-cper")
    comp.trigger_check()
    backend.state[0]["cper"].append(
        {"severity": 2, "severity_name": "non_fatal_corrected", "record_id": "cc03"}
    )
    comp.trigger_check()
    # "restart": fresh component instance over the same stores + driver cache
    import dataclasses

    from prometheus_client import CollectorRegistry

    from gpud_amd.components.accelerator import cper as cper_mod

    inst = dataclasses.replace(
        mock_core.registry.gpud_instance, metrics_registry=CollectorRegistry()
    )
    comp2 = cper_mod.new(inst)
    if comp2._bucket is not None:
        for ev in comp2._bucket.get(utcnow() - datetime.timedelta(days=7)):
            import re as _re

            m = _re.search(r"record_id=([^\s,]+)", ev.message or "")
            if m:
                comp2._seen.add(m.group(1))
    cr = comp2.trigger_check()  # cursor 0 ⇒ full replay, but dedup holds
    uuid0 = mock_core.smi_instance.device_uuids()[0]
    assert comp2._counts[uuid0][2] == 0  # corrected count not re-incremented
    assert cr.health == HealthStateType.HEALTHY


def test_partition_policy_hot_settable_via_updateconfig(mock_core):
    """Control-plane updateConfig changes the partition policy without a
    component restart (live Config read-through)."""
    from gpud_amd.session import Session

    comp = mock_core.registry.get("accelerator-amd-partition")
    assert comp.trigger_check().health == HealthStateType.HEALTHY
    s = Session(
        mock_core,
        endpoint="unused",
        open_reader=lambda: iter(()),
        send_response=lambda f: None,
    )
    resp = s.process_request(
        {
            "req_id": "u",
            "method": "updateConfig",
            "data": {"expected_compute_partition": "CPX"},
        }
    )
    assert "expected_compute_partition" in resp["data"]["applied"]
    mock_core.shared_snapshots.refresh()
    assert comp.trigger_check().health == HealthStateType.UNHEALTHY
    s.process_request(
        {
            "req_id": "u2",
            "method": "updateConfig",
            "data": {"expected_compute_partition": ""},
        }
    )
    mock_core.shared_snapshots.refresh()
    assert comp.trigger_check().health == HealthStateType.HEALTHY


def test_pcie_healthy_then_downtrained(mock_core):
    backend = mock_core.smi_instance._b
    comp = mock_core.registry.get("accelerator-amd-pcie")
    cr = comp.trigger_check()
    assert cr.health == HealthStateType.HEALTHY
    assert "PCIe link healthy" in cr.reason
    backend.state[0]["pcie_width"] = 8  # down-trained from x16
    cr = comp.trigger_check()
    assert cr.health == HealthStateType.DEGRADED
    assert "down-trained" in cr.reason and "x8" in cr.reason
    del backend.state[0]["pcie_width"]


def test_pcie_error_counter_deltas(mock_core):
    from gpud_amd.components.accelerator.pcie import EVENT_NAME

    backend = mock_core.smi_instance._b
    comp = mock_core.registry.get("accelerator-amd-pcie")
    comp.trigger_check()  # prime counters
    backend.state[1]["pcie_replay_count"] = 42
    backend.state[1]["pcie_nak_sent_count"] = 3
    cr = comp.trigger_check()
    evs = comp.events(utcnow() - datetime.timedelta(minutes=5))
    assert any(EVENT_NAME == e.name and "replay +42" in e.message for e in evs)
    # sustained rate over the window degrades
    bucket = mock_core.event_store.bucket("accelerator-amd-pcie")
    from gpud_amd.apiv1.types import Event, EventType

    now = utcnow()
    for i in range(8):
        bucket.insert(
            Event(
                time=now - datetime.timedelta(minutes=i),
                component=comp.name,
                name=EVENT_NAME,
                type=EventType.WARNING,
                message=f"synthetic {i}",
            )
        )
    comp.get_now = lambda: now
    cr = comp.trigger_check()
    assert cr.health == HealthStateType.DEGRADED
    assert "sustained PCIe" in cr.reason


def test_partition_ttl_cache(mock_core):
    """Partition reads are TTL-cached (poll-cycle cost control) but a
    zero TTL reads through — the seam tests use."""
    backend = mock_core.smi_instance._b
    comp = mock_core.registry.get("accelerator-amd-partition")
    comp.trigger_check()
    backend.state[0]["compute_partition"] = "CPX"
    # within the TTL the cached SPX answer stands
    cr = comp.trigger_check()
    assert "CPX" not in cr.reason
    # expiring the cache picks up the live mode
    comp.cache_ttl_seconds = 0.0
    comp._cache = {}
    cr = comp.trigger_check()
    assert "CPX" in cr.reason
    backend.state[0]["compute_partition"] = "SPX"
    comp._cache = {}


def test_bad_envs_detection(mock_core):
    """Dangerous global GPU env vars degrade; per-process device hiding in
    the daemon scope is tolerated (launchers set it legitimately)."""
    from gpud_amd.components.accelerator.bad_envs import scan_bad_envs

    comp = mock_core.registry.get("accelerator-amd-bad-envs")
    comp.get_scopes = lambda: {"/etc/environment": {}, "pid1": {}, "daemon": {}}
    assert comp.trigger_check().health == HealthStateType.HEALTHY

    comp.get_scopes = lambda: {
        "/etc/environment": {"AMD_SERIALIZE_KERNEL": "3"},
        "pid1": {},
        "daemon": {},
    }
    cr = comp.trigger_check()
    assert cr.health == HealthStateType.DEGRADED
    assert "AMD_SERIALIZE_KERNEL" in cr.reason

    # value-sensitive rules
    assert scan_bad_envs({"pid1": {"HSA_ENABLE_SDMA": "1"}}) == []
    assert scan_bad_envs({"pid1": {"HSA_ENABLE_SDMA": "0"}}) != []
    assert scan_bad_envs({"pid1": {"NCCL_P2P_DISABLE": "0"}}) == []
    assert scan_bad_envs({"pid1": {"NCCL_P2P_DISABLE": "1"}}) != []
    # device hiding: global scope bad, daemon scope tolerated
    assert scan_bad_envs({"pid1": {"HIP_VISIBLE_DEVICES": "0"}}) != []
    assert scan_bad_envs({"daemon": {"HIP_VISIBLE_DEVICES": "0"}}) == []


def test_bad_envs_etc_environment_parse(tmp_path):
    from gpud_amd.components.accelerator.bad_envs import read_etc_environment

    f = tmp_path / "environment"
    f.write_text(
        '# comment\nPATH="/usr/bin"\nAMD_SERIALIZE_KERNEL=3\nBROKENLINE\n'
    )
    env = read_etc_environment(str(f))
    assert env["AMD_SERIALIZE_KERNEL"] == "3"
    assert env["PATH"] == "/usr/bin"
    assert "BROKENLINE" not in env


def test_bad_envs_ttl_and_live_daemon_scope(mock_core, tmp_path, monkeypatch):
    """Global scopes are TTL-cached; the daemon's own env is always live."""
    comp = mock_core.registry.get("accelerator-amd-bad-envs")
    comp.get_scopes = comp._default_scopes  # real path
    monkeypatch.delenv("AMD_SERIALIZE_KERNEL", raising=False)
    base = comp.trigger_check()
    monkeypatch.setenv("AMD_SERIALIZE_KERNEL", "3")
    cr = comp.trigger_check()  # daemon scope is read live, no TTL wait
    assert cr.health == HealthStateType.DEGRADED
    assert "daemon" in cr.reason
    monkeypatch.delenv("AMD_SERIALIZE_KERNEL")
    assert comp.trigger_check().health == base.health


def test_diag_components_report_unhealthy_without_extension(monkeypatch, tmp_path):
    """On a GPU host (/dev/kfd present) with the HIP extension missing the
    diag components must be UNHEALTHY, never a silent eager fallback (the
    round-end 'native code not loaded' check)."""
    from gpud_amd.components.accelerator import diag as diag_mod

    def broken_load():
        raise ImportError("extension not built")

    monkeypatch.setattr(diag_mod, "_load_diag", broken_load)
    monkeypatch.setattr(diag_mod, "gpu_present", lambda: True)
    from gpud_amd.components.base import GPUdInstance

    comps = [f(GPUdInstance()) for f in diag_mod.init_funcs()]
    assert comps
    for comp in comps:
        cr = comp.trigger_check()
        assert cr.health == "Unhealthy", (comp.name, cr.reason)
        blob = (cr.reason + cr.error).lower()
        # kernels: missing extension is named; fabric: the real binary runs
        # and reports the missing device — loud either way
        assert "extension" in blob or "not built" in blob or "fabric" in blob


def test_cper_section_fru_attribution(mock_core):
    """Decoded CPER section descriptors carry section type + FRU text into
    the event message (FRU-level attribution)."""
    backend = mock_core.smi_instance._b
    comp = mock_core.registry.get("accelerator-amd-cper")
    comp.trigger_check()
    backend.state[0]["cper"].append(
        {
            "severity": 1,
            "severity_name": "fatal",
            "record_id": "bb02",
            "notify_type": "MCE",
            "section_count": 2,
            "sections": [
                {
                    "type_guid": "a5bc1114-6f64-4ede-b863-3e83ed7c83b1",
                    "type_name": "memory",
                    "severity": 1,
                    "fru_text": "OAM3 HBM stack 2",
                },
                {
                    "type_guid": "d995e954-bbc1-430f-ad91-b44dcb3c6f35",
                    "type_name": "pcie",
                    "severity": 0,
                },
            ],
        }
    )
    cr = comp.trigger_check()
    assert cr.health == HealthStateType.UNHEALTHY
    evs = comp.events(utcnow() - datetime.timedelta(minutes=5))
    msg = next(e.message for e in evs if "bb02" in e.message)
    assert "memory fru=OAM3 HBM stack 2" in msg
    assert "pcie" in msg
