"""Integration story: a GPU degrades over its lifecycle and the daemon
tracks every stage — CE creep, UE, kernel RAS event, reboot, recurrence
escalation, manual clear. Exercises ecc + error-ras + event store + reboot
store + kmsg replay together (the flagship detection pipeline)."""

import datetime

import pytest

from gpud_amd.apiv1.types import Event, EventType, HealthStateType, RepairActionType, utcnow
from gpud_amd.pkg.host import EVENT_NAME_REBOOT, REBOOT_BUCKET
from gpud_amd.pkg.kmsg.parser import Message


@pytest.fixture()
def core(monkeypatch, tmp_path):
    monkeypatch.setenv("GPUD_AMDSMI_MOCK", "1")
    monkeypatch.setenv("GPUD_AMDSMI_MOCK_GPUS", "2")
    from gpud_amd.bootstrap import build_core
    from gpud_amd.pkg.config import Config

    c = build_core(
        Config(data_dir=str(tmp_path)),
        in_memory_db=True,
        kmsg_writable=False,
        record_reboot=False,
    )
    yield c
    c.close()


def test_gpu_failure_lifecycle(core):
    backend = core.smi_instance._b
    uuid = core.smi_instance.device_uuids()[0]
    ecc = core.registry.get("accelerator-amd-ecc")
    ras = core.registry.get("accelerator-amd-error-ras")
    os_bucket = core.event_store.bucket(REBOOT_BUCKET)
    t0 = utcnow() - datetime.timedelta(hours=12)

    # stage 0: healthy
    core.shared_snapshots.refresh()
    assert ecc.trigger_check().health == HealthStateType.HEALTHY
    assert ras.trigger_check().health == HealthStateType.HEALTHY

    # stage 1: correctable errors creep up -> still healthy, but an event
    backend.state[0]["ecc_correctable"] = 3
    core.shared_snapshots.refresh()
    cr = ecc.trigger_check()
    assert cr.health == HealthStateType.HEALTHY
    backend.state[0]["ecc_correctable"] = 9
    core.shared_snapshots.refresh()
    ecc.trigger_check()
    evs = ecc.events(t0)
    assert any(e.name == "amd_ecc_correctable_increase" for e in evs)

    # stage 2: the kernel logs a RAS UE (replayed through the kmsg matcher
    # exactly as the live follower would deliver it)
    from gpud_amd.components.accelerator.error_ras import _match
    from gpud_amd.pkg.kmsg.syncer import Syncer
    from gpud_amd.pkg.kmsg.watcher import Watcher

    ras_bucket = core.event_store.bucket("accelerator-amd-error-ras")
    syncer = Syncer(Watcher(path="/nonexistent"), _match, ras_bucket)
    syncer.replay(
        [
            Message(
                message="amdgpu 0000:0a:00.0: amdgpu: uncorrectable hardware "
                "error(ERREVENT_ATHUB_INTERRUPT) detected!",
                time=utcnow() - datetime.timedelta(hours=10),
            )
        ]
    )
    cr = ras.trigger_check()
    assert cr.health == HealthStateType.UNHEALTHY
    assert RepairActionType.REBOOT_SYSTEM in cr.suggested_actions.repair_actions

    # stage 3: UE also visible in the SMI counters
    backend.state[0]["ecc_uncorrectable"] = 2
    core.shared_snapshots.refresh()
    cr = ecc.trigger_check()
    assert cr.health == HealthStateType.UNHEALTHY
    assert uuid in cr.reason

    # stage 4: operator reboots -> RAS state machine clears
    os_bucket.insert(
        Event(
            time=utcnow() - datetime.timedelta(hours=8),
            component="os",
            name=EVENT_NAME_REBOOT,
            type=EventType.WARNING,
            message="reboot",
        )
    )
    assert ras.trigger_check().health == HealthStateType.HEALTHY

    # stage 5: the UE comes BACK after a second reboot -> escalation to
    # hardware inspection (reference xid/health_state.go:61-97 semantics)
    os_bucket.insert(
        Event(
            time=utcnow() - datetime.timedelta(hours=4),
            component="os",
            name=EVENT_NAME_REBOOT,
            type=EventType.WARNING,
            message="reboot",
        )
    )
    syncer.replay(
        [
            Message(
                message="amdgpu 0000:0a:00.0: amdgpu: uncorrectable hardware "
                "error(ERREVENT_ATHUB_INTERRUPT) detected!",
                time=utcnow(),
            )
        ]
    )
    cr = ras.trigger_check()
    assert cr.health == HealthStateType.UNHEALTHY
    assert cr.suggested_actions.repair_actions == [
        RepairActionType.HARDWARE_INSPECTION
    ]

    # stage 6: after the board swap, the operator clears the state
    ras.set_healthy()
    assert ras.last_check_result().health == HealthStateType.HEALTHY


def test_per_event_escalation_threshold(monkeypatch, tmp_path):
    """Per-event-name reboot thresholds override the global one
    (reference: per-Xid thresholds, xid/threshold.go)."""
    import datetime

    monkeypatch.setenv("GPUD_AMDSMI_MOCK", "1")
    from gpud_amd.apiv1.types import Event, EventType, RepairActionType, utcnow
    from gpud_amd.bootstrap import build_core
    from gpud_amd.pkg.config import Config

    cfg = Config(
        data_dir=str(tmp_path),
        ras_event_thresholds={"amdgpu_ring_timeout": 1},
    )
    core = build_core(cfg, in_memory_db=True, kmsg_writable=False, record_reboot=False)
    try:
        comp = core.registry.get("accelerator-amd-error-ras")
        bucket = core.event_store.bucket("accelerator-amd-error-ras")
        now = utcnow()
        # event first seen 2h ago, ONE reboot since; global threshold (2)
        # would still suggest reboot, but the override escalates at 1
        for dt in (120, 5):
            bucket.insert(
                Event(
                    time=now - datetime.timedelta(minutes=dt),
                    component=comp.name,
                    name="amdgpu_ring_timeout",
                    type=EventType.CRITICAL,
                    message="ring gfx_0.0.0 timeout",
                )
            )
        core.reboot_event_store._bucket.insert(
            Event(
                time=now - datetime.timedelta(minutes=60),
                component="os",
                name="reboot",
                type=EventType.WARNING,
                message="reboot detected",
            )
        )
        comp.get_now = lambda: now
        cr = comp.trigger_check()
        assert cr.health == "Unhealthy"
        assert cr.suggested_actions.repair_actions == [
            RepairActionType.HARDWARE_INSPECTION
        ], cr.suggested_actions
    finally:
        core.close()


def test_disk_and_nfs_kmsg_through_file_seam(tmp_path, monkeypatch):
    """Injected disk/NFS kernel lines travel the full seam path —
    FileSeamWriter -> watcher poll-follow -> per-component syncer ->
    component event bucket -> component events() — proving the new
    per-component matchers are wired into the daemon, not just unit-level
    (reference: disk/nfs kmsg_matcher wiring in their components)."""
    import time

    monkeypatch.setenv("GPUD_AMDSMI_MOCK", "1")
    from gpud_amd.apiv1.types import utcnow
    from gpud_amd.bootstrap import build_core
    from gpud_amd.pkg.config import Config
    from gpud_amd.pkg.fault_injector import KernelMessage, Request

    seam = tmp_path / "kmsg-seam"
    seam.write_text("")
    cfg = Config(data_dir=str(tmp_path / "data"), kmsg_path=str(seam))
    core = build_core(cfg, in_memory_db=True)
    try:
        disk = core.registry.get("disk")
        nfs = core.registry.get("nfs")
        disk.start()
        nfs.start()
        core.kmsg_watcher.start(from_start=True)
        since = utcnow() - datetime.timedelta(minutes=1)
        for msg in (
            "Buffer I/O error on dev sda1, logical block 77, lost async "
            "page write",
            "nfs: server fileserver01 not responding, still trying",
        ):
            err = core.fault_injector.inject(
                Request(kernel_message=KernelMessage(message=msg, priority=2))
            )
            assert err is None
        deadline = time.time() + 10
        got_disk = got_nfs = False
        while time.time() < deadline and not (got_disk and got_nfs):
            got_disk = any(
                e.name == "buffer_io_error" for e in disk.events(since))
            got_nfs = any(
                e.name == "nfs_server_not_responding"
                for e in nfs.events(since))
            time.sleep(0.2)
        assert got_disk, "disk kmsg event did not reach the disk bucket"
        assert got_nfs, "nfs kmsg event did not reach the nfs bucket"
    finally:
        core.close()


def test_full_daemon_restart_state_survival(tmp_path, monkeypatch):
    """Checkpoint/resume end-to-end: a daemon core with a FILE-backed
    state DB accumulates RAS events, link history, metadata and a
    set-healthy tombstone; a freshly-built core on the same data dir
    sees all of it (reference: single gpud.state SQLite — SURVEY §5
    checkpoint/resume row)."""
    monkeypatch.setenv("GPUD_AMDSMI_MOCK", "1")
    from gpud_amd.apiv1.types import Event, utcnow
    from gpud_amd.bootstrap import build_core
    from gpud_amd.pkg import metadata
    from gpud_amd.pkg.config import Config

    data = tmp_path / "data"
    cfg = Config(data_dir=str(data))
    now = utcnow()

    core1 = build_core(cfg, kmsg_writable=False, record_reboot=False)
    try:
        bucket = core1.event_store.bucket("accelerator-amd-error-ras",
                                          disable_purge=True)
        bucket.insert(Event(time=now, name="amdgpu_ring_timeout",
                            type="Critical", message="ring gfx timeout"))
        metadata.set_value(core1.db_rw, "machine_id", "m-123")
        xgmi = core1.registry.get("accelerator-amd-xgmi")
        if getattr(xgmi, "_store", None) is not None:
            xgmi._store.set_tombstone(now.timestamp())
    finally:
        core1.close()

    core2 = build_core(cfg, kmsg_writable=False, record_reboot=False)
    try:
        bucket = core2.event_store.bucket("accelerator-amd-error-ras",
                                          disable_purge=True)
        evs = bucket.get(now - datetime.timedelta(minutes=5))
        assert any(e.name == "amdgpu_ring_timeout" for e in evs)
        assert metadata.get_value(core2.db_ro, "machine_id") == "m-123"
        xgmi = core2.registry.get("accelerator-amd-xgmi")
        if getattr(xgmi, "_store", None) is not None:
            assert xgmi._store.get_tombstone() >= int(now.timestamp())
    finally:
        core2.close()
