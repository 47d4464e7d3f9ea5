"""nfs — NFS member-visibility checker + client-health kernel events.

Reference: components/nfs + pkg/nfs-checker (write/read a per-machine file
in each configured NFS group dir, verify member visibility; configs pushed
from the control plane — nfs/component.go:31). NFS client kernel
messages land in this component's bucket (reference:
nfs/kmsg_matcher.go — server not responding / recovered, lock-reclaim
failure, writeback-path hang stack frames).
"""

from __future__ import annotations

import os
import re
from dataclasses import dataclass
from typing import List

from ...apiv1.types import HealthStateType
from ...pkg import host as pkghost
from ...pkg.kmsg.syncer import MatchResult, Syncer
from ..base import CheckResult, Component, GPUdInstance, TickerComponent

NAME = "nfs"

# kmsg matcher (reference: components/nfs/kmsg_matcher.go)
_KMSG_RULES = (
    ("nfs_server_not_responding",
     re.compile(r"nfs: server (?P<server>\S+) not responding"), "Warning"),
    ("nfs_server_ok",
     re.compile(r"nfs: server (?P<server>\S+) OK"), "Info"),
    ("nfs_lock_reclaim_failed",
     re.compile(r"nfs4_reclaim_open_state: Lock reclaim failed"),
     "Warning"),
    ("nfs_writeback_hang",
     re.compile(r"(?:^|\s)(?:nfs_lock_and_join_requests|nfs_wb_all|"
                r"nfs_page_async_flush|nfs_writepages_callback)"
                r"\+0x[0-9a-f]+"),
     "Critical"),
)


def collect_nfs_hang_events(events):
    """(hang_events, reason) from a window of this component's kmsg
    events: every lock-reclaim failure, every writeback-path hang trace,
    and every server whose LAST response event is still not-responding
    (a later "server OK" resolves it) — reference:
    nfs/hang_evaluator.go collectNFSHangEvents."""
    lock = [e for e in events if e.name == "nfs_lock_reclaim_failed"]
    wb = [e for e in events if e.name == "nfs_writeback_hang"]
    by_server = {}
    server_rx = re.compile(r"nfs: server (\S+)")
    for e in events:
        if e.name in ("nfs_server_not_responding", "nfs_server_ok"):
            m = server_rx.search(e.message or "")
            server = m.group(1) if m else "?"
            by_server.setdefault(server, []).append(e)
    hang = list(lock) + list(wb)
    parts = []
    if lock:
        parts.append(f"{len(lock)} lock reclaim failure(s)")
    unresolved = []
    for server, evs in sorted(by_server.items()):
        evs_sorted = sorted(evs, key=lambda e: e.time)
        if evs_sorted and evs_sorted[-1].name == "nfs_server_not_responding":
            unresolved.append(server)
            hang.extend(e for e in evs_sorted
                        if e.name == "nfs_server_not_responding")
    if unresolved:
        parts.append("server(s) not responding: " + ", ".join(unresolved))
    if wb:
        parts.append(f"{len(wb)} writeback-path hang trace(s)")
    return hang, "; ".join(parts)


def match_nfs_kmsg(line):
    for name, rx, event_type in _KMSG_RULES:
        m = rx.search(line)
        if m:
            extra = {k: v for k, v in m.groupdict().items() if v}
            return MatchResult(name=name, event_type=event_type,
                               message=line, extra_info=extra or None)
    return None


@dataclass
class GroupConfig:
    """One NFS group dir every member machine writes its marker into."""

    volume_path: str
    dir_name: str = ".gpud-nfs-checker"
    file_contents: str = ""


class NFSComponent(TickerComponent):
    def __init__(self, inst: GPUdInstance):
        super().__init__()
        self._bucket = (
            inst.event_store.bucket(NAME)
            if inst.event_store is not None else None
        )
        self._kmsg = inst.kmsg_reader
        self._syncer = None
        self.configs: List[GroupConfig] = [
            c if isinstance(c, GroupConfig) else GroupConfig(**c)
            for c in (inst.nfs_checker_configs or [])
        ]
        self.machine_id = pkghost.machine_id() or pkghost.hostname()

    @property
    def name(self) -> str:
        return NAME

    def start(self) -> None:
        if self._kmsg is not None and self._bucket is not None:
            self._syncer = Syncer(self._kmsg, match_nfs_kmsg, self._bucket)
        super().start()

    def events(self, since):
        return self._bucket.get(since) if self._bucket is not None else []

    def tags(self) -> list:
        return [NAME]

    def set_configs(self, configs: List[GroupConfig]) -> None:
        """Control-plane push path (reference: configs pushed via session)."""
        self.configs = configs

    def check(self) -> CheckResult:
        if not self.configs:
            return CheckResult(NAME, reason="no NFS group configs")
        problems = []
        extra = {}
        for cfg in self.configs:
            d = os.path.join(cfg.volume_path, cfg.dir_name)
            marker = os.path.join(d, self.machine_id)
            try:
                os.makedirs(d, exist_ok=True)
                contents = cfg.file_contents or self.machine_id
                with open(marker, "w") as f:
                    f.write(contents)
                with open(marker) as f:
                    back = f.read()
                if back != contents:
                    problems.append(f"{cfg.volume_path}: read-back mismatch")
                    continue
                members = [
                    n for n in os.listdir(d) if not n.startswith(".")
                ]
                extra[cfg.volume_path] = f"{len(members)} member(s) visible"
            except OSError as e:
                problems.append(f"{cfg.volume_path}: {e}")
        if problems:
            return CheckResult(
                NAME,
                health=HealthStateType.UNHEALTHY,
                reason="; ".join(problems),
                extra_info=extra,
            )
        if self._bucket is not None:
            import datetime as _dt

            window = _dt.datetime.now(_dt.timezone.utc) - _dt.timedelta(
                minutes=10)
            hang, hang_reason = collect_nfs_hang_events(
                self._bucket.get(window))
            if hang:
                return CheckResult(
                    NAME,
                    health=HealthStateType.DEGRADED,
                    reason="NFS client hang indicators: " + hang_reason,
                    extra_info=extra,
                )
        return CheckResult(
            NAME,
            reason=f"all {len(self.configs)} NFS group dir(s) writable and readable",
            extra_info=extra,
        )


def new(inst: GPUdInstance) -> Component:
    return NFSComponent(inst)
