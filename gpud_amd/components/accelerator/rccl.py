"""accelerator-amd-rccl — RCCL crash-signature monitor (monitor-only).

Reference: components/accelerator/nvidia/nccl — monitor-only kmsg regex
``segfault at .* in libnccl.so`` → event, no collectives ever issued
(nccl/kmsg_matcher.go:12-48). AMD-first: the signature is librccl; the
active RCCL fabric diagnostic lives in the separate manual-run diag
component (components/accelerator/diag_fabric.py).
"""

from __future__ import annotations

import datetime
import re
from typing import Optional

from ...apiv1.types import EventType, HealthStateType
from ...pkg.kmsg.syncer import MatchResult, Syncer
from ..base import CheckResult, Component, GPUdInstance, TickerComponent

NAME = "accelerator-amd-rccl"
EVENT_NAME = "amd_rccl_segfault_in_librccl"

_RE = re.compile(r"segfault at .* in librccl\.so")


def match_rccl(line: str) -> Optional[MatchResult]:
    if _RE.search(line):
        return MatchResult(
            name=EVENT_NAME, event_type=EventType.WARNING, message=line
        )
    return None


class RCCLComponent(TickerComponent):
    def __init__(self, inst: GPUdInstance):
        super().__init__()
        self._bucket = (
            inst.event_store.bucket(NAME) if inst.event_store is not None else None
        )
        self._kmsg = inst.kmsg_reader
        self._syncer: Optional[Syncer] = None

    @property
    def name(self) -> str:
        return NAME

    def tags(self) -> list:
        return ["accelerator", "amd", "gpu", NAME]

    def start(self) -> None:
        if self._kmsg is not None and self._bucket is not None:
            self._syncer = Syncer(self._kmsg, match_rccl, self._bucket)
        super().start()

    def events(self, since: datetime.datetime):
        return self._bucket.get(since) if self._bucket is not None else []

    def check(self) -> CheckResult:
        if self._bucket is None:
            return CheckResult(NAME, reason="no event store; kmsg matching disabled")
        since = datetime.datetime.now(datetime.timezone.utc) - datetime.timedelta(
            days=3
        )
        recent = self._bucket.find_by_name_since(EVENT_NAME, since)
        if recent:
            return CheckResult(
                NAME,
                health=HealthStateType.HEALTHY,  # monitor-only, like the reference
                reason=f"{len(recent)} librccl segfault(s) in the last 3 days",
                extra_info={"latest": recent[0].message[:200]},
            )
        return CheckResult(NAME, reason="no librccl crash signatures observed")


def new(inst: GPUdInstance) -> Component:
    return RCCLComponent(inst)
