"""kmsg → eventstore bridge with dedup.

Reference: pkg/kmsg/syncer.go:13-49 — a match function classifies each kmsg
record; matches become events in a component's event bucket, deduplicated
within a time window (reference deduper.go:76-99).
"""

from __future__ import annotations

import datetime
import threading
from dataclasses import dataclass
from typing import Callable, Dict, Optional, Tuple

from ...apiv1.types import Event
from ..eventstore import Bucket
from ..log import logger
from .parser import Message
from .watcher import Watcher

DEFAULT_DEDUP_WINDOW = datetime.timedelta(minutes=5)


@dataclass
class MatchResult:
    """What a match function returns for a recognised kernel message."""

    name: str
    event_type: str  # apiv1 EventType value
    message: str
    extra_info: Optional[Dict[str, str]] = None


MatchFunc = Callable[[str], Optional[MatchResult]]


class Syncer:
    def __init__(
        self,
        watcher: Watcher,
        match_fn: MatchFunc,
        bucket: Bucket,
        dedup_window: datetime.timedelta = DEFAULT_DEDUP_WINDOW,
    ):
        self._watcher = watcher
        self._match = match_fn
        self._bucket = bucket
        self._window = dedup_window
        self._lock = threading.Lock()
        # (name, message) -> last insert time
        self._seen: Dict[Tuple[str, str], datetime.datetime] = {}
        watcher.register(self._on_message)

    def _on_message(self, m: Message) -> None:
        try:
            res = self._match(m.message)
        except Exception:
            logger.exception("kmsg match function failed")
            return
        if res is None:
            return
        ts = m.time
        key = (res.name, res.message)
        with self._lock:
            last = self._seen.get(key)
            if last is not None and ts is not None and ts - last < self._window:
                return
            if ts is not None:
                self._seen[key] = ts
            # bound the dedup cache
            if len(self._seen) > 4096:
                cutoff = ts - self._window if ts else None
                if cutoff:
                    self._seen = {
                        k: v for k, v in self._seen.items() if v >= cutoff
                    }
        self._bucket.insert(
            Event(
                time=ts or datetime.datetime.now(datetime.timezone.utc),
                component=self._bucket.component_name,
                name=res.name,
                type=res.event_type,
                message=res.message,
            ),
            extra_info=res.extra_info,
        )

    def replay(self, messages) -> int:
        """Feed historical messages (from Watcher.read_all) through the matcher."""
        n = 0
        for m in messages:
            before = len(self._seen)
            self._on_message(m)
            if len(self._seen) != before:
                n += 1
        return n
