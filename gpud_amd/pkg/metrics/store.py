"""SQLite metrics store.

Schema-compatible with the reference table (reference:
pkg/metrics/store/sqlite.go:47-106): ``gpud_metrics_v0_5`` with primary key
``(unix_milliseconds, component_name, metric_name, metric_labels)`` WITHOUT
ROWID; labels persisted as sorted-key JSON.
"""

from __future__ import annotations

import datetime
import json
from typing import Dict, List, Optional

from ...apiv1.types import Metric
from ..sqlite_util import Conn
from .scraper import ScrapedMetric

SCHEMA_VERSION = "v0_5"
DEFAULT_TABLE_NAME = f"gpud_metrics_{SCHEMA_VERSION}"


class MetricsStore:
    def __init__(self, db_rw: Conn, db_ro: Conn, table: str = DEFAULT_TABLE_NAME):
        self._db_rw = db_rw
        self._db_ro = db_ro
        self.table = table
        self._create_table()

    def _create_table(self) -> None:
        self._db_rw.executescript(
            f"""
CREATE TABLE IF NOT EXISTS {self.table} (
    unix_milliseconds INTEGER NOT NULL,
    component_name TEXT NOT NULL,
    metric_name TEXT NOT NULL,
    metric_labels TEXT,
    metric_value REAL NOT NULL,
    PRIMARY KEY (unix_milliseconds, component_name, metric_name, metric_labels)
) WITHOUT ROWID;
"""
        )

    def record(self, metrics: List[ScrapedMetric]) -> None:
        if not metrics:
            return
        self._db_rw.executemany(
            f"INSERT OR REPLACE INTO {self.table}"
            " (unix_milliseconds, component_name, metric_name, metric_labels,"
            " metric_value) VALUES (?, ?, ?, ?, ?)",
            [
                (
                    m.unix_ms,
                    m.component,
                    m.name,
                    json.dumps(m.labels, sort_keys=True) if m.labels else "",
                    m.value,
                )
                for m in metrics
            ],
        )

    def read(
        self,
        since: Optional[datetime.datetime] = None,
        components: Optional[List[str]] = None,
    ) -> Dict[str, List[Metric]]:
        """Historical metrics grouped by component (serves /v1/metrics)."""
        sql = (
            f"SELECT unix_milliseconds, component_name, metric_name,"
            f" metric_labels, metric_value FROM {self.table}"
        )
        where, params = [], []
        if since is not None:
            where.append("unix_milliseconds >= ?")
            params.append(int(since.timestamp() * 1000))
        if components:
            where.append(
                "component_name IN (%s)" % ",".join("?" * len(components))
            )
            params.extend(components)
        if where:
            sql += " WHERE " + " AND ".join(where)
        sql += " ORDER BY unix_milliseconds ASC"
        out: Dict[str, List[Metric]] = {}
        for unix_ms, comp, name, labels_json, value in self._db_ro.query(
            sql, params
        ):
            labels = json.loads(labels_json) if labels_json else None
            out.setdefault(comp, []).append(
                Metric(
                    unix_seconds=unix_ms // 1000,
                    name=name,
                    labels=labels,
                    value=value,
                )
            )
        return out

    def purge(self, before: datetime.datetime) -> int:
        cutoff = int(before.timestamp() * 1000)
        row = self._db_ro.query_one(
            f"SELECT COUNT(*) FROM {self.table} WHERE unix_milliseconds < ?",
            (cutoff,),
        )
        self._db_rw.execute(
            f"DELETE FROM {self.table} WHERE unix_milliseconds < ?", (cutoff,)
        )
        return int(row[0]) if row else 0
