"""Scrape the Prometheus registry into record rows.

Reference: pkg/metrics/scraper/prometheus.go:28 — gather all metric families
from the custom registry and keep only samples carrying the
``gpud_component`` label; that label becomes the component column and is
stripped from the persisted label set.
"""

from __future__ import annotations

import time
from dataclasses import dataclass, field
from typing import Dict, List

from prometheus_client import CollectorRegistry

from .registry import LABEL_COMPONENT


@dataclass
class ScrapedMetric:
    unix_ms: int
    component: str
    name: str
    labels: Dict[str, str] = field(default_factory=dict)
    value: float = 0.0


class Scraper:
    def __init__(self, registry: CollectorRegistry):
        self._registry = registry

    def scrape(self) -> List[ScrapedMetric]:
        now_ms = int(time.time() * 1000)
        out: List[ScrapedMetric] = []
        for family in self._registry.collect():
            for sample in family.samples:
                labels = dict(sample.labels)
                component = labels.pop(LABEL_COMPONENT, None)
                if component is None:
                    continue
                # skip created-series noise (histogram/gauge _created samples)
                if sample.name.endswith("_created"):
                    continue
                out.append(
                    ScrapedMetric(
                        unix_ms=now_ms,
                        component=component,
                        name=sample.name,
                        labels=labels,
                        value=float(sample.value),
                    )
                )
        return out
