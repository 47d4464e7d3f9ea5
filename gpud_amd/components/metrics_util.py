"""Prometheus gauge helpers for components.

Metric naming follows the reference convention: no
namespace, subsystem = component name with underscores (e.g.
``accelerator_amd_temperature_current_celsius``), every metric carrying the
curried ``gpud_component`` label plus ``uuid`` for per-GPU gauges
(reference: components/accelerator/nvidia/temperature/metrics.go:10-35).
"""

from __future__ import annotations

from typing import Dict, List, Optional

from prometheus_client import CollectorRegistry, Gauge

from ..pkg.metrics.registry import LABEL_COMPONENT


class ComponentGauges:
    """Lazily-created gauges bound to one component and one registry.

    Child (labelled) gauge objects are cached: prometheus_client's
    ``labels()`` re-validates and re-hashes label tuples on every call,
    which dominates a tight poll cycle with hundreds of per-GPU gauge
    updates — the cache turns each update into a dict hit + ``set()``.
    """

    def __init__(self, component: str, registry: Optional[CollectorRegistry]):
        self.component = component
        self.registry = registry
        self._gauges: Dict[str, Gauge] = {}
        self._children: Dict[tuple, object] = {}

    def gauge(self, name: str, doc: str, extra_labels: List[str] = ()) -> Optional[Gauge]:
        if self.registry is None:
            return None
        g = self._gauges.get(name)
        if g is None:
            g = Gauge(
                name,
                doc,
                [LABEL_COMPONENT, *extra_labels],
                registry=self.registry,
            )
            self._gauges[name] = g
        return g

    def set(self, name: str, doc: str, value: float, **labels: str) -> None:
        if self.registry is None:
            return
        # hot path: most gauges carry zero or one extra label (uuid)
        if len(labels) < 2:
            key = (name, *labels.items())
        else:
            key = (name, *sorted(labels.items()))
        child = self._children.get(key)
        if child is None:
            g = self.gauge(name, doc, sorted(labels.keys()))
            if g is None:
                return
            child = g.labels(**{LABEL_COMPONENT: self.component, **labels})
            self._children[key] = child
        child.set(value)
