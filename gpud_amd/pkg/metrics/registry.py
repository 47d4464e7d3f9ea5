"""Prometheus custom registry (reference: pkg/metrics/registry.go:5-24).

Every gpud metric carries the ``gpud_component`` label (reference:
pkg/metrics/types.go:9); the scraper keeps only metrics having that label so
third-party collectors never leak into ``/v1/metrics``.
"""

from __future__ import annotations

from prometheus_client import CollectorRegistry

LABEL_COMPONENT = "gpud_component"


def create_registry() -> CollectorRegistry:
    """A fresh custom registry — never the process-global default one."""
    return CollectorRegistry()
