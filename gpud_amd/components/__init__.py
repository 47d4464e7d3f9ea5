from .base import (
    DEFAULT_POLL_INTERVAL_SECONDS,
    CheckResult,
    Component,
    GPUdInstance,
    Registry,
    TickerComponent,
)

__all__ = [
    "DEFAULT_POLL_INTERVAL_SECONDS",
    "CheckResult",
    "Component",
    "GPUdInstance",
    "Registry",
    "TickerComponent",
]
