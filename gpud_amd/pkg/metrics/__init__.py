from .registry import create_registry, LABEL_COMPONENT
from .scraper import Scraper, ScrapedMetric
from .store import MetricsStore, DEFAULT_TABLE_NAME
from .syncer import Syncer
from .recorder import Recorder

__all__ = [
    "create_registry",
    "LABEL_COMPONENT",
    "Scraper",
    "ScrapedMetric",
    "MetricsStore",
    "DEFAULT_TABLE_NAME",
    "Syncer",
    "Recorder",
]
