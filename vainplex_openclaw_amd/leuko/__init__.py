"""Leuko: cognitive immune system — health checks, anomaly detection,
situation reports (capability rebuild of the external Leuko plugin with
the in-repo Sitrep package as concrete spec — SURVEY.md §2.5/§2.7).
"""

from .aggregator import generate_sitrep, write_sitrep
from .anomaly import AnomalyDetector, MetricHistory
from .collectors import BUILT_IN_COLLECTORS, run_custom_collector, safe_collect
from .plugin import LeukoPlugin, create_plugin, default_config, resolve_config

__all__ = [
    "generate_sitrep",
    "write_sitrep",
    "AnomalyDetector",
    "MetricHistory",
    "BUILT_IN_COLLECTORS",
    "run_custom_collector",
    "safe_collect",
    "LeukoPlugin",
    "create_plugin",
    "default_config",
    "resolve_config",
]
