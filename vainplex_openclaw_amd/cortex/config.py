"""Top-level cortex plugin config: defaults + resolver.

Parity target: cortex `src/config.ts` — DEFAULTS (threadTracker /
decisionTracker / commitmentTracker / bootContext / llm / traceAnalyzer
sections) and resolveConfig(raw): section-wise defaults resolution; the
traceAnalyzer section delegates to its own typed resolver
(trace-analyzer/config.ts).
"""

from __future__ import annotations

import copy
from typing import Any, Dict, Optional

from ..core.config import resolve_defaults
from .trace.config import TRACE_ANALYZER_DEFAULTS, resolve_trace_analyzer_config

DEFAULTS: Dict[str, Any] = {
    "language": "both",
    "threadTracker": {"enabled": True, "pruneDays": 14, "maxThreads": 50},
    "decisionTracker": {"enabled": True},
    "commitmentTracker": {"enabled": True},
    "bootContext": {"enabled": True},
    "llm": {"enabled": False},
    "traceAnalyzer": copy.deepcopy(TRACE_ANALYZER_DEFAULTS),
}


def resolve_config(raw: Optional[Dict[str, Any]] = None) -> Dict[str, Any]:
    raw = raw if isinstance(raw, dict) else {}
    out = resolve_defaults(raw, DEFAULTS)
    out["traceAnalyzer"] = resolve_trace_analyzer_config(raw.get("traceAnalyzer"))
    return out
