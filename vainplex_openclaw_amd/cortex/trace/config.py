"""Trace-analyzer config: defaults + typed resolver.

Parity target: cortex `src/trace-analyzer/config.ts` —
TRACE_ANALYZER_DEFAULTS (`:98-150`) and resolveTraceAnalyzerConfig:
per-field typed resolution that falls back to the default on a wrong
type, per-signal enable/severity overrides keyed by SIG-* id (unknown
ids dropped, `:178-205`), nats / llm(+optional triage) / output
sub-resolvers, and string-filtered redactPatterns.
"""

from __future__ import annotations

import copy
from typing import Any, Dict, List, Optional

# SIG-* id -> this build's detector name (signals.DETECTORS keys)
SIGNAL_IDS = {
    "SIG-CORRECTION": "correction",
    "SIG-TOOL-FAIL": "tool_fail",
    "SIG-DOOM-LOOP": "doom_loop",
    "SIG-DISSATISFIED": "dissatisfied",
    "SIG-REPEAT-FAIL": "repeat_fail",
    "SIG-HALLUCINATION": "hallucination",
    "SIG-UNVERIFIED-CLAIM": "unverified_claim",
}

TRACE_ANALYZER_DEFAULTS: Dict[str, Any] = {
    "enabled": False,
    "nats": {
        "url": "nats://localhost:4222",
        "stream": "openclaw-events",
        "subjectPrefix": "openclaw.events",
        "credentials": "",
        "user": "",
        "password": "",
    },
    "schedule": {"enabled": False, "intervalHours": 24},
    "chainGapMinutes": 30,
    "fetchBatchSize": 500,
    # SIG-UNVERIFIED-CLAIM is the only detector off by default (noisy)
    "signals": {
        "SIG-CORRECTION": {"enabled": True, "severity": "medium"},
        "SIG-TOOL-FAIL": {"enabled": True, "severity": "medium"},
        "SIG-DOOM-LOOP": {"enabled": True, "severity": "high"},
        "SIG-DISSATISFIED": {"enabled": True, "severity": "high"},
        "SIG-REPEAT-FAIL": {"enabled": True, "severity": "medium"},
        "SIG-HALLUCINATION": {"enabled": True, "severity": "high"},
        "SIG-UNVERIFIED-CLAIM": {"enabled": False, "severity": "low"},
    },
    "llm": {
        "enabled": False,
        "endpoint": "",
        "model": "",
        "apiKey": "",
        "timeoutMs": 15000,
        "triage": None,
    },
    "redactPatterns": [],
    "output": {"maxFindings": 1000, "reportPath": ""},
}


def _bool(v: Any, default: bool) -> bool:
    return v if isinstance(v, bool) else default


def _num(v: Any, default: float) -> float:
    return v if isinstance(v, (int, float)) and not isinstance(v, bool) else default


def _str(v: Any, default: str) -> str:
    return v if isinstance(v, str) else default


def _resolve_signals(raw: Any) -> Dict[str, Dict[str, Any]]:
    out = copy.deepcopy(TRACE_ANALYZER_DEFAULTS["signals"])
    if not isinstance(raw, dict):
        return out
    for sig_id, defaults in out.items():  # unknown SIG-* ids are dropped
        override = raw.get(sig_id)
        if not isinstance(override, dict):
            continue
        defaults["enabled"] = _bool(override.get("enabled"), defaults["enabled"])
        sev = override.get("severity")
        if sev in ("low", "medium", "high", "critical"):
            defaults["severity"] = sev
    return out


def _resolve_nats(raw: Any) -> Dict[str, str]:
    d = TRACE_ANALYZER_DEFAULTS["nats"]
    raw = raw if isinstance(raw, dict) else {}
    return {key: _str(raw.get(key), d[key]) for key in d}


def _resolve_llm(raw: Any) -> Dict[str, Any]:
    d = TRACE_ANALYZER_DEFAULTS["llm"]
    raw = raw if isinstance(raw, dict) else {}
    out = {
        "enabled": _bool(raw.get("enabled"), d["enabled"]),
        "endpoint": _str(raw.get("endpoint"), d["endpoint"]),
        "model": _str(raw.get("model"), d["model"]),
        "apiKey": _str(raw.get("apiKey"), d["apiKey"]),
        "timeoutMs": _num(raw.get("timeoutMs"), d["timeoutMs"]),
        "triage": None,
    }
    triage = raw.get("triage")
    if isinstance(triage, dict):
        out["triage"] = {
            "endpoint": _str(triage.get("endpoint"), ""),
            "model": _str(triage.get("model"), ""),
            "timeoutMs": _num(triage.get("timeoutMs"), 5000),
        }
    return out


def resolve_trace_analyzer_config(raw: Optional[Dict[str, Any]] = None) -> Dict[str, Any]:
    raw = raw if isinstance(raw, dict) else {}
    d = TRACE_ANALYZER_DEFAULTS
    schedule_raw = raw.get("schedule") if isinstance(raw.get("schedule"), dict) else {}
    output_raw = raw.get("output") if isinstance(raw.get("output"), dict) else {}
    patterns = raw.get("redactPatterns")
    return {
        "enabled": _bool(raw.get("enabled"), d["enabled"]),
        "nats": _resolve_nats(raw.get("nats")),
        "schedule": {
            "enabled": _bool(schedule_raw.get("enabled"), d["schedule"]["enabled"]),
            "intervalHours": _num(schedule_raw.get("intervalHours"),
                                  d["schedule"]["intervalHours"]),
        },
        "chainGapMinutes": _num(raw.get("chainGapMinutes"), d["chainGapMinutes"]),
        "fetchBatchSize": _num(raw.get("fetchBatchSize"), d["fetchBatchSize"]),
        "signals": _resolve_signals(raw.get("signals")),
        "llm": _resolve_llm(raw.get("llm")),
        "redactPatterns": [p for p in patterns if isinstance(p, str)]
        if isinstance(patterns, list) else [],
        "output": {
            "maxFindings": _num(output_raw.get("maxFindings"),
                                d["output"]["maxFindings"]),
            "reportPath": _str(output_raw.get("reportPath"),
                               d["output"]["reportPath"]),
        },
    }


def enabled_detectors(resolved: Dict[str, Any]) -> List[str]:
    """Detector names (signals.DETECTORS keys) enabled by the config."""
    return [SIGNAL_IDS[sig] for sig, c in resolved["signals"].items()
            if c.get("enabled")]
