"""Bridge: cortex trace-analyzer findings -> governance fact registry.

Parity target: governance `src/trace-to-facts-bridge.ts` (211 LoC) —
converts trace-analysis report findings (hallucination / unverified-claim
signals carrying subject-predicate-value payloads) into FactRegistry
entries so repeated hallucinations become checkable contradictions.
"""

from __future__ import annotations

from typing import Any, Dict, List, Optional

from .facts import FactRegistry

FACT_SIGNAL_TYPES = {"hallucination", "unverified_claim", "unverified-claim"}


def findings_to_facts(report: Dict[str, Any], min_confidence: float = 0.5) -> List[Dict[str, Any]]:
    """Extract SPO facts from an AnalysisReport's findings."""
    facts: List[Dict[str, Any]] = []
    for finding in report.get("findings", []):
        if str(finding.get("signalType", "")) not in FACT_SIGNAL_TYPES:
            continue
        if float(finding.get("confidence", 0)) < min_confidence:
            continue
        payload = finding.get("payload") or finding.get("evidence") or {}
        subject = payload.get("subject")
        predicate = payload.get("predicate")
        value = payload.get("actualValue") or payload.get("value")
        if subject and predicate and value is not None:
            facts.append(
                {
                    "subject": str(subject),
                    "predicate": str(predicate),
                    "value": str(value),
                    "source": f"trace-analyzer:{finding.get('id', 'unknown')}",
                }
            )
    return facts


def apply_report_to_registry(
    report: Dict[str, Any], registry: FactRegistry, min_confidence: float = 0.5
) -> int:
    facts = findings_to_facts(report, min_confidence)
    for f in facts:
        registry.add_fact(f)
    return len(facts)


# -- file-level bridge (trace-to-facts-bridge.ts:60-211) --------------------

import json
import os
import threading


def extract_facts_from_file(path: str, min_confidence: float = 0.5) -> List[Dict[str, Any]]:
    """Read one trace report file -> facts. Accepts the full report
    shape, a bare findings array, or {findings: [...]}; missing files and
    invalid JSON return [] (the bridge never raises)."""
    try:
        with open(path, "r", encoding="utf-8") as fh:
            data = json.load(fh)
    except (OSError, json.JSONDecodeError, ValueError):
        return []
    if isinstance(data, list):
        data = {"findings": data}
    if not isinstance(data, dict) or not isinstance(data.get("findings"), list):
        return []
    out = []
    for f in findings_to_facts(data, min_confidence):
        if not f.get("predicate"):
            f["predicate"] = "value"  # default predicate
        out.append(f)
    return out


def _fact_key(f: Dict[str, Any]) -> str:
    return f"{str(f.get('subject', '')).lower()}|{str(f.get('predicate', '')).lower()}"


def merge_facts(existing: List[Dict[str, Any]], new: List[Dict[str, Any]]) -> List[Dict[str, Any]]:
    """Dedupe by subject|predicate, case-insensitive, LATER wins (new
    facts override existing, and later entries within each list win)."""
    merged: Dict[str, Dict[str, Any]] = {}
    for f in list(existing) + list(new):
        if isinstance(f, dict) and f.get("subject"):
            merged[_fact_key(f)] = f
    return list(merged.values())


def run_bridge(report_paths: List[str], out_path: str,
               min_confidence: float = 0.5) -> int:
    """Extract facts from every report file, merge with the existing
    output file (if any), write atomically; returns the fact count in
    the output (existing count when nothing new was extracted)."""
    new: List[Dict[str, Any]] = []
    for p in report_paths:
        new.extend(extract_facts_from_file(p, min_confidence))
    existing: List[Dict[str, Any]] = []
    try:
        with open(out_path, "r", encoding="utf-8") as fh:
            data = json.load(fh)
        if isinstance(data, dict) and isinstance(data.get("facts"), list):
            existing = [f for f in data["facts"] if isinstance(f, dict)]
        elif isinstance(data, list):
            existing = [f for f in data if isinstance(f, dict)]
    except (OSError, json.JSONDecodeError, ValueError):
        pass
    merged = merge_facts(existing, new)
    os.makedirs(os.path.dirname(os.path.abspath(out_path)), exist_ok=True)
    tmp = out_path + ".tmp"
    with open(tmp, "w", encoding="utf-8") as fh:
        json.dump({"facts": merged}, fh, indent=2)
    os.replace(tmp, out_path)
    return len(merged)


class BridgeTimer:
    """Interval runner for the bridge (start/stop idempotent)."""

    def __init__(self, interval_s: float, fn):
        self.interval_s = interval_s
        self.fn = fn
        self._timer: Optional[threading.Timer] = None
        self._running = False

    def start(self) -> None:
        if self._running:
            return
        self._running = True
        self._schedule()

    def _schedule(self) -> None:
        if not self._running:
            return
        self._timer = threading.Timer(self.interval_s, self._tick)
        self._timer.daemon = True
        self._timer.start()

    def _tick(self) -> None:
        try:
            self.fn()
        except Exception:
            pass
        self._schedule()

    def stop(self) -> None:
        self._running = False
        if self._timer is not None:
            self._timer.cancel()
            self._timer = None
