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
