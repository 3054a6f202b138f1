"""Fact checking — Stage 2 of output validation.

Parity target: governance `src/fact-checker.ts` — FactRegistry
(Map "subject|predicate" -> Fact, subject index, inline or file-loaded
facts `:67-123`); claim-type -> predicate strategy table (`:130-136`);
verdicts verified / contradicted / unverified; fuzzy numeric matching
("255908 items" == "255908", `:207-230`); boolean normalization
yes/no/1/0 -> true/false.
"""

from __future__ import annotations

import re
from typing import Any, Dict, List, Optional

from ..utils.storage import read_json

Fact = Dict[str, Any]
Claim = Dict[str, Any]

CLAIM_TO_FACT_PREDICATE: Dict[str, Optional[Any]] = {
    "system_state": "state",
    "existence": "exists",
    "entity_name": None,  # match by subject, any predicate
    "operational_status": ["count", "metric", "percentage"],
    "self_referential": None,  # match by "self" subject
}

_NUM_RX = re.compile(r"^[\d,]+(\.\d+)?")


def _make_key(subject: str, predicate: str) -> str:
    return f"{subject.lower()}|{predicate.lower()}"


def _normalize_value(v: str) -> str:
    t = str(v).strip().lower()
    if t in ("yes", "1"):
        return "true"
    if t in ("no", "0"):
        return "false"
    return t


def _extract_number(v: str) -> Optional[float]:
    m = _NUM_RX.match(str(v).strip())
    if not m:
        return None
    try:
        return float(m.group(0).replace(",", ""))
    except ValueError:
        return None


def values_match(claim_value: str, fact_value: str) -> bool:
    return _normalize_value(claim_value) == _normalize_value(fact_value)


def values_match_fuzzy(claim_value: str, fact_value: str) -> bool:
    if values_match(claim_value, fact_value):
        return True
    cn = _extract_number(claim_value)
    fn = _extract_number(fact_value)
    if cn is not None and fn is not None:
        return cn == fn
    return False


def _load_facts_from_file(path: str) -> List[Fact]:
    data = read_json(path)
    if isinstance(data, dict) and isinstance(data.get("facts"), list):
        return [f for f in data["facts"] if isinstance(f, dict)]
    return []


class FactRegistry:
    """O(1) subject|predicate lookup; later registries override earlier
    (fact-checker.ts:67-123)."""

    def __init__(self, configs: Optional[List[Dict[str, Any]]] = None):
        self._index: Dict[str, Fact] = {}
        self._subject_index: Dict[str, List[Fact]] = {}
        for cfg in configs or []:
            facts = (
                _load_facts_from_file(cfg["filePath"])
                if cfg.get("filePath")
                else cfg.get("facts") or []
            )
            for fact in facts:
                if not isinstance(fact, dict) or "subject" not in fact or "predicate" not in fact:
                    continue
                self._index[_make_key(fact["subject"], fact["predicate"])] = fact
                self._subject_index.setdefault(str(fact["subject"]).lower(), []).append(fact)

    def lookup(self, subject: str, predicate: str) -> Optional[Fact]:
        return self._index.get(_make_key(subject, predicate))

    def lookup_by_subject(self, subject: str) -> List[Fact]:
        return self._subject_index.get(subject.lower(), [])

    @property
    def size(self) -> int:
        return len(self._index)

    def get_all_facts(self) -> List[Fact]:
        return list(self._index.values())

    def add_fact(self, fact: Fact) -> None:
        self._index[_make_key(fact["subject"], fact["predicate"])] = fact
        self._subject_index.setdefault(str(fact["subject"]).lower(), []).append(fact)


def check_claim(claim: Claim, registry: FactRegistry) -> Dict[str, Any]:
    # 1. exact predicate strategy for the claim type (fuzzy numeric values)
    direct = CLAIM_TO_FACT_PREDICATE.get(str(claim.get("type")))
    if direct:
        for pred in direct if isinstance(direct, list) else [direct]:
            fact = registry.lookup(str(claim["subject"]), pred)
            if fact:
                status = "verified" if values_match_fuzzy(str(claim["value"]), str(fact["value"])) else "contradicted"
                return {"claim": claim, "fact": fact, "status": status}
    # 2. claim's own predicate (exact values)
    fact = registry.lookup(str(claim["subject"]), str(claim["predicate"]))
    if fact:
        status = "verified" if values_match(str(claim["value"]), str(fact["value"])) else "contradicted"
        return {"claim": claim, "fact": fact, "status": status}
    # 3. self_referential -> "self" subject
    if claim.get("type") == "self_referential":
        fact = registry.lookup("self", str(claim["predicate"]))
        if fact:
            status = "verified" if values_match(str(claim["value"]), str(fact["value"])) else "contradicted"
            return {"claim": claim, "fact": fact, "status": status}
    return {"claim": claim, "fact": None, "status": "unverified"}


def check_claims(claims: List[Claim], registry: FactRegistry) -> List[Dict[str, Any]]:
    return [check_claim(c, registry) for c in claims]
