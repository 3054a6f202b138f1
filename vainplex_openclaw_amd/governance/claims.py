"""Claim detection — Stage 1 of output validation.

Parity target: governance `src/claim-detector.ts` — 5 synchronous regex
detectors (system_state `:33-59`, entity_name `:64-87`, existence
pos/neg/there-is `:93-151`, operational_status, self_referential
`:20-26`), a common-word false-positive filter, dedupe by
(type, offset, subject). The same pattern families are compiled into the
GPU multi-pattern DFA (`ops/dfa.py` + `csrc/pattern_scan.hip`) for the
batched firewall path; this module is the semantic reference.
"""

from __future__ import annotations

import re
from typing import Dict, List, Optional

Claim = Dict[str, object]

_SUBJ = r"([\w][\w.:-]{0,60})"

SYSTEM_STATE_RX = re.compile(
    _SUBJ
    + r"\s+(?:is|are)\s+(running|stopped|online|offline|active|inactive|enabled|"
    r"disabled|up|down|started|paused|healthy|unhealthy)\b",
    re.IGNORECASE,
)

ENTITY_NAME_RX = re.compile(
    r"\bthe\s+(agent|service|server|container|process|pod|node|instance|database|"
    r"cluster|daemon|plugin|module)\s+(?:named|called|known as|labelled|labeled)?"
    r"\s*[\"`']?([\w][\w.:-]{0,60})[\"`']?\b",
    re.IGNORECASE,
)

EXISTENCE_POS_RX = re.compile(
    _SUBJ
    + r"\s+(?:exists|is available|is present|is configured|is installed|"
    r"is deployed|is registered)\b",
    re.IGNORECASE,
)
EXISTENCE_NEG_RX = re.compile(
    _SUBJ
    + r"\s+(?:does(?:n't| not) exist|is not available|is not present|"
    r"is not configured|is not installed|is not deployed|is not registered|"
    r"doesn't exist)\b",
    re.IGNORECASE,
)
THERE_IS_RX = re.compile(r"\bthere\s+(?:is|are)\s+(no\s+)?([\w][\w.:-]{0,60})\b", re.IGNORECASE)

METRIC_RX = re.compile(
    _SUBJ
    + r"\s+(?:has|contains|uses|consumes|shows|reports)\s+(\d+[\d,.]*)\s*"
    r"(items?|entries|records|connections|requests|errors|GB|MB|KB|%|nodes?|pods?|"
    r"replicas?|instances?|processes?)?\b",
    re.IGNORECASE,
)
PERCENTAGE_RX = re.compile(_SUBJ + r"\s+is\s+at\s+(\d+[\d,.]*)\s*%", re.IGNORECASE)
COUNT_RX = re.compile(_SUBJ + r"\s+count\s+is\s+(\d+[\d,.]*)\b", re.IGNORECASE)

SELF_IDENTITY_RX = re.compile(r"\bI\s+am\s+([\w][\w\s.:-]{0,60}?)\s*[.,!?\n]", re.IGNORECASE)
MY_NAME_RX = re.compile(r"\bmy\s+name\s+is\s+([\w][\w\s.:-]{0,60}?)\s*[.,!?\n]", re.IGNORECASE)
I_HAVE_RX = re.compile(r"\bI\s+(?:have|possess|contain)\s+([\w][\w\s.:-]{0,60}?)\s*[.,!?\n]", re.IGNORECASE)

COMMON_WORDS = {
    "it", "this", "that", "the", "a", "an", "they", "we", "he", "she",
    "what", "which", "who", "how", "there", "here", "then", "now",
    "everything", "nothing", "something", "anything",
    "one", "two", "three", "all", "some", "none",
    "yes", "no", "not", "also", "very", "just", "still",
}


def _is_common(word: str) -> bool:
    return word.lower() in COMMON_WORDS


def _detect_system_state(text: str) -> List[Claim]:
    out: List[Claim] = []
    for m in SYSTEM_STATE_RX.finditer(text):
        subject = m.group(1).strip()
        if _is_common(subject):
            continue
        out.append({
            "type": "system_state", "subject": subject, "predicate": "state",
            "value": m.group(2).lower(), "source": m.group(0), "offset": m.start(),
        })
    return out


def _detect_entity_name(text: str) -> List[Claim]:
    out: List[Claim] = []
    for m in ENTITY_NAME_RX.finditer(text):
        out.append({
            "type": "entity_name", "subject": m.group(2).strip(),
            "predicate": "entity_type", "value": m.group(1).lower(),
            "source": m.group(0), "offset": m.start(),
        })
    return out


def _detect_existence(text: str) -> List[Claim]:
    out: List[Claim] = []
    for rx, value in ((EXISTENCE_POS_RX, "true"), (EXISTENCE_NEG_RX, "false")):
        for m in rx.finditer(text):
            subject = m.group(1).strip()
            if _is_common(subject):
                continue
            out.append({
                "type": "existence", "subject": subject, "predicate": "exists",
                "value": value, "source": m.group(0), "offset": m.start(),
            })
    for m in THERE_IS_RX.finditer(text):
        subject = m.group(2).strip()
        if _is_common(subject):
            continue
        out.append({
            "type": "existence", "subject": subject, "predicate": "exists",
            "value": "false" if m.group(1) else "true",
            "source": m.group(0), "offset": m.start(),
        })
    return out


def _detect_operational_status(text: str) -> List[Claim]:
    out: List[Claim] = []
    for m in METRIC_RX.finditer(text):
        subject = m.group(1).strip()
        if _is_common(subject):
            continue
        unit = m.group(3) or ""
        out.append({
            "type": "operational_status", "subject": subject, "predicate": "metric",
            "value": f"{m.group(2)} {unit}" if unit else m.group(2),
            "source": m.group(0), "offset": m.start(),
        })
    for m in PERCENTAGE_RX.finditer(text):
        subject = m.group(1).strip()
        if _is_common(subject):
            continue
        out.append({
            "type": "operational_status", "subject": subject, "predicate": "percentage",
            "value": f"{m.group(2)}%", "source": m.group(0), "offset": m.start(),
        })
    for m in COUNT_RX.finditer(text):
        subject = m.group(1).strip()
        if _is_common(subject):
            continue
        out.append({
            "type": "operational_status", "subject": subject, "predicate": "count",
            "value": m.group(2), "source": m.group(0), "offset": m.start(),
        })
    return out


def _detect_self_referential(text: str) -> List[Claim]:
    padded = text + "\n"
    out: List[Claim] = []
    for rx, predicate in (
        (SELF_IDENTITY_RX, "identity"),
        (MY_NAME_RX, "name"),
        (I_HAVE_RX, "capability"),
    ):
        for m in rx.finditer(padded):
            out.append({
                "type": "self_referential", "subject": "self", "predicate": predicate,
                "value": m.group(1).strip(), "source": m.group(0).strip(), "offset": m.start(),
            })
    return out


BUILTIN_DETECTORS = {
    "system_state": _detect_system_state,
    "entity_name": _detect_entity_name,
    "existence": _detect_existence,
    "operational_status": _detect_operational_status,
    "self_referential": _detect_self_referential,
}


def get_builtin_detector_ids() -> List[str]:
    return list(BUILTIN_DETECTORS.keys())


def detect_claims(text: str, enabled: Optional[List[str]] = None) -> List[Claim]:
    if not text:
        return []
    ids = enabled if enabled is not None else list(BUILTIN_DETECTORS.keys())
    claims: List[Claim] = []
    for det_id in ids:
        fn = BUILTIN_DETECTORS.get(det_id)
        if fn:
            claims.extend(fn(text))
    seen = set()
    result: List[Claim] = []
    for c in claims:
        key = (c["type"], c["offset"], c["subject"])
        if key not in seen:
            seen.add(key)
            result.append(c)
    return result
