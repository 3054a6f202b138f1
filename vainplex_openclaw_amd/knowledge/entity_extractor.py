"""Regex entity extraction with canonicalization and importance scoring.

Parity target: reference `openclaw-knowledge-engine/src/entity-extractor.ts`
— extract (`:34-54`), canonicalize strips org suffixes (`:90-96`), id is
`type:value-slug` (`:66`), type-based initial importance (org .8, person
.7, product .6, location .5, date/email/url .4, multiword .5 else .3,
`:101-112`), mergeEntities (`:117-136`).

The batched GPU path (ENTITY DFA family + 4-gram encoder in
`ops/pattern_sets.py` / `csrc/pattern_scan.hip`) detects which messages
contain entities at wire speed; this host module produces the full Entity
records for those messages.
"""

from __future__ import annotations

import re
import time
from dataclasses import dataclass, field
from typing import Dict, List, Optional

from .patterns import iter_matches

_ORG_SUFFIX = re.compile(r",?\s?(?:Inc\.|LLC|Corp\.|GmbH|AG|Ltd\.)$", re.IGNORECASE)
_TRAIL_PUNCT = re.compile(r"[.,!?;:]$")
_SLUG_WS = re.compile(r"\s+")

INITIAL_IMPORTANCE = {
    "organization": 0.8,
    "person": 0.7,
    "product": 0.6,
    "location": 0.5,
    "date": 0.4,
    "email": 0.4,
    "url": 0.4,
}


def _iso_now(clock=time.time) -> str:
    return time.strftime("%Y-%m-%dT%H:%M:%S", time.gmtime(clock())) + "Z"


@dataclass
class Entity:
    id: str
    type: str
    value: str
    mentions: List[str] = field(default_factory=list)
    count: int = 0
    importance: float = 0.0
    last_seen: str = ""
    source: List[str] = field(default_factory=list)

    def to_dict(self) -> Dict:
        return {
            "id": self.id,
            "type": self.type,
            "value": self.value,
            "mentions": list(self.mentions),
            "count": self.count,
            "importance": self.importance,
            "lastSeen": self.last_seen,
            "source": list(self.source),
        }


def canonicalize(value: str, etype: str) -> str:
    """Strip org suffixes / trailing punctuation (entity-extractor.ts:90-96)."""
    if etype == "organization":
        return _ORG_SUFFIX.sub("", value).strip()
    return _TRAIL_PUNCT.sub("", value).strip()


def initial_importance(etype: str, value: str) -> float:
    """Type table, multiword bonus for unknowns (entity-extractor.ts:101-112)."""
    if etype in INITIAL_IMPORTANCE:
        return INITIAL_IMPORTANCE[etype]
    return 0.5 if len(re.split(r"\s|-", value)) > 1 else 0.3


def entity_id(etype: str, canonical: str) -> str:
    return f"{etype}:{_SLUG_WS.sub('-', canonical.lower())}"


class EntityExtractor:
    def __init__(self, logger=None, clock=time.time):
        self._log = logger
        self._clock = clock

    def extract(self, text: str) -> List[Entity]:
        found: Dict[str, Entity] = {}
        for _family, etype, value in iter_matches(text):
            canonical = canonicalize(value, etype)
            if not canonical:
                continue
            eid = entity_id(etype, canonical)
            existing = found.get(eid)
            if existing is not None:
                if value not in existing.mentions:
                    existing.mentions.append(value)
                existing.count += 1
                if "regex" not in existing.source:
                    existing.source.append("regex")
            else:
                found[eid] = Entity(
                    id=eid,
                    type=etype,
                    value=canonical,
                    mentions=[value],
                    count=1,
                    importance=initial_importance(etype, value),
                    last_seen=_iso_now(self._clock),
                    source=["regex"],
                )
        return list(found.values())


def merge_entities(list_a: List[Entity], list_b: List[Entity], clock=time.time) -> List[Entity]:
    """Merge by id: sum counts, union mentions/sources, max importance,
    later lastSeen wins (entity-extractor.ts:117-136)."""
    merged: Dict[str, Entity] = {}
    for e in list_a:
        merged[e.id] = Entity(**{**e.__dict__, "mentions": list(e.mentions), "source": list(e.source)})
    now = _iso_now(clock)
    for e in list_b:
        ex = merged.get(e.id)
        if ex is not None:
            ex.count += e.count
            ex.mentions = list(dict.fromkeys([*ex.mentions, *e.mentions]))
            ex.source = list(dict.fromkeys([*ex.source, *e.source]))
            if now > ex.last_seen:
                ex.last_seen = now
            ex.importance = max(ex.importance, e.importance)
        else:
            merged[e.id] = Entity(**{**e.__dict__, "mentions": list(e.mentions), "source": list(e.source)})
    return list(merged.values())
