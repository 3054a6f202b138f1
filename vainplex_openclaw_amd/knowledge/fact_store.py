"""SPO-triple fact store with dedupe, relevance boost/decay, prune, and
debounced atomic persistence to `<workspace>/facts.json`.

Parity target: reference `openclaw-knowledge-engine/src/fact-store.ts` —
dedupe on (subject, predicate, object) with relevance boost (`:82-94`),
new facts start at relevance 1.0 (`:96-102`), boost pushes 50% closer to
1.0, decay multiplies by (1-rate) with a 0.1 floor, prune drops the least
relevant (then oldest-accessed) facts over maxFacts, debounced atomic
writes (`:25-34`), unembedded tracking for the embeddings sync.
"""

from __future__ import annotations

import os
import time
import uuid
from typing import Callable, Dict, List, Optional

from ..utils.storage import DebouncedSaver, atomic_write_json, read_json

MIN_RELEVANCE = 0.1  # decay floor (fact-store.ts decayFacts)


def _iso_now(clock: Callable[[], float]) -> str:
    return time.strftime("%Y-%m-%dT%H:%M:%S", time.gmtime(clock())) + "Z"


def boost_relevance(current: float) -> float:
    """Push relevance 50% closer to 1.0 (fact-store.ts boostRelevance)."""
    return min(1.0, current + (1.0 - current) * 0.5)


class FactStore:
    def __init__(
        self,
        workspace: str,
        max_facts: int = 10_000,
        write_debounce_ms: int = 250,
        logger=None,
        clock: Callable[[], float] = time.time,
    ):
        self.workspace = workspace
        self.max_facts = max_facts
        self._log = logger
        self._clock = clock
        self.facts: Dict[str, Dict] = {}
        self.is_loaded = False
        self._saver = DebouncedSaver(self._persist, delay=write_debounce_ms / 1000.0)

    # -- persistence -------------------------------------------------------
    @property
    def path(self) -> str:
        return os.path.join(self.workspace, "facts.json")

    def load(self) -> None:
        if self.is_loaded:
            return
        data = read_json(self.path)
        if isinstance(data, dict) and isinstance(data.get("facts"), list):
            self.facts = {f["id"]: f for f in data["facts"] if isinstance(f, dict) and "id" in f}
        else:
            self.facts = {}
        self.is_loaded = True

    def _persist(self) -> None:
        if not self.is_loaded:
            return
        atomic_write_json(
            self.path,
            {"updated": _iso_now(self._clock), "facts": list(self.facts.values())},
        )

    def commit(self) -> None:
        self._saver.mark_dirty()

    def flush(self) -> None:
        if self.is_loaded:
            self._saver.flush()
            self._persist()

    # -- mutation ----------------------------------------------------------
    def add_fact(self, subject: str, predicate: str, obj: str, source: str = "ingested") -> Dict:
        """Dedupe on (s,p,o): existing fact gets a relevance boost + fresh
        lastAccessed instead of a duplicate (fact-store.ts:82-94)."""
        if not self.is_loaded:
            raise RuntimeError("FactStore has not been loaded yet. Call load() first.")
        now = _iso_now(self._clock)
        for fact in self.facts.values():
            if (
                fact["subject"] == subject
                and fact["predicate"] == predicate
                and fact["object"] == obj
            ):
                fact["relevance"] = boost_relevance(fact["relevance"])
                fact["lastAccessed"] = now
                self.commit()
                return fact
        fact = {
            "id": str(uuid.uuid4()),
            "subject": subject,
            "predicate": predicate,
            "object": obj,
            "source": source,
            "createdAt": now,
            "lastAccessed": now,
            "relevance": 1.0,
        }
        self.facts[fact["id"]] = fact
        self._prune()
        self.commit()
        return fact

    def get_fact(self, fact_id: str) -> Optional[Dict]:
        fact = self.facts.get(fact_id)
        if fact is not None:
            fact["lastAccessed"] = _iso_now(self._clock)
            fact["relevance"] = boost_relevance(fact["relevance"])
            self.commit()
        return fact

    def query(
        self,
        subject: Optional[str] = None,
        predicate: Optional[str] = None,
        obj: Optional[str] = None,
    ) -> List[Dict]:
        out = [
            f
            for f in self.facts.values()
            if (subject is None or f["subject"] == subject)
            and (predicate is None or f["predicate"] == predicate)
            and (obj is None or f["object"] == obj)
        ]
        return sorted(out, key=lambda f: -f["relevance"])

    def decay_facts(self, rate: float) -> int:
        """relevance *= (1-rate), floored at 0.1; returns decayed count."""
        decayed = 0
        for fact in self.facts.values():
            new_rel = fact["relevance"] * (1.0 - rate)
            if new_rel != fact["relevance"]:
                fact["relevance"] = max(MIN_RELEVANCE, new_rel)
                decayed += 1
        if decayed:
            self.commit()
        return decayed

    def _prune(self) -> None:
        overflow = len(self.facts) - self.max_facts
        if overflow <= 0:
            return
        victims = sorted(
            self.facts.values(), key=lambda f: (f["relevance"], f["lastAccessed"])
        )[:overflow]
        for f in victims:
            del self.facts[f["id"]]

    # -- embeddings sync support -------------------------------------------
    def unembedded_facts(self) -> List[Dict]:
        return [f for f in self.facts.values() if not f.get("embedded")]

    def mark_embedded(self, fact_ids: List[str]) -> None:
        now = _iso_now(self._clock)
        touched = 0
        for fid in fact_ids:
            fact = self.facts.get(fid)
            if fact is not None:
                fact["embedded"] = now
                touched += 1
        if touched:
            self.commit()
