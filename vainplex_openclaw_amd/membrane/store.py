"""Membrane memory store: episodic / semantic / working memory records
with per-agent isolation, salience, sensitivity and organic decay.

Capability target (the Membrane plugin is an external repo; surface from
reference `README.md:17` "salience-based recall with organic decay",
brainplex `README.md` §Membrane: episodic/semantic/working memory, agent
isolation, salience filtering; config keys in brainplex
`configurator.ts:137-148`).

Records persist to `<workspace>/memory/membrane/<agent>.json` (per-agent
files = hard isolation on disk, mirroring the suite's atomic-JSON L3
persistence style).
"""

from __future__ import annotations

import os
import time
import uuid
from typing import Dict, List, Optional

from ..utils.storage import DebouncedSaver, atomic_write_json, read_json

SENSITIVITY_ORDER = {"low": 0, "medium": 1, "high": 2}
KINDS = ("episodic", "semantic", "working")

# Organic decay: salience halves every DECAY_HALF_LIFE_H hours without
# reinforcement; recall reinforces by RECALL_BOOST toward 1.0.
DECAY_HALF_LIFE_H = 72.0
RECALL_BOOST = 0.25
MIN_SALIENCE = 0.01


class MemoryRecord(dict):
    """Plain dict subclass so records serialize directly to JSON."""

    @staticmethod
    def make(
        agent: str,
        text: str,
        kind: str = "episodic",
        sensitivity: str = "low",
        salience: float = 1.0,
        meta: Optional[Dict] = None,
        ts: Optional[float] = None,
    ) -> "MemoryRecord":
        if kind not in KINDS:
            raise ValueError(f"kind must be one of {KINDS}")
        if sensitivity not in SENSITIVITY_ORDER:
            raise ValueError("sensitivity must be low|medium|high")
        return MemoryRecord(
            id=str(uuid.uuid4()),
            agent=agent,
            text=text,
            kind=kind,
            sensitivity=sensitivity,
            salience=float(salience),
            meta=meta or {},
            createdTs=ts if ts is not None else time.time(),
            lastRecallTs=ts if ts is not None else time.time(),
            recalls=0,
        )


class MemoryStore:
    def __init__(self, workspace: str, clock=time.time, write_debounce_ms: int = 250):
        self.workspace = workspace
        self._clock = clock
        self._by_agent: Dict[str, Dict[str, MemoryRecord]] = {}
        self._savers: Dict[str, DebouncedSaver] = {}
        self._debounce = write_debounce_ms / 1000.0

    def _dir(self) -> str:
        return os.path.join(self.workspace, "memory", "membrane")

    def _path(self, agent: str) -> str:
        safe = "".join(c if c.isalnum() or c in "-_" else "_" for c in agent)
        return os.path.join(self._dir(), f"{safe}.json")

    def _agent_map(self, agent: str) -> Dict[str, MemoryRecord]:
        if agent not in self._by_agent:
            data = read_json(self._path(agent))
            recs = {}
            if isinstance(data, dict) and isinstance(data.get("memories"), list):
                for r in data["memories"]:
                    if isinstance(r, dict) and "id" in r:
                        recs[r["id"]] = MemoryRecord(r)
            self._by_agent[agent] = recs
            self._savers[agent] = DebouncedSaver(
                lambda a=agent: self._persist(a), delay=self._debounce
            )
        return self._by_agent[agent]

    def _persist(self, agent: str) -> None:
        atomic_write_json(
            self._path(agent),
            {"updated": self._clock(), "memories": list(self._agent_map(agent).values())},
        )

    # -- API ---------------------------------------------------------------
    def add(self, record: MemoryRecord) -> MemoryRecord:
        m = self._agent_map(record["agent"])
        m[record["id"]] = record
        self._savers[record["agent"]].mark_dirty()
        return record

    def get(self, agent: str, record_id: str) -> Optional[MemoryRecord]:
        return self._agent_map(agent).get(record_id)

    def all(self, agent: str) -> List[MemoryRecord]:
        """Per-agent isolation: only this agent's memories, ever."""
        return list(self._agent_map(agent).values())

    def count(self, agent: str) -> int:
        return len(self._agent_map(agent))

    def decayed_salience(self, rec: MemoryRecord, now: Optional[float] = None) -> float:
        """Effective salience after organic decay since last recall."""
        now = self._clock() if now is None else now
        dt_h = max(0.0, (now - rec.get("lastRecallTs", now)) / 3600.0)
        return max(MIN_SALIENCE, rec["salience"] * (0.5 ** (dt_h / DECAY_HALF_LIFE_H)))

    def reinforce(self, agent: str, record_id: str) -> None:
        """Recall reinforcement: commit the decayed value, then boost
        toward 1.0 (organic decay + use-it-or-lose-it)."""
        rec = self.get(agent, record_id)
        if rec is None:
            return
        now = self._clock()
        cur = self.decayed_salience(rec, now)
        rec["salience"] = min(1.0, cur + RECALL_BOOST * (1.0 - cur))
        rec["lastRecallTs"] = now
        rec["recalls"] = rec.get("recalls", 0) + 1
        self._savers[agent].mark_dirty()

    def prune(self, agent: str, min_salience: float = MIN_SALIENCE * 2, max_records: int = 0) -> int:
        """Drop fully-decayed records; optionally cap the store size
        (lowest effective salience evicted first)."""
        m = self._agent_map(agent)
        now = self._clock()
        victims = [rid for rid, r in m.items() if self.decayed_salience(r, now) <= min_salience]
        for rid in victims:
            del m[rid]
        if max_records and len(m) > max_records:
            ranked = sorted(m.values(), key=lambda r: self.decayed_salience(r, now))
            for r in ranked[: len(m) - max_records]:
                del m[r["id"]]
                victims.append(r["id"])
        if victims:
            self._savers[agent].mark_dirty()
        return len(victims)

    def flush(self) -> None:
        for agent, saver in self._savers.items():
            saver.flush()
