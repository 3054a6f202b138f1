"""Membrane engine: ingest buffering + salience-filtered retrieval.

Config surface parity (brainplex `configurator.ts:137-148`):
  buffer_size 10, default_sensitivity "low", retrieve_limit 2,
  retrieve_min_salience 0.1, retrieve_max_sensitivity "medium",
  retrieve_timeout_ms 30000.

Retrieval score = cosine similarity x effective (decayed) salience; only
memories with sensitivity <= retrieve_max_sensitivity and effective
salience >= retrieve_min_salience are eligible; recalled memories are
reinforced (organic decay + reinforcement, reference `README.md:17`).
"""

from __future__ import annotations

import time
from typing import Dict, List, Optional

from .index import SalienceIndex
from .store import SENSITIVITY_ORDER, MemoryRecord, MemoryStore

DEFAULT_CONFIG = {
    "buffer_size": 10,
    "default_sensitivity": "low",
    "retrieve_limit": 2,
    "retrieve_min_salience": 0.1,
    "retrieve_max_sensitivity": "medium",
    "retrieve_timeout_ms": 30000,
}


class MembraneEngine:
    def __init__(
        self,
        workspace: str,
        config: Optional[Dict] = None,
        device: Optional[str] = None,
        dim: int = 1024,
        clock=time.time,
    ):
        self.config = {**DEFAULT_CONFIG, **(config or {})}
        self.store = MemoryStore(workspace, clock=clock)
        self.index = SalienceIndex(dim=dim, device=device)
        self._clock = clock
        self._buffer: List[MemoryRecord] = []
        self.stats = {"ingested": 0, "retrieved": 0, "retrievals": 0}

    # -- ingest ------------------------------------------------------------
    def remember(
        self,
        agent: str,
        text: str,
        kind: str = "episodic",
        sensitivity: Optional[str] = None,
        salience: float = 1.0,
        meta: Optional[Dict] = None,
    ) -> MemoryRecord:
        """Buffered ingest: records hit the store immediately, the GPU
        index in batches of `buffer_size` (amortizes encode+append)."""
        rec = MemoryRecord.make(
            agent,
            text,
            kind=kind,
            sensitivity=sensitivity or self.config["default_sensitivity"],
            salience=salience,
            meta=meta,
            ts=self._clock(),
        )
        self.store.add(rec)
        self._buffer.append(rec)
        self.stats["ingested"] += 1
        if len(self._buffer) >= int(self.config["buffer_size"]):
            self.flush_buffer()
        return rec

    def ingest(self, agent: str, texts: List[str], metas: Optional[List[Dict]] = None) -> None:
        """Bulk ingest (used by knowledge.embeddings.LocalGpuEmbedder)."""
        for i, t in enumerate(texts):
            self.remember(agent, t, kind="semantic", meta=(metas[i] if metas else None))

    def flush_buffer(self) -> None:
        if not self._buffer:
            return
        by_agent: Dict[str, List[MemoryRecord]] = {}
        for rec in self._buffer:
            by_agent.setdefault(rec["agent"], []).append(rec)
        for agent, recs in by_agent.items():
            self.index.add(agent, [r["id"] for r in recs], [r["text"] for r in recs])
        self._buffer = []

    # -- retrieve ----------------------------------------------------------
    def retrieve(
        self,
        agent: str,
        query: str,
        limit: Optional[int] = None,
        min_salience: Optional[float] = None,
        max_sensitivity: Optional[str] = None,
    ) -> List[Dict]:
        self.flush_buffer()  # queries see everything remembered so far
        limit = int(limit if limit is not None else self.config["retrieve_limit"])
        min_sal = float(
            min_salience if min_salience is not None else self.config["retrieve_min_salience"]
        )
        max_sens = SENSITIVITY_ORDER[
            max_sensitivity or self.config["retrieve_max_sensitivity"]
        ]
        now = self._clock()
        candidates = self.index.search(agent, query, k=max(limit * 4, limit))
        out = []
        for rec_id, cosine in candidates:
            rec = self.store.get(agent, rec_id)
            if rec is None:
                continue
            if SENSITIVITY_ORDER[rec["sensitivity"]] > max_sens:
                continue
            eff = self.store.decayed_salience(rec, now)
            if eff < min_sal:
                continue
            out.append({"record": rec, "cosine": cosine, "salience": eff,
                        "score": cosine * eff})
        out.sort(key=lambda r: -r["score"])
        out = out[:limit]
        for r in out:
            self.store.reinforce(agent, r["record"]["id"])
        self.stats["retrieved"] += len(out)
        self.stats["retrievals"] += 1
        return out

    def format_context(self, results: List[Dict]) -> str:
        """Inject-into-prompt block (suite README data flow: 'Membrane
        (inject relevant memories)')."""
        if not results:
            return ""
        lines = ["## Relevant memories"]
        for r in results:
            rec = r["record"]
            lines.append(f"- [{rec['kind']}|{r['score']:.2f}] {rec['text']}")
        return "\n".join(lines)

    def decay_pass(self, agent: str) -> int:
        """Periodic maintenance: prune fully-decayed memories."""
        return self.store.prune(agent)

    def flush(self) -> None:
        self.flush_buffer()
        self.store.flush()
