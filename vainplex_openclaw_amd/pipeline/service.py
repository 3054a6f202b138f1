"""FirewallService: micro-batching bridge between the per-message hook
API and the batched GPU pipeline.

SURVEY.md §7 "hard parts": the hook contract is per-message and
`before_message_write` is SYNCHRONOUS, but the GPU wants 4096-message
batches. This service aggregates concurrent submissions into
micro-batches (size- or deadline-triggered), runs one pipeline step, and
resolves each caller's future with its own verdict — a sync fast-path
with bounded added latency (`max_wait_ms`), batched throughput underneath.

Host-side callers: `submit(...)` returns a Future; `check(...)` is the
synchronous wrapper the hook handlers use. Works against the real
`FirewallPipeline` on a GPU or any stub exposing `step(batch)` (CPU
tests inject one).
"""

from __future__ import annotations

import threading
import time
from concurrent.futures import Future
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional

import numpy as np

from .synth import SynthBatch

VERDICT_NAMES = {0: "allow", 1: "audit", 2: "2fa", 3: "deny"}


@dataclass
class _Pending:
    message: bytes
    agent_idx: int
    tool_risk: float
    future: Future = field(default_factory=Future)


class FirewallService:
    def __init__(
        self,
        pipeline: Any,
        max_batch: int = 4096,
        max_wait_ms: float = 2.0,
        clock=time.time,
    ):
        self.pipeline = pipeline
        self.max_batch = max_batch
        self.max_wait_ms = max_wait_ms
        self._clock = clock
        self._lock = threading.Lock()
        self._cv = threading.Condition(self._lock)
        self._queue: List[_Pending] = []
        self._stop = False
        self._thread = threading.Thread(target=self._run, daemon=True)
        self._thread.start()
        self.stats = {"batches": 0, "messages": 0, "maxBatch": 0}

    # -- submission --------------------------------------------------------
    def submit(self, message: bytes, agent_idx: int = 0, tool_risk: float = 0.0) -> Future:
        if isinstance(message, str):
            message = message.encode("utf-8", "replace")
        p = _Pending(message, int(agent_idx), float(tool_risk))
        with self._cv:
            if self._stop:
                raise RuntimeError("FirewallService is closed")
            self._queue.append(p)
            self._cv.notify()
        return p.future

    def check(self, message: bytes, agent_idx: int = 0, tool_risk: float = 0.0,
              timeout: float = 5.0) -> Dict[str, Any]:
        """Synchronous gate (the before_message_write contract): blocks
        for this message's slot in the next micro-batch."""
        return self.submit(message, agent_idx, tool_risk).result(timeout=timeout)

    # -- batcher -----------------------------------------------------------
    def _take_batch(self) -> List[_Pending]:
        with self._cv:
            while not self._queue and not self._stop:
                self._cv.wait(0.05)
            if not self._queue:
                return []
            # deadline: give co-arriving messages max_wait_ms to pile up
            deadline = self._clock() + self.max_wait_ms / 1000.0
            while (len(self._queue) < self.max_batch
                   and self._clock() < deadline and not self._stop):
                self._cv.wait(max(0.0, deadline - self._clock()))
            batch = self._queue[: self.max_batch]
            del self._queue[: len(batch)]
            return batch

    def _run(self) -> None:
        while True:
            batch = self._take_batch()
            if not batch:
                if self._stop:
                    return
                continue
            try:
                out = self._process(batch)
                for i, p in enumerate(batch):
                    # a caller may have cancelled / timed out its future;
                    # that must never poison the rest of the batch
                    if not p.future.done():
                        try:
                            p.future.set_result(out[i])
                        except Exception:
                            pass
            except Exception as exc:
                for p in batch:
                    if not p.future.done():
                        try:
                            p.future.set_exception(exc)
                        except Exception:
                            pass

    def _process(self, batch: List[_Pending]) -> List[Dict[str, Any]]:
        sb = SynthBatch(
            messages=[p.message for p in batch],
            agent_idx=np.array([p.agent_idx for p in batch], dtype=np.int32),
            tool_risk=np.array([p.tool_risk for p in batch], dtype=np.float32),
            labels=None,
        )
        res = self.pipeline.step(sb)
        verdict = res["verdict"].to("cpu").numpy()
        risk = res["risk"].to("cpu").numpy()
        hits = {k: v.to("cpu").numpy() for k, v in res.get("hits", {}).items()}
        recall_ids = res.get("recall_ids")
        recall_ids = recall_ids.to("cpu").numpy() if recall_ids is not None else None
        self.stats["batches"] += 1
        self.stats["messages"] += len(batch)
        self.stats["maxBatch"] = max(self.stats["maxBatch"], len(batch))
        out = []
        for i in range(len(batch)):
            out.append({
                "verdict": VERDICT_NAMES.get(int(verdict[i]), "allow"),
                "risk": float(risk[i]),
                "hits": {k: int(v[i]) for k, v in hits.items()},
                "recallIds": recall_ids[i].tolist() if recall_ids is not None else [],
            })
        return out

    def close(self) -> None:
        with self._cv:
            self._stop = True
            self._cv.notify_all()
        self._thread.join(timeout=5.0)
