"""AgentProof REST reputation + feedback-signal client.

Parity target: governance `src/security/agentproof-rest.ts` — ring-buffer
signal queue (capacity 1000), 5 s background flusher, circuit breaker on
repeated failures, file-based bearer key (`:77-110`). HTTP transport is
injected; the queue/breaker logic is the tested surface.
"""

from __future__ import annotations

import os
import threading
import time
from typing import Any, Callable, Dict, List, Optional

QUEUE_CAPACITY = 1000
FLUSH_INTERVAL_S = 5.0
BREAKER_THRESHOLD = 5
BREAKER_RESET_S = 60.0


class CircuitBreaker:
    def __init__(self, threshold: int = BREAKER_THRESHOLD, reset_s: float = BREAKER_RESET_S, clock=time.time):
        self.threshold = threshold
        self.reset_s = reset_s
        self.clock = clock
        self.failures = 0
        self.opened_at: Optional[float] = None

    @property
    def is_open(self) -> bool:
        if self.opened_at is None:
            return False
        if self.clock() - self.opened_at >= self.reset_s:
            # half-open: allow a retry
            return False
        return True

    def record_success(self) -> None:
        self.failures = 0
        self.opened_at = None

    def record_failure(self) -> None:
        self.failures += 1
        if self.failures >= self.threshold:
            self.opened_at = self.clock()


class AgentProofRestClient:
    def __init__(
        self,
        base_url: str = "https://api.agentproof.example",
        api_key_file: Optional[str] = None,
        http_post: Optional[Callable[[str, Dict[str, str], Any], Any]] = None,
        http_get: Optional[Callable[[str, Dict[str, str]], Any]] = None,
        clock=time.time,
    ):
        self.base_url = base_url.rstrip("/")
        self.api_key_file = api_key_file
        self.http_post = http_post
        self.http_get = http_get
        self.clock = clock
        self.breaker = CircuitBreaker(clock=clock)
        self._queue: List[Dict[str, Any]] = []
        self._lock = threading.Lock()
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None
        self.dropped = 0
        self.flushed = 0
        self._key_cached: Optional[str] = None

    def _bearer(self) -> Optional[str]:
        # API key is read once and cached (agentproof-rest.ts:35-80);
        # missing or empty files mean unauthenticated requests
        if self._key_cached is not None:
            return self._key_cached or None
        if not self.api_key_file or not os.path.isfile(self.api_key_file):
            self._key_cached = ""
            return None
        with open(self.api_key_file, "r", encoding="utf-8") as fh:
            self._key_cached = fh.read().strip()
        return self._key_cached or None

    def _headers(self) -> Dict[str, str]:
        key = self._bearer()
        # the REST API authenticates via X-API-Key (agentproof-rest.ts)
        return {"X-API-Key": key} if key else {}

    # -- signal queue ------------------------------------------------------
    def enqueue_signal(self, agent_id: str, signal: str, detail: Optional[Dict[str, Any]] = None) -> bool:
        with self._lock:
            if len(self._queue) >= QUEUE_CAPACITY:
                self._queue.pop(0)  # ring: drop oldest
                self.dropped += 1
            self._queue.append(
                {"agentId": agent_id, "signal": signal, "detail": detail or {}, "ts": self.clock()}
            )
        return True

    def flush(self) -> int:
        with self._lock:
            batch = self._queue
            self._queue = []
        if not batch:
            return 0
        if self.http_post is None or self.breaker.is_open:
            # no transport / breaker open: drop silently (fire-and-forget)
            self.dropped += len(batch)
            return 0
        headers = self._headers()
        try:
            self.http_post(f"{self.base_url}/v1/signals", headers, {"signals": batch})
            self.breaker.record_success()
            self.flushed += len(batch)
            return len(batch)
        except Exception:
            self.breaker.record_failure()
            with self._lock:
                # requeue up to capacity for retry
                requeue = batch[-(QUEUE_CAPACITY - len(self._queue)) :]
                self._queue = requeue + self._queue
            return 0

    def lookup_reputation(self, agent_id: str) -> Optional[Dict[str, Any]]:
        if self.http_get is None or self.breaker.is_open:
            return None
        try:
            out = self.http_get(f"{self.base_url}/v1/reputation/{agent_id}", self._headers())
            self.breaker.record_success()
            return out
        except Exception:
            self.breaker.record_failure()
            return None

    def get_agent_profile(self, agent_id: str) -> Optional[Dict[str, Any]]:
        """GET /trust/{agentId} (agentproof-rest.ts getAgentProfile):
        None on HTTP/network error or malformed response (no agentId);
        score clamped to 0-100; tier classified from the REST data."""
        from .erc8004 import classify_reputation

        if self.http_get is None:
            return None
        try:
            out = self.http_get(f"{self.base_url}/trust/{agent_id}", self._headers())
        except Exception:
            return None
        if not isinstance(out, dict) or not out.get("agentId"):
            return None
        score = min(100, max(0, int(out.get("reputationScore") or 0)))
        feedback = int(out.get("feedbackCount") or 0)
        return {
            "agentId": out["agentId"],
            "registered": bool(out.get("registered", True)),
            "score": score,
            "feedbackCount": feedback,
            "tier": classify_reputation(bool(out.get("registered", True)), feedback, score),
        }

    def batch_lookup(self, agent_ids: List[str]) -> List[Optional[Dict[str, Any]]]:
        """POST /trust/batch (agentproof-rest.ts batchLookup): empty in ->
        empty out; any failure or missing results array -> all None."""
        if not agent_ids:
            return []
        if self.http_post is None:
            return [None] * len(agent_ids)
        try:
            out = self.http_post(f"{self.base_url}/trust/batch", self._headers(),
                                 {"agentIds": list(agent_ids)})
        except Exception:
            return [None] * len(agent_ids)
        results = out.get("results") if isinstance(out, dict) else None
        if not isinstance(results, list):
            return [None] * len(agent_ids)
        by_id = {r.get("agentId"): r for r in results if isinstance(r, dict)}
        return [by_id.get(a) for a in agent_ids]

    # -- background flusher ------------------------------------------------
    def start(self) -> None:
        if self._thread is not None:
            return
        self._stop.clear()

        def run() -> None:
            while not self._stop.wait(FLUSH_INTERVAL_S):
                try:
                    self.flush()
                except Exception:
                    pass

        self._thread = threading.Thread(target=run, daemon=True)
        self._thread.start()

    def stop(self) -> None:
        self._stop.set()
        if self._thread is not None:
            self._thread.join(timeout=2.0)
            self._thread = None
        self.flush()

    @property
    def queue_depth(self) -> int:
        with self._lock:
            return len(self._queue)
