"""Fixed-capacity action-frequency ring buffer.

Parity target: governance `src/frequency-tracker.ts` — ring of
{ts, agentId, sessionKey, toolName}; `count(windowSeconds, scope, ...)`
is a linear scan over live entries (`:16-46`). Scopes: "agent" (same
agentId), "session" (same sessionKey), "tool" (same agentId+toolName),
"global".
"""

from __future__ import annotations

import threading
import time
from typing import List, Optional, Tuple


class FrequencyTracker:
    def __init__(self, capacity: int = 1000, clock=time.time):
        self.capacity = capacity
        self.clock = clock
        self._buf: List[Optional[Tuple[float, str, str, str]]] = [None] * capacity
        self._head = 0
        self._size = 0
        # the reference is single-threaded; here the service micro-batcher
        # and hook threads may record concurrently — a lost update means a
        # rate limit silently under-counts, so writes take a lock
        self._lock = threading.Lock()

    def record(self, agent_id: str, session_key: str = "", tool_name: str = "") -> None:
        with self._lock:
            self._buf[self._head] = (self.clock(), agent_id, session_key, tool_name)
            self._head = (self._head + 1) % self.capacity
            self._size = min(self._size + 1, self.capacity)

    def count(
        self,
        window_seconds: float,
        scope: str = "agent",
        agent_id: str = "",
        session_key: str = "",
        tool_name: str = "",
    ) -> int:
        cutoff = self.clock() - window_seconds
        n = 0
        for entry in self._buf:
            if entry is None:
                continue
            ts, a, s, t = entry
            if ts < cutoff:
                continue
            if scope == "agent" and a != agent_id:
                continue
            if scope == "session" and s != session_key:
                continue
            if scope == "tool" and (a != agent_id or t != tool_name):
                continue
            n += 1
        return n
