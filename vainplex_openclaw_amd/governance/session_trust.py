"""Ephemeral per-session trust.

Parity target: governance `src/session-trust-manager.ts` — session score
seeded at floor(agentScore × seedFactor) (`:59`), per-signal deltas with a
clean-streak bonus, ceiling = floor(agentScore × ceilingFactor) capped at
100, floor 0, 500-session LRU-by-creation eviction (`:10,23-34`).
"""

from __future__ import annotations

import math
import time
from dataclasses import dataclass, field
from typing import Any, Dict, Optional

from .trust import TrustManager
from .util import score_to_tier

MAX_SESSIONS = 500

DEFAULT_SIGNALS = {
    "success": 0.5,
    "policyBlock": -5.0,
    "credentialViolation": -15.0,
    "validationFailure": -3.0,
    "cleanStreakThreshold": 10,
    "cleanStreakBonus": 2.0,
}


@dataclass
class SessionTrustConfig:
    enabled: bool = True
    seed_factor: float = 0.7
    ceiling_factor: float = 1.2
    signals: Dict[str, float] = field(default_factory=lambda: dict(DEFAULT_SIGNALS))

    @classmethod
    def from_dict(cls, d: Optional[Dict[str, Any]]) -> "SessionTrustConfig":
        d = d or {}
        sig = dict(DEFAULT_SIGNALS)
        sig.update(d.get("signals", {}))
        return cls(
            enabled=bool(d.get("enabled", True)),
            seed_factor=float(d.get("seedFactor", 0.7)),
            ceiling_factor=float(d.get("ceilingFactor", 1.2)),
            signals=sig,
        )


class SessionTrustManager:
    def __init__(self, config: Optional[SessionTrustConfig], trust: TrustManager, clock=time.time):
        self.config = config or SessionTrustConfig()
        self.trust = trust
        self.clock = clock
        self._sessions: Dict[str, Dict[str, Any]] = {}

    def _evict_if_needed(self) -> None:
        if len(self._sessions) <= MAX_SESSIONS:
            return
        oldest = min(self._sessions, key=lambda k: self._sessions[k]["createdAt"])
        del self._sessions[oldest]

    def initialize(self, session_id: str, agent_id: str) -> Dict[str, Any]:
        agent_score = self.trust.score(agent_id)
        if not self.config.enabled:
            st = {
                "sessionId": session_id,
                "agentId": agent_id,
                "score": agent_score,
                "tier": self.trust.tier(agent_id),
                "cleanStreak": 0,
                "createdAt": self.clock(),
            }
            self._sessions[session_id] = st
            return st
        score = math.floor(agent_score * self.config.seed_factor)
        st = {
            "sessionId": session_id,
            "agentId": agent_id,
            "score": score,
            "tier": score_to_tier(score),
            "cleanStreak": 0,
            "createdAt": self.clock(),
        }
        self._sessions[session_id] = st
        self._evict_if_needed()
        return st

    def get(self, session_id: str, agent_id: str) -> Dict[str, Any]:
        st = self._sessions.get(session_id)
        if st is not None:
            return st
        return self.initialize(session_id, agent_id)

    def apply_signal(self, session_id: str, agent_id: str, signal: str) -> Dict[str, Any]:
        if not self.config.enabled:
            return self.get(session_id, agent_id)
        st = self.get(session_id, agent_id)
        delta = float(self.config.signals.get(signal, 0))
        if signal == "success":
            st["cleanStreak"] += 1
            if st["cleanStreak"] >= self.config.signals.get("cleanStreakThreshold", 10):
                delta += float(self.config.signals.get("cleanStreakBonus", 0))
                st["cleanStreak"] = 0
        else:
            st["cleanStreak"] = 0
        return self.set_score(session_id, agent_id, st["score"] + delta)

    def set_score(self, session_id: str, agent_id: str, new_score: float) -> Dict[str, Any]:
        if not self.config.enabled:
            return self.get(session_id, agent_id)
        st = self.get(session_id, agent_id)
        ceiling = min(100, math.floor(self.trust.score(agent_id) * self.config.ceiling_factor))
        st["score"] = max(0, min(new_score, ceiling))
        st["tier"] = score_to_tier(st["score"])
        return st

    def destroy(self, session_id: str) -> None:
        self._sessions.pop(session_id, None)

    def sessions(self) -> Dict[str, Dict[str, Any]]:
        return dict(self._sessions)
