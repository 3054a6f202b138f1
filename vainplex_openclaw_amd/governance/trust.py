"""Persistent per-agent trust scoring.

Parity target: governance `src/trust-manager.ts` —
score = clamp(min(ageDays*0.5, 20) + min(success*0.1, 30) - 2*violations
              + min(cleanStreak*0.3, 20) + manualAdjustment, 0, 100)
(`trust-manager.ts:15-43`), decay ×0.95 after 30 days of inactivity
(`:151-168`), tier lock/floor (`:223-245`), fresh-agent manualAdjustment
migration (`:116-149`), periodic flush of `<workspace>/governance/trust.json`
(`:78, :291-324`). On-disk format is kept compatible: a `TrustStore`
{version, updated, agents: {id: AgentTrust}}.
"""

from __future__ import annotations

import datetime as _dt
import os
import time
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional

from ..core.api import PluginLogger, NullLogger
from ..utils.storage import atomic_write_json, read_json, IntervalFlusher
from .util import clamp, score_to_tier, tier_ordinal

DEFAULT_WEIGHTS = {
    "agePerDay": 0.5,
    "ageMax": 20.0,
    "successPerAction": 0.1,
    "successMax": 30.0,
    "violationPenalty": -2.0,
    "cleanStreakPerDay": 0.3,
    "cleanStreakMax": 20.0,
}

DAY_MS = 86_400_000


def _iso_now() -> str:
    return _dt.datetime.now(_dt.timezone.utc).isoformat().replace("+00:00", "Z")


def compute_score(signals: Dict[str, float], weights: Optional[Dict[str, float]] = None) -> float:
    """The reference trust formula (trust-manager.ts:30-43)."""
    w = dict(DEFAULT_WEIGHTS)
    if weights:
        w.update(weights)
    base = min(signals.get("ageDays", 0) * w["agePerDay"], w["ageMax"])
    success = min(signals.get("successCount", 0) * w["successPerAction"], w["successMax"])
    violations = signals.get("violationCount", 0) * w["violationPenalty"]
    streak = min(signals.get("cleanStreak", 0) * w["cleanStreakPerDay"], w["cleanStreakMax"])
    raw = base + success + violations + streak + signals.get("manualAdjustment", 0)
    return clamp(raw, 0, 100)


def _new_agent(agent_id: str, initial_score: float) -> Dict[str, Any]:
    now = _iso_now()
    return {
        "agentId": agent_id,
        "score": clamp(initial_score, 0, 100),
        "tier": score_to_tier(initial_score),
        "signals": {
            "successCount": 0,
            "violationCount": 0,
            "ageDays": 0,
            "cleanStreak": 0,
            # Backfilled so a fresh agent doesn't drop to 0 on first
            # recalculate (migration note, trust-manager.ts:116-149).
            "manualAdjustment": initial_score,
        },
        "history": [],
        "lastEvaluation": now,
        "created": now,
        "lastActivity": now,
    }


@dataclass
class TrustConfig:
    default_score: float = 40.0
    initial_scores: Dict[str, float] = field(default_factory=dict)
    weights: Dict[str, float] = field(default_factory=dict)
    decay_after_days: float = 30.0
    decay_factor: float = 0.95
    tier_locks: Dict[str, str] = field(default_factory=dict)  # agentId -> locked tier
    tier_floors: Dict[str, str] = field(default_factory=dict)  # agentId -> min tier
    history_limit: int = 50
    flush_interval: float = 5.0

    @classmethod
    def from_dict(cls, d: Optional[Dict[str, Any]]) -> "TrustConfig":
        """Accepts both the resolved reference shape (config.ts resolveTrust:
        defaults / decay.{inactivityDays,rate} / persistIntervalSeconds /
        maxHistoryPerAgent) and the flat internal spelling."""
        d = d or {}
        defaults = d.get("initialScores") or d.get("defaults") or {}
        default_score = d.get("defaultScore")
        if default_score is None:
            default_score = defaults.get("*", 40)
        decay = d.get("decay") if isinstance(d.get("decay"), dict) else {}
        return cls(
            default_score=float(default_score),
            initial_scores={k: float(v) for k, v in defaults.items() if k != "*"},
            weights=dict(d.get("weights") or {}),
            decay_after_days=float(d.get("decayAfterDays", decay.get("inactivityDays", 30))),
            decay_factor=float(d.get("decayFactor", decay.get("rate", 0.95))),
            tier_locks=dict(d.get("tierLocks") or {}),
            tier_floors=dict(d.get("tierFloors") or {}),
            history_limit=int(d.get("historyLimit", d.get("maxHistoryPerAgent", 50))),
            flush_interval=float(d.get("flushIntervalSeconds", d.get("persistIntervalSeconds", 5))),
        )


TIER_FLOOR_SCORE = {"untrusted": 0, "restricted": 20, "standard": 40, "trusted": 60, "elevated": 80}


class TrustManager:
    """Agent trust store with the reference's score math + file format."""

    def __init__(
        self,
        config: Optional[TrustConfig] = None,
        workspace: str = ".",
        logger: Optional[PluginLogger] = None,
        clock=time.time,
    ):
        self.config = config or TrustConfig()
        self.workspace = workspace
        self.file_path = os.path.join(workspace, "governance", "trust.json")
        self.logger = logger or NullLogger()
        self.clock = clock
        self.store: Dict[str, Any] = {"version": 1, "updated": _iso_now(), "agents": {}}
        self.flusher = IntervalFlusher(self._persist, self.config.flush_interval)

    # -- lifecycle ---------------------------------------------------------
    def load(self) -> None:
        data = read_json(self.file_path)
        if isinstance(data, dict) and isinstance(data.get("agents"), dict):
            self.store = data
            self._drop_unknown_agent()
            self._apply_decay()
            self._migrate_fresh_agents()
            self._refresh_age_days()

    def _drop_unknown_agent(self) -> None:
        """Bug 3 (trust-manager.ts): activity recorded before agent-id
        resolution landed under the literal id 'unknown' — drop it on load
        so its stats never pollute a real agent."""
        ghost = self.store["agents"].pop("unknown", None)
        if ghost is not None:
            s = ghost.get("signals", {})
            self.logger.warn(
                "[trust] Trust migration: removing 'unknown' agent "
                "(%s successes, %s violations) — activity predates agent-id resolution",
                s.get("successCount", 0), s.get("violationCount", 0),
            )
            self.flusher.mark_dirty()

    def start(self) -> None:
        self.load()
        self.flusher.start()

    def stop(self) -> None:
        self.flusher.stop()

    def flush(self) -> None:
        self.flusher.flush()

    def _iso(self) -> str:
        return (
            _dt.datetime.fromtimestamp(self.clock(), _dt.timezone.utc)
            .isoformat()
            .replace("+00:00", "Z")
        )

    def _persist(self) -> None:
        self.store["updated"] = self._iso()
        atomic_write_json(self.file_path, self.store)

    # -- migrations / maintenance (trust-manager.ts:116-168) ---------------
    def _migrate_fresh_agents(self) -> None:
        for agent in self.store["agents"].values():
            s = agent.get("signals", {})
            if (
                not s.get("successCount")
                and not s.get("violationCount")
                and not s.get("cleanStreak")
                and not s.get("manualAdjustment")
                and agent.get("score", 0) > 0
            ):
                s["manualAdjustment"] = agent["score"]

    def _refresh_age_days(self) -> None:
        now_ms = self.clock() * 1000
        for agent in self.store["agents"].values():
            try:
                created = _dt.datetime.fromisoformat(agent["created"].replace("Z", "+00:00")).timestamp() * 1000
            except (KeyError, ValueError):
                continue
            agent["signals"]["ageDays"] = int((now_ms - created) // DAY_MS)

    def _apply_decay(self) -> None:
        """×decay_factor on score after decay_after_days of inactivity
        (trust-manager.ts:151-168); decay acts through manualAdjustment so
        the formula stays consistent."""
        now_ms = self.clock() * 1000
        horizon = self.config.decay_after_days * DAY_MS
        for agent in self.store["agents"].values():
            last = agent.get("lastActivity") or agent.get("lastEvaluation")
            try:
                last_ms = _dt.datetime.fromisoformat(str(last).replace("Z", "+00:00")).timestamp() * 1000
            except (TypeError, ValueError):
                continue
            if now_ms - last_ms > horizon:
                old = agent.get("score", 0)
                new = clamp(old * self.config.decay_factor, 0, 100)
                delta = new - compute_score(agent["signals"], self.config.weights)
                agent["signals"]["manualAdjustment"] = agent["signals"].get("manualAdjustment", 0) + delta
                self._recalculate(agent, "decay", f"inactive > {self.config.decay_after_days}d")

    # -- core API ----------------------------------------------------------
    def get(self, agent_id: str) -> Dict[str, Any]:
        agents = self.store["agents"]
        if agent_id not in agents:
            initial = self.config.initial_scores.get(agent_id, self.config.default_score)
            a = _new_agent(agent_id, initial)
            a["created"] = a["lastEvaluation"] = a["lastActivity"] = self._iso()
            agents[agent_id] = a
            self.flusher.mark_dirty()
        return agents[agent_id]

    def score(self, agent_id: str) -> float:
        return float(self.get(agent_id)["score"])

    def tier(self, agent_id: str) -> str:
        return str(self.get(agent_id)["tier"])

    def known_agents(self) -> List[str]:
        return list(self.store["agents"].keys())

    def _record_event(self, agent: Dict[str, Any], kind: str, reason: str, delta: float) -> None:
        hist: List[Any] = agent.setdefault("history", [])
        hist.append({"ts": self._iso(), "type": kind, "reason": reason, "delta": round(delta, 4)})
        limit = self.config.history_limit
        if len(hist) > limit:
            del hist[: len(hist) - limit]

    def _recalculate(self, agent: Dict[str, Any], kind: str, reason: str) -> None:
        old = agent.get("score", 0)
        new = compute_score(agent["signals"], self.config.weights)
        # tier lock / floor (trust-manager.ts:223-245)
        aid = agent["agentId"]
        lock = self.config.tier_locks.get(aid)
        if lock in TIER_FLOOR_SCORE:
            lo = TIER_FLOOR_SCORE[lock]
            hi = lo + 19.999 if lock != "elevated" else 100
            new = clamp(new, lo, hi)
        floor = self.config.tier_floors.get(aid)
        if floor in TIER_FLOOR_SCORE:
            new = max(new, TIER_FLOOR_SCORE[floor])
        # per-agent numeric floor / runtime tier lock (trust-manager.ts
        # setFloor/lockTier) — the floor also binds during decay
        if isinstance(agent.get("floor"), (int, float)):
            new = max(new, float(agent["floor"]))
        agent["score"] = round(new, 4)
        agent["tier"] = agent["locked"] if agent.get("locked") else score_to_tier(new)
        agent["lastEvaluation"] = self._iso()
        agent["lastActivity"] = agent["lastEvaluation"]
        self._record_event(agent, kind, reason, new - old)
        self.flusher.mark_dirty()

    def record_success(self, agent_id: str, n: int = 1) -> None:
        a = self.get(agent_id)
        a["signals"]["successCount"] = a["signals"].get("successCount", 0) + n
        self._recalculate(a, "success", f"+{n} successful action(s)")

    def record_violation(self, agent_id: str, reason: str = "policy violation") -> None:
        a = self.get(agent_id)
        a["signals"]["violationCount"] = a["signals"].get("violationCount", 0) + 1
        a["signals"]["cleanStreak"] = 0
        self._recalculate(a, "violation", reason)

    def record_clean_day(self, agent_id: str, days: int = 1) -> None:
        a = self.get(agent_id)
        a["signals"]["cleanStreak"] = a["signals"].get("cleanStreak", 0) + days
        self._recalculate(a, "clean_streak", f"+{days} clean day(s)")

    def adjust(self, agent_id: str, delta: float, reason: str = "manual adjustment") -> None:
        a = self.get(agent_id)
        a["signals"]["manualAdjustment"] = a["signals"].get("manualAdjustment", 0) + delta
        self._recalculate(a, "manual", reason)

    def set_score(self, agent_id: str, score: float, reason: str = "manual set") -> None:
        """Pin the score exactly (trust-manager.ts setScore): shifts
        manualAdjustment so the formula lands on `score`."""
        a = self.get(agent_id)
        target = clamp(score, 0, 100)
        current = compute_score(a["signals"], self.config.weights)
        a["signals"]["manualAdjustment"] = a["signals"].get("manualAdjustment", 0) + (target - current)
        self._recalculate(a, "manual", reason)

    def lock_tier(self, agent_id: str, tier: str) -> None:
        a = self.get(agent_id)
        a["locked"] = tier
        a["tier"] = tier
        self.flusher.mark_dirty()

    def unlock_tier(self, agent_id: str) -> None:
        a = self.get(agent_id)
        a.pop("locked", None)
        a["tier"] = score_to_tier(a.get("score", 0))
        self.flusher.mark_dirty()

    def set_floor(self, agent_id: str, floor: float) -> None:
        """Numeric per-agent floor; binds immediately and during decay."""
        a = self.get(agent_id)
        a["floor"] = float(floor)
        self._recalculate(a, "manual", f"floor set to {floor}")

    def reset_history(self, agent_id: str) -> None:
        a = self.get(agent_id)
        a["history"] = []
        a["signals"]["successCount"] = 0
        self.flusher.mark_dirty()

    def get_store(self) -> Dict[str, Any]:
        return self.store

    def snapshot(self) -> Dict[str, Any]:
        return {
            aid: {"score": a["score"], "tier": a["tier"], "signals": dict(a["signals"])}
            for aid, a in self.store["agents"].items()
        }
