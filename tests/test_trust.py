"""Trust math, session trust, tiers — golden vectors from the reference
formula (trust-manager.ts:30-43, session-trust-manager.ts, util.ts:192-198)."""

import json
import os

import pytest

from vainplex_openclaw_amd.governance.trust import TrustManager, TrustConfig, compute_score
from vainplex_openclaw_amd.governance.session_trust import SessionTrustManager, SessionTrustConfig
from vainplex_openclaw_amd.governance.util import score_to_tier, tier_ordinal, extract_agent_id, resolve_agent_id


# table-style golden vectors: (signals, expected score)
SCORE_VECTORS = [
    ({"ageDays": 0, "successCount": 0, "violationCount": 0, "cleanStreak": 0, "manualAdjustment": 0}, 0),
    ({"ageDays": 10, "successCount": 0, "violationCount": 0, "cleanStreak": 0, "manualAdjustment": 0}, 5),
    ({"ageDays": 100, "successCount": 0, "violationCount": 0, "cleanStreak": 0, "manualAdjustment": 0}, 20),  # age capped
    ({"ageDays": 0, "successCount": 100, "violationCount": 0, "cleanStreak": 0, "manualAdjustment": 0}, 10),
    ({"ageDays": 0, "successCount": 1000, "violationCount": 0, "cleanStreak": 0, "manualAdjustment": 0}, 30),  # capped
    ({"ageDays": 0, "successCount": 0, "violationCount": 3, "cleanStreak": 0, "manualAdjustment": 20}, 14),
    ({"ageDays": 0, "successCount": 0, "violationCount": 0, "cleanStreak": 100, "manualAdjustment": 0}, 20),  # capped
    ({"ageDays": 40, "successCount": 300, "violationCount": 0, "cleanStreak": 66.67, "manualAdjustment": 30}, 100),  # clamp
    ({"ageDays": 0, "successCount": 0, "violationCount": 60, "cleanStreak": 0, "manualAdjustment": 40}, 0),  # clamp low
]


@pytest.mark.parametrize("signals,expected", SCORE_VECTORS)
def test_compute_score_golden(signals, expected):
    assert compute_score(signals) == pytest.approx(expected, abs=0.01)


def test_score_to_tier_boundaries():
    assert score_to_tier(80) == "elevated"
    assert score_to_tier(79.9) == "trusted"
    assert score_to_tier(60) == "trusted"
    assert score_to_tier(59.9) == "standard"
    assert score_to_tier(40) == "standard"
    assert score_to_tier(20) == "restricted"
    assert score_to_tier(19.9) == "untrusted"
    assert tier_ordinal("elevated") == 4
    assert tier_ordinal("untrusted") == 0


def test_trust_manager_lifecycle(workspace):
    tm = TrustManager(TrustConfig(default_score=40), workspace)
    agent = tm.get("alpha")
    assert agent["score"] == 40
    assert agent["tier"] == "standard"
    assert agent["signals"]["manualAdjustment"] == 40  # fresh-agent backfill

    tm.record_violation("alpha")
    assert tm.score("alpha") == 38
    tm.record_success("alpha", 10)
    assert tm.score("alpha") == pytest.approx(39)

    tm.flush()
    path = os.path.join(workspace, "governance", "trust.json")
    assert os.path.isfile(path)
    with open(path) as fh:
        store = json.load(fh)
    assert store["version"] == 1
    assert "alpha" in store["agents"]

    # reload round-trip
    tm2 = TrustManager(TrustConfig(), workspace)
    tm2.load()
    assert tm2.score("alpha") == pytest.approx(39)


def test_trust_decay_after_inactivity(workspace):
    t = [1_000_000.0]
    tm = TrustManager(TrustConfig(default_score=60), workspace, clock=lambda: t[0])
    tm.get("lazy")
    tm.flush()
    t[0] += 40 * 86400  # 40 days later
    tm2 = TrustManager(TrustConfig(default_score=60), workspace, clock=lambda: t[0])
    tm2.load()
    assert tm2.score("lazy") == pytest.approx(60 * 0.95, abs=0.1)


def test_tier_lock_and_floor(workspace):
    cfg = TrustConfig(default_score=90, tier_locks={"locked": "standard"}, tier_floors={"floored": "trusted"})
    tm = TrustManager(cfg, workspace)
    tm.get("locked")
    tm.record_success("locked", 100)
    assert tm.tier("locked") == "standard"
    tm2 = TrustManager(TrustConfig(default_score=10, tier_floors={"floored": "trusted"}), workspace)
    tm2.get("floored")
    tm2.record_violation("floored")
    assert tm2.score("floored") >= 60


def test_session_trust_seed_and_ceiling(workspace):
    tm = TrustManager(TrustConfig(default_score=50), workspace)
    stm = SessionTrustManager(SessionTrustConfig(), tm)
    st = stm.initialize("s1", "agent-a")
    assert st["score"] == 35  # floor(50 * 0.7)
    # ceiling: floor(50*1.2)=60
    stm.set_score("s1", "agent-a", 1000)
    assert stm.get("s1", "agent-a")["score"] == 60
    stm.set_score("s1", "agent-a", -5)
    assert stm.get("s1", "agent-a")["score"] == 0


def test_session_trust_signals_and_streak(workspace):
    tm = TrustManager(TrustConfig(default_score=50), workspace)
    cfg = SessionTrustConfig()
    cfg.signals.update({"success": 1, "cleanStreakThreshold": 3, "cleanStreakBonus": 5, "policyBlock": -5})
    stm = SessionTrustManager(cfg, tm)
    stm.initialize("s1", "a")
    s0 = stm.get("s1", "a")["score"]
    stm.apply_signal("s1", "a", "success")
    stm.apply_signal("s1", "a", "success")
    assert stm.get("s1", "a")["cleanStreak"] == 2
    stm.apply_signal("s1", "a", "success")  # hits threshold: +1+5, streak reset
    assert stm.get("s1", "a")["cleanStreak"] == 0
    assert stm.get("s1", "a")["score"] == min(s0 + 8, 60)
    stm.apply_signal("s1", "a", "policyBlock")
    assert stm.get("s1", "a")["cleanStreak"] == 0


def test_session_eviction(workspace):
    tm = TrustManager(TrustConfig(), workspace)
    t = [0.0]
    stm = SessionTrustManager(SessionTrustConfig(), tm, clock=lambda: t[0])
    for i in range(501):
        t[0] += 1
        stm.initialize(f"s{i}", "a")
    assert len(stm.sessions()) == 500
    assert "s0" not in stm.sessions()  # oldest evicted


def test_agent_id_resolution():
    assert extract_agent_id("agent:main") == "main"
    assert extract_agent_id("agent:main:subagent:forge:abc123") == "forge"
    assert extract_agent_id(None, "explicit") == "explicit"
    assert extract_agent_id(None) == "unknown"
    assert resolve_agent_id({"sessionKey": "agent:main"}) == "main"
    assert resolve_agent_id({}) == "unresolved"
    assert resolve_agent_id({}, {"metadata": {"agentId": "meta"}}) == "meta"
