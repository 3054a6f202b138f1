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


# -- reference trust-manager.test.ts mirrors ------------------------------

def _write_store(workspace, agents):
    import datetime
    path = os.path.join(workspace, "governance", "trust.json")
    os.makedirs(os.path.dirname(path), exist_ok=True)
    with open(path, "w") as fh:
        json.dump({"version": 1, "updated": "2020-01-01T00:00:00Z", "agents": agents}, fh)
    return path


def _agent_record(aid, score, ts, signals=None, **extra):
    rec = {
        "agentId": aid, "score": score,
        "tier": score_to_tier(score),
        "signals": {"successCount": 0, "violationCount": 0, "ageDays": 0,
                    "cleanStreak": 0, "manualAdjustment": 0, **(signals or {})},
        "history": [], "lastEvaluation": ts, "created": ts, "lastActivity": ts,
    }
    rec.update(extra)
    return rec


def test_set_score_manually(workspace):
    tm = TrustManager(TrustConfig(default_score=10), workspace)
    tm.set_score("test", 75)
    a = tm.get("test")
    assert a["score"] == 75
    assert a["tier"] == "trusted"
    # survives a recalculate
    tm.record_success("test")
    assert tm.score("test") == pytest.approx(75.1)


def test_set_score_maps_all_tier_ranges(workspace):
    tm = TrustManager(TrustConfig(), workspace)
    for score, tier in [(5, "untrusted"), (25, "restricted"), (45, "standard"),
                        (65, "trusted"), (85, "elevated")]:
        tm.set_score(f"t{score}", score)
        assert tm.tier(f"t{score}") == tier


def test_runtime_lock_unlock_tier(workspace):
    tm = TrustManager(TrustConfig(default_score=10), workspace)
    tm.lock_tier("test", "elevated")
    a = tm.get("test")
    assert a["tier"] == "elevated" and a["locked"] == "elevated"
    tm.record_success("test")          # recalc keeps the locked tier
    assert tm.tier("test") == "elevated"
    tm.unlock_tier("test")
    assert "locked" not in tm.get("test")
    assert tm.tier("test") == score_to_tier(tm.score("test"))


def test_set_floor_binds_now_and_on_decay(workspace):
    tm = TrustManager(TrustConfig(default_score=10), workspace)
    tm.set_floor("test", 30)
    a = tm.get("test")
    assert a["floor"] == 30 and a["score"] == 30

    # floor respected while decaying (reference: 50*0.95=47.5 -> floor 48)
    stale = "2020-01-01T00:00:00Z"
    _write_store(workspace, {"floored": _agent_record("floored", 50, stale, floor=48)})
    t = [1_900_000_000.0]
    tm2 = TrustManager(TrustConfig(), workspace, clock=lambda: t[0])
    tm2.load()
    assert tm2.get("floored")["score"] == 48


def test_history_trim_and_reset(workspace):
    tm = TrustManager(TrustConfig(history_limit=5), workspace)
    for _ in range(10):
        tm.record_success("test")
    assert len(tm.get("test")["history"]) <= 5
    tm.reset_history("test")
    assert tm.get("test")["history"] == []
    assert tm.get("test")["signals"]["successCount"] == 0


def test_get_store(workspace):
    tm = TrustManager(TrustConfig(), workspace)
    tm.get("main")
    store = tm.get_store()
    assert store["version"] == 1 and "main" in store["agents"]


def test_unknown_agent_removed_on_load_with_warning(workspace):
    from vainplex_openclaw_amd.core.api import PluginLogger

    _write_store(workspace, {
        "unknown": _agent_record("unknown", 20, "2020-01-01T00:00:00Z",
                                 signals={"successCount": 340, "violationCount": 32}),
        "main": _agent_record("main", 60, "2020-01-01T00:00:00Z"),
    })
    warnings = []
    logger = PluginLogger("t", lambda lvl, msg: warnings.append(msg) if lvl == "warn" else None)
    tm = TrustManager(TrustConfig(), workspace, logger=logger)
    tm.load()
    assert "unknown" not in tm.get_store()["agents"]
    assert "main" in tm.get_store()["agents"]
    assert any("Trust migration" in w for w in warnings)
    assert any("340 successes" in w for w in warnings)


def test_fresh_agent_migration_on_load(workspace):
    # fresh agent (no activity, manualAdjustment=0) gets its score
    # backfilled into manualAdjustment; active agents are left alone
    _write_store(workspace, {
        "fresh": _agent_record("fresh", 60, "2020-01-01T00:00:00Z"),
        "active": _agent_record("active", 55, "2020-01-01T00:00:00Z",
                                signals={"successCount": 100, "manualAdjustment": 45}),
    })
    t = [1_577_836_800.0 + 3600]  # 1h after the store timestamp: no decay
    tm = TrustManager(TrustConfig(), workspace, clock=lambda: t[0])
    tm.load()
    assert tm.get_store()["agents"]["fresh"]["signals"]["manualAdjustment"] == 60
    assert tm.get_store()["agents"]["active"]["signals"]["manualAdjustment"] == 45
    # and the migrated score survives a recalculate
    tm.record_success("fresh")
    assert tm.score("fresh") == pytest.approx(60.1, abs=0.2)


def test_wildcard_default_for_unknown_agents(workspace):
    cfg = TrustConfig.from_dict({"initialScores": {"*": 33, "vip": 80}})
    tm = TrustManager(cfg, workspace)
    assert tm.score("random-agent") == 33
    assert tm.score("vip") == 80


def test_age_days_refreshed_on_load(workspace):
    _write_store(workspace, {
        "aged": _agent_record("aged", 40, "2020-01-01T00:00:00Z",
                              signals={"manualAdjustment": 40}),
    })
    # 2020-01-01 + 10 days
    t = [1_577_836_800.0 + 10 * 86400]
    tm = TrustManager(TrustConfig(), workspace, clock=lambda: t[0])
    tm.load()
    assert tm.get_store()["agents"]["aged"]["signals"]["ageDays"] == 10


# ===========================================================================
# cross-agent.test.ts depth (13 its)
# ===========================================================================

def _ca(workspace):
    from vainplex_openclaw_amd.governance.cross_agent import CrossAgentManager
    from vainplex_openclaw_amd.governance.trust import TrustConfig, TrustManager

    tm = TrustManager(TrustConfig.from_dict(None), workspace)
    tm.load()
    return CrossAgentManager(tm), tm


PARENT = "agent:main"
CHILD = "agent:main:subagent:forge:abc"


def test_cross_agent_relationship_lifecycle(workspace):
    ca, _ = _ca(workspace)
    ca.register_relationship(PARENT, CHILD)
    p = ca.get_parent(CHILD)
    assert p and p["parentAgentId"] == "main"
    assert [c["childAgentId"] for c in ca.get_children(PARENT)] == ["forge"]
    ca.remove_relationship(CHILD)
    assert ca.get_children(PARENT) == []


def test_cross_agent_root_has_no_parent(workspace):
    ca, _ = _ca(workspace)
    assert ca.get_parent(PARENT) is None


def test_cross_agent_enrich_sub_vs_root(workspace):
    ca, _ = _ca(workspace)
    ca.register_relationship(PARENT, CHILD)
    sub = ca.enrich_context({"sessionKey": CHILD, "agentId": "forge"})
    assert sub.get("crossAgent", {}).get("parentAgentId") == "main"
    root = ca.enrich_context({"sessionKey": PARENT, "agentId": "main"})
    assert "crossAgent" not in root or not root["crossAgent"].get("parentAgentId")


def test_cross_agent_trust_ceiling_caps_child_only(workspace):
    ca, tm = _ca(workspace)
    tm.set_score("main", 55.0)
    tm.set_score("forge", 90.0)
    ca.register_relationship(PARENT, CHILD)
    assert ca.compute_trust_ceiling(CHILD) <= 55.0
    # root agents are uncapped
    assert ca.compute_trust_ceiling(PARENT) >= 90.0 or \
        ca.compute_trust_ceiling(PARENT) == tm.get("main")["score"]


def test_cross_agent_policy_cascade(workspace):
    from vainplex_openclaw_amd.governance.policies import build_policy_index

    ca, _ = _ca(workspace)
    ca.register_relationship(PARENT, CHILD)
    idx = build_policy_index({"policies": [
        {"id": "parent-only", "scope": {"agents": ["main"], "hooks": ["before_tool_call"]},
         "rules": [{"id": "r", "conditions": [], "effect": {"action": "deny"}}]},
        {"id": "global", "scope": {"hooks": ["before_tool_call"]},
         "rules": [{"id": "r", "conditions": [], "effect": {"action": "audit"}}]},
        {"id": "other-agent", "scope": {"agents": ["zeta"], "hooks": ["before_tool_call"]},
         "rules": [{"id": "r", "conditions": [], "effect": {"action": "deny"}}]},
    ]})
    # sub-agent inherits parent-scoped + global, not other agents'
    sub_pols = {p["id"] for p in ca.resolve_effective_policies(
        {"sessionKey": CHILD, "agentId": "forge", "hook": "before_tool_call"}, idx)}
    assert "parent-only" in sub_pols and "global" in sub_pols
    assert "other-agent" not in sub_pols
    # root gets only own + global
    root_pols = {p["id"] for p in ca.resolve_effective_policies(
        {"sessionKey": "agent:zeta", "agentId": "zeta", "hook": "before_tool_call"}, idx)}
    assert root_pols == {"other-agent", "global"}


def test_cross_agent_implicit_subagent_detection(workspace):
    ca, _ = _ca(workspace)
    # no explicit registration: the session key alone identifies the parent
    ctx = ca.enrich_context({"sessionKey": CHILD, "agentId": "forge"})
    assert ctx.get("crossAgent", {}).get("parentAgentId") == "main"


def test_cross_agent_graph_summary(workspace):
    ca, _ = _ca(workspace)
    ca.register_relationship(PARENT, CHILD)
    ca.register_relationship(PARENT, "agent:main:subagent:scout:xyz")
    g = ca.graph_summary()
    assert len(g["relationships"]) == 2
    kids = {r["childAgentId"] for r in g["relationships"]}
    assert kids == {"forge", "scout"}
