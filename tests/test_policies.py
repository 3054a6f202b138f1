"""Policy evaluation: conditions, builtins, aggregation, risk — golden
semantics from policy-evaluator.ts / conditions/* / builtin-policies.ts /
risk-assessor.ts."""

import pytest

from vainplex_openclaw_amd.governance.conditions import (
    ConditionDeps,
    evaluate_condition,
    glob_to_regex,
    is_in_time_range,
    parse_time_to_minutes,
)
from vainplex_openclaw_amd.governance.evaluator import PolicyEvaluator, aggregate_matches, sort_policies
from vainplex_openclaw_amd.governance.frequency import FrequencyTracker
from vainplex_openclaw_amd.governance.policies import build_policy_index, get_builtin_policies
from vainplex_openclaw_amd.governance.risk import RiskAssessor, score_to_risk_level


def ctx_base(**kw):
    ctx = {
        "hook": "before_tool_call",
        "agentId": "a1",
        "sessionKey": "agent:a1",
        "toolName": "exec",
        "toolParams": {"command": "ls"},
        "time": {"hour": 12, "minute": 0, "dayOfWeek": "Mon"},
        "trust": {
            "agent": {"score": 50, "tier": "standard"},
            "session": {"score": 35, "tier": "untrusted"},
        },
    }
    ctx.update(kw)
    return ctx


def test_glob_and_time_utils():
    assert glob_to_regex("mem*").match("memory_search")
    assert not glob_to_regex("mem*").match("x_memory")
    assert glob_to_regex("a?c").match("abc")
    assert parse_time_to_minutes("23:00") == 1380
    assert parse_time_to_minutes("bogus") == -1
    assert is_in_time_range(100, 50, 200)
    # midnight wrap 23:00-06:00
    assert is_in_time_range(1400, 1380, 360)
    assert is_in_time_range(100, 1380, 360)
    assert not is_in_time_range(700, 1380, 360)


def test_tool_condition_matchers():
    deps = ConditionDeps()
    ctx = ctx_base(toolParams={"command": "cat .env", "n": 5})
    assert evaluate_condition({"type": "tool", "name": "exec"}, ctx, deps)
    assert evaluate_condition({"type": "tool", "name": ["read", "exec"]}, ctx, deps)
    assert not evaluate_condition({"type": "tool", "name": "read"}, ctx, deps)
    assert evaluate_condition({"type": "tool", "params": {"command": {"contains": ".env"}}}, ctx, deps)
    assert evaluate_condition({"type": "tool", "params": {"command": {"matches": r"cat.*\.env"}}}, ctx, deps)
    assert evaluate_condition({"type": "tool", "params": {"command": {"startsWith": "cat"}}}, ctx, deps)
    assert evaluate_condition({"type": "tool", "params": {"n": {"equals": 5}}}, ctx, deps)
    assert evaluate_condition({"type": "tool", "params": {"n": {"in": [4, 5]}}}, ctx, deps)
    assert not evaluate_condition({"type": "tool", "params": {"missing": {"contains": "x"}}}, ctx, deps)


def test_time_condition_inline_and_window():
    deps = ConditionDeps(time_windows={"night": {"start": "23:00", "end": "06:00"}})
    ctx = ctx_base(time={"hour": 23, "minute": 30, "dayOfWeek": "Mon"})
    assert evaluate_condition({"type": "time", "after": "23:00", "before": "08:00"}, ctx, deps)
    assert evaluate_condition({"type": "time", "window": "night"}, ctx, deps)
    day_ctx = ctx_base(time={"hour": 12, "minute": 0, "dayOfWeek": "Mon"})
    assert not evaluate_condition({"type": "time", "after": "23:00", "before": "08:00"}, day_ctx, deps)
    assert evaluate_condition({"type": "time", "days": ["Mon"]}, day_ctx, deps)
    assert not evaluate_condition({"type": "time", "days": ["Sun"]}, day_ctx, deps)


def test_agent_risk_frequency_composite_conditions():
    freq = FrequencyTracker()
    for _ in range(5):
        freq.record("a1", "agent:a1", "exec")
    deps = ConditionDeps(risk={"level": "high", "score": 60}, frequency_tracker=freq)
    ctx = ctx_base()
    assert evaluate_condition({"type": "agent", "id": "a1"}, ctx, deps)
    assert evaluate_condition({"type": "agent", "id": "a*"}, ctx, deps)
    assert evaluate_condition({"type": "agent", "trustTier": ["standard"]}, ctx, deps)  # agent tier, not session
    assert evaluate_condition({"type": "agent", "minScore": 40}, ctx, deps)
    assert not evaluate_condition({"type": "agent", "maxScore": 40}, ctx, deps)
    assert evaluate_condition({"type": "risk", "minRisk": "medium"}, ctx, deps)
    assert not evaluate_condition({"type": "risk", "maxRisk": "medium"}, ctx, deps)
    assert evaluate_condition({"type": "frequency", "maxCount": 5, "windowSeconds": 60}, ctx, deps)
    assert not evaluate_condition({"type": "frequency", "maxCount": 6, "windowSeconds": 60}, ctx, deps)
    assert evaluate_condition(
        {"type": "any", "conditions": [{"type": "agent", "id": "zzz"}, {"type": "agent", "id": "a1"}]}, ctx, deps
    )
    assert evaluate_condition({"type": "not", "condition": {"type": "agent", "id": "zzz"}}, ctx, deps)


def test_context_condition():
    deps = ConditionDeps()
    ctx = ctx_base(messageContent="please deploy to prod", channel="slack", metadata={"k": 1})
    assert evaluate_condition({"type": "context", "messageContains": "deploy"}, ctx, deps)
    assert evaluate_condition({"type": "context", "messageContains": ["nope", "dep.oy"]}, ctx, deps)
    assert not evaluate_condition({"type": "context", "messageContains": "xyz"}, ctx, deps)
    assert evaluate_condition({"type": "context", "channel": ["slack"]}, ctx, deps)
    assert evaluate_condition({"type": "context", "hasMetadata": "k"}, ctx, deps)
    assert evaluate_condition({"type": "context", "sessionKey": "agent:*"}, ctx, deps)


def test_aggregation_precedence():
    def m(action):
        return {"policyId": "p", "ruleId": "r", "effect": {"action": action}, "controls": []}

    assert aggregate_matches([m("allow"), m("deny"), m("2fa")])["action"] == "deny"
    assert aggregate_matches([m("allow"), m("2fa")])["action"] == "2fa"
    assert aggregate_matches([m("audit"), m("allow")])["action"] == "allow"
    assert aggregate_matches([m("audit")])["reason"] == "Allowed with audit logging"
    assert aggregate_matches([])["reason"] == "No matching policies"


def test_sort_by_priority_then_specificity():
    p1 = {"id": "1", "priority": 50, "scope": {}}
    p2 = {"id": "2", "priority": 100, "scope": {}}
    p3 = {"id": "3", "priority": 100, "scope": {"agents": ["a"], "hooks": ["h"]}}
    assert [p["id"] for p in sort_policies([p1, p2, p3])] == ["3", "2", "1"]


def test_builtin_night_mode():
    policies = get_builtin_policies({"nightMode": True})
    ev = PolicyEvaluator()
    night = ctx_base(time={"hour": 23, "minute": 30, "dayOfWeek": "Mon"})
    res = ev.evaluate(night, policies, {"level": "low", "score": 0})
    assert res["action"] == "deny"
    assert "Night mode" in res["reason"]
    # read-only tool allowed at night
    night_read = ctx_base(toolName="read", time={"hour": 23, "minute": 30, "dayOfWeek": "Mon"})
    assert ev.evaluate(night_read, policies, {"level": "low", "score": 0})["action"] == "allow"
    # daytime allowed
    assert ev.evaluate(ctx_base(), policies, {"level": "low", "score": 0})["action"] == "allow"


def test_builtin_credential_guard():
    policies = get_builtin_policies({"credentialGuard": True})
    ev = PolicyEvaluator()
    bad = ctx_base(toolName="read", toolParams={"file_path": "/app/.env"})
    assert ev.evaluate(bad, policies, {"level": "low", "score": 0})["action"] == "deny"
    bad2 = ctx_base(toolParams={"command": "cat secrets/prod.pem"})
    assert ev.evaluate(bad2, policies, {"level": "low", "score": 0})["action"] == "deny"
    ok = ctx_base(toolName="read", toolParams={"file_path": "/app/readme.md"})
    assert ev.evaluate(ok, policies, {"level": "low", "score": 0})["action"] == "allow"


def test_builtin_production_safeguard_tier_exemption():
    policies = get_builtin_policies({"productionSafeguard": True})
    ev = PolicyEvaluator()
    cmd = {"command": "git push origin main"}
    low = ctx_base(toolParams=cmd)
    assert ev.evaluate(low, policies, {"level": "low", "score": 0})["action"] == "deny"
    trusted = ctx_base(
        toolParams=cmd,
        trust={"agent": {"score": 70, "tier": "trusted"}, "session": {"score": 49, "tier": "standard"}},
    )
    assert ev.evaluate(trusted, policies, {"level": "low", "score": 0})["action"] == "allow"


def test_builtin_rate_limiter():
    policies = get_builtin_policies({"rateLimiter": {"maxPerMinute": 3}})
    freq = FrequencyTracker()
    ev = PolicyEvaluator()
    deps_ctx = ctx_base()
    from vainplex_openclaw_amd.governance.conditions import ConditionDeps as CD

    deps = CD(risk={"level": "low", "score": 0}, frequency_tracker=freq)
    assert ev.evaluate(deps_ctx, policies, deps=deps)["action"] == "allow"
    for _ in range(3):
        freq.record("a1", "agent:a1", "exec")
    assert ev.evaluate(deps_ctx, policies, deps=deps)["action"] == "deny"
    # trusted gets 2x
    trusted = ctx_base(trust={"agent": {"score": 70, "tier": "trusted"}, "session": {"score": 60, "tier": "trusted"}})
    assert ev.evaluate(trusted, policies, deps=deps)["action"] == "allow"
    for _ in range(3):
        freq.record("a1", "agent:a1", "exec")
    assert ev.evaluate(trusted, policies, deps=deps)["action"] == "deny"


def test_min_max_trust_rule_guards():
    ev = PolicyEvaluator()
    policy = {
        "id": "p",
        "priority": 1,
        "scope": {},
        "rules": [
            {"id": "r", "minTrust": "trusted", "conditions": [{"type": "tool", "name": "exec"}], "effect": {"action": "deny"}}
        ],
    }
    low = ctx_base()  # session tier untrusted
    assert ev.evaluate(low, [policy], {"level": "low", "score": 0})["action"] == "allow"
    high = ctx_base(trust={"agent": {"score": 70, "tier": "trusted"}, "session": {"score": 65, "tier": "trusted"}})
    assert ev.evaluate(high, [policy], {"level": "low", "score": 0})["action"] == "deny"


def test_scope_filtering():
    ev = PolicyEvaluator()
    policy = {
        "id": "p",
        "scope": {"excludeAgents": ["a1"]},
        "rules": [{"id": "r", "conditions": [], "effect": {"action": "deny"}}],
    }
    assert ev.evaluate(ctx_base(), [policy], {"level": "low", "score": 0})["action"] == "allow"
    other = ctx_base(agentId="a2")
    assert ev.evaluate(other, [policy], {"level": "low", "score": 0})["action"] == "deny"


def test_risk_assessor_factors():
    ra = RiskAssessor({})
    freq = FrequencyTracker()
    ctx = ctx_base()  # exec risk 70, hour 12, session 35
    res = ra.assess(ctx, freq)
    # tool 70/100*30=21 + time 0 + trust (100-35)/100*20=13 + freq 0 + scope 0 = 34
    assert res["score"] == 34
    assert res["level"] == "medium"
    off = ctx_base(time={"hour": 23, "minute": 0, "dayOfWeek": "Mon"}, messageTo="x@y.z")
    res2 = ra.assess(off, freq)
    assert res2["score"] == 34 + 15 + 20
    assert res2["level"] == "high"
    assert score_to_risk_level(76) == "critical"
    assert score_to_risk_level(25) == "low"


def test_policy_index_by_hook_and_agent():
    idx = build_policy_index({"builtinPolicies": {"credentialGuard": True, "rateLimiter": True}})
    assert any(p["id"] == "builtin-credential-guard" for p in idx.for_hook("before_tool_call"))
    assert idx.by_id("builtin-rate-limiter") is not None
    assert "builtin-credential-guard" in [p["id"] for p in idx.by_agent["*"]]


# -- policy-evaluator.test.ts mirrors ------------------------------------

def test_channel_scope():
    ev = PolicyEvaluator()
    policy = {"id": "p", "scope": {"channels": ["slack"]},
              "rules": [{"id": "r", "conditions": [], "effect": {"action": "deny"}}]}
    assert ev.evaluate(ctx_base(channel="slack"), [policy], {"level": "low", "score": 0})["action"] == "deny"
    assert ev.evaluate(ctx_base(channel="email"), [policy], {"level": "low", "score": 0})["action"] == "allow"
    # no channel in ctx -> scoped policy skipped
    assert ev.evaluate(ctx_base(), [policy], {"level": "low", "score": 0})["action"] == "allow"


def test_max_trust_rule_guard():
    ev = PolicyEvaluator()
    policy = {"id": "p", "priority": 1, "scope": {}, "rules": [
        {"id": "r", "maxTrust": "standard",
         "conditions": [{"type": "tool", "name": "exec"}], "effect": {"action": "deny"}}]}
    low = ctx_base()  # session untrusted <= standard: rule applies
    assert ev.evaluate(low, [policy], {"level": "low", "score": 0})["action"] == "deny"
    high = ctx_base(trust={"agent": {"score": 70, "tier": "trusted"},
                           "session": {"score": 65, "tier": "trusted"}})
    assert ev.evaluate(high, [policy], {"level": "low", "score": 0})["action"] == "allow"


def test_first_match_within_policy():
    ev = PolicyEvaluator()
    policy = {"id": "p", "scope": {}, "rules": [
        {"id": "r1", "conditions": [{"type": "tool", "name": "exec"}], "effect": {"action": "audit"}},
        {"id": "r2", "conditions": [], "effect": {"action": "deny"}},
    ]}
    res = ev.evaluate(ctx_base(), [policy], {"level": "low", "score": 0})
    # r1 matched first -> audit, r2 never consulted
    assert res["action"] == "allow"
    assert res["matches"][0]["ruleId"] == "r1"
    # different tool -> falls through to r2
    res2 = ev.evaluate(ctx_base(toolName="write"), [policy], {"level": "low", "score": 0})
    assert res2["action"] == "deny" and res2["matches"][0]["ruleId"] == "r2"


def test_deny_wins_across_policies():
    ev = PolicyEvaluator()
    allow_p = {"id": "a", "priority": 100, "scope": {},
               "rules": [{"id": "r", "conditions": [], "effect": {"action": "allow"}}]}
    deny_p = {"id": "d", "priority": 1, "scope": {},
              "rules": [{"id": "r", "conditions": [], "effect": {"action": "deny",
                         "reason": "blocked"}}]}
    res = ev.evaluate(ctx_base(), [allow_p, deny_p], {"level": "low", "score": 0})
    assert res["action"] == "deny" and res["reason"] == "blocked"


def test_controls_propagate_into_matches():
    """Bug 4: ISO/SOC controls on the policy ride along on each match."""
    ev = PolicyEvaluator()
    with_controls = {"id": "p", "scope": {}, "controls": ["SOC2-CC6.1", "A.5.15"],
                     "rules": [{"id": "r", "conditions": [], "effect": {"action": "audit"}}]}
    res = ev.evaluate(ctx_base(), [with_controls], {"level": "low", "score": 0})
    assert res["matches"][0]["controls"] == ["SOC2-CC6.1", "A.5.15"]
    without = {"id": "p2", "scope": {},
               "rules": [{"id": "r", "conditions": [], "effect": {"action": "audit"}}]}
    res2 = ev.evaluate(ctx_base(), [without], {"level": "low", "score": 0})
    assert res2["matches"][0]["controls"] == []


# ===========================================================================
# policy-loader.test.ts depth: file loading, disabled filter, merge,
# hook/agent indexing, regex cache robustness
# ===========================================================================

def _pol(pid, hooks=None, agents=None, enabled=True, action="deny", pattern=None):
    cond = []
    if pattern is not None:
        cond = [{"type": "tool", "params": {"cmd": {"matches": pattern}}}]
    p = {"id": pid, "enabled": enabled,
         "rules": [{"id": "r", "conditions": cond, "effect": {"action": action}}]}
    scope = {}
    if hooks is not None:
        scope["hooks"] = hooks
    if agents is not None:
        scope["agents"] = agents
    if scope:
        p["scope"] = scope
    return p


def test_loader_policy_dir_files(tmp_path):
    import json

    from vainplex_openclaw_amd.governance.policies import load_policies

    d = tmp_path / "policies"
    d.mkdir()
    (d / "one.json").write_text(json.dumps(_pol("from-file")))
    (d / "many.json").write_text(json.dumps([_pol("list-a"), _pol("list-b")]))
    (d / "junk.json").write_text("{broken")
    (d / "noid.json").write_text(json.dumps({"rules": []}))
    (d / "notes.txt").write_text("ignored")
    try:
        pols = load_policies({"policies": [_pol("inline")]}, policy_dir=str(d))
    except Exception as exc:
        import pytest as _pt

        _pt.fail(f"loader must tolerate junk files: {exc}")
    ids = {p["id"] for p in pols}
    assert {"inline", "from-file", "list-a", "list-b"} <= ids
    assert "noid" not in ids


def test_loader_disabled_policies_filtered_at_index(tmp_path):
    from vainplex_openclaw_amd.governance.policies import build_policy_index

    idx = build_policy_index({"policies": [
        _pol("on", hooks=["before_tool_call"]),
        _pol("off", hooks=["before_tool_call"], enabled=False),
    ]})
    hooked = {p["id"] for p in idx.for_hook("before_tool_call")}
    assert "on" in hooked and "off" not in hooked


def test_index_by_hook_and_global(tmp_path):
    from vainplex_openclaw_amd.governance.policies import build_policy_index

    idx = build_policy_index({"policies": [
        _pol("tool-only", hooks=["before_tool_call"]),
        _pol("msg-only", hooks=["message_sending"]),
        _pol("everywhere"),                       # no hooks = all hooks
    ]})
    t = {p["id"] for p in idx.for_hook("before_tool_call")}
    m = {p["id"] for p in idx.for_hook("message_sending")}
    assert t == {"tool-only", "everywhere"}
    assert m == {"msg-only", "everywhere"}


def test_index_regex_cache_and_invalid_regex(tmp_path):
    from vainplex_openclaw_amd.governance.policies import build_policy_index

    idx = build_policy_index({"policies": [
        _pol("good", hooks=["before_tool_call"], pattern=r"rm\s+-rf"),
        _pol("bad", hooks=["before_tool_call"], pattern="(unclosed"),
    ]})
    assert idx.regex_cache.get(r"rm\s+-rf") is not None
    assert idx.regex_cache.get("(unclosed") is None  # cached as invalid, no crash


# -- audit-trail query depth -------------------------------------------------

def test_audit_query_filters(workspace):
    from vainplex_openclaw_amd.governance.audit import AuditTrail

    t = [1_700_000_000.0]
    trail = AuditTrail({}, workspace, clock=lambda: t[0])
    trail.load()
    for i, (agent, verdict) in enumerate([("a1", "allow"), ("a1", "deny"),
                                          ("a2", "allow"), ("a2", "2fa")]):
        t[0] += 10
        trail.record(verdict, "r", {"hook": "h", "agentId": agent}, {}, {}, [], 5)
    trail.flush()
    assert len(trail.query({"agentId": "a1"})) == 2
    assert len(trail.query({"verdict": "allow"})) == 2
    assert len(trail.query({"agentId": "a2", "verdict": "2fa"})) == 1
    assert len(trail.query({"limit": 3})) == 3
    # time-range filter
    mid = 1_700_000_000.0 + 25
    after = trail.query({"after": mid * 1000})
    assert all(r["timestamp"] >= mid * 1000 for r in after)
    assert len(trail.query({})) == 4
