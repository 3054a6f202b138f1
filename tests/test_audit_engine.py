"""Audit trail (JSONL + Merkle), engine pipeline, output validation."""

import hashlib
import json
import os

import pytest

from vainplex_openclaw_amd.governance.audit import AuditTrail, derive_controls, merkle_root
from vainplex_openclaw_amd.governance.engine import GovernanceEngine
from vainplex_openclaw_amd.governance.output_validator import OutputValidator, more_restrictive
from vainplex_openclaw_amd.governance.claims import detect_claims
from vainplex_openclaw_amd.governance.facts import FactRegistry, check_claims


def test_merkle_root_shapes():
    # single leaf: root == sha256(leaf-hash || leaf-hash)? No: single level -> leaf hash itself
    leaf = b"hello"
    assert merkle_root([leaf]) == hashlib.sha256(leaf).hexdigest()
    # two leaves
    h0 = hashlib.sha256(b"a").digest()
    h1 = hashlib.sha256(b"b").digest()
    assert merkle_root([b"a", b"b"]) == hashlib.sha256(h0 + h1).hexdigest()
    # odd count duplicates last
    h2 = hashlib.sha256(b"c").digest()
    inner0 = hashlib.sha256(h0 + h1).digest()
    inner1 = hashlib.sha256(h2 + h2).digest()
    assert merkle_root([b"a", b"b", b"c"]) == hashlib.sha256(inner0 + inner1).hexdigest()


def test_derive_controls_deny_baseline():
    mp = [{"controls": ["A.8.6", "A.7.1"]}]
    assert derive_controls(mp, "deny") == ["A.5.24", "A.5.28", "A.7.1", "A.8.6"]
    assert derive_controls(mp, "allow") == ["A.7.1", "A.8.6"]


def test_audit_record_format_and_flush(workspace):
    trail = AuditTrail({}, workspace)
    trail.load()
    rec = trail.record(
        "deny",
        "test reason",
        {"hook": "before_tool_call", "agentId": "a1", "toolParams": {"token": "api_key=supersecret123"}},
        {"score": 35, "tier": "untrusted"},
        {"level": "high", "score": 60},
        [{"policyId": "p", "ruleId": "r", "effect": {"action": "deny"}, "controls": ["A.8.6"]}],
        123,
    )
    for key in ("id", "timestamp", "timestampIso", "verdict", "reason", "context",
                "trust", "risk", "matchedPolicies", "evaluationUs", "controls"):
        assert key in rec
    # redaction applied to context
    assert "supersecret123" not in json.dumps(rec["context"])
    trail.flush()
    files = os.listdir(os.path.join(workspace, "governance", "audit"))
    day_files = [f for f in files if f.endswith(".jsonl") and ".merkle." not in f]
    assert len(day_files) == 1
    day = day_files[0].replace(".jsonl", "")
    assert trail.verify_merkle(day)


def test_audit_query_filters(workspace):
    trail = AuditTrail({}, workspace)
    trail.load()
    for i, verdict in enumerate(["allow", "deny", "allow"]):
        trail.record(verdict, "r", {"agentId": f"a{i % 2}"}, {"score": 0, "tier": "untrusted"},
                     {"level": "low", "score": 0}, [], 1)
    trail.flush()
    assert len(trail.query({"verdict": "deny"})) == 1
    assert len(trail.query({"agentId": "a0"})) == 2
    assert len(trail.query({"limit": 1})) == 1


def test_audit_merkle_chain_continuity(workspace):
    trail = AuditTrail({}, workspace)
    trail.load()
    trail.record("allow", "r1", {}, {"score": 0, "tier": "untrusted"}, {"level": "low", "score": 0}, [], 1)
    trail.flush()
    trail.record("allow", "r2", {}, {"score": 0, "tier": "untrusted"}, {"level": "low", "score": 0}, [], 1)
    trail.flush()
    audit_dir = os.path.join(workspace, "governance", "audit")
    mfile = [f for f in os.listdir(audit_dir) if f.endswith(".merkle.jsonl")][0]
    with open(os.path.join(audit_dir, mfile)) as fh:
        entries = [json.loads(ln) for ln in fh if ln.strip()]
    assert len(entries) == 2
    assert entries[1]["prevRoot"] == entries[0]["chained"]


def test_engine_deny_records_violation_and_audit(workspace):
    engine = GovernanceEngine(
        {"builtinPolicies": {"credentialGuard": True}, "trust": {"enabled": True, "defaultScore": 40}},
        workspace,
    )
    engine.start()
    try:
        ctx = engine.build_context(
            "before_tool_call", "a1", tool_name="read", tool_params={"file_path": "/x/.env"}
        )
        verdict = engine.evaluate(ctx)
        assert verdict["action"] == "deny"
        assert engine.trust_manager.score("a1") == 38  # violation -2
        assert engine.stats["denies"] == 1
        assert verdict["evaluationUs"] > 0
        engine.audit_trail.flush()
        recs = engine.audit_trail.query({"verdict": "deny"})
        assert len(recs) == 1
        assert "A.5.24" in recs[0]["controls"]
    finally:
        engine.stop()


def test_engine_night_mode_no_trust_penalty(workspace):
    t = [0.0]

    # fixed clock at 23:30 local time
    import datetime as dt
    base = dt.datetime(2026, 1, 1, 23, 30).timestamp()
    engine = GovernanceEngine(
        {"builtinPolicies": {"nightMode": True}},
        workspace,
        clock=lambda: base,
    )
    engine.start()
    try:
        ctx = engine.build_context("before_tool_call", "a1", tool_name="exec", tool_params={"command": "ls"})
        verdict = engine.evaluate(ctx)
        assert verdict["action"] == "deny"
        assert engine.trust_manager.score("a1") == 40  # night-mode deny: no violation
    finally:
        engine.stop()


def test_engine_fail_modes(workspace):
    engine = GovernanceEngine({"failMode": "closed"}, workspace)
    engine.start()
    try:
        # force a pipeline crash
        engine.risk_assessor = None
        verdict = engine.evaluate(engine.build_context("before_tool_call", "a1", tool_name="exec"))
        assert verdict["action"] == "deny"
        assert "fail-closed" in verdict["reason"]
    finally:
        engine.stop()
    engine2 = GovernanceEngine({"failMode": "open"}, workspace)
    engine2.start()
    try:
        engine2.risk_assessor = None
        verdict = engine2.evaluate(engine2.build_context("before_tool_call", "a1", tool_name="exec"))
        assert verdict["action"] == "allow"
    finally:
        engine2.stop()


def test_claim_detection_families():
    text = (
        "The nginx-service is running. The server named web-01 is fine. "
        "backup.db does not exist. The queue has 255,908 items. disk is at 80%. "
        "I am DeployBot, and I have admin capabilities.\n"
    )
    claims = detect_claims(text)
    types = {c["type"] for c in claims}
    assert "system_state" in types
    assert "entity_name" in types
    assert "existence" in types
    assert "operational_status" in types
    assert "self_referential" in types
    # common word filter
    assert not any(c["subject"] == "it" for c in detect_claims("it is running"))


def test_fact_check_verdicts():
    reg = FactRegistry([{"facts": [
        {"subject": "nginx", "predicate": "state", "value": "running"},
        {"subject": "queue", "predicate": "count", "value": "255908"},
    ]}])
    claims = detect_claims("nginx is stopped. queue count is 255908.")
    results = check_claims(claims, reg)
    by_subject = {r["claim"]["subject"]: r["status"] for r in results}
    assert by_subject["nginx"] == "contradicted"
    assert by_subject["queue"] == "verified"  # fuzzy numeric
    unv = check_claims(detect_claims("mystery-svc is running."), reg)
    assert unv[0]["status"] == "unverified"


def test_output_validator_trust_proportional():
    cfg = {"factRegistries": [{"facts": [{"subject": "nginx", "predicate": "state", "value": "running"}]}]}
    ov = OutputValidator(cfg)
    text = "nginx is stopped."
    assert ov.validate(text, trust_score=30)["verdict"] == "block"
    assert ov.validate(text, trust_score=50)["verdict"] == "flag"
    assert ov.validate(text, trust_score=70)["verdict"] == "pass"
    assert ov.validate("all good here", trust_score=10)["verdict"] == "pass"
    assert more_restrictive("flag", "block") == "block"
    assert more_restrictive("pass", "flag") == "flag"


# -- LLM validator (llm-validator.ts) ----------------------------------------

def test_llm_validator_external_detection():
    from vainplex_openclaw_amd.governance.llm_validator import is_external_comm

    assert is_external_comm(channel="twitter")
    assert is_external_comm(tool_name="send_email")
    assert is_external_comm(command="bird tweet hello world")
    assert not is_external_comm(channel="internal-slack", command="ls -la")


def test_llm_validator_cache_ttl_and_verdicts():
    from vainplex_openclaw_amd.governance.llm_validator import LlmValidator

    calls = []

    def fake_llm(prompt):
        calls.append(prompt)
        return '{"verdict": "flag", "reason": "unverifiable claim"}'

    t = [1000.0]
    v = LlmValidator(fake_llm, clock=lambda: t[0])
    facts = [{"subject": "system", "predicate": "status", "value": "online"}]
    r1 = v.validate("our system has 1M users", facts, True)
    assert r1["verdict"] == "flag" and not r1["cached"]
    r2 = v.validate("our system has 1M users", facts, True)
    assert r2["cached"] and len(calls) == 1  # djb2 cache hit
    t[0] += 301.0
    r3 = v.validate("our system has 1M users", facts, True)
    assert not r3["cached"] and len(calls) == 2  # TTL expired
    # non-external never hits the LLM
    assert v.validate("internal note", facts, False)["verdict"] == "pass"
    assert len(calls) == 2


def test_llm_validator_fail_open_and_bad_json():
    from vainplex_openclaw_amd.governance.llm_validator import LlmValidator

    v = LlmValidator(lambda p: "not json")
    assert v.validate("x", [], True)["verdict"] == "pass"

    def boom(p):
        raise OSError("llm down")

    v2 = LlmValidator(boom)
    r = v2.validate("y", [], True)
    assert r["verdict"] == "pass" and "llm-error" in r["reason"]


# -- full config resolution (config.ts) --------------------------------------

def test_resolve_governance_config_defaults():
    from vainplex_openclaw_amd.governance.config import resolve_config

    cfg = resolve_config({})
    assert cfg["failMode"] == "open" and cfg["timezone"] == "UTC"
    assert cfg["trust"]["defaults"] == {"main": 60, "*": 10}
    assert cfg["trust"]["decay"] == {"enabled": True, "inactivityDays": 30, "rate": 0.95}
    assert cfg["trust"]["sessionTrust"]["seedFactor"] == 0.7
    assert cfg["trust"]["sessionTrust"]["signals"]["credentialViolation"] == -10
    assert cfg["audit"] == {"enabled": True, "retentionDays": 90,
                            "redactPatterns": [], "level": "standard"}
    assert set(cfg["outputValidation"]["detectors"]) == {
        "system_state", "entity_name", "existence",
        "operational_status", "self_referential"}
    assert cfg["outputValidation"]["llmValidator"]["externalChannels"] == [
        "twitter", "linkedin", "email"]
    assert cfg["approval2fa"] is None  # needs enabled + totpSecret
    assert not cfg["erc8004"]["enabled"]


def test_resolve_governance_config_overrides_and_gates():
    from vainplex_openclaw_amd.governance.config import resolve_config

    cfg = resolve_config({
        "failMode": "closed",
        "trust": {"defaults": {"boss": 80, "bad": "nope"},
                  "decay": {"rate": 0.9}},
        "audit": {"level": "bogus", "retentionDays": 7},
        "approval2fa": {"enabled": True, "totpSecret": "JBSWY3DP"},
        "agentFirewall": {"erc8004": {"enabled": True, "agentMapping": {"main": 3}}},
        "builtinPolicies": {"nightMode": True},
    })
    assert cfg["failMode"] == "closed"
    assert cfg["trust"]["defaults"] == {"boss": 80}  # non-numeric dropped
    assert cfg["trust"]["decay"]["rate"] == 0.9
    assert cfg["trust"]["decay"]["inactivityDays"] == 30  # default kept
    assert cfg["audit"]["level"] == "standard"  # invalid -> default
    assert cfg["audit"]["retentionDays"] == 7
    assert cfg["approval2fa"]["totpSecret"] == "JBSWY3DP"
    assert cfg["approval2fa"]["sessionApprovalMinutes"] == 10
    assert cfg["erc8004"]["enabled"] and cfg["erc8004"]["agentMapping"] == {"main": 3}
    assert cfg["builtinPolicies"]["nightMode"] is True
    # 2fa without secret stays None
    assert resolve_config({"approval2fa": {"enabled": True}})["approval2fa"] is None


# -- engine.test.ts mirrors ------------------------------------------------

def _engine(workspace, **cfg):
    base = {"trust": {"enabled": True, "defaultScore": 40}}
    base.update(cfg)
    return GovernanceEngine(base, workspace)


def test_engine_start_stop_and_status(workspace):
    e = _engine(workspace)
    e.start()
    try:
        e.evaluate(e.build_context("before_tool_call", "a1", tool_name="read"))
        st = e.status()
        assert st["stats"]["evaluations"] == 1
        assert isinstance(st["policies"], list)
        assert "a1" in st["agents"]
        assert "audit" in st and "crossAgent" in st
    finally:
        e.stop()


def test_engine_allow_when_no_policies(workspace):
    e = _engine(workspace)
    e.start()
    try:
        v = e.evaluate(e.build_context("before_tool_call", "a1", tool_name="read"))
        assert v["action"] == "allow"
    finally:
        e.stop()


def test_engine_record_outcomes_success_only_on_record(workspace):
    """Bug 3: evaluate() allow must NOT bump successCount; only an
    explicit record_outcome(success=True) does."""
    e = _engine(workspace)
    e.start()
    try:
        e.evaluate(e.build_context("before_tool_call", "a1", tool_name="read"))
        assert e.trust_manager.get("a1")["signals"]["successCount"] == 0
        e.record_outcome("a1", "agent:a1", True)
        assert e.trust_manager.get("a1")["signals"]["successCount"] == 1
    finally:
        e.stop()


def test_engine_denial_increments_violation_but_not_night(workspace):
    e = _engine(workspace, builtinPolicies={"credentialGuard": True})
    e.start()
    try:
        e.evaluate(e.build_context("before_tool_call", "a1", tool_name="read",
                                   tool_params={"file_path": "/app/.env"}))
        assert e.trust_manager.get("a1")["signals"]["violationCount"] == 1
    finally:
        e.stop()


def test_engine_denial_audit_includes_reason_and_policy_controls(workspace):
    """Bug 2 + Bug 4: the audit record carries the aggregated reason and
    controls derived from the MATCHED POLICY (plus the deny baseline),
    not from the hook name."""
    e = _engine(workspace, builtinPolicies={"credentialGuard": True})
    e.start()
    try:
        v = e.evaluate(e.build_context("before_tool_call", "a1", tool_name="read",
                                       tool_params={"file_path": "/x/.env"}))
        assert v["action"] == "deny" and v["reason"]
        e.audit_trail.flush()
        rec = e.audit_trail.query({"verdict": "deny"})[0]
        assert rec["reason"] == v["reason"]
        assert "A.5.24" in rec["controls"] and "A.5.28" in rec["controls"]
        # the credential guard's own ISO tags ride along
        policy_controls = [c for c in rec["controls"] if c not in ("A.5.24", "A.5.28")]
        assert policy_controls, rec["controls"]
        # allow records carry a reason too
        e.evaluate(e.build_context("before_tool_call", "a1", tool_name="read",
                                   tool_params={"file_path": "/x/notes.md"}))
        e.audit_trail.flush()
        allows = e.audit_trail.query({"verdict": "allow"})
        assert allows and allows[0]["reason"]
    finally:
        e.stop()


def test_engine_evaluation_stats_running_average(workspace):
    e = _engine(workspace)
    e.start()
    try:
        for _ in range(3):
            e.evaluate(e.build_context("before_tool_call", "a1", tool_name="read"))
        assert e.stats["evaluations"] == 3
        assert e.stats["allows"] == 3
        assert e.stats["avgEvaluationUs"] > 0
    finally:
        e.stop()


def test_engine_known_agents_auto_registered_and_kept(workspace):
    e = _engine(workspace, trust={"enabled": True, "defaults": {"main": 60, "*": 25}})
    e.start()
    try:
        e.set_known_agents(["main", "forge"])
        assert e.trust_manager.score("main") == 60
        assert e.trust_manager.score("forge") == 25  # wildcard default
        # unspecified agent also gets the wildcard
        assert e.trust_manager.score("random") == 25
        # removing an agent keeps its trust data
        e.set_known_agents(["main"])
        assert "forge" in e.trust_manager.get_store()["agents"]
    finally:
        e.stop()


# ===========================================================================
# audit-redactor.test.ts depth (9 its): recursive redaction, defaults,
# custom/invalid patterns, non-string preservation
# ===========================================================================

def test_audit_redactor_default_patterns():
    from vainplex_openclaw_amd.governance.audit_redactor import create_redactor

    red = create_redactor()
    ctx = {
        "toolParams": {"cmd": "export api_key=abc123secret && run"},
        "note": "uses sk-" + "a" * 20,
        "hdr": "Authorization: Bearer abc.def.ghi",
    }
    out = red(ctx)
    blob = str(out)
    assert "abc123secret" not in blob
    assert "sk-" + "a" * 20 not in blob
    assert "abc.def.ghi" not in blob
    assert "[REDACTED]" in blob


def test_audit_redactor_recursive_and_type_preserving():
    from vainplex_openclaw_amd.governance.audit_redactor import create_redactor

    red = create_redactor()
    ctx = {"deep": {"list": ["password: hunter22", 42, None, True],
                    "n": 3.5}}
    out = red(ctx)
    assert "hunter22" not in str(out)
    assert out["deep"]["list"][1] == 42
    assert out["deep"]["list"][2] is None
    assert out["deep"]["list"][3] is True
    assert out["deep"]["n"] == 3.5


def test_audit_redactor_custom_and_invalid_patterns():
    from vainplex_openclaw_amd.governance.audit_redactor import create_redactor

    red = create_redactor([r"TICKET-\d+", "(unclosed"])
    out = red({"msg": "see TICKET-123 and password: x9x9x9x9"})
    assert "TICKET-123" not in out["msg"]
    # custom list REPLACES defaults: the password pattern no longer applies
    assert "x9x9x9x9" in out["msg"]
    # empty pattern list redacts nothing
    red2 = create_redactor([])
    assert red2({"a": "password: keepme99"})["a"] == "password: keepme99"


def test_audit_redactor_does_not_mutate_input():
    from vainplex_openclaw_amd.governance.audit_redactor import create_redactor

    red = create_redactor()
    ctx = {"k": "password: original9"}
    red(ctx)
    assert ctx["k"] == "password: original9"


# -- core config loader depth (config-loader.ts, 14 its) ---------------------

def test_plugin_config_file_first_and_fallback(tmp_path):
    import json

    from vainplex_openclaw_amd.core.config import load_plugin_config

    # `home` here is the .openclaw directory itself (openclaw_home())
    home = tmp_path
    d = home / "plugins" / "p1"
    d.mkdir(parents=True)
    (d / "config.json").write_text(json.dumps({"from": "file"}))
    assert load_plugin_config("p1", {"from": "api"}, home=str(home)) == {"from": "file"}
    # no file -> api.pluginConfig fallback
    assert load_plugin_config("p2", {"from": "api"}, home=str(home)) == {"from": "api"}
    # neither -> empty dict
    assert load_plugin_config("p3", home=str(home)) == {}


def test_plugin_config_invalid_json_falls_back(tmp_path):
    from vainplex_openclaw_amd.core.config import load_plugin_config

    d = tmp_path / "plugins" / "p1"
    d.mkdir(parents=True)
    (d / "config.json").write_text("{broken json")
    assert load_plugin_config("p1", {"ok": 1}, home=str(tmp_path)) == {"ok": 1}


def test_resolve_defaults_recursion_and_type_guards():
    from vainplex_openclaw_amd.core.config import resolve_defaults

    defaults = {"a": 1, "nested": {"x": "dx", "y": True}, "keep": "d"}
    out = resolve_defaults({"a": 5, "nested": {"x": "cx"}, "extra": 9}, defaults)
    assert out["a"] == 5
    assert out["nested"] == {"x": "cx", "y": True}
    assert out["keep"] == "d"
