"""Full-suite integration: one Gateway, all seven plugins, a scripted
conversation driven through the hook bus.

Mirrors the reference's data flow (suite README.md:60-106):
message -> Governance (gate) -> Membrane (recall/inject) -> agent ->
Cortex + KE + Membrane-ingest; all components -> EventStore -> Leuko.
"""

import json
import os

import pytest

from vainplex_openclaw_amd.core.api import NullLogger
from vainplex_openclaw_amd.core.gateway import Gateway
from vainplex_openclaw_amd.cortex.hooks import create_plugin as create_cortex
from vainplex_openclaw_amd.eventstore import EventJournal
from vainplex_openclaw_amd.eventstore.plugin import create_plugin as create_eventstore
from vainplex_openclaw_amd.governance.plugin import create_plugin as create_governance
from vainplex_openclaw_amd.knowledge.hooks import create_plugin as create_knowledge
from vainplex_openclaw_amd.leuko.plugin import create_plugin as create_leuko
from vainplex_openclaw_amd.membrane.hooks import create_plugin as create_membrane


@pytest.fixture()
def suite(tmp_path, monkeypatch):
    monkeypatch.setenv("HOME", str(tmp_path))  # isolate ~/.openclaw
    ws = str(tmp_path / "workspace")
    os.makedirs(ws, exist_ok=True)
    gw = Gateway(config={"agents": [{"id": "main"}, {"id": "forge"}]},
                 logger=NullLogger(), home=str(tmp_path))
    journal = EventJournal(durable=False)
    gw.load(create_governance(workspace=ws), {})
    gw.load(create_cortex(workspace=ws), {})
    gw.load(create_knowledge(workspace=ws), {})
    gw.load(create_membrane(workspace=ws), {})
    gw.load(create_eventstore(journal=journal), {})
    gw.load(create_leuko(workspace=ws, journal=journal), {})
    return gw, journal, ws


def _msg(gw, content, agent="main"):
    return gw.bus.emit("message_received", {
        "content": content, "from": "user",
        "ctx": {"agentId": agent, "sessionKey": f"{agent}:test:1"},
    })


def test_suite_loads_all_plugins(suite):
    gw, journal, ws = suite
    assert set(gw.plugins) == {
        "openclaw-governance", "openclaw-cortex", "openclaw-knowledge-engine",
        "openclaw-membrane", "nats-eventstore", "openclaw-leuko",
    }


def test_message_flow_fans_out_to_all_plugins(suite):
    gw, journal, ws = suite
    gw.bus.emit("session_start", {"sessionId": "s1", "ctx": {"sessionKey": "main:test:1"}})
    _msg(gw, "We decided to use postgres for the storage layer.")
    _msg(gw, "Contact ada@example.org at Acme Corp. about the migration")
    ev = _msg(gw, "what did we decide about storage?")

    # Membrane injected the earlier memory
    assert "postgres" in ev.get("membrane_context", "")
    # Cortex tracked a thread (threads.json persisted per message)
    cortex = gw.plugins["openclaw-cortex"]
    ws_state = cortex.hooks.ws(ws)
    assert ws_state.threads.threads or ws_state.decisions.decisions
    # KE extracted entities
    ke = gw.plugins["openclaw-knowledge-engine"]
    assert any(e.type == "email" for e in ke.hooks.entities.values())
    # EventStore captured envelopes for every message
    types = [e["canonicalType"] for _s, e in journal.replay()]
    assert types.count("message.in.received") == 3
    assert "session.started" in types


def test_tool_call_gate_and_audit_flow(suite):
    gw, journal, ws = suite
    ev = gw.bus.emit("before_tool_call", {
        "toolName": "exec", "params": {"command": "ls /tmp"},
        "ctx": {"agentId": "main", "sessionKey": "main:test:1", "toolCallId": "t1"},
    })
    assert not ev.get("block")  # benign command passes
    # eventstore saw the tool call request
    assert any(e["canonicalType"] == "tool.call.requested" for _s, e in journal.replay())


def test_redaction_protects_outbound(suite):
    gw, journal, ws = suite
    ev = gw.bus.emit("message_sending", {
        "content": "here is the key sk-abcdefghijklmnopqrst1234",
        "to": "user", "ctx": {"agentId": "main", "sessionKey": "main:test:1"},
    })
    out = ev.get("content", "")
    assert "sk-abcdefghijklmnopqrst1234" not in out
    assert "[REDACTED:" in out


def test_leuko_sees_journal_and_reports(suite):
    gw, journal, ws = suite
    _msg(gw, "hello")
    leuko = gw.plugins["openclaw-leuko"]
    report = leuko.run_once()
    assert report["version"] == 1
    assert "nats" in report["collectors"]
    # journal collector enabled via injected journal
    assert report["health"]["details"]["nats"] in ("ok", "warn")


def test_gateway_stop_flushes_everything(suite):
    gw, journal, ws = suite
    gw.bus.emit("session_start", {"sessionId": "s1", "ctx": {"sessionKey": "main:test:1"}})
    _msg(gw, "We decided to ship v2 on friday. I will prepare the notes.")
    gw.bus.emit("gateway_stop", {})
    # KE fact store flushed to disk
    assert os.path.exists(os.path.join(ws, "facts.json"))
    # membrane store flushed
    membrane = gw.plugins["openclaw-membrane"]
    assert membrane.engine.stats["ingested"] >= 1


def test_suite_cli_demo(tmp_path, monkeypatch, capsys):
    """`python -m vainplex_openclaw_amd --demo` end-to-end."""
    monkeypatch.setenv("HOME", str(tmp_path))
    from vainplex_openclaw_amd.__main__ import main

    rc = main(["--demo", "--workspace", str(tmp_path / "ws")])
    assert rc == 0
    out = capsys.readouterr().out
    assert "membrane_context" in out and "postgres" in out


def test_cortex_trace_analyzer_wired_to_journal(tmp_path, monkeypatch):
    """Cortex registers the trace analyzer over the shared journal and
    cortex.analyze produces a report from real hook traffic."""
    monkeypatch.setenv("HOME", str(tmp_path))
    from vainplex_openclaw_amd.core.api import HookBus, NullLogger, PluginApi
    from vainplex_openclaw_amd.cortex.hooks import create_plugin as mk_cortex
    from vainplex_openclaw_amd.eventstore import EventJournal
    from vainplex_openclaw_amd.eventstore.plugin import create_plugin as mk_es

    ws = str(tmp_path / "ws")
    bus = HookBus()
    journal = EventJournal(durable=False)
    for plugin, pid in ((mk_es(journal=journal), "nats-eventstore"),
                        (mk_cortex(workspace=ws, journal=journal), "openclaw-cortex")):
        api = PluginApi(id=pid, plugin_config={}, logger=NullLogger(), config={}, bus=bus)
        plugin.register(api)
        if pid == "openclaw-cortex":
            cortex_api = api
    # drive a doom loop through the hooks -> journal -> analyzer
    for i in range(4):
        bus.emit("before_tool_call", {"toolName": "exec", "params": {"command": "make build"},
                                      "ctx": {"sessionKey": "main:t:1", "toolCallId": f"t{i}"}})
        bus.emit("after_tool_call", {"toolName": "exec", "params": {"command": "make build"},
                                     "error": "exit 1",
                                     "ctx": {"sessionKey": "main:t:1", "toolCallId": f"t{i}"}})
    report = cortex_api.gateway_methods["cortex.analyze"]()
    assert report["eventsAnalyzed"] >= 8
    assert any(f["signalType"] in ("doom_loop", "repeat_fail", "tool_fail")
               for f in report["findings"])


# -- governance integration.test.ts mirrors -------------------------------

def test_subagent_policy_cascade_and_trust_cap(workspace):
    """Parent deny policies cascade to spawned sub-agents; child trust is
    capped at the parent's (integration.test.ts:297-364)."""
    from vainplex_openclaw_amd.core.gateway import Gateway
    from vainplex_openclaw_amd.governance.plugin import GovernancePlugin

    gw = Gateway(config={"agents": ["parent"]})
    plugin = GovernancePlugin(workspace)
    gw.load(plugin, plugin_config={
        "workspace": workspace,
        "trust": {"enabled": True, "defaults": {"parent": 70, "*": 40}},
        "policies": [{
            "id": "parent-no-deploy", "priority": 50,
            "scope": {"agents": ["parent"]},
            "rules": [{"id": "r", "conditions": [{"type": "tool", "name": "deploy"}],
                       "effect": {"action": "deny", "reason": "parent forbids deploys"}}],
        }],
    })
    gw.start()
    try:
        # spawn a sub-agent under the parent session
        gw.emit("after_tool_call", {
            "agentId": "parent", "sessionKey": "agent:parent",
            "toolName": "sessions_spawn",
            "result": {"sessionId": "agent:parent:subagent:w:1"},
        })
        ca = plugin.engine.cross_agent
        assert ca.get_parent("agent:parent:subagent:w:1") is not None
        # the child's effective policies include the parent's deny
        eff = ca.resolve_effective_policies(
            {"sessionKey": "agent:parent:subagent:w:1", "agentId": "w",
             "hook": "before_tool_call"},
            plugin.engine.policy_index)
        assert any(p["id"] == "parent-no-deploy" for p in eff)
        # child trust ceiling <= parent score
        cap = ca.compute_trust_ceiling("agent:parent:subagent:w:1")
        assert cap <= plugin.engine.trust_manager.score("parent")
    finally:
        gw.stop()


def test_policy_eval_and_frequency_performance(workspace):
    """10 regex policies evaluate fast; 1000 frequency entries don't
    degrade (integration.test.ts:389-442; bounds relaxed for CPython)."""
    import time
    from vainplex_openclaw_amd.governance.engine import GovernanceEngine

    policies = [{
        "id": f"p{i}", "priority": i, "scope": {},
        "rules": [{"id": "r", "conditions": [
            {"type": "tool", "params": {"command": {"matches": rf"cmd-{i}-\d+"}}}],
            "effect": {"action": "deny"}}],
    } for i in range(10)]
    engine = GovernanceEngine({"policies": policies}, workspace)
    engine.start()
    try:
        ctx = engine.build_context("before_tool_call", "a1",
                                   tool_name="exec", tool_params={"command": "harmless"})
        start = time.perf_counter()
        for _ in range(50):
            engine.evaluate(ctx)
        per_eval_ms = (time.perf_counter() - start) / 50 * 1000
        assert per_eval_ms < 50, per_eval_ms
        for i in range(1000):
            engine.frequency.record(f"a{i % 4}", f"agent:a{i % 4}", "exec")
        start = time.perf_counter()
        engine.evaluate(ctx)
        assert (time.perf_counter() - start) < 0.1
    finally:
        engine.stop()


def test_output_validation_end_to_end_with_audit(workspace):
    """Output validation through the ENGINE entry point, including the
    trust-proportional verdict and an audit trail record
    (integration.test.ts:443-604)."""
    from vainplex_openclaw_amd.governance.engine import GovernanceEngine

    engine = GovernanceEngine({
        "trust": {"enabled": True, "defaults": {"low": 20, "high": 80, "*": 40}},
        "outputValidation": {
            "factRegistries": [{"facts": [
                {"subject": "nginx", "predicate": "state", "value": "stopped"}]}],
        },
    }, workspace)
    engine.start()
    try:
        engine.trust_manager.get("low")
        engine.trust_manager.get("high")
        blocked = engine.validate_output("nginx is running", "low")
        assert blocked["verdict"] == "block"
        passed = engine.validate_output("nginx is running", "high")
        assert passed["verdict"] == "pass"
        ok = engine.validate_output("nginx is stopped", "low")
        assert ok["verdict"] == "pass"
        # unverified ignored by default
        assert engine.validate_output("mystery-svc is running", "low")["verdict"] == "pass"
    finally:
        engine.stop()


# ===========================================================================
# integration.test.ts depth: deny-wins, tier gates, night mode flow,
# fail-open engine errors, validation trust thresholds end to end
# ===========================================================================

def _engine(workspace, config=None):
    from vainplex_openclaw_amd.governance.engine import GovernanceEngine

    eng = GovernanceEngine(config or {}, workspace)
    eng.start()
    return eng


def test_deny_wins_across_policies(workspace):
    eng = _engine(workspace, {"policies": [
        {"id": "p-allow", "scope": {"hooks": ["before_tool_call"]}, "priority": 10,
         "rules": [{"id": "r", "conditions": [], "effect": {"action": "audit"}}]},
        {"id": "p-deny", "scope": {"hooks": ["before_tool_call"]}, "priority": 1,
         "rules": [{"id": "r", "conditions": [{"type": "tool", "name": "exec"}],
                    "effect": {"action": "deny"}}]},
    ]})
    out = eng.evaluate(eng.build_context("before_tool_call", "a1", tool_name="exec"))
    assert out["action"] == "deny"
    out2 = eng.evaluate(eng.build_context("before_tool_call", "a1", tool_name="read"))
    # audit aggregates to allow-with-logging (policy-evaluator.ts:44-78)
    assert out2["action"] == "allow" and "audit logging" in out2["reason"]


def test_trust_tier_gate_on_rule(workspace):
    eng = _engine(workspace, {"policies": [
        {"id": "p", "scope": {"hooks": ["before_tool_call"]},
         "rules": [{"id": "r", "minTrust": "standard",
                    "conditions": [{"type": "tool", "name": "deploy"}],
                    "effect": {"action": "allow"}},
                   {"id": "r2",
                    "conditions": [{"type": "tool", "name": "deploy"}],
                    "effect": {"action": "deny",
                               "reason": "untrusted deploy"}}]},
    ]})
    # minTrust is a TIER gate on the SESSION tier
    # (policy-evaluator.ts:134-139): agent 40 seeds session 28 ->
    # "restricted" < "standard" -> the allow rule is skipped, deny hits
    out = eng.evaluate(eng.build_context("before_tool_call", "lowT", tool_name="deploy"))
    assert out["action"] == "deny"
    eng.trust_manager.set_score("highT", 90.0)  # session 63 -> "trusted"
    out2 = eng.evaluate(eng.build_context("before_tool_call", "highT", tool_name="deploy"))
    assert out2["action"] == "allow"


def test_night_mode_flow_and_violation_skip(workspace):
    t = [0.0]

    def clock():
        return t[0]

    from vainplex_openclaw_amd.governance.engine import GovernanceEngine

    eng = GovernanceEngine({"builtinPolicies": {"nightMode": True}}, workspace,
                           clock=clock)
    eng.start()
    # 03:00 UTC: destructive exec denied by builtin-night-mode
    t[0] = 1_700_000_000.0
    import time as _time

    # find a timestamp at 03:00 UTC
    base = 1_700_000_000.0
    while _time.gmtime(base).tm_hour != 3:
        base += 3600
    t[0] = base
    ctx = eng.build_context("before_tool_call", "nite", tool_name="exec",
                            tool_params={"command": "rm -rf /var/data"})
    out = eng.evaluate(ctx)
    assert out["action"] == "deny"
    assert any(m.get("policyId") == "builtin-night-mode" for m in out["matchedPolicies"])
    # night-mode denials do NOT count trust violations (engine.ts:248-263)
    assert eng.trust_manager.get("nite")["signals"].get("violations", 0) == 0
    # same call at 14:00 passes the time window
    while _time.gmtime(base).tm_hour != 14:
        base += 3600
    t[0] = base
    out2 = eng.evaluate(eng.build_context("before_tool_call", "nite", tool_name="exec",
                                          tool_params={"command": "rm -rf /var/data"}))
    assert out2["action"] != "deny" or not any(
        m.get("policyId") == "builtin-night-mode" for m in out2["matchedPolicies"])


def test_engine_error_fail_open_and_closed(workspace):
    eng = _engine(workspace, {"failMode": "open"})
    eng.policy_index = None          # force an internal failure
    out = eng.evaluate({"hook": "before_tool_call", "agentId": "x", "toolName": "t"})
    assert out["action"] == "allow" and "error" in str(out.get("reason", "")).lower()

    eng2 = _engine(workspace + "/c", {"failMode": "closed"})
    eng2.policy_index = None
    out2 = eng2.evaluate({"hook": "before_tool_call", "agentId": "x", "toolName": "t"})
    assert out2["action"] == "deny"


def test_validation_trust_thresholds_end_to_end(workspace):
    cfg = {"outputValidation": {
        "enabled": True,
        "factRegistries": [{"facts": [
            {"subject": "nginx", "predicate": "state", "value": "running"}]}],
    }}
    eng = _engine(workspace, cfg)
    text = "nginx is stopped."
    eng.trust_manager.set_score("low", 20.0)
    v_low = eng.validate_output(text, "low")
    assert v_low["verdict"] == "block"
    eng.trust_manager.set_score("mid", 50.0)
    assert eng.validate_output(text, "mid")["verdict"] == "flag"
    eng.trust_manager.set_score("high", 85.0)
    assert eng.validate_output(text, "high")["verdict"] == "pass"
    # matching claims pass at any trust
    assert eng.validate_output("nginx is running.", "low")["verdict"] == "pass"


def test_trace_findings_become_checkable_facts_end_to_end(tmp_path):
    """Capstone loop (SURVEY §1 flow): journal events -> trace analyzer
    detects a hallucination with an SPO payload -> trace-to-facts bridge
    registers the fact -> the governance output validator then catches an
    agent REPEATING the hallucinated claim, trust-proportionally."""
    from vainplex_openclaw_amd.cortex.trace.analyzer import JournalTraceSource, TraceAnalyzer
    from vainplex_openclaw_amd.eventstore import EventJournal
    from vainplex_openclaw_amd.governance.facts import FactRegistry
    from vainplex_openclaw_amd.governance.output_validator import OutputValidator
    from vainplex_openclaw_amd.governance.trace_to_facts import apply_report_to_registry

    j = EventJournal(durable=False)
    base = {"actor": {"id": "forge"}, "scope": {"sessionKey": "agent:forge:s1"}}
    j.publish("s.1", {**base, "id": "m1", "ts": 1000.0,
                      "canonicalType": "message.out.sent",
                      "data": {"content": "nginx-service is running"}})
    j.publish("s.2", {**base, "id": "t1", "ts": 1001.0,
                      "canonicalType": "tool.call.failed",
                      "data": {"toolName": "exec",
                               "error": "nginx-service: connection refused"}})

    report = TraceAnalyzer(str(tmp_path), JournalTraceSource(j)).run()
    halls = [f for f in report["findings"] if f["signalType"] == "hallucination"]
    assert halls and halls[0]["evidence"]["subject"] == "nginx-service"

    registry = FactRegistry()
    assert apply_report_to_registry(report, registry) >= 1
    fact = registry.lookup("nginx-service", "state")
    assert fact and fact["value"] == "error"

    validator = OutputValidator({"enabled": True})
    validator.fact_registry = registry
    low = validator.validate("the nginx-service is running fine", trust_score=20)
    assert low["verdict"] == "block"                  # contradiction + low trust
    high = validator.validate("the nginx-service is running fine", trust_score=90)
    assert high["verdict"] == "pass"                  # trust-proportional
