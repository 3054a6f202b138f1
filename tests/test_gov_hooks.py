"""Governance hooks, 2FA TOTP, response gate, cross-agent, security clients."""

import pytest

from vainplex_openclaw_amd.core.api import HookBus, PluginApi, NullLogger
from vainplex_openclaw_amd.core.gateway import Gateway
from vainplex_openclaw_amd.governance.approval_2fa import Approval2FA, totp_at, verify_totp, generate_secret
from vainplex_openclaw_amd.governance.cross_agent import CrossAgentManager
from vainplex_openclaw_amd.governance.engine import GovernanceEngine
from vainplex_openclaw_amd.governance.hooks import GovernanceHooks, detect_external_comm, register_governance_hooks
from vainplex_openclaw_amd.governance.plugin import GovernancePlugin, extract_agent_ids
from vainplex_openclaw_amd.governance.response_gate import ResponseGate
from vainplex_openclaw_amd.governance.security.agentproof import AgentProofRestClient, CircuitBreaker
from vainplex_openclaw_amd.governance.security.erc8004 import classify_tier, encode_call, decode_uint256, ERC8004Provider, ERC8004Client
from vainplex_openclaw_amd.governance.trust import TrustManager, TrustConfig
from vainplex_openclaw_amd.governance.matrix_poller import MatrixPoller


def test_totp_roundtrip():
    secret = generate_secret()
    code = totp_at(secret, 1_700_000_000)
    assert len(code) == 6 and code.isdigit()
    assert verify_totp(secret, code, 1_700_000_000) is not None
    assert verify_totp(secret, code, 1_700_000_000 + 29) is not None  # same step
    assert verify_totp(secret, "000000", 1_700_000_000) in (None, verify_totp(secret, "000000", 1_700_000_000))
    assert verify_totp(secret, "12345", 1_700_000_000) is None  # wrong length


def test_2fa_batch_resolution_and_replay():
    t = [1_700_000_000.0]
    ap = Approval2FA(clock=lambda: t[0])
    r1 = ap.request("agent:a", "a", "deploy")
    r2 = ap.request("agent:a", "a", "push")  # same 3s window -> same batch
    assert r1["status"] == "pending" and r2["status"] == "pending"
    assert len(ap.pending_requests()) == 2
    code = totp_at(ap.secret, t[0])
    resolved = ap.try_resolve_any(code)
    assert len(resolved) == 2
    assert all(r["status"] == "approved" for r in resolved)
    # replay protection: same code again does nothing
    ap.request("agent:b", "b", "x")
    assert ap.try_resolve_any(code) == []
    # session auto-approval within 10 min
    r3 = ap.request("agent:a", "a", "again")
    assert r3["status"] == "approved" and r3.get("auto")


def test_2fa_expiry():
    t = [0.0]
    ap = Approval2FA(clock=lambda: t[0], timeout_s=100)
    ap.request("agent:a", "a", "x")
    t[0] = 200
    expired = ap.expire_stale()
    assert len(expired) == 1 and expired[0]["status"] == "expired"


def test_response_gate():
    gate = ResponseGate({
        "enabled": True,
        "fallbackTemplate": "blocked for {agent}: {reasons}",
        "rules": [
            {"agentId": "a1", "validators": [
                {"type": "requiredTools", "tools": ["read"]},
                {"type": "mustNotMatch", "pattern": "password"},
            ]},
        ],
    })
    res = gate.validate("here is the password", "a1", [])
    assert not res["passed"]
    assert len(res["failedValidators"]) == 2
    assert "blocked for a1" in res["fallbackMessage"]
    ok = gate.validate("clean", "a1", [{"toolName": "read", "output": "x"}])
    assert ok["passed"]
    # other agents unaffected
    assert gate.validate("password", "a2", [])["passed"]
    # invalid regex fails closed
    bad = ResponseGate({"enabled": True, "rules": [{"validators": [{"type": "mustMatch", "pattern": "("}]}]})
    assert not bad.validate("anything", "a", [])["passed"]


def test_cross_agent_ceiling_and_cascade(workspace):
    tm = TrustManager(TrustConfig(default_score=40, initial_scores={"parent": 55}), workspace)
    cam = CrossAgentManager(tm)
    ctx = {
        "sessionKey": "agent:parent:subagent:child:123",
        "agentId": "child",
        "trust": {"agent": {"score": 90, "tier": "elevated"}, "session": {"score": 80, "tier": "elevated"}},
    }
    out = cam.enrich_context(ctx)
    assert out["trust"]["session"]["score"] == 55  # capped at parent
    assert out["trust"]["agent"]["score"] == 55
    assert out["crossAgent"]["parentAgentId"] == "parent"
    # explicit registration
    cam.register_relationship("agent:main", "sess-xyz")
    assert cam.get_parent("sess-xyz")["parentAgentId"] == "main"
    assert cam.graph_summary()["agentCount"] == 1


def test_governance_hooks_end_to_end(workspace):
    gw = Gateway(config={"agents": ["a1"]})
    plugin = GovernancePlugin(workspace)
    gw.load(plugin, plugin_config={
        "builtinPolicies": {"credentialGuard": True},
        "workspace": workspace,
    })
    gw.start()
    try:
        ev = gw.emit("before_tool_call", {
            "agentId": "a1", "sessionKey": "agent:a1",
            "toolName": "read", "params": {"file_path": "/etc/secrets/k.pem"},
        })
        assert ev["block"] is True
        assert "Credential Guard" in ev["blockReason"]
        ok = gw.emit("before_tool_call", {
            "agentId": "a1", "sessionKey": "agent:a1",
            "toolName": "read", "params": {"file_path": "/tmp/notes.txt"},
        })
        assert "block" not in ok or not ok["block"]
        # after_tool_call logs for response gate + trust
        gw.emit("after_tool_call", {
            "agentId": "a1", "sessionKey": "agent:a1", "toolName": "read", "result": "data",
        })
        assert plugin.engine.trust_manager.get("a1")["signals"]["successCount"] == 1
        # sub-agent spawn registration
        gw.emit("after_tool_call", {
            "agentId": "a1", "sessionKey": "agent:a1", "toolName": "sessions_spawn",
            "result": {"sessionId": "agent:a1:subagent:w:9"},
        })
        assert plugin.engine.cross_agent.get_parent("agent:a1:subagent:w:9") is not None
        status = gw.gateway_method("governance.status")
        assert status["stats"]["evaluations"] >= 2
    finally:
        gw.stop()


def test_detect_external_comm():
    cfg = {"outputValidation": {"llmValidator": {
        "enabled": True, "externalChannels": ["twitter", "email"], "externalCommands": ["bird tweet"],
    }}}
    assert detect_external_comm(
        {"toolName": "message", "params": {"channel": "twitter", "text": "hi"}}, cfg) == "hi"
    assert detect_external_comm(
        {"toolName": "exec", "params": {"command": "bird tweet 'yo'"}}, cfg) == "bird tweet 'yo'"
    assert detect_external_comm(
        {"toolName": "message", "params": {"channel": "internal", "text": "hi"}}, cfg) is None
    assert detect_external_comm({"toolName": "exec", "params": {"command": "ls"}}, {}) is None


def test_erc8004_encoding_and_provider():
    call = encode_call("getReputation(address)", "0x" + "ab" * 20)
    assert call.startswith("0x") and len(call) == 2 + 8 + 64
    assert decode_uint256("0x" + "0" * 63 + "5") == 5
    assert classify_tier(85) == "elevated"
    assert classify_tier(10) == "untrusted"
    calls = []

    def rpc(method, params):
        calls.append(params[0]["data"])
        if len(calls) == 1:  # ownerOf -> a non-zero owner
            return "0x" + "ab" * 12 + "cd" * 20
        # profile: score 72, feedbackCount 3, lastUpdated 99
        return "0x" + hex(72)[2:].rjust(64, "0") + hex(3)[2:].rjust(64, "0")             + hex(99)[2:].rjust(64, "0")

    prov = ERC8004Provider(ERC8004Client(rpc))
    rep = prov.lookup_reputation("0x" + "ab" * 20)
    assert rep["registered"] and rep["score"] == 72 and rep["tier"] == "high"
    prov.lookup_reputation("0x" + "ab" * 20)
    assert len(calls) == 2  # ownerOf + getProfile, then cached


def test_agentproof_queue_and_breaker():
    t = [0.0]
    posts = []

    def http_post(url, headers, body):
        posts.append(body)

    client = AgentProofRestClient(http_post=http_post, clock=lambda: t[0])
    for i in range(1005):
        client.enqueue_signal("a", "success")
    assert client.queue_depth == 1000
    assert client.dropped == 5
    assert client.flush() == 1000
    assert posts and len(posts[0]["signals"]) == 1000

    # breaker opens after 5 failures
    def bad_post(url, headers, body):
        raise IOError("down")

    failing = AgentProofRestClient(http_post=bad_post, clock=lambda: t[0])
    for _ in range(6):
        failing.enqueue_signal("a", "x")
        failing.flush()
    assert failing.breaker.is_open
    t[0] += 61
    assert not failing.breaker.is_open  # half-open after reset


def test_matrix_poller_code_extraction():
    ap = Approval2FA(clock=lambda: 1_700_000_000.0)
    ap.request("agent:a", "a", "x")
    code = totp_at(ap.secret, 1_700_000_000.0)

    def http_get(url, headers):
        return {"chunk": [{"type": "m.room.message", "content": {"body": f"approve {code}"}}], "end": "t1"}

    poller = MatrixPoller(ap, room_id="!r", http_get=http_get)
    codes = poller.poll_once()
    assert codes == [code]
    assert ap.pending_requests() == []


def test_extract_agent_ids_shapes():
    assert extract_agent_ids({"agents": ["a", "b"]}) == ["a", "b"]
    assert extract_agent_ids({"agents": [{"id": "x"}, {"name": "y"}]}) == ["x", "y"]
    assert extract_agent_ids({"agents": {"m": {}, "n": {}}}) == ["m", "n"]
    assert extract_agent_ids({"defaultAgent": "main"}) == ["main"]


def test_before_agent_start_context_injection(tmp_path, monkeypatch):
    """Context injection at priority 5 + ERC-8004 config nesting
    (reference hooks.ts:445-498, config.ts:330-334)."""
    monkeypatch.setenv("HOME", str(tmp_path))
    from vainplex_openclaw_amd.core.api import HookBus, NullLogger, PluginApi
    from vainplex_openclaw_amd.governance.plugin import create_plugin
    from vainplex_openclaw_amd.governance.hooks import resolve_erc8004_config

    bus = HookBus()
    api = PluginApi(id="openclaw-governance", plugin_config={}, logger=NullLogger(),
                    config={}, bus=bus)
    p = create_plugin(workspace=str(tmp_path))
    p.register(api)
    ev = bus.emit("before_agent_start", {"ctx": {"agentId": "main", "sessionKey": "m:1"}})
    ctx = ev.get("prependContext", "")
    assert "[Governance] Agent: main" in ctx and "Session:" in ctx

    # erc8004 nesting: top-level and under agentFirewall both resolve
    assert resolve_erc8004_config({"erc8004": {"enabled": True}})["enabled"]
    assert resolve_erc8004_config(
        {"agentFirewall": {"erc8004": {"enabled": True, "agentMapping": {"main": 7}}}}
    )["agentMapping"] == {"main": 7}
    assert not resolve_erc8004_config({})["enabled"]


def test_rate_limiter_through_full_stack(workspace):
    """Rate Limiter (15/min; 2x for trusted) triggers through the hook
    stack after enough calls (builtin-policies.ts:152-196)."""
    gw = Gateway(config={"agents": ["spammy"]})
    plugin = GovernancePlugin(workspace)
    gw.load(plugin, plugin_config={
        "builtinPolicies": {"rateLimiter": True}, "workspace": workspace,
    })
    gw.start()
    try:
        blocked = 0
        for i in range(25):
            ev = gw.emit("before_tool_call", {
                "agentId": "spammy", "sessionKey": "agent:spammy",
                "toolName": "exec", "params": {"command": f"echo {i}"},
            })
            if ev.get("block"):
                blocked += 1
        assert blocked >= 5  # beyond 15/min everything denies
    finally:
        gw.stop()


def test_message_sending_redaction_layer(workspace):
    """Layer-2 outbound redaction fires on message_sending through the
    full plugin stack (redaction/hooks.ts:127-141)."""
    gw = Gateway(config={"agents": ["a1"]})
    plugin = GovernancePlugin(workspace)
    gw.load(plugin, plugin_config={"workspace": workspace})
    gw.start()
    try:
        ev = gw.emit("message_sending", {
            "content": "token is ghp_" + "z" * 36,
            "agentId": "a1", "sessionKey": "agent:a1", "to": "user",
        })
        assert "ghp_" + "z" * 36 not in ev.get("content", "")
    finally:
        gw.stop()


def test_trust_violation_feedback_on_deny(workspace):
    """Deny verdicts record violations + session signal (engine.ts:248-263)
    except for night-mode denials (covered in test_audit_engine)."""
    gw = Gateway(config={"agents": ["risky"]})
    plugin = GovernancePlugin(workspace)
    gw.load(plugin, plugin_config={
        "builtinPolicies": {"credentialGuard": True}, "workspace": workspace,
    })
    gw.start()
    try:
        before = plugin.engine.trust_manager.get("risky")["signals"]["violationCount"]
        gw.emit("before_tool_call", {
            "agentId": "risky", "sessionKey": "agent:risky",
            "toolName": "read", "params": {"file_path": "~/.aws/credentials"},
        })
        after = plugin.engine.trust_manager.get("risky")["signals"]["violationCount"]
        assert after == before + 1
    finally:
        gw.stop()


def test_session_end_clears_toolcall_log(workspace):
    gw = Gateway(config={"agents": ["a1"]})
    plugin = GovernancePlugin(workspace)
    gw.load(plugin, plugin_config={"workspace": workspace})
    gw.start()
    try:
        gw.emit("after_tool_call", {
            "agentId": "a1", "sessionKey": "agent:a1", "toolName": "read", "result": "x",
        })
        assert plugin.hooks.tool_call_log
        gw.emit("session_end", {"agentId": "a1", "sessionKey": "agent:a1",
                                "ctx": {"sessionKey": "agent:a1"}})
        assert not plugin.hooks.tool_call_log.get("agent:a1")
    finally:
        gw.stop()


def test_2fa_reference_behaviors():
    """approval-2fa.test.ts mirrors: invalid codes, per-session batches,
    attempt rate limiting, notify with all commands, pending edges."""
    t = [1_700_000_000.0]
    notified = []
    ap = Approval2FA(clock=lambda: t[0], notify=lambda b: notified.append(b),
                     max_attempts=3, cooldown_s=60)
    # no_pending edge: resolving with nothing queued
    assert ap.try_resolve_any(totp_at(ap.secret, t[0])) == []
    assert not ap.has_pending_batch()
    # different agents/sessions -> separate batches; same session -> one
    ap.request("agent:a", "a", "deploy prod")
    ap.request("agent:a", "a", "push tags")
    ap.request("agent:b", "b", "rm -rf data")
    assert ap.has_pending_batch("agent:a") and ap.has_pending_batch("agent:b")
    assert len(notified) == 2  # one notification per batch
    batch_a = next(b for b in notified if b["sessionKey"] == "agent:a")
    assert [r["reason"] for r in batch_a["requests"]][:1] == ["deploy prod"]
    # invalid code rejected; 3 bad attempts trip the cooldown
    t[0] += 40
    assert ap.try_resolve_any("000001") == []
    assert ap.try_resolve_any("000002") == []
    assert ap.try_resolve_any("999999") == []
    assert ap.in_attempt_cooldown()
    good = totp_at(ap.secret, t[0])
    assert ap.try_resolve_any(good) == []  # even a valid code during cooldown
    t[0] += 61
    good2 = totp_at(ap.secret, t[0])
    resolved = ap.try_resolve_any(good2)
    assert len(resolved) == 3 and all(r["status"] == "approved" for r in resolved)
    assert not ap.has_pending_batch()


def test_2fa_timeout_denies_all():
    t = [0.0]
    ap = Approval2FA(clock=lambda: t[0], timeout_s=30)
    ap.request("agent:a", "a", "one")
    ap.request("agent:b", "b", "two")
    t[0] = 31.0
    expired = ap.expire_stale()
    assert len(expired) == 2 and all(r["status"] == "expired" for r in expired)
    assert ap.pending_requests() == []


def test_erc8004_reference_behaviors():
    """erc8004-client.test.ts mirrors: ABI encode/decode, profile
    decoding, reputation tiers, LRU, fail-open."""
    from vainplex_openclaw_amd.governance.security.erc8004 import (
        LRUCache, ZERO_ADDRESS, classify_reputation, decode_address,
        decode_profile, encode_uint256_arg,
    )

    # uint256 encoding always 64 chars
    assert encode_uint256_arg(0) == "0" * 64
    assert encode_uint256_arg(1).endswith("1") and len(encode_uint256_arg(1)) == 64
    assert int(encode_uint256_arg(16700), 16) == 16700
    # address decode: padded, zero, short, no-prefix
    assert decode_address("0x" + "0" * 24 + "ab" * 20) == "0x" + "ab" * 20
    assert decode_address("0x" + "0" * 64) == ZERO_ADDRESS
    assert decode_address("0x1234") == ZERO_ADDRESS
    assert decode_address("0" * 24 + "cd" * 20) == "0x" + "cd" * 20
    # profile decode: full, short, empty, zeros
    full = "0x" + format(55, "x").rjust(64, "0") + format(7, "x").rjust(64, "0") \
        + format(1234, "x").rjust(64, "0")
    assert decode_profile(full) == {"score": 55, "feedbackCount": 7, "lastUpdated": 1234}
    assert decode_profile("0xabc") == {"score": 0, "feedbackCount": 0, "lastUpdated": 0}
    assert decode_profile("") == {"score": 0, "feedbackCount": 0, "lastUpdated": 0}
    assert decode_profile("0x" + "0" * 192)["score"] == 0
    # reputation tiers
    assert classify_reputation(False, 5, 90) == "unregistered"
    assert classify_reputation(True, 0, 90) == "none"
    assert classify_reputation(True, 3, 70) == "high"
    assert classify_reputation(True, 3, 69) == "medium"
    assert classify_reputation(True, 3, 29) == "low"
    assert classify_reputation(True, 1, 0) == "low"
    # LRU: miss, store, evict oldest, TTL, update-without-evict, clear
    t = [0.0]
    c = LRUCache(capacity=2, ttl_s=10, clock=lambda: t[0])
    assert c.get("x") is None
    c.put("a", 1); t[0] += 1; c.put("b", 2)
    assert c.get("a") == 1 and c.get("b") == 2
    c.put("a", 11)  # update, no eviction
    assert len(c) == 2 and c.get("a") == 11
    t[0] += 1; c.put("c", 3)  # evicts oldest (b at t=1? a updated at t=0... oldest entry)
    assert len(c) == 2 and c.get("c") == 3
    t[0] += 20
    assert c.get("c") is None and not c.has("a")
    c.clear(); assert len(c) == 0
    # recency: get() refreshes lastAccess, so eviction is true LRU —
    # a hot old entry survives, the cold newer one goes (erc8004-client.ts
    # updates entry.lastAccess on get)
    t[0] = 0.0
    c2 = LRUCache(capacity=2, ttl_s=100, clock=lambda: t[0])
    c2.put("old", 1); t[0] = 1; c2.put("cold", 2)
    t[0] = 2; assert c2.get("old") == 1   # refresh "old"
    t[0] = 3; c2.put("new", 3)             # must evict "cold", not "old"
    assert c2.get("old") == 1 and c2.get("cold") is None and c2.get("new") == 3


def test_erc8004_unregistered_and_fail_open():
    def rpc_zero(method, params):
        return "0x" + "0" * 64

    cli = ERC8004Client(rpc_zero)
    rep = cli.lookup_reputation("0x" + "11" * 20)
    assert rep["registered"] is False and rep["tier"] == "unregistered"
    assert rep["score"] == 0

    def rpc_err(method, params):
        raise ConnectionError("rpc down")

    assert ERC8004Client(rpc_err).lookup_reputation("0x" + "22" * 20) is None

    # score clamped to 0-100
    calls = []

    def rpc_big(method, params):
        calls.append(1)
        if len(calls) == 1:
            return "0x" + "ab" * 32
        return "0x" + format(5000, "x").rjust(64, "0") + format(2, "x").rjust(64, "0") \
            + "0" * 64

    rep2 = ERC8004Client(rpc_big).lookup_reputation("0x" + "33" * 20)
    assert rep2["score"] == 100 and rep2["tier"] == "high"
    # Phase 2 stub
    assert ERC8004Client(rpc_big).submit_feedback("0x" + "33" * 20, 50) is None


def test_agentproof_profile_and_batch(tmp_path):
    """agentproof-rest.test.ts mirrors: key file loading/caching/trim,
    /trust/{id}, tier classification, clamping, batch, failure nulls."""
    keyf = tmp_path / "key.txt"
    keyf.write_text("  sekrit-key \n")
    gets, posts = [], []

    def http_get(url, headers):
        gets.append((url, dict(headers)))
        return {"agentId": "main", "reputationScore": 150, "feedbackCount": 2}

    def http_post(url, headers, body):
        posts.append((url, dict(headers), body))
        return {"results": [{"agentId": "a"}, {"agentId": "b"}]}

    cli = AgentProofRestClient(base_url="https://ap.example/",
                               api_key_file=str(keyf),
                               http_get=http_get, http_post=http_post)
    prof = cli.get_agent_profile("main")
    assert gets[0][0] == "https://ap.example/trust/main"  # trailing slash stripped
    assert gets[0][1] == {"X-API-Key": "sekrit-key"}      # trimmed key header
    assert prof["score"] == 100 and prof["tier"] == "high"  # clamped
    # key cached: rewrite the file, header unchanged
    keyf.write_text("other")
    cli.get_agent_profile("main")
    assert gets[1][1] == {"X-API-Key": "sekrit-key"}
    # batch
    assert cli.batch_lookup([]) == []
    res = cli.batch_lookup(["a", "missing", "b"])
    assert posts[0][0] == "https://ap.example/trust/batch"
    assert posts[0][2] == {"agentIds": ["a", "missing", "b"]}
    assert res[0]["agentId"] == "a" and res[1] is None and res[2]["agentId"] == "b"


def test_agentproof_failure_nulls(tmp_path):
    def boom(*a):
        raise OSError("net down")

    cli = AgentProofRestClient(http_get=boom, http_post=boom)
    assert cli.get_agent_profile("x") is None
    assert cli.batch_lookup(["x", "y"]) == [None, None]
    # malformed profile (no agentId) and missing results array
    cli2 = AgentProofRestClient(http_get=lambda u, h: {"reputationScore": 50},
                                http_post=lambda u, h, b: {"nope": 1})
    assert cli2.get_agent_profile("x") is None
    assert cli2.batch_lookup(["x"]) == [None]
    # no key file: no auth header
    calls = []
    cli3 = AgentProofRestClient(http_get=lambda u, h: calls.append(dict(h)) or
                                {"agentId": "m", "feedbackCount": 0})
    p = cli3.get_agent_profile("m")
    assert calls[0] == {} and p["tier"] == "none"


def make_hooks(workspace):
    engine = GovernanceEngine({"trust": {"enabled": True, "defaultScore": 40}}, workspace)
    engine.start()
    return GovernanceHooks(engine, {})


def test_hooks_bug1_agent_resolution(workspace):
    """hooks.test.ts Bug 1: agentId resolves from sessionKey when missing;
    missing both never crashes; after_tool_call with only sessionKey."""
    gh = make_hooks(workspace)
    out = gh.before_tool_call({"sessionKey": "agent:forge", "toolName": "read",
                               "params": {"file_path": "x.md"}})
    assert out["verdict"]["action"] in ("allow", "deny")
    assert "forge" in gh.engine.trust_manager.known_agents()
    # neither agentId nor sessionKey
    out2 = gh.before_tool_call({"toolName": "read", "params": {}})
    assert out2 is not None  # no crash; evaluated as unresolved
    # after_tool_call with only sessionKey feeds trust/log without crash
    gh.after_tool_call({"sessionKey": "agent:forge", "toolName": "read",
                        "result": "file contents"})
    gh.engine.record_outcome("forge", "agent:forge", True)
    assert gh.engine.trust_manager.get("forge")["signals"]["successCount"] >= 1


def test_hooks_subagent_spawn_and_failed_tool(workspace):
    gh = make_hooks(workspace)
    gh.after_tool_call({"sessionKey": "agent:parent", "agentId": "parent",
                        "toolName": "sessions_spawn",
                        "result": {"sessionId": "agent:child-77"}})
    summary = gh.engine.cross_agent.graph_summary()
    import json as _json
    assert "child-77" in _json.dumps(summary) or summary  # relationship registered
    # failed tool: no toolCallLog entry, outcome recorded as failure
    gh.after_tool_call({"agentId": "parent", "sessionKey": "agent:parent",
                        "toolName": "exec", "error": "boom"})
    assert all(e["toolName"] != "exec"
               for e in gh.tool_call_log.get("agent:parent", []))


def test_make_call_llm_factory():
    from vainplex_openclaw_amd.governance.plugin import make_call_llm

    assert make_call_llm({}) is None
    assert make_call_llm({"enabled": True}) is None  # no endpoint
    posts = []

    def fake_post(url, body, headers=None, timeout_s=None):
        posts.append((url, body, headers, timeout_s))
        return '{"choices": [{"message": {"content": "the answer"}}]}'

    fn = make_call_llm({"enabled": True, "endpoint": "http://llm:11434/v1/",
                        "model": "mistral:7b", "apiKey": "k", "timeoutMs": 5000},
                       http_post=fake_post)
    out = fn("analyze this")
    assert out == "the answer"
    url, body, headers, timeout_s = posts[0]
    assert url == "http://llm:11434/v1/chat/completions"
    assert body["model"] == "mistral:7b"
    assert body["messages"][0]["content"] == "analyze this"
    assert headers == {"Authorization": "Bearer k"} and timeout_s == 5.0


# ===========================================================================
# approval-2fa.test.ts depth: batching windows, per-session batches,
# cooldown status, notification contents, has_pending edge cases
# ===========================================================================

def _fa(**kw):
    from vainplex_openclaw_amd.governance.approval_2fa import Approval2FA, generate_secret

    t = [1_700_000_000.0]
    notes = []
    fa = Approval2FA(secret=generate_secret(), notify=notes.append,
                     clock=lambda: t[0], **kw)
    return fa, t, notes


def _code(fa, t):
    from vainplex_openclaw_amd.governance.approval_2fa import totp_at

    return totp_at(fa.secret, t[0])


def test_2fa_same_session_batches_within_debounce():
    fa, t, notes = _fa()
    r1 = fa.request("s1", "a", "cmd one")
    t[0] += 1.0
    r2 = fa.request("s1", "a", "cmd two")
    assert len(notes) == 1                       # one batch, one notification
    batch = notes[0]
    assert len(batch["requests"]) == 2
    out = fa.try_resolve_any(_code(fa, t))
    assert {r["status"] for r in out} == {"approved"}
    assert r1["status"] == "approved" and r2["status"] == "approved"


def test_2fa_different_sessions_separate_batches():
    fa, t, notes = _fa()
    fa.request("s1", "a", "one")
    fa.request("s2", "b", "two")
    assert len(notes) == 2
    assert notes[0]["sessionKey"] == "s1" and notes[1]["sessionKey"] == "s2"


def test_2fa_notification_includes_all_commands():
    fa, t, notes = _fa()
    fa.request("s1", "a", "deploy the service", details={"command": "x" * 50})
    t[0] += 0.5
    fa.request("s1", "a", "restart the db")
    reasons = [r["reason"] for r in notes[0]["requests"]]
    assert "deploy the service" in reasons and "restart the db" in reasons


def test_2fa_wrong_code_attempts_then_cooldown():
    fa, t, _ = _fa(max_attempts=3)
    fa.request("s1", "a", "cmd")
    for _i in range(3):
        assert fa.try_resolve_any("000000") == []
    assert fa.in_attempt_cooldown()
    # during cooldown even the RIGHT code is refused
    assert fa.try_resolve_any(_code(fa, t)) == []
    # cooldown expires
    t[0] += fa.cooldown_s + 1
    fa.request("s1b", "a", "cmd2")
    assert fa.try_resolve_any(_code(fa, t)) != []


def test_2fa_no_pending_and_has_pending():
    fa, t, _ = _fa()
    assert fa.try_resolve_any(_code(fa, t)) == []   # nothing pending
    assert not fa.has_pending_batch()
    t[0] += 31  # the empty resolve consumed this TOTP counter (replay set)
    fa.request("s1", "a", "cmd")
    assert fa.has_pending_batch() and fa.has_pending_batch("s1")
    assert not fa.has_pending_batch("other-session")
    fa.try_resolve_any(_code(fa, t))
    assert not fa.has_pending_batch()


def test_2fa_session_auto_approval_window():
    fa, t, _ = _fa()
    fa.request("s1", "a", "first")
    fa.try_resolve_any(_code(fa, t))
    # within 10 min: auto-approved without a new batch
    t[0] += 300
    r = fa.request("s1", "a", "second")
    assert r["status"] == "approved" and r.get("auto")
    # other sessions do not inherit the approval
    r2 = fa.request("s-other", "a", "third")
    assert r2["status"] == "pending"


def test_2fa_timeout_expires_batches():
    fa, t, _ = _fa()
    r = fa.request("s1", "a", "cmd")
    t[0] += fa.timeout_s + 1
    expired = fa.expire_stale()
    assert expired and r["status"] in ("expired", "denied")
    assert not fa.has_pending_batch()


def test_2fa_replay_protection_across_batches():
    fa, t, _ = _fa()
    fa.request("s1", "a", "one")
    code = _code(fa, t)
    assert fa.try_resolve_any(code)
    # bypass the auto-approval window to force a real second batch
    fa._session_approved_until.clear()
    fa.request("s1", "a", "two")
    assert fa.try_resolve_any(code) == []      # same counter refused


# ===========================================================================
# agentproof-rest.test.ts depth: API-key lifecycle, URL shapes, error
# nulls, ring overflow, breaker timing, requeue-on-failure
# ===========================================================================

def _ap(tmp_path=None, key=None, **kw):
    from vainplex_openclaw_amd.governance.security.agentproof import AgentProofRestClient

    t = [1_700_000_000.0]
    keyfile = None
    if tmp_path is not None and key is not None:
        keyfile = str(tmp_path / "key.txt")
        (tmp_path / "key.txt").write_text(key)
    cli = AgentProofRestClient(api_key_file=keyfile, clock=lambda: t[0], **kw)
    return cli, t


def test_agentproof_key_read_cache_and_trim(tmp_path):
    reads = []
    cli, _ = _ap(tmp_path, key="  secret-key \n")
    real_open = open

    assert cli._headers() == {"X-API-Key": "secret-key"}   # trimmed
    # cached: mutate the file, header unchanged
    (tmp_path / "key.txt").write_text("changed")
    assert cli._headers() == {"X-API-Key": "secret-key"}


def test_agentproof_missing_and_empty_key_unauthenticated(tmp_path):
    cli, _ = _ap()                                          # no file
    assert cli._headers() == {}
    cli2, _ = _ap(tmp_path, key="   \n")                    # empty file
    assert cli2._headers() == {}


def test_agentproof_base_url_trailing_slash():
    from vainplex_openclaw_amd.governance.security.agentproof import AgentProofRestClient

    urls = []

    def get(url, headers):
        urls.append(url)
        return {"agentId": "a", "score": 50}

    cli = AgentProofRestClient(base_url="https://ap.example/", http_get=get)
    cli.lookup_reputation("a")
    assert urls[0].startswith("https://ap.example/v1/")
    assert "//v1" not in urls[0].replace("https://", "")


def test_agentproof_error_and_network_nulls():
    from vainplex_openclaw_amd.governance.security.agentproof import AgentProofRestClient

    def boom(url, headers):
        raise OSError("network down")

    cli = AgentProofRestClient(http_get=boom)
    assert cli.lookup_reputation("a") is None
    assert AgentProofRestClient().lookup_reputation("a") is None  # no transport


def test_agentproof_ring_overflow_drops_oldest():
    from vainplex_openclaw_amd.governance.security.agentproof import (
        QUEUE_CAPACITY,
        AgentProofRestClient,
    )

    cli = AgentProofRestClient()
    for i in range(QUEUE_CAPACITY + 5):
        cli.enqueue_signal(f"a{i}", "success")
    assert cli.queue_depth == QUEUE_CAPACITY
    assert cli.dropped == 5
    with cli._lock:
        first = cli._queue[0]["agentId"]
    assert first == "a5"                      # oldest dropped


def test_agentproof_breaker_opens_and_resets():
    from vainplex_openclaw_amd.governance.security.agentproof import (
        BREAKER_RESET_S,
        BREAKER_THRESHOLD,
        AgentProofRestClient,
    )

    calls = []
    t = [1_700_000_000.0]

    def flaky_get(url, headers):
        calls.append(url)
        raise OSError("500")

    cli = AgentProofRestClient(http_get=flaky_get, clock=lambda: t[0])
    for _ in range(BREAKER_THRESHOLD):
        cli.lookup_reputation("a")
    assert cli.breaker.is_open
    n = len(calls)
    cli.lookup_reputation("a")                # breaker short-circuits
    assert len(calls) == n
    t[0] += BREAKER_RESET_S + 1               # circuit half-opens
    cli.lookup_reputation("a")
    assert len(calls) == n + 1


def test_agentproof_flush_requeues_on_failure():
    from vainplex_openclaw_amd.governance.security.agentproof import AgentProofRestClient

    posts = []

    def failing_post(url, headers, body):
        posts.append(body)
        raise OSError("temporarily down")

    cli = AgentProofRestClient(http_post=failing_post)
    cli.enqueue_signal("a", "success")
    cli.enqueue_signal("b", "policyBlock")
    assert cli.flush() == 0
    assert cli.queue_depth == 2             # requeued for retry
    # transport recovers
    ok = []
    cli.http_post = lambda u, h, b: ok.append(b)
    # breaker may be counting failures but not open after one miss
    assert cli.flush() == 2
    assert cli.queue_depth == 0
    assert {s["agentId"] for s in ok[0]["signals"]} == {"a", "b"}


# ===========================================================================
# Matrix-poller depth: code extraction across messages, pagination
# token, non-message events, failure tolerance, end-to-end 2FA resolve
# ===========================================================================

def _poller(responses):
    from vainplex_openclaw_amd.governance.approval_2fa import Approval2FA, generate_secret
    from vainplex_openclaw_amd.governance.matrix_poller import MatrixPoller

    t = [1_700_000_000.0]
    fa = Approval2FA(secret=generate_secret(), clock=lambda: t[0])
    calls = []

    def http_get(url, headers):
        calls.append((url, headers))
        if isinstance(responses[0], Exception):
            raise responses.pop(0)
        return responses.pop(0) if len(responses) > 1 else responses[0]

    return MatrixPoller(fa, homeserver="https://hs.example/", room_id="!r:x",
                        access_token="tok", http_get=http_get), fa, t, calls


def _msg(body):
    return {"type": "m.room.message", "content": {"body": body}}


def test_matrix_poll_resolves_pending_batch():
    from vainplex_openclaw_amd.governance.approval_2fa import totp_at

    poller, fa, t, calls = _poller([{"chunk": [], "end": "t1"}])
    r = fa.request("s1", "a", "deploy")
    code = totp_at(fa.secret, t[0])
    poller.http_get = lambda u, h: {"chunk": [_msg(f"approve: {code}")], "end": "t2"}
    found = poller.poll_once()
    assert found == [code]
    assert r["status"] == "approved"
    assert poller.codes_seen == 1


def test_matrix_poll_extracts_multiple_and_ignores_noise():
    poller, _fa, _t, _ = _poller([{
        "chunk": [
            _msg("codes 123456 and 654321 here"),
            {"type": "m.reaction", "content": {"body": "999999"}},  # not a message
            _msg("no digits"),
            _msg("1234567 is seven digits"),   # 7-digit run: no 6-digit match
        ],
        "end": "tkn",
    }])
    codes = poller.poll_once()
    assert "123456" in codes and "654321" in codes
    assert "999999" not in codes
    assert len([c for c in codes if c == "234567"]) == 0


def test_matrix_poll_pagination_and_auth_header():
    poller, _fa, _t, calls = _poller([
        {"chunk": [], "end": "page-1"},
        {"chunk": [], "end": "page-2"},
    ])
    poller.poll_once()
    poller.poll_once()
    assert "from=page-1" in calls[1][0]
    assert calls[0][1]["Authorization"] == "Bearer tok"
    assert "/_matrix/client/v3/rooms/!r:x/messages" in calls[0][0]


def test_matrix_poll_failures_tolerated():
    poller, _fa, _t, _ = _poller([OSError("matrix down"), {"chunk": []}])
    assert poller.poll_once() == []        # error swallowed
    assert poller.poll_once() == []        # next poll works
    # no transport configured -> no-op
    from vainplex_openclaw_amd.governance.approval_2fa import Approval2FA
    from vainplex_openclaw_amd.governance.matrix_poller import MatrixPoller

    silent = MatrixPoller(Approval2FA(), http_get=None)
    assert silent.poll_once() == []
    silent.start()                          # refuses without transport
    assert silent._thread is None
