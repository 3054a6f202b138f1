"""Response-gate tables mirroring `test/response-gate.test.ts` (27 its)."""

import pytest

from vainplex_openclaw_amd.governance.response_gate import (
    ResponseGate,
    resolve_response_gate,
)

LOG = [{"toolName": "read"}, {"toolName": "search"}]


def gate(rules=None, **extra):
    cfg = {"enabled": True, "rules": rules or []}
    cfg.update(extra)
    return ResponseGate(cfg)


def test_disabled_always_passes():
    g = ResponseGate({"enabled": False, "rules": [
        {"validators": [{"type": "mustMatch", "pattern": "^nope$"}]}]})
    assert g.validate("anything", "a", [])["passed"]
    assert ResponseGate(None).validate("x", "a", [])["passed"]


# -- requiredTools ------------------------------------------------------------

def test_required_tools_pass_and_block():
    rules = [{"validators": [{"type": "requiredTools", "tools": ["read"]}]}]
    assert gate(rules).validate("x", "a", LOG)["passed"]
    r = gate(rules).validate("x", "a", [])
    assert not r["passed"]
    assert "read" in r["reasons"][0]


def test_required_tools_partial_blocks():
    rules = [{"validators": [{"type": "requiredTools",
                              "tools": ["read", "verify"]}]}]
    r = gate(rules).validate("x", "a", LOG)
    assert not r["passed"] and "verify" in r["reasons"][0]
    assert "read" not in r["reasons"][0].split(":")[-1]  # only missing listed


def test_required_tools_custom_message():
    rules = [{"validators": [{"type": "requiredTools", "tools": ["verify"],
                              "message": "verify before replying"}]}]
    r = gate(rules).validate("x", "a", [])
    assert r["reasons"] == ["verify before replying"]


# -- mustMatch / mustNotMatch -------------------------------------------------

@pytest.mark.parametrize("vtype,pattern,content,want", [
    ("mustMatch", r"\bsources?:", "sources: a, b", True),
    ("mustMatch", r"\bsources?:", "no citations here", False),
    ("mustNotMatch", r"(?i)as an ai", "I checked the logs", True),
    ("mustNotMatch", r"(?i)as an ai", "As an AI, I cannot", False),
])
def test_match_validators(vtype, pattern, content, want):
    rules = [{"validators": [{"type": vtype, "pattern": pattern}]}]
    assert gate(rules).validate(content, "a", [])["passed"] == want


# -- agent scoping ------------------------------------------------------------

def test_rule_agent_scoping_single_list_wildcard():
    rules = [{"agentId": "main",
              "validators": [{"type": "mustMatch", "pattern": "x"}]}]
    assert not gate(rules).validate("yyy", "main", [])["passed"]
    assert gate(rules).validate("yyy", "other", [])["passed"]

    rules2 = [{"agentId": ["a", "b"],
               "validators": [{"type": "mustMatch", "pattern": "x"}]}]
    assert not gate(rules2).validate("yyy", "a", [])["passed"]
    assert not gate(rules2).validate("yyy", "b", [])["passed"]
    assert gate(rules2).validate("yyy", "c", [])["passed"]

    rules3 = [{"validators": [{"type": "mustMatch", "pattern": "x"}]}]
    assert not gate(rules3).validate("yyy", "anyone", [])["passed"]


def test_all_validators_must_pass_and_failures_accumulate():
    rules = [{"validators": [
        {"type": "mustMatch", "pattern": "alpha"},
        {"type": "mustNotMatch", "pattern": "beta"},
        {"type": "requiredTools", "tools": ["verify"]},
    ]}]
    r = gate(rules).validate("beta only", "a", [])
    assert not r["passed"]
    assert len(r["failedValidators"]) == 3
    assert len(r["reasons"]) == 3


# -- invalid regex = fail-closed ---------------------------------------------

@pytest.mark.parametrize("vtype", ["mustMatch", "mustNotMatch"])
def test_invalid_regex_blocks(vtype):
    rules = [{"validators": [{"type": vtype, "pattern": "(unclosed"}]}]
    r = gate(rules).validate("anything", "a", [])
    assert not r["passed"]
    assert "fail-closed" in r["reasons"][0]


def test_regex_cache_reused():
    rules = [{"validators": [{"type": "mustMatch", "pattern": "cachepat"}]}]
    g = gate(rules)
    g.validate("cachepat here", "a", [])
    g.validate("cachepat again", "a", [])
    assert len(g._regex_cache) == 1


# -- config resolver ----------------------------------------------------------

def test_resolver_defaults_and_garbage():
    assert resolve_response_gate(None)["enabled"] is False
    assert resolve_response_gate("garbage")["rules"] == []
    assert resolve_response_gate({"rules": "nope"})["rules"] == []
    cfg = resolve_response_gate({"enabled": True, "rules": [{"x": 1}],
                                 "fallbackMessage": "m", "fallbackTemplate": "t"})
    assert cfg["enabled"] and cfg["rules"] == [{"x": 1}]
    assert cfg["fallbackMessage"] == "m" and cfg["fallbackTemplate"] == "t"


def test_resolver_drops_non_string_fallbacks():
    cfg = resolve_response_gate({"fallbackMessage": 42, "fallbackTemplate": ["x"]})
    assert cfg["fallbackMessage"] is None and cfg["fallbackTemplate"] is None


# -- fallback messaging -------------------------------------------------------

FAIL_RULES = [{"validators": [{"type": "mustMatch", "pattern": "zzz"}]}]


def test_no_fallback_when_unconfigured_or_passing():
    r = gate(FAIL_RULES).validate("no match", "a", [])
    assert not r["passed"] and "fallbackMessage" not in r
    g = gate(FAIL_RULES, fallbackMessage="static msg")
    ok = g.validate("zzz present", "a", [])
    assert ok["passed"] and "fallbackMessage" not in ok


def test_static_fallback_message_verbatim():
    g = gate(FAIL_RULES, fallbackMessage="blocked: {reasons}")
    r = g.validate("no match", "a", [])
    # fallbackMessage is static — template vars NOT rendered
    assert r["fallbackMessage"] == "blocked: {reasons}"


def test_template_renders_variables():
    g = gate(FAIL_RULES, fallbackTemplate="agent={agent} failed={validators} why={reasons}")
    r = g.validate("no match", "agent7", [])
    fb = r["fallbackMessage"]
    assert "agent=agent7" in fb
    assert "mustMatch:zzz" in fb
    assert "why=" in fb and "{reasons}" not in fb


def test_message_takes_precedence_over_template():
    g = gate(FAIL_RULES, fallbackMessage="static", fallbackTemplate="t {agent}")
    assert g.validate("no match", "a", [])["fallbackMessage"] == "static"
