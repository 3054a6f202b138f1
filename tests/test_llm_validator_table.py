"""LLM-validator tables mirroring `test/llm-validator.test.ts` (32 its):
prompt construction, response parsing (issues schema, fences, severity
mapping, malformed inputs), caching/TTL, retries, fail modes.
"""

import json

import pytest

from vainplex_openclaw_amd.governance.llm_validator import (
    DEFAULT_EXTERNAL_CHANNELS,
    LlmValidator,
    djb2,
    is_external_comm,
    parse_llm_response,
)


# -- prompt construction -----------------------------------------------------

def _prompt_of(text, facts):
    captured = {}

    def grab(p):
        captured["p"] = p
        return '{"verdict": "pass"}'

    LlmValidator(grab).validate(text, facts, True)
    return captured["p"]


def test_prompt_includes_text_and_facts():
    p = _prompt_of("the queue has 42 items",
                   [{"subject": "queue", "predicate": "count", "value": "42"}])
    assert "the queue has 42 items" in p
    assert "queue count: 42" in p


def test_prompt_no_facts_marker():
    p = _prompt_of("hello world", [])
    assert "(none)" in p


def test_prompt_requests_json():
    assert "JSON" in _prompt_of("x", [])


def test_prompt_caps_fact_list():
    facts = [{"subject": f"s{i}", "predicate": "p", "value": "v"} for i in range(80)]
    p = _prompt_of("x", facts)
    assert "s49" in p and "s50" not in p  # first 50 only


# -- parse_llm_response table ------------------------------------------------

@pytest.mark.parametrize("raw,verdict", [
    ('{"verdict": "pass"}', "pass"),
    ('{"verdict": "flag", "reason": "r"}', "flag"),
    ('{"verdict": "block"}', "block"),
    ('{"verdict": "banana"}', "pass"),                     # unknown -> pass
    ('{"issues": []}', "pass"),
    ('{"issues": [{"severity": "critical", "description": "bad"}]}', "block"),
    ('{"issues": [{"severity": "high"}]}', "flag"),
    ('{"issues": [{"severity": "medium"}]}', "flag"),
    ('{"issues": [{"severity": "low"}]}', "pass"),
    ('{"issues": [{"severity": "wat"}]}', "flag"),         # unknown sev -> medium
    ('{"issues": [{"severity": "low"}, {"severity": "critical"}]}', "block"),
    ('```json\n{"issues": [{"severity": "high"}]}\n```', "flag"),
    ('```\n{"verdict": "block"}\n```', "block"),
    ("total garbage", "pass"),
    ('{"no_issues_field": 1}', "pass"),
    ('{"issues": "not-an-array"}', "pass"),
    ('{"issues": [42, {"severity": "critical"}]}', "block"),  # malformed skipped
    ("[1,2,3]", "pass"),
])
def test_parse_response_table(raw, verdict):
    assert parse_llm_response(raw)["verdict"] == verdict, raw


def test_parse_response_collects_descriptions():
    raw = json.dumps({"issues": [
        {"severity": "high", "description": "claim A unverified"},
        {"severity": "medium", "description": "claim B odd"},
    ]})
    r = parse_llm_response(raw)
    assert "claim A unverified" in r["reason"] and "claim B odd" in r["reason"]


# -- validate(): gating, cache, retries, fail modes --------------------------

def test_disabled_and_internal_pass_without_call():
    calls = []
    v = LlmValidator(None)
    assert v.validate("x", [], True)["verdict"] == "pass"

    v2 = LlmValidator(lambda p: calls.append(p) or '{"verdict": "block"}')
    assert v2.validate("x", [], False)["verdict"] == "pass"   # internal
    assert v2.validate("", [], True)["verdict"] == "pass"     # empty
    assert calls == []


def test_block_on_critical_issue():
    v = LlmValidator(lambda p: '{"issues": [{"severity": "critical"}]}')
    assert v.validate("tweet this", [], True)["verdict"] == "block"


def test_cache_same_text_and_ttl_expiry():
    t = [0.0]
    calls = []

    def llm(p):
        calls.append(p)
        return '{"verdict": "flag"}'

    v = LlmValidator(llm, cache_ttl_s=300, clock=lambda: t[0])
    r1 = v.validate("same text", [], True)
    r2 = v.validate("same text", [], True)
    assert len(calls) == 1 and r2["cached"] and r2["verdict"] == "flag"
    # different text misses
    v.validate("other text", [], True)
    assert len(calls) == 2
    # expiry re-calls
    t[0] = 301
    v.validate("same text", [], True)
    assert len(calls) == 3
    assert v.stats["cache_hits"] == 1


def test_retry_then_success():
    n = [0]

    def flaky(p):
        n[0] += 1
        if n[0] == 1:
            raise OSError("transient")
        return '{"verdict": "pass"}'

    v = LlmValidator(flaky, max_attempts=2)
    r = v.validate("x", [], True)
    assert r["verdict"] == "pass" and "llm-error" not in r["reason"]
    assert v.stats["retries"] == 1 and v.stats["errors"] == 0


def test_retries_exhausted_fail_open_and_closed():
    def dead(p):
        raise OSError("down")

    v = LlmValidator(dead, max_attempts=3)
    r = v.validate("x", [], True)
    assert r["verdict"] == "pass" and "llm-error" in r["reason"]
    assert v.stats["retries"] == 2 and v.stats["errors"] == 1

    v2 = LlmValidator(dead, fail_mode="closed")
    assert v2.validate("x", [], True)["verdict"] == "flag"


def test_failure_not_cached():
    n = [0]

    def once_dead(p):
        n[0] += 1
        if n[0] <= 2:  # both attempts of the first validate fail
            raise OSError("down")
        return '{"verdict": "block"}'

    v = LlmValidator(once_dead)
    assert v.validate("x", [], True)["verdict"] == "pass"
    # recovery is visible (the error result was not cached)
    assert v.validate("x", [], True)["verdict"] == "block"


def test_cache_sweep_bounds_size():
    t = [0.0]
    v = LlmValidator(lambda p: '{"verdict": "pass"}', cache_ttl_s=10,
                     clock=lambda: t[0])
    for i in range(1030):
        v.validate(f"text {i}", [], True)
        t[0] += 0.02
    assert len(v._cache) <= 1026  # expired entries swept past the bound


# -- djb2 / external detection ----------------------------------------------

def test_djb2_deterministic_distinct():
    assert djb2("abc") == djb2("abc")
    assert djb2("abc") != djb2("abd")
    assert isinstance(djb2(""), int)


@pytest.mark.parametrize("channel,tool,command,want", [
    ("twitter", None, None, True),
    ("linkedin", None, None, True),
    ("email", None, None, True),
    ("internal", None, None, False),
    (None, "exec", "bird tweet hello", True),
    (None, "exec", "ls -la", False),
    (None, None, None, False),
])
def test_is_external_comm_table(channel, tool, command, want):
    assert is_external_comm(channel, tool, command,
                            DEFAULT_EXTERNAL_CHANNELS,
                            ("bird tweet",)) == want
