"""Governance utility tables mirroring `test/util.test.ts` (47 its):
time parsing/windows, glob compilation, tier mapping, agent-id
resolution fallback chains, sub-agent session keys, clamp/hash helpers.
"""

import pytest

from vainplex_openclaw_amd.governance.audit import sha256_hex
from vainplex_openclaw_amd.governance.conditions import (
    glob_to_regex,
    parse_time_to_minutes,
)
from vainplex_openclaw_amd.governance.util import (
    clamp,
    extract_agent_id,
    is_sub_agent,
    parent_session_key,
    resolve_agent_id,
    score_to_tier,
    tier_ordinal,
)
from vainplex_openclaw_amd.utils.storage import now_us


# -- parseTimeToMinutes ------------------------------------------------------

@pytest.mark.parametrize("text,want", [
    ("00:00", 0), ("01:30", 90), ("12:00", 720), ("23:59", 1439),
    ("09:05", 545), ("18:45", 1125),
])
def test_parse_time_valid(text, want):
    assert parse_time_to_minutes(text) == want


@pytest.mark.parametrize("text", ["", "24:00", "12:60", "noon",
                                  "-1:00", "ab:cd", "25:99"])
def test_parse_time_invalid(text):
    assert parse_time_to_minutes(text) == -1


def test_parse_time_extra_parts_ignored():
    # util.ts Number(parts[0])/Number(parts[1]): trailing parts ignored
    assert parse_time_to_minutes("1:5:9") == 65


# -- isInTimeRange (util.ts:16-26; [after, before) with midnight wrap) ------

@pytest.mark.parametrize("cur,after,before,want", [
    (600, 540, 1020, True),    # 10:00 in 09:00-17:00
    (539, 540, 1020, False),   # just before
    (1021, 540, 1020, False),  # just after
    (540, 540, 1020, True),    # inclusive start
    (1019, 540, 1020, True),   # last included minute
    (1020, 540, 1020, False),  # exclusive end
    (1410, 1320, 360, True),   # 23:30 in 22:00-06:00 wrap
    (180, 1320, 360, True),    # 03:00 in wrap
    (720, 1320, 360, False),   # noon outside wrap
    (600, 480, 480, False),    # equal start/end = EMPTY range (util.test.ts:46-48)
])
def test_is_in_time_range_table(cur, after, before, want):
    from vainplex_openclaw_amd.governance.conditions import is_in_time_range

    assert is_in_time_range(cur, after, before) == want


def test_time_condition_after_before_through_evaluator():
    from vainplex_openclaw_amd.governance.conditions import (
        ConditionDeps,
        evaluate_conditions,
    )

    deps = ConditionDeps()

    def at(h, m=0):
        return {"time": {"hour": h, "minute": m, "dayOfWeek": "Mon"}}

    cond = [{"type": "time", "after": "09:00", "before": "17:00"}]
    assert evaluate_conditions(cond, at(10), deps)
    assert not evaluate_conditions(cond, at(18), deps)
    wrap = [{"type": "time", "after": "22:00", "before": "06:00"}]
    assert evaluate_conditions(wrap, at(23, 30), deps)
    assert not evaluate_conditions(wrap, at(12), deps)


# -- globToRegex -------------------------------------------------------------

@pytest.mark.parametrize("glob,text,want", [
    ("exec", "exec", True),
    ("exec", "exec2", False),
    ("ex*", "exec", True),
    ("*", "anything", True),
    ("file-?.txt", "file-1.txt", True),
    ("file-?.txt", "file-12.txt", False),
    ("a.b", "a.b", True),
    ("a.b", "axb", False),          # dot must be escaped, not wildcard
    ("a+b", "a+b", True),           # regex specials escaped
    ("(x)", "(x)", True),
    ("tool_*", "tool_read", True),
    ("tool_*", "mytool_read", False),  # anchored match
])
def test_glob_to_regex_table(glob, text, want):
    assert bool(glob_to_regex(glob).fullmatch(text)) == want


# -- sha256 / clamp / nowUs --------------------------------------------------

def test_sha256_hex_shape_and_determinism():
    h = sha256_hex(b"governance")
    assert len(h) == 64 and all(c in "0123456789abcdef" for c in h)
    assert h == sha256_hex(b"governance")
    assert h != sha256_hex(b"governance2")


@pytest.mark.parametrize("v,lo,hi,want", [
    (5, 0, 10, 5), (-1, 0, 10, 0), (11, 0, 10, 10),
    (0, 0, 10, 0), (10, 0, 10, 10), (3.5, 1.0, 2.0, 2.0),
])
def test_clamp_table(v, lo, hi, want):
    assert clamp(v, lo, hi) == want


def test_now_us_positive_and_monotonic_enough():
    a = now_us()
    b = now_us()
    assert a > 0 and b >= a


# -- scoreToTier / tierOrdinal (util.ts:192-198) -----------------------------

@pytest.mark.parametrize("score,tier", [
    (100, "elevated"), (80, "elevated"), (79.9, "trusted"), (60, "trusted"),
    (59.9, "standard"), (40, "standard"), (39.9, "restricted"),
    (20, "restricted"), (19.9, "untrusted"), (0, "untrusted"),
])
def test_score_to_tier_table(score, tier):
    assert score_to_tier(score) == tier


def test_tier_ordinal_strictly_increasing():
    tiers = ["untrusted", "restricted", "standard", "trusted", "elevated"]
    ords = [tier_ordinal(t) for t in tiers]
    assert ords == sorted(ords) and len(set(ords)) == 5


# -- extractAgentId ----------------------------------------------------------

@pytest.mark.parametrize("session_key,agent_id,want", [
    (None, "explicit", "explicit"),
    ("agent:main", "explicit", "explicit"),       # explicit wins
    ("agent:main", None, "main"),
    ("agent:main:subagent:forge:abc", None, "forge"),
    (None, None, "unknown"),
    ("", None, "unknown"),
    ("nocolonshere", None, "unknown"),
    ("agent:", None, "unknown"),
])
def test_extract_agent_id_table(session_key, agent_id, want):
    assert extract_agent_id(session_key, agent_id) == want


# -- isSubAgent / parentSessionKey ------------------------------------------

@pytest.mark.parametrize("key,want", [
    ("agent:main:subagent:forge:abc", True),
    ("agent:main", False),
    (None, False),
    ("", False),
])
def test_is_sub_agent_table(key, want):
    assert is_sub_agent(key) == want


def test_parent_session_key():
    assert parent_session_key("agent:main:subagent:forge:abc") == "agent:main"
    assert parent_session_key("agent:main") is None


# -- resolveAgentId fallback chain (util.ts resolveAgentId, 12 its) ----------

@pytest.mark.parametrize("ctx,event,want", [
    ({"agentId": "a1"}, None, "a1"),
    ({"sessionKey": "agent:root"}, None, "root"),
    ({"sessionKey": "agent:main:subagent:kid:1"}, None, "kid"),
    ({}, None, "unresolved"),
    ({"sessionKey": "8c2e7b90-1b2c-4d5e"}, None, "unresolved"),  # UUID-ish
    ({"sessionId": "agent:viaid"}, None, "viaid"),
    ({}, {"metadata": {"agentId": "meta"}}, "meta"),
    ({"agentId": "a1", "sessionKey": "agent:other"}, None, "a1"),
    ({"sessionKey": "agent:sk", "sessionId": "agent:sid"}, None, "sk"),
    ({"sessionId": "agent:sid"}, {"metadata": {"agentId": "meta"}}, "sid"),
    ({"agentId": ""}, None, "unresolved"),       # empty string is not an id
    ({"sessionKey": None, "sessionId": None}, {"metadata": {}}, "unresolved"),
])
def test_resolve_agent_id_chain(ctx, event, want):
    assert resolve_agent_id(ctx, event) == want
