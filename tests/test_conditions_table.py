"""Condition-evaluator tables mirroring governance
`test/conditions/{simple,tool,time,context}.test.ts` (65 its):
agent/risk/frequency/any/not (simple), tool name+param matchers, time
windows with midnight wrap and day filters, and the five context
matchers — all through the same evaluator map the policy evaluator uses.
"""

import pytest

from vainplex_openclaw_amd.governance.conditions import (
    ConditionDeps,
    RegexCache,
    evaluate_condition,
    evaluate_conditions,
)
from vainplex_openclaw_amd.governance.util import tier_ordinal


def ctx(**over):
    base = {
        "hook": "before_tool_call",
        "agentId": "forge",
        "sessionKey": "agent:main:subagent:forge:abc",
        "time": {"hour": 12, "minute": 0, "dayOfWeek": 3},
        "trust": {"agent": {"agentId": "forge", "score": 45, "tier": "standard"},
                  "session": {"score": 45, "tier": "standard"}},
        "toolName": "exec",
    }
    base.update(over)
    return base


def deps(**over):
    d = ConditionDeps(risk=over.pop("risk", {"level": "medium", "score": 50}),
                      time_windows=over.pop("time_windows", {}),
                      frequency_tracker=over.pop("frequency_tracker", None))
    return d


class _Freq:
    def __init__(self, n):
        self.n = n

    def count(self, *a, **k):
        return self.n


# -- agent --------------------------------------------------------------------

@pytest.mark.parametrize("cond,context,want", [
    ({"type": "agent", "id": "forge"}, {}, True),
    ({"type": "agent", "id": "forge"}, {"agentId": "main"}, False),
    ({"type": "agent", "id": "forge*"}, {}, True),
    ({"type": "agent", "id": ["forge", "cerberus"]}, {}, True),
    ({"type": "agent", "id": ["forge", "cerberus"]}, {"agentId": "main"}, False),
    ({"type": "agent", "trustTier": "standard"}, {}, True),
    ({"type": "agent", "trustTier": ["standard", "trusted"]}, {}, True),
    ({"type": "agent", "minScore": 40}, {}, True),     # 45 >= 40
    ({"type": "agent", "maxScore": 50}, {}, True),     # 45 <= 50
    ({"type": "agent"}, {}, True),                     # empty = any agent
])
def test_agent_condition_matrix(cond, context, want):
    assert evaluate_condition(cond, ctx(**context), deps()) is want


def test_agent_tier_and_score_reject():
    high = {"trust": {"agent": {"score": 80, "tier": "elevated"},
                      "session": {"score": 80, "tier": "elevated"}}}
    assert evaluate_condition({"type": "agent", "trustTier": "standard"},
                              ctx(**high), deps()) is False
    low = {"trust": {"agent": {"score": 30, "tier": "restricted"},
                     "session": {"score": 30, "tier": "restricted"}}}
    assert evaluate_condition({"type": "agent", "minScore": 40},
                              ctx(**low), deps()) is False
    assert evaluate_condition({"type": "agent", "maxScore": 50},
                              ctx(**high), deps()) is False


# -- risk ---------------------------------------------------------------------

@pytest.mark.parametrize("cond,level,want", [
    ({"type": "risk", "minRisk": "medium", "maxRisk": "high"}, "medium", True),
    ({"type": "risk", "minRisk": "medium", "maxRisk": "high"}, "high", True),
    ({"type": "risk", "minRisk": "high"}, "low", False),
    ({"type": "risk", "maxRisk": "medium"}, "critical", False),
    ({"type": "risk"}, "low", True),                   # unconstrained
])
def test_risk_condition_matrix(cond, level, want):
    assert evaluate_condition(cond, ctx(), deps(risk={"level": level})) is want


# -- frequency ----------------------------------------------------------------

@pytest.mark.parametrize("count,max_count,want", [
    (5, 5, True),       # at limit -> exceeded
    (3, 5, False),      # under
    (10, 10, True),     # exactly at limit, session scope
])
def test_frequency_condition_matrix(count, max_count, want):
    cond = {"type": "frequency", "maxCount": max_count, "windowSeconds": 60}
    assert evaluate_condition(cond, ctx(), deps(frequency_tracker=_Freq(count))) is want


# -- any / not ----------------------------------------------------------------

def test_any_is_or():
    cond = {"type": "any", "conditions": [
        {"type": "tool", "name": "read"}, {"type": "tool", "name": "exec"}]}
    assert evaluate_condition(cond, ctx(), deps()) is True
    cond2 = {"type": "any", "conditions": [
        {"type": "tool", "name": "read"}, {"type": "tool", "name": "write"}]}
    assert evaluate_condition(cond2, ctx(), deps()) is False


def test_not_negates_and_nests():
    assert evaluate_condition({"type": "not", "condition": {"type": "tool", "name": "exec"}},
                              ctx(), deps()) is False
    assert evaluate_condition({"type": "not", "condition": {"type": "tool", "name": "read"}},
                              ctx(), deps()) is True
    nested = {"type": "not", "condition": {"type": "any", "conditions": [
        {"type": "tool", "name": "read"}, {"type": "tool", "name": "write"}]}}
    assert evaluate_condition(nested, ctx(), deps()) is True


def test_tier_ordinal_helpers():
    assert tier_ordinal("trusted") >= tier_ordinal("standard")
    assert tier_ordinal("restricted") < tier_ordinal("standard")
    assert tier_ordinal("standard") <= tier_ordinal("trusted")
    assert tier_ordinal("elevated") > tier_ordinal("trusted")


# -- tool ---------------------------------------------------------------------

@pytest.mark.parametrize("cond,context,want", [
    ({"type": "tool", "name": "exec"}, {}, True),
    ({"type": "tool", "name": "ex*"}, {}, True),
    ({"type": "tool", "name": ["read", "exec"]}, {}, True),
    ({"type": "tool", "name": "read"}, {}, False),
    ({"type": "tool"}, {}, True),                       # any tool
    ({"type": "tool", "name": "exec"}, {"toolName": None}, False),
    ({"type": "tool", "params": {"command": {"contains": "rm"}}},
     {"toolParams": {"command": "rm -rf /tmp/x"}}, True),
    ({"type": "tool", "params": {"command": {"equals": "ls"}}},
     {"toolParams": {"command": "ls"}}, True),
    ({"type": "tool", "params": {"command": {"matches": r"^rm\s+-rf"}}},
     {"toolParams": {"command": "rm -rf /"}}, True),
    ({"type": "tool", "params": {"command": {"startsWith": "sudo"}}},
     {"toolParams": {"command": "sudo reboot"}}, True),
    ({"type": "tool", "params": {"env": {"in": ["prod", "staging"]}}},
     {"toolParams": {"env": "prod"}}, True),
    ({"type": "tool", "params": {"command": {"contains": "rm"}}}, {}, False),
    # non-string values never match string matchers
    ({"type": "tool", "params": {"n": {"matches": r"\d+"}}},
     {"toolParams": {"n": 42}}, False),
    ({"type": "tool", "params": {"n": {"contains": "4"}}},
     {"toolParams": {"n": 42}}, False),
    ({"type": "tool", "params": {"n": {"startsWith": "4"}}},
     {"toolParams": {"n": 42}}, False),
    # invalid regex is dropped by the cache -> no match, no raise
    ({"type": "tool", "params": {"c": {"matches": "(unclosed"}}},
     {"toolParams": {"c": "anything"}}, False),
])
def test_tool_condition_matrix(cond, context, want):
    assert evaluate_condition(cond, ctx(**context), deps()) is want


def test_regex_cache_reuses_and_caches_invalid():
    cache = RegexCache()
    rx1 = cache.get(r"\d+")
    rx2 = cache.get(r"\d+")
    assert rx1 is rx2
    assert cache.get("(unclosed") is None
    assert cache.get("(unclosed") is None


# -- time ---------------------------------------------------------------------

def t_at(hour, minute=0, dow=3):
    return ctx(time={"hour": hour, "minute": minute, "dayOfWeek": dow})


@pytest.mark.parametrize("cond,hour,want", [
    ({"type": "time", "after": "09:00", "before": "17:00"}, 12, True),
    ({"type": "time", "after": "09:00", "before": "17:00"}, 8, False),
    ({"type": "time", "after": "22:00", "before": "06:00"}, 23, True),   # wrap
    ({"type": "time", "after": "22:00", "before": "06:00"}, 3, True),    # wrap
    ({"type": "time", "after": "22:00", "before": "06:00"}, 12, False),
    ({"type": "time", "after": "14:00"}, 15, True),                      # after-only
    ({"type": "time", "after": "14:00"}, 13, False),
    ({"type": "time", "before": "14:00"}, 13, True),                     # before-only
    ({"type": "time", "before": "14:00"}, 15, False),
    ({"type": "time"}, 12, True),                                        # empty
])
def test_time_condition_matrix(cond, hour, want):
    assert evaluate_condition(cond, t_at(hour), deps()) is want


def test_time_day_filter_and_days_only():
    cond = {"type": "time", "after": "09:00", "before": "17:00", "days": [1, 2, 3, 4, 5]}
    assert evaluate_condition(cond, t_at(12, dow=3), deps()) is True
    assert evaluate_condition(cond, t_at(12, dow=0), deps()) is False
    days_only = {"type": "time", "days": [3]}
    assert evaluate_condition(days_only, t_at(2, dow=3), deps()) is True
    assert evaluate_condition(days_only, t_at(2, dow=4), deps()) is False


def test_time_named_windows():
    windows = {"night": {"start": "23:00", "end": "08:00"},
               "weekday-work": {"start": "09:00", "end": "17:00", "days": [1, 2, 3, 4, 5]}}
    d = deps(time_windows=windows)
    assert evaluate_condition({"type": "time", "window": "night"}, t_at(2), d) is True
    assert evaluate_condition({"type": "time", "window": "night"}, t_at(12), d) is False
    assert evaluate_condition({"type": "time", "window": "ghost"}, t_at(2), d) is False
    assert evaluate_condition({"type": "time", "window": "weekday-work"}, t_at(10, dow=2), d) is True
    assert evaluate_condition({"type": "time", "window": "weekday-work"}, t_at(10, dow=6), d) is False


def test_time_equal_start_end_window_is_empty():
    d = deps(time_windows={"zero": {"start": "09:00", "end": "09:00"}})
    assert evaluate_condition({"type": "time", "window": "zero"}, t_at(9), d) is False
    assert evaluate_condition({"type": "time", "window": "zero"}, t_at(12), d) is False


# -- context ------------------------------------------------------------------

@pytest.mark.parametrize("cond,context,want", [
    ({"type": "context", "conversationContains": r"deploy\s+to\s+prod"},
     {"conversationContext": ["please deploy to prod now"]}, True),
    ({"type": "context", "conversationContains": "missing-phrase"},
     {"conversationContext": ["unrelated text"]}, False),
    ({"type": "context", "conversationContains": "x"},
     {"conversationContext": []}, False),
    ({"type": "context", "conversationContains": ["alpha", "deploy"]},
     {"conversationContext": ["we deploy tonight"]}, True),
    ({"type": "context", "messageContains": "secret"},
     {"messageContent": "the secret plan"}, True),
    ({"type": "context", "messageContains": "secret"},
     {"messageContent": "nothing here"}, False),
    ({"type": "context", "messageContains": "secret"}, {}, False),
    ({"type": "context", "hasMetadata": "source"},
     {"metadata": {"source": "api"}}, True),
    ({"type": "context", "hasMetadata": ["source", "user"]},
     {"metadata": {"source": "api", "user": "bob"}}, True),
    ({"type": "context", "hasMetadata": ["source", "user"]},
     {"metadata": {"source": "api"}}, False),
    ({"type": "context", "channel": "slack"}, {"channel": "slack"}, True),
    ({"type": "context", "channel": ["slack", "matrix"]}, {"channel": "matrix"}, True),
    ({"type": "context", "channel": "slack"}, {"channel": "email"}, False),
    ({"type": "context", "sessionKey": "agent:main:*"}, {}, True),
    ({"type": "context", "sessionKey": "agent:other:*"}, {}, False),
    ({"type": "context"}, {}, True),                     # no constraints
])
def test_context_condition_matrix(cond, context, want):
    assert evaluate_condition(cond, ctx(**context), deps()) is want


# -- AND combination + unknown types ------------------------------------------

def test_conditions_are_anded():
    conds = [{"type": "tool", "name": "exec"},
             {"type": "agent", "id": "forge"},
             {"type": "risk", "minRisk": "medium"}]
    assert evaluate_conditions(conds, ctx(), deps()) is True
    conds.append({"type": "time", "after": "20:00", "before": "23:00"})
    assert evaluate_conditions(conds, ctx(), deps()) is False


def test_unknown_condition_type_never_matches():
    assert evaluate_condition({"type": "quantum"}, ctx(), deps()) is False
    assert evaluate_conditions([], ctx(), deps()) is True   # vacuous AND
