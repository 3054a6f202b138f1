"""Policy rule conditions: 8 condition types, AND-combined per rule.

Parity target: governance `src/conditions/{index,simple,tool,time,context}.ts`
— `tool` (name globs + param matchers equals/contains/matches/startsWith/in),
`time` (named windows + after/before with midnight wrap + days), `context`
(conversationContains/messageContains/hasMetadata/channel/sessionKey),
`agent` (id globs, trustTier against the persistent agent tier, min/maxScore),
`risk` (min/maxRisk ordinals), `frequency` (count >= maxCount in window),
`any` (OR), `not` — recursive via evaluator map (`conditions/index.ts:20-47`).
"""

from __future__ import annotations

import re
from typing import Any, Callable, Dict, List, Optional, Sequence

from .util import tier_ordinal

RISK_ORDINAL = {"low": 0, "medium": 1, "high": 2, "critical": 3}


def glob_to_regex(pattern: str) -> "re.Pattern[str]":
    """`*` -> `.*`, `?` -> `.`, everything else escaped, fully anchored
    (util.ts:68-74)."""
    escaped = re.sub(r"[.+^${}()|\[\]\\]", lambda m: "\\" + m.group(0), pattern)
    escaped = escaped.replace("*", ".*").replace("?", ".")
    return re.compile(f"^{escaped}$")


def parse_time_to_minutes(text: str) -> int:
    parts = str(text).split(":")
    try:
        h = int(parts[0])
        m = int(parts[1]) if len(parts) > 1 else 0
    except (ValueError, IndexError):
        return -1
    if h < 0 or h > 23 or m < 0 or m > 59:
        return -1
    return h * 60 + m


def is_in_time_range(current: int, after: int, before: int) -> bool:
    """[after, before) with midnight wrap (util.ts:16-26)."""
    if after <= before:
        return after <= current < before
    return current >= after or current < before


class RegexCache:
    def __init__(self) -> None:
        self._cache: Dict[str, Optional["re.Pattern[str]"]] = {}

    def get(self, pattern: str) -> Optional["re.Pattern[str]"]:
        if pattern not in self._cache:
            try:
                self._cache[pattern] = re.compile(pattern)
            except re.error:
                self._cache[pattern] = None
        return self._cache[pattern]


def _as_list(v: Any) -> List[Any]:
    return list(v) if isinstance(v, (list, tuple)) else [v]


def _match_glob_any(patterns: Any, value: Optional[str]) -> bool:
    if not value:
        return False
    for p in _as_list(patterns):
        p = str(p)
        if "*" in p or "?" in p:
            if glob_to_regex(p).match(value):
                return True
        elif p == value:
            return True
    return False


def _match_param(matcher: Dict[str, Any], value: Any, cache: RegexCache) -> bool:
    if "equals" in matcher:
        return value == matcher["equals"]
    if "contains" in matcher:
        return isinstance(value, str) and matcher["contains"] in value
    if "matches" in matcher:
        if not isinstance(value, str):
            return False
        rx = cache.get(matcher["matches"])
        return bool(rx and rx.search(value))
    if "startsWith" in matcher:
        return isinstance(value, str) and value.startswith(matcher["startsWith"])
    if "in" in matcher:
        return value in matcher["in"]
    return False


def _matches_any_text(patterns: Any, texts: Sequence[str], cache: RegexCache) -> bool:
    for pattern in _as_list(patterns):
        pattern = str(pattern)
        rx = cache.get(pattern)
        if rx is not None:
            if any(rx.search(t) for t in texts):
                return True
        else:
            if any(pattern in t for t in texts):
                return True
    return False


# --- individual evaluators -------------------------------------------------

def eval_tool(cond: Dict[str, Any], ctx: Dict[str, Any], deps: "ConditionDeps") -> bool:
    if "name" in cond and not _match_glob_any(cond["name"], ctx.get("toolName")):
        return False
    params = cond.get("params")
    if params:
        tool_params = ctx.get("toolParams")
        if not tool_params:
            return False
        for key, matcher in params.items():
            if not _match_param(matcher, tool_params.get(key), deps.regex_cache):
                return False
    return True


def eval_time(cond: Dict[str, Any], ctx: Dict[str, Any], deps: "ConditionDeps") -> bool:
    t = ctx.get("time") or {}
    current = int(t.get("hour", 0)) * 60 + int(t.get("minute", 0))
    if cond.get("window"):
        win = deps.time_windows.get(cond["window"])
        if not win:
            return False
        start = parse_time_to_minutes(win.get("start", ""))
        end = parse_time_to_minutes(win.get("end", ""))
        if start < 0 or end < 0 or not is_in_time_range(current, start, end):
            return False
        days = win.get("days")
        if days and t.get("dayOfWeek") not in days:
            return False
        return True
    after, before = cond.get("after"), cond.get("before")
    if after is not None and before is not None:
        a, b = parse_time_to_minutes(after), parse_time_to_minutes(before)
        if a < 0 or b < 0 or not is_in_time_range(current, a, b):
            return False
    elif after is not None:
        a = parse_time_to_minutes(after)
        if a < 0 or current < a:
            return False
    elif before is not None:
        b = parse_time_to_minutes(before)
        if b < 0 or current >= b:
            return False
    days = cond.get("days")
    if days and t.get("dayOfWeek") not in days:
        return False
    return True


def eval_context(cond: Dict[str, Any], ctx: Dict[str, Any], deps: "ConditionDeps") -> bool:
    if "conversationContains" in cond:
        convo = ctx.get("conversationContext") or []
        if not convo or not _matches_any_text(cond["conversationContains"], convo, deps.regex_cache):
            return False
    if "messageContains" in cond:
        msg = ctx.get("messageContent")
        if not msg or not _matches_any_text(cond["messageContains"], [msg], deps.regex_cache):
            return False
    if "hasMetadata" in cond:
        meta = ctx.get("metadata") or {}
        if not all(k in meta for k in _as_list(cond["hasMetadata"])):
            return False
    if "channel" in cond:
        if not ctx.get("channel") or ctx["channel"] not in _as_list(cond["channel"]):
            return False
    if "sessionKey" in cond:
        sk = ctx.get("sessionKey")
        if not sk or not glob_to_regex(str(cond["sessionKey"])).match(sk):
            return False
    return True


def eval_agent(cond: Dict[str, Any], ctx: Dict[str, Any], deps: "ConditionDeps") -> bool:
    if "id" in cond and not _match_glob_any(cond["id"], ctx.get("agentId")):
        return False
    agent_trust = (ctx.get("trust") or {}).get("agent") or {}
    # trustTier checks the PERSISTENT agent tier, not the session tier
    # (conditions/simple.ts comment).
    if "trustTier" in cond and agent_trust.get("tier") not in _as_list(cond["trustTier"]):
        return False
    if "minScore" in cond and agent_trust.get("score", 0) < cond["minScore"]:
        return False
    if "maxScore" in cond and agent_trust.get("score", 0) > cond["maxScore"]:
        return False
    return True


def eval_risk(cond: Dict[str, Any], ctx: Dict[str, Any], deps: "ConditionDeps") -> bool:
    cur = RISK_ORDINAL.get(deps.risk.get("level", "low"), 0)
    if "minRisk" in cond and cur < RISK_ORDINAL.get(cond["minRisk"], 0):
        return False
    if "maxRisk" in cond and cur > RISK_ORDINAL.get(cond["maxRisk"], 3):
        return False
    return True


def eval_frequency(cond: Dict[str, Any], ctx: Dict[str, Any], deps: "ConditionDeps") -> bool:
    scope = cond.get("scope", "agent")
    count = deps.frequency_tracker.count(
        cond.get("windowSeconds", 60),
        scope,
        ctx.get("agentId", ""),
        ctx.get("sessionKey", ""),
        ctx.get("toolName", ""),
    )
    return count >= cond.get("maxCount", 0)


def eval_any(cond: Dict[str, Any], ctx: Dict[str, Any], deps: "ConditionDeps") -> bool:
    return any(evaluate_condition(c, ctx, deps) for c in cond.get("conditions", []))


def eval_not(cond: Dict[str, Any], ctx: Dict[str, Any], deps: "ConditionDeps") -> bool:
    inner = cond.get("condition")
    if inner is None:
        return False
    return not evaluate_condition(inner, ctx, deps)


EVALUATORS: Dict[str, Callable[[Dict[str, Any], Dict[str, Any], "ConditionDeps"], bool]] = {
    "tool": eval_tool,
    "time": eval_time,
    "context": eval_context,
    "agent": eval_agent,
    "risk": eval_risk,
    "frequency": eval_frequency,
    "any": eval_any,
    "not": eval_not,
}


class ConditionDeps:
    def __init__(
        self,
        regex_cache: Optional[RegexCache] = None,
        time_windows: Optional[Dict[str, Dict[str, Any]]] = None,
        risk: Optional[Dict[str, Any]] = None,
        frequency_tracker: Any = None,
    ):
        self.regex_cache = regex_cache or RegexCache()
        self.time_windows = time_windows or {}
        self.risk = risk or {"level": "low", "score": 0}

        class _NullFreq:
            def count(self, *a: Any, **k: Any) -> int:
                return 0

        self.frequency_tracker = frequency_tracker or _NullFreq()


def evaluate_condition(cond: Dict[str, Any], ctx: Dict[str, Any], deps: ConditionDeps) -> bool:
    fn = EVALUATORS.get(cond.get("type", ""))
    if fn is None:
        return False
    return fn(cond, ctx, deps)


def evaluate_conditions(conds: Sequence[Dict[str, Any]], ctx: Dict[str, Any], deps: ConditionDeps) -> bool:
    """AND of all conditions (conditions/index.ts:36-47)."""
    return all(evaluate_condition(c, ctx, deps) for c in conds)
