"""Risk assessment: 5 weighted factors -> 0-100 score -> level.

Parity target: governance `src/risk-assessor.ts:10-99`:
- tool_sensitivity (weight 30): static tool-risk table, default 30,
  value = toolRisk/100*30
- time_of_day (15): off-hours = hour < 8 or hour >= 23
- trust_deficit (20): (100 - sessionScore)/100*20
- frequency (15): min(count60s/20, 1)*15
- target_scope (20): external target (messageTo set, host != "sandbox",
  or elevated param) -> 20
Level thresholds (`scoreToRiskLevel`): <=25 low, <=50 medium, <=75 high,
else critical. Score is rounded.
"""

from __future__ import annotations

from typing import Any, Dict, List, Optional

from .frequency import FrequencyTracker
from .util import clamp

DEFAULT_TOOL_RISK: Dict[str, int] = {
    "gateway": 95,
    "cron": 90,
    "elevated": 95,
    "exec": 70,
    "write": 65,
    "edit": 60,
    "sessions_spawn": 45,
    "sessions_send": 50,
    "browser": 40,
    "message": 40,
    "read": 10,
    "memory_search": 5,
    "memory_get": 5,
    "web_search": 15,
    "web_fetch": 20,
    "image": 10,
    "canvas": 15,
}


def score_to_risk_level(score: float) -> str:
    if score <= 25:
        return "low"
    if score <= 50:
        return "medium"
    if score <= 75:
        return "high"
    return "critical"


def is_external_target(ctx: Dict[str, Any]) -> bool:
    if ctx.get("messageTo"):
        return True
    params = ctx.get("toolParams")
    if not params:
        return False
    host = params.get("host")
    if isinstance(host, str) and host != "sandbox":
        return True
    return params.get("elevated") is True


class RiskAssessor:
    def __init__(self, tool_risk_overrides: Optional[Dict[str, int]] = None):
        self.overrides = dict(tool_risk_overrides or {})

    def tool_risk(self, tool_name: Optional[str]) -> int:
        if not tool_name:
            return 30
        if tool_name in self.overrides:
            return self.overrides[tool_name]
        return DEFAULT_TOOL_RISK.get(tool_name, 30)

    def assess(self, ctx: Dict[str, Any], freq: FrequencyTracker) -> Dict[str, Any]:
        factors = self._factors(ctx, freq)
        total = clamp(sum(f["value"] for f in factors), 0, 100)
        return {
            "level": score_to_risk_level(total),
            "score": round(total),
            "factors": factors,
        }

    def _factors(self, ctx: Dict[str, Any], freq: FrequencyTracker) -> List[Dict[str, Any]]:
        tool_name = ctx.get("toolName")
        tool_raw = self.tool_risk(tool_name)
        hour = int((ctx.get("time") or {}).get("hour", 12))
        is_off = hour < 8 or hour >= 23
        session_score = float(((ctx.get("trust") or {}).get("session") or {}).get("score", 0))
        recent = freq.count(
            60, "agent", ctx.get("agentId", ""), ctx.get("sessionKey", "")
        )
        external = is_external_target(ctx)
        return [
            {
                "name": "tool_sensitivity",
                "weight": 30,
                "value": (tool_raw / 100) * 30,
                "description": f"Tool {tool_name or 'unknown'} risk={tool_raw}",
            },
            {
                "name": "time_of_day",
                "weight": 15,
                "value": 15 if is_off else 0,
                "description": "Off-hours operation" if is_off else "Business hours",
            },
            {
                "name": "trust_deficit",
                "weight": 20,
                "value": ((100 - session_score) / 100) * 20,
                "description": f"Trust score {session_score}/100",
            },
            {
                "name": "frequency",
                "weight": 15,
                "value": min(recent / 20, 1) * 15,
                "description": f"{recent} actions in last 60s",
            },
            {
                "name": "target_scope",
                "weight": 20,
                "value": 20 if external else 0,
                "description": "External target" if external else "Internal target",
            },
        ]
