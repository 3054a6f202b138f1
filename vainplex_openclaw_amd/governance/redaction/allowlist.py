"""Redaction allowlist: exempt values/channels — credentials never.

Parity target: governance `src/redaction/allowlist.ts` — credentials are
NEVER allowlisted (code-level invariant); tool exemption (Layer 1), agent
exemption (Layer 2), channel-specific pii/financial allowlists.
"""

from __future__ import annotations

from typing import Any, Dict, List, Optional

DEFAULT_ALLOWLIST: Dict[str, List[str]] = {
    "exemptTools": [],
    "exemptAgents": [],
    "piiAllowedChannels": [],
    "financialAllowedChannels": [],
}


def normalize_allowlist(cfg: Optional[Dict[str, Any]]) -> Dict[str, List[str]]:
    out = {k: list(v) for k, v in DEFAULT_ALLOWLIST.items()}
    for k in out:
        if cfg and isinstance(cfg.get(k), list):
            out[k] = list(cfg[k])
    return out


def evaluate_allowlist(category: str, context: Dict[str, Any], allowlist: Dict[str, List[str]]) -> Dict[str, Any]:
    if category == "credential":
        return {"allowed": False, "reason": "Credentials are never allowlisted"}
    tool = context.get("toolName")
    if tool and tool in allowlist.get("exemptTools", []):
        return {"allowed": True, "reason": f'Tool "{tool}" is exempt from redaction'}
    agent = context.get("agentId")
    if agent and agent in allowlist.get("exemptAgents", []):
        return {"allowed": True, "reason": f'Agent "{agent}" is exempt from outbound redaction'}
    channel = context.get("channel")
    if category == "pii" and channel and channel in allowlist.get("piiAllowedChannels", []):
        return {"allowed": True, "reason": f'PII allowed on channel "{channel}"'}
    if category == "financial" and channel and channel in allowlist.get("financialAllowedChannels", []):
        return {"allowed": True, "reason": f'Financial data allowed on channel "{channel}"'}
    return {"allowed": False, "reason": "No allowlist match"}


def is_tool_exempt(tool_name: str, allowlist: Dict[str, List[str]]) -> bool:
    return tool_name in allowlist.get("exemptTools", [])


def is_agent_exempt(agent_id: str, allowlist: Dict[str, List[str]]) -> bool:
    return agent_id in allowlist.get("exemptAgents", [])


def get_redactable_categories(
    categories: List[str], context: Dict[str, Any], allowlist: Dict[str, List[str]]
) -> List[str]:
    return [c for c in categories if not evaluate_allowlist(c, context, allowlist)["allowed"]]
