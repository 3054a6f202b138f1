"""Shared governance helpers (tier mapping, clamps, session keys).

Parity: governance `src/util.ts:192-198` (scoreToTier) and the
`:subagent:` session-key convention used by cross-agent tracking.
"""

from __future__ import annotations

from typing import Optional

TIERS = ("untrusted", "restricted", "standard", "trusted", "elevated")


def clamp(v: float, lo: float, hi: float) -> float:
    return lo if v < lo else hi if v > hi else v


def score_to_tier(score: float) -> str:
    """>=80 elevated, >=60 trusted, >=40 standard, >=20 restricted, else
    untrusted (util.ts:192-198)."""
    if score >= 80:
        return "elevated"
    if score >= 60:
        return "trusted"
    if score >= 40:
        return "standard"
    if score >= 20:
        return "restricted"
    return "untrusted"


def tier_ordinal(tier: str) -> int:
    try:
        return TIERS.index(tier)
    except ValueError:
        return 0


def parent_session_key(session_key: str) -> Optional[str]:
    """Strip a ':subagent:' suffix to find the parent session (util.ts)."""
    idx = session_key.find(":subagent:")
    if idx == -1:
        return None
    return session_key[:idx]


def is_sub_agent(session_key: Optional[str]) -> bool:
    return bool(session_key) and ":subagent:" in str(session_key)


def _parse_agent_from_session_key(key: str) -> Optional[str]:
    """"agent:NAME" -> NAME; "agent:NAME:subagent:CHILD:..." -> CHILD
    (util.ts parseAgentFromSessionKey)."""
    parts = key.split(":")
    if len(parts) >= 2 and parts[0] == "agent":
        if len(parts) >= 4 and parts[2] == "subagent":
            return parts[3] or None
        return parts[1] or None
    return None


def extract_agent_id(session_key: Optional[str] = None, agent_id: Optional[str] = None) -> str:
    """Legacy extraction: "agent:main:subagent:forge:abc" -> "forge";
    "agent:main" -> "main"; fallback "unknown" (util.ts:90-106)."""
    if agent_id:
        return agent_id
    if not session_key:
        return "unknown"
    parts = session_key.split(":")
    if len(parts) >= 4 and parts[2] == "subagent":
        return parts[3] or "unknown"
    return parts[1] if len(parts) > 1 and parts[1] else "unknown"


def resolve_agent_id(hook_ctx: dict, event: Optional[dict] = None) -> str:
    """Multi-source fallback; "unresolved" when all fail (util.ts resolveAgentId)."""
    if hook_ctx.get("agentId"):
        return str(hook_ctx["agentId"])
    for key in ("sessionKey", "sessionId"):
        v = hook_ctx.get(key)
        if v:
            parsed = _parse_agent_from_session_key(str(v))
            if parsed:
                return parsed
    meta = (event or {}).get("metadata") or {}
    if isinstance(meta.get("agentId"), str):
        return meta["agentId"]
    return "unresolved"
