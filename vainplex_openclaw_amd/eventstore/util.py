"""Subject scheme + agent-id extraction.

Parity target: reference `openclaw-nats-eventstore/src/util.ts:1-24` —
agent priority ctx.agentId -> sessionKey first segment -> "main";
subject `<prefix>.<agent>.<type with dots -> underscores>`.
"""

from __future__ import annotations

from typing import Dict, Optional


def extract_agent_id(ctx: Dict) -> str:
    agent_id = ctx.get("agentId")
    if isinstance(agent_id, str) and agent_id and agent_id != "main":
        return agent_id
    session_key = ctx.get("sessionKey")
    if isinstance(session_key, str) and session_key:
        if session_key == "main":
            return "main"
        return session_key.split(":")[0] or "main"
    return "main"


def build_subject(prefix: str, agent: str, event_type: str) -> str:
    """`openclaw.events.main.msg_in` (util.ts:22-24)."""
    return f"{prefix}.{agent}.{event_type.replace('.', '_')}"
