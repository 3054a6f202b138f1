"""Trace analyzer normalized events.

Parity target: cortex `src/trace-analyzer/events.ts` — NormalizedEvent
{id, ts, agent, session, type, payload, seq} (`:27-42`) with unified
payload fields (`:44-52`); the analyzer normalizes Schema A
(nats-eventstore ClawEvent envelopes) and Schema B (session-sync) into
this shape so detectors never see raw payloads.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any, Dict, Optional

EVENT_TYPES = (
    "msg.in", "msg.out", "tool.call", "tool.result",
    "session.start", "session.end", "run.start", "run.end", "run.error",
)

# Schema A (eventstore canonical + legacy types, eventstore/events.py)
# -> analyzer types
SCHEMA_A_MAP = {
    "message.in.received": "msg.in",
    "message.out.sending": "msg.out",
    "message.out.sent": "msg.out",
    "tool.call.requested": "tool.call",
    "tool.call.executed": "tool.result",
    "tool.call.failed": "tool.result",
    "session.started": "session.start",
    "session.ended": "session.end",
    "run.started": "run.start",
    "run.ended": "run.end",
    "run.failed": "run.error",
    # legacy aliases (the envelope's `type` field)
    "msg.in": "msg.in",
    "msg.out": "msg.out",
    "msg.sending": "msg.out",
    "tool.call": "tool.call",
    "tool.result": "tool.result",
    "session.start": "session.start",
    "session.end": "session.end",
    "run.start": "run.start",
    "run.end": "run.end",
    "run.error": "run.error",
}


@dataclass
class NormalizedEvent:
    id: str
    ts: float  # ms since epoch
    agent: str
    session: str
    type: str
    payload: Dict[str, Any] = field(default_factory=dict)
    seq: int = 0


def _norm_session(raw: Optional[str]) -> str:
    """"agent:main:uuid" -> "uuid" (events.ts session normalization)."""
    if not raw:
        return "unknown"
    parts = str(raw).split(":")
    return parts[-1] if len(parts) > 1 else str(raw)


def normalize_schema_a(ev: Dict[str, Any], seq: int = 0) -> Optional[NormalizedEvent]:
    """ClawEvent envelope (eventstore/envelope.py) -> NormalizedEvent."""
    etype = SCHEMA_A_MAP.get(
        str(ev.get("canonicalType") or ev.get("type", ""))
    ) or SCHEMA_A_MAP.get(str(ev.get("type", "")))
    if etype is None:
        return None
    data = ev.get("data") or ev.get("payload") or {}
    actor = ev.get("actor") or {}
    scope = ev.get("scope") or {}
    payload: Dict[str, Any] = {}
    if etype in ("msg.in", "msg.out"):
        payload["content"] = data.get("content") or data.get("text") or ""
        payload["role"] = "user" if etype == "msg.in" else "assistant"
        payload["from"] = data.get("from") or actor.get("id")
        payload["to"] = data.get("to")
        payload["channel"] = data.get("channel")
    elif etype == "tool.call":
        payload["toolName"] = data.get("toolName") or data.get("tool")
        payload["toolParams"] = data.get("params") or data.get("toolParams") or {}
    elif etype == "tool.result":
        payload["toolName"] = data.get("toolName") or data.get("tool")
        payload["toolResult"] = data.get("result")
        payload["toolError"] = data.get("error")
        payload["toolIsError"] = bool(data.get("error")) or bool(data.get("isError"))
    ts = ev.get("ts") or ev.get("timestamp") or 0
    return NormalizedEvent(
        id=str(ev.get("id", "")),
        ts=float(ts),
        agent=str(actor.get("id") or actor.get("agentId") or ev.get("agent") or "unknown"),
        session=_norm_session(scope.get("sessionKey") or ev.get("session")),
        type=etype,
        payload=payload,
        seq=seq,
    )


def normalize_schema_b(ev: Dict[str, Any], seq: int = 0) -> Optional[NormalizedEvent]:
    """session-sync shape: {kind, sessionId, agentId, ts, body}."""
    kind_map = {
        "user_message": "msg.in",
        "assistant_message": "msg.out",
        "tool_use": "tool.call",
        "tool_result": "tool.result",
        "session_start": "session.start",
        "session_end": "session.end",
    }
    etype = kind_map.get(str(ev.get("kind", "")))
    if etype is None:
        return None
    body = ev.get("body") or {}
    payload: Dict[str, Any] = {}
    if etype in ("msg.in", "msg.out"):
        payload["content"] = body.get("text", "")
        payload["role"] = "user" if etype == "msg.in" else "assistant"
    elif etype == "tool.call":
        payload["toolName"] = body.get("name")
        payload["toolParams"] = body.get("input") or {}
    elif etype == "tool.result":
        payload["toolName"] = body.get("name")
        payload["toolResult"] = body.get("output")
        payload["toolError"] = body.get("error")
        payload["toolIsError"] = bool(body.get("error"))
    return NormalizedEvent(
        id=str(ev.get("id", "")),
        ts=float(ev.get("ts", 0)),
        agent=str(ev.get("agentId", "unknown")),
        session=_norm_session(ev.get("sessionId")),
        type=etype,
        payload=payload,
        seq=seq,
    )


def normalize_event(ev, seq: int = 0) -> Optional[NormalizedEvent]:
    """Schema-sniffing normalizer (events.ts normalizeEvent): Schema B
    carries a `kind` field; everything else is tried as a Schema A
    ClawEvent envelope."""
    if not isinstance(ev, dict):
        return None
    if "kind" in ev:
        return normalize_schema_b(ev, seq)
    return normalize_schema_a(ev, seq)
