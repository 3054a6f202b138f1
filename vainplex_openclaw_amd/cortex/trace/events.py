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
        err = _extract_error_from_result(data)
        payload["toolError"] = err.get("error") if err["isError"] else data.get("error")
        payload["toolIsError"] = err["isError"] or bool(data.get("isError"))
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


# session-sync "conversation.*" types (events.ts:90-93)
CONVERSATION_MAP = {
    "conversation.message.in": "msg.in",
    "conversation.message.out": "msg.out",
    "conversation.tool_call": "tool.call",
    "conversation.tool_result": "tool.result",
}


def detect_schema(ev: Dict[str, Any]) -> Optional[str]:
    """Schema sniffing (events.ts detectSchema): "B" = session-sync
    (conversation.* type, meta.source == "session-sync", or a `timestamp`
    field instead of `ts`), "A" = eventstore envelope, None = unknown.
    This build's kind-based session-sync shape also reads as "B"."""
    if not isinstance(ev, dict):
        return None
    if "kind" in ev:
        return "B"
    etype = ev.get("type")
    if not isinstance(etype, str):
        if ev.get("canonicalType"):
            return "A"
        return None
    if etype.startswith("conversation."):
        return "B"
    meta = ev.get("meta")
    if isinstance(meta, dict) and meta.get("source") == "session-sync":
        return "B"
    known = etype in SCHEMA_A_MAP
    if isinstance(ev.get("ts"), (int, float)) and known:
        return "A"
    if isinstance(ev.get("timestamp"), (int, float)):
        return "B"
    return "A" if known else None


def _extract_error_from_result(payload: Dict[str, Any]) -> Dict[str, Any]:
    """Nested tool-result error extraction (events.ts:218-247):
    top-level error > result.details.{error,status,exitCode} >
    result.isError with content[0].text."""
    top = payload.get("error")
    if isinstance(top, str) and top:
        return {"error": top, "isError": True}
    result = payload.get("result")
    if isinstance(result, dict):
        details = result.get("details")
        if isinstance(details, dict):
            derr = details.get("error")
            if isinstance(derr, str) and derr:
                return {"error": derr, "isError": True}
            if details.get("status") == "error":
                return {"error": "status: error", "isError": True}
            code = details.get("exitCode")
            if isinstance(code, (int, float)) and code > 0:
                return {"error": f"exit code {int(code)}", "isError": True}
        if result.get("isError") is True:
            text = None
            content = result.get("content")
            if isinstance(content, list) and content and isinstance(content[0], dict):
                t = content[0].get("text")
                if isinstance(t, str):
                    text = t[:500]
            if text is None and isinstance(result.get("result"), str):
                text = result["result"][:500]
            return {"error": text or "unknown error", "isError": True}
    return {"isError": False}


def normalize_schema_b_conv(ev: Dict[str, Any], seq: int = 0) -> Optional[NormalizedEvent]:
    """session-sync conversation.* shape (events.ts Schema B paths):
    content from payload.text_preview[0].text, tool call/result under
    payload.data {name, args, result, isError}."""
    etype = CONVERSATION_MAP.get(str(ev.get("type", "")))
    if etype is None:
        return None
    ts = ev.get("ts") if isinstance(ev.get("ts"), (int, float)) else ev.get("timestamp")
    if not isinstance(ts, (int, float)) or ts == 0:
        return None
    raw_payload = ev.get("payload") if isinstance(ev.get("payload"), dict) else {}
    payload: Dict[str, Any] = {}
    if etype in ("msg.in", "msg.out"):
        payload["role"] = "user" if etype == "msg.in" else "assistant"
        tp = raw_payload.get("text_preview")
        if isinstance(tp, list) and tp and isinstance(tp[0], dict) \
                and isinstance(tp[0].get("text"), str):
            payload["content"] = tp[0]["text"]
        else:
            payload["content"] = ""
        if isinstance(raw_payload.get("sessionId"), str):
            payload["sessionId"] = raw_payload["sessionId"]
    elif etype == "tool.call":
        data = raw_payload.get("data") if isinstance(raw_payload.get("data"), dict) else {}
        payload["toolName"] = data.get("name") if isinstance(data.get("name"), str) else None
        payload["toolParams"] = data.get("args") if isinstance(data.get("args"), dict) else {}
    elif etype == "tool.result":
        data = raw_payload.get("data") if isinstance(raw_payload.get("data"), dict) else {}
        payload["toolName"] = data.get("name") if isinstance(data.get("name"), str) else None
        payload["toolResult"] = data.get("result")
        is_err = data.get("isError") is True
        payload["toolIsError"] = is_err
        payload["toolError"] = data.get("result") \
            if is_err and isinstance(data.get("result"), str) else None
    session_raw = ev.get("session") if isinstance(ev.get("session"), str) else "unknown"
    return NormalizedEvent(
        id=str(ev.get("id", "")),
        ts=float(ts),
        agent=str(ev.get("agent") or "unknown"),
        session=_norm_session(session_raw) if session_raw != "unknown" else "unknown",
        type=etype,
        payload=payload,
        seq=seq,
    )


def normalize_event(ev, seq: int = 0) -> Optional[NormalizedEvent]:
    """Schema-sniffing normalizer (events.ts normalizeEvent): routes by
    detect_schema — the kind-based and conversation.* session-sync
    shapes to the Schema B normalizers, everything else to Schema A."""
    schema = detect_schema(ev)
    if schema is None:
        return None
    if schema == "B":
        if "kind" in ev:
            return normalize_schema_b(ev, seq)
        return normalize_schema_b_conv(ev, seq)
    return normalize_schema_a(ev, seq)
