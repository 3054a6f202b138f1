"""Envelope construction + hook registration for the event store.

Parity target: reference `openclaw-nats-eventstore/src/hooks.ts` —
deterministic event id `evt-` + sha256(session:type:stableSourceId)[:16]
else uuid4 (`:67-98`); actor/scope/trace blocks (`:100-129`); full
ClawEvent envelope schemaVersion 1 (`:131-159`); fire-and-forget publish
(`:161-181`); include/exclude hook filter (`:42-50`); system events use
agent/session "system".
"""

from __future__ import annotations

import hashlib
import time
import uuid
from typing import Any, Dict, List, Optional

from .events import SCHEMA_VERSION
from .mappings import EXTRA_EMITTERS, HOOK_MAPPINGS, HookMapping
from .util import build_subject, extract_agent_id


def _first_string(*values: Any) -> Optional[str]:
    for v in values:
        if isinstance(v, str) and v:
            return v
    return None


def derive_event_id(canonical_type: str, session: str, payload: Dict, ctx: Dict) -> str:
    """Deterministic id from the first stable source id (hooks.ts:67-98)."""
    oe = ctx.get("originalEvent") or {}
    stable = _first_string(
        ctx.get("runId"), payload.get("runId"), oe.get("runId"),
        ctx.get("messageId"), payload.get("messageId"), oe.get("messageId"),
        payload.get("toolCallId"), oe.get("toolCallId"),
        ctx.get("jobId"), payload.get("jobId"), oe.get("jobId"),
        oe.get("id"),
    )
    if stable:
        h = hashlib.sha256(f"{session}:{canonical_type}:{stable}".encode()).hexdigest()[:16]
        return f"evt-{h}"
    return str(uuid.uuid4())


def build_actor(agent: str, ctx: Dict) -> Dict:
    return {
        "agentId": None if agent == "system" else agent,
        "userId": _first_string(ctx.get("senderId")),
        "channel": _first_string(ctx.get("channelId")),
    }


def build_scope(payload: Dict, ctx: Dict) -> Dict:
    oe = ctx.get("originalEvent") or {}
    return {
        "sessionKey": _first_string(ctx.get("sessionKey"), oe.get("sessionKey")),
        "sessionId": _first_string(ctx.get("sessionId"), oe.get("sessionId")),
        "runId": _first_string(ctx.get("runId"), payload.get("runId"), oe.get("runId")),
        "toolCallId": _first_string(payload.get("toolCallId"), oe.get("toolCallId")),
        "messageId": _first_string(ctx.get("messageId"), payload.get("messageId"), oe.get("messageId")),
        "jobId": _first_string(ctx.get("jobId"), payload.get("jobId"), oe.get("jobId")),
    }


def build_trace(payload: Dict, ctx: Dict) -> Dict:
    oe = ctx.get("originalEvent") or {}
    trace = ctx.get("trace") or {}
    return {
        "traceId": _first_string(ctx.get("traceId"), trace.get("traceId"), oe.get("traceId")),
        "spanId": _first_string(ctx.get("spanId"), trace.get("spanId"), oe.get("spanId")),
        "parentSpanId": _first_string(
            ctx.get("parentSpanId"), trace.get("parentSpanId"), oe.get("parentSpanId")
        ),
        "causationId": _first_string(payload.get("causationId"), oe.get("causationId")),
        "correlationId": _first_string(
            ctx.get("runId"), ctx.get("sessionId"), ctx.get("sessionKey"),
            oe.get("runId"), oe.get("sessionId"), oe.get("sessionKey"),
        ),
    }


def build_envelope(
    canonical_type: str,
    agent: str,
    session: str,
    payload: Dict,
    legacy_type: Optional[str] = None,
    visibility: Optional[str] = None,
    redaction: Optional[Dict] = None,
    ctx: Optional[Dict] = None,
    original_event: Optional[Dict] = None,
    clock=time.time,
) -> Dict:
    c = dict(ctx or {})
    c["originalEvent"] = original_event
    env = {
        "id": derive_event_id(canonical_type, session, payload, c),
        "ts": int(clock() * 1000),
        "agent": agent,
        "session": session,
        "type": legacy_type or canonical_type,
        "canonicalType": canonical_type,
        "legacyType": legacy_type,
        "schemaVersion": SCHEMA_VERSION,
        "source": {"plugin": "nats-eventstore"},
        "actor": build_actor(agent, c),
        "scope": build_scope(payload, c),
        "trace": build_trace(payload, c),
        "visibility": visibility or "internal",
        "payload": payload,
    }
    if redaction is not None:
        env["redaction"] = redaction
    return env


def should_publish(hook_name: str, include: List[str], exclude: List[str]) -> bool:
    if include:
        return hook_name in include
    if exclude:
        return hook_name not in exclude
    return True


class EventPublisher:
    """Wires the mapping table to a journal/client (hooks.ts registerEventHooks)."""

    def __init__(self, journal, config: Dict, logger=None, clock=time.time):
        self.journal = journal
        self.config = config
        self._log = logger
        self._clock = clock

    def _publish(self, mapping: HookMapping, event: Dict, ctx: Dict) -> None:
        payload = mapping.mapper(event, ctx)
        etype = mapping.resolve_type(event, ctx)
        if mapping.system_event:
            agent = session = "system"
        else:
            agent = extract_agent_id(ctx)
            session = ctx.get("sessionKey") or ctx.get("sessionId") or "unknown"
        env = build_envelope(
            etype, agent, session, payload,
            legacy_type=mapping.legacy_type,
            visibility=mapping.visibility,
            redaction=mapping.redaction,
            ctx=ctx, original_event=event, clock=self._clock,
        )
        subject = build_subject(self.config.get("subjectPrefix", "openclaw.events"), agent, env["type"])
        try:
            self.journal.publish(subject, env)
        except Exception as exc:  # fire-and-forget: never break the hook
            if self._log is not None:
                self._log.warn("[nats-eventstore] Publish %s failed: %s", etype, exc)

    def handle(self, mapping: HookMapping, extras: List, ev: Dict) -> None:
        ctx = ev.get("ctx") or {k: v for k, v in ev.items() if k != "hook"}
        try:
            self._publish(mapping, ev, ctx)
            for extra in extras:
                if extra.condition(ev):
                    self._publish(extra, ev, ctx)
        except Exception as exc:
            if self._log is not None:
                self._log.warn("[nats-eventstore] Hook %s error: %s", mapping.hook_name, exc)

    def register(self, api) -> None:
        include = self.config.get("includeHooks", [])
        exclude = self.config.get("excludeHooks", [])
        extras_by_hook: Dict[str, List] = {}
        for extra in EXTRA_EMITTERS:
            extras_by_hook.setdefault(extra.hook_name, []).append(extra)
        for mapping in HOOK_MAPPINGS:
            if not should_publish(mapping.hook_name, include, exclude):
                continue
            extras = extras_by_hook.get(mapping.hook_name, [])

            def handler(ev: Dict, _m=mapping, _x=extras) -> None:
                self.handle(_m, _x, ev)

            api.on(mapping.hook_name, handler, priority=10)
