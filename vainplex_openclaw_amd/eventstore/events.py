"""Canonical + legacy event type taxonomy (schema v1).

Parity target: reference `openclaw-nats-eventstore/src/events.ts` — 18
canonical nervous-system types, 16 legacy aliases, visibility tiers
public/internal/confidential/secret, ClawEvent envelope shape
{id, ts, agent, session, type, canonicalType, legacyType, schemaVersion,
source, actor, scope, trace, visibility, redaction?, payload}.
"""

from __future__ import annotations

CANONICAL_EVENT_TYPES = (
    "message.in.received",
    "message.out.sending",
    "message.out.sent",
    "tool.call.requested",
    "tool.call.executed",
    "tool.call.failed",
    "run.started",
    "run.ended",
    "run.failed",
    "model.input.observed",
    "model.output.observed",
    "session.started",
    "session.ended",
    "session.compaction.started",
    "session.compaction.ended",
    "session.reset",
    "gateway.started",
    "gateway.stopped",
)

LEGACY_EVENT_TYPES = (
    "msg.in",
    "msg.out",
    "msg.sending",
    "tool.call",
    "tool.result",
    "run.start",
    "run.end",
    "run.error",
    "llm.input",
    "llm.output",
    "session.start",
    "session.end",
    "session.compaction_start",
    "session.compaction_end",
    "gateway.start",
    "gateway.stop",
)

ALL_EVENT_TYPES = CANONICAL_EVENT_TYPES + LEGACY_EVENT_TYPES

VISIBILITIES = ("public", "internal", "confidential", "secret")

SCHEMA_VERSION = 1
