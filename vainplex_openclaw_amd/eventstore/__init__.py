"""EventStore: every hook -> ClawEvent envelope; audit/replay backbone.

MI355X-native rebuild of reference `packages/openclaw-nats-eventstore`
(SURVEY.md §2.4). NATS JetStream over TCP is replaced by an embedded
append-only journal with the same stream/subject/retention/replay
semantics (no network in this environment); the envelope schema,
deterministic event ids, hook->type mapping table and status surface are
byte-level parity.
"""

from .events import ALL_EVENT_TYPES, CANONICAL_EVENT_TYPES, LEGACY_EVENT_TYPES
from .hooks import EventPublisher, build_envelope, derive_event_id
from .journal import EventJournal
from .plugin import EventStorePlugin, create_plugin
from .util import build_subject, extract_agent_id

__all__ = [
    "ALL_EVENT_TYPES",
    "CANONICAL_EVENT_TYPES",
    "LEGACY_EVENT_TYPES",
    "EventPublisher",
    "build_envelope",
    "derive_event_id",
    "EventJournal",
    "EventStorePlugin",
    "create_plugin",
    "build_subject",
    "extract_agent_id",
]
