"""EventStore config defaults + per-field resolution.

Parity target: reference `openclaw-nats-eventstore/src/config.ts:18-59` —
defaults nats://localhost:4222, stream `openclaw-events`, prefix
`openclaw.events`, unlimited retention (-1/-1/0), include/exclude hook
filters; each field resolved individually with type checks.

MI355X addition: `journalDir` — when set (or when no NATS server is
reachable, always true here: no network) events go to the embedded
journal (journal.py) instead of a remote JetStream.
"""

from __future__ import annotations

from typing import Any, Dict, Optional

PLUGIN_ID = "nats-eventstore"

DEFAULTS: Dict[str, Any] = {
    "enabled": True,
    "natsUrl": "nats://localhost:4222",
    "streamName": "openclaw-events",
    "subjectPrefix": "openclaw.events",
    "retention": {"maxMessages": -1, "maxBytes": -1, "maxAgeHours": 0},
    "publishTimeoutMs": 5000,
    "connectTimeoutMs": 5000,
    "drainTimeoutMs": 5000,
    "includeHooks": [],
    "excludeHooks": [],
    "journalDir": None,
}


def resolve_config(plugin_config: Optional[Dict[str, Any]] = None) -> Dict[str, Any]:
    raw = plugin_config or {}
    ret = raw.get("retention") if isinstance(raw.get("retention"), dict) else {}
    d = DEFAULTS

    def pick(key: str, typ) -> Any:
        v = raw.get(key)
        return v if isinstance(v, typ) else d[key]

    return {
        "enabled": pick("enabled", bool),
        "natsUrl": pick("natsUrl", str),
        "streamName": pick("streamName", str),
        "subjectPrefix": pick("subjectPrefix", str),
        "retention": {
            "maxMessages": int(ret.get("maxMessages", d["retention"]["maxMessages"])),
            "maxBytes": int(ret.get("maxBytes", d["retention"]["maxBytes"])),
            "maxAgeHours": float(ret.get("maxAgeHours", d["retention"]["maxAgeHours"])),
        },
        "publishTimeoutMs": pick("publishTimeoutMs", (int, float)),
        "connectTimeoutMs": pick("connectTimeoutMs", (int, float)),
        "drainTimeoutMs": pick("drainTimeoutMs", (int, float)),
        "includeHooks": pick("includeHooks", list),
        "excludeHooks": pick("excludeHooks", list),
        "journalDir": raw.get("journalDir") if isinstance(raw.get("journalDir"), str) else None,
    }
