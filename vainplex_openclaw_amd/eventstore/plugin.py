"""EventStore plugin entry.

Parity target: reference `openclaw-nats-eventstore/index.ts` — register
service (connect/drain lifecycle), hooks, `/eventstatus` command and
`eventstore.status` gateway method.
"""

from __future__ import annotations

from typing import Any, Dict, Optional

from ..core.api import PluginApi
from ..core.config import load_raw_layered
from .config import PLUGIN_ID, resolve_config
from .hooks import EventPublisher
from .journal import EventJournal


class EventStorePlugin:
    id = PLUGIN_ID
    name = "NATS Event Store"
    description = "Publish agent events for audit, replay, and multi-agent sharing"
    version = "0.1.0"

    def __init__(self, journal: Optional[EventJournal] = None, journal_dir: Optional[str] = None,
                 nats_transport: Any = None):
        self.journal = journal
        self.journal_dir = journal_dir
        self.nats_transport = nats_transport  # injectable (tests); None -> TCP
        self.publisher: Optional[EventPublisher] = None
        self.config: Dict[str, Any] = {}

    def register(self, api: PluginApi) -> None:
        cfg = resolve_config(load_raw_layered(PLUGIN_ID, api.plugin_config))
        self.config = cfg
        if not cfg["enabled"]:
            api.logger.info("[nats-eventstore] Disabled via config")
            return
        if self.journal is None and (cfg.get("useNats") or self.nats_transport is not None):
            # live NATS backend over the wire-protocol client; connect +
            # ensure-stream can fail without egress -> fall back to the
            # embedded journal (fire-and-forget philosophy)
            from .nats_client import JetStreamClient, NatsPublishAdapter

            try:
                client = JetStreamClient(cfg, logger=api.logger,
                                         transport=self.nats_transport)
                client.connect()
                self.journal = NatsPublishAdapter(client)
            except Exception as exc:
                api.logger.warn("[nats-eventstore] NATS connect failed (%s); "
                                "using embedded journal", exc)
        if self.journal is None:
            ret = cfg["retention"]
            self.journal = EventJournal(
                directory=self.journal_dir or cfg.get("journalDir"),
                stream=cfg["streamName"],
                subject_prefix=cfg["subjectPrefix"],
                max_messages=ret["maxMessages"],
                max_bytes=ret["maxBytes"],
                max_age_hours=ret["maxAgeHours"],
            )
        self.publisher = EventPublisher(self.journal, cfg, logger=api.logger)
        self.publisher.register(api)

        api.register_service(
            {
                "id": self.id,
                "start": lambda *a: None,
                "stop": lambda *a: self.journal.close() if self.journal is not None else None,
            }
        )

        def eventstatus(*a, **kw) -> Dict[str, Any]:
            status = self.journal.status() if self.journal is not None else {
                "connected": False, "stream": None, "disconnectCount": 0, "publishFailures": 0,
            }
            return {
                "text": "\n".join(
                    [
                        "**NATS Event Store**",
                        f"Connected: {'yes' if status['connected'] else 'no'}",
                        f"Stream: {status.get('stream') or 'n/a'}",
                        f"Disconnects: {status['disconnectCount']}",
                        f"Publish failures: {status['publishFailures']}",
                    ]
                )
            }

        api.register_command("eventstatus", eventstatus)
        api.register_gateway_method(
            "eventstore.status",
            lambda *a, **kw: self.journal.status() if self.journal is not None else {"connected": False},
        )


def create_plugin(**kw) -> EventStorePlugin:
    return EventStorePlugin(**kw)
