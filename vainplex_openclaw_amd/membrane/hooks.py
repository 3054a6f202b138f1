"""Membrane plugin: hook wiring.

Data-flow parity (suite `README.md:60-106`, brainplex README §Flow):
message -> Governance (gate) -> **Membrane (inject relevant memories)**
-> agent -> Membrane ingest. So:
  - message_received (priority 500, after governance's 1000 gate but
    before the agent): retrieve + inject `membrane_context`,
  - message_sent / message_received: ingest episodic memories,
  - session_end / gateway_stop: flush,
  - working memory: before_compaction snapshots current task state.
"""

from __future__ import annotations

import time
from typing import Any, Dict, Optional

from ..core.api import PluginApi, PluginLogger
from ..core.config import load_plugin_config
from .engine import DEFAULT_CONFIG, MembraneEngine

PLUGIN_ID = "openclaw-membrane"


class MembranePlugin:
    id = PLUGIN_ID
    name = "Membrane"
    description = "Episodic memory: salience-based recall with organic decay"
    version = "0.1.0"

    def __init__(self, workspace: Optional[str] = None, device: Optional[str] = None):
        self.workspace = workspace
        self.device = device
        self.engine: Optional[MembraneEngine] = None

    def register(self, api: PluginApi) -> None:
        cfg = {**DEFAULT_CONFIG, **load_plugin_config(PLUGIN_ID, fallback=api.plugin_config)}
        ws = self.workspace or cfg.get("workspace") or "."
        self.engine = MembraneEngine(ws, config=cfg, device=self.device)
        eng = self.engine

        def agent_of(ev: Dict[str, Any]) -> str:
            ctx = ev.get("ctx") or {}
            agent = ctx.get("agentId") or ev.get("agentId")
            if isinstance(agent, str) and agent:
                return agent
            sk = ctx.get("sessionKey") or ev.get("sessionKey") or ""
            return sk.split(":")[0] if sk else "main"

        def on_message_received(ev: Dict[str, Any]):
            text = ev.get("content") or ev.get("message") or ev.get("text")
            if not isinstance(text, str) or not text.strip():
                return None
            agent = agent_of(ev)
            results = eng.retrieve(agent, text)
            eng.remember(agent, text, kind="episodic")
            if results:
                return {"membrane_context": eng.format_context(results)}
            return None

        def on_message_sent(ev: Dict[str, Any]) -> None:
            text = ev.get("content") or ev.get("message") or ev.get("text")
            if isinstance(text, str) and text.strip():
                eng.remember(agent_of(ev), text, kind="episodic")

        def on_before_compaction(ev: Dict[str, Any]) -> None:
            snapshot = ev.get("summary") or ev.get("snapshot")
            if isinstance(snapshot, str) and snapshot.strip():
                eng.remember(agent_of(ev), snapshot, kind="working", salience=1.0)

        def on_stop(ev: Dict[str, Any]) -> None:
            eng.flush()

        api.on("message_received", on_message_received, priority=500)
        api.on("message_sent", on_message_sent, priority=100)
        api.on("before_compaction", on_before_compaction, priority=100)
        api.on("session_end", on_stop, priority=900)
        api.on("gateway_stop", on_stop, priority=900)

        api.register_command("membranestatus", lambda *a, **kw: {
            "text": f"**Membrane**\nIngested: {eng.stats['ingested']}\n"
                    f"Retrievals: {eng.stats['retrievals']} "
                    f"({eng.stats['retrieved']} memories injected)\n"
                    f"Index rows: {eng.index.size}",
        })
        api.register_gateway_method("membrane.stats", lambda *a, **kw: dict(eng.stats))


def create_plugin(workspace: Optional[str] = None, device: Optional[str] = None) -> MembranePlugin:
    return MembranePlugin(workspace, device)
