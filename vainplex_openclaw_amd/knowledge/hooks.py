"""Knowledge-engine hook manager + plugin entry.

Parity target: reference `openclaw-knowledge-engine/src/hooks.ts` —
session_start loads the fact store and starts maintenance (priority 200),
message_received/message_sent run regex extraction + optional LLM batch
(priority 100), gateway_stop flushes and stops timers (priority 900);
`index.ts` register().
"""

from __future__ import annotations

import time
from typing import Any, Dict, List, Optional

from ..core.api import NullLogger, PluginApi, PluginLogger
from .config import PLUGIN_ID, resolve_config
from .embeddings import Embeddings
from .entity_extractor import Entity, EntityExtractor, merge_entities
from .fact_store import FactStore
from .llm_enhancer import LlmEnhancer
from .maintenance import Maintenance


class HookManager:
    def __init__(
        self,
        config: Dict[str, Any],
        workspace: str,
        logger: Optional[PluginLogger] = None,
        call_llm=None,
        http_post=None,
        clock=time.time,
    ):
        self.config = config
        self.workspace = workspace
        self._log = logger or NullLogger()
        self._clock = clock
        self.extractor = EntityExtractor(self._log, clock=clock)
        st = config.get("storage", {})
        self.fact_store = FactStore(
            workspace,
            max_facts=st.get("maxFacts", 10000),
            write_debounce_ms=st.get("writeDebounceMs", 250),
            logger=self._log,
            clock=clock,
        )
        self.entities: Dict[str, Entity] = {}  # in-memory entity graph
        llm_cfg = config.get("extraction", {}).get("llm", {})
        self.llm: Optional[LlmEnhancer] = None
        if llm_cfg.get("enabled") and call_llm is not None:
            self.llm = LlmEnhancer(
                call_llm,
                batch_size=llm_cfg.get("batchSize", 3),
                cooldown_ms=llm_cfg.get("cooldownMs", 5000),
                logger=self._log,
                clock=clock,
            )
            self.llm.set_result_handler(self._apply_llm_result)
        emb_cfg = config.get("embeddings", {})
        self.embeddings = Embeddings(
            emb_cfg.get("endpoint", ""),
            emb_cfg.get("collectionName", "openclaw-facts"),
            enabled=bool(emb_cfg.get("enabled")),
            http_post=http_post,
            logger=self._log,
        )
        self.maintenance: Optional[Maintenance] = None

    # -- hook handlers -----------------------------------------------------
    def on_session_start(self, ev: Dict[str, Any]) -> None:
        self.fact_store.load()
        self.maintenance = Maintenance(
            self.config, self.fact_store, self.embeddings, logger=self._log
        )
        self.maintenance.start()

    def on_message(self, ev: Dict[str, Any]) -> None:
        text = ev.get("content") or ev.get("message") or ev.get("text")
        if not isinstance(text, str) or not text.strip():
            return
        if self.config.get("extraction", {}).get("regex", {}).get("enabled", True):
            found = self.extractor.extract(text)
            if found:
                self._merge_into_graph(found)
        if self.llm is not None:
            result = self.llm.add_to_batch(f"msg-{int(self._clock() * 1000)}", text)
            if result is not None:
                self._apply_llm_result(result)

    def on_shutdown(self, ev: Dict[str, Any]) -> None:
        if self.maintenance is not None:
            self.maintenance.stop()
        if self.llm is not None:
            self.llm.clear_timers()
        self.fact_store.flush()

    # -- internals ---------------------------------------------------------
    def _merge_into_graph(self, found: List[Entity]) -> None:
        merged = merge_entities(list(self.entities.values()), found, clock=self._clock)
        max_entities = self.config.get("storage", {}).get("maxEntities", 5000)
        if len(merged) > max_entities:
            merged = sorted(merged, key=lambda e: -e.importance)[:max_entities]
        self.entities = {e.id: e for e in merged}

    def _apply_llm_result(self, result: Dict) -> None:
        if not self.fact_store.is_loaded:
            self.fact_store.load()
        for f in result.get("facts", ()):
            self.fact_store.add_fact(
                f["subject"], f["predicate"], f["object"], source="extracted-llm"
            )

    def register(self, api: PluginApi) -> None:
        if not self.config.get("enabled", True):
            return
        api.on("session_start", self.on_session_start, priority=200)
        api.on("message_received", self.on_message, priority=100)
        api.on("message_sent", self.on_message, priority=100)
        api.on("gateway_stop", self.on_shutdown, priority=900)


class KnowledgePlugin:
    id = PLUGIN_ID
    name = "Knowledge Engine"
    description = "Entity + SPO-fact extraction, fact store, embedding sync"
    version = "0.1.0"

    def __init__(self, workspace: Optional[str] = None, call_llm=None, http_post=None):
        self.workspace = workspace
        self.call_llm = call_llm
        self.http_post = http_post
        self.hooks: Optional[HookManager] = None

    def register(self, api: PluginApi) -> None:
        cfg = resolve_config(api.plugin_config)
        ws = self.workspace or cfg.get("workspace") or "."
        self.hooks = HookManager(
            cfg, ws, logger=api.logger, call_llm=self.call_llm, http_post=self.http_post
        )
        self.hooks.register(api)


def create_plugin(workspace: Optional[str] = None, **kw) -> KnowledgePlugin:
    return KnowledgePlugin(workspace, **kw)
