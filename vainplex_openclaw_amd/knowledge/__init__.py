"""Knowledge Engine: entity + SPO-fact extraction, fact store, embeddings.

MI355X-native rebuild of reference `packages/openclaw-knowledge-engine`
(SURVEY.md §2.3). The per-message host path lives here; the batched GPU
path (ENTITY DFA family + 4-gram encoder) lives in `ops/` + `csrc/`.
"""

from .entity_extractor import Entity, EntityExtractor, merge_entities
from .fact_store import FactStore
from .hooks import HookManager, KnowledgePlugin, create_plugin
from .llm_enhancer import LlmEnhancer
from .maintenance import Maintenance

__all__ = [
    "Entity",
    "EntityExtractor",
    "merge_entities",
    "FactStore",
    "HookManager",
    "KnowledgePlugin",
    "create_plugin",
    "LlmEnhancer",
    "Maintenance",
]
