"""Membrane: episodic/semantic/working memory with salience-based recall
and organic decay (capability rebuild of the external Membrane plugin —
reference `README.md:17`, brainplex README §Membrane).

The recall hot path is the MFMA streaming top-k kernel
(`csrc/topk_recall.hip`) over an HBM-resident bf16 embedding matrix.
"""

from .engine import DEFAULT_CONFIG, MembraneEngine
from .hooks import MembranePlugin, create_plugin
from .index import SalienceIndex
from .store import MemoryRecord, MemoryStore

__all__ = [
    "DEFAULT_CONFIG",
    "MembraneEngine",
    "MembranePlugin",
    "create_plugin",
    "SalienceIndex",
    "MemoryRecord",
    "MemoryStore",
]
