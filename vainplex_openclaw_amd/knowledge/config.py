"""Knowledge-engine config defaults + resolution.

Parity target: reference `openclaw-knowledge-engine/src/config.ts` /
`config-loader.ts` — external `~/.openclaw/plugins/<id>/config.json`
first, inline pluginConfig fallback, inline `enabled` override, per-field
defaults (`types.ts KnowledgeConfig`).
"""

from __future__ import annotations

from typing import Any, Dict, Optional

from ..core.config import load_plugin_config, resolve_defaults

PLUGIN_ID = "openclaw-knowledge-engine"

DEFAULT_CONFIG: Dict[str, Any] = {
    "enabled": True,
    "workspace": "",
    "extraction": {
        "regex": {"enabled": True},
        "llm": {
            "enabled": False,
            "model": "mistral:7b",
            "endpoint": "http://localhost:11434/api/generate",
            "batchSize": 3,
            "cooldownMs": 5000,
        },
    },
    "decay": {"enabled": True, "intervalHours": 24, "rate": 0.05},
    "embeddings": {
        "enabled": False,
        "endpoint": "http://localhost:8000/api/v2/collections/{name}/upsert",
        "collectionName": "openclaw-facts",
        "syncIntervalMinutes": 30,
    },
    "storage": {"maxEntities": 5000, "maxFacts": 10000, "writeDebounceMs": 250},
}


def resolve_config(
    plugin_config: Optional[Dict[str, Any]] = None, home: Optional[str] = None,
    logger: Any = None, bootstrap: bool = False,
) -> Dict[str, Any]:
    """Full config-loader.ts semantics via the shared layered loader:
    legacy full inline config wins outright; a minimal {enabled?,
    configPath?} pointer loads (and optionally bootstraps) the external
    file; inline `enabled` overrides the file's."""
    from ..core.config import load_layered_config

    return load_layered_config(PLUGIN_ID, plugin_config, DEFAULT_CONFIG,
                               home=home, logger=logger, bootstrap=bootstrap)
