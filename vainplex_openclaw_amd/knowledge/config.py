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
    plugin_config: Optional[Dict[str, Any]] = None, home: Optional[str] = None
) -> Dict[str, Any]:
    inline = dict(plugin_config or {})
    raw = load_plugin_config(PLUGIN_ID, fallback=inline, home=home)
    cfg = resolve_defaults(raw, DEFAULT_CONFIG)
    # inline `enabled` always wins (config-loader.ts applyInlineOverrides)
    if isinstance(inline.get("enabled"), bool):
        cfg["enabled"] = inline["enabled"]
    return cfg
