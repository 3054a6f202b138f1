"""Per-plugin manifests (`openclaw.plugin.json` parity).

Parity target: each reference package ships an `openclaw.plugin.json`
with {id, name, description, version, configSchema}
(`openclaw-governance/openclaw.plugin.json:1-24`). Here the manifests are
code (one source of truth next to the plugins); `write_manifest()` emits
the JSON file for hosts that expect it on disk.
"""

from __future__ import annotations

import json
import os
from typing import Any, Dict, Optional

_BASE_SCHEMA: Dict[str, Any] = {
    "type": "object",
    "additionalProperties": True,
    "properties": {
        "enabled": {
            "type": "boolean",
            "default": True,
            "description": "Enable/disable the plugin",
        },
        "configPath": {
            "type": "string",
            "description": "Path to external config file "
                           "(default: ~/.openclaw/plugins/<id>/config.json)",
        },
    },
}


def _manifest(pid: str, name: str, description: str,
              extra_props: Optional[Dict[str, Any]] = None) -> Dict[str, Any]:
    schema = json.loads(json.dumps(_BASE_SCHEMA))
    if extra_props:
        schema["properties"].update(extra_props)
    return {
        "id": pid,
        "name": name,
        "description": description,
        "version": "0.1.0",
        "configSchema": schema,
    }


MANIFESTS: Dict[str, Dict[str, Any]] = {
    m["id"]: m
    for m in [
        _manifest(
            "openclaw-governance", "OpenClaw Governance",
            "Contextual, learning, cross-agent governance for AI agents",
        ),
        _manifest(
            "openclaw-cortex", "OpenClaw Cortex",
            "Conversation intelligence: threads, decisions, commitments, boot context, trace analysis",
        ),
        _manifest(
            "openclaw-knowledge-engine", "OpenClaw Knowledge Engine",
            "Entity and relationship extraction, SPO fact store",
        ),
        _manifest(
            "nats-eventstore", "NATS Event Store",
            "Publish agent events for audit, replay, and multi-agent sharing",
        ),
        _manifest(
            "openclaw-membrane", "Membrane",
            "Episodic memory: salience-based recall with organic decay",
            {"buffer_size": {"type": "number", "default": 10},
             "retrieve_limit": {"type": "number", "default": 2}},
        ),
        _manifest(
            "openclaw-leuko", "Leuko",
            "Cognitive immune system: health checks, anomaly detection, sitrep",
        ),
    ]
}


def get_manifest(plugin_id: str) -> Dict[str, Any]:
    return MANIFESTS[plugin_id]


def write_manifest(plugin_id: str, directory: str) -> str:
    path = os.path.join(directory, "openclaw.plugin.json")
    os.makedirs(directory, exist_ok=True)
    with open(path, "w", encoding="utf-8") as fh:
        json.dump(MANIFESTS[plugin_id], fh, indent=2)
        fh.write("\n")
    return path
