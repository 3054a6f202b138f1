"""Brainplex configurator: generated plugin configs + trust heuristics.

Parity target: reference `brainplex/src/configurator.ts` — name-heuristic
trust scores (admin/root 70, main 60, review/cerberus 50, forge/build 45,
default 40, `*` 10; first match wins, case-insensitive; `:11-19`), the
generated governance/cortex/membrane/leuko/KE configs (`:47-199`)
including the Membrane keys the suite documents (buffer_size 10,
default_sensitivity low, retrieve_limit 2, retrieve_min_salience 0.1,
retrieve_max_sensitivity medium, retrieve_timeout_ms 30000).
"""

from __future__ import annotations

import time
from typing import Any, Dict, List


def compute_trust_score(agent_name: str) -> int:
    name = agent_name.lower()
    if name == "*":
        return 10
    if "admin" in name or "root" in name:
        return 70
    if "main" in name:
        return 60
    if "review" in name or "cerberus" in name:
        return 50
    if "forge" in name or "build" in name:
        return 45
    return 40


def build_trust_defaults(agents: List[str]) -> Dict[str, int]:
    out = {a: compute_trust_score(a) for a in agents}
    out["*"] = 10
    return out


def detect_timezone() -> str:
    try:
        return time.tzname[0] or "UTC"
    except Exception:
        return "UTC"


def generate_governance_config(agents: List[str], timezone: str) -> Dict[str, Any]:
    return {
        "enabled": True,
        "timezone": timezone,
        "failMode": "open",
        "trust": {
            "enabled": True,
            "defaults": build_trust_defaults(agents),
            "persistIntervalSeconds": 60,
            "decay": {"enabled": True, "inactivityDays": 30, "rate": 0.95},
            "sessionTrust": {"enabled": True, "seedFactor": 0.7, "ceilingFactor": 1.2},
        },
        "nightMode": {"enabled": True, "start": "23:00", "end": "06:00"},
        "credentialGuard": {"enabled": True},
        "productionSafeguard": {"enabled": True},
        "rateLimiter": {"enabled": True, "maxPerMinute": 15},
        "builtinPolicies": {
            "credentialGuard": True,
            "productionSafeguard": True,
            "nightMode": True,
        },
        "audit": {"enabled": True},
        "policies": [],
        "responseGate": {"enabled": True},
    }


def generate_cortex_config() -> Dict[str, Any]:
    return {
        "enabled": True,
        "threadTracker": {"enabled": True, "pruneDays": 7, "maxThreads": 50},
        "decisionTracker": {"enabled": True, "maxDecisions": 100, "dedupeWindowHours": 24},
        "bootContext": {
            "enabled": True,
            "maxChars": 16000,
            "onSessionStart": True,
            "maxThreadsInBoot": 7,
            "maxDecisionsInBoot": 10,
            "decisionRecencyDays": 14,
        },
        "preCompaction": {"enabled": True, "maxSnapshotMessages": 15},
        "narrative": {"enabled": True},
        "patterns": {"language": "both"},
    }


def generate_membrane_config() -> Dict[str, Any]:
    return {
        "enabled": True,
        "buffer_size": 10,
        "default_sensitivity": "low",
        "retrieve_enabled": True,
        "retrieve_limit": 2,
        "retrieve_min_salience": 0.1,
        "retrieve_max_sensitivity": "medium",
        "retrieve_timeout_ms": 30000,
    }


def generate_leuko_config() -> Dict[str, Any]:
    return {"enabled": True}


def generate_knowledge_engine_config() -> Dict[str, Any]:
    return {"enabled": True, "entityExtraction": False}


def generate_eventstore_config() -> Dict[str, Any]:
    return {"enabled": True}


def generate_configs(agents: List[str], timezone: str, full: bool = False) -> List[Dict[str, Any]]:
    configs = [
        {"pluginId": "openclaw-governance", "config": generate_governance_config(agents, timezone)},
        {"pluginId": "openclaw-cortex", "config": generate_cortex_config()},
        {"pluginId": "openclaw-membrane", "config": generate_membrane_config()},
        {"pluginId": "openclaw-leuko", "config": generate_leuko_config()},
        {"pluginId": "nats-eventstore", "config": generate_eventstore_config()},
    ]
    if full:
        configs.append({
            "pluginId": "openclaw-knowledge-engine",
            "config": generate_knowledge_engine_config(),
        })
    return configs
