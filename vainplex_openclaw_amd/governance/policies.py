"""Policy model, loader/index, and the 4 builtin policies.

Parity target: governance `src/policy-loader.ts` (user + builtin load,
PolicyIndex with byHook map + regexCache) and `src/builtin-policies.ts`:
Night Mode (`:3-58`), Credential Guard (`:60-100`), Production Safeguard
with trust-tier exemption (`:110-150`), Rate Limiter 15/min, 2x for
trusted (`:152-196`). ISO-27001 control tags preserved per policy.

Policies are plain dicts:
  {id, name, version, description, scope:{hooks, agents?, excludeAgents?,
   channels?}, priority, controls:[...], rules:[{id, conditions:[...],
   effect:{action, reason?}, minTrust?, maxTrust?}]}
"""

from __future__ import annotations

import os
from typing import Any, Dict, List, Optional

from ..utils.storage import read_json
from .conditions import RegexCache

READONLY_NIGHT_TOOLS = ["read", "memory_search", "memory_get", "web_search"]


def night_mode_policy(cfg: Any) -> Optional[Dict[str, Any]]:
    if not cfg:
        return None
    raw = cfg if isinstance(cfg, dict) else {}
    after = raw.get("after") or raw.get("start") or "23:00"
    before = raw.get("before") or raw.get("end") or "08:00"
    return {
        "id": "builtin-night-mode",
        "name": "Night Mode",
        "version": "1.0.0",
        "description": f"Restricts non-critical operations between {after} and {before}",
        "scope": {"hooks": ["before_tool_call", "message_sending"]},
        "priority": 100,
        "controls": ["A.7.1", "A.6.2"],
        "rules": [
            {
                "id": "allow-critical-at-night",
                "conditions": [
                    {"type": "time", "after": after, "before": before},
                    {"type": "tool", "name": READONLY_NIGHT_TOOLS},
                ],
                "effect": {"action": "allow"},
            },
            {
                "id": "deny-non-critical-at-night",
                "conditions": [
                    {"type": "time", "after": after, "before": before},
                    {"type": "not", "condition": {"type": "tool", "name": READONLY_NIGHT_TOOLS}},
                ],
                "effect": {
                    "action": "deny",
                    "reason": f"Night mode active ({after}-{before}). Only critical operations allowed.",
                },
            },
        ],
    }


_CRED_FILE_RX = r"\.(env|pem|key)$"


def credential_guard_policy(enabled: Any) -> Optional[Dict[str, Any]]:
    if not enabled:
        return None
    return {
        "id": "builtin-credential-guard",
        "name": "Credential Guard",
        "version": "1.0.0",
        "description": "Prevents access to credential files and secrets",
        "scope": {"hooks": ["before_tool_call"]},
        "priority": 200,
        "controls": ["A.8.11", "A.8.4", "A.5.33"],
        "rules": [
            {
                "id": "block-credential-read",
                "conditions": [
                    {"type": "tool", "name": ["read", "exec", "write", "edit"]},
                    {
                        "type": "any",
                        "conditions": [
                            {"type": "tool", "params": {"file_path": {"matches": _CRED_FILE_RX}}},
                            {"type": "tool", "params": {"path": {"matches": _CRED_FILE_RX}}},
                            {"type": "tool", "params": {"command": {"matches": r"(cat|less|head|tail|cp|mv|grep|find|scp|rsync|docker\s+cp).*\.(env|pem|key)"}}},
                            {"type": "tool", "params": {"command": {"matches": r"(cp|mv|scp|rsync|docker\s+cp).*(credentials|secrets|\.env|\.pem|\.key)"}}},
                            {"type": "tool", "params": {"command": {"matches": r"(grep|find).*(password|token|secret|credential)"}}},
                            {"type": "tool", "params": {"file_path": {"contains": "credentials"}}},
                            {"type": "tool", "params": {"path": {"contains": "credentials"}}},
                            {"type": "tool", "params": {"file_path": {"contains": "secrets"}}},
                            {"type": "tool", "params": {"path": {"contains": "secrets"}}},
                        ],
                    },
                ],
                "effect": {
                    "action": "deny",
                    "reason": "Credential Guard: Access to credential files is restricted",
                },
            }
        ],
    }


def _production_ops_conditions() -> List[Dict[str, Any]]:
    return [
        {"type": "tool", "name": "exec", "params": {"command": {"matches": r"(docker push|docker-compose.*prod|systemctl.*(restart|stop|enable|disable))"}}},
        {"type": "tool", "name": "exec", "params": {"command": {"matches": r"git push.*(origin|upstream).*(main|master|prod)"}}},
        {"type": "tool", "name": "gateway", "params": {"action": {"matches": r"(restart|config\.apply|update\.run)"}}},
    ]


def production_safeguard_policy(enabled: Any) -> Optional[Dict[str, Any]]:
    if not enabled:
        return None
    trusted = {"type": "agent", "trustTier": ["trusted", "elevated"]}
    return {
        "id": "builtin-production-safeguard",
        "name": "Production Safeguard",
        "version": "1.2.0",
        "description": "Restricts production-impacting operations (trusted+ agents exempt)",
        "scope": {"hooks": ["before_tool_call"], "excludeAgents": ["unresolved"]},
        "priority": 150,
        "controls": ["A.8.31", "A.8.32", "A.8.9"],
        "rules": [
            {
                "id": "allow-production-ops-trusted",
                "conditions": [trusted, {"type": "any", "conditions": _production_ops_conditions()}],
                "effect": {"action": "allow"},
            },
            {
                "id": "block-production-ops",
                "conditions": [
                    {"type": "not", "condition": trusted},
                    {"type": "any", "conditions": _production_ops_conditions()},
                ],
                "effect": {
                    "action": "deny",
                    "reason": "Production Safeguard: This operation requires explicit approval (trusted+ agents only)",
                },
            },
        ],
    }


def rate_limiter_policy(cfg: Any) -> Optional[Dict[str, Any]]:
    if not cfg:
        return None
    max_per_minute = cfg.get("maxPerMinute", 15) if isinstance(cfg, dict) else 15
    trusted_limit = max_per_minute * 2
    trusted = {"type": "agent", "trustTier": ["trusted", "elevated"]}
    return {
        "id": "builtin-rate-limiter",
        "name": "Rate Limiter",
        "version": "1.1.0",
        "description": f"Limits agents to {max_per_minute}/min (trusted+: {trusted_limit}/min)",
        "scope": {"hooks": ["before_tool_call"]},
        "priority": 50,
        "controls": ["A.8.6"],
        "rules": [
            {
                "id": "rate-limit-trusted",
                "conditions": [
                    trusted,
                    {"type": "frequency", "maxCount": trusted_limit, "windowSeconds": 60, "scope": "agent"},
                ],
                "effect": {"action": "deny", "reason": f"Rate limit exceeded ({trusted_limit}/min for trusted agents)"},
            },
            {
                "id": "rate-limit-default",
                "conditions": [
                    {"type": "not", "condition": trusted},
                    {"type": "frequency", "maxCount": max_per_minute, "windowSeconds": 60, "scope": "agent"},
                ],
                "effect": {"action": "deny", "reason": f"Rate limit exceeded ({max_per_minute}/min)"},
            },
        ],
    }


def get_builtin_policies(cfg: Optional[Dict[str, Any]]) -> List[Dict[str, Any]]:
    cfg = cfg or {}
    out = []
    for p in (
        night_mode_policy(cfg.get("nightMode")),
        credential_guard_policy(cfg.get("credentialGuard")),
        production_safeguard_policy(cfg.get("productionSafeguard")),
        rate_limiter_policy(cfg.get("rateLimiter")),
    ):
        if p:
            out.append(p)
    return out


class PolicyIndex:
    """byHook map + shared regex cache (policy-loader.ts)."""

    def __init__(self, policies: List[Dict[str, Any]]):
        self.policies = policies
        self.regex_cache = RegexCache()
        self.by_hook: Dict[str, List[Dict[str, Any]]] = {}
        self.by_agent: Dict[str, List[Dict[str, Any]]] = {}
        self.unscoped: List[Dict[str, Any]] = []
        for p in policies:
            scope = p.get("scope") or {}
            hooks = scope.get("hooks")
            if hooks:
                for h in hooks:
                    self.by_hook.setdefault(h, []).append(p)
            else:
                self.unscoped.append(p)
            # policies without an agents scope are global ("*")
            agents = scope.get("agents") or ["*"]
            for a in agents:
                self.by_agent.setdefault(a, []).append(p)

    def for_hook(self, hook: str) -> List[Dict[str, Any]]:
        return self.by_hook.get(hook, []) + self.unscoped

    def by_id(self, pid: str) -> Optional[Dict[str, Any]]:
        for p in self.policies:
            if p.get("id") == pid:
                return p
        return None


def load_policies(
    config: Optional[Dict[str, Any]] = None,
    policy_dir: Optional[str] = None,
) -> List[Dict[str, Any]]:
    """User policies (inline config list + *.json files in policy_dir) plus
    builtins (policy-loader.ts)."""
    config = config or {}

    def _usable(p: Any) -> bool:
        # explicit enabled:false is filtered at load (policy-loader.ts:80-82)
        return isinstance(p, dict) and bool(p.get("id")) and p.get("enabled") is not False

    policies: List[Dict[str, Any]] = []
    for p in config.get("policies", []):
        if _usable(p):
            policies.append(p)
    if policy_dir and os.path.isdir(policy_dir):
        for name in sorted(os.listdir(policy_dir)):
            if not name.endswith(".json"):
                continue
            data = read_json(os.path.join(policy_dir, name))
            if isinstance(data, dict) and _usable(data):
                policies.append(data)
            elif isinstance(data, list):
                policies.extend(d for d in data if _usable(d))
    policies.extend(get_builtin_policies(config.get("builtinPolicies")))
    return policies


def build_policy_index(
    config: Optional[Dict[str, Any]] = None,
    policy_dir: Optional[str] = None,
) -> PolicyIndex:
    return PolicyIndex(load_policies(config, policy_dir))
