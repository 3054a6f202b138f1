"""Redaction hook wiring: the two-layer scan.

Parity target: governance `src/redaction/hooks.ts` — Layer 1: redact tool
output via `tool_result_persist` before it enters LLM context
(`:104-126,154`); Layer 2: outbound `message_sending` + synchronous
`before_message_write` scans (`:127-141`); fail-closed mode blocks when a
scan errors.
"""

from __future__ import annotations

from typing import Any, Dict, Optional

from ...core.api import PluginApi, PluginLogger, NullLogger
from .allowlist import normalize_allowlist, is_tool_exempt, is_agent_exempt
from .engine import RedactionEngine
from .registry import PatternRegistry
from .vault import RedactionVault


class RedactionState:
    def __init__(self, config: Optional[Dict[str, Any]] = None, logger: Optional[PluginLogger] = None):
        config = config or {}
        self.config = config
        self.logger = logger or NullLogger()
        self.enabled = bool(config.get("enabled", True))
        self.fail_closed = bool(config.get("failClosed", False))
        self.allowlist = normalize_allowlist(config.get("allowlist"))
        self.registry = PatternRegistry(
            config.get("enabledCategories"), config.get("customPatterns"), self.logger
        )
        self.vault = RedactionVault(float(config.get("vaultExpirySeconds", 3600)))
        self.engine = RedactionEngine(self.registry, self.vault)
        self.stats = {"layer1Scans": 0, "layer2Scans": 0, "redactions": 0, "errors": 0}

    # Layer 1: tool output entering context
    def on_tool_result_persist(self, ev: Dict[str, Any]) -> Optional[Dict[str, Any]]:
        if not self.enabled:
            return None
        tool = str(ev.get("toolName") or "")
        if tool and is_tool_exempt(tool, self.allowlist):
            # exempt tools: credentials are STILL redacted
            return self._scan_credentials_only(ev)
        try:
            self.stats["layer1Scans"] += 1
            result = self.engine.scan(ev.get("result"))
            self.stats["redactions"] += result["redactionCount"]
            if result["redactionCount"]:
                return {"result": result["output"], "redactionCount": result["redactionCount"]}
            return None
        except Exception as exc:
            self.stats["errors"] += 1
            self.logger.error("[redaction] Layer-1 scan failed: %s", exc)
            if self.fail_closed:
                return {"result": "[REDACTION-ERROR: output withheld]", "block": True}
            return None

    def _scan_credentials_only(self, ev: Dict[str, Any]) -> Optional[Dict[str, Any]]:
        cred_registry = PatternRegistry(["credential"], None, self.logger)
        engine = RedactionEngine(cred_registry, self.vault)
        result = engine.scan(ev.get("result"))
        if result["redactionCount"]:
            self.stats["redactions"] += result["redactionCount"]
            return {"result": result["output"], "redactionCount": result["redactionCount"]}
        return None

    # Layer 2: outbound messages
    def on_message_sending(self, ev: Dict[str, Any]) -> Optional[Dict[str, Any]]:
        return self._scan_outbound(ev)

    def on_before_message_write(self, ev: Dict[str, Any]) -> Optional[Dict[str, Any]]:
        return self._scan_outbound(ev)

    def _scan_outbound(self, ev: Dict[str, Any]) -> Optional[Dict[str, Any]]:
        if not self.enabled:
            return None
        agent = str(ev.get("agentId") or "")
        content = ev.get("content")
        if not isinstance(content, str) or not content:
            return None
        try:
            self.stats["layer2Scans"] += 1
            if agent and is_agent_exempt(agent, self.allowlist):
                cred_registry = PatternRegistry(["credential"], None, self.logger)
                engine = RedactionEngine(cred_registry, self.vault)
                result = engine.scan_string(content)
            else:
                result = self.engine.scan_string(content)
            self.stats["redactions"] += result["redactionCount"]
            if result["redactionCount"]:
                return {"content": result["output"], "redactionCount": result["redactionCount"]}
            return None
        except Exception as exc:
            self.stats["errors"] += 1
            self.logger.error("[redaction] Layer-2 scan failed: %s", exc)
            if self.fail_closed:
                return {"block": True, "blockReason": "Redaction scan failed (fail-closed)"}
            return None


def register_redaction_hooks(api: PluginApi, config: Optional[Dict[str, Any]] = None) -> RedactionState:
    state = RedactionState(config, api.logger)
    api.on("tool_result_persist", state.on_tool_result_persist, priority=1100)
    api.on("message_sending", state.on_message_sending, priority=1100)
    api.on("before_message_write", state.on_before_message_write, priority=1100)
    return state
