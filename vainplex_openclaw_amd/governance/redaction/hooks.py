"""Redaction hook wiring: the two-layer scan plus vault resolution.

Parity target: governance `src/redaction/hooks.ts` — Layer 1: redact tool
output via `tool_result_persist` before it enters LLM context
(`:104-126,154`); vault RESOLUTION on `before_tool_call` (placeholders
in tool params are swapped back to originals; an unresolvable
placeholder blocks the call, `:238-343`); Layer 2: outbound
`message_sending` with channel-aware PII allowlisting (credentials are
never allowlisted) and the synchronous `before_message_write` scan which
runs the FULL engine but only enforces when credential/financial
categories are present — returning the fully-redacted output, PII
included (hooks.ts:405-456; RFC-007 §5.4 — PII-only content passes);
exempt agents bypass the sync gate entirely; fail-closed mode blocks
when a scan errors.
"""

from __future__ import annotations

from typing import Any, Dict, List, Optional

from ...core.api import PluginApi, PluginLogger, NullLogger
from .allowlist import (
    get_redactable_categories,
    is_agent_exempt,
    is_tool_exempt,
    normalize_allowlist,
)
from .engine import RedactionEngine
from .registry import CATEGORY_ORDER, PatternRegistry
from .vault import PLACEHOLDER_RX, RedactionVault


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
        self.stats = {"layer1Scans": 0, "layer2Scans": 0, "redactions": 0,
                      "resolutions": 0, "errors": 0}
        self._engines: Dict[tuple, RedactionEngine] = {}

    def _engine_for(self, categories: List[str]) -> RedactionEngine:
        key = tuple(sorted(categories))
        if key not in self._engines:
            self._engines[key] = RedactionEngine(
                PatternRegistry(list(categories), self.config.get("customPatterns"),
                                self.logger),
                self.vault,
            )
        return self._engines[key]

    # Vault resolution: placeholders in tool params -> originals
    def on_before_tool_call(self, ev: Dict[str, Any]) -> Optional[Dict[str, Any]]:
        if not self.enabled:
            return None
        params = ev.get("params") if ev.get("params") is not None else ev.get("toolParams")
        if params is None:
            return None
        unresolved: List[str] = []

        def walk(value):
            if isinstance(value, str):
                if PLACEHOLDER_RX.search(value):
                    out = self.vault.resolve(value)
                    for m in PLACEHOLDER_RX.finditer(out):
                        unresolved.append(m.group(0))
                    self.stats["resolutions"] += 1
                    return out
                return value
            if isinstance(value, dict):
                return {k: walk(v) for k, v in value.items()}
            if isinstance(value, list):
                return [walk(v) for v in value]
            return value

        resolved = walk(params)
        if unresolved:
            return {
                "block": True,
                "blockReason": f"Unresolvable redaction placeholder(s): {unresolved[0]}",
            }
        if resolved != params:
            return {"params": resolved}
        return None

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

    # Layer 2: outbound messages (channel-aware PII allowlisting)
    def on_message_sending(self, ev: Dict[str, Any]) -> Optional[Dict[str, Any]]:
        cats = get_redactable_categories(
            list(CATEGORY_ORDER),
            {"channel": ev.get("channel"), "agentId": ev.get("agentId")},
            self.allowlist,
        )
        return self._scan_outbound(ev, cats)

    def on_before_message_write(self, ev: Dict[str, Any]) -> Optional[Dict[str, Any]]:
        # Synchronous gate mirroring hooks.ts handleBeforeMessageWrite: exempt
        # agents pass entirely; the FULL engine scans, but enforcement only
        # triggers when credential/financial categories are present — and then
        # the fully-redacted output (PII included) is returned (RFC-007 §5.4:
        # PII-only content passes untouched).
        if not self.enabled:
            return None
        content = ev.get("content")
        if not isinstance(content, str) or not content:
            return None
        agent = str(ev.get("agentId") or "")
        if agent and is_agent_exempt(agent, self.allowlist):
            return None
        try:
            self.stats["layer2Scans"] += 1
            result = self.engine.scan_string(content)
            if result["redactionCount"] == 0:
                return None
            cats = result.get("categories") or set()
            if "credential" not in cats and "financial" not in cats:
                return None
            self.stats["redactions"] += result["redactionCount"]
            return {"content": result["output"], "redactionCount": result["redactionCount"]}
        except Exception as exc:
            self.stats["errors"] += 1
            self.logger.error("[redaction] L2-sync scan failed: %s", exc)
            if self.fail_closed:
                return {"block": True, "blockReason": "Redaction scan failed (fail-closed)"}
            return None

    def _scan_outbound(self, ev: Dict[str, Any], categories: List[str]) -> Optional[Dict[str, Any]]:
        if not self.enabled:
            return None
        agent = str(ev.get("agentId") or "")
        content = ev.get("content")
        if not isinstance(content, str) or not content:
            return None
        try:
            self.stats["layer2Scans"] += 1
            if agent and is_agent_exempt(agent, self.allowlist):
                categories = ["credential"]  # exemptions never cover credentials
            engine = (
                self.engine
                if tuple(sorted(categories)) == tuple(sorted(CATEGORY_ORDER))
                else self._engine_for(categories)
            )
            result = engine.scan_string(content)
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
    api.on("before_tool_call", state.on_before_tool_call, priority=1100)
    api.on("message_sending", state.on_message_sending, priority=1100)
    api.on("before_message_write", state.on_before_message_write, priority=1100)
    return state
