"""Host plugin API: the contract every plugin programs against.

Parity target: reference `packages/openclaw-governance/src/types.ts:10-26`
(`OpenClawPluginApi`) and the hook bus semantics observed across the suite
(governance `src/hooks.ts:883-919`, nats-eventstore `src/hook-mappings.ts:33-209`,
cortex `src/hooks.ts:119-212`).

Hook names are free-form strings; the canonical set used by the suite:

    before_tool_call, after_tool_call, message_received, message_sending,
    message_sent, before_message_write, before_agent_start, agent_end,
    session_start, session_end, gateway_start, gateway_stop,
    before_compaction, after_compaction, before_reset, llm_input,
    llm_output, tool_result_persist

Handlers registered with ``api.on(hook, handler, priority=N)`` run in
descending priority order (enforcement hooks use 1000, trust feedback 900,
context injection 5 — reference governance `src/hooks.ts:883-919`).
A handler may return a dict; returned keys are shallow-merged into the
event, so e.g. a governance handler returning ``{"block": True,
"blockReason": ...}`` short-circuits the tool call.
"""

from __future__ import annotations

import time
import traceback
from dataclasses import dataclass, field
from typing import Any, Callable, Dict, List, Optional

CANONICAL_HOOKS = (
    "before_tool_call",
    "after_tool_call",
    "message_received",
    "message_sending",
    "message_sent",
    "before_message_write",
    "before_agent_start",
    "agent_end",
    "session_start",
    "session_end",
    "gateway_start",
    "gateway_stop",
    "before_compaction",
    "after_compaction",
    "before_reset",
    "llm_input",
    "llm_output",
    "tool_result_persist",
)


class PluginLogger:
    """Minimal injected logger (reference passes a PluginLogger everywhere;
    tests use no-op fakes — `test/integration.test.ts:47`)."""

    def __init__(self, name: str = "plugin", sink: Optional[Callable[[str, str], None]] = None):
        self.name = name
        self._sink = sink

    def _emit(self, level: str, msg: str) -> None:
        if self._sink is not None:
            self._sink(level, msg)

    def info(self, msg: str, *a: Any) -> None:
        self._emit("info", msg % a if a else msg)

    def warn(self, msg: str, *a: Any) -> None:
        self._emit("warn", msg % a if a else msg)

    warning = warn

    def error(self, msg: str, *a: Any) -> None:
        self._emit("error", msg % a if a else msg)

    def debug(self, msg: str, *a: Any) -> None:
        self._emit("debug", msg % a if a else msg)


class NullLogger(PluginLogger):
    def __init__(self) -> None:
        super().__init__("null", None)


@dataclass
class _HookEntry:
    priority: int
    seq: int
    plugin_id: str
    handler: Callable[[Dict[str, Any]], Any]


@dataclass
class HookDiagnostics:
    """Per-hook fire/error counters (reference cortex `src/hooks.ts:31-77`)."""

    fires: int = 0
    errors: int = 0
    last_fired: float = 0.0
    last_error: str = ""


class HookBus:
    """Priority-ordered hook dispatch.

    Descending priority; stable registration order within a priority.
    Dict results are shallow-merged into the event. Exceptions are caught,
    counted and re-raised only when ``fail_closed`` is set on dispatch
    (mirrors the reference's fail-open default — governance
    `src/hooks.ts:232-241`).
    """

    def __init__(self, logger: Optional[PluginLogger] = None):
        self._hooks: Dict[str, List[_HookEntry]] = {}
        self._seq = 0
        self._log = logger or NullLogger()
        self.diagnostics: Dict[str, HookDiagnostics] = {}

    def on(
        self,
        hook: str,
        handler: Callable[[Dict[str, Any]], Any],
        priority: int = 0,
        plugin_id: str = "",
    ) -> None:
        self._seq += 1
        entry = _HookEntry(priority=priority, seq=self._seq, plugin_id=plugin_id, handler=handler)
        lst = self._hooks.setdefault(hook, [])
        lst.append(entry)
        lst.sort(key=lambda e: (-e.priority, e.seq))

    def handlers(self, hook: str) -> List[_HookEntry]:
        return list(self._hooks.get(hook, ()))

    def emit(self, hook: str, event: Optional[Dict[str, Any]] = None, fail_closed: bool = False) -> Dict[str, Any]:
        ev: Dict[str, Any] = dict(event or {})
        ev.setdefault("hook", hook)
        diag = self.diagnostics.setdefault(hook, HookDiagnostics())
        for entry in self._hooks.get(hook, ()):
            diag.fires += 1
            diag.last_fired = time.time()
            try:
                out = entry.handler(ev)
            except Exception as exc:  # fail-open by default
                diag.errors += 1
                diag.last_error = f"{type(exc).__name__}: {exc}"
                self._log.error(
                    "hook %s handler (plugin=%s) failed: %s",
                    hook,
                    entry.plugin_id,
                    traceback.format_exc(limit=3),
                )
                if fail_closed:
                    raise
                continue
            if isinstance(out, dict):
                ev.update(out)
                if ev.get("block"):
                    break  # short-circuit: enforcement verdict stands
        return ev


@dataclass
class PluginApi:
    """What ``plugin.register(api)`` receives.

    Reference: `OpenClawPluginApi` — governance `src/types.ts:10-26`:
    { id, pluginConfig, logger, config, registerService, registerCommand,
      registerGatewayMethod, on(hookName, handler, {priority}) }.
    """

    id: str
    plugin_config: Dict[str, Any]
    logger: PluginLogger
    config: Dict[str, Any]  # full host openclaw.json config
    bus: HookBus
    services: Dict[str, Any] = field(default_factory=dict)
    commands: Dict[str, Callable[..., Any]] = field(default_factory=dict)
    gateway_methods: Dict[str, Callable[..., Any]] = field(default_factory=dict)

    def register_service(self, service: Dict[str, Any]) -> None:
        """service = {"start": fn, "stop": fn} (governance `index.ts:90-94`)."""
        self.services[service.get("id", self.id)] = service

    def register_command(self, name: str, handler: Callable[..., Any]) -> None:
        self.commands[name] = handler

    def register_gateway_method(self, name: str, handler: Callable[..., Any]) -> None:
        self.gateway_methods[name] = handler

    def on(self, hook: str, handler: Callable[[Dict[str, Any]], Any], priority: int = 0) -> None:
        self.bus.on(hook, handler, priority=priority, plugin_id=self.id)
