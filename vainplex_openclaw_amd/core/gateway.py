"""The host gateway: loads plugins, dispatches lifecycle hooks.

The reference relies on an external OpenClaw gateway host; this module is
the in-framework equivalent so the suite is standalone. Plugins are objects
(or modules) with ``id``/``name``/``version`` attributes and a
``register(api)`` callable (reference plugin shape:
`openclaw-governance/index.ts:66-116`, `openclaw-cortex/index.ts:11-33`).
"""

from __future__ import annotations

from typing import Any, Dict, List, Optional

from .api import HookBus, PluginApi, PluginLogger, NullLogger
from .config import load_plugin_config, plugin_enabled


class Gateway:
    def __init__(
        self,
        config: Optional[Dict[str, Any]] = None,
        logger: Optional[PluginLogger] = None,
        home: Optional[str] = None,
    ):
        self.config = config or {}
        self.logger = logger or NullLogger()
        self.home = home
        self.bus = HookBus(self.logger)
        self.plugins: Dict[str, Any] = {}
        self.apis: Dict[str, PluginApi] = {}
        self._started = False

    def load(self, plugin: Any, plugin_config: Optional[Dict[str, Any]] = None) -> PluginApi:
        pid = getattr(plugin, "id", None) or getattr(plugin, "ID", None)
        if not pid:
            raise ValueError("plugin has no id")
        cfg = plugin_config
        if cfg is None:
            entry = ((self.config.get("plugins") or {}).get("entries") or {}).get(pid) or {}
            cfg = load_plugin_config(pid, fallback=entry.get("config"), home=self.home)
        api = PluginApi(
            id=pid,
            plugin_config=cfg,
            logger=self.logger,
            config=self.config,
            bus=self.bus,
        )
        plugin.register(api)
        self.plugins[pid] = plugin
        self.apis[pid] = api
        return api

    def load_enabled(self, available: Dict[str, Any]) -> List[str]:
        loaded = []
        for pid, plugin in available.items():
            if plugin_enabled(self.config, pid):
                self.load(plugin)
                loaded.append(pid)
        return loaded

    def start(self) -> None:
        if self._started:
            return
        self._started = True
        for api in self.apis.values():
            for svc in api.services.values():
                start = svc.get("start")
                if callable(start):
                    start()
        self.emit("gateway_start", {})

    def stop(self) -> None:
        if not self._started:
            return
        self.emit("gateway_stop", {})
        for api in self.apis.values():
            for svc in api.services.values():
                stop = svc.get("stop")
                if callable(stop):
                    stop()
        self._started = False

    def emit(self, hook: str, event: Optional[Dict[str, Any]] = None, fail_closed: bool = False) -> Dict[str, Any]:
        return self.bus.emit(hook, event, fail_closed=fail_closed)

    def command(self, name: str, *args: Any, **kw: Any) -> Any:
        for api in self.apis.values():
            if name in api.commands:
                return api.commands[name](*args, **kw)
        raise KeyError(f"unknown command {name!r}")

    def gateway_method(self, name: str, *args: Any, **kw: Any) -> Any:
        for api in self.apis.values():
            if name in api.gateway_methods:
                return api.gateway_methods[name](*args, **kw)
        raise KeyError(f"unknown gateway method {name!r}")
