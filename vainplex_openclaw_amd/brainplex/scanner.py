"""Brainplex scanner: discover openclaw.json, agents, installed plugins.

Parity target: reference `brainplex/src/scanner.ts` — JSON5-tolerant
parse (`:15-25`), config discovery walk-up with `.openclaw/` nesting and
`~/.openclaw` fallback (`:30-53`), agent extraction from 4 config shapes
(`:58-91`), installed/configured plugin detection.
"""

from __future__ import annotations

import json
import os
import re
from typing import Any, Dict, List, Optional, Set

from ..core.config import parse_jsonc

META_KEYS = {"definitions", "defaults", "list"}


def parse_config(content: str) -> Dict[str, Any]:
    """Strict JSON first, then comment/trailing-comma-tolerant."""
    try:
        return json.loads(content)
    except json.JSONDecodeError:
        return parse_jsonc(content)


def find_config(start_dir: str, home: Optional[str] = None) -> Optional[str]:
    d = os.path.abspath(start_dir)
    while True:
        direct = os.path.join(d, "openclaw.json")
        if os.path.isfile(direct):
            return direct
        nested = os.path.join(d, ".openclaw", "openclaw.json")
        if os.path.isfile(nested):
            return nested
        parent = os.path.dirname(d)
        if parent == d:
            break
        d = parent
    fallback = os.path.join(home or os.path.expanduser("~"), ".openclaw", "openclaw.json")
    return fallback if os.path.isfile(fallback) else None


def _names_from_list(items: List[Any]) -> List[str]:
    out = []
    for a in items:
        if isinstance(a, str):
            out.append(a)
        elif isinstance(a, dict):
            v = a.get("id") or a.get("name")
            if isinstance(v, str):
                out.append(v)
    return out


def extract_agents(config: Dict[str, Any]) -> List[str]:
    """4 shapes: flat array, agents.list, agents.definitions, named keys."""
    agents = config.get("agents")
    if not agents:
        return []
    if isinstance(agents, list):
        return _names_from_list(agents)
    if isinstance(agents, dict):
        if isinstance(agents.get("list"), list):
            return _names_from_list(agents["list"])
        if isinstance(agents.get("definitions"), list):
            return _names_from_list(agents["definitions"])
        return [k for k in agents.keys() if k not in META_KEYS]
    return []


def detect_installed_plugins(config: Dict[str, Any], config_dir: str) -> Set[str]:
    """Installed = plugins.entries keys + plugins.allow strings +
    plugins.installs keys (dict) or items (list) + directories under the
    extensions path (scanner.ts detectInstalledPlugins)."""
    out: Set[str] = set()
    plugins = config.get("plugins")
    if isinstance(plugins, dict):
        entries = plugins.get("entries")
        if isinstance(entries, dict):
            out.update(entries.keys())
        allow = plugins.get("allow")
        if isinstance(allow, list):
            out.update(x for x in allow if isinstance(x, str))
        installs = plugins.get("installs")
        if isinstance(installs, dict):
            out.update(installs.keys())
        elif isinstance(installs, list):
            out.update(x for x in installs if isinstance(x, str))
    ext = os.path.join(config_dir, "extensions")
    if os.path.isdir(ext):
        try:
            out.update(d for d in os.listdir(ext)
                       if os.path.isdir(os.path.join(ext, d)))
        except OSError:
            pass
    return out


def detect_configured_plugins(home: Optional[str] = None) -> Set[str]:
    """Plugins with an external ~/.openclaw/plugins/<id>/config.json."""
    base = os.path.join(home or os.path.expanduser("~"), ".openclaw", "plugins")
    if not os.path.isdir(base):
        return set()
    out = set()
    for pid in os.listdir(base):
        if os.path.isfile(os.path.join(base, pid, "config.json")):
            out.add(pid)
    return out


class ScanResult:
    def __init__(self, config_path: Optional[str], config: Dict[str, Any],
                 agents: List[str], installed: Set[str], configured: Set[str]):
        self.config_path = config_path
        self.config = config
        self.agents = agents
        self.installed_plugins = installed
        self.configured_plugins = configured


def scan(start_dir: str = ".", config_path: Optional[str] = None,
         home: Optional[str] = None) -> ScanResult:
    path = config_path or find_config(start_dir, home=home)
    config: Dict[str, Any] = {}
    if path and os.path.isfile(path):
        try:
            with open(path, "r", encoding="utf-8") as fh:
                config = parse_config(fh.read())
        except (OSError, json.JSONDecodeError, ValueError):
            config = {}
    return ScanResult(
        path,
        config,
        extract_agents(config),
        detect_installed_plugins(config, os.path.dirname(path or ".")),
        detect_configured_plugins(home=home),
    )
