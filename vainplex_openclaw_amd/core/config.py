"""Layered plugin config loading.

Parity target: each reference plugin loads an external JSON config from
``~/.openclaw/plugins/<plugin-id>/config.json`` with ``api.pluginConfig``
as fallback (governance `src/config-loader.ts`, `index.ts:73-76`; same
pattern in knowledge-engine / nats / sitrep), and plugin activation comes
from ``openclaw.json -> plugins.entries.<id>.enabled`` plus a
``plugins.allow`` list (brainplex `src/writer.ts:141-173`).

JSON5-tolerant parsing (comments, trailing commas) mirrors brainplex
`src/scanner.ts:15-25`.
"""

from __future__ import annotations

import json
import os
import re
from typing import Any, Dict, Optional


def _strip_json5(text: str) -> str:
    """Tolerate // and /* */ comments and trailing commas (scanner.ts:15-25)."""
    out = []
    i, n = 0, len(text)
    in_str = False
    quote = ""
    while i < n:
        c = text[i]
        if in_str:
            out.append(c)
            if c == "\\" and i + 1 < n:
                out.append(text[i + 1])
                i += 2
                continue
            if c == quote:
                in_str = False
            i += 1
            continue
        if c in ('"', "'"):
            in_str = True
            quote = c
            out.append(c)
            i += 1
            continue
        if c == "/" and i + 1 < n and text[i + 1] == "/":
            while i < n and text[i] != "\n":
                i += 1
            continue
        if c == "/" and i + 1 < n and text[i + 1] == "*":
            i += 2
            while i + 1 < n and not (text[i] == "*" and text[i + 1] == "/"):
                i += 1
            i += 2
            continue
        out.append(c)
        i += 1
    s = "".join(out)
    s = re.sub(r",(\s*[}\]])", r"\1", s)  # trailing commas
    return s


def parse_jsonc(text: str) -> Any:
    try:
        return json.loads(text)
    except json.JSONDecodeError:
        return json.loads(_strip_json5(text))


def load_json_file(path: str) -> Optional[Dict[str, Any]]:
    if not os.path.isfile(path):
        return None
    with open(path, "r", encoding="utf-8") as fh:
        data = parse_jsonc(fh.read())
    return data if isinstance(data, dict) else None


def openclaw_home(env: Optional[Dict[str, str]] = None) -> str:
    env = env if env is not None else dict(os.environ)
    if env.get("OPENCLAW_HOME"):
        return env["OPENCLAW_HOME"]
    return os.path.join(env.get("HOME", os.path.expanduser("~")), ".openclaw")


def plugin_config_path(plugin_id: str, home: Optional[str] = None) -> str:
    home = home or openclaw_home()
    return os.path.join(home, "plugins", plugin_id, "config.json")


def load_plugin_config(
    plugin_id: str,
    fallback: Optional[Dict[str, Any]] = None,
    home: Optional[str] = None,
) -> Dict[str, Any]:
    """External file first, api.pluginConfig fallback (index.ts:73-76).
    A corrupt external file falls back instead of breaking plugin
    registration (config-loader.ts handles parse errors gracefully)."""
    try:
        data = load_json_file(plugin_config_path(plugin_id, home))
    except Exception:
        data = None
    if data is not None:
        return data
    return dict(fallback or {})


def resolve_defaults(config: Dict[str, Any], defaults: Dict[str, Any]) -> Dict[str, Any]:
    """Per-field defaults-resolution, recursive for dict values
    (hand-rolled per field in the reference — nats `src/config.ts:36-59`)."""
    out: Dict[str, Any] = {}
    for key, dval in defaults.items():
        cval = config.get(key)
        if isinstance(dval, dict) and isinstance(cval, dict):
            out[key] = resolve_defaults(cval, dval)
        elif cval is None:
            out[key] = dval
        else:
            out[key] = cval
    for key, cval in config.items():
        if key not in out:
            out[key] = cval
    return out


def plugin_enabled(openclaw_config: Dict[str, Any], plugin_id: str) -> bool:
    """plugins.entries.<id>.enabled plus plugins.allow (writer.ts:141-173)."""
    plugins = openclaw_config.get("plugins") or {}
    entries = plugins.get("entries") or {}
    entry = entries.get(plugin_id) or {}
    if not entry.get("enabled", False):
        return False
    allow = plugins.get("allow")
    if isinstance(allow, list):
        return plugin_id in allow
    return True


def load_layered_config(
    plugin_id: str,
    plugin_config: Optional[Dict[str, Any]],
    defaults: Dict[str, Any],
    home: Optional[str] = None,
    logger: Any = None,
    bootstrap: bool = True,
) -> Dict[str, Any]:
    """The shared config-loader pattern (knowledge-engine / nats /
    sitrep `src/config-loader.ts`): the host's pluginConfig is either a
    LEGACY full inline config (any key beyond enabled/configPath -> use
    it directly) or a minimal pointer {enabled?, configPath?}; the real
    config lives in an external JSON file (configPath override or
    ~/.openclaw/plugins/<id>/config.json), which is BOOTSTRAPPED from
    the defaults when missing; a malformed or non-object file falls
    back to defaults; an inline `enabled` boolean always overrides the
    file's. Result is defaults-resolved."""
    data = load_raw_layered(plugin_id, plugin_config, home=home,
                            logger=logger,
                            bootstrap_defaults=defaults if bootstrap else None)
    return resolve_defaults(data, defaults)


def load_raw_layered(
    plugin_id: str,
    plugin_config: Optional[Dict[str, Any]],
    home: Optional[str] = None,
    logger: Any = None,
    bootstrap_defaults: Optional[Dict[str, Any]] = None,
) -> Dict[str, Any]:
    """The layered loading WITHOUT defaults-resolution — for plugins
    whose resolve_config does its own per-field typed picks (the
    eventstore / sitrep pattern)."""
    raw = plugin_config if isinstance(plugin_config, dict) else {}
    if raw and any(k not in ("enabled", "configPath") for k in raw):
        return dict(raw)  # legacy inline config
    path = raw.get("configPath") if isinstance(raw.get("configPath"), str) \
        else plugin_config_path(plugin_id, home)
    data: Optional[Dict[str, Any]] = None
    try:
        data = load_json_file(path)
        if data is None and os.path.isfile(path) and logger is not None:
            logger.warn(f"[{plugin_id}] Config file is not an object: {path}")
    except Exception as exc:
        if logger is not None:
            logger.warn(f"[{plugin_id}] Failed to read config file {path}: {exc}")
        data = None
    if data is None and bootstrap_defaults is not None and not os.path.exists(path):
        try:
            os.makedirs(os.path.dirname(os.path.abspath(path)), exist_ok=True)
            tmp = path + ".tmp"
            with open(tmp, "w", encoding="utf-8") as fh:
                json.dump(bootstrap_defaults, fh, indent=2)
            os.replace(tmp, path)
            data = dict(bootstrap_defaults)
        except OSError:
            data = None
    if data is None:
        data = {}
    if isinstance(raw.get("enabled"), bool):
        data = {**data, "enabled": raw["enabled"]}
    return data
