"""Brainplex writer: atomic config writes that never clobber.

Parity target: reference `brainplex/src/writer.ts` — atomic tmp+rename
(`:14-37`), `.bak` backups (`:40-49`), never overwrites existing plugin
configs (`:55`), merges `plugins.entries` + `plugins.allow` into
openclaw.json without touching existing entries (`:117-173`).
"""

from __future__ import annotations

import json
import os
from typing import Any, Dict, List, Optional

from ..utils.storage import atomic_write_text, backup_then_write


def plugin_config_dir(plugin_id: str, home: Optional[str] = None) -> str:
    return os.path.join(home or os.path.expanduser("~"), ".openclaw", "plugins", plugin_id)


def write_configs(configs: List[Dict[str, Any]], home: Optional[str] = None,
                  dry_run: bool = False) -> Dict[str, List[str]]:
    """Write each plugin's config.json; existing files are NEVER
    overwritten (writer.ts:55)."""
    result = {"written": [], "skipped": []}
    for pc in configs:
        pid = pc["pluginId"]
        path = os.path.join(plugin_config_dir(pid, home), "config.json")
        if os.path.isfile(path):
            result["skipped"].append(pid)
            continue
        if not dry_run:
            atomic_write_text(path, json.dumps(pc["config"], indent=2) + "\n")
        result["written"].append(pid)
    return result


def update_openclaw_config(
    config_path: str,
    config: Dict[str, Any],
    plugin_ids: List[str],
    dry_run: bool = False,
) -> Dict[str, Any]:
    """Merge entries/allow; backup first; preserve everything else."""
    result = {"updated": False, "backed_up": False, "added_entries": [], "added_allow": []}
    if not plugin_ids:
        return result
    cfg = dict(config)
    plugins = cfg.get("plugins")
    if not isinstance(plugins, dict):
        plugins = {}
    else:
        plugins = dict(plugins)
    cfg["plugins"] = plugins
    entries = plugins.get("entries")
    entries = dict(entries) if isinstance(entries, dict) else {}
    plugins["entries"] = entries
    allow = plugins.get("allow")
    allow = list(allow) if isinstance(allow, list) else []
    plugins["allow"] = allow

    for pid in plugin_ids:
        if pid not in entries:
            entries[pid] = {"enabled": True}
            result["added_entries"].append(pid)
    have = set(allow)
    for pid in plugin_ids:
        if pid not in have:
            allow.append(pid)
            result["added_allow"].append(pid)

    if not result["added_entries"] and not result["added_allow"]:
        return result
    if not dry_run:
        bak = backup_then_write(config_path, json.dumps(cfg, indent=2) + "\n")
        result["backed_up"] = bak is not None
    result["updated"] = True
    result["config"] = cfg
    return result
