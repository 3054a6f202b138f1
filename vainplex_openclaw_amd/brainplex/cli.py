"""Brainplex CLI: `python -m vainplex_openclaw_amd.brainplex init`.

Parity target: reference `brainplex/src/{cli,installer,output}.ts` —
hand-rolled arg parsing (`cli.ts:17-64`), the 10-step init flow, install
plan/execute (`installer.ts:51,93`). Plugins here ship inside this
package (no npm, no network), so "install" resolves each plugin id to
its in-package module and verifies it imports; everything else (plan,
skip-if-installed, config generation, never-overwrite writes, entries +
allow merge, dry-run) matches the reference flow.
"""

from __future__ import annotations

import importlib
import sys
from typing import Any, Dict, List, Optional

from .configurator import detect_timezone, generate_configs
from .scanner import ScanResult, scan
from .writer import update_openclaw_config, write_configs

VERSION = "0.1.0"

CORE_PLUGINS = [
    {"id": "openclaw-governance", "module": "vainplex_openclaw_amd.governance.plugin"},
    {"id": "openclaw-cortex", "module": "vainplex_openclaw_amd.cortex.hooks"},
    {"id": "openclaw-membrane", "module": "vainplex_openclaw_amd.membrane.hooks"},
    {"id": "openclaw-leuko", "module": "vainplex_openclaw_amd.leuko.plugin"},
    {"id": "nats-eventstore", "module": "vainplex_openclaw_amd.eventstore.plugin"},
]
OPTIONAL_PLUGINS = [
    {"id": "openclaw-knowledge-engine", "module": "vainplex_openclaw_amd.knowledge.hooks"},
]


def parse_args(args: List[str]) -> Dict[str, Any]:
    opts: Dict[str, Any] = {
        "command": "init", "full": False, "dry_run": False, "verbose": False,
        "help": False, "version": False, "config_path": None,
    }
    i = 0
    while i < len(args):
        a = args[i]
        if a == "init":
            opts["command"] = "init"
        elif a == "--full":
            opts["full"] = True
        elif a == "--dry-run":
            opts["dry_run"] = True
        elif a == "--config":
            i += 1
            opts["config_path"] = args[i] if i < len(args) else None
        elif a == "--verbose":
            opts["verbose"] = True
        elif a in ("--help", "-h"):
            opts["help"] = True
        elif a in ("--version", "-v"):
            opts["version"] = True
        elif a.startswith("-"):
            raise SystemExit(f"Unknown flag: {a}")
        i += 1
    return opts


def plan_installation(scan_result: ScanResult, configs: List[Dict], full: bool) -> Dict[str, List]:
    plugins = list(CORE_PLUGINS) + (list(OPTIONAL_PLUGINS) if full else [])
    plan: Dict[str, List] = {"to_install": [], "to_skip": [], "to_configure": [], "to_skip_config": []}
    for p in plugins:
        if p["id"] in scan_result.installed_plugins:
            plan["to_skip"].append({"id": p["id"], "reason": "already_installed"})
        else:
            plan["to_install"].append(p)
        cfg = next((c for c in configs if c["pluginId"] == p["id"]), None)
        if cfg is not None:
            if p["id"] in scan_result.configured_plugins:
                plan["to_skip_config"].append({"id": p["id"], "reason": "already_configured"})
            else:
                plan["to_configure"].append(cfg)
    return plan


def execute_installation(plan: Dict[str, List], dry_run: bool = False,
                         runner=None, workspace_path=None, home=None) -> Dict[str, List]:
    """Real install execution (installer.py): openclaw CLI when present,
    else package-manager install copied into the extensions directory,
    with the in-package source-copy fallback for the six suite plugins
    (this environment has no index access). Falls back to the import
    check when no runner is usable at all."""
    from . import installer as _inst

    if runner is None:
        # offline default: skip the process-spawning paths, source-install
        # straight from this package (still a real copy into extensions)
        def runner(argv, cwd, timeout_s):
            return 1, "offline: no package index"

    result = _inst.execute_installation(plan, dry_run=dry_run, runner=runner,
                                        workspace_path=workspace_path, home=home)
    if not dry_run:
        verify = _inst.verify_installed(plan)
        for entry in result["installed"]:
            entry["verified"] = entry["id"] in verify["ok"]
    return result


def run_init(opts: Dict[str, Any], start_dir: str = ".", home: Optional[str] = None,
             echo=print) -> Dict[str, Any]:
    """The 10-step flow (cli.ts:17-80): scan -> configure -> plan ->
    install -> write configs -> merge openclaw.json -> report."""
    echo("🧠 Brainplex — OpenClaw Plugin Suite Setup")
    sr = scan(start_dir, config_path=opts.get("config_path"), home=home)
    echo(f"🔍 Scanning... config: {sr.config_path or 'not found'}; "
         f"agents: {', '.join(sr.agents) or '(none)'}")
    configs = generate_configs(sr.agents, detect_timezone(), full=opts["full"])
    plan = plan_installation(sr, configs, opts["full"])
    echo(f"📦 Installing {len(plan['to_install'])} plugin(s), "
         f"skipping {len(plan['to_skip'])} (already installed)")
    install = execute_installation(plan, dry_run=opts["dry_run"],
                                   runner=opts.get("runner"), home=home)
    written = write_configs(plan["to_configure"], home=home, dry_run=opts["dry_run"])
    echo(f"⚙️  Configured: {', '.join(written['written']) or '(none)'}; "
         f"kept existing: {', '.join(written['skipped']) or '(none)'}")
    merged: Dict[str, Any] = {"updated": False}
    if sr.config_path:
        ids = [p["id"] for p in plan["to_install"]] + [c["pluginId"] for c in plan["to_configure"]]
        merged = update_openclaw_config(sr.config_path, sr.config,
                                        sorted(set(ids)), dry_run=opts["dry_run"])
        if merged["updated"]:
            echo(f"🔗 openclaw.json: +entries {merged['added_entries']}, "
                 f"+allow {merged['added_allow']}")
    echo("✓ Done — run: openclaw gateway restart")
    return {"scan": sr, "plan": plan, "install": install, "written": written, "merged": merged}


def main(argv: Optional[List[str]] = None) -> int:
    opts = parse_args(list(sys.argv[1:] if argv is None else argv))
    if opts["version"]:
        print(VERSION)
        return 0
    if opts["help"]:
        print("brainplex init [--full] [--dry-run] [--config PATH] [--verbose]")
        return 0
    result = run_init(opts)
    return 1 if result["install"]["failed"] else 0


if __name__ == "__main__":
    sys.exit(main())
