"""Plugin installation execution (installer.ts:40-165 parity).

The reference installs npm packages: prefer the `openclaw plugins
install` CLI when present, else run the package manager in a temp
workspace (npm 10.x idealTree workaround) and copy the installed package
into `<workspace>/.openclaw/extensions`. The MI355X build mirrors that
shape with the Python toolchain: prefer the openclaw CLI, else
`pip install --target <tmp>` and copy into the extensions directory; the
six in-suite plugins additionally support a no-network source install
(copied from this package), since this environment has no index access.

The process runner is injectable so tests drive the full execute path
without spawning real installs.
"""

from __future__ import annotations

import importlib
import os
import shutil
import subprocess
import tempfile
from typing import Any, Callable, Dict, List, Optional

INSTALL_TIMEOUT_S = 120.0  # 2-minute per-plugin timeout (installer.ts:127)

# Runner contract: (argv, cwd, timeout_s) -> (returncode, output_str)
Runner = Callable[[List[str], Optional[str], float], Any]


def default_runner(argv: List[str], cwd: Optional[str], timeout_s: float):
    proc = subprocess.run(
        argv, cwd=cwd, timeout=timeout_s,
        stdout=subprocess.PIPE, stderr=subprocess.STDOUT,
    )
    return proc.returncode, proc.stdout.decode(errors="replace")


def has_openclaw_cli(runner: Runner = default_runner) -> bool:
    """`which openclaw` probe (installer.ts:40-46)."""
    try:
        rc, _ = runner(["which", "openclaw"], None, 10.0)
        return rc == 0
    except Exception:
        return False


def _extensions_dir(workspace_path: Optional[str], home: Optional[str]) -> str:
    base = workspace_path or os.path.join(home or os.path.expanduser("~"), ".openclaw")
    return os.path.join(base, "extensions")


def install_plugin(
    plugin: Dict[str, Any],
    use_openclaw: bool,
    runner: Runner = default_runner,
    workspace_path: Optional[str] = None,
    home: Optional[str] = None,
    verbose: bool = False,
) -> Dict[str, Any]:
    """One plugin install; never raises (installer.ts installPlugin)."""
    pkg = plugin.get("package") or plugin["id"]
    try:
        if use_openclaw:
            rc, out = runner(["openclaw", "plugins", "install", pkg], None,
                             INSTALL_TIMEOUT_S)
            if rc != 0:
                raise RuntimeError(f"openclaw install failed ({rc}): {out[-200:]}")
            return {"id": plugin["id"], "success": True, "method": "openclaw"}

        # package-manager path in a temp workspace, then copy into the
        # extensions dir (installer.ts:128-165 npm idealTree workaround)
        install_dir = tempfile.mkdtemp(prefix="brainplex-install-")
        try:
            rc, out = runner(
                ["pip", "install", "--no-deps", "--target", install_dir, pkg],
                install_dir, INSTALL_TIMEOUT_S,
            )
            ext_dir = _extensions_dir(workspace_path, home)
            os.makedirs(ext_dir, exist_ok=True)
            if rc == 0:
                src = os.path.join(install_dir, pkg.replace("-", "_"))
                if os.path.isdir(src):
                    dst = os.path.join(ext_dir, os.path.basename(src))
                    if os.path.isdir(dst):
                        shutil.rmtree(dst)
                    shutil.copytree(src, dst)
                return {"id": plugin["id"], "success": True, "method": "pip"}
            # no-network fallback: the six suite plugins ship inside this
            # package — source-install by copying the module directory
            mod_name = plugin.get("module")
            if mod_name:
                mod = importlib.import_module(mod_name)
                src = os.path.dirname(os.path.abspath(mod.__file__))
                dst = os.path.join(ext_dir, plugin["id"])
                if os.path.isdir(dst):
                    shutil.rmtree(dst)
                shutil.copytree(src, dst, ignore=shutil.ignore_patterns("__pycache__"))
                return {"id": plugin["id"], "success": True, "method": "in-package"}
            raise RuntimeError(f"pip install failed ({rc}): {out[-200:]}")
        finally:
            shutil.rmtree(install_dir, ignore_errors=True)
    except Exception as exc:
        return {"id": plugin["id"], "success": False, "error": str(exc)}


def execute_installation(
    plan: Dict[str, List],
    dry_run: bool = False,
    runner: Runner = default_runner,
    workspace_path: Optional[str] = None,
    home: Optional[str] = None,
    verbose: bool = False,
) -> Dict[str, List]:
    """Run the plan; dry-run returns empty results (installer.ts:93-115)."""
    result: Dict[str, List] = {"installed": [], "failed": []}
    if dry_run or not plan.get("to_install"):
        return result
    use_openclaw = has_openclaw_cli(runner)
    for plugin in plan["to_install"]:
        entry = install_plugin(plugin, use_openclaw, runner=runner,
                               workspace_path=workspace_path, home=home,
                               verbose=verbose)
        (result["installed"] if entry["success"] else result["failed"]).append(entry)
    return result


def verify_installed(plan: Dict[str, List]) -> Dict[str, List]:
    """Post-install import verification (each suite plugin module exposes
    create_plugin); kept from the round-1 in-package checker."""
    out: Dict[str, List] = {"ok": [], "broken": []}
    for p in plan.get("to_install", []):
        try:
            mod = importlib.import_module(p["module"])
            if not hasattr(mod, "create_plugin"):
                raise AttributeError(f"{p['module']} has no create_plugin")
            out["ok"].append(p["id"])
        except Exception as exc:
            out["broken"].append({"id": p["id"], "error": str(exc)})
    return out
