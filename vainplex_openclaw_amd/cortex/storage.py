"""Cortex persistence: <workspace>/memory/reboot/ JSON + markdown files.

Parity target: cortex `src/storage.ts` — rebootDir, atomic tmp+rename
saveJson/saveText, loadJson tolerant of missing/corrupt files,
isFileOlderThan helpers.
"""

from __future__ import annotations

import json
import os
import time
from typing import Any, Optional

from ..utils.storage import atomic_write_text


def reboot_dir(workspace: str) -> str:
    return os.path.join(workspace, "memory", "reboot")


def ensure_reboot_dir(workspace: str) -> bool:
    try:
        os.makedirs(reboot_dir(workspace), exist_ok=True)
        return True
    except OSError:
        return False


def load_json(path: str) -> dict:
    try:
        with open(path, "r", encoding="utf-8") as fh:
            data = json.load(fh)
        return data if isinstance(data, dict) else {}
    except (OSError, json.JSONDecodeError):
        return {}


def save_json(path: str, data: Any) -> bool:
    try:
        atomic_write_text(path, json.dumps(data, indent=2, ensure_ascii=False) + "\n")
        return True
    except OSError:
        return False


def load_text(path: str) -> str:
    try:
        with open(path, "r", encoding="utf-8") as fh:
            return fh.read()
    except OSError:
        return ""


def save_text(path: str, content: str) -> bool:
    try:
        atomic_write_text(path, content)
        return True
    except OSError:
        return False


def file_mtime(path: str) -> Optional[float]:
    try:
        return os.stat(path).st_mtime
    except OSError:
        return None


def is_file_older_than(path: str, hours: float, now: Optional[float] = None) -> bool:
    mtime = file_mtime(path)
    if mtime is None:
        return True
    return (now if now is not None else time.time()) - mtime > hours * 3600
