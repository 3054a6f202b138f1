"""Atomic file persistence + debounced savers.

Parity target: knowledge-engine `src/storage.ts` (tmp+rename atomic writes,
debounce helper), brainplex `src/writer.ts:14-37` (atomic tmp+rename,
`.bak` backups, never-overwrite), and the dirty-flag + interval flush
pattern of governance `src/trust-manager.ts:291-324`.
"""

from __future__ import annotations

import json
import os
import threading
import time
from typing import Any, Callable, Optional


def atomic_write_text(path: str, text: str) -> None:
    """tmp + os.replace in the destination directory (writer.ts:14-37)."""
    d = os.path.dirname(os.path.abspath(path))
    os.makedirs(d, exist_ok=True)
    tmp = os.path.join(d, f".{os.path.basename(path)}.tmp.{os.getpid()}.{threading.get_ident()}")
    with open(tmp, "w", encoding="utf-8") as fh:
        fh.write(text)
        fh.flush()
        os.fsync(fh.fileno())
    os.replace(tmp, path)


def atomic_write_json(path: str, obj: Any, indent: int = 2) -> None:
    atomic_write_text(path, json.dumps(obj, indent=indent, ensure_ascii=False))


def read_json(path: str, default: Any = None) -> Any:
    if not os.path.isfile(path):
        return default
    try:
        with open(path, "r", encoding="utf-8") as fh:
            return json.load(fh)
    except (json.JSONDecodeError, OSError):
        return default


def backup_then_write(path: str, text: str) -> Optional[str]:
    """Write with a `.bak` of any pre-existing file (writer.ts:40-49)."""
    bak = None
    if os.path.isfile(path):
        bak = path + ".bak"
        with open(path, "rb") as src, open(bak, "wb") as dst:
            dst.write(src.read())
    atomic_write_text(path, text)
    return bak


class DebouncedSaver:
    """Debounced persistence: calls `save_fn` at most once per `delay`
    seconds after the last `mark_dirty()` (fact-store.ts:25-34 debounced
    AtomicStorage; commitment-tracker 15 s debounced saves)."""

    def __init__(self, save_fn: Callable[[], None], delay: float = 0.25):
        self._save = save_fn
        self._delay = delay
        self._timer: Optional[threading.Timer] = None
        self._lock = threading.Lock()
        self.dirty = False

    def mark_dirty(self) -> None:
        with self._lock:
            self.dirty = True
            if self._timer is not None:
                self._timer.cancel()
            self._timer = threading.Timer(self._delay, self.flush)
            self._timer.daemon = True
            self._timer.start()

    def flush(self) -> None:
        with self._lock:
            if self._timer is not None:
                self._timer.cancel()
                self._timer = None
            if not self.dirty:
                return
            self.dirty = False
        self._save()

    def close(self) -> None:
        self.flush()


class IntervalFlusher:
    """Dirty-flag + periodic flush thread (trust-manager.ts:291-324).
    Synchronous `flush()` always available; the thread is optional so tests
    stay deterministic."""

    def __init__(self, save_fn: Callable[[], None], interval: float = 5.0):
        self._save = save_fn
        self.interval = interval
        self.dirty = False
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None
        self._lock = threading.Lock()

    def mark_dirty(self) -> None:
        self.dirty = True

    def flush(self) -> None:
        with self._lock:
            if not self.dirty:
                return
            self.dirty = False
        self._save()

    def start(self) -> None:
        if self._thread is not None:
            return
        self._stop.clear()

        def run() -> None:
            while not self._stop.wait(self.interval):
                try:
                    self.flush()
                except Exception:
                    pass

        self._thread = threading.Thread(target=run, daemon=True)
        self._thread.start()

    def stop(self) -> None:
        self._stop.set()
        if self._thread is not None:
            self._thread.join(timeout=2.0)
            self._thread = None
        self.flush()


def now_ms() -> int:
    return int(time.time() * 1000)


def now_us() -> int:
    return int(time.time() * 1_000_000)
