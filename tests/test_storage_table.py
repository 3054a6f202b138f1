"""Storage-layer tables mirroring cortex `test/storage.test.ts` (21 its)
plus the shared utils/storage atomic-write/debounce/interval helpers
(the reference's AtomicStorage + per-plugin storage.ts files).
"""

import json
import os
import threading
import time

import pytest

from vainplex_openclaw_amd.cortex import storage as cst
from vainplex_openclaw_amd.utils.storage import (
    DebouncedSaver,
    IntervalFlusher,
    atomic_write_json,
    atomic_write_text,
    backup_then_write,
    read_json,
)


# -- cortex reboot dir --------------------------------------------------------

def test_reboot_dir_path_and_creation(tmp_path):
    ws = str(tmp_path)
    d = cst.reboot_dir(ws)
    assert d.endswith(os.path.join("memory", "reboot"))
    assert cst.ensure_reboot_dir(ws) is True
    assert os.path.isdir(d)
    assert cst.ensure_reboot_dir(ws) is True     # already exists


# -- load/save JSON -----------------------------------------------------------

def test_load_json_matrix(tmp_path):
    p = str(tmp_path / "f.json")
    with open(p, "w") as fh:
        json.dump({"a": 1}, fh)
    assert cst.load_json(p) == {"a": 1}
    assert cst.load_json(str(tmp_path / "ghost.json")) == {}
    with open(p, "w") as fh:
        fh.write("{corrupt")
    assert cst.load_json(p) == {}
    open(p, "w").close()                          # empty file
    assert cst.load_json(p) == {}


def test_save_json_atomic_and_dirs(tmp_path):
    p = str(tmp_path / "deep" / "dir" / "out.json")
    assert cst.save_json(p, {"x": [1, 2]}) is True
    assert json.load(open(p)) == {"x": [1, 2]}
    leftovers = [f for f in os.listdir(os.path.dirname(p)) if ".tmp" in f]
    assert leftovers == []
    # pretty-printed, 2-space indent
    text = open(p).read()
    assert '\n  "x"' in text


# -- load/save text -----------------------------------------------------------

def test_text_roundtrip_and_missing(tmp_path):
    p = str(tmp_path / "sub" / "note.md")
    assert cst.save_text(p, "# hello\nbody") is True
    assert cst.load_text(p) == "# hello\nbody"
    assert cst.load_text(str(tmp_path / "nope.md")) == ""


# -- mtime / staleness --------------------------------------------------------

def test_mtime_and_staleness(tmp_path):
    p = str(tmp_path / "f.txt")
    assert cst.file_mtime(p) is None
    assert cst.is_file_older_than(p, 1.0) is True     # missing = stale
    with open(p, "w") as fh:
        fh.write("x")
    assert isinstance(cst.file_mtime(p), float)
    assert cst.is_file_older_than(p, 1.0) is False    # fresh
    old = time.time() - 7200
    os.utime(p, (old, old))
    assert cst.is_file_older_than(p, 1.0) is True
    assert cst.is_file_older_than(p, 3.0) is False


# -- utils: atomic writes, backup, debounce, interval ------------------------

def test_atomic_write_text_and_json(tmp_path):
    p = str(tmp_path / "a" / "b.txt")
    atomic_write_text(p, "first")
    atomic_write_text(p, "second")
    assert open(p).read() == "second"
    j = str(tmp_path / "a" / "c.json")
    atomic_write_json(j, {"k": True})
    assert read_json(j) == {"k": True}
    assert read_json(str(tmp_path / "missing.json"), default=42) == 42


def test_backup_then_write(tmp_path):
    p = str(tmp_path / "cfg.json")
    atomic_write_text(p, "v1")
    bak = backup_then_write(p, "v2")
    assert open(p).read() == "v2"
    assert bak and open(bak).read() == "v1" and bak.endswith(".bak")
    # first write of a new file makes no backup
    p2 = str(tmp_path / "new.json")
    assert backup_then_write(p2, "x") is None
    assert open(p2).read() == "x"


def test_debounced_saver_coalesces():
    saves = []
    s = DebouncedSaver(lambda: saves.append(time.time()), delay=0.05)
    for _ in range(10):
        s.mark_dirty()
    time.sleep(0.15)
    assert len(saves) == 1                         # ten marks, one save
    s.mark_dirty()
    s.flush()                                      # flush forces now
    assert len(saves) == 2
    s.close()


def test_interval_flusher_periodic_and_stop():
    saves = []
    f = IntervalFlusher(lambda: saves.append(1), interval=0.04)
    f.start()
    f.start()                                      # idempotent
    f.mark_dirty()
    time.sleep(0.15)
    n = len(saves)
    assert n >= 1
    # clean-flag skip: no dirt, no save
    time.sleep(0.1)
    f.flush()
    assert len(saves) == n
    f.mark_dirty()
    f.stop()
    f.flush()                                      # sync flush still works
    final = len(saves)
    assert final == n + 1
    time.sleep(0.1)
    assert len(saves) == final                     # thread stopped
