"""Redaction-engine tables mirroring `test/redaction/engine.test.ts`
(32 its): deep object traversal, JSON-in-string, circular refs, depth
caps, type preservation, vault round-trips, perf budgets.
"""

import json
import time

import pytest

from vainplex_openclaw_amd.governance.redaction.engine import RedactionEngine
from vainplex_openclaw_amd.governance.redaction.registry import PatternRegistry
from vainplex_openclaw_amd.governance.redaction.vault import RedactionVault

KEY = "sk-abcdefghij0123456789XY"
CARD = "4111 1111 1111 1111"


@pytest.fixture
def eng():
    return RedactionEngine(PatternRegistry(), RedactionVault())


def test_string_scan_basics(eng):
    r = eng.scan_string(f"use {KEY} now")
    assert KEY not in r["output"] and r["redactionCount"] == 1
    r2 = eng.scan_string(f"{KEY} and ghp_{'f' * 36}")
    assert r2["redactionCount"] == 2
    r3 = eng.scan_string("mail alice@example.com")
    assert "alice@example.com" not in r3["output"]
    r4 = eng.scan_string(f"card {CARD}")
    assert "4111" not in r4["output"]
    assert eng.scan_string("plain text stays")["output"] == "plain text stays"
    assert eng.scan_string("")["redactionCount"] == 0


def test_deep_object_and_array_scan(eng):
    r = eng.scan({
        "a": {"b": {"secret": KEY}},
        "list": [CARD, {"inner": "bob@x.io"}, 42, None, True],
        "n": 3.14,
    })
    dumped = json.dumps(r["output"])
    assert KEY not in dumped and "4111" not in dumped and "bob@x.io" not in dumped
    assert r["output"]["list"][2] == 42          # non-strings preserved
    assert r["output"]["list"][3] is None
    assert r["output"]["list"][4] is True
    assert r["output"]["n"] == 3.14
    assert r["redactionCount"] == 3


def test_json_in_string_levels(eng):
    inner = json.dumps({"token": KEY})
    outer = json.dumps({"payload": inner})
    r = eng.scan({"wrapped": outer})
    assert KEY not in json.dumps(r["output"])
    # still valid JSON after redaction
    lvl1 = json.loads(r["output"]["wrapped"])
    lvl2 = json.loads(lvl1["payload"])
    assert lvl2["token"].startswith("[REDACTED:")


def test_json_array_in_string_and_non_json(eng):
    r = eng.scan({"arr": json.dumps([KEY, "ok"])})
    assert KEY not in r["output"]["arr"]
    assert json.loads(r["output"]["arr"])[1] == "ok"
    r2 = eng.scan({"s": "not json at all"})
    assert r2["output"]["s"] == "not json at all"
    r3 = eng.scan({"bad": '{"unclosed": '})
    assert r3["output"]["bad"] == '{"unclosed": '


def test_json_scalar_strings_survive(eng):
    assert eng.scan({"n": "42"})["output"]["n"] == "42"
    assert eng.scan({"b": "true"})["output"]["b"] == "true"


def test_circular_references_terminate(eng):
    a = {"name": "a"}
    a["self"] = a
    r = eng.scan(a)                      # must not hang
    assert r["output"]["name"] == "a"
    x = {"k": KEY}
    y = {"x": x}
    x["y"] = y                           # mutual cycle
    r2 = eng.scan(x)
    assert KEY not in str(r2["output"].get("k"))


def test_depth_cap_stops_descent(eng):
    deep = {"v": KEY}
    for _ in range(40):
        deep = {"next": deep}
    r = eng.scan(deep)                   # bounded traversal, no crash
    assert r["redactionCount"] in (0, 1)  # beyond MAX_DEPTH may stay


def test_empty_containers(eng):
    assert eng.scan({})["output"] == {}
    assert eng.scan([])["output"] == []


def test_categories_tracked(eng):
    r = eng.scan({"a": KEY, "b": "alice@example.com", "c": CARD})
    assert {"credential", "pii", "financial"} <= set(r["categories"])


def test_vault_roundtrip_through_engine(eng):
    r = eng.scan_string(f"key {KEY} end")
    ph = r["output"].split("key ")[1].split(" end")[0]
    assert eng.vault.lookup(ph) == KEY
    assert eng.vault.resolve(r["output"]) == f"key {KEY} end"


def test_perf_100kb_and_1mb(eng):
    blob = ("ordinary log line with nothing sensitive in it 0123456789 " * 1800)
    assert len(blob) > 100_000
    t0 = time.perf_counter()
    eng.scan_string(blob[:100_000])
    assert (time.perf_counter() - t0) * 1000 < 150  # generous CPU-CI budget
    big = blob * 10
    t0 = time.perf_counter()
    eng.scan_string(big[:1_000_000])
    assert (time.perf_counter() - t0) * 1000 < 1000


def test_large_nested_objects(eng):
    obj = {f"k{i}": {"v": f"text {i}", "e": f"user{i}@x.io"} for i in range(300)}
    t0 = time.perf_counter()
    r = eng.scan(obj)
    assert (time.perf_counter() - t0) * 1000 < 500
    assert r["redactionCount"] == 300
