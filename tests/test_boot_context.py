"""Boot-context mirror tests: thread ordering, integrity staleness,
mood display, recency filters, truncation, write/shouldGenerate — from
the reference's `test/boot-context.test.ts`."""

import datetime
import json
import os

import pytest

from vainplex_openclaw_amd.cortex.boot_context import (
    BootContextConfig,
    BootContextGenerator,
    get_open_threads,
    integrity_warning,
)

NOW = 1_700_000_000.0  # fixed clock for the whole module


def iso(offset_s=0.0):
    return (
        datetime.datetime.fromtimestamp(NOW + offset_s, datetime.timezone.utc)
        .isoformat()
        .replace("+00:00", "Z")
    )


def write_threads(workspace, threads, mood="neutral", integrity_ts="fresh"):
    d = os.path.join(workspace, "memory", "reboot")
    os.makedirs(d, exist_ok=True)
    data = {"version": 2, "threads": threads, "session_mood": mood}
    if integrity_ts == "fresh":
        data["integrity"] = {"last_event_timestamp": iso()}
    elif integrity_ts is not None:
        data["integrity"] = {"last_event_timestamp": integrity_ts}
    with open(os.path.join(d, "threads.json"), "w") as fh:
        json.dump(data, fh)
    return d


def thread(title, priority="medium", status="open", age_s=0.0, **kw):
    t = {"id": title, "title": title, "priority": priority, "status": status,
         "last_activity": iso(-age_s)}
    t.update(kw)
    return t


def gen(workspace, **cfg):
    return BootContextGenerator(workspace, BootContextConfig(**cfg), clock=lambda: NOW)


# -- getOpenThreads ----------------------------------------------------------

def test_open_threads_filter_and_order():
    data = {"threads": [
        thread("done", status="resolved"),
        thread("low old", priority="low", age_s=9000),
        thread("crit", priority="critical", age_s=5000),
        thread("high new", priority="high", age_s=10),
        thread("high old", priority="high", age_s=8000),
    ]}
    out = get_open_threads(data, 10)
    assert [t["title"] for t in out] == ["crit", "high new", "high old", "low old"]
    assert [t["title"] for t in get_open_threads(data, 2)] == ["crit", "high new"]
    assert get_open_threads({}, 5) == []


# -- integrityWarning --------------------------------------------------------

def test_integrity_warning_tiers():
    assert "No integrity data" in integrity_warning({}, NOW)
    assert integrity_warning({"integrity": {"last_event_timestamp": iso()}}, NOW) == ""
    w3h = integrity_warning({"integrity": {"last_event_timestamp": iso(-3 * 3600)}}, NOW)
    assert w3h.startswith("⚠️ Data staleness") and "3h" in w3h
    w9h = integrity_warning({"integrity": {"last_event_timestamp": iso(-9 * 3600)}}, NOW)
    assert w9h.startswith("🚨 STALE DATA") and "9h" in w9h
    assert "Could not parse" in integrity_warning(
        {"integrity": {"last_event_timestamp": "not-a-date"}}, NOW)


# -- generate ----------------------------------------------------------------

def test_generate_header_mode_and_footer(workspace):
    write_threads(workspace, [thread("alpha")])
    out = gen(workspace).generate()
    assert out.startswith("# BOOT CONTEXT")
    assert "Execution mode: **" in out
    assert "_Boot context | 1 active threads | 0 recent decisions_" in out


def test_generate_mood_shown_only_when_not_neutral(workspace):
    write_threads(workspace, [], mood="frustrated")
    out = gen(workspace).generate()
    assert "Last session mood: frustrated 😤" in out
    write_threads(workspace, [], mood="neutral")
    assert "session mood" not in gen(workspace).generate()


def test_generate_threads_section_with_priority_emoji(workspace):
    write_threads(workspace, [thread("urgent fix", priority="critical"),
                              thread("later", priority="low")])
    out = gen(workspace).generate()
    assert "## Open threads" in out
    assert out.index("urgent fix") < out.index("later")
    assert "🔴 [critical] urgent fix" in out


def test_generate_decision_recency_filter(workspace):
    d = write_threads(workspace, [])
    old_date = datetime.datetime.fromtimestamp(NOW - 30 * 86400).strftime("%Y-%m-%d")
    new_date = datetime.datetime.fromtimestamp(NOW - 3600).strftime("%Y-%m-%d")
    with open(os.path.join(d, "decisions.json"), "w") as fh:
        json.dump({"decisions": [
            {"date": old_date, "what": "ancient choice", "impact": "high"},
            {"date": new_date, "what": "fresh choice", "impact": "critical", "why": "because"},
        ]}, fh)
    out = gen(workspace).generate()
    assert "fresh choice" in out and "ancient choice" not in out
    assert "🔴" in out and "Why: because" in out
    assert "| 1 recent decisions_" in out


def test_generate_hot_snapshot_and_narrative_staleness(workspace):
    d = write_threads(workspace, [])
    for name in ("hot-snapshot.md", "narrative.md"):
        with open(os.path.join(d, name), "w") as fh:
            fh.write(f"content of {name}")
    g = gen(workspace)
    out = g.generate()
    assert "Hot snapshot" in out and "content of hot-snapshot.md" in out
    assert "Narrative" in out and "content of narrative.md" in out
    # age the files: snapshot >1h stale, narrative survives until 36h
    old = NOW - 2 * 3600
    for name in ("hot-snapshot.md",):
        os.utime(os.path.join(d, name), (old, old))
    out2 = g.generate()
    assert "Hot snapshot" not in out2 and "Narrative" in out2
    ancient = NOW - 40 * 3600
    os.utime(os.path.join(d, "narrative.md"), (ancient, ancient))
    assert "Narrative" not in g.generate()


def test_generate_handles_empty_state(workspace):
    out = gen(workspace).generate()
    assert out.startswith("# BOOT CONTEXT")
    assert "0 active threads" in out
    assert "No integrity data" in out


def test_truncation_budget(workspace):
    write_threads(workspace, [thread("t" * 80 + str(i)) for i in range(40)])
    out = gen(workspace, max_chars=500).generate()
    assert len(out) < 560
    assert out.endswith("_[truncated to token budget]_")
    # within budget -> no marker
    out2 = gen(workspace, max_chars=100_000).generate()
    assert "truncated" not in out2


def test_write_bootstrap_md(workspace):
    write_threads(workspace, [thread("alpha")])
    g = gen(workspace)
    assert g.write() is True
    path = os.path.join(workspace, "BOOTSTRAP.md")
    with open(path) as fh:
        assert "# BOOT CONTEXT" in fh.read()
    # overwrites
    assert g.write() is True


def test_should_generate_gate(workspace):
    assert gen(workspace).should_generate() is True
    assert gen(workspace, enabled=False).should_generate() is False
    assert gen(workspace, on_session_start=False).should_generate() is False


def test_legacy_array_format(workspace):
    d = os.path.join(workspace, "memory", "reboot")
    os.makedirs(d, exist_ok=True)
    with open(os.path.join(d, "threads.json"), "w") as fh:
        json.dump([thread("legacy-thread")], fh)
    out = gen(workspace).generate()
    assert "legacy-thread" in out
