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


# ===========================================================================
# boot-context.test.ts depth: execution-mode table, per-tier staleness
# messages, sort stability, thread caps, footer stats, overwrite
# ===========================================================================

from vainplex_openclaw_amd.cortex.boot_context import execution_mode


@pytest.mark.parametrize("hour,mode", [
    (2, "night-watch"), (5, "night-watch"), (7, "night-watch"),
    (8, "business-hours"), (13, "business-hours"), (17, "business-hours"),
    (18, "evening"), (22, "evening"), (23, "night-watch"),
])
def test_execution_mode_by_hour(hour, mode):
    # boot-context.ts:18-24 tiers: 8-18 business, 18-23 evening, else night
    assert execution_mode(hour) == mode


def test_open_threads_priority_then_recency():
    data = {"threads": [
        thread("old-critical", "critical", age_s=7200),
        thread("new-critical", "critical", age_s=60),
        thread("new-high", "high", age_s=10),
        thread("new-low", "low", age_s=5),
    ]}
    got = [t["title"] for t in get_open_threads(data, 10)]
    assert got == ["new-critical", "old-critical", "new-high", "new-low"]


def test_open_threads_excludes_closed_and_respects_limit():
    data = {"threads": [
        thread("a", status="closed"),
        thread("b"), thread("c"), thread("d"),
    ]}
    got = get_open_threads(data, 2)
    assert len(got) == 2 and all(t["status"] == "open" for t in got)


def test_open_threads_missing_file_shape():
    assert get_open_threads({}, 5) == []
    assert get_open_threads({"threads": []}, 5) == []


@pytest.mark.parametrize("age_h,expect", [
    (0.5, ""),                 # fresh -> no warning
    (3.0, "h old"),            # > 2h -> aging warning
    (9.0, "STALE"),            # > 8h -> STALE DATA
])
def test_integrity_tier_messages(age_h, expect):
    data = {"integrity": {"last_event_timestamp": iso(-age_h * 3600)}}
    w = integrity_warning(data, NOW)
    if expect == "":
        assert w == ""
    else:
        assert expect in w, w


def test_integrity_no_data_warns():
    assert integrity_warning({}, NOW) != ""


def test_generate_includes_footer_stats(workspace):
    write_threads(workspace, [thread("t1"), thread("t2")])
    out = gen(workspace).generate()
    assert out.startswith("#")          # markdown header
    assert "2" in out                   # thread count surfaces in footer


def test_generate_thread_cap(workspace):
    write_threads(workspace, [thread(f"t{i}") for i in range(20)])
    out = gen(workspace, max_threads=3).generate()
    listed = [ln for ln in out.splitlines() if ln.strip().startswith(("-", "*"))
              and " t" in ln]
    assert len([ln for ln in listed if any(f"t{i}" in ln for i in range(20))]) <= 3


def test_write_overwrites_existing(workspace):
    write_threads(workspace, [thread("first-run")])
    g = gen(workspace)
    assert g.write()
    path = os.path.join(workspace, "BOOTSTRAP.md")
    first = open(path).read()
    assert "first-run" in first
    write_threads(workspace, [thread("second-run")])
    assert gen(workspace).write()
    second = open(path).read()
    assert "second-run" in second and "first-run" not in second


def test_mood_emoji_for_frustrated(workspace):
    write_threads(workspace, [thread("t")], mood="frustrated")
    out = gen(workspace).generate()
    assert "frustrated" in out.lower()


def test_narrative_included_only_when_fresh(workspace):
    d = write_threads(workspace, [thread("t")])
    with open(os.path.join(d, "narrative.md"), "w") as fh:
        fh.write("NARRATIVE-CONTENT-HERE")
    # fresh mtime -> included
    out = gen(workspace).generate()
    assert "NARRATIVE-CONTENT-HERE" in out
    # stale (>36h) -> excluded
    old = NOW - 37 * 3600
    os.utime(os.path.join(d, "narrative.md"), (old, old))
    out2 = gen(workspace).generate()
    assert "NARRATIVE-CONTENT-HERE" not in out2


def test_hot_snapshot_included_only_within_1h(workspace):
    d = write_threads(workspace, [thread("t")])
    snap = os.path.join(d, "hot-snapshot.md")
    with open(snap, "w") as fh:
        fh.write("SNAPSHOT-LINES")
    out = gen(workspace).generate()
    assert "SNAPSHOT-LINES" in out
    old = NOW - 2 * 3600
    os.utime(snap, (old, old))
    assert "SNAPSHOT-LINES" not in gen(workspace).generate()
