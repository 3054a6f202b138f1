"""Decision tracker mirror tests from the reference's
`test/decision-tracker.test.ts`: impact inference, extraction, dedupe,
cap, persistence/load, recency filter."""

import json
import os

import pytest

from vainplex_openclaw_amd.cortex.decision_tracker import (
    DecisionTracker,
    DecisionTrackerConfig,
    infer_impact,
)

T0 = 1_700_000_000.0


@pytest.mark.parametrize("word", [
    "architecture", "security", "migration", "delete", "production",
    "deploy", "critical", "strategy", "architektur", "löschen",
])
def test_infer_impact_high_keywords(word):
    assert infer_impact(f"we decided on the {word} change") == "high"


def test_infer_impact_medium_default():
    assert infer_impact("a perfectly ordinary choice") == "medium"
    assert infer_impact("") == "medium"


def make_tracker(workspace, **cfg):
    t = [T0]
    dt = DecisionTracker(workspace, DecisionTrackerConfig(**cfg), clock=lambda: t[0])
    return dt, t


def test_extract_english_and_german(workspace):
    dt, _ = make_tracker(workspace)
    assert dt.decisions == []
    assert dt.process_message("After discussion we decided to use postgres for storage") >= 1
    assert any("postgres" in d["what"] for d in dt.decisions)
    assert dt.process_message("Wir haben uns entschieden, die API zu versionieren") >= 1


def test_decision_record_shape(workspace):
    dt, _ = make_tracker(workspace)
    dt.process_message("we decided to ship on friday")
    d = dt.decisions[0]
    assert len(d["date"]) == 10 and d["date"][4] == "-" and d["date"][7] == "-"
    assert "T" in d["timestamp"] and d["timestamp"].endswith("Z")
    assert d["impact"] in ("high", "medium")
    # unique ids
    dt.process_message("we decided to also ship the docs")
    ids = [x["id"] for x in dt.decisions]
    assert len(set(ids)) == len(ids)


def test_no_extraction_from_unrelated_or_empty(workspace):
    dt, _ = make_tracker(workspace)
    assert dt.process_message("the weather is nice today") == 0
    assert dt.process_message("") == 0
    assert dt.decisions == []


def test_dedupe_within_window_and_distinct_allowed(workspace):
    dt, t = make_tracker(workspace)
    assert dt.process_message("we decided to use redis here") == 1
    assert dt.process_message("we decided to use redis here") == 0  # same window
    assert dt.process_message("we decided to use kafka instead") == 1
    t[0] += 25 * 3600  # outside the 24h window
    assert dt.process_message("we decided to use redis here") == 1


def test_impact_assignment_through_tracker(workspace):
    dt, _ = make_tracker(workspace)
    dt.process_message("we decided to rework the security architecture")
    assert dt.decisions[-1]["impact"] == "high"
    dt.process_message("we decided to rename the button")
    assert dt.decisions[-1]["impact"] == "medium"


def test_max_decisions_cap_drops_oldest(workspace):
    dt, t = make_tracker(workspace, max_decisions=5)
    for i in range(8):
        t[0] += 3600 * 25
        assert dt.process_message(f"we decided to adopt tool number {i} today") == 1
    assert len(dt.decisions) == 5
    assert "number 7" in dt.decisions[-1]["what"]
    assert all(f"number {i}" not in json.dumps(dt.decisions) for i in (0, 1, 2))


def test_persistence_and_load(workspace):
    dt, _ = make_tracker(workspace)
    dt.process_message("we decided to persist this decision")
    path = os.path.join(workspace, "memory", "reboot", "decisions.json")
    assert os.path.isfile(path)
    dt2, _ = make_tracker(workspace)
    assert any("persist this decision" in d["what"] for d in dt2.decisions)


def test_corrupt_and_missing_files(workspace):
    path = os.path.join(workspace, "memory", "reboot")
    os.makedirs(path, exist_ok=True)
    with open(os.path.join(path, "decisions.json"), "w") as fh:
        fh.write("{corrupt!")
    dt, _ = make_tracker(workspace)  # must not raise
    assert dt.decisions == []
    assert dt.process_message("we decided to recover gracefully") == 1


def test_recent_within_days_and_limit(workspace):
    dt, t = make_tracker(workspace)
    dt.process_message("we decided on the ancient thing")
    t[0] += 30 * 86400
    for i in range(4):
        t[0] += 86400 * 1.01
        dt.process_message(f"we decided on fresh thing {i}")
    recent = dt.recent_within(days=7, limit=10)
    assert len(recent) == 4
    assert all("fresh" in d["what"] for d in recent)
    assert len(dt.recent_within(days=7, limit=2)) == 2
    assert len(dt.recent_within(days=365, limit=100)) == 5


def test_multiple_decisions_one_message(workspace):
    dt, _ = make_tracker(workspace)
    n = dt.process_message(
        "we decided to use rust for the agent. separately, we agreed to "
        "postpone the migration until May."
    )
    assert n >= 2


# ===========================================================================
# decision-tracker.test.ts depth: context windows, multi-decision
# messages, impact assignment per scenario, cap eviction order, corrupt
# and missing files, recency/limit interplay
# ===========================================================================

def test_no_extract_from_unrelated_or_empty(workspace):
    dt, _ = make_tracker(workspace)
    assert dt.process_message("the weather is nice today") == 0
    assert dt.process_message("") == 0
    assert dt.process_message("   ") == 0
    assert dt.decisions == []


def test_context_window_for_why(workspace):
    dt, _ = make_tracker(workspace)
    before = "Because latency dominated the eval results and p99 spiked, "
    dt.process_message(before + "we decided to move to the rust rewrite")
    d = dt.decisions[0]
    ctx = d.get("why") or d.get("context") or ""
    assert "latency" in str(ctx), d


@pytest.mark.parametrize("text,impact", [
    ("we decided on the new architecture for auth", "high"),
    ("we decided to tighten security reviews", "high"),
    ("we decided to start the data migration", "high"),
    ("we decided to pick blue as the accent color", "medium"),
    ("we decided to rename the helper function", "medium"),
])
def test_impact_assignment_scenarios(workspace, text, impact):
    dt, _ = make_tracker(workspace)
    dt.process_message(text)
    assert dt.decisions and dt.decisions[0]["impact"] == impact, text


def test_dedupe_window_expires(workspace):
    dt, t = make_tracker(workspace)
    assert dt.process_message("we decided to use postgres") == 1
    # identical within 24 h window -> deduped
    t[0] += 3600
    assert dt.process_message("we decided to use postgres") == 0
    # after the window it is a new decision again
    t[0] += 25 * 3600
    assert dt.process_message("we decided to use postgres") == 1
    assert len(dt.decisions) == 2


def test_cap_evicts_oldest_first(workspace):
    dt, t = make_tracker(workspace, max_decisions=3)
    for i in range(5):
        t[0] += 90000  # outside the dedupe window each time
        dt.process_message(f"we decided to adopt plan-{i} next")
    assert len(dt.decisions) == 3
    whats = " ".join(d["what"] for d in dt.decisions)
    assert "plan-4" in whats and "plan-0" not in whats and "plan-1" not in whats


def test_load_roundtrip_and_missing(workspace):
    dt, t = make_tracker(workspace)
    dt.process_message("we decided to archive old logs")
    dt.flush()
    dt2, _ = make_tracker(workspace)
    assert any("archive" in d["what"] for d in dt2.decisions)
    # missing file in a fresh workspace
    import tempfile

    dt3, _ = make_tracker(tempfile.mkdtemp(prefix="dt-missing-"))
    assert dt3.decisions == []


def test_corrupt_file_graceful(workspace):
    d = os.path.join(workspace, "memory", "reboot")
    os.makedirs(d, exist_ok=True)
    with open(os.path.join(d, "decisions.json"), "w") as fh:
        fh.write("{not json!!")
    dt, _ = make_tracker(workspace)
    assert dt.decisions == []
    assert dt.process_message("we decided to recover gracefully") == 1


def test_recency_filter_and_limit(workspace):
    dt, t = make_tracker(workspace)
    dt.process_message("we decided on the ancient plan")
    t[0] += 10 * 86400
    for i in range(4):
        t[0] += 90000
        dt.process_message(f"we decided on recent plan {i}")
    recent = dt.recent_within(days=7.0, limit=10)
    assert all("recent" in d["what"] for d in recent)
    assert len(dt.recent_within(days=7.0, limit=2)) == 2
    assert len(dt.recent(2)) == 2


def test_multiple_decisions_in_one_message(workspace):
    dt, _ = make_tracker(workspace)
    n = dt.process_message(
        "we decided to use postgres. also agreed the plan is to ship monthly."
    )
    assert n >= 1 and len(dt.decisions) == n
