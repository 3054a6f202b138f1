"""Cortex: patterns, trackers, boot context, pre-compaction, hooks, demo."""

import json
import os

import pytest

from vainplex_openclaw_amd.cortex import patterns as P
from vainplex_openclaw_amd.cortex.boot_context import BootContextGenerator, execution_mode
from vainplex_openclaw_amd.cortex.commitment_tracker import CommitmentTracker, detect_commitments
from vainplex_openclaw_amd.cortex.decision_tracker import DecisionTracker, infer_impact
from vainplex_openclaw_amd.cortex.demo import run_demo
from vainplex_openclaw_amd.cortex.hooks import CortexHooks, CortexPlugin, CortexWorkspace
from vainplex_openclaw_amd.cortex.pre_compaction import PreCompaction
from vainplex_openclaw_amd.cortex.thread_tracker import ThreadTracker, extract_signals, matches_thread
from vainplex_openclaw_amd.core.gateway import Gateway


def test_pattern_registry_langs():
    assert len(P.language_codes()) == 10
    for code in P.language_codes():
        reg = P.get_registry(code)
        assert reg.get_patterns("decision"), code
        assert reg.moods, code


def test_detect_mood_last_match_wins():
    assert P.detect_mood("this sucks but now it works, awesome") in ("excited", "productive")
    # "awesome" (excited) is the last match
    assert P.detect_mood("it works. awesome") == "excited"
    assert P.detect_mood("plain text") == "neutral"
    assert P.detect_mood("das ist kaputt und nervt") == "frustrated"


def test_noise_topic():
    assert P.is_noise_topic("it")
    assert P.is_noise_topic("xy")
    assert P.is_noise_topic("something here" * 20)  # too long
    assert not P.is_noise_topic("database migration")


def test_extract_signals_en_de():
    s = extract_signals("We decided to use postgres. Waiting for ops approval. Let's talk about the billing rewrite")
    assert s["decisions"] and s["waits"] and s["topics"]
    assert any("billing rewrite" in t for t in s["topics"])
    s2 = extract_signals("Das ist erledigt ✅")
    assert s2["closures"]


def test_matches_thread_overlap():
    t = {"title": "database migration plan"}
    assert matches_thread(t, "the migration of the database is ready")
    assert not matches_thread(t, "completely unrelated text")


def test_matches_thread_reference_cases():
    """Mirrors reference thread-tracker.test.ts matching block."""
    t = {"title": "database migration plan"}
    # 2+ title words in text -> match; 1 word -> no match
    assert matches_thread(t, "the database migration starts tomorrow")
    assert not matches_thread(t, "the database is slow")
    # case-insensitive
    assert matches_thread(t, "DATABASE MIGRATION done")
    # words <3 chars ignored on both sides
    assert not matches_thread({"title": "go to it"}, "go to it now yes")
    # custom min_overlap
    assert matches_thread(t, "the database is slow", min_overlap=1)
    # degenerate inputs never match
    assert not matches_thread({"title": ""}, "anything at all")
    assert not matches_thread(t, "")


def test_thread_tracker_lifecycle(workspace):
    tt = ThreadTracker(workspace)
    tt.process_message("Let's talk about the payment gateway integration", "user")
    threads = tt.get_threads()
    assert len(threads) == 1
    assert threads[0]["status"] == "open"
    tt.process_message("the payment gateway integration is done", "agent")
    assert tt.get_threads()[0]["status"] == "closed"
    # persisted v2 format with integrity block
    with open(os.path.join(workspace, "memory", "reboot", "threads.json")) as fh:
        data = json.load(fh)
    assert data["version"] == 2
    assert data["integrity"]["events_processed"] == 2
    assert "session_mood" in data


def test_thread_priority_and_waits(workspace):
    tt = ThreadTracker(workspace)
    tt.process_message("let's discuss the production security audit", "user")
    t = tt.get_threads()[0]
    assert t["priority"] == "high"  # "production"/"security" keywords
    tt.process_message("the security audit is waiting for the vendor report", "user")
    assert tt.get_threads()[0]["waiting_for"]


def test_thread_prune_and_cap(workspace):
    t = [1_000_000.0]
    from vainplex_openclaw_amd.cortex.thread_tracker import ThreadTrackerConfig

    tt = ThreadTracker(workspace, ThreadTrackerConfig(max_threads=5, prune_days=1), clock=lambda: t[0])
    for i in range(8):
        tt.threads.append({
            "id": str(i), "title": f"topic {i} alpha beta", "status": "closed",
            "priority": "medium", "summary": "", "decisions": [], "waiting_for": None,
            "mood": "neutral", "last_activity": "1970-01-10T00:00:00Z", "created": "1970-01-10T00:00:00Z",
        })
    tt.process_message("let's talk about the fresh new subject", "user")
    assert len(tt.get_threads()) <= 5


def test_decision_tracker_dedupe(workspace):
    t = [1000.0]
    dt = DecisionTracker(workspace, clock=lambda: t[0])
    assert dt.process_message("We decided to adopt kubernetes for deploys", "user") == 1
    assert dt.process_message("We decided to adopt kubernetes for deploys", "user") == 0  # dup in window
    t[0] += 25 * 3600
    assert dt.process_message("We decided to adopt kubernetes for deploys", "user") == 1
    assert infer_impact("delete the production database") == "high"
    assert infer_impact("rename a variable") == "medium"


def test_commitment_tracker(workspace):
    t = [1000.0]
    ct = CommitmentTracker(workspace, clock=lambda: t[0])
    assert ct.process_message("I'll send the report tomorrow", "agent") >= 1
    assert ct.process_message("Ich kümmere mich um das Backup heute", "agent") >= 1
    assert len(ct.open_commitments()) >= 2
    assert ct.overdue() == []
    t[0] += 8 * 86400
    assert len(ct.overdue()) >= 2
    cid = ct.open_commitments()[0]["id"]
    assert ct.complete(cid)
    ct.flush()
    with open(os.path.join(workspace, "memory", "reboot", "commitments.json")) as fh:
        data = json.load(fh)
    assert any(c["status"] == "done" for c in data["commitments"])
    assert detect_commitments("consider it done")


def test_boot_context(workspace):
    ws = CortexWorkspace(workspace, {})
    ws.process_message("let's talk about the quarterly planning review", "user")
    ws.process_message("We decided to push the release to friday", "user")
    ws.process_message("I'll draft the announcement", "agent")
    ws.commitments.flush()  # boot context reads persisted state
    ctx = ws.boot.generate()
    assert "BOOT CONTEXT" in ctx
    assert "quarterly planning review" in ctx
    assert "Open commitments" in ctx
    assert os.path.isfile(os.path.join(workspace, "memory", "reboot", "BOOT-CONTEXT.md"))
    assert execution_mode(10) == "business-hours"
    assert execution_mode(23) == "night-watch"
    assert execution_mode(19) == "evening"


def test_pre_compaction_snapshot(workspace):
    ws = CortexWorkspace(workspace, {})
    for i in range(5):
        ws.process_message(f"message number {i} about the data pipeline", "user")
    snap = ws.pre_compaction.run(ws.threads, ws.decisions, ws.commitments)
    assert "Hot snapshot" in snap
    d = os.path.join(workspace, "memory", "reboot")
    assert os.path.isfile(os.path.join(d, "hot-snapshot.md"))
    assert os.path.isfile(os.path.join(d, "narrative.md"))
    assert os.path.isfile(os.path.join(d, "BOOT-CONTEXT.md"))


def test_cortex_plugin_hooks(workspace):
    gw = Gateway(config={})
    plugin = CortexPlugin(workspace)
    gw.load(plugin, plugin_config={"language": "both"})
    gw.start()
    gw.emit("message_received", {"content": "Let's talk about the search indexing rework"})
    gw.emit("message_sent", {"content": "I'll benchmark the tokenizer first"})
    out = gw.emit("session_start", {})
    assert "bootContext" in out
    gw.emit("before_compaction", {})
    status = gw.command("cortexstatus")
    assert status["threads"]["open"] >= 1
    assert status["commitments"]["open"] >= 1
    found = gw.command("cortex.search", "indexing")
    assert found["threads"]
    gw.stop()


def test_demo_runs():
    out = run_demo()
    st = out["status"]
    assert st["threads"]["open"] + st["threads"]["closed"] >= 1
    assert st["decisions"] >= 1
    assert st["commitments"]["open"] >= 1
    assert "BOOT CONTEXT" in out["boot_context"]


def test_llm_enhance_parse_and_merge(workspace):
    from vainplex_openclaw_amd.cortex.llm_enhance import LlmEnhancer, parse_analysis

    calls = []

    def fake_llm(prompt):
        calls.append(prompt)
        return 'noise {"threads": [{"title": "vector search rollout", "status": "open"}], "closures": [], "decisions": [], "mood": "productive"} trailing'

    enh = LlmEnhancer({"enabled": True, "batchSize": 2}, call_llm=fake_llm)
    assert enh.add_message("first") is None
    analysis = enh.add_message("second")
    assert analysis and analysis["threads"][0]["title"] == "vector search rollout"
    tt = ThreadTracker(workspace)
    tt.apply_llm_analysis(analysis)
    assert any(t["title"] == "vector search rollout" for t in tt.get_threads())
    assert parse_analysis("not json") is None


def test_thread_tracker_dedupe_mood_and_decisions(workspace):
    tt = ThreadTracker(workspace)
    tt.process_message("let's talk about the billing system rework", "user")
    tt.process_message("more thoughts regarding the billing system rework", "user")
    assert len(tt.get_threads()) == 1  # same topic -> no duplicate
    # decision appended to the matching thread
    tt.process_message("we decided the billing system rework ships in Q2", "user")
    th = tt.get_threads()[0]
    assert th.get("decisions"), th
    # mood propagates to matching threads and the session
    tt.process_message("the billing system rework is broken and annoying", "user")
    assert tt.session_mood == "frustrated"
    assert tt.get_threads()[0].get("mood") == "frustrated"
    assert tt.events_processed == 4
    # empty content skipped
    tt.process_message("", "user")
    assert tt.events_processed == 4


def test_thread_tracker_loads_existing_state(workspace):
    tt = ThreadTracker(workspace)
    tt.process_message("let's talk about the observability dashboard", "user")
    tt.flush()
    tt2 = ThreadTracker(workspace)
    assert any("observability" in t["title"] for t in tt2.get_threads())
    assert tt2.session_mood in ("neutral", "exploratory", "productive")


def test_extract_signals_families():
    s = extract_signals("we decided to use rust. the migration is done. "
                        "waiting for the security review. let's talk about budgets.")
    assert s["decisions"] and s["closures"] and s["waits"] and s["topics"]
    assert extract_signals("nothing interesting here")["decisions"] == []


def test_custom_patterns_via_workspace_config(workspace):
    try:
        ws = CortexWorkspace(workspace, {"customPatterns": {
            "decision": ["committee greenlit"], "mode": "extend"}})
        ws.process_message("the committee greenlit the rollout", "user")
        assert ws.decisions.decisions, "custom decision pattern should fire"
    finally:
        P.set_custom_patterns(None)


# -- commitment-tracker.test.ts depth ---------------------------------------

def _ct(workspace, **cfg):
    from vainplex_openclaw_amd.cortex.commitment_tracker import (
        CommitmentTracker,
        CommitmentTrackerConfig,
    )

    t = [1_700_000_000.0]
    return CommitmentTracker(workspace, CommitmentTrackerConfig(**cfg),
                             clock=lambda: t[0]), t


def test_commitment_detection_and_what_capture(workspace):
    ct, _ = _ct(workspace)
    n = ct.process_message("I'll send the report by friday", "agent")
    assert n == 1
    c = ct.open_commitments()[0]
    assert "report" in c["action"]
    assert c["status"] == "open" and c["by"] == "agent"
    assert c["language"] == "en" and c["completed"] is None


def test_commitment_dedupe_same_message(workspace):
    ct, _ = _ct(workspace)
    ct.process_message("I'll update the docs. I'll update the docs.", "agent")
    assert len(ct.open_commitments()) == 1


def test_commitment_non_commitment_messages(workspace):
    ct, _ = _ct(workspace)
    assert ct.process_message("the weather is nice", "agent") == 0
    assert ct.process_message("", "agent") == 0
    assert ct.open_commitments() == []


def test_commitment_overdue_marking(workspace):
    ct, t = _ct(workspace)
    ct.process_message("I'll fix the flaky test tomorrow", "agent")
    assert ct.overdue() == []                 # fresh: not overdue
    t[0] += 8 * 86400                          # past the 7-day window
    od = ct.overdue()
    assert len(od) == 1 and od[0]["status"] in ("open", "overdue")
    # recent commitment not flagged
    ct.process_message("I'll write the changelog", "agent")
    assert len(ct.overdue()) == 1


def test_commitment_done_not_overdue(workspace):
    ct, t = _ct(workspace)
    ct.process_message("I'll deploy the fix", "agent")
    cid = ct.open_commitments()[0]["id"]
    assert ct.complete(cid)
    assert not ct.complete("ghost")
    t[0] += 30 * 86400
    assert ct.overdue() == []


def test_commitment_flush_roundtrip(workspace):
    import json
    import os

    ct, _ = _ct(workspace)
    ct.process_message("I'll benchmark the kernel", "agent")
    ct.flush()
    path = os.path.join(workspace, "memory", "reboot", "commitments.json")
    data = json.load(open(path))
    blob = json.dumps(data)
    assert "benchmark" in blob
    ct2, _ = _ct(workspace)
    assert any("benchmark" in c["action"] for c in ct2.open_commitments())


def test_cortex_plugin_layered_config(tmp_path, monkeypatch):
    """config-loader.ts semantics through the cortex plugin: configPath
    pointer + inline enabled, legacy inline config wins outright."""
    import json

    from vainplex_openclaw_amd.core.api import HookBus, NullLogger, PluginApi
    from vainplex_openclaw_amd.cortex.hooks import CortexPlugin

    monkeypatch.setenv("OPENCLAW_HOME", str(tmp_path))
    ext = tmp_path / "cortex.json"
    ext.write_text(json.dumps({"workspace": str(tmp_path),
                               "threadTracker": {"maxThreads": 7}}))
    api = PluginApi(id="openclaw-cortex", plugin_config={"configPath": str(ext)},
                    logger=NullLogger(), config={}, bus=HookBus())
    p = CortexPlugin(str(tmp_path))
    p.register(api)
    assert p.hooks is not None
    ws = p.hooks.ws(str(tmp_path))
    assert ws.threads.config.max_threads == 7
