"""Cortex auxiliary tables mirroring `test/noise-filter.test.ts` (9 its),
`narrative-generator.test.ts` (11), `pre-compaction.test.ts` (13) and
`tools.test.ts` (9): noise-topic rules, narrative content/persistence,
hot-snapshot window + tracker flush, and agent-tool filters.
"""

import os

import pytest

from vainplex_openclaw_amd.cortex.patterns import is_noise_topic
from vainplex_openclaw_amd.cortex.pre_compaction import NarrativeGenerator, PreCompaction


# -- noise filter -------------------------------------------------------------

@pytest.mark.parametrize("topic,noisy", [
    ("api", True),                                     # too short (<4)
    ("x", True),
    ("something", True),                               # single blacklisted word
    ("tomorrow", True),
    ("the it a", True),                                # all blacklisted/short
    ("it should work now", True),                      # pronoun prefix
    ("he said so", True),
    ("multi\nline topic", True),                       # newline
    ("x" * 61, True),                                  # overlong
    ("deployment pipeline", False),                    # valid
    ("database migration", False),
    ("auth service rollout", False),
])
def test_noise_topic_matrix(topic, noisy):
    assert is_noise_topic(topic) is noisy


def test_noise_topic_german():
    assert is_noise_topic("datenbank migration") is False
    assert is_noise_topic("nichts gepostet habe", language="de") is True


# -- narrative generator ------------------------------------------------------

def _threads(*specs):
    return [{"title": t, "status": s, **extra} for t, s, extra in specs]


def test_narrative_empty_workspace(tmp_path):
    gen = NarrativeGenerator(str(tmp_path), clock=lambda: 1_700_000_000.0)
    text = gen.generate([], [], "neutral")
    assert "mood is neutral" in text
    assert "thread" not in text                        # nothing to report


def test_narrative_sections_and_persistence(tmp_path):
    gen = NarrativeGenerator(str(tmp_path), clock=lambda: 1_700_000_000.0)
    threads = _threads(("deploy api", "open", {}),
                       ("fix tests", "open", {"waiting_for": "CI run"}),
                       ("old topic", "closed", {}))
    decisions = [{"what": "use postgres"}, {"what": "ship friday"}]
    text = gen.generate(threads, decisions, "focused")
    assert "mood is focused" in text
    assert "2 thread(s) remain open" in text and "deploy api" in text
    assert "1 thread(s) were closed" in text
    assert "Most recent decision: ship friday" in text
    assert "Blocked on: fix tests (CI run)" in text
    path = os.path.join(str(tmp_path), "memory", "reboot", "narrative.md")
    assert open(path).read() == text


def test_narrative_includes_date_header(tmp_path):
    gen = NarrativeGenerator(str(tmp_path), clock=lambda: 1_700_000_000.0)
    text = gen.generate([], [], "neutral")
    assert "As of 2023-11-" in text                    # 1.7e9 epoch => Nov 2023


# -- pre-compaction -----------------------------------------------------------

class _FakeThreads:
    session_mood = "tense"

    def __init__(self):
        self.flushed = 0

    def flush(self):
        self.flushed += 1

    def get_threads(self):
        return [{"title": "hotfix", "status": "open"}]


class _FakeDecisions:
    decisions = [{"what": "rollback"}]

    def __init__(self):
        self.flushed = 0

    def flush(self):
        self.flushed += 1


def test_pre_compaction_window_caps_and_snapshot(tmp_path):
    pc = PreCompaction(str(tmp_path), window=3, clock=lambda: 1_700_000_000.0)
    for i in range(6):
        pc.observe("user" if i % 2 == 0 else "agent", f"message {i}")
    pc.observe("user", "")                             # empty ignored
    assert len(pc.recent) == 3                         # rolling window
    snap = pc.run()
    assert "# Hot snapshot" in snap
    assert "message 5" in snap and "message 2" not in snap
    assert os.path.isfile(os.path.join(str(tmp_path), "memory", "reboot",
                                       "hot-snapshot.md"))


def test_pre_compaction_flushes_trackers_and_regenerates(tmp_path):
    pc = PreCompaction(str(tmp_path), clock=lambda: 1_700_000_000.0)
    th, de = _FakeThreads(), _FakeDecisions()
    pc.observe("user", "please fix prod")
    pc.run(thread_tracker=th, decision_tracker=de)
    assert th.flushed == 1 and de.flushed == 1
    reboot = os.path.join(str(tmp_path), "memory", "reboot")
    narrative = open(os.path.join(reboot, "narrative.md")).read()
    assert "mood is tense" in narrative and "hotfix" in narrative
    assert "rollback" in narrative
    assert os.path.isfile(os.path.join(reboot, "BOOT-CONTEXT.md"))


def test_pre_compaction_content_truncated_at_400(tmp_path):
    pc = PreCompaction(str(tmp_path))
    pc.observe("user", "x" * 1000)
    assert len(pc.recent[0][1]) == 400


# -- agent tools --------------------------------------------------------------

def _hooks(tmp_path):
    from vainplex_openclaw_amd.cortex.hooks import CortexHooks

    h = CortexHooks({}, str(tmp_path))
    for msg, sender in [
        ("we decided to deploy the payment api tomorrow", "user"),
        ("I will update the payment docs", "agent"),
    ]:
        h.ws().process_message(msg, sender)
    return h


def test_tools_threads_status_filter(tmp_path):
    h = _hooks(tmp_path)
    all_threads = h.tool_threads()
    open_only = h.tool_threads(status="open")
    assert all(t["status"] == "open" for t in open_only)
    assert len(open_only) <= len(all_threads)


def test_tools_search_hits_across_stores(tmp_path):
    h = _hooks(tmp_path)
    res = h.tool_search("payment")
    assert set(res) == {"threads", "decisions", "commitments"}
    assert any("payment" in str(d.get("what", "")).lower() for d in res["decisions"])
    assert any("payment" in str(c.get("action", "")).lower() for c in res["commitments"])
    empty = h.tool_search("zzz-no-such-topic")
    assert empty == {"threads": [], "decisions": [], "commitments": []}


def test_tools_status_shape(tmp_path):
    h = _hooks(tmp_path)
    st = h.tool_status()
    assert st["threads"]["open"] >= 0
    assert "mood" in st["threads"] and "decisions" in st
    assert set(st["commitments"]) == {"open", "overdue"}
