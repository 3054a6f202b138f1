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


# -- commitment patterns (commitment-patterns.test.ts, 11 its) ----------------

@pytest.mark.parametrize("text,lang", [
    ("I'll update the docs tomorrow", "en"),
    ("I will deploy the fix tonight", "en"),
    ("let me check the logs first", "en"),
    ("I promise to review it today", "en"),
    ("ich werde das morgen erledigen", "de"),
    ("mach ich", "de"),
    ("je vais corriger le bug", "fr"),
    ("lo haré mañana", "es"),
    ("pode deixar", "pt"),
    ("ci penso io", "it"),
    ("我会处理这个问题", "zh"),
    ("対応します", "ja"),
    ("제가 할게요", "ko"),
    ("я займусь этим", "ru"),
])
def test_commitment_detection_per_language(text, lang):
    from vainplex_openclaw_amd.cortex.commitment_tracker import detect_commitments

    hits = detect_commitments(text)
    assert hits and any(h["language"] == lang for h in hits)


@pytest.mark.parametrize("text", ["sounds good", "agreed", "", "the weather is nice"])
def test_commitment_no_false_positives(text):
    from vainplex_openclaw_amd.cortex.commitment_tracker import detect_commitments

    assert detect_commitments(text) == []


def test_commitment_patterns_cover_ten_languages():
    from vainplex_openclaw_amd.cortex.commitment_tracker import COMMITMENT_PATTERNS

    langs = {lang for _, lang in COMMITMENT_PATTERNS}
    assert langs >= {"en", "de", "fr", "es", "pt", "it", "zh", "ja", "ko", "ru"}


# -- LLM enhancer (llm-enhance.test.ts, 9 its) --------------------------------

def test_enhancer_disabled_returns_none():
    from vainplex_openclaw_amd.cortex.llm_enhance import LlmEnhancer

    e = LlmEnhancer({"enabled": False}, call_llm=lambda p: "{}")
    assert e.enabled is False
    assert e.add_message("anything") is None
    assert e.flush() is None


def test_enhancer_requires_call_llm():
    from vainplex_openclaw_amd.cortex.llm_enhance import LlmEnhancer

    e = LlmEnhancer({"enabled": True}, call_llm=None)
    assert e.enabled is False


def test_enhancer_buffers_until_batch_size():
    import json

    from vainplex_openclaw_amd.cortex.llm_enhance import LlmEnhancer

    calls = []

    def llm(prompt):
        calls.append(prompt)
        return json.dumps({"threads": [], "decisions": [], "closures": [],
                           "mood": "focused"})

    e = LlmEnhancer({"enabled": True, "batchSize": 3}, call_llm=llm)
    assert e.add_message("one") is None
    assert e.add_message("two") is None
    out = e.add_message("three")
    assert out is not None and out.get("mood") == "focused"
    assert len(calls) == 1
    assert "one" in calls[0] and "three" in calls[0]


def test_enhancer_flush_empty_and_error():
    from vainplex_openclaw_amd.cortex.llm_enhance import LlmEnhancer

    e = LlmEnhancer({"enabled": True}, call_llm=lambda p: "{}")
    assert e.flush() is None                            # nothing buffered

    def boom(p):
        raise ConnectionError("down")

    e2 = LlmEnhancer({"enabled": True, "batchSize": 1}, call_llm=boom)
    assert e2.add_message("msg") is None                # error swallowed


def test_enhancer_config_defaults_merge():
    from vainplex_openclaw_amd.cortex.llm_enhance import DEFAULT_CONFIG, LlmEnhancer

    e = LlmEnhancer({"endpoint": "http://cloud/v1"})
    assert e.config["endpoint"] == "http://cloud/v1"
    for key, val in DEFAULT_CONFIG.items():
        if key != "endpoint":
            assert e.config[key] == val
