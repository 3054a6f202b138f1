"""Trace analyzer: events, chains, 7 detectors, classifier, outputs, run()."""

import pytest

from vainplex_openclaw_amd.cortex.trace.analyzer import (
    ChainRedactor,
    FindingClassifier,
    MockTraceSource,
    TraceAnalyzer,
    generate_outputs,
)
from vainplex_openclaw_amd.cortex.trace.chains import reconstruct_chains
from vainplex_openclaw_amd.cortex.trace.events import (
    NormalizedEvent,
    normalize_schema_a,
    normalize_schema_b,
)
from vainplex_openclaw_amd.cortex.trace.signals import (
    detect_all_signals,
    jaccard_similarity,
    levenshtein,
)

_seq = [0]
_ts = [1_000_000.0]


def ev(etype, agent="main", session="s1", gap_ms=1000, **payload):
    _seq[0] += 1
    _ts[0] += gap_ms
    return NormalizedEvent(
        id=f"e{_seq[0]}", ts=_ts[0], agent=agent, session=session,
        type=etype, payload=payload, seq=_seq[0],
    )


def tool_fail_pair(cmd="make build", err="exit 1", tool="exec"):
    return [
        ev("tool.call", toolName=tool, toolParams={"command": cmd}),
        ev("tool.result", toolName=tool, toolError=err, toolIsError=True),
    ]


def test_normalize_schema_a_and_b():
    env = {
        "id": "evt-1", "type": "tool.call.requested", "ts": 123,
        "actor": {"id": "forge"}, "scope": {"sessionKey": "agent:forge:abc"},
        "data": {"toolName": "exec", "params": {"command": "ls"}},
    }
    ne = normalize_schema_a(env)
    assert ne.type == "tool.call" and ne.agent == "forge" and ne.session == "abc"
    assert ne.payload["toolName"] == "exec"
    nb = normalize_schema_b({"kind": "user_message", "ts": 5, "agentId": "m", "sessionId": "x", "body": {"text": "hi"}})
    assert nb.type == "msg.in" and nb.payload["content"] == "hi"
    assert normalize_schema_a({"type": "unknown.kind"}) is None


def test_chain_reconstruction_splits():
    events = []
    events.append(ev("msg.in", content="start"))
    events.append(ev("msg.out", content="ok"))
    events.append(ev("msg.in", content="later", gap_ms=31 * 60 * 1000))  # 31-min gap
    events.append(ev("session.start", gap_ms=100))  # lifecycle split
    events.append(ev("msg.in", content="fresh"))
    # different agent bucket
    events.append(ev("msg.in", agent="other", content="x"))
    chains = reconstruct_chains(events)
    # 1-event chains ("later" alone, "other" agent, post-lifecycle tail)
    # are dropped per the reference's min-2-events rule
    assert len(chains) == 2
    assert all(len(c.events) >= 2 for c in chains)
    # with the filter disabled the old buckets reappear
    raw = reconstruct_chains(events, min_events=1)
    assert ("other", 1) in [(c.agent, len(c.events)) for c in raw]
    # dedupe by id
    dup = events[0]
    chains2 = reconstruct_chains(events + [dup])
    assert sum(len(c.events) for c in chains2) == sum(len(c.events) for c in chains)


def test_similarity_helpers():
    assert jaccard_similarity({"a": 1, "timeout": 5}, {"a": 1, "timeout": 9}) == 1.0
    assert jaccard_similarity({"a": 1}, {"a": 2}) == 0.0
    assert levenshtein("kitten", "sitting") == 3
    assert levenshtein("same", "same") == 0


def test_doom_loop_detector():
    events = [ev("msg.in", content="please build")]
    for _ in range(4):
        events.extend(tool_fail_pair("make build -j8"))
    chains = reconstruct_chains(events)
    findings = detect_all_signals(chains, ["doom_loop"])
    assert len(findings) == 1
    assert findings[0].signal_type == "doom_loop"
    assert findings[0].evidence["count"] == 4
    # dissimilar commands break the loop
    events2 = [ev("msg.in", content="x")]
    events2.extend(tool_fail_pair("make build"))
    events2.extend(tool_fail_pair("rm -rf /tmp/cache && fetch artifacts --all"))
    events2.extend(tool_fail_pair("curl http://somewhere/else/entirely"))
    assert detect_all_signals(reconstruct_chains(events2), ["doom_loop"]) == []


def test_correction_and_dissatisfied():
    events = [
        ev("msg.in", content="deploy please"),
        ev("msg.out", content="deployed to prod-a"),
        ev("msg.in", content="no, that's wrong — I wanted prod-b"),
        ev("msg.in", content="this is not helpful at all, useless"),
    ]
    chains = reconstruct_chains(events)
    types = {f.signal_type for f in detect_all_signals(chains, ["correction", "dissatisfied"])}
    assert types == {"correction", "dissatisfied"}


def test_tool_fail_and_repeat_fail():
    events = [ev("msg.in", content="go")]
    events.extend(tool_fail_pair("a b c", err="timeout"))
    events.append(ev("tool.call", toolName="read", toolParams={"p": 1}))
    events.append(ev("tool.result", toolName="read", toolResult="ok"))
    events.extend(tool_fail_pair("x y z", err="timeout"))
    events.extend(tool_fail_pair("q r s", err="timeout"))
    chains = reconstruct_chains(events)
    findings = detect_all_signals(chains, ["tool_fail", "repeat_fail"])
    types = [f.signal_type for f in findings]
    assert "tool_fail" in types
    assert "repeat_fail" in types


def test_hallucination_and_unverified():
    events = [
        ev("msg.out", content="nginx is running and healthy, all good"),
        ev("tool.call", toolName="exec", toolParams={"command": "systemctl status nginx"}),
        ev("tool.result", toolName="exec", toolError="nginx: service not found", toolIsError=True),
        ev("msg.out", content="This will definitely work, 100% guaranteed."),
    ]
    chains = reconstruct_chains(events)
    types = {f.signal_type for f in detect_all_signals(chains, ["hallucination", "unverified_claim"])}
    assert "hallucination" in types
    assert "unverified_claim" in types


def test_redactor_strips_credentials():
    events = [ev("msg.in", content="token ghp_" + "f" * 36 + " here"),
              ev("msg.out", content="noted")]
    chains = reconstruct_chains(events)
    red = ChainRedactor().redact_chain(chains[0])
    assert "ghp_" + "f" * 36 not in str(red)


def test_classifier_with_llm_triage():
    events = [ev("msg.in", content="x")]
    for _ in range(3):
        events.extend(tool_fail_pair())
    chains = reconstruct_chains(events)
    findings = detect_all_signals(chains, ["doom_loop"])

    def fake_llm(prompt):
        if "triage" in prompt.lower() or '"keep"' in prompt:
            return '{"keep": true, "severity": "high"}'
        return '{"rootCause": "stale cache", "actionText": "clear the cache before retrying", "confidence": 0.9}'

    out = FindingClassifier(fake_llm).classify(findings, {c.id: c for c in chains})
    assert out[0]["actionText"] == "clear the cache before retrying"
    assert out[0]["confidence"] == 0.9
    # no-LLM path uses default action text
    out2 = FindingClassifier(None).classify(findings, {})
    assert out2[0]["actionText"]


def test_output_grouping():
    classified = [
        {"id": "1", "signalType": "doom_loop", "agent": "a", "actionText": "Stop retrying X", "confidence": 0.8},
        {"id": "2", "signalType": "doom_loop", "agent": "b", "actionText": "stop retrying  x", "confidence": 0.7},
        {"id": "3", "signalType": "hallucination", "agent": "a", "actionText": "Verify before asserting", "confidence": 0.6},
    ]
    outputs = generate_outputs(classified)
    assert len(outputs) == 2
    gp = next(o for o in outputs if o["kind"] == "governance_policy")
    assert gp["occurrences"] == 2 and set(gp["agents"]) == {"a", "b"}
    assert any(o["kind"] == "soul_rule" for o in outputs)


def test_analyzer_run_and_incremental_state(workspace):
    events = [ev("msg.in", content="go")]
    for _ in range(3):
        events.extend(tool_fail_pair())
    source = MockTraceSource(events)
    analyzer = TraceAnalyzer(workspace, source)
    report = analyzer.run()
    assert report["eventsAnalyzed"] == len(events)
    assert report["findings"]
    assert report["outputs"]
    assert analyzer.state["runsCompleted"] == 1
    # second run resumes from state (context window re-reads the tail only)
    analyzer2 = TraceAnalyzer(workspace, source)
    report2 = analyzer2.run()
    assert analyzer2.state["runsCompleted"] == 2
    assert report2["eventsAnalyzed"] <= len(events)


def test_trace_to_facts_bridge(workspace):
    from vainplex_openclaw_amd.governance.facts import FactRegistry
    from vainplex_openclaw_amd.governance.trace_to_facts import apply_report_to_registry

    report = {"findings": [
        {"id": "f1", "signalType": "hallucination", "confidence": 0.7,
         "evidence": {"subject": "nginx", "predicate": "state", "value": "error"}},
        {"id": "f2", "signalType": "correction", "confidence": 0.9, "evidence": {}},
    ]}
    reg = FactRegistry([])
    n = apply_report_to_registry(report, reg)
    assert n == 1
    assert reg.lookup("nginx", "state")["value"] == "error"


SIGNAL_I18N_SAMPLES = {
    "de": ("nein, das ist falsch", "das hilft nicht, frustrierend"),
    "es": ("no, eso está mal", "inútil, esto no ayuda"),
    "fr": ("non, c'est faux", "inutile, ça n'aide pas"),
    "it": ("no, è sbagliato", "inutile, non aiuta"),
    "pt": ("não, está errado", "inútil, isso não ajuda"),
    "ru": ("нет, это неверно", "бесполезно, это не помогает"),
    "ja": ("違います、やり直して", "役に立たない"),
    "ko": ("아니요, 틀렸습니다", "쓸모없어요"),
    "zh": ("不对，这是错的", "没用，这没有帮助"),
}


@pytest.mark.parametrize("lang", sorted(SIGNAL_I18N_SAMPLES))
def test_correction_and_dissatisfied_i18n(lang):
    """Localized phrase packs fire in all 10 languages
    (signals/lang/signal-lang-*.ts parity)."""
    corr, dissat = SIGNAL_I18N_SAMPLES[lang]
    events = [
        ev("msg.out", content="here is the result"),
        ev("msg.in", content=corr),
        ev("msg.out", content="another answer"),
        ev("msg.in", content=dissat),
    ]
    chains = reconstruct_chains(events)
    kinds = {f.signal_type for f in detect_all_signals(chains, ["correction", "dissatisfied"])}
    assert "correction" in kinds, lang
    assert "dissatisfied" in kinds, lang


def test_chain_run_boundary_split_and_metadata():
    """run.end -> run.start with >5 min gap splits; <5 min does not;
    boundary_type and type_counts mirror chain-reconstructor.ts."""
    events = [
        ev("msg.in", content="a"),
        ev("run.ended", gap_ms=1000),
        ev("run.started", gap_ms=6 * 60 * 1000),  # >5 min after run end
        ev("msg.out", content="b", gap_ms=1000),
    ]
    chains = reconstruct_chains(events)
    assert len(chains) == 2
    assert chains[0].type_counts == {"msg.in": 1, "run.ended": 1}
    assert chains[0].boundary_type == "gap"

    # <5 min: stays one chain
    events2 = [
        ev("msg.in", content="a"),
        ev("run.ended", gap_ms=1000),
        ev("run.started", gap_ms=2 * 60 * 1000),
        ev("msg.out", content="b", gap_ms=1000),
    ]
    assert len(reconstruct_chains(events2)) == 1

    # lifecycle boundary marks the chain
    events3 = [
        ev("session.start"),
        ev("msg.in", content="x", gap_ms=100),
        ev("msg.out", content="y", gap_ms=100),
    ]
    c3 = reconstruct_chains(events3)
    assert c3[0].boundary_type == "lifecycle"


def test_chain_configurable_gap_and_cap():
    events = [ev("msg.in", content="a"), ev("msg.out", content="b", gap_ms=10 * 60 * 1000),
              ev("msg.in", content="c", gap_ms=10 * 60 * 1000),
              ev("msg.out", content="d", gap_ms=10 * 60 * 1000)]
    assert len(reconstruct_chains(events)) == 1  # default 30-min gap
    small = reconstruct_chains(events, gap_minutes=5.0)
    assert len(small) == 0 or all(len(c.events) >= 2 for c in small)
    # cap: 5 events with max_events=2 -> 2-event chains
    burst = [ev("msg.in", content=str(i), gap_ms=10) for i in range(6)]
    capped = reconstruct_chains(burst, max_events=2)
    assert all(len(c.events) == 2 for c in capped)
    assert sum(len(c.events) for c in capped) == 6


def test_chain_deterministic_ids_and_empty():
    assert reconstruct_chains([]) == []
    events = [ev("msg.in", content="a"), ev("msg.out", content="b", gap_ms=50)]
    a = reconstruct_chains(list(events))
    b = reconstruct_chains(list(events))
    assert a[0].id == b[0].id and len(a[0].id) == 16


def test_report_aggregates_and_incremental_totals(workspace):
    """report.test.ts mirrors: stats, signalStats, topAgents, timeRange,
    incremental totals across runs."""
    events = [ev("msg.in", content="x")]
    for _ in range(3):
        events.extend(tool_fail_pair())
    src = MockTraceSource(events)
    an = TraceAnalyzer(workspace, src)
    rep = an.run()
    assert rep["stats"]["events"] == len(events)
    assert rep["stats"]["chains"] == rep["chains"]
    assert rep["stats"]["findings"] == len(rep["findings"])
    assert rep["stats"]["findingsClassified"] <= rep["stats"]["findings"]
    assert sum(rep["signalStats"].values()) == len(rep["findings"])
    for sig, agents in rep["topAgents"].items():
        assert sig in rep["signalStats"]
        assert agents and agents[0]["count"] >= agents[-1]["count"]
    assert rep["timeRange"] is None or rep["timeRange"]["end"] >= rep["timeRange"]["start"]
    # totals accumulate
    t1 = an.state["totalEventsAnalyzed"]
    an.run()
    assert an.state["totalEventsAnalyzed"] >= t1
    assert an.state["runsCompleted"] == 2
    # empty stream handled
    empty = TraceAnalyzer(workspace, MockTraceSource([]))
    rep2 = empty.run()
    assert rep2["stats"]["findings"] == 0 and rep2["timeRange"] is None


def test_resolve_analyzer_llm_config():
    from vainplex_openclaw_amd.cortex.trace.analyzer import resolve_analyzer_llm_config

    TOP = {"endpoint": "http://localhost:11434/v1", "model": "mistral:7b",
           "apiKey": "", "timeoutMs": 15000}
    assert resolve_analyzer_llm_config(TOP, {"enabled": False})["enabled"] is False
    assert resolve_analyzer_llm_config(TOP, None)["enabled"] is False
    r = resolve_analyzer_llm_config(TOP, {"enabled": True})
    assert r == {"enabled": True, "endpoint": "http://localhost:11434/v1",
                 "model": "mistral:7b", "apiKey": "", "timeoutMs": 15000}
    r2 = resolve_analyzer_llm_config(TOP, {"enabled": True, "endpoint": "http://c/v1",
                                           "model": "gpt-4o", "apiKey": "sk-test",
                                           "timeoutMs": 30000})
    assert r2["endpoint"] == "http://c/v1" and r2["model"] == "gpt-4o"
    assert r2["apiKey"] == "sk-test" and r2["timeoutMs"] == 30000
    r3 = resolve_analyzer_llm_config(TOP, {"enabled": True, "model": "gpt-4o"})
    assert r3["model"] == "gpt-4o" and r3["endpoint"] == TOP["endpoint"]
    assert r3["apiKey"] == ""


# ===========================================================================
# chain-reconstructor.test.ts + events.test.ts depth tables
# ===========================================================================

def _nev(i, ts, agent="a1", session="s1", etype="msg.in", payload=None):
    from vainplex_openclaw_amd.cortex.trace.events import NormalizedEvent

    return NormalizedEvent(id=f"e{i}", ts=ts, agent=agent, session=session,
                           type=etype, payload=payload or {}, seq=i)


def _rc(events, **kw):
    from vainplex_openclaw_amd.cortex.trace.chains import reconstruct_chains

    return reconstruct_chains(events, **kw)


MIN = 60_000.0  # one minute in ms


def test_chains_group_by_session_and_agent():
    evs = [
        _nev(1, 1 * MIN, agent="a1", session="s1"),
        _nev(2, 2 * MIN, agent="a1", session="s1"),
        _nev(3, 1 * MIN, agent="a2", session="s1"),
        _nev(4, 2 * MIN, agent="a2", session="s1"),
        _nev(5, 1 * MIN, agent="a1", session="s2"),
        _nev(6, 2 * MIN, agent="a1", session="s2"),
    ]
    chains = _rc(evs)
    keys = {(c.session, c.agent) for c in chains}
    assert keys == {("s1", "a1"), ("s1", "a2"), ("s2", "a1")}


def test_chain_orders_events_by_ts():
    evs = [_nev(1, 5 * MIN), _nev(2, 1 * MIN), _nev(3, 3 * MIN)]
    chains = _rc(evs)
    assert len(chains) == 1
    assert [e.ts for e in chains[0].events] == [1 * MIN, 3 * MIN, 5 * MIN]


@pytest.mark.parametrize("split_type", ["session.start", "session.end"])
def test_chain_splits_on_lifecycle(split_type):
    evs = [
        _nev(1, 1 * MIN), _nev(2, 2 * MIN),
        _nev(3, 3 * MIN, etype=split_type),
        _nev(4, 4 * MIN),
    ]
    chains = _rc(evs)
    assert len(chains) == 2
    assert chains[1].boundary_type == "lifecycle"


def test_chain_gap_split_boundary_exact():
    evs = [_nev(1, 0.0), _nev(2, 29 * MIN), _nev(3, 29 * MIN + 29 * MIN),
           _nev(4, 29 * MIN + 29 * MIN + 31 * MIN),
           _nev(5, 29 * MIN + 29 * MIN + 32 * MIN)]
    chains = _rc(evs)
    # <=30 min gaps keep one chain; the 31-min gap splits
    assert len(chains) == 2
    assert chains[0].boundary_type == "gap"


def test_chain_min_two_events_filter():
    evs = [_nev(1, 0.0), _nev(2, 40 * MIN), _nev(3, 41 * MIN)]
    chains = _rc(evs)
    # the singleton before the gap is dropped
    assert len(chains) == 1 and len(chains[0].events) == 2


def test_chain_id_deterministic_16_hex():
    from vainplex_openclaw_amd.cortex.trace.chains import chain_id

    a = chain_id("s1", "a1", 12345.0)
    assert a == chain_id("s1", "a1", 12345.0)
    assert len(a) == 16 and all(c in "0123456789abcdef" for c in a)
    assert a != chain_id("s1", "a2", 12345.0)
    assert a != chain_id("s1", "a1", 99999.0)


def test_chain_type_counts_and_ts_range():
    evs = [_nev(1, 1 * MIN), _nev(2, 2 * MIN, etype="tool.call"),
           _nev(3, 3 * MIN, etype="tool.call")]
    c = _rc(evs)[0]
    assert c.type_counts == {"msg.in": 1, "tool.call": 2}
    assert c.start_ts == 1 * MIN and c.end_ts == 3 * MIN


def test_chain_cap_max_events():
    evs = [_nev(i, i * 1000.0) for i in range(50)]
    chains = _rc(evs, max_events=10)
    assert all(len(c.events) <= 10 for c in chains)
    assert sum(len(c.events) for c in chains) == 50


def test_chain_empty_and_unknown_session():
    assert _rc([]) == []
    evs = [_nev(1, 1 * MIN, session="unknown"), _nev(2, 2 * MIN, session="unknown")]
    chains = _rc(evs)
    assert len(chains) == 1 and chains[0].session == "unknown"


def test_chain_dedupes_same_event_id():
    e1 = _nev(1, 1 * MIN)
    e1b = _nev(1, 1 * MIN)  # same id
    e2 = _nev(2, 2 * MIN)
    chains = _rc([e1, e1b, e2])
    assert len(chains[0].events) == 2


def test_chain_interleaved_agents():
    evs = []
    for i in range(6):
        evs.append(_nev(10 + i, i * MIN, agent="a1"))
        evs.append(_nev(20 + i, i * MIN + 1000, agent="a2"))
    chains = _rc(evs)
    assert len(chains) == 2
    for c in chains:
        assert all(e.agent == c.agent for e in c.events)


# -- events.test.ts depth ----------------------------------------------------

def _schema_a(etype, **kw):
    ev = {"id": "evt-1", "ts": 1000.0, "agent": "main",
          "actor": {"id": "main"}, "scope": {"sessionKey": "agent:main:abc123"},
          "type": etype, "payload": kw.pop("payload", {})}
    ev.update(kw)
    return ev


def test_schema_a_session_uuid_extraction():
    from vainplex_openclaw_amd.cortex.trace.events import normalize_schema_a

    ev = normalize_schema_a(_schema_a("message.in.received"))
    assert ev is not None and ev.session == "abc123"
    # plain session strings pass through
    ev2 = normalize_schema_a({"id": "e", "ts": 1.0, "type": "message.in.received",
                              "session": "plain", "payload": {}})
    assert ev2.session == "plain"


def test_schema_a_msg_roles():
    from vainplex_openclaw_amd.cortex.trace.events import normalize_schema_a

    assert normalize_schema_a(_schema_a("message.in.received")).payload["role"] == "user"
    out = normalize_schema_a(_schema_a("message.out.sent"))
    assert out.payload["role"] == "assistant"


def test_schema_a_tool_call_and_result_payloads():
    from vainplex_openclaw_amd.cortex.trace.events import normalize_schema_a

    call = normalize_schema_a(_schema_a(
        "tool.call.requested", payload={"toolName": "exec", "params": {"cmd": "ls"}}))
    assert call.payload["toolName"] == "exec"
    assert call.payload["toolParams"] == {"cmd": "ls"}
    res = normalize_schema_a(_schema_a(
        "tool.call.failed", payload={"toolName": "exec", "error": "boom"}))
    assert res.payload["toolIsError"] is True
    ok = normalize_schema_a(_schema_a(
        "tool.call.executed", payload={"toolName": "exec", "result": "fine"}))
    assert ok.payload["toolIsError"] is False


def test_schema_a_unknown_type_is_none():
    from vainplex_openclaw_amd.cortex.trace.events import normalize_schema_a

    assert normalize_schema_a(_schema_a("totally.unknown.type")) is None


def test_schema_a_agent_defaults_unknown():
    from vainplex_openclaw_amd.cortex.trace.events import normalize_schema_a

    ev = normalize_schema_a({"id": "e", "ts": 5.0,
                             "type": "message.in.received", "payload": {}})
    assert ev.agent == "unknown" and ev.session == "unknown"


def test_schema_b_kinds_and_fields():
    from vainplex_openclaw_amd.cortex.trace.events import normalize_schema_b

    ev = normalize_schema_b({"kind": "tool_use", "sessionId": "s-9",
                             "agentId": "forge", "ts": 77.0,
                             "body": {"name": "read", "args": {"p": 1}}})
    assert ev is not None
    assert ev.type == "tool.call" and ev.agent == "forge"
    msg = normalize_schema_b({"kind": "assistant_message", "sessionId": "s",
                              "agentId": "a", "ts": 1.0,
                              "body": {"text": "hello"}})
    assert msg.type == "msg.out" and msg.payload.get("content")


def test_schema_b_unknown_kind_none():
    from vainplex_openclaw_amd.cortex.trace.events import normalize_schema_b

    assert normalize_schema_b({"kind": "nope", "ts": 1.0}) is None


# -- multilang-detectors.test.ts depth ---------------------------------------

COMPLETION_WORDS = {
    "fr": "c'est terminé", "ja": "タスクは完了です", "ko": "작업 완료 했어요",
    "es": "está completado", "ru": "всё готово", "de": "das ist erledigt",
    "pt": "está concluído", "zh": "任务完成", "it": "tutto completato",
    "en": "the task is done",
}


@pytest.mark.parametrize("lang", sorted(COMPLETION_WORDS))
def test_hallucination_completion_claim_after_tool_error(lang):
    events = [
        ev("msg.out", content=COMPLETION_WORDS[lang]),
        ev("tool.result", toolError="command failed with exit 1", toolIsError=True),
    ]
    chains = reconstruct_chains(events)
    kinds = {f.signal_type for f in detect_all_signals(chains, ["hallucination"])}
    assert "hallucination" in kinds, lang


def test_hallucination_completion_claim_without_error_is_clean():
    events = [
        ev("msg.out", content="c'est terminé"),
        ev("tool.result", toolResult="ok"),
    ]
    chains = reconstruct_chains(events)
    assert not detect_all_signals(chains, ["hallucination"])


STATE_CLAIMS = [
    ("there are 5 errors in the log", True),
    ("es gibt 5 fehler im log", True),
    ("il y a 3 erreurs", True),
    ("hay 5 errores en total", True),
    ("ci sono 2 errori", True),
    ("есть 4 ошибки", True),
    ("je crois qu'il y a 3 erreurs", False),      # opinion exclusion
    ("ich glaube es gibt 5 fehler", False),
    ("creo que hay 5 errores", False),
    ("наверное есть 4 ошибки", False),
    ("the weather is nice", False),
]


@pytest.mark.parametrize("text,want", STATE_CLAIMS)
def test_unverified_state_claims_with_opinion_exclusions(text, want):
    events = [
        ev("msg.in", content="status?"),
        ev("msg.out", content=text),
    ]
    chains = reconstruct_chains(events)
    kinds = {f.signal_type for f in detect_all_signals(chains, ["unverified_claim"])}
    assert ("unverified_claim" in kinds) == want, text


def test_state_claim_backed_by_tool_result_passes():
    events = [
        ev("tool.call", toolName="grep"),
        ev("tool.result", toolResult="5 matches"),
        ev("msg.out", content="there are 5 errors in the log"),
    ]
    chains = reconstruct_chains(events)
    assert not detect_all_signals(chains, ["unverified_claim"])


# -- trace-to-facts-bridge.test.ts file-level depth -------------------------

def _report(findings):
    return {"findings": findings}


def _finding(subject="svc", predicate="status", value="running",
             conf=0.8, sig="hallucination"):
    return {"id": "f1", "signalType": sig, "confidence": conf,
            "evidence": {"subject": subject, "predicate": predicate,
                         "value": value}}


def test_bridge_file_formats(tmp_path):
    import json

    from vainplex_openclaw_amd.governance.trace_to_facts import extract_facts_from_file

    p = tmp_path / "r.json"
    p.write_text(json.dumps(_report([_finding()])))
    assert extract_facts_from_file(str(p))[0]["subject"] == "svc"
    # bare array
    p2 = tmp_path / "arr.json"
    p2.write_text(json.dumps([_finding(subject="x")]))
    assert extract_facts_from_file(str(p2))[0]["subject"] == "x"
    # missing / invalid / wrong shape
    assert extract_facts_from_file(str(tmp_path / "ghost.json")) == []
    bad = tmp_path / "bad.json"
    bad.write_text("{nope")
    assert extract_facts_from_file(str(bad)) == []
    odd = tmp_path / "odd.json"
    odd.write_text('{"not_findings": 1}')
    assert extract_facts_from_file(str(odd)) == []


def test_bridge_confidence_and_signal_filter(tmp_path):
    import json

    from vainplex_openclaw_amd.governance.trace_to_facts import extract_facts_from_file

    p = tmp_path / "r.json"
    p.write_text(json.dumps(_report([
        _finding(conf=0.3),                         # below threshold
        _finding(sig="doom_loop"),                  # wrong signal type
        _finding(subject=None),                     # no factCorrection payload
        _finding(subject="keeper"),
    ])))
    facts = extract_facts_from_file(str(p))
    assert [f["subject"] for f in facts] == ["keeper"]


def test_bridge_merge_dedupe_later_wins():
    from vainplex_openclaw_amd.governance.trace_to_facts import merge_facts

    existing = [{"subject": "svc", "predicate": "status", "value": "old"}]
    new = [{"subject": "SVC", "predicate": "Status", "value": "new"},
           {"subject": "svc", "predicate": "port", "value": "443"}]
    out = merge_facts(existing, new)
    assert len(out) == 2                          # case-insensitive dedupe
    by_pred = {f["predicate"].lower(): f for f in out}
    assert by_pred["status"]["value"] == "new"    # later wins
    assert by_pred["port"]["value"] == "443"


def test_bridge_run_write_and_merge(tmp_path):
    import json

    from vainplex_openclaw_amd.governance.trace_to_facts import run_bridge

    r1 = tmp_path / "r1.json"
    r1.write_text(json.dumps(_report([_finding(subject="a", value="1")])))
    r2 = tmp_path / "r2.json"
    r2.write_text(json.dumps(_report([_finding(subject="b", value="2")])))
    out = tmp_path / "deep" / "facts.json"        # dir created on demand
    n = run_bridge([str(r1), str(r2)], str(out))
    assert n == 2 and out.is_file()
    data = json.loads(out.read_text())
    assert {f["subject"] for f in data["facts"]} == {"a", "b"}
    # second run merges + dedupes (same subject|predicate overrides)
    r3 = tmp_path / "r3.json"
    r3.write_text(json.dumps(_report([_finding(subject="a", value="9")])))
    n2 = run_bridge([str(r3)], str(out))
    assert n2 == 2
    data2 = json.loads(out.read_text())
    va = next(f["value"] for f in data2["facts"] if f["subject"] == "a")
    assert va == "9"
    # no new facts -> existing count returned
    empty = tmp_path / "empty.json"
    empty.write_text(json.dumps(_report([])))
    assert run_bridge([str(empty)], str(out)) == 2


def test_bridge_timer_idempotent():
    import time as _time

    from vainplex_openclaw_amd.governance.trace_to_facts import BridgeTimer

    hits = []
    t = BridgeTimer(0.02, lambda: hits.append(1))
    t.start()
    t.start()                                     # idempotent
    _time.sleep(0.08)
    t.stop()
    t.stop()                                      # safe twice
    n = len(hits)
    assert n >= 1
    _time.sleep(0.05)
    assert len(hits) == n                         # actually stopped


# -- redactor.test.ts depth --------------------------------------------------

def _red_chain(payload, etype="msg.in"):
    from vainplex_openclaw_amd.cortex.trace.analyzer import ChainRedactor
    from vainplex_openclaw_amd.cortex.trace.chains import ConversationChain
    from vainplex_openclaw_amd.cortex.trace.events import NormalizedEvent

    ev = NormalizedEvent(id="e", ts=1.0, agent="a", session="s",
                         type=etype, payload=payload)
    chain = ConversationChain(id="c", session="s", agent="a", events=[ev])
    return ChainRedactor().redact_chain(chain)[0]["payload"], ev


REDACT_CASES = [
    ("sk-" + "a1B2" * 6, True),                       # OpenAI-style
    ("pk_live_" + "x" * 20, True),                    # Stripe publishable
    ("sk_test_" + "y" * 20, True),                    # Stripe secret
    ("Bearer " + "t" * 24, True),
    ("https://bob:s3cretpw@db.example/x", True),      # url-embedded password
    ("-----BEGIN RSA PRIVATE KEY-----", True),
    ("PASSWORD=hunter22x", True),
    ("ghp_" + "f" * 36, True),
    ("eyJhbGciOiJIUzI1NiJ9.eyJzdWIiOiIxMjMifQ.sig-part-here", True),  # JWT
    ("perfectly ordinary sentence with no secrets", False),
    ("", False),
]


@pytest.mark.parametrize("text,redacts", REDACT_CASES)
def test_trace_redactor_content_table(text, redacts):
    out, ev = _red_chain({"content": f"before {text} after"})
    changed = out["content"] != f"before {text} after"
    assert changed == redacts, (text, out["content"])


def test_trace_redactor_tool_fields_and_nesting():
    key = "sk-abcdefghij0123456789XY"
    out, ev = _red_chain({
        "toolError": f"auth failed for {key}",
        "toolResult": {"nested": {"token": key}, "list": [key, "ok"]},
        "toolParams": {"cmd": f"use {key}"},
    }, etype="tool.result")
    import json as _json

    blob = _json.dumps(out)
    assert key not in blob
    assert out["toolResult"]["list"][1] == "ok"
    # original event untouched
    assert key in ev.payload["toolError"]
    assert key in ev.payload["toolParams"]["cmd"]


def test_trace_redactor_multiple_types_one_text():
    out, _ = _red_chain({"content":
        "key sk-abcdefghij0123456789XY then ghp_" + "f" * 36 +
        " and Bearer " + "b" * 24})
    c = out["content"]
    assert "sk-abcdefghij" not in c and "ghp_" + "f" * 36 not in c
    assert "Bearer " + "b" * 24 not in c


def test_trace_redactor_custom_patterns_and_invalid():
    from vainplex_openclaw_amd.cortex.trace.analyzer import ChainRedactor
    from vainplex_openclaw_amd.cortex.trace.chains import ConversationChain
    from vainplex_openclaw_amd.cortex.trace.events import NormalizedEvent

    red = ChainRedactor(custom_patterns=[
        {"name": "ticket", "regex": r"TICKET-\d{4}"},
        {"name": "bad", "regex": "(unclosed"},        # skipped silently
    ])
    ev = NormalizedEvent(id="e", ts=1.0, agent="a", session="s", type="msg.in",
                         payload={"content": "see TICKET-9999 now"})
    chain = ConversationChain(id="c", session="s", agent="a", events=[ev])
    out = red.redact_chain(chain)[0]["payload"]["content"]
    assert "TICKET-9999" not in out
