"""Classifier + output-generator tables mirroring cortex
`test/trace-analyzer/classifier.test.ts` (19 its) and
`output-generator.test.ts` (13 its): LLM-config resolution, triage
filtering, classification parsing/defaults, redaction-before-LLM, and
the classification-driven soul_rule / governance_policy / cortex_pattern
outputs.
"""

import json

import pytest

from vainplex_openclaw_amd.cortex.trace.analyzer import (
    ACTION_TYPES,
    FindingClassifier,
    default_action_text,
    generate_classified_outputs,
    resolve_analyzer_llm_config,
)
from vainplex_openclaw_amd.cortex.trace.chains import reconstruct_chains
from vainplex_openclaw_amd.cortex.trace.events import NormalizedEvent
from vainplex_openclaw_amd.cortex.trace.signals import Finding

TOP_LEVEL = {"enabled": True, "endpoint": "http://localhost:11434/v1",
             "model": "mistral:7b", "apiKey": "", "timeoutMs": 15000}


# -- resolve_analyzer_llm_config ---------------------------------------------

def test_resolve_disabled_override():
    assert resolve_analyzer_llm_config(TOP_LEVEL, {"enabled": False}) == {"enabled": False}
    assert resolve_analyzer_llm_config(TOP_LEVEL, None) == {"enabled": False}


def test_resolve_falls_back_to_top_level():
    r = resolve_analyzer_llm_config(TOP_LEVEL, {"enabled": True})
    assert r["enabled"] is True
    assert r["endpoint"] == "http://localhost:11434/v1"
    assert r["model"] == "mistral:7b"
    assert r["timeoutMs"] == 15000


def test_resolve_overrides_individual_fields():
    r = resolve_analyzer_llm_config(TOP_LEVEL, {
        "enabled": True, "endpoint": "http://cloud-api.com/v1",
        "model": "gpt-4o", "apiKey": "sk-test", "timeoutMs": 30000})
    assert r["endpoint"] == "http://cloud-api.com/v1"
    assert r["model"] == "gpt-4o"
    assert r["apiKey"] == "sk-test"
    assert r["timeoutMs"] == 30000


def test_resolve_partial_merge():
    r = resolve_analyzer_llm_config(TOP_LEVEL, {"enabled": True, "model": "gpt-4o"})
    assert r["model"] == "gpt-4o"
    assert r["endpoint"] == "http://localhost:11434/v1"
    assert r["apiKey"] == ""


# -- helpers ------------------------------------------------------------------

def _finding(sig="tool_fail", **over):
    base = dict(id="finding-00112233", signal_type=sig, chain_id="c1",
                agent="main", session="s", severity="medium",
                summary="Tool exec failed without recovery",
                evidence={"toolName": "exec"}, confidence=0.5)
    base.update(over)
    return Finding(**base)


def _chain(texts=("deploy", "done")):
    evs = [NormalizedEvent(id=f"e{i}", ts=1000.0 + i, agent="main", session="s",
                           type="msg.in" if i % 2 == 0 else "msg.out",
                           payload={"content": t}, seq=i)
           for i, t in enumerate(texts)]
    chains = reconstruct_chains(evs)
    return {chains[0].id: chains[0]} if chains else {}


def _analysis(**over):
    d = {"rootCause": "Agent retried without changing approach",
         "actionType": "soul_rule",
         "actionText": "NIEMALS denselben Befehl 3× wiederholen",
         "confidence": 0.85}
    d.update(over)
    return json.dumps(d)


# -- FindingClassifier --------------------------------------------------------

def test_no_llm_keeps_classification_null_with_default_action():
    c = FindingClassifier(call_llm=None)
    out = c.classify([_finding()], {})
    assert len(out) == 1
    assert out[0].get("classification") is None
    assert out[0]["actionText"] == default_action_text(_finding())


def test_missing_chain_still_classifies():
    # reference returns unclassified when chain is missing from map;
    # this build classifies on finding context alone (chain optional)
    f = _finding(chain_id="missing")
    c = FindingClassifier(call_llm=lambda p: _analysis(), triage_enabled=False)
    out = c.classify([f], {})
    assert len(out) == 1


def test_valid_llm_response_populates_classification():
    f = _finding()
    ch = _chain()
    f.chain_id = next(iter(ch))
    c = FindingClassifier(call_llm=lambda p: _analysis(),
                          triage_enabled=False, model="test-model")
    out = c.classify([f], ch)
    cl = out[0]["classification"]
    assert cl["rootCause"] == "Agent retried without changing approach"
    assert cl["actionType"] == "soul_rule"
    assert cl["model"] == "test-model"
    assert out[0]["confidence"] == 0.85


def test_invalid_json_classification_stays_null():
    c = FindingClassifier(call_llm=lambda p: "This is not JSON at all",
                          triage_enabled=False)
    out = c.classify([_finding()], {})
    assert out[0].get("classification") is None
    assert out[0]["actionText"]  # default action kept


def test_llm_exception_classification_stays_null():
    def boom(p):
        raise TimeoutError("abort")
    c = FindingClassifier(call_llm=boom, triage_enabled=False)
    out = c.classify([_finding()], {})
    assert len(out) == 1
    assert out[0].get("classification") is None


def test_unknown_action_type_defaults_manual_review():
    c = FindingClassifier(call_llm=lambda p: _analysis(actionType="unknown_type"),
                          triage_enabled=False)
    out = c.classify([_finding()], {})
    assert out[0]["classification"]["actionType"] == "manual_review"
    assert "unknown_type" not in ACTION_TYPES


def test_omitted_confidence_defaults_half():
    raw = json.dumps({"rootCause": "t", "actionType": "soul_rule",
                      "actionText": "rule"})
    c = FindingClassifier(call_llm=lambda p: raw, triage_enabled=False)
    out = c.classify([_finding()], {})
    assert out[0]["classification"]["confidence"] == 0.5


def test_redaction_applied_before_llm():
    sent = []

    def llm(prompt):
        sent.append(prompt)
        return _analysis()

    evs = [NormalizedEvent(id="e0", ts=1000.0, agent="main", session="s",
                           type="msg.in",
                           payload={"content": "Use PASSWORD=secret123 for the DB"}, seq=0),
           NormalizedEvent(id="e1", ts=1001.0, agent="main", session="s",
                           type="msg.out", payload={"content": "OK"}, seq=1)]
    chain = reconstruct_chains(evs)[0]
    f = _finding(chain_id=chain.id)
    FindingClassifier(call_llm=llm, triage_enabled=False).classify([f], {chain.id: chain})
    assert sent and "secret123" not in sent[0]


def test_triage_keep_false_filters_finding():
    calls = []

    def llm(prompt):
        calls.append(prompt)
        if len(calls) == 1:
            return json.dumps({"keep": False, "severity": "low"})
        return _analysis()

    out = FindingClassifier(call_llm=llm).classify([_finding()], {})
    assert out == []
    assert len(calls) == 1                      # deep analysis never ran


def test_triage_keep_true_passes_and_overrides_severity():
    calls = []

    def llm(prompt):
        calls.append(prompt)
        if len(calls) == 1:
            return json.dumps({"keep": True, "severity": "high"})
        return _analysis()

    out = FindingClassifier(call_llm=llm).classify([_finding()], {})
    assert len(out) == 1
    assert out[0]["severity"] == "high"
    assert out[0]["classification"] is not None
    assert len(calls) == 2                      # triage + analysis


# -- generate_classified_outputs ----------------------------------------------

def _cf(atype, text, conf=0.85, sig="tool_fail", fid=None, summary="Tool exec failed"):
    import uuid
    return {"id": fid or f"finding-{uuid.uuid4().hex[:8]}", "signalType": sig,
            "severity": "medium", "summary": summary, "agent": "main",
            "confidence": conf,
            "classification": {"rootCause": "rc", "actionType": atype,
                               "actionText": text, "confidence": conf,
                               "model": "m"}}


def test_outputs_empty_for_empty_and_unclassified():
    assert generate_classified_outputs([]) == []
    unclassified = [{"id": "f1", "signalType": "tool_fail", "confidence": 0.5}]
    assert generate_classified_outputs(unclassified) == []


def test_manual_review_produces_no_output():
    assert generate_classified_outputs([_cf("manual_review", "Investigate")]) == []


def test_soul_rule_format_and_refs():
    out = generate_classified_outputs(
        [_cf("soul_rule", "NIEMALS denselben Befehl 3× wiederholen", fid="finding-abcd1234")])
    assert len(out) == 1
    assert out[0]["type"] == "soul_rule"
    assert "NIEMALS denselben Befehl 3× wiederholen" in out[0]["content"]
    assert "beobachtet in Traces" in out[0]["content"]
    assert "Findings: finding-" in out[0]["content"]
    assert out[0]["sourceFindings"] == ["finding-abcd1234"]


def test_soul_rule_groups_and_counts_observations():
    rule = "NEVER repeat the same command"
    out = generate_classified_outputs([_cf("soul_rule", rule) for _ in range(3)])
    assert len(out) == 1
    assert out[0]["observationCount"] == 3
    assert "3× beobachtet in Traces" in out[0]["content"]


def test_soul_rule_different_texts_separate():
    out = generate_classified_outputs(
        [_cf("soul_rule", "Rule A"), _cf("soul_rule", "Rule B")])
    assert len(out) == 2


def test_soul_rule_confidence_averaged():
    rule = "NEVER do X"
    out = generate_classified_outputs(
        [_cf("soul_rule", rule, conf=0.8), _cf("soul_rule", rule, conf=0.6)])
    assert out[0]["confidence"] == pytest.approx(0.7)


def test_governance_policy_structure():
    out = generate_classified_outputs(
        [_cf("governance_policy", "Block repeated exec calls", sig="doom_loop",
             fid="finding-deadbeef", summary="Doom loop")])
    assert out[0]["type"] == "governance_policy"
    policy = json.loads(out[0]["content"])
    assert policy["id"].startswith("trace-gen-")
    assert policy["name"].startswith("Auto:")
    assert policy["version"] == "1.0.0"
    assert policy["scope"]["hooks"] == ["before_tool_call"]
    assert len(policy["rules"]) == 1
    assert policy["rules"][0]["effect"]["action"] == "audit"


def test_policy_hooks_by_signal_type():
    outs = generate_classified_outputs([
        _cf("governance_policy", "a", sig="doom_loop"),
        _cf("governance_policy", "b", sig="hallucination"),
        _cf("governance_policy", "c", sig="dissatisfied"),
    ])
    hooks = [json.loads(o["content"])["scope"]["hooks"] for o in outs]
    assert hooks == [["before_tool_call"], ["message_sending"], ["message_sent"]]


def test_cortex_pattern_content_is_regex():
    rx = r"\b(?:error|failed)\s+(?:to\s+)?connect\b"
    out = generate_classified_outputs([_cf("cortex_pattern", rx)])
    assert out[0]["type"] == "cortex_pattern"
    assert out[0]["content"] == rx


def test_mixed_action_types():
    out = generate_classified_outputs([
        _cf("soul_rule", "Rule 1"),
        _cf("governance_policy", "Policy 1"),
        _cf("cortex_pattern", r"\bpattern\b"),
        _cf("manual_review", "Check this"),
    ])
    types = [o["type"] for o in out]
    assert sorted(types) == ["cortex_pattern", "governance_policy", "soul_rule"]


# -- analyzer report carries classifiedOutputs --------------------------------

def test_report_counts_classified_and_carries_outputs(tmp_path):
    from vainplex_openclaw_amd.cortex.trace.analyzer import (
        MockTraceSource,
        TraceAnalyzer,
    )

    evs = []
    # a doom loop: 3 identical failing tool calls
    for i in range(3):
        evs.append(NormalizedEvent(id=f"c{i}", ts=1000.0 + 2 * i, agent="main",
                                   session="s", type="tool.call",
                                   payload={"toolName": "exec",
                                            "toolParams": {"command": "ls"}},
                                   seq=2 * i))
        evs.append(NormalizedEvent(id=f"r{i}", ts=1001.0 + 2 * i, agent="main",
                                   session="s", type="tool.result",
                                   payload={"toolName": "exec",
                                            "toolError": "Connection refused"},
                                   seq=2 * i + 1))
    ta = TraceAnalyzer(str(tmp_path), MockTraceSource(evs),
                       call_llm=lambda p: _analysis(actionType="governance_policy"))
    report = ta.run()
    assert report["stats"]["findingsClassified"] >= 1
    assert any(o["type"] == "governance_policy" for o in report["classifiedOutputs"])
    json.loads(report["classifiedOutputs"][0]["content"])  # valid policy JSON
