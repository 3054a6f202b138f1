"""Trace-analyzer + top-level cortex config tables mirroring
`test/trace-analyzer/config.test.ts` (23 its): defaults, typed
resolution with graceful fallback, per-signal overrides with unknown-id
filtering, nats/llm/triage/redactPatterns/output sections, and the
top-level resolveConfig integration.
"""

import pytest

from vainplex_openclaw_amd.cortex.config import DEFAULTS, resolve_config
from vainplex_openclaw_amd.cortex.trace.config import (
    SIGNAL_IDS,
    TRACE_ANALYZER_DEFAULTS,
    enabled_detectors,
    resolve_trace_analyzer_config,
)


# -- defaults -----------------------------------------------------------------

def test_defaults_disabled():
    assert TRACE_ANALYZER_DEFAULTS["enabled"] is False


def test_defaults_nats():
    n = TRACE_ANALYZER_DEFAULTS["nats"]
    assert n["url"] == "nats://localhost:4222"
    assert n["stream"] == "openclaw-events"
    assert n["subjectPrefix"] == "openclaw.events"


def test_defaults_schedule():
    s = TRACE_ANALYZER_DEFAULTS["schedule"]
    assert s["enabled"] is False and s["intervalHours"] == 24


def test_defaults_chain_gap_and_batch():
    assert TRACE_ANALYZER_DEFAULTS["chainGapMinutes"] == 30
    assert TRACE_ANALYZER_DEFAULTS["fetchBatchSize"] == 500


def test_defaults_unverified_claim_off_rest_on():
    sigs = TRACE_ANALYZER_DEFAULTS["signals"]
    assert sigs["SIG-UNVERIFIED-CLAIM"]["enabled"] is False
    for sid in ("SIG-CORRECTION", "SIG-TOOL-FAIL", "SIG-DOOM-LOOP",
                "SIG-DISSATISFIED", "SIG-REPEAT-FAIL", "SIG-HALLUCINATION"):
        assert sigs[sid]["enabled"] is True


# -- resolver -----------------------------------------------------------------

def test_resolve_no_config_gives_defaults():
    c = resolve_trace_analyzer_config()
    assert c["enabled"] is False
    assert c["nats"]["url"] == "nats://localhost:4222"
    assert c["chainGapMinutes"] == 30
    assert resolve_trace_analyzer_config(None) == c


def test_resolve_partial_merge():
    c = resolve_trace_analyzer_config({"enabled": True, "chainGapMinutes": 60})
    assert c["enabled"] is True
    assert c["chainGapMinutes"] == 60
    assert c["nats"]["url"] == "nats://localhost:4222"
    assert c["fetchBatchSize"] == 500


def test_resolve_nats_full():
    c = resolve_trace_analyzer_config({"nats": {
        "url": "nats://prod:4222", "stream": "my-events",
        "subjectPrefix": "my.events", "credentials": "/path/to/creds",
        "user": "admin", "password": "secret"}})
    assert c["nats"]["url"] == "nats://prod:4222"
    assert c["nats"]["stream"] == "my-events"
    assert c["nats"]["credentials"] == "/path/to/creds"
    assert c["nats"]["user"] == "admin"
    assert c["nats"]["password"] == "secret"


def test_resolve_per_signal_enable_disable():
    c = resolve_trace_analyzer_config({"signals": {
        "SIG-CORRECTION": {"enabled": False},
        "SIG-DOOM-LOOP": {"enabled": True, "severity": "critical"}}})
    assert c["signals"]["SIG-CORRECTION"]["enabled"] is False
    assert c["signals"]["SIG-DOOM-LOOP"]["enabled"] is True
    assert c["signals"]["SIG-DOOM-LOOP"]["severity"] == "critical"
    assert c["signals"]["SIG-TOOL-FAIL"]["enabled"] is True   # untouched


def test_resolve_signal_severity_override():
    c = resolve_trace_analyzer_config({"signals": {
        "SIG-TOOL-FAIL": {"enabled": True, "severity": "high"}}})
    assert c["signals"]["SIG-TOOL-FAIL"]["severity"] == "high"


def test_resolve_invalid_severity_keeps_default():
    c = resolve_trace_analyzer_config({"signals": {
        "SIG-TOOL-FAIL": {"severity": "catastrophic"}}})
    assert c["signals"]["SIG-TOOL-FAIL"]["severity"] == "medium"


def test_resolve_ignores_invalid_signal_ids():
    c = resolve_trace_analyzer_config({"signals": {
        "SIG-INVALID": {"enabled": True},
        "SIG-CORRECTION": {"enabled": False}}})
    assert c["signals"]["SIG-CORRECTION"]["enabled"] is False
    assert "SIG-INVALID" not in c["signals"]


def test_resolve_llm_with_triage():
    c = resolve_trace_analyzer_config({"llm": {
        "enabled": True, "endpoint": "https://api.openai.com/v1",
        "model": "gpt-4o", "apiKey": "sk-test", "timeoutMs": 30000,
        "triage": {"endpoint": "http://localhost:11434/v1",
                   "model": "mistral:7b"}}})
    assert c["llm"]["enabled"] is True
    assert c["llm"]["endpoint"] == "https://api.openai.com/v1"
    assert c["llm"]["model"] == "gpt-4o"
    assert c["llm"]["triage"]["endpoint"] == "http://localhost:11434/v1"
    assert c["llm"]["triage"]["model"] == "mistral:7b"


def test_resolve_triage_none_when_absent():
    c = resolve_trace_analyzer_config({"llm": {"enabled": True}})
    assert c["llm"]["triage"] is None


def test_resolve_redact_patterns():
    c = resolve_trace_analyzer_config({"redactPatterns": [r"secret-\d+", r"password=\S+"]})
    assert c["redactPatterns"] == [r"secret-\d+", r"password=\S+"]


def test_resolve_redact_patterns_filters_non_strings():
    c = resolve_trace_analyzer_config({"redactPatterns": ["valid", 123, None, "also-valid"]})
    assert c["redactPatterns"] == ["valid", "also-valid"]


def test_resolve_output():
    c = resolve_trace_analyzer_config({"output": {
        "maxFindings": 500, "reportPath": "/custom/path/report.json"}})
    assert c["output"]["maxFindings"] == 500
    assert c["output"]["reportPath"] == "/custom/path/report.json"


def test_resolve_invalid_types_fall_back():
    c = resolve_trace_analyzer_config({
        "enabled": "yes", "chainGapMinutes": "thirty", "fetchBatchSize": None})
    assert c["enabled"] is False
    assert c["chainGapMinutes"] == 30
    assert c["fetchBatchSize"] == 500


def test_enabled_detectors_mapping():
    c = resolve_trace_analyzer_config({"signals": {
        "SIG-UNVERIFIED-CLAIM": {"enabled": True},
        "SIG-CORRECTION": {"enabled": False}}})
    dets = enabled_detectors(c)
    assert "unverified_claim" in dets and "correction" not in dets
    assert set(SIGNAL_IDS.values()) >= set(dets)


# -- top-level resolveConfig integration --------------------------------------

def test_top_level_includes_trace_analyzer():
    c = resolve_config()
    assert c["traceAnalyzer"]["enabled"] is False
    assert DEFAULTS["traceAnalyzer"]["enabled"] is False


def test_top_level_resolves_trace_analyzer_section():
    c = resolve_config({"traceAnalyzer": {"enabled": True, "chainGapMinutes": 45}})
    assert c["traceAnalyzer"]["enabled"] is True
    assert c["traceAnalyzer"]["chainGapMinutes"] == 45


def test_top_level_other_sections_unaffected():
    c = resolve_config({"traceAnalyzer": {"enabled": True}})
    assert c["threadTracker"]["enabled"] is True
    assert c["decisionTracker"]["enabled"] is True
    assert c["bootContext"]["enabled"] is True


# -- from_config wiring -------------------------------------------------------

def test_analyzer_from_config_wires_detectors_gap_and_report_path(tmp_path):
    from vainplex_openclaw_amd.cortex.trace.analyzer import (
        MockTraceSource,
        TraceAnalyzer,
    )

    resolved = resolve_trace_analyzer_config({
        "enabled": True,
        "chainGapMinutes": 7,
        "signals": {"SIG-CORRECTION": {"enabled": False}},
        "redactPatterns": [r"hunter\d"],
        "output": {"maxFindings": 3,
                   "reportPath": str(tmp_path / "custom-report.json")},
    })
    ta = TraceAnalyzer.from_config(str(tmp_path), MockTraceSource([]), resolved)
    assert ta.config.chain_gap_minutes == 7
    assert "correction" not in ta.config.detectors
    assert "doom_loop" in ta.config.detectors
    assert ta.config.max_findings == 3
    assert ta.report_path.endswith("custom-report.json")
    # custom redact pattern active in the classifier's redactor
    assert ta.classifier.redactor.engine.scan_string("hunter2")["output"] != "hunter2"
    report = ta.run()
    assert report["stats"]["findings"] == 0


# -- top-level cortex resolve_config depth (cortex config.test.ts) -----------

def test_top_level_defaults_none_and_empty():
    for raw in (None, {}):
        c = resolve_config(raw)
        assert c["threadTracker"] == {"enabled": True, "pruneDays": 14,
                                      "maxThreads": 50}
        assert c["language"] == "both"
        assert c["llm"] == {"enabled": False}


def test_top_level_partial_nested_merge():
    c = resolve_config({"threadTracker": {"maxThreads": 9},
                        "bootContext": {"enabled": False}})
    assert c["threadTracker"]["maxThreads"] == 9
    assert c["threadTracker"]["pruneDays"] == 14       # untouched default
    assert c["bootContext"]["enabled"] is False
    assert c["decisionTracker"]["enabled"] is True     # other section intact


def test_top_level_unknown_keys_preserved_and_language():
    c = resolve_config({"language": "de", "workspace": "/w",
                        "customPatterns": {"en": {}}})
    assert c["language"] == "de"
    assert c["workspace"] == "/w"                      # extra keys pass through
    assert c["customPatterns"] == {"en": {}}


def test_top_level_all_features_disabled():
    c = resolve_config({"threadTracker": {"enabled": False},
                        "decisionTracker": {"enabled": False},
                        "commitmentTracker": {"enabled": False},
                        "bootContext": {"enabled": False},
                        "traceAnalyzer": {"enabled": False}})
    for section in ("threadTracker", "decisionTracker", "commitmentTracker",
                    "bootContext", "traceAnalyzer"):
        assert c[section]["enabled"] is False
