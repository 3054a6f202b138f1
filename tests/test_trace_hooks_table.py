"""Trace-analyzer hook registration tables mirroring cortex
`test/trace-analyzer/hooks.test.ts` (11 its): command + gateway
registration when enabled, schedule timer setup/skip, cleanup
idempotence, command handlers returning results, signal-registry
preload logging, resolved-config wiring.
"""

import json
import time

import pytest

from vainplex_openclaw_amd.core.api import HookBus, PluginApi
from vainplex_openclaw_amd.cortex.hooks import CortexPlugin
from vainplex_openclaw_amd.eventstore import EventJournal


class _Log:
    def __init__(self):
        self.lines = []

    def info(self, m, *a):
        self.lines.append(("info", m % a if a else m))

    def warn(self, m, *a):
        self.lines.append(("warn", m % a if a else m))

    def error(self, m, *a):
        self.lines.append(("error", m % a if a else m))

    def debug(self, m, *a):
        self.lines.append(("debug", m % a if a else m))


def _api(plugin_config):
    return PluginApi(id="openclaw-cortex", plugin_config=plugin_config,
                     logger=_Log(), config={}, bus=HookBus())


def _plugin(tmp_path, ta=None, with_journal=True, monkeypatch=None):
    j = EventJournal(durable=False) if with_journal else None
    cfg = {"workspace": str(tmp_path)}
    if ta is not None:
        cfg["traceAnalyzer"] = ta
    p = CortexPlugin(str(tmp_path), journal=j)
    api = _api(cfg)
    p.register(api)
    return p, api, j


def test_registers_analyze_and_status_commands(tmp_path):
    p, api, j = _plugin(tmp_path, ta={"enabled": True})
    assert "cortexanalyze" in api.commands
    assert "cortextracestatus" in api.commands
    assert "trace-analyze" in api.commands           # reference names
    assert "trace-status" in api.commands
    assert "cortex.analyze" in api.gateway_methods
    assert "cortex.trace.status" in api.gateway_methods


def test_logs_signal_registry_preload(tmp_path):
    p, api, j = _plugin(tmp_path, ta={"enabled": True})
    assert any("Loaded signal patterns for" in m and "en" in m and "zh" in m
               for _, m in api.logger.lines)


def test_disabled_registers_nothing(tmp_path):
    p, api, j = _plugin(tmp_path, ta={"enabled": False})
    assert "cortexanalyze" not in api.commands
    assert p.analyzer is None


def test_no_timer_without_schedule(tmp_path):
    p, api, j = _plugin(tmp_path, ta={"enabled": True})
    assert p._analyzer_timer is None


def test_schedule_enabled_sets_timer_and_logs(tmp_path):
    p, api, j = _plugin(tmp_path, ta={
        "enabled": True, "schedule": {"enabled": True, "intervalHours": 24}})
    try:
        assert p._analyzer_timer is not None
        assert p._analyzer_timer.daemon is True          # never blocks exit
        assert any("Scheduled analysis every 24h" in m for _, m in api.logger.lines)
    finally:
        p.cleanup_trace_analyzer()


def test_cleanup_clears_timer_and_analyzer_and_is_idempotent(tmp_path):
    p, api, j = _plugin(tmp_path, ta={
        "enabled": True, "schedule": {"enabled": True, "intervalHours": 1}})
    assert p.analyzer is not None and p._analyzer_timer is not None
    p.cleanup_trace_analyzer()
    assert p.analyzer is None and p._analyzer_timer is None
    p.cleanup_trace_analyzer()                           # safe again
    assert p.analyzer is None


def test_gateway_stop_triggers_cleanup(tmp_path):
    p, api, j = _plugin(tmp_path, ta={
        "enabled": True, "schedule": {"enabled": True, "intervalHours": 1}})
    api.bus.emit("gateway_stop", {})
    assert p._analyzer_timer is None and p.analyzer is None


def test_analyze_command_runs_and_returns_report(tmp_path):
    p, api, j = _plugin(tmp_path, ta={"enabled": True})
    for i in range(3):
        j.publish("s.evt", {"id": f"c{i}", "ts": 1000.0 + 2 * i,
                            "canonicalType": "tool.call.requested",
                            "actor": {"id": "main"},
                            "scope": {"sessionKey": "agent:main:s1"},
                            "data": {"toolName": "exec", "params": {"c": "ls"}}})
        j.publish("s.evt", {"id": f"r{i}", "ts": 1001.0 + 2 * i,
                            "canonicalType": "tool.call.failed",
                            "actor": {"id": "main"},
                            "scope": {"sessionKey": "agent:main:s1"},
                            "data": {"toolName": "exec", "error": "refused"}})
    report = api.commands["cortexanalyze"]()
    assert report["eventsAnalyzed"] == 6
    assert any(f["signalType"] == "doom_loop" for f in report["findings"])


def test_status_command_before_and_after_run(tmp_path):
    p, api, j = _plugin(tmp_path, ta={"enabled": True})
    st0 = api.commands["cortextracestatus"]()
    assert st0["enabled"] is True and st0["runsCompleted"] == 0
    api.commands["cortexanalyze"]()
    st1 = api.commands["cortextracestatus"]()
    assert st1["runsCompleted"] == 1
    assert st1["reportPath"].endswith("trace-analysis-report.json")
    p.cleanup_trace_analyzer()
    assert api.commands["cortextracestatus"]() == {"enabled": False}


def test_resolved_config_shape_wires_from_config(tmp_path):
    ta = {"enabled": True,
          "signals": {"SIG-CORRECTION": {"enabled": False}},
          "chainGapMinutes": 7,
          "output": {"maxFindings": 5, "reportPath": str(tmp_path / "r.json")}}
    p, api, j = _plugin(tmp_path, ta=ta)
    assert p.analyzer.config.chain_gap_minutes == 7
    assert "correction" not in p.analyzer.config.detectors
    assert p.analyzer.report_path.endswith("r.json")


def test_journal_absent_falls_back_to_nats_then_none(tmp_path):
    # no journal + unreachable NATS -> graceful no-registration
    p = CortexPlugin(str(tmp_path), journal=None)
    api = _api({"workspace": str(tmp_path),
                "traceAnalyzer": {"enabled": True,
                                  "nats": {"url": "nats://127.0.0.1:1"}}})
    p.register(api)
    assert p.analyzer is None
    assert "cortexanalyze" not in api.commands
    assert any("no event source available" in m for _, m in api.logger.lines)
