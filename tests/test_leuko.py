"""Leuko tests: collectors, custom thresholds, aggregator (health/
categories/delta/summary), anomaly detection, plugin surface."""

import json
import time

import pytest

from vainplex_openclaw_amd.core.api import HookBus, NullLogger, PluginApi
from vainplex_openclaw_amd.eventstore import EventJournal
from vainplex_openclaw_amd.leuko import (
    AnomalyDetector,
    LeukoPlugin,
    MetricHistory,
    generate_sitrep,
    resolve_config,
    run_custom_collector,
    safe_collect,
    write_sitrep,
)
from vainplex_openclaw_amd.leuko.collectors import (
    collect_errors,
    collect_goals,
    collect_journal,
    collect_threads,
    result,
)


class FakeClock:
    def __init__(self, t=1_700_000_000.0):
        self.t = t

    def __call__(self):
        return self.t


def _iso(ts):
    return time.strftime("%Y-%m-%dT%H:%M:%S", time.gmtime(ts)) + "Z"


# -- safe_collect ------------------------------------------------------------

def test_safe_collect_disabled_and_error():
    assert safe_collect("x", lambda c: 1 / 0, {"enabled": False})["summary"] == "disabled"
    r = safe_collect("x", lambda c: 1 / 0, {"enabled": True}, NullLogger())
    assert r["status"] == "error" and "division" in r["error"]


def test_safe_collect_times():
    r = safe_collect("x", lambda c: result("ok", [], "fine"), {"enabled": True})
    assert r["status"] == "ok" and r["duration_ms"] >= 0


# -- collectors --------------------------------------------------------------

def test_collect_goals_red_zone_and_stale(tmp_path):
    now = time.time()
    p = tmp_path / "goals.json"
    p.write_text(json.dumps({"goals": [
        {"id": "g1", "title": "Deploy", "zone": "red", "status": "approved"},
        {"id": "g2", "title": "Old", "status": "proposed", "proposed_at": _iso(now - 100 * 3600)},
        {"id": "g3", "title": "Fresh", "status": "proposed", "proposed_at": _iso(now)},
    ]}))
    r = collect_goals({"enabled": True, "goalsPath": str(p), "staleHours": 48})
    ids = [i["id"] for i in r["items"]]
    assert "goal-g1-red-approved" in ids and "goal-g2-stale" in ids
    assert "goal-g3-stale" not in ids
    assert r["status"] == "warn"


def test_collect_threads_stale_and_high_priority(tmp_path):
    now = time.time()
    p = tmp_path / "threads.json"
    p.write_text(json.dumps({"threads": [
        {"id": "t1", "topic": "old stuff", "status": "open", "last_activity": _iso(now - 10 * 86400)},
        {"id": "t2", "topic": "urgent", "status": "open", "priority": "high", "last_activity": _iso(now)},
        {"id": "t3", "topic": "done", "status": "closed", "last_activity": _iso(now - 30 * 86400)},
    ]}))
    r = collect_threads({"enabled": True, "threadsPath": str(p), "staleDays": 7})
    ids = [i["id"] for i in r["items"]]
    assert "thread-t1-stale" in ids and "thread-t2-high" in ids
    assert not any("t3" in i for i in ids)


def test_collect_errors_recent_critical(tmp_path):
    now = time.time()
    p = tmp_path / "errors.json"
    p.write_text(json.dumps([
        {"id": "e1", "pattern": "OOM", "severity": "critical", "last_seen": _iso(now - 3600)},
        {"id": "e2", "pattern": "old", "severity": "critical", "last_seen": _iso(now - 48 * 3600)},
        {"id": "e3", "pattern": "meh", "severity": "low", "last_seen": _iso(now)},
    ]))
    r = collect_errors({"enabled": True, "patternsPath": str(p), "recentHours": 24})
    assert [i["id"] for i in r["items"]] == ["error-e1"]
    assert r["status"] == "critical"


def test_collect_journal_counts_and_staleness():
    j = EventJournal(durable=False)
    j.publish("s", {"ts": (time.time() - 7200) * 1000})
    r = collect_journal({"enabled": True, "journal": j, "maxAgeMins": 60})
    assert any(i["id"] == "journal-stale" for i in r["items"])
    j2 = EventJournal(durable=False)
    j2.publish("s", {"ts": time.time() * 1000})
    r2 = collect_journal({"enabled": True, "journal": j2, "maxAgeMins": 60})
    assert r2["status"] == "ok" and "1 message(s)" in r2["summary"]


def test_custom_collector_threshold_and_flags():
    r = run_custom_collector({"id": "disk", "command": "echo 85", "warnThreshold": "80",
                              "criticalThreshold": "95"})
    assert r["items"][0]["severity"] == "warn"
    r2 = run_custom_collector({"id": "disk", "command": "echo 97", "warnThreshold": "80",
                               "criticalThreshold": "95"})
    assert r2["items"][0]["severity"] == "critical"
    assert r2["items"][0]["category"] == "needs_owner"
    r3 = run_custom_collector({"id": "log", "command": "echo bad", "warnIfOutput": True})
    assert r3["items"][0]["id"] == "custom-log-output"
    r4 = run_custom_collector({"id": "hb", "command": "true", "warnIfNoOutput": True})
    assert r4["items"][0]["id"] == "custom-hb-no-output"


# -- aggregator --------------------------------------------------------------

def _min_cfg(tmp_path, **collectors):
    return {
        "enabled": True,
        "outputPath": str(tmp_path / "sitrep.json"),
        "previousPath": str(tmp_path / "sitrep-previous.json"),
        "collectors": collectors,
        "customCollectors": [],
        "summaryMaxChars": 2000,
    }


def test_generate_sitrep_schema_and_categories(tmp_path):
    now = time.time()
    gp = tmp_path / "goals.json"
    gp.write_text(json.dumps({"goals": [
        {"id": "g1", "title": "X", "zone": "red", "status": "approved"}]}))
    cfg = _min_cfg(tmp_path, goals={"enabled": True, "goalsPath": str(gp)})
    rep = generate_sitrep(cfg)
    assert rep["version"] == 1 and rep["generated"].endswith("Z")
    assert rep["health"]["overall"] == "warn"
    assert rep["health"]["details"]["goals"] == "warn"
    assert rep["health"]["details"]["errors"] == "disabled"
    assert len(rep["categories"]["needs_owner"]) == 1
    assert "need owner attention" in rep["summary"]


def test_sitrep_delta_tracking(tmp_path):
    cfg = _min_cfg(tmp_path)
    rep1 = generate_sitrep(cfg)
    write_sitrep(rep1, cfg["outputPath"], cfg["previousPath"])
    gp = tmp_path / "goals.json"
    gp.write_text(json.dumps({"goals": [
        {"id": "g1", "title": "X", "zone": "red", "status": "approved"}]}))
    cfg2 = _min_cfg(tmp_path, goals={"enabled": True, "goalsPath": str(gp)})
    cfg2["previousPath"] = cfg["outputPath"]  # delta vs last written
    rep2 = generate_sitrep(cfg2)
    assert rep2["delta"]["new_items"] == 1
    assert rep2["delta"]["previous_generated"] == rep1["generated"]


def test_sitrep_all_nominal_summary(tmp_path):
    rep = generate_sitrep(_min_cfg(tmp_path))
    assert rep["summary"] == "All systems nominal."
    assert rep["health"]["overall"] == "ok"


# -- anomaly detection -------------------------------------------------------

def test_metric_history_slope(tmp_path):
    clock = FakeClock()
    h = MetricHistory(str(tmp_path / "m.jsonl"), clock=clock)
    for i in range(10):
        h.record("x", 100.0 + i * 50.0, ts=clock.t + i * 3600.0)
    slope = h.slope_per_hour("x")
    assert slope == pytest.approx(50.0, rel=1e-6)


def test_metric_history_persists(tmp_path):
    p = str(tmp_path / "m.jsonl")
    h = MetricHistory(p)
    h.record("y", 1.0, ts=1.0)
    h.record("y", 2.0, ts=3600.0)
    h2 = MetricHistory(p)
    assert len(h2.series("y")) == 2


def test_directory_growth_detection(tmp_path):
    clock = FakeClock()
    h = MetricHistory(str(tmp_path / "m.jsonl"), clock=clock)
    det = AnomalyDetector(h, clock=clock)
    d = tmp_path / "grow"
    d.mkdir()
    # simulate 200 MB/h growth: fabricated history below the current (0 B)
    # sample so the trend continues through the real measurement
    name = f"dirsize:{d}"
    for i in range(5):
        h.record(name, -(5 - i) * 2e8, ts=clock.t - (5 - i) * 3600.0)
    it = det.check_directory_growth(str(d), warn_mb_per_hour=100.0)
    assert it is not None and "growing" in it["title"]


def test_declining_metric_detection(tmp_path):
    clock = FakeClock()
    h = MetricHistory(str(tmp_path / "m.jsonl"), clock=clock)
    det = AnomalyDetector(h, clock=clock)
    it = None
    for i in range(8):
        clock.t += 3600.0
        it = det.check_declining_metric("throughput", 1000.0 - i * 100.0, 50.0)
    assert it is not None and "declining" in it["title"]


def test_failure_correlation():
    clock = FakeClock()
    j = EventJournal(durable=False)
    for i in range(4):
        j.publish("s", {"ts": clock.t * 1000, "canonicalType": "tool.call.failed",
                        "agent": f"a{i % 2}"})
    det = AnomalyDetector(MetricHistory("/nonexistent/never.jsonl"), clock=clock)
    it = det.correlate_failures(j, window_s=600.0, threshold=3)
    assert it is not None and it["severity"] == "critical"
    assert "2 agent(s)" in it["title"]


def test_bootstrap_integrity(tmp_path):
    (tmp_path / "governance").mkdir()
    (tmp_path / "governance" / "trust.json").write_text("{not json")
    det = AnomalyDetector(MetricHistory(str(tmp_path / "m.jsonl")))
    items = det.check_bootstrap_integrity(str(tmp_path))
    assert len(items) == 1 and "corrupt" in items[0]["title"]


# -- plugin ------------------------------------------------------------------

def test_plugin_run_once_and_command(tmp_path, monkeypatch):
    monkeypatch.setenv("HOME", str(tmp_path))
    bus = HookBus()
    api = PluginApi(id="openclaw-leuko",
                    plugin_config={"collectors": {"systemd_timers": {"enabled": False},
                                                  "gpu_health": {"enabled": False}}},
                    logger=NullLogger(), config={}, bus=bus)
    p = LeukoPlugin(workspace=str(tmp_path), journal=EventJournal(durable=False))
    p.register(api)
    out = api.commands["sitrep"]()
    assert "Sitrep" in out["text"]
    assert (tmp_path / ".openclaw" / "sitrep" / "sitrep.json").exists()
    rep = api.gateway_methods["leuko.report"]()
    assert rep["version"] == 1
    assert "leuko_anomaly" in rep["collectors"]


def test_resolve_config_merges_collectors(tmp_path):
    cfg = resolve_config({"collectors": {"goals": {"enabled": True, "goalsPath": "/x"}},
                          "intervalMinutes": 5}, home=str(tmp_path))
    assert cfg["collectors"]["goals"]["enabled"] is True
    assert cfg["collectors"]["systemd_timers"]["enabled"] is True
    assert cfg["intervalMinutes"] == 5


# ===========================================================================
# Anomaly-detector depth: growth slope, declining metric, failure
# correlation window/threshold, bootstrap integrity, metric history
# ===========================================================================

def _det(tmp_path):
    from vainplex_openclaw_amd.leuko.anomaly import AnomalyDetector, MetricHistory

    t = [1_700_000_000.0]
    h = MetricHistory(str(tmp_path / "hist.json"), clock=lambda: t[0])
    return AnomalyDetector(h, clock=lambda: t[0]), h, t


def test_metric_history_roundtrip_and_cap(tmp_path):
    from vainplex_openclaw_amd.leuko.anomaly import MetricHistory

    t = [0.0]
    h = MetricHistory(str(tmp_path / "m.json"), max_points=5, clock=lambda: t[0])
    for i in range(8):
        t[0] += 60
        h.record("x", float(i))
    assert len(h.series("x")) == 5           # capped
    h2 = MetricHistory(str(tmp_path / "m.json"), max_points=5, clock=lambda: t[0])
    assert len(h2.series("x")) == 5          # persisted (cap applied on load)
    h3 = MetricHistory(str(tmp_path / "m.json"), clock=lambda: t[0])
    assert len(h3.series("x")) == 8          # full JSONL kept on disk


def test_slope_per_hour_math(tmp_path):
    from vainplex_openclaw_amd.leuko.anomaly import MetricHistory

    t = [0.0]
    h = MetricHistory(str(tmp_path / "m.json"), clock=lambda: t[0])
    for i in range(10):
        h.record("lin", 100.0 * i, ts=i * 3600.0)  # +100/hour
    slope = h.slope_per_hour("lin")
    assert slope is not None and abs(slope - 100.0) < 1e-6
    assert h.slope_per_hour("missing") is None


def test_declining_metric_thresholds(tmp_path):
    det, h, t = _det(tmp_path)
    # decline of 50/h, threshold 10/h -> warn
    for i in range(10):
        t[0] = i * 3600.0
        it = det.check_declining_metric("throughput", 1000.0 - 50.0 * i, 10.0)
    assert it is not None and "declining" in it["title"]
    # rising metric never warns
    for i in range(10):
        t[0] = 100_000 + i * 3600.0
        it2 = det.check_declining_metric("rising", 10.0 * i, 10.0)
    assert it2 is None


def test_failure_correlation_window_and_threshold(tmp_path):
    from vainplex_openclaw_amd.eventstore import EventJournal

    det, _h, t = _det(tmp_path)
    j = EventJournal(durable=False, clock=lambda: t[0])
    t[0] = 1_700_000_000.0
    for i, agent in enumerate(("a", "b", "a")):
        j.publish(f"s.{agent}", {"ts": t[0] * 1000, "canonicalType": "tool.call.failed",
                                 "agent": agent})
    it = det.correlate_failures(j, window_s=600, threshold=3)
    assert it is not None and it["severity"] == "critical"
    assert "2 agent(s)" in it["title"]
    # below threshold -> None
    j2 = EventJournal(durable=False, clock=lambda: t[0])
    j2.publish("s.a", {"ts": t[0] * 1000, "canonicalType": "tool.call.failed", "agent": "a"})
    assert det.correlate_failures(j2, window_s=600, threshold=3) is None
    # outside the window -> None
    j3 = EventJournal(durable=False, clock=lambda: t[0])
    for i in range(3):
        j3.publish("s.a", {"ts": (t[0] - 1200) * 1000,
                           "canonicalType": "tool.call.failed", "agent": "a"})
    assert det.correlate_failures(j3, window_s=600, threshold=3) is None
    assert det.correlate_failures(None) is None


def test_bootstrap_integrity_matrix(tmp_path):
    import os

    det, _h, _t = _det(tmp_path)
    ws = tmp_path / "ws"
    (ws / "governance").mkdir(parents=True)
    (ws / "memory" / "reboot").mkdir(parents=True)
    # missing files = first boot = fine
    assert det.check_bootstrap_integrity(str(ws)) == []
    # valid file = fine
    (ws / "governance" / "trust.json").write_text('{"agents": {}}')
    assert det.check_bootstrap_integrity(str(ws)) == []
    # corrupt file = critical
    (ws / "memory" / "reboot" / "threads.json").write_text("{broken")
    items = det.check_bootstrap_integrity(str(ws))
    assert len(items) == 1 and items[0]["severity"] == "critical"
    assert "threads.json" in items[0]["title"]


def test_directory_growth_warn(tmp_path):
    det, h, t = _det(tmp_path)
    d = tmp_path / "grow"
    d.mkdir()
    assert det.check_directory_growth(str(d), warn_mb_per_hour=100.0) is None
    (d / "blob1.bin").write_bytes(b"x" * 2_000_000)   # +2 MB
    t[0] += 36.0                                       # 0.01 h
    assert det.check_directory_growth(str(d), warn_mb_per_hour=100.0) is None  # 2 pts: no slope yet
    (d / "blob2.bin").write_bytes(b"x" * 2_000_000)   # +2 MB more
    t[0] += 36.0                                       # ~200 MB/h over 3 points
    it = det.check_directory_growth(str(d), warn_mb_per_hour=100.0)
    assert it is not None and "growing" in it["title"]


# -- resolve_config tables (sitrep test/config.test.ts, 10 its) ---------------

def test_resolve_config_defaults_for_none_and_empty(tmp_path):
    from vainplex_openclaw_amd.leuko.plugin import default_config, resolve_config

    d = default_config(str(tmp_path))
    for raw in (None, {}):
        c = resolve_config(raw, home=str(tmp_path))
        assert c["enabled"] == d["enabled"]
        assert set(c["collectors"]) == set(d["collectors"])


def test_resolve_config_scalar_overrides(tmp_path):
    from vainplex_openclaw_amd.leuko.plugin import resolve_config

    c = resolve_config({"enabled": False, "outputPath": "/x/sitrep.md",
                        "intervalMinutes": 15}, home=str(tmp_path))
    assert c["enabled"] is False
    assert c["outputPath"] == "/x/sitrep.md"
    assert c["intervalMinutes"] == 15


def test_resolve_config_merges_collector_configs(tmp_path):
    from vainplex_openclaw_amd.leuko.plugin import default_config, resolve_config

    c = resolve_config({"collectors": {"goals": {"enabled": True,
                                                 "staleDays": 9}}},
                       home=str(tmp_path))
    assert c["collectors"]["goals"]["enabled"] is True
    assert c["collectors"]["goals"]["staleDays"] == 9
    # untouched collectors keep their defaults
    d = default_config(str(tmp_path))
    for name, base in d["collectors"].items():
        if name != "goals":
            assert c["collectors"][name] == base


def test_resolve_config_ignores_non_object_collectors(tmp_path):
    from vainplex_openclaw_amd.leuko.plugin import default_config, resolve_config

    c = resolve_config({"collectors": {"goals": "yes", "errors": 3}},
                       home=str(tmp_path))
    d = default_config(str(tmp_path))
    assert c["collectors"]["goals"] == d["collectors"]["goals"]
    assert c["collectors"]["errors"] == d["collectors"]["errors"]


def test_resolve_config_scoring_and_custom_collectors(tmp_path):
    from vainplex_openclaw_amd.leuko.plugin import resolve_config

    custom = [{"name": "disk", "command": "df -h", "threshold": 90}]
    c = resolve_config({"scoring": {"critical": 40},
                        "customCollectors": custom}, home=str(tmp_path))
    assert c["scoring"]["critical"] == 40
    assert c["customCollectors"] == custom


def test_resolve_config_unknown_collector_added(tmp_path):
    from vainplex_openclaw_amd.leuko.plugin import resolve_config

    c = resolve_config({"collectors": {"brandnew": {"enabled": True}}},
                       home=str(tmp_path))
    assert c["collectors"]["brandnew"]["enabled"] is True


# -- goal quality checks + LLM recommendations (SURVEY §2.7 Leuko surface) ----

def _goals_file(tmp_path, goals):
    import json

    p = tmp_path / "goals.json"
    p.write_text(json.dumps({"goals": goals}))
    return str(p)


def test_goal_quality_flags_vague_unmeasurable_unowned(tmp_path):
    from vainplex_openclaw_amd.leuko.quality import check_goal_quality

    path = _goals_file(tmp_path, [
        {"id": "g1", "title": "improve things", "status": "open"},
        {"id": "g2", "title": "", "status": "open"},
        {"id": "g3", "title": "reduce p95 latency below 200ms by friday",
         "status": "open"},
        {"id": "g4", "title": "migrate billing database to postgres 16",
         "zone": "red", "status": "approved"},
    ])
    r = check_goal_quality({"goalsPath": path})
    ids = [i["id"] for i in r["items"]]
    assert "goalq-g1-vague" in ids
    assert "goalq-g2-untitled" in ids
    assert "goalq-g4-red-unowned" in ids
    assert not any("g3" in i for i in ids)          # measurable goal is clean
    assert r["status"] == "warn"


def test_goal_quality_overload_and_clean(tmp_path):
    from vainplex_openclaw_amd.leuko.quality import check_goal_quality

    many = [{"id": f"g{i}", "title": f"ship feature {i} to 100% of users",
             "status": "open", "success_criteria": "rollout complete"}
            for i in range(9)]
    r = check_goal_quality({"goalsPath": _goals_file(tmp_path, many),
                            "maxOpenGoals": 7})
    assert any(i["id"] == "goalq-overload" for i in r["items"])
    few = many[:3]
    r2 = check_goal_quality({"goalsPath": _goals_file(tmp_path / "b"
                             if (tmp_path / "b").mkdir() is None else tmp_path,
                             few)})
    assert r2["items"] == [] and r2["status"] == "ok"


def test_goal_quality_missing_file_and_unconfigured(tmp_path):
    from vainplex_openclaw_amd.leuko.quality import check_goal_quality

    assert check_goal_quality({})["status"] == "ok"
    assert check_goal_quality({"goalsPath": str(tmp_path / "ghost.json")})["status"] == "ok"


def test_llm_recommendations_parse_validate_and_fail_soft():
    import json

    from vainplex_openclaw_amd.leuko.quality import llm_recommendations

    report = {"summary": "2 warnings", "health": 80,
              "items": [{"title": "timer failed", "severity": "warn",
                         "category": "auto_fixable"}]}
    good = json.dumps({"recommendations": [
        {"action": "restart the failed timer", "priority": "high", "reason": "stuck"},
        {"action": "archive stale goals", "priority": "weird", "reason": ""},
    ]})
    recs = llm_recommendations(report, lambda p: good)
    assert len(recs) == 2
    assert recs[0]["priority"] == "high"
    assert recs[1]["priority"] == "medium"            # invalid -> medium
    assert llm_recommendations(report, None) == []
    assert llm_recommendations(report, lambda p: "not json") == []

    def boom(p):
        raise TimeoutError("llm down")

    assert llm_recommendations(report, boom) == []
    capped = json.dumps({"recommendations": [
        {"action": f"a{i}"} for i in range(20)]})
    assert len(llm_recommendations(report, lambda p: capped, max_items=5)) == 5


def test_plugin_attaches_recommendations(tmp_path):
    import json

    from vainplex_openclaw_amd.leuko.plugin import LeukoPlugin, resolve_config

    p = LeukoPlugin(str(tmp_path), clock=lambda: 1_700_000_000.0,
                    call_llm=lambda prompt: json.dumps(
                        {"recommendations": [{"action": "do the thing"}]}))
    p.config = resolve_config({
        "outputPath": str(tmp_path / "sitrep.json"),
        "previousPath": str(tmp_path / "prev.json"),
        "collectors": {"systemd_timers": {"enabled": False},
                       "gpu_health": {"enabled": False}},
        "anomaly": {"enabled": False},
    }, home=str(tmp_path))
    report = p.run_once()
    assert report["recommendations"] == [
        {"action": "do the thing", "priority": "medium", "reason": ""}]
    on_disk = json.load(open(str(tmp_path / "sitrep.json")))
    assert on_disk["recommendations"][0]["action"] == "do the thing"


def test_goal_quality_collector_registered(tmp_path):
    from vainplex_openclaw_amd.leuko.collectors import BUILT_IN_COLLECTORS

    assert "goal_quality" in BUILT_IN_COLLECTORS
    r = BUILT_IN_COLLECTORS["goal_quality"]({"goalsPath": ""})
    assert r["status"] == "ok"
