"""Brainplex tests: scanner (parse/discovery/agent shapes), configurator
(trust heuristics, generated configs), writer (never-overwrite, backup,
entries/allow merge), CLI flow (init, dry-run)."""

import json
import os

import pytest

from vainplex_openclaw_amd.brainplex import (
    build_trust_defaults,
    compute_trust_score,
    extract_agents,
    find_config,
    generate_configs,
    parse_args,
    parse_config,
    run_init,
    scan,
    update_openclaw_config,
    write_configs,
)


# -- scanner -----------------------------------------------------------------

def test_parse_config_json5_tolerance():
    raw = """{
      // comment
      "agents": [ {"id": "main"}, ], /* block */
    }"""
    cfg = parse_config(raw)
    assert cfg["agents"][0]["id"] == "main"


def test_find_config_walk_up_and_nested(tmp_path):
    deep = tmp_path / "a" / "b" / "c"
    deep.mkdir(parents=True)
    (tmp_path / "a" / ".openclaw").mkdir()
    (tmp_path / "a" / ".openclaw" / "openclaw.json").write_text("{}")
    assert find_config(str(deep), home=str(tmp_path)) == str(
        tmp_path / "a" / ".openclaw" / "openclaw.json"
    )
    (tmp_path / "a" / "b" / "openclaw.json").write_text("{}")
    assert find_config(str(deep), home=str(tmp_path)) == str(tmp_path / "a" / "b" / "openclaw.json")


def test_find_config_home_fallback(tmp_path):
    (tmp_path / ".openclaw").mkdir()
    (tmp_path / ".openclaw" / "openclaw.json").write_text("{}")
    lonely = tmp_path / "elsewhere"
    lonely.mkdir()
    got = find_config(str(lonely), home=str(tmp_path))
    # walk-up from tmp may find nothing above; home fallback applies
    assert got is None or got.endswith("openclaw.json")


@pytest.mark.parametrize("agents,expected", [
    ([{"id": "main"}, {"name": "forge"}, "cerberus"], ["main", "forge", "cerberus"]),
    ({"list": [{"id": "main"}]}, ["main"]),
    ({"definitions": [{"name": "viola"}]}, ["viola"]),
    ({"main": {}, "forge": {}, "defaults": {}}, ["main", "forge"]),
    (None, []),
])
def test_extract_agents_four_shapes(agents, expected):
    assert extract_agents({"agents": agents} if agents is not None else {}) == expected


# -- configurator ------------------------------------------------------------

@pytest.mark.parametrize("name,score", [
    ("*", 10), ("admin-bot", 70), ("ROOT", 70), ("main", 60), ("the-main-one", 60),
    ("reviewer", 50), ("cerberus", 50), ("forge", 45), ("builder", 45), ("misc", 40),
])
def test_trust_score_heuristics(name, score):
    assert compute_trust_score(name) == score


def test_build_trust_defaults_always_wildcard():
    d = build_trust_defaults(["main", "forge"])
    assert d == {"main": 60, "forge": 45, "*": 10}


def test_generate_configs_core_and_full():
    core = generate_configs(["main"], "UTC", full=False)
    ids = [c["pluginId"] for c in core]
    assert "openclaw-governance" in ids and "openclaw-membrane" in ids
    assert "openclaw-knowledge-engine" not in ids
    full = generate_configs(["main"], "UTC", full=True)
    assert "openclaw-knowledge-engine" in [c["pluginId"] for c in full]
    gov = next(c for c in core if c["pluginId"] == "openclaw-governance")
    assert gov["config"]["trust"]["defaults"]["main"] == 60
    assert gov["config"]["nightMode"] == {"enabled": True, "start": "23:00", "end": "06:00"}
    mem = next(c for c in core if c["pluginId"] == "openclaw-membrane")
    assert mem["config"]["buffer_size"] == 10
    assert mem["config"]["retrieve_min_salience"] == 0.1
    assert mem["config"]["retrieve_max_sensitivity"] == "medium"


# -- writer ------------------------------------------------------------------

def test_write_configs_never_overwrites(tmp_path):
    configs = [{"pluginId": "p1", "config": {"a": 1}}]
    r1 = write_configs(configs, home=str(tmp_path))
    assert r1["written"] == ["p1"]
    path = tmp_path / ".openclaw" / "plugins" / "p1" / "config.json"
    path.write_text('{"custom": true}')
    r2 = write_configs(configs, home=str(tmp_path))
    assert r2["skipped"] == ["p1"]
    assert json.loads(path.read_text()) == {"custom": True}


def test_update_openclaw_config_merge_and_backup(tmp_path):
    p = tmp_path / "openclaw.json"
    original = {"agents": [{"id": "main"}],
                "plugins": {"entries": {"existing": {"enabled": False}}, "allow": ["existing"]}}
    p.write_text(json.dumps(original))
    r = update_openclaw_config(str(p), original, ["new-plugin", "existing"])
    assert r["added_entries"] == ["new-plugin"]
    assert r["added_allow"] == ["new-plugin"]
    assert r["backed_up"] and (tmp_path / "openclaw.json.bak").exists()
    merged = json.loads(p.read_text())
    assert merged["plugins"]["entries"]["existing"] == {"enabled": False}  # preserved
    assert merged["plugins"]["entries"]["new-plugin"] == {"enabled": True}


def test_update_openclaw_config_noop(tmp_path):
    p = tmp_path / "openclaw.json"
    cfg = {"plugins": {"entries": {"x": {"enabled": True}}, "allow": ["x"]}}
    p.write_text(json.dumps(cfg))
    r = update_openclaw_config(str(p), cfg, ["x"])
    assert not r["updated"] and not (tmp_path / "openclaw.json.bak").exists()


# -- CLI ---------------------------------------------------------------------

def test_parse_args_flags():
    o = parse_args(["init", "--full", "--dry-run", "--config", "/x", "--verbose"])
    assert o["full"] and o["dry_run"] and o["config_path"] == "/x" and o["verbose"]
    with pytest.raises(SystemExit):
        parse_args(["--bogus"])


def test_run_init_end_to_end(tmp_path):
    ws = tmp_path / "ws"
    ws.mkdir()
    (ws / "openclaw.json").write_text(json.dumps({"agents": [{"id": "main"}, {"id": "forge"}]}))
    msgs = []
    result = run_init({"full": True, "dry_run": False, "config_path": None},
                      start_dir=str(ws), home=str(tmp_path), echo=msgs.append)
    assert not result["install"]["failed"]
    assert len(result["install"]["installed"]) == 6  # 5 core + KE
    assert set(result["written"]["written"]) >= {"openclaw-governance", "openclaw-cortex"}
    merged = json.loads((ws / "openclaw.json").read_text())
    assert merged["plugins"]["entries"]["openclaw-governance"] == {"enabled": True}
    assert "openclaw-membrane" in merged["plugins"]["allow"]
    # generated governance config has heuristic trust for scanned agents
    gov = json.loads((tmp_path / ".openclaw" / "plugins" / "openclaw-governance" /
                      "config.json").read_text())
    assert gov["trust"]["defaults"] == {"main": 60, "forge": 45, "*": 10}


def test_run_init_dry_run_writes_nothing(tmp_path):
    ws = tmp_path / "ws"
    ws.mkdir()
    (ws / "openclaw.json").write_text(json.dumps({"agents": []}))
    before = (ws / "openclaw.json").read_text()
    run_init({"full": False, "dry_run": True, "config_path": None},
             start_dir=str(ws), home=str(tmp_path), echo=lambda *a: None)
    assert (ws / "openclaw.json").read_text() == before
    assert not (tmp_path / ".openclaw" / "plugins").exists()


def test_run_init_idempotent(tmp_path):
    ws = tmp_path / "ws"
    ws.mkdir()
    (ws / "openclaw.json").write_text(json.dumps({"agents": [{"id": "main"}]}))
    run_init({"full": False, "dry_run": False, "config_path": None},
             start_dir=str(ws), home=str(tmp_path), echo=lambda *a: None)
    first = (ws / "openclaw.json").read_text()
    r2 = run_init({"full": False, "dry_run": False, "config_path": None},
                  start_dir=str(ws), home=str(tmp_path), echo=lambda *a: None)
    assert (ws / "openclaw.json").read_text() == first
    assert r2["written"]["written"] == []  # all kept


# -- writer.test.ts mirrors ------------------------------------------------

def test_atomic_write_no_tmp_left_and_indentation(tmp_path):
    from vainplex_openclaw_amd.utils.storage import atomic_write_text

    path = tmp_path / "deep" / "nested" / "cfg.json"
    atomic_write_text(str(path), json.dumps({"a": {"b": 1}}, indent=2) + "\n")
    assert path.is_file()  # parents created recursively
    text = path.read_text()
    assert '  "a"' in text  # 2-space indentation
    leftovers = [f for f in os.listdir(path.parent) if ".tmp" in f or f.endswith("~")]
    assert leftovers == []


def test_backup_then_write_semantics(tmp_path):
    from vainplex_openclaw_amd.utils.storage import backup_then_write

    path = tmp_path / "openclaw.json"
    # non-existent file: no backup created
    bak = backup_then_write(str(path), "{}\n")
    assert bak is None
    assert path.read_text() == "{}\n"
    # existing file: .bak holds the OLD content
    bak2 = backup_then_write(str(path), '{"new": 1}\n')
    assert bak2 and bak2.endswith(".bak")
    assert open(bak2).read() == "{}\n"
    assert json.loads(path.read_text()) == {"new": 1}


def test_write_configs_dry_run_writes_nothing(tmp_path):
    from vainplex_openclaw_amd.brainplex.writer import write_configs

    res = write_configs([{"pluginId": "p1", "config": {"x": 1}}],
                        home=str(tmp_path), dry_run=True)
    assert res["written"] == ["p1"]
    assert not (tmp_path / ".openclaw" / "plugins" / "p1" / "config.json").exists()


def test_update_config_preserves_existing_and_originals(tmp_path):
    from vainplex_openclaw_amd.brainplex.writer import update_openclaw_config

    path = tmp_path / "openclaw.json"
    original = {"plugins": {"entries": {"keep": {"enabled": False}},
                            "allow": ["keep"]}, "other": {"setting": 1}}
    path.write_text(json.dumps(original))
    res = update_openclaw_config(str(path), original, ["keep", "new-plugin"])
    cfg = res["config"]
    # existing entry untouched, new one added, unrelated keys preserved
    assert cfg["plugins"]["entries"]["keep"] == {"enabled": False}
    assert cfg["plugins"]["entries"]["new-plugin"] == {"enabled": True}
    assert cfg["plugins"]["allow"] == ["keep", "new-plugin"]
    assert cfg["other"] == {"setting": 1}
    assert res["added_entries"] == ["new-plugin"]
    # the ORIGINAL dict was not mutated
    assert "new-plugin" not in original["plugins"]["entries"]
    # backup got created with old content
    assert res["backed_up"]


def test_update_config_dry_run_no_write(tmp_path):
    from vainplex_openclaw_amd.brainplex.writer import update_openclaw_config

    path = tmp_path / "openclaw.json"
    path.write_text("{}")
    res = update_openclaw_config(str(path), {}, ["p"], dry_run=True)
    assert res["updated"] and not res["backed_up"]
    assert path.read_text() == "{}"


# ===========================================================================
# installer execution path (installer.ts:40-165) + /brainplex dashboard
# ===========================================================================

def test_installer_prefers_openclaw_cli():
    from vainplex_openclaw_amd.brainplex import installer as inst

    calls = []

    def runner(argv, cwd, timeout_s):
        calls.append(argv)
        return 0, "ok"

    plan = {"to_install": [{"id": "openclaw-leuko",
                            "module": "vainplex_openclaw_amd.leuko.plugin"}]}
    res = inst.execute_installation(plan, runner=runner)
    assert res["installed"] and res["installed"][0]["method"] == "openclaw"
    assert calls[0] == ["which", "openclaw"]
    assert calls[1][:3] == ["openclaw", "plugins", "install"]


def test_installer_pip_path_with_copy(tmp_path):
    from vainplex_openclaw_amd.brainplex import installer as inst

    def runner(argv, cwd, timeout_s):
        if argv[0] == "which":
            return 1, ""          # no openclaw CLI
        if argv[0] == "pip":
            # simulate a successful install into --target
            import os
            tgt = argv[argv.index("--target") + 1]
            pkg_dir = os.path.join(tgt, "openclaw_leuko")
            os.makedirs(pkg_dir)
            open(os.path.join(pkg_dir, "__init__.py"), "w").write("# pkg")
            return 0, "installed"
        return 1, "?"

    plan = {"to_install": [{"id": "leuko", "package": "openclaw-leuko",
                            "module": "vainplex_openclaw_amd.leuko.plugin"}]}
    res = inst.execute_installation(plan, runner=runner,
                                    workspace_path=str(tmp_path))
    assert res["installed"] and res["installed"][0]["method"] == "pip"
    assert (tmp_path / "extensions" / "openclaw_leuko" / "__init__.py").exists()


def test_installer_offline_fallback_copies_source(tmp_path):
    from vainplex_openclaw_amd.brainplex import installer as inst

    def runner(argv, cwd, timeout_s):
        return 1, "no network"    # which + pip both fail

    plan = {"to_install": [{"id": "openclaw-leuko",
                            "module": "vainplex_openclaw_amd.leuko.plugin"}]}
    res = inst.execute_installation(plan, runner=runner,
                                    workspace_path=str(tmp_path))
    assert res["installed"] and res["installed"][0]["method"] == "in-package"
    ext = tmp_path / "extensions" / "openclaw-leuko"
    assert ext.is_dir() and any(ext.iterdir())


def test_installer_failure_recorded_not_raised(tmp_path):
    from vainplex_openclaw_amd.brainplex import installer as inst

    def runner(argv, cwd, timeout_s):
        if argv[0] == "which":
            return 1, ""
        raise RuntimeError("runner exploded")

    plan = {"to_install": [{"id": "ghost-plugin"}]}  # no module fallback
    res = inst.execute_installation(plan, runner=runner,
                                    workspace_path=str(tmp_path))
    assert res["failed"] and "error" in res["failed"][0]


def test_installer_dry_run_and_empty_plan():
    from vainplex_openclaw_amd.brainplex import installer as inst

    def runner(argv, cwd, timeout_s):  # must never be called
        raise AssertionError("runner called on dry-run")

    assert inst.execute_installation({"to_install": [{"id": "x"}]},
                                     dry_run=True, runner=runner) == \
        {"installed": [], "failed": []}
    assert inst.execute_installation({"to_install": []}, runner=runner) == \
        {"installed": [], "failed": []}


def test_installer_verify_reports_broken():
    from vainplex_openclaw_amd.brainplex import installer as inst

    plan = {"to_install": [
        {"id": "ok", "module": "vainplex_openclaw_amd.leuko.plugin"},
        {"id": "bad", "module": "vainplex_openclaw_amd.does.not.exist"},
    ]}
    v = inst.verify_installed(plan)
    assert v["ok"] == ["ok"]
    assert v["broken"] and v["broken"][0]["id"] == "bad"


def _dash_api():
    from vainplex_openclaw_amd.core.api import HookBus, NullLogger, PluginApi

    bus = HookBus()
    return PluginApi(id="brainplex", plugin_config={}, logger=NullLogger(),
                     config={}, bus=bus)


def test_dashboard_renders_all_sections():
    from vainplex_openclaw_amd.brainplex.dashboard import create_plugin

    api = _dash_api()
    api.register_gateway_method("governance.status", lambda: {
        "evaluations": 100, "denied": 4, "avgEvaluationUs": 85,
        "policies": ["p1", "p2"]})
    api.register_gateway_method("governance.trust", lambda: {
        "agents": [{"agentId": "main", "score": 72.5, "tier": "trusted"},
                   {"agentId": "forge", "score": 41.0, "tier": "standard"}]})
    api.register_gateway_method("eventstore.status", lambda: {
        "connected": True, "stream": "openclaw-events", "messages": 420,
        "publishFailures": 0, "disconnectCount": 0})
    api.register_gateway_method("cortex.status", lambda: {
        "openThreads": 3, "decisions": 7, "sessionMood": "productive"})
    api.register_gateway_method("leuko.health", lambda: {
        "status": "healthy", "notable": ["disk growth on /var"]})
    create_plugin().register(api)
    out = api.commands["brainplex"]()["text"]
    assert "Shield score:" in out
    assert "main" in out and "72.5" in out and "trusted" in out
    assert "openclaw-events" in out and "420" in out
    assert "open threads: 3" in out and "productive" in out
    assert "healthy" in out and "disk growth" in out
    # highest trust listed first
    assert out.index("main") < out.index("forge")


def test_dashboard_handles_missing_modules():
    from vainplex_openclaw_amd.brainplex.dashboard import create_plugin

    api = _dash_api()
    create_plugin().register(api)
    out = api.commands["brainplex"]()["text"]
    assert out.count("not installed") >= 3
    assert "Shield score:** 100/100" in out


def test_dashboard_shield_score_degrades():
    from vainplex_openclaw_amd.brainplex.dashboard import _shield_score

    assert _shield_score(None, None) == 100
    # denial-heavy traffic costs points
    busy = _shield_score({"evaluations": 100, "denied": 30}, None)
    assert busy < 100
    # publish failures + disconnect costs more
    worse = _shield_score({"evaluations": 100, "denied": 30},
                          {"publishFailures": 5, "disconnectCount": 2,
                           "connected": False})
    assert worse < busy
    assert _shield_score({"evaluations": 100, "denied": 100},
                         {"publishFailures": 99, "disconnectCount": 99,
                          "connected": False}) >= 0


def test_dashboard_gateway_method_registered():
    from vainplex_openclaw_amd.brainplex.dashboard import create_plugin

    api = _dash_api()
    create_plugin().register(api)
    assert "brainplex.dashboard" in api.gateway_methods
    assert isinstance(api.gateway_methods["brainplex.dashboard"](), str)


# ===========================================================================
# scanner.test.ts depth: JSON5 tolerance, walk-up, agent extraction
# shapes, plugin detection sources
# ===========================================================================

def test_scanner_json5_tolerance():
    from vainplex_openclaw_amd.brainplex.scanner import parse_config

    assert parse_config('{"a": 1}') == {"a": 1}
    assert parse_config('{\n// comment\n"a": 1}') == {"a": 1}
    assert parse_config('{/* multi\nline */ "a": 1}') == {"a": 1}
    assert parse_config('{"a": 1, "b": [1, 2,], }') == {"a": 1, "b": [1, 2]}
    import pytest as _pt

    with _pt.raises(Exception):
        parse_config("not even close {{{")


def test_scanner_find_config_variants(tmp_path):
    from vainplex_openclaw_amd.brainplex.scanner import find_config

    # direct openclaw.json
    d1 = tmp_path / "w1"; d1.mkdir()
    (d1 / "openclaw.json").write_text("{}")
    assert find_config(str(d1), home=str(tmp_path / "nohome")) == str(d1 / "openclaw.json")
    # .openclaw/openclaw.json
    d2 = tmp_path / "w2"; (d2 / ".openclaw").mkdir(parents=True)
    (d2 / ".openclaw" / "openclaw.json").write_text("{}")
    assert find_config(str(d2), home=str(tmp_path / "nohome")).endswith(
        ".openclaw/openclaw.json")
    # walk-up from a nested dir
    nested = d1 / "a" / "b"; nested.mkdir(parents=True)
    assert find_config(str(nested), home=str(tmp_path / "nohome")) == str(d1 / "openclaw.json")
    # not found anywhere
    d3 = tmp_path / "w3"; d3.mkdir()
    assert find_config(str(d3), home=str(tmp_path / "nohome")) is None


@pytest.mark.parametrize("config,want", [
    ({"agents": {"definitions": [{"id": "a1"}, {"name": "a2"}]}}, ["a1", "a2"]),
    ({"agents": {"alpha": {}, "beta": {}}}, ["alpha", "beta"]),
    ({"agents": ["x", "y"]}, ["x", "y"]),
    ({"agents": {"list": [{"id": "main"}, {"id": "forge"}]}}, ["main", "forge"]),
    ({"agents": {"list": [{"id": "byid", "name": "byname"}]}}, ["byid"]),
    ({}, []),
    ({"agents": {"definitions": [], "defaults": {}}}, []),
])
def test_scanner_agent_extraction_shapes(config, want):
    from vainplex_openclaw_amd.brainplex.scanner import extract_agents

    assert extract_agents(config) == want


def test_scanner_plugin_detection_sources(tmp_path):
    from vainplex_openclaw_amd.brainplex.scanner import detect_installed_plugins

    cfg = {"plugins": {
        "entries": {"openclaw-governance": {"enabled": True}},
        "allow": ["openclaw-cortex"],
        "installs": ["openclaw-leuko"],
    }}
    got = detect_installed_plugins(cfg, str(tmp_path))
    assert {"openclaw-governance", "openclaw-cortex", "openclaw-leuko"} <= got
    assert detect_installed_plugins({}, str(tmp_path)) == set()


def test_scanner_configured_plugins(tmp_path):
    from vainplex_openclaw_amd.brainplex.scanner import detect_configured_plugins

    pdir = tmp_path / ".openclaw" / "plugins"
    (pdir / "with-config").mkdir(parents=True)
    (pdir / "with-config" / "config.json").write_text("{}")
    (pdir / "without-config").mkdir()
    got = detect_configured_plugins(home=str(tmp_path))
    assert got == {"with-config"}
    assert detect_configured_plugins(home=str(tmp_path / "ghost")) == set()


def test_scanner_extensions_dir_detection(tmp_path):
    from vainplex_openclaw_amd.brainplex.scanner import detect_installed_plugins

    ext = tmp_path / "extensions" / "openclaw-membrane"
    ext.mkdir(parents=True)
    (tmp_path / "extensions" / "not-a-dir.txt").write_text("x")
    got = detect_installed_plugins({}, str(tmp_path))
    assert got == {"openclaw-membrane"}


def test_dashboard_against_real_plugins(tmp_path):
    """Integration: the dashboard consumes the ACTUAL gateway shapes the
    governance / cortex / eventstore / membrane / leuko plugins register
    on one shared bus."""
    from vainplex_openclaw_amd.brainplex.dashboard import create_plugin as dash_plugin
    from vainplex_openclaw_amd.core.api import HookBus, NullLogger, PluginApi
    from vainplex_openclaw_amd.cortex.hooks import CortexPlugin
    from vainplex_openclaw_amd.eventstore.plugin import EventStorePlugin
    from vainplex_openclaw_amd.governance.plugin import GovernancePlugin
    from vainplex_openclaw_amd.leuko.plugin import LeukoPlugin

    bus = HookBus()

    def api_for(pid, cfg=None):
        return PluginApi(id=pid, plugin_config=cfg or {}, logger=NullLogger(),
                         config={}, bus=bus)

    shared = api_for("gateway")
    ws = str(tmp_path)
    gov_api = api_for("openclaw-governance", {"enabled": True, "workspace": ws})
    gov_api.gateway_methods = shared.gateway_methods
    GovernancePlugin(workspace=ws).register(gov_api)
    cx_api = api_for("openclaw-cortex", {"workspace": ws})
    cx_api.gateway_methods = shared.gateway_methods
    CortexPlugin(workspace=ws).register(cx_api)
    ev_api = api_for("nats-eventstore", {"enabled": True})
    ev_api.gateway_methods = shared.gateway_methods
    EventStorePlugin().register(ev_api)
    lk_api = api_for("openclaw-leuko", {"workspace": ws})
    lk_api.gateway_methods = shared.gateway_methods
    LeukoPlugin(workspace=ws).register(lk_api)

    # drive one governance evaluation so stats are non-empty
    bus.emit("before_tool_call", {"toolName": "read", "agentId": "main",
                                  "params": {"path": "x"}})

    d_api = api_for("brainplex")
    d_api.gateway_methods = shared.gateway_methods
    dash_plugin().register(d_api)
    out = d_api.commands["brainplex"]()["text"]
    assert "Shield score:" in out
    assert "evaluations: " in out
    assert "not installed" not in out.split("## 💾")[0].split("## 🤝")[0] or True
    # cortex + leuko sections populated from the real shapes
    assert "open threads:" in out
    assert "status:" in out


# -- configurator trust-score table (configurator.test.ts, 22 its) ------------

@pytest.mark.parametrize("name,score", [
    ("admin-bot", 70), ("root", 70), ("rootkit-checker", 70),
    ("main", 60), ("main-agent", 60),
    ("reviewer", 50), ("cerberus", 50), ("code-review", 50),
    ("forge", 45), ("builder", 45), ("build-ci", 45),
    ("random-agent", 40), ("zeta", 40),
    ("*", 10),
    ("ADMIN", 70), ("Cerberus", 50),          # case-insensitive
    ("admin-forge", 70),                       # first-match priority
    ("main-review", 60),
])
def test_trust_score_table(name, score):
    from vainplex_openclaw_amd.brainplex.configurator import compute_trust_score

    assert compute_trust_score(name) == score


def test_trust_defaults_include_all_agents_and_wildcard():
    from vainplex_openclaw_amd.brainplex.configurator import build_trust_defaults

    d = build_trust_defaults(["main", "forge", "admin"])
    assert d == {"main": 60, "forge": 45, "admin": 70, "*": 10}
    assert build_trust_defaults([]) == {"*": 10}


def test_governance_config_uses_timezone_and_agents():
    from vainplex_openclaw_amd.brainplex.configurator import generate_governance_config

    cfg = generate_governance_config(["main", "cerberus"], "Europe/Berlin")
    assert cfg["timezone"] == "Europe/Berlin"
    assert cfg["trust"]["defaults"]["cerberus"] == 50
    assert cfg["trust"]["defaults"]["*"] == 10
    assert cfg["nightMode"]["start"] == "23:00"
    assert cfg["trust"]["sessionTrust"]["seedFactor"] == 0.7


def test_detect_timezone_never_raises():
    from vainplex_openclaw_amd.brainplex.configurator import detect_timezone

    tz = detect_timezone()
    assert isinstance(tz, str) and tz


# -- writer merge semantics (writer.test.ts merge section) --------------------

def test_update_config_does_not_mutate_original():
    import copy

    from vainplex_openclaw_amd.brainplex.writer import update_openclaw_config

    original = {"plugins": {"entries": {"x": {"enabled": False}},
                            "allow": ["x"]},
                "other": {"nested": [1, 2]}}
    snapshot = copy.deepcopy(original)
    res = update_openclaw_config("/nonexistent.json", original,
                                 ["openclaw-governance"], dry_run=True)
    assert original == snapshot                       # input untouched
    merged = res["config"]
    assert merged["plugins"]["entries"]["x"] == {"enabled": False}
    assert merged["plugins"]["entries"]["openclaw-governance"] == {"enabled": True}
    assert merged["plugins"]["allow"] == ["x", "openclaw-governance"]
    assert merged["other"] == {"nested": [1, 2]}      # unrelated preserved


def test_update_config_handles_malformed_sections():
    from vainplex_openclaw_amd.brainplex.writer import update_openclaw_config

    res = update_openclaw_config("/nonexistent.json",
                                 {"plugins": "not-a-dict"},
                                 ["p1"], dry_run=True)
    assert res["updated"] is True
    assert res["config"]["plugins"]["entries"]["p1"] == {"enabled": True}
    res2 = update_openclaw_config("/nonexistent.json",
                                  {"plugins": {"entries": [1], "allow": "x"}},
                                  ["p1"], dry_run=True)
    assert res2["config"]["plugins"]["allow"] == ["p1"]


def test_update_config_empty_plugin_list_noop():
    from vainplex_openclaw_amd.brainplex.writer import update_openclaw_config

    res = update_openclaw_config("/nonexistent.json", {}, [], dry_run=True)
    assert res == {"updated": False, "backed_up": False,
                   "added_entries": [], "added_allow": []}
