"""Governance config-resolution tables mirroring `test/config.test.ts`
(27 its): defaults, nested overrides, invalid-value rejection,
mixed-type tolerance, output-validation and llmValidator resolution.
"""

import pytest

from vainplex_openclaw_amd.governance.config import (
    resolve_audit,
    resolve_config,
    resolve_llm_validator,
    resolve_output_validation,
    resolve_performance,
    resolve_trust,
)


@pytest.mark.parametrize("raw", [None, {}])
def test_full_defaults(raw):
    cfg = resolve_config(raw)
    assert cfg["enabled"] is True
    assert cfg["failMode"] == "open"
    assert cfg["timezone"] == "UTC"
    assert cfg["policies"] == []
    assert cfg["trust"]["enabled"] is True
    assert cfg["audit"]["enabled"] is True
    for key in ("builtinPolicies", "outputValidation", "performance",
                "responseGate", "redaction"):
        assert key in cfg


def test_trust_defaults_and_nested_override():
    t = resolve_trust(None)
    assert t["enabled"] is True and "decay" in t
    t2 = resolve_trust({"decay": {"enabled": False},
                        "sessionTrust": {"seedFactor": 0.5}})
    assert t2["decay"]["enabled"] is False
    assert t2["sessionTrust"]["seedFactor"] == 0.5
    # untouched siblings keep defaults
    assert t2["enabled"] is True


def test_audit_defaults_and_override():
    a = resolve_audit(None)
    assert a["enabled"] is True and isinstance(a.get("retentionDays"), (int, float))
    a2 = resolve_audit({"retentionDays": 7, "level": "minimal"})
    assert a2["retentionDays"] == 7 and a2["level"] == "minimal"


def test_invalid_fail_mode_and_level_rejected():
    assert resolve_config({"failMode": "yolo"})["failMode"] == "open"
    assert resolve_config({"failMode": "closed"})["failMode"] == "closed"
    a = resolve_audit({"level": "verbose-nonsense"})
    assert a["level"] in ("standard", "full", "minimal")


def test_non_record_inputs_graceful():
    for garbage in ("string", 42, [1, 2], True):
        cfg = resolve_config(garbage if isinstance(garbage, dict) else {})
        assert cfg["enabled"] is True
    assert resolve_trust("nope")["enabled"] is True
    assert resolve_audit(123)["enabled"] is True
    assert resolve_performance([])  # returns default dict


def test_non_typed_values_fall_back():
    cfg = resolve_config({"enabled": "yes", "timezone": 5})
    assert cfg["enabled"] is True        # non-bool -> default
    assert cfg["timezone"] == "UTC"      # non-string -> default


def test_output_validation_defaults_and_custom():
    ov = resolve_output_validation(None)
    assert isinstance(ov["enabled"], bool)
    assert set(ov["detectors"]) <= {
        "system_state", "entity_name", "existence",
        "operational_status", "self_referential"}
    assert ov["unverifiedClaimPolicy"] == "ignore"
    ov2 = resolve_output_validation({
        "enabled": True,
        "claimDetectors": ["system_state", "fake_detector", 42],
        "unverifiedClaimPolicy": "annotate-nonsense",
        "selfReferentialPolicy": "block",
    })
    assert ov2["enabled"] is True
    assert ov2["detectors"] == ["system_state"]        # invalid ids filtered
    assert ov2["unverifiedClaimPolicy"] == "ignore"    # invalid -> ignore
    assert ov2["selfReferentialPolicy"] == "block"
    # reference's enabledDetectors spelling wins when present
    ov3 = resolve_output_validation({"enabledDetectors": ["existence"]})
    assert ov3["detectors"] == ["existence"]


def test_llm_validator_resolution():
    default = resolve_llm_validator(None)
    assert default.get("enabled") in (False, None)
    custom = resolve_llm_validator({
        "enabled": True, "model": "mistral:7b",
        "externalChannels": ["twitter", 42, "email"],
    })
    assert custom["enabled"] is True
    assert custom["model"] == "mistral:7b"
    assert custom["externalChannels"] == ["twitter", "email"]  # 42 filtered


def test_policies_array_passthrough_and_garbage():
    cfg = resolve_config({"policies": [{"id": "p1"}]})
    assert cfg["policies"] == [{"id": "p1"}]
    assert resolve_config({"policies": "nope"})["policies"] == []


def test_builtin_policy_toggles():
    cfg = resolve_config({"builtinPolicies": {"nightMode": True}})
    assert cfg["builtinPolicies"]["nightMode"] is True
    assert cfg["builtinPolicies"]["credentialGuard"] is False


def test_performance_defaults():
    p = resolve_performance(None)
    assert isinstance(p, dict) and p


# -- layered loading through the governance plugin (config-loader.test.ts) ---

def test_gov_plugin_legacy_inline_config(tmp_path, monkeypatch):
    import json

    from vainplex_openclaw_amd.core.api import HookBus, NullLogger, PluginApi
    from vainplex_openclaw_amd.governance.plugin import GovernancePlugin

    monkeypatch.setenv("OPENCLAW_HOME", str(tmp_path))
    api = PluginApi(id="openclaw-governance",
                    plugin_config={"enabled": True, "failMode": "closed"},
                    logger=NullLogger(), config={}, bus=HookBus())
    p = GovernancePlugin(workspace=str(tmp_path))
    p.register(api)
    assert p.engine.config["failMode"] == "closed"   # legacy inline won


def test_gov_plugin_config_path_pointer(tmp_path, monkeypatch):
    import json
    import os

    from vainplex_openclaw_amd.core.api import HookBus, NullLogger, PluginApi
    from vainplex_openclaw_amd.governance.plugin import GovernancePlugin

    monkeypatch.setenv("OPENCLAW_HOME", str(tmp_path))
    ext = tmp_path / "gov.json"
    ext.write_text(json.dumps({"enabled": True, "failMode": "closed",
                               "audit": {"enabled": False}}))
    api = PluginApi(id="openclaw-governance",
                    plugin_config={"configPath": str(ext)},
                    logger=NullLogger(), config={}, bus=HookBus())
    p = GovernancePlugin(workspace=str(tmp_path))
    p.register(api)
    assert p.engine.config["failMode"] == "closed"
    assert p.engine.config["audit"]["enabled"] is False


def test_gov_plugin_external_file_default_location(tmp_path, monkeypatch):
    import json
    import os

    from vainplex_openclaw_amd.core.api import HookBus, NullLogger, PluginApi
    from vainplex_openclaw_amd.governance.plugin import GovernancePlugin

    monkeypatch.setenv("OPENCLAW_HOME", str(tmp_path))
    d = tmp_path / "plugins" / "openclaw-governance"
    os.makedirs(d)
    (d / "config.json").write_text(json.dumps({"failMode": "closed"}))
    # pointer-only inline: enabled override applies on the file config
    api = PluginApi(id="openclaw-governance", plugin_config={"enabled": True},
                    logger=NullLogger(), config={}, bus=HookBus())
    p = GovernancePlugin(workspace=str(tmp_path))
    p.register(api)
    assert p.engine.config["failMode"] == "closed"
    assert p.engine.config["enabled"] is True
