"""Output validator mirror tests: trust-proportional verdicts, claim
policies, 3-stage merge — from the reference's `test/output-validator.test.ts`."""

import pytest

from vainplex_openclaw_amd.governance.output_validator import (
    DEFAULT_CONFIG,
    OutputValidator,
    more_restrictive,
)

NGINX_FACTS = [{"id": "t", "facts": [
    {"subject": "nginx", "predicate": "state", "value": "stopped"},
]}]


def make_validator(config=None, facts=NGINX_FACTS):
    cfg = {"factRegistries": facts}
    cfg.update(config or {})
    return OutputValidator(cfg)


def test_pass_when_disabled_or_empty():
    v = OutputValidator({"enabled": False})
    assert v.validate("nginx is running", 0)["verdict"] == "pass"
    v2 = make_validator()
    assert v2.validate("", 0)["verdict"] == "pass"
    assert v2.validate("just a friendly chat about nothing", 0)["verdict"] == "pass"


def test_pass_when_no_facts_configured():
    v = OutputValidator({"factRegistries": []})
    res = v.validate("nginx is running", 0)
    assert res["verdict"] == "pass"  # unverified ignored by default
    assert res["claims"]


TRUST_TABLE = [
    (0, "block"),
    (39, "block"),
    (40, "flag"),
    (59, "flag"),
    (60, "pass"),
    (100, "pass"),
]


@pytest.mark.parametrize("trust,want", TRUST_TABLE)
def test_contradiction_trust_thresholds(trust, want):
    v = make_validator()
    res = v.validate("nginx is running", trust)
    assert res["verdict"] == want, res["reason"]
    assert res["contradictions"]


def test_custom_thresholds():
    v = make_validator({"contradictionThresholds": {"flagAbove": 90, "blockBelow": 10}})
    assert v.validate("nginx is running", 9)["verdict"] == "block"
    assert v.validate("nginx is running", 50)["verdict"] == "flag"
    assert v.validate("nginx is running", 95)["verdict"] == "pass"


def test_verified_claim_passes():
    v = make_validator(facts=[{"id": "t", "facts": [
        {"subject": "nginx", "predicate": "state", "value": "running"}]}])
    res = v.validate("nginx is running", 0)
    assert res["verdict"] == "pass"
    assert any(r["status"] == "verified" for r in res["factCheckResults"])


def test_multiple_contradictions_reported():
    v = make_validator(facts=[{"id": "t", "facts": [
        {"subject": "nginx", "predicate": "state", "value": "stopped"},
        {"subject": "redis", "predicate": "state", "value": "stopped"},
    ]}])
    res = v.validate("nginx is running. redis is running.", 0)
    assert res["verdict"] == "block"
    assert len(res["contradictions"]) == 2
    assert "nginx" in res["reason"] and "redis" in res["reason"]


def test_unverified_claim_policy():
    flag = make_validator({"unverifiedClaimPolicy": "flag"}, facts=[])
    assert flag.validate("mystery-svc is running", 80)["verdict"] == "flag"
    block = make_validator({"unverifiedClaimPolicy": "block"}, facts=[])
    assert block.validate("mystery-svc is running", 80)["verdict"] == "block"
    # verified claims don't trip the unverified policy
    ok = make_validator({"unverifiedClaimPolicy": "flag"}, facts=[{"id": "t", "facts": [
        {"subject": "nginx", "predicate": "state", "value": "running"}]}])
    assert ok.validate("nginx is running", 80)["verdict"] == "pass"


def test_default_config_shape():
    assert DEFAULT_CONFIG["unverifiedClaimPolicy"] == "ignore"
    assert DEFAULT_CONFIG["selfReferentialPolicy"] == "ignore"
    assert DEFAULT_CONFIG["contradictionThresholds"] == {"flagAbove": 60, "blockBelow": 40}
    assert DEFAULT_CONFIG["enabled"] is True


def test_result_structure_and_perf():
    import time
    v = make_validator()
    res = v.validate("nginx is running", 50)
    assert isinstance(res["evaluationUs"], int) and res["evaluationUs"] >= 0
    assert isinstance(res["claims"], list) and isinstance(res["factCheckResults"], list)
    start = time.perf_counter()
    v.validate("nginx is running and the deploy finished with 3 errors", 50)
    assert (time.perf_counter() - start) < 0.1


# -- Stage 3: LLM merge ------------------------------------------------------

class StubLLM:
    def __init__(self, verdict="pass", reason="", raise_exc=False):
        self.verdict, self.reason, self.raise_exc = verdict, reason, raise_exc
        self.calls = 0

    def validate(self, text, facts, is_external):
        self.calls += 1
        if self.raise_exc:
            raise RuntimeError("llm down")
        return {"verdict": self.verdict, "reason": self.reason}


def llm_validator(stub, facts=NGINX_FACTS, **cfg_extra):
    cfg = {"llmValidator": {"enabled": True}}
    cfg.update(cfg_extra)
    v = make_validator(cfg, facts=facts)
    v.set_llm_validator(stub)
    return v


def test_llm_not_called_when_internal():
    stub = StubLLM()
    v = llm_validator(stub)
    v.validate("nginx is running", 80, is_external=False)
    v.validate("nginx is running", 80)  # is_external defaults False
    assert stub.calls == 0


def test_llm_called_for_external_and_block_overrides_pass():
    stub = StubLLM(verdict="block", reason="fabricated numbers")
    v = llm_validator(stub, facts=[])
    res = v.validate("all systems nominal at 99.99% uptime", 100, is_external=True)
    assert stub.calls == 1
    assert res["verdict"] == "block"
    assert "LLM: fabricated numbers" in res["reason"]
    assert res["llmResult"]["verdict"] == "block"


def test_stage12_block_prevails_over_llm_pass():
    stub = StubLLM(verdict="pass")
    v = llm_validator(stub)
    res = v.validate("nginx is running", 0, is_external=True)
    assert res["verdict"] == "block"


def test_most_restrictive_merge_flag():
    stub = StubLLM(verdict="flag", reason="tone")
    v = llm_validator(stub, facts=[])
    res = v.validate("everything shipped", 100, is_external=True)
    assert res["verdict"] == "flag"


def test_llm_failure_fails_open():
    stub = StubLLM(raise_exc=True)
    v = llm_validator(stub, facts=[])
    res = v.validate("everything shipped", 100, is_external=True)
    assert res["verdict"] == "pass"
    assert stub.calls == 1


def test_llm_skipped_when_disabled_or_absent():
    stub = StubLLM(verdict="block")
    v = llm_validator(stub, llmValidator={"enabled": False})
    assert v.validate("nginx is running", 100, is_external=True)["verdict"] == "pass"
    assert stub.calls == 0
    v2 = make_validator({"llmValidator": {"enabled": True}})  # no validator set
    assert v2.validate("nginx is running", 100, is_external=True)["verdict"] == "pass"


def test_more_restrictive_order():
    assert more_restrictive("pass", "flag") == "flag"
    assert more_restrictive("block", "flag") == "block"
    assert more_restrictive("pass", "pass") == "pass"
    assert more_restrictive("flag", "block") == "block"
