"""Fact-checker mirror tests: registry indexing/lookup, claim verdicts,
value normalization, file loading, perf — from the reference's
`test/fact-checker.test.ts` case list."""

import json
import os
import time

from vainplex_openclaw_amd.governance.facts import (
    FactRegistry,
    check_claim,
    check_claims,
)


def make_claim(**overrides):
    c = {"type": "system_state", "subject": "nginx", "predicate": "state",
         "value": "running", "source": "nginx is running", "offset": 0}
    c.update(overrides)
    return c


def make_registry(facts):
    return FactRegistry([{"id": "test", "facts": facts}])


# -- registry ----------------------------------------------------------------

def test_empty_registry():
    assert FactRegistry([]).size == 0
    assert FactRegistry().size == 0


def test_indexes_multiple_configs():
    reg = FactRegistry([
        {"id": "a", "facts": [{"subject": "nginx", "predicate": "state", "value": "running"}]},
        {"id": "b", "facts": [{"subject": "redis", "predicate": "state", "value": "stopped"}]},
    ])
    assert reg.size == 2


def test_lookup_case_insensitive():
    reg = make_registry([{"subject": "Nginx", "predicate": "State", "value": "running"}])
    assert reg.lookup("nginx", "state") is not None
    assert reg.lookup("NGINX", "STATE") is not None
    assert reg.lookup("unknown", "state") is None
    assert reg.lookup("nginx", "unknown") is None


def test_later_configs_override():
    reg = FactRegistry([
        {"id": "a", "facts": [{"subject": "nginx", "predicate": "state", "value": "running"}]},
        {"id": "b", "facts": [{"subject": "nginx", "predicate": "state", "value": "stopped"}]},
    ])
    assert reg.lookup("nginx", "state")["value"] == "stopped"
    assert reg.size == 1


def test_lookup_by_subject():
    reg = make_registry([
        {"subject": "nginx", "predicate": "state", "value": "running"},
        {"subject": "nginx", "predicate": "port", "value": "80"},
        {"subject": "redis", "predicate": "state", "value": "stopped"},
    ])
    assert len(reg.lookup_by_subject("NGINX")) == 2
    assert reg.lookup_by_subject("mystery") == []


# -- checkClaim verdicts -----------------------------------------------------

def test_system_state_verdicts():
    reg = make_registry([{"subject": "nginx", "predicate": "state", "value": "running"}])
    r = check_claim(make_claim(), reg)
    assert r["status"] == "verified" and r["fact"] is not None

    reg2 = make_registry([{"subject": "nginx", "predicate": "state", "value": "stopped"}])
    r2 = check_claim(make_claim(value="running"), reg2)
    assert r2["status"] == "contradicted" and r2["fact"]["value"] == "stopped"

    r3 = check_claim(make_claim(), make_registry([]))
    assert r3["status"] == "unverified" and r3["fact"] is None


def test_existence_claims():
    reg = make_registry([{"subject": "config.yaml", "predicate": "exists", "value": "true"}])
    claim = make_claim(type="existence", subject="config.yaml", predicate="exists", value="true")
    assert check_claim(claim, reg)["status"] == "verified"

    reg2 = make_registry([{"subject": "config.yaml", "predicate": "exists", "value": "false"}])
    assert check_claim(claim, reg2)["status"] == "contradicted"


def test_self_referential_claims():
    reg = make_registry([{"subject": "self", "predicate": "name", "value": "Forge"}])
    ok = make_claim(type="self_referential", subject="self", predicate="name", value="Forge")
    assert check_claim(ok, reg)["status"] == "verified"
    wrong = make_claim(type="self_referential", subject="self", predicate="name", value="Atlas")
    assert check_claim(wrong, reg)["status"] == "contradicted"


def test_value_normalization():
    # case-insensitive
    reg = make_registry([{"subject": "nginx", "predicate": "state", "value": "Running"}])
    assert check_claim(make_claim(value="running"), reg)["status"] == "verified"
    # whitespace trimmed
    reg2 = make_registry([{"subject": "nginx", "predicate": "state", "value": " running "}])
    assert check_claim(make_claim(value="running"), reg2)["status"] == "verified"
    # yes/no -> true/false
    reg3 = make_registry([{"subject": "config.yaml", "predicate": "exists", "value": "yes"}])
    claim = make_claim(type="existence", subject="config.yaml", predicate="exists", value="true")
    assert check_claim(claim, reg3)["status"] == "verified"


def test_fuzzy_numeric_and_predicate_fallback():
    reg = make_registry([{"subject": "queue", "predicate": "metric", "value": "150 items"}])
    claim = make_claim(type="operational_status", subject="queue",
                       predicate="metric", value="150 items")
    assert check_claim(claim, reg)["status"] == "verified"
    # fuzzy: "150" == "150 items"
    fuzzy = make_claim(type="operational_status", subject="queue",
                       predicate="metric", value="150")
    assert check_claim(fuzzy, reg)["status"] == "verified"
    # comma-grouped numbers
    reg2 = make_registry([{"subject": "db", "predicate": "count", "value": "255,908"}])
    c2 = make_claim(type="operational_status", subject="db", predicate="count",
                    value="255908 items")
    assert check_claim(c2, reg2)["status"] == "verified"


def test_check_claims_batch():
    reg = make_registry([
        {"subject": "nginx", "predicate": "state", "value": "running"},
        {"subject": "redis", "predicate": "state", "value": "stopped"},
    ])
    claims = [
        make_claim(subject="nginx", value="running"),
        make_claim(subject="redis", value="running"),
        make_claim(subject="mysql", value="running"),
    ]
    results = check_claims(claims, reg)
    assert [r["status"] for r in results] == ["verified", "contradicted", "unverified"]
    assert check_claims([], reg) == []


def test_lookup_performance():
    facts = [{"subject": f"service-{i}", "predicate": "state", "value": "running"}
             for i in range(1000)]
    reg = FactRegistry([{"id": "perf", "facts": facts}])
    claims = [make_claim(subject=f"service-{i}") for i in range(100)]
    start = time.perf_counter()
    check_claims(claims, reg)
    assert (time.perf_counter() - start) < 0.05  # reference bound is 2ms in JS


# -- file loading ------------------------------------------------------------

def test_file_loading(workspace):
    path = os.path.join(workspace, "facts.json")
    with open(path, "w") as fh:
        json.dump({"facts": [{"subject": "nginx", "predicate": "state", "value": "running"}]}, fh)
    reg = FactRegistry([{"id": "f", "filePath": path}])
    assert reg.size == 1
    assert reg.lookup("nginx", "state")["value"] == "running"


def test_file_loading_bad_inputs(workspace):
    missing = FactRegistry([{"id": "f", "filePath": os.path.join(workspace, "nope.json")}])
    assert missing.size == 0

    bad = os.path.join(workspace, "bad.json")
    with open(bad, "w") as fh:
        fh.write("{not json")
    assert FactRegistry([{"id": "f", "filePath": bad}]).size == 0

    noarr = os.path.join(workspace, "noarr.json")
    with open(noarr, "w") as fh:
        json.dump({"version": 1}, fh)
    assert FactRegistry([{"id": "f", "filePath": noarr}]).size == 0

    notobj = os.path.join(workspace, "notobj.json")
    with open(notobj, "w") as fh:
        json.dump([1, 2, 3], fh)
    assert FactRegistry([{"id": "f", "filePath": notobj}]).size == 0

    # neither facts nor filePath
    assert FactRegistry([{"id": "empty"}]).size == 0


def test_file_overrides_inline(workspace):
    path = os.path.join(workspace, "facts.json")
    with open(path, "w") as fh:
        json.dump({"facts": [{"subject": "nginx", "predicate": "state", "value": "stopped"}]}, fh)
    reg = FactRegistry([
        {"id": "inline", "facts": [{"subject": "nginx", "predicate": "state", "value": "running"},
                                   {"subject": "redis", "predicate": "state", "value": "up"}]},
        {"id": "file", "filePath": path},
    ])
    assert reg.lookup("nginx", "state")["value"] == "stopped"
    assert reg.size == 2


def test_get_all_facts_deduped():
    reg = FactRegistry([
        {"id": "a", "facts": [{"subject": "s", "predicate": "p", "value": "1"}]},
        {"id": "b", "facts": [{"subject": "s", "predicate": "p", "value": "2"}]},
    ])
    allf = reg.get_all_facts()
    assert len(allf) == 1 and allf[0]["value"] == "2"
    assert FactRegistry([]).get_all_facts() == []


# -- GPU fact-probe CPU mirror semantics ------------------------------------

def test_fact_probe_reference_semantics():
    from vainplex_openclaw_amd.ops import gpu as g
    from vainplex_openclaw_amd.ops import pattern_sets as ps

    facts = [("nginx-service", "state", "running"),
             ("backup.db", "exists", "replica")]
    msgs = [
        b"the nginx-service is running fine",     # verified
        b"the nginx-service is stopped now",      # contradicted
        b"no claims at all here",                 # no claim hit
        b"the unknown-svc is running today",      # claim, no fact -> unverified
    ]
    masks = [ps.get_family("claims").scan(m) for m in msgs]
    v, c = g.reference_fact_probe(msgs, masks, facts)
    assert list(v) == [1, 0, 0, 0]
    assert list(c) == [0, 1, 0, 0]


def test_fact_table_build_and_collisions():
    from vainplex_openclaw_amd.ops import gpu as g

    facts = [(f"svc-{i}", "state", "running") for i in range(40)]
    tk, tv, ph, pw = g.build_fact_table(facts)
    keys = tk.numpy().view("uint64")
    assert (keys != 0).sum() == 40           # all inserted
    assert 1 << pw >= 80                      # >= 2x occupancy headroom
    # predicate hash vector has entries only on claim bits
    pred = ph.numpy().view("uint64")
    assert pred[0] != 0 and pred[7] != 0
    assert pred[1] == 0 and pred[8] == 0  # subject-only strategies skipped


def test_fact_probe_tokenizer_matches_kernel_classes():
    from vainplex_openclaw_amd.ops.gpu import _tokenize_fact

    assert _tokenize_fact(b"The Nginx-Service IS running!") == [
        b"the", b"nginx-service", b"is", b"running"]
    assert _tokenize_fact(b"a b") == []          # 1-char tokens dropped
    assert _tokenize_fact(b"x1 .. y_2") == [b"x1", b"..", b"y_2"]
