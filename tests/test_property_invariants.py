"""Hypothesis property tests for core invariants that table tests can't
sweep: vault placeholder determinism, glob compilation vs fnmatch,
fact-merge algebra, journal ordering, banded-recall merge optimality.
"""

import fnmatch
import string

import torch
from hypothesis import given, settings
from hypothesis import strategies as st

# -- redaction vault ----------------------------------------------------------

secrets = st.text(alphabet=string.ascii_letters + string.digits + "-_", min_size=4, max_size=40)


@given(st.lists(st.tuples(secrets, st.sampled_from(["credential", "pii", "financial"])),
                min_size=1, max_size=30))
@settings(max_examples=60, deadline=None)
def test_vault_placeholder_deterministic_and_resolvable(pairs):
    from vainplex_openclaw_amd.governance.redaction.vault import RedactionVault

    v = RedactionVault()
    seen = {}
    for value, cat in pairs:
        ph = v.store(value, cat)
        assert ph.startswith("[REDACTED:")
        assert v.lookup(ph) == value                  # always resolvable
        # same value -> same placeholder, every time
        assert seen.setdefault(value, ph) == ph
    # distinct values never collide
    inv = {}
    for value, ph in seen.items():
        assert inv.setdefault(ph, value) == value


# -- glob compilation ---------------------------------------------------------

glob_pat = st.text(alphabet=string.ascii_lowercase + "*?-.", min_size=1, max_size=12)
candidate = st.text(alphabet=string.ascii_lowercase + "-.", min_size=0, max_size=16)


@given(glob_pat, candidate)
@settings(max_examples=200, deadline=None)
def test_glob_to_regex_matches_fnmatch(pattern, value):
    from vainplex_openclaw_amd.governance.conditions import glob_to_regex

    # fnmatch treats '.' specially only in [] groups, which we exclude
    want = fnmatch.fnmatchcase(value, pattern)
    got = bool(glob_to_regex(pattern).match(value))
    assert got == want, (pattern, value)


# -- fact merge algebra -------------------------------------------------------

fact = st.fixed_dictionaries({
    "subject": st.sampled_from(["api", "db", "Cache", "QUEUE"]),
    "predicate": st.sampled_from(["state", "count", "exists"]),
    "value": st.text(alphabet=string.ascii_lowercase + string.digits, min_size=1, max_size=8),
})


@given(st.lists(fact, max_size=20), st.lists(fact, max_size=20))
@settings(max_examples=100, deadline=None)
def test_merge_facts_later_wins_and_key_unique(existing, new):
    from vainplex_openclaw_amd.governance.trace_to_facts import _fact_key, merge_facts

    merged = merge_facts(existing, new)
    keys = [_fact_key(f) for f in merged]
    assert len(keys) == len(set(keys))                # dedupe by key
    # later-wins: for every key, the merged value equals the LAST
    # occurrence across existing + new
    last = {}
    for f in list(existing) + list(new):
        last[_fact_key(f)] = f["value"]
    for f in merged:
        assert f["value"] == last[_fact_key(f)]
    # idempotent: merging the result with itself changes nothing
    assert sorted((_fact_key(f), f["value"]) for f in merge_facts(merged, [])) == \
           sorted((_fact_key(f), f["value"]) for f in merged)


# -- journal ordering ---------------------------------------------------------

@given(st.lists(st.integers(min_value=0, max_value=3), min_size=1, max_size=40))
@settings(max_examples=40, deadline=None)
def test_journal_replay_seq_monotone_under_mixed_publishes(kinds):
    from vainplex_openclaw_amd.eventstore import EventJournal

    j = EventJournal(durable=False)
    expected = 0
    for i, kind in enumerate(kinds):
        if kind == 3:
            j.publish_block(f"s.{i}", b'{"id":"x"}\n' * 2, 2)
            expected += 2
        else:
            j.publish(f"s.{i}", {"ts": i, "id": f"e{i}"})
            expected += 1
    seqs = [s for s, _ in j.replay()]
    assert len(seqs) == expected == len(j)
    assert seqs == sorted(seqs) and len(set(seqs)) == len(seqs)


# -- banded recall merge ------------------------------------------------------

@given(st.integers(min_value=0, max_value=64), st.integers(min_value=1, max_value=8),
       st.integers(min_value=0, max_value=10_000))
@settings(max_examples=30, deadline=None)
def test_banded_equals_dense_optimum_with_exact_cold_path(nh, k, seed):
    import vainplex_openclaw_amd.ops.gpu as g

    torch.manual_seed(seed)
    nq, nx, d = 4, 96, 16
    Q = torch.nn.functional.normalize(torch.randn(nq, d), dim=1)
    X = torch.nn.functional.normalize(torch.randn(nx, d), dim=1)
    sal = torch.rand(nx).clamp(min=0.01)
    hot_idx = torch.topk(sal, min(nh, nx)).indices

    def exact_cold(Qc, Xc, kc, salience=None, **kw):
        dense = torch.matmul(Qc.float(), Xc.float().T) * salience
        top = torch.topk(dense, kc, dim=1)
        return top.values, top.indices.to(torch.int32)

    orig = g.topk_recall_threshold
    g.topk_recall_threshold = exact_cold
    try:
        s, i = g.topk_recall_threshold_banded(Q, X, k, salience=sal, hot_idx=hot_idx)
    finally:
        g.topk_recall_threshold = orig
    opt = torch.topk(torch.matmul(Q.float(), X.float().T) * sal, k, dim=1).values
    # with an exact cold path the banded merge must be exactly optimal,
    # for ANY hot band (including empty) — and never duplicate an id
    assert torch.allclose(s, opt, atol=1e-4)
    for row in i:
        assert len(set(row.tolist())) == k
