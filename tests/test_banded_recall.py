"""CPU-side logic tests for salience-banded recall
(ops.gpu.topk_recall_threshold_banded): hot-band exact scoring, cold-band
delegation, duplicate masking, empty-band delegation. The cold-band scan
is monkeypatched to a pure-torch reference so no HIP extension is needed;
GPU quality under Zipf salience is covered in test_gpu_kernels.py.
"""

import pytest
import torch

import vainplex_openclaw_amd.ops.gpu as g


def _dense_weighted_topk(Q, X, k, salience):
    dense = torch.matmul(Q.float(), X.float().T) * salience
    top = torch.topk(dense, k, dim=1)
    return top.values, top.indices.to(torch.int32)


@pytest.fixture
def patched_cold(monkeypatch):
    """Cold path = exact dense weighted top-k (stands in for the fp4
    threshold scan, which needs the HIP extension)."""
    def fake_threshold(Q, X, k, salience=None, **kw):
        sal = salience if salience is not None else torch.ones(X.shape[0])
        return _dense_weighted_topk(Q, X, k, sal)

    monkeypatch.setattr(g, "topk_recall_threshold", fake_threshold)
    return fake_threshold


def _setup(nq=8, nx=512, d=32, seed=3):
    torch.manual_seed(seed)
    Q = torch.nn.functional.normalize(torch.randn(nq, d), dim=1)
    X = torch.nn.functional.normalize(torch.randn(nx, d), dim=1)
    sal = torch.rand(nx).clamp(min=0.01)
    return Q, X, sal


def test_banded_matches_dense_weighted_optimum(patched_cold):
    Q, X, sal = _setup()
    k = 8
    hot_idx = torch.topk(sal, 32).indices
    s, i = g.topk_recall_threshold_banded(Q, X, k, salience=sal, hot_idx=hot_idx)
    want_s, _ = _dense_weighted_topk(Q, X, k, sal)
    assert torch.allclose(s, want_s, atol=1e-4)


def test_banded_no_duplicate_ids(patched_cold):
    Q, X, sal = _setup(seed=5)
    # hot band deliberately contains the global cosine winners so the
    # cold scan would re-report them without the mask
    hot_idx = torch.topk(sal, 64).indices
    s, i = g.topk_recall_threshold_banded(Q, X, 8, salience=sal, hot_idx=hot_idx)
    for row in i:
        assert len(set(row.tolist())) == len(row)


def test_banded_empty_hot_band_delegates(patched_cold):
    Q, X, sal = _setup()
    empty = torch.empty(0, dtype=torch.long)
    s, i = g.topk_recall_threshold_banded(Q, X, 4, salience=sal, hot_idx=empty)
    want_s, want_i = _dense_weighted_topk(Q, X, 4, sal)
    assert torch.allclose(s, want_s, atol=1e-4)


def test_banded_recovers_skewed_optimum_beyond_cosine_overfetch(monkeypatch):
    """The scenario the banded mode exists for: a high-salience row whose
    cosine rank is beyond the cold path's overfetch. A bounded-overfetch
    cold path (cosine-top-16 then weight) misses it; the hot band scores
    it exactly."""
    torch.manual_seed(7)
    nq, nx, d, k = 4, 256, 32, 4
    Q = torch.nn.functional.normalize(torch.randn(nq, d), dim=1)
    X = torch.nn.functional.normalize(torch.randn(nx, d), dim=1)
    sal = torch.full((nx,), 0.01)
    # row 0: low-ish cosine to every query but enormous salience
    sal[0] = 1.0

    def bounded_cold(Qc, Xc, kc, salience=None, **kw):
        dense = torch.matmul(Qc.float(), Xc.float().T)
        cv, ci = torch.topk(dense, 16, dim=1)           # bounded overfetch
        w = cv * salience[ci]
        top = torch.topk(w, kc, dim=1)
        return top.values, torch.gather(ci, 1, top.indices).to(torch.int32)

    monkeypatch.setattr(g, "topk_recall_threshold", bounded_cold)
    dense = torch.matmul(Q.float(), X.float().T)
    # ensure row 0 is outside every query's cosine-top-16 (else reshuffle)
    assert all(0 not in torch.topk(dense[q], 16).indices for q in range(nq))

    cold_s, cold_i = bounded_cold(Q, X, k, salience=sal)
    assert not any(0 in row for row in cold_i.tolist())  # cold path misses it

    hot_idx = torch.topk(sal, 8).indices
    s, i = g.topk_recall_threshold_banded(Q, X, k, salience=sal, hot_idx=hot_idx)
    opt_s, opt_i = _dense_weighted_topk(Q, X, k, sal)
    if any(0 in row for row in opt_i.tolist()):          # optimum includes row 0
        assert any(0 in row for row in i.tolist())       # banded finds it
    assert s.sum().item() >= cold_s.sum().item() - 1e-5
    assert torch.allclose(s, opt_s, atol=1e-4)


def test_banded_hot_cache_params(patched_cold):
    Q, X, sal = _setup()
    hot_idx = torch.topk(sal, 32).indices
    hot_X = X[hot_idx]
    is_hot = torch.zeros(X.shape[0], dtype=torch.bool)
    is_hot[hot_idx] = True
    s1, i1 = g.topk_recall_threshold_banded(Q, X, 8, salience=sal, hot_idx=hot_idx)
    s2, i2 = g.topk_recall_threshold_banded(Q, X, 8, salience=sal, hot_idx=hot_idx,
                                            hot_X=hot_X, is_hot=is_hot)
    assert torch.equal(i1, i2) and torch.allclose(s1, s2)
