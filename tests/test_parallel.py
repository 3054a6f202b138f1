"""Multi-process CPU tests of the distributed path (gloo, world_size=2).

The driver has no GPU here; these tests prove the DP collective logic
(query all-gather, candidate merge, shard-id handling, root combination,
max-elapsed) is correct by construction before the same code runs over
RCCL/xGMI on the 8-GPU node (SURVEY.md §4 "env-gated GPU tests + gloo
multi-process CPU tests").
"""

import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from vainplex_openclaw_amd.parallel.collectives import (
    allgather_queries,
    allgather_roots,
    globalize_ids,
    local_shard_ids,
    max_elapsed,
    merge_topk_candidates,
)

WORLD = 2


def _run_worker(rank, fn, port):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=WORLD)
    try:
        fn(rank)
    finally:
        dist.destroy_process_group()


def _spawn(fn, port):
    mp.spawn(_run_worker, args=(fn, port), nprocs=WORLD, join=True)


# -- single-process behavior (world=1 short-circuits) -------------------------

def test_world1_passthrough():
    feats = torch.randn(4, 8)
    assert allgather_queries(feats, 1) is feats
    s, i = merge_topk_candidates(torch.randn(4, 3), torch.zeros(4, 3, dtype=torch.int32), 0, 4, 1, 3)
    assert s.shape == (4, 3)
    assert allgather_roots(torch.zeros(32, dtype=torch.uint8), 1).shape == (1, 32)


def test_globalize_and_localize_ids():
    ids = torch.tensor([[0, 5, -1], [2, -1, 9]], dtype=torch.int32)
    g = globalize_ids(ids, rank=1, shard_rows=100)
    assert g.tolist() == [[100, 105, -1], [102, -1, 109]]
    loc = local_shard_ids(g.reshape(-1), rank=1, shard_rows=100)
    assert sorted(loc.tolist()) == [0, 2, 5, 9]
    assert local_shard_ids(g.reshape(-1), rank=0, shard_rows=100).numel() == 0


def test_max_elapsed_no_group():
    assert max_elapsed(1.5, "cpu") == 1.5


# -- gloo world_size=2 --------------------------------------------------------

def _check_allgather_queries(rank):
    feats = torch.full((3, 4), float(rank))
    out = allgather_queries(feats, WORLD)
    assert out.shape == (6, 4)
    assert out[:3].eq(0.0).all() and out[3:].eq(1.0).all()


def test_gloo_allgather_queries():
    _spawn(_check_allgather_queries, 29611)


def _check_merge_topk(rank):
    # B=2 queries per rank, k=2, each rank owns shard of 10 rows.
    # rank r scores all 4 queries against its shard: score = global_id/100
    # so the true global top-2 for every query = ids 19, 18 (rank 1 shard).
    B, k, shard = 2, 2, 10
    nq = WORLD * B
    local_ids = torch.tensor([[9, 8], [9, 8], [9, 8], [9, 8]], dtype=torch.int32)
    gids = globalize_ids(local_ids, rank, shard)
    scores = gids.float() / 100.0
    s, i = merge_topk_candidates(scores, gids, rank, B, WORLD, k)
    assert s.shape == (B, k) and i.shape == (B, k)
    assert i.tolist() == [[19, 18], [19, 18]]
    assert torch.allclose(s, torch.tensor([[0.19, 0.18], [0.19, 0.18]]))


def test_gloo_merge_topk_candidates():
    _spawn(_check_merge_topk, 29613)


def _check_roots_and_elapsed(rank):
    root = torch.full((32,), rank, dtype=torch.uint8)
    roots = allgather_roots(root, WORLD)
    assert roots.shape == (WORLD, 32)
    assert roots[0].eq(0).all() and roots[1].eq(1).all()
    e = max_elapsed(1.0 + rank, "cpu")
    assert e == 2.0  # MAX over ranks


def test_gloo_roots_and_max_elapsed():
    _spawn(_check_roots_and_elapsed, 29615)


def _check_dp_recall_parity(rank):
    """End-to-end DP recall parity vs a single-process reference:
    sharded index + all-gather + merge == full-index top-k."""
    torch.manual_seed(7)
    B, D, k, shard = 4, 16, 3, 32
    full_x = torch.nn.functional.normalize(torch.randn(WORLD * shard, D), dim=1)
    queries = torch.nn.functional.normalize(torch.randn(WORLD * B, D), dim=1)
    my_q = queries[rank * B : (rank + 1) * B]
    my_shard = full_x[rank * shard : (rank + 1) * shard]

    q_all = allgather_queries(my_q, WORLD)
    assert torch.allclose(q_all, queries)
    scores = q_all @ my_shard.T
    top = torch.topk(scores, k, dim=1)
    gids = globalize_ids(top.indices.to(torch.int32), rank, shard)
    s, i = merge_topk_candidates(top.values, gids, rank, B, WORLD, k)

    ref = torch.topk(my_q @ full_x.T, k, dim=1)
    assert torch.allclose(s, ref.values, atol=1e-5)
    assert i.tolist() == ref.indices.to(torch.int32).tolist()


def test_gloo_dp_recall_parity():
    _spawn(_check_dp_recall_parity, 29617)
