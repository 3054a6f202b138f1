"""Distributed collectives for the firewall pipeline (DP over RCCL/xGMI).

One process per GPU, torch.distributed with backend "nccl" (= RCCL on
ROCm) over xGMI; the same code runs under "gloo" on CPU for the
multi-process tests (no-GPU CI covers the distributed path
with gloo world_size=2).

Design for xGMI (7 p2p links x ~153 GB/s per GPU):
- recall queries are all-gathered (bf16, B x D per rank — a large
  contiguous tensor, ring-friendly),
- per-rank top-k candidates are all-gathered and merged host-side-free
  (each rank keeps only its own query rows),
- per-batch Merkle roots (32 B) are combined with a single all-gather —
  a latency-bound small payload, never a ring reduction.
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch
import torch.distributed as dist


def init_process_group_from_env(backend: Optional[str] = None) -> Tuple[int, int, int]:
    """Read RANK/LOCAL_RANK/WORLD_SIZE from the env (torchrun contract)
    and init the group. Returns (rank, local_rank, world)."""
    import os

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    if world > 1 and not dist.is_initialized():
        if backend is None:
            backend = "nccl" if torch.cuda.is_available() else "gloo"
        dist.init_process_group(backend, rank=rank, world_size=world)
    return rank, local_rank, world


def allgather_queries(feats: torch.Tensor, world: int) -> torch.Tensor:
    """[B, D] per rank -> [world*B, D]: every rank searches the FULL
    index for every query (the index is sharded)."""
    if world <= 1:
        return feats
    out = torch.empty(world * feats.shape[0], feats.shape[1],
                      dtype=feats.dtype, device=feats.device)
    dist.all_gather_into_tensor(out, feats.contiguous())
    return out


def globalize_ids(ids: torch.Tensor, rank: int, shard_rows: int) -> torch.Tensor:
    """Shard-local row ids -> global ids (invalid -1 slots preserved)."""
    out = ids + rank * shard_rows
    return torch.where(ids < 0, ids, out)


def merge_topk_candidates(
    scores: torch.Tensor, ids: torch.Tensor, rank: int, batch: int, world: int, k: int
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Each rank scored ALL (world*batch) queries against ITS shard.
    All-gather candidates, keep my query rows, merge to final top-k.

    scores/ids: [world*batch, k] local candidates (ids already global).
    Returns [batch, k] for THIS rank's queries.
    """
    if world <= 1:
        return scores[:batch], ids[:batch]
    nq = scores.shape[0]
    cand_s = torch.empty(world, nq, k, dtype=scores.dtype, device=scores.device)
    cand_i = torch.empty(world, nq, k, dtype=ids.dtype, device=ids.device)
    dist.all_gather_into_tensor(cand_s, scores.unsqueeze(0).contiguous())
    dist.all_gather_into_tensor(cand_i, ids.unsqueeze(0).contiguous())
    my0 = rank * batch
    mine_s = cand_s[:, my0 : my0 + batch].permute(1, 0, 2).reshape(batch, world * k)
    mine_i = cand_i[:, my0 : my0 + batch].permute(1, 0, 2).reshape(batch, world * k)
    top = torch.topk(mine_s, k, dim=1)
    return top.values, torch.gather(mine_i, 1, top.indices)


def local_shard_ids(flat_ids: torch.Tensor, rank: int, shard_rows: int) -> torch.Tensor:
    """Global recall ids -> this rank's shard-local rows (for salience
    reinforcement); drops other ranks' ids and unfilled (-1) slots."""
    base = rank * shard_rows
    mask = (flat_ids >= base) & (flat_ids < base + shard_rows)
    return flat_ids[mask] - base


def allgather_roots(root: torch.Tensor, world: int) -> torch.Tensor:
    """Per-rank 32-B Merkle roots -> [world, 32] (tiny payload: one-shot
    all-gather, the global root is the Merkle parent of the gathered
    leaves)."""
    if world <= 1:
        return root.unsqueeze(0)
    out = torch.empty(world, root.numel(), dtype=root.dtype, device=root.device)
    dist.all_gather_into_tensor(out, root.unsqueeze(0).contiguous())
    return out


def max_elapsed(elapsed: float, device) -> float:
    """MAX over ranks of a wall-clock measurement (bench contract)."""
    if not (dist.is_available() and dist.is_initialized()):
        return elapsed
    t = torch.tensor([elapsed], dtype=torch.float64,
                     device=device if torch.cuda.is_available() else "cpu")
    dist.all_reduce(t, op=dist.ReduceOp.MAX)
    return float(t.item())
