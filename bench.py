#!/usr/bin/env python3
"""Flagship benchmark: messages/sec through the full
Governance -> Membrane -> Cortex/KE firewall pipeline (BASELINE.json
headline metric).

Single GPU:   python bench.py --steps 10 --warmup 3
Multi GPU:    python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
                  --master-addr 127.0.0.1 bench.py --gpus N --steps K --warmup W

Each rank owns one GPU (RCCL over xGMI), a shard of the 50M x 1024-d
Membrane index (shard = total/world: per-GPU recall work is constant as
ranks grow -> weak scaling), and processes its own batch of 4096 synthetic
messages per step through every stage: DFA pattern scans, encoder,
classifier head, full-index cosine top-k recall (queries all-gathered),
verdict + trust update, audit Merkle root (roots combined across ranks).

Rank 0 prints ONE JSON line with the whole-job aggregate messages/sec.
"""

import argparse
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from vainplex_openclaw_amd.pipeline.engine import FirewallPipeline, PipelineConfig
from vainplex_openclaw_amd.pipeline.synth import synthetic_batch

BASELINE_MSG_S = 9000.0  # reference headline: NATS concurrent publish ~9k msg/s
                         # (openclaw-nats-eventstore/README.md:261, other hardware)

TOTAL_INDEX = 50_000_000  # BASELINE config: cosine-kNN over 50M x 1024-d


def main() -> int:
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=10)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--batch", type=int, default=4096)
    ap.add_argument("--dim", type=int, default=1024)
    ap.add_argument("--index", type=int, default=TOTAL_INDEX, help="TOTAL index rows across ranks")
    ap.add_argument("--topk", type=int, default=16)
    ap.add_argument("--pool", type=int, default=4, help="pre-generated batch pool size")
    ap.add_argument("--recall-mode", default="threshold",
                    choices=["threshold", "threshold_banded", "two_stage", "direct"])
    ap.add_argument("--no-fp8", action="store_true")
    ap.add_argument("--no-mx", action="store_true")
    ap.add_argument("--no-fp4", action="store_true")
    ap.add_argument("--no-audit-sink", action="store_true",
                    help="disable the async audit writer + event journal (on by default: "
                    "the timed path includes audit record D2H + JSONL manifest + journal envelopes)")
    ap.add_argument("--profile", action="store_true",
                    help="after the timed region, run 3 instrumented steps and print per-stage ms (stderr)")
    args = ap.parse_args()

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    distributed = world > 1
    if distributed:
        torch.distributed.init_process_group("nccl", rank=rank, world_size=world)
    device = f"cuda:{local_rank}"
    torch.cuda.set_device(device)

    shard = max(1, args.index // world)
    # round shard to a multiple of the 128-row recall tile
    shard = (shard // 128) * 128
    cfg = PipelineConfig(
        batch=args.batch, dim=args.dim, index_size=shard, topk=args.topk,
        recall_mode=args.recall_mode, recall_fp8=not args.no_fp8,
        recall_mx=not args.no_mx, recall_fp4=not args.no_fp4,
    )
    t0 = time.time()
    pipe = FirewallPipeline(cfg, device=device, world_size=world, rank=rank)
    torch.cuda.synchronize()
    setup_s = time.time() - t0

    # BASELINE config #5: the full pipeline INCLUDING the event-store leg —
    # async audit writer (binary records + Merkle manifest JSONL) and ONE
    # ClawEvent envelope PER MESSAGE (C++ batched builder) published into
    # the embedded journal as EventBlocks. The writer/journal are drained
    # INSIDE the timed region so the per-message envelope cost is measured.
    writer = None
    journal = None
    if not args.no_audit_sink:
        import tempfile

        from vainplex_openclaw_amd.eventstore import EventJournal
        from vainplex_openclaw_amd.pipeline.engine import AsyncAuditWriter

        audit_dir = tempfile.mkdtemp(prefix=f"bench-audit-r{rank}-")
        journal = EventJournal(durable=False)
        writer = AsyncAuditWriter(audit_dir, journal=journal, session=f"rank{rank}")
        pipe.audit_sink = writer

    # pre-GENERATED synthetic batch pool (message bytes only); staging
    # (CPU pack + pinned H2D on the copy stream) runs INSIDE the timed
    # loop, double-buffered and overlapped with compute
    from vainplex_openclaw_amd.pipeline.engine import StagingRing

    pool = [synthetic_batch(args.batch, seed=1000 * rank + i, n_agents=cfg.n_agents)
            for i in range(args.pool)]
    ring = StagingRing(pipe, max_bytes=8 << 20, max_msgs=max(args.batch, 8192) + 1)

    def barrier():
        if distributed:
            torch.distributed.barrier()

    # warmup
    staged_next = ring.stage(pool[0])
    for i in range(args.warmup):
        staged = staged_next
        staged_next = ring.stage(pool[(i + 1) % len(pool)])
        pipe.step(None, staged=staged)
    if writer is not None:
        writer.drain()
    pre_events = writer.events_published if writer is not None else 0
    barrier()
    torch.cuda.synchronize()

    # timed region: exactly --steps full pipeline steps, tokenize -> verdict
    # -> audit -> per-message envelopes end to end
    t_start = time.perf_counter()
    for i in range(args.steps):
        staged = staged_next
        staged_next = ring.stage(pool[(i + 1) % len(pool)])
        pipe.step(None, staged=staged)
    torch.cuda.synchronize()
    if writer is not None:
        writer.drain()  # envelope + journal work inside the timed region
    barrier()
    torch.cuda.synchronize()
    elapsed = time.perf_counter() - t_start
    timed_events = (writer.events_published - pre_events) if writer is not None else 0

    # MAX elapsed over ranks
    if distributed:
        t = torch.tensor([elapsed], device=device)
        torch.distributed.all_reduce(t, op=torch.distributed.ReduceOp.MAX)
        elapsed = t.item()

    total_msgs = world * args.batch * args.steps
    msg_per_s = total_msgs / elapsed
    ms_per_step = elapsed / args.steps * 1000

    if rank == 0:
        print(json.dumps({
            "metric": "messages/sec through Governance->Membrane->Cortex/KE pipeline",
            "value": round(msg_per_s, 2),
            "unit": "msg/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": round(msg_per_s / BASELINE_MSG_S, 3),
            "dtype": "bf16",
            "data": "synthetic",
            "setup_s": round(setup_s, 1),
            "config": {
                "model": "firewall-pipeline (fused dfa-scan x5 families + fact probe + cortex fold + encoder + classifier head + cosine-kNN recall + merkle audit + per-message envelopes)",
                "global_batch": world * args.batch,
                "seq_len": 500,
                "parallelism": f"dp{world}",
                "index_rows_total": shard * world,
                "index_dim": args.dim,
                "topk": args.topk,
                "families": list(cfg.families),
                "recall_mode": cfg.recall_mode,
                # scan operand precision; final recall scores are always
                # EXACT fp32 rescored cosines (ops/gpu.py threshold path)
                "recall_scan_operands": (
                    "mxfp4" if (cfg.recall_fp4 and cfg.recall_mode
                                in ("threshold", "threshold_banded"))
                    else ("fp8-mx" if (cfg.recall_fp8 and cfg.recall_mx)
                          else ("fp8" if cfg.recall_fp8 else "bf16"))
                ),
                "recall_rescore": "exact-fp32 x salience",
                "salience_weighting": cfg.salience_weighting,
                "per_message_envelopes": journal is not None,
            },
        }))
    if args.profile:
        # hipEvent per-stage timing, outside the timed region (the
        # profiler synchronizes per step)
        pipe.profiler.enabled = torch.cuda.is_available()
        for i in range(3):
            s2 = ring.stage(pool[i % len(pool)])
            pipe.step(None, staged=s2)
            pipe.profiler.commit()
        if rank == 0:
            print("stage ms/step: " + json.dumps(
                {k: round(v, 3) for k, v in pipe.profiler.summary().items()}
            ), file=sys.stderr)

    if writer is not None:
        writer.close()
        if rank == 0:
            replayed = sum(1 for _ in journal.replay())
            expected = (args.steps + args.warmup) * args.batch
            print(json.dumps({
                "audit_batches_written": writer.batches_written,
                "events_published": writer.events_published,
                "journal_events_replayed": replayed,
                "journal_expected": expected,
                "journal_events_timed": timed_events,
                "journal_msg_s": round(timed_events / elapsed, 1) if elapsed else None,
            }), file=sys.stderr)

    if distributed:
        torch.distributed.destroy_process_group()
    return 0


if __name__ == "__main__":
    sys.exit(main())
