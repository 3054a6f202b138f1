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
                    choices=["threshold", "two_stage", "direct"])
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
    # async audit writer (binary records + Merkle manifest JSONL) and a
    # per-batch envelope into the embedded journal
    writer = None
    journal = None
    if not args.no_audit_sink:
        import tempfile

        from vainplex_openclaw_amd.eventstore import EventJournal
        from vainplex_openclaw_amd.pipeline.engine import AsyncAuditWriter

        audit_dir = tempfile.mkdtemp(prefix=f"bench-audit-r{rank}-")
        writer = AsyncAuditWriter(audit_dir)
        journal = EventJournal(durable=False)

        def sink(records, root):
            writer(records, root)
            journal.publish(
                f"openclaw.events.rank{rank}.batch",
                {"ts": int(time.time() * 1000), "type": "batch.audited",
                 "agent": f"rank{rank}", "session": "bench",
                 "payload": {"count": int(records.shape[0])}},
            )

        pipe.audit_sink = sink

    # pre-generated, pre-staged synthetic batch pool (rotated; every step
    # still runs the full pipeline on real message bytes)
    pool = []
    for i in range(args.pool):
        batch = synthetic_batch(args.batch, seed=1000 * rank + i, n_agents=cfg.n_agents)
        pool.append((batch, pipe.stage(batch)))

    def barrier():
        if distributed:
            torch.distributed.barrier()

    # warmup
    for i in range(args.warmup):
        b, s = pool[i % len(pool)]
        pipe.step(b, staged=s)
    barrier()
    torch.cuda.synchronize()

    # timed region: exactly --steps full pipeline steps
    t_start = time.perf_counter()
    for i in range(args.steps):
        b, s = pool[i % len(pool)]
        pipe.step(b, staged=s)
    barrier()
    torch.cuda.synchronize()
    elapsed = time.perf_counter() - t_start

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
                "model": "firewall-pipeline (dfa-scan + encoder + classifier head + cosine-kNN recall + merkle audit)",
                "global_batch": world * args.batch,
                "seq_len": 500,
                "parallelism": f"dp{world}",
                "index_rows_total": shard * world,
                "index_dim": args.dim,
                "topk": args.topk,
            },
        }))
    if args.profile:
        # hipEvent per-stage timing, outside the timed region (the
        # profiler synchronizes per step)
        pipe.profiler.enabled = torch.cuda.is_available()
        for i in range(3):
            b, s2 = pool[i % len(pool)]
            pipe.step(b, staged=s2)
            pipe.profiler.commit()
        if rank == 0:
            print("stage ms/step: " + json.dumps(
                {k: round(v, 3) for k, v in pipe.profiler.summary().items()}
            ), file=sys.stderr)

    if writer is not None:
        writer.close()
        if rank == 0:
            replayed = sum(1 for _ in journal.replay())
            print(json.dumps({"audit_batches_written": writer.batches_written,
                              "journal_events_replayed": replayed}), file=sys.stderr)

    if distributed:
        torch.distributed.destroy_process_group()
    return 0


if __name__ == "__main__":
    sys.exit(main())
