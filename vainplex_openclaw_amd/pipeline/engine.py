"""The batched GPU firewall pipeline: Governance -> Membrane -> Cortex/KE.

This is the MI355X-native high-throughput mode of the suite (the Python
plugin engines in governance/, cortex/, knowledge/ are the per-message
interactive mode with identical semantics). Per step, for a batch of
messages:

  1. pack + H2D copy of message bytes
  2. DFA scans (csrc/pattern_scan.hip): redaction, injection, claims,
     entity families -> u64 hit masks per message
  3. encoder (csrc/encoder.hip): 4-gram hash features [B, D] bf16
  4. classifier head (csrc/gemm_nt.hip, fused sigmoid): injection /
     threat logits
  5. Membrane recall (csrc/topk_recall.hip): cosine top-k against the
     HBM-resident embedding index shard
  6. firewall verdict + trust updates (csrc/firewall.hip) implementing
     the reference's risk weights / verdict precedence / trust formula
  7. audit: 64-B records packed on GPU, SHA-256 leaves + Merkle root
     (csrc/sha256_merkle.hip)

Multi-GPU (one process per GPU, torch.distributed over RCCL/xGMI):
the index is sharded across ranks; recall queries are all-gathered so
every message searches the FULL index, per-rank top-k candidates are
all-gathered and merged back, and the per-batch Merkle roots are combined
into a global root. Per-GPU work is constant as ranks grow (weak scaling):
each rank scores (world * batch) queries against (total / world) index
rows.
"""

from __future__ import annotations

import threading
import time
from dataclasses import dataclass
from typing import Any, Dict, List, Optional

import numpy as np
import torch

from ..ops import gpu as g
from ..ops import pattern_sets
from ..parallel import collectives as coll
from .synth import SynthBatch


@dataclass
class PipelineConfig:
    batch: int = 4096
    dim: int = 1024
    vocab: int = 65536
    n_classes: int = 8
    n_agents: int = 64
    index_size: int = 6_250_000  # per-GPU shard (8 GPUs x 6.25M = 50M)
    topk: int = 16
    # recall modes: "direct" = bf16 streaming top-k kernel;
    # "two_stage" = fp8 scan (k2=32) + exact rescore;
    # "threshold" = Gaussian-tail threshold scan + select (+ fp8 rescore
    # when recall_fp8) — no in-kernel top-k maintenance
    recall_mode: str = "threshold"
    recall_fp8: bool = True
    # MX-scaled x128 scan for the fp8 threshold path (same bytes, higher
    # MFMA issue rate — csrc topk_scan_mx_kernel); needs dim % 128 == 0
    recall_mx: bool = True
    # MXFP4 X operand for the threshold scan (half the fp8 index bytes;
    # csrc topk_scan_mx4_kernel + the probed fragment permutation)
    recall_fp4: bool = True
    inj_threshold: float = 0.9
    seed: int = 1234
    families: tuple = ("redaction", "injection", "claims", "entity")


class StageProfiler:
    """hipEvent (torch.cuda.Event) timing per pipeline stage (SURVEY.md §5:
    keep the reference's µs self-timing; add device-side stage timing).
    Enabled via FirewallPipeline(..., profile=True); `summary()` returns
    average ms per stage over recorded steps."""

    def __init__(self, enabled: bool = False):
        self.enabled = enabled and torch.cuda.is_available()
        self._events: List = []
        self.totals: Dict[str, float] = {}
        self.steps = 0

    def mark(self, name: str) -> None:
        if not self.enabled:
            return
        ev = torch.cuda.Event(enable_timing=True)
        ev.record()
        self._events.append((name, ev))

    def commit(self) -> None:
        """Call after a step (outside the hot loop) to fold event deltas."""
        if not self.enabled or len(self._events) < 2:
            self._events = []
            return
        torch.cuda.synchronize()
        for (n0, e0), (_n1, e1) in zip(self._events, self._events[1:]):
            self.totals[n0] = self.totals.get(n0, 0.0) + e0.elapsed_time(e1)
        self._events = []
        self.steps += 1

    def summary(self) -> Dict[str, float]:
        if not self.steps:
            return {}
        return {k: v / self.steps for k, v in self.totals.items()}


class FirewallPipeline:
    def __init__(self, cfg: PipelineConfig, device: str = "cuda:0", world_size: int = 1, rank: int = 0, profile: bool = False):
        self.cfg = cfg
        self.device = torch.device(device)
        self.world_size = world_size
        self.rank = rank
        self.batch_seq = 0
        self.profiler = StageProfiler(profile)
        torch.manual_seed(cfg.seed + rank)

        with torch.cuda.device(self.device):
            # model state (random init; no network for checkpoints)
            self.embed = (torch.randn(cfg.vocab, cfg.dim, dtype=torch.float32, device=self.device) * 0.05).bfloat16()
            self.head = (torch.randn(cfg.n_classes, cfg.dim, dtype=torch.float32, device=self.device) * 0.05).bfloat16()
            self.head_bias = torch.zeros(cfg.n_classes, device=self.device)

            # Membrane index shard: L2-normalized rows. Preallocate the bf16
            # tensor and fill in chunks so peak extra memory is one fp32
            # chunk (4 GB), not a second copy of the whole index.
            chunk = 1_000_000
            gen = torch.Generator(device="cuda")
            gen.manual_seed(cfg.seed * 7919 + rank)
            self.index = torch.empty(cfg.index_size, cfg.dim, dtype=torch.bfloat16, device=self.device)
            for i in range(0, cfg.index_size, chunk):
                n = min(chunk, cfg.index_size - i)
                x = torch.randn(n, cfg.dim, generator=gen, dtype=torch.float32, device=self.device)
                x = torch.nn.functional.normalize(x, dim=1)
                self.index[i : i + n] = x.bfloat16()
                del x

            # low-precision scan copies of the index. fp4 (MXFP4, half
            # the fp8 bytes) is the default threshold-scan operand; the
            # e4m3 copy serves the two_stage mode and the fp8/MX scans.
            self.index8 = None
            self.index4 = None
            if cfg.recall_fp4 and cfg.recall_mode == "threshold" and cfg.dim % 128 == 0:
                x4 = torch.empty(cfg.index_size, cfg.dim // 2, dtype=torch.uint8,
                                 device=self.device)
                xs = torch.empty(cfg.index_size, cfg.dim // 32, dtype=torch.uint8,
                                 device=self.device)
                for i in range(0, cfg.index_size, chunk):
                    n = min(chunk, cfg.index_size - i)
                    c4, cs = g.to_fp4_mx(self.index[i : i + n])
                    x4[i : i + n] = c4
                    xs[i : i + n] = cs
                self.index4 = (x4, xs)
            elif cfg.recall_fp8 and cfg.recall_mode in ("two_stage", "threshold"):
                self.index8 = torch.empty(
                    cfg.index_size, cfg.dim, dtype=torch.uint8, device=self.device
                )
                for i in range(0, cfg.index_size, chunk):
                    n = min(chunk, cfg.index_size - i)
                    self.index8[i : i + n] = g.to_fp8_bytes(self.index[i : i + n])

            # salience state: recall strength + decay (Membrane semantics)
            self.salience = torch.ones(cfg.index_size, device=self.device)

            # trust state vectors (mirrors governance TrustManager fields)
            A = cfg.n_agents
            self.trust_state = {
                "success": torch.zeros(A, device=self.device),
                "violation": torch.zeros(A, device=self.device),
                "age_days": torch.zeros(A, device=self.device),
                "clean_streak": torch.zeros(A, device=self.device),
                "manual_adj": torch.full((A,), 40.0, device=self.device),
                "score": torch.full((A,), 40.0, device=self.device),
            }
            # warm the DFA tables onto the device
            for fam in cfg.families:
                g.device_family(fam, self.device)

        self.audit_sink: Optional[Any] = None  # callable(records_cpu, root_hex)

    # -- input staging -----------------------------------------------------
    def stage(self, batch: SynthBatch) -> Dict[str, torch.Tensor]:
        b, o = g.pack_messages(batch.messages, device=self.device)
        return {
            "bytes": b,
            "offsets": o,
            "agent_idx": torch.from_numpy(batch.agent_idx).to(self.device),
            "tool_risk": torch.from_numpy(batch.tool_risk).to(self.device),
        }

    # -- the step ----------------------------------------------------------
    def step(self, batch, staged: Optional[Dict[str, torch.Tensor]] = None) -> Dict[str, Any]:
        cfg = self.cfg
        s = staged if staged is not None else self.stage(batch)
        bytes_t, offsets = s["bytes"], s["offsets"]
        agent_idx, tool_risk = s["agent_idx"], s["tool_risk"]
        B = offsets.numel() - 1

        prof = self.profiler
        prof.mark("dfa_scan")
        # 2. pattern scans
        hits = {fam: g.dfa_scan(bytes_t, offsets, fam) for fam in cfg.families}

        prof.mark("encoder")
        # 3. encoder
        feats = g.encode_messages(bytes_t, offsets, self.embed, normalize=True)

        prof.mark("classifier")
        # 4. classifier head (fused sigmoid)
        logits = g.gemm_nt(feats, self.head, bias=self.head_bias, act=1)

        prof.mark("recall")
        # 5. Membrane recall (full index across ranks; parallel/collectives)
        def local_recall(queries):
            if cfg.recall_mode == "threshold":
                return g.topk_recall_threshold(
                    queries, self.index, cfg.topk, X8=self.index8,
                    mx=cfg.recall_mx, X4=self.index4
                )
            if cfg.recall_mode == "two_stage" and self.index8 is not None:
                return g.topk_recall_two_stage(queries, self.index, self.index8, cfg.topk)
            return g.topk_recall(queries, self.index, cfg.topk)

        if self.world_size > 1 and torch.distributed.is_initialized():
            q_all = coll.allgather_queries(feats, self.world_size)
            scores, ids = local_recall(q_all)
            ids = coll.globalize_ids(ids.to(torch.int32), self.rank, cfg.index_size)
            recall_scores, recall_ids = coll.merge_topk_candidates(
                scores, ids, self.rank, B, self.world_size, cfg.topk
            )
        else:
            recall_scores, recall_ids = local_recall(feats)

        # salience reinforcement + decay (Membrane recall semantics)
        if self.world_size > 1:
            flat_local = coll.local_shard_ids(
                recall_ids.reshape(-1), self.rank, cfg.index_size
            )
        else:
            flat_local = recall_ids.reshape(-1)
            flat_local = flat_local[flat_local >= 0]  # -1 = unfilled slot
        self.salience.mul_(0.9999)
        self.salience.index_add_(
            0, flat_local.long(), torch.full((flat_local.numel(),), 0.01, device=self.device)
        )

        prof.mark("verdict_trust")
        # 6. verdict + trust
        hour = time.localtime().tm_hour
        freq = torch.bincount(agent_idx.long(), minlength=cfg.n_agents).to(torch.int32)
        freq_count = freq[agent_idx.long()]
        verdict, risk, sdelta, vdelta = g.firewall_verdict(
            hits["injection"], hits["redaction"], logits, agent_idx.to(torch.int32),
            self.trust_state["score"], tool_risk, freq_count, hour, cfg.n_agents,
            inj_threshold=cfg.inj_threshold,
        )
        g.trust_recompute(self.trust_state, sdelta, vdelta)

        prof.mark("audit_merkle")
        # 7. audit Merkle
        inj_score = logits.max(dim=1).values
        records = g.audit_pack(
            verdict, risk, hits["injection"], hits["redaction"], agent_idx.to(torch.int32),
            self.trust_state["score"], inj_score,
            ts_ms=int(time.time() * 1000), msg_id0=self.batch_seq * B, batch_seq=self.batch_seq,
        )
        flat = records.reshape(-1)
        offs64 = torch.arange(0, (B + 1) * 64, 64, dtype=torch.int32, device=self.device)
        leaves = g.sha256_leaves(flat, offs64)
        root = g.merkle_root(leaves)

        if self.world_size > 1 and torch.distributed.is_initialized():
            roots = coll.allgather_roots(root, self.world_size)
            root = g.merkle_root(roots)

        prof.mark("end")
        self.batch_seq += 1
        if self.audit_sink is not None:
            self.audit_sink(records, root)

        return {
            "verdict": verdict,
            "risk": risk,
            "logits": logits,
            "hits": hits,
            "features": feats,
            "recall_scores": recall_scores,
            "recall_ids": recall_ids,
            "merkle_root": root,
            "trust_scores": self.trust_state["score"],
        }


class AsyncAuditWriter:
    """Background writer: binary audit records + JSONL batch manifest with
    the Merkle chain (byte format shared with governance.audit)."""

    def __init__(self, audit_dir: str):
        import os

        self.audit_dir = audit_dir
        os.makedirs(audit_dir, exist_ok=True)
        self._q: List = []
        self._cv = threading.Condition()
        self._stop = False
        self._thread = threading.Thread(target=self._run, daemon=True)
        self._thread.start()
        self.batches_written = 0

    def __call__(self, records: torch.Tensor, root: torch.Tensor) -> None:
        rec_cpu = records.to("cpu", non_blocking=True)
        root_cpu = root.to("cpu", non_blocking=True)
        with self._cv:
            self._q.append((rec_cpu, root_cpu, time.time()))
            self._cv.notify()

    def _run(self) -> None:
        import json
        import os

        bin_path = os.path.join(self.audit_dir, "audit-records.bin")
        manifest = os.path.join(self.audit_dir, "audit-manifest.jsonl")
        while True:
            with self._cv:
                while not self._q and not self._stop:
                    self._cv.wait(0.2)
                if self._stop and not self._q:
                    return
                items = self._q
                self._q = []
            with open(bin_path, "ab") as bf, open(manifest, "a") as mf:
                for rec, root, ts in items:
                    if torch.cuda.is_available():
                        torch.cuda.synchronize()  # non_blocking D2H complete
                    raw = rec.numpy().tobytes()
                    bf.write(raw)
                    mf.write(
                        json.dumps(
                            {"ts": int(ts * 1000), "count": rec.shape[0], "root": bytes(root.numpy()).hex()}
                        )
                        + "\n"
                    )
                    self.batches_written += 1

    def close(self) -> None:
        with self._cv:
            self._stop = True
            self._cv.notify()
        self._thread.join(timeout=5.0)
