#!/usr/bin/env python3
"""Microbenchmark for the recall kernels: isolates scan dtype (bf16 vs
fp8), candidate width k, and queue effects. Prints one line per config.
"""

import argparse
import sys
import time

import torch

sys.path.insert(0, __import__("os").path.dirname(__import__("os").path.abspath(__file__)) + "/..")

from vainplex_openclaw_amd.ops import gpu as g


def timed(fn, iters=3, warmup=1):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1000


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--rows", type=int, default=4_194_304)
    ap.add_argument("--nq", type=int, default=4096)
    ap.add_argument("--dim", type=int, default=1024)
    ap.add_argument("--iters", type=int, default=3)
    ap.add_argument("--short", action="store_true", help="1 iter of the two main configs")
    args = ap.parse_args()

    torch.manual_seed(0)
    dev = "cuda:0"
    Q = torch.nn.functional.normalize(torch.randn(args.nq, args.dim, device=dev), dim=1).bfloat16()
    X = torch.nn.functional.normalize(torch.randn(args.rows, args.dim, device=dev), dim=1).bfloat16()
    X8 = g.to_fp8_bytes(X)
    Q8 = g.to_fp8_bytes(Q)

    configs = [
        ("bf16 k=16", lambda: g.topk_recall(Q, X, 16)),
        ("fp8  k=16", lambda: g.topk_recall_fp8(Q8, X8, 16)),
    ]
    if not args.short:
        configs += [
            ("bf16 k=32", lambda: g.topk_recall(Q, X, 32)),
            ("fp8  k=32", lambda: g.topk_recall_fp8(Q8, X8, 32)),
            ("two-stage k=16 (of=4)", lambda: g.topk_recall_two_stage(Q, X, X8, 16)),
            ("two-stage k=16 (of=2)", lambda: g.topk_recall_two_stage(Q, X, X8, 16, overfetch=2)),
        ]
    iters = 1 if args.short else args.iters
    for name, fn in configs:
        ms = timed(fn, iters=iters, warmup=1)
        gb = args.rows * args.dim * (1 if "fp8" in name else 2) / 1e9
        work_tf = 2 * args.nq * args.rows * args.dim / 1e12
        print(f"{name:24s} {ms:9.2f} ms   {work_tf/ms*1000:7.1f} TF/s   index {gb:.1f} GB")




def scan_only_bench():
    import torch, time
    from vainplex_openclaw_amd.ops import gpu as g
    torch.manual_seed(0)
    dev = "cuda:0"
    rows, nq, dim = 4_194_304, 4096, 1024
    Q = torch.nn.functional.normalize(torch.randn(nq, dim, device=dev), dim=1).bfloat16()
    X = torch.nn.functional.normalize(torch.randn(rows, dim, device=dev), dim=1).bfloat16()
    X8 = g.to_fp8_bytes(X); Q8 = g.to_fp8_bytes(Q)
    for name, fn in [
        ("scan-only bf16 S=16", lambda: g.ext().topk_scan_only(Q, X, 16, False)),
        ("scan-only fp8  S=16", lambda: g.ext().topk_scan_only(Q8, X8, 16, True)),
        ("scan-only MX   S=16", lambda: g.ext().topk_scan_only(Q8, X8, 16, True, True)),
        ("scan-only MX   S=8",  lambda: g.ext().topk_scan_only(Q8, X8, 8, True, True)),
        ("scan-only bf16 S=8",  lambda: g.ext().topk_scan_only(Q, X, 8, False)),
        ("scan-only bf16 S=32", lambda: g.ext().topk_scan_only(Q, X, 32, False)),
    ]:
        ms = timed(fn, iters=3, warmup=1)
        print(f"{name:24s} {ms:9.2f} ms   {2*nq*rows*dim/1e12/ms*1000:7.1f} TF/s")




def threshold_bench():
    import torch
    from vainplex_openclaw_amd.ops import gpu as g
    torch.manual_seed(0)
    dev = "cuda:0"
    rows, nq, dim = 4_194_304, 4096, 1024
    Q = torch.nn.functional.normalize(torch.randn(nq, dim, device=dev), dim=1).bfloat16()
    X = torch.nn.functional.normalize(torch.randn(rows, dim, device=dev), dim=1).bfloat16()
    X8 = g.to_fp8_bytes(X)
    X4 = g.to_fp4_mx(X)
    for name, fn in [
        ("thresh bf16 k=16", lambda: g.topk_recall_threshold(Q, X, 16)),
        ("thresh fp8  k=16", lambda: g.topk_recall_threshold(Q, X, 16, X8=X8, mx=False)),
        ("thresh MX   k=16", lambda: g.topk_recall_threshold(Q, X, 16, X8=X8)),
        ("thresh fp4  k=16", lambda: g.topk_recall_threshold(Q, X, 16, X4=X4, q4=False)),
        ("thresh fp4x4 k=16", lambda: g.topk_recall_threshold(Q, X, 16, X4=X4)),
        ("thresh bf16 k=32", lambda: g.topk_recall_threshold(Q, X, 32)),
        ("direct bf16 k=16", lambda: g.topk_recall(Q, X, 16)),
    ]:
        ms = timed(fn, iters=2, warmup=1)
        print(f"{name:24s} {ms:9.2f} ms   {2*nq*rows*dim/1e12/ms*1000:7.1f} TF/s")


if __name__ == "__main__":
    import sys as _sys
    if "--scan-only" in _sys.argv:
        scan_only_bench()
    elif "--threshold" in _sys.argv:
        threshold_bench()
    else:
        main()
