"""Minimal fp4x4 scan driver for rocprofv3 PMC runs: a few iterations of
just the threshold-scan kernel at the kernbench shape."""

import sys

sys.path.insert(0, __import__("os").path.dirname(__import__("os").path.abspath(__file__)) + "/..")

import torch

from vainplex_openclaw_amd.ops import gpu as g


def main():
    torch.manual_seed(0)
    rows, nq, dim = 4_194_304, 4096, 1024
    Q = torch.nn.functional.normalize(torch.randn(nq, dim, device="cuda"), dim=1).bfloat16()
    X = torch.nn.functional.normalize(torch.randn(rows, dim, device="cuda"), dim=1).bfloat16()
    Q4, QS = g.to_fp4_mx(Q)
    X4, XS = g.to_fp4_mx(X)
    sample = torch.matmul(Q, X[:131072].T).float()
    theta = (sample.mean(1) + 4.0 * sample.std(1)).contiguous()
    for _ in range(3):
        g.ext().topk_scan_threshold_fp4x4(Q4, QS, X4, XS, theta, 1024, 0)
    torch.cuda.synchronize()
    print("done")


if __name__ == "__main__":
    main()
