"""A/B the fp4x4 threshold scan v1 (monolithic reads) vs v2 (pipelined):
bit-exact candidate parity + kernel timing at a mid-size index."""
import os
import sys
import time

import torch

sys.path.insert(0, ".")
from vainplex_openclaw_amd.ops import gpu as g


def run(nx=4_194_304, nq=4096, d=1024, cap=1024):
    torch.manual_seed(7)
    X = torch.nn.functional.normalize(
        torch.randn(nx, d, device="cuda"), dim=1).bfloat16()
    Q = torch.nn.functional.normalize(
        torch.randn(nq, d, device="cuda"), dim=1).bfloat16()
    X4, XS = g.to_fp4_mx(X)
    Q4, QS = g.to_fp4_mx(Q)
    theta = (torch.matmul(Q, X[:65536].T).float().mean(dim=1)
             + 4.2 * torch.matmul(Q, X[:65536].T).float().std(dim=1)).contiguous()

    def once(tag):
        torch.cuda.synchronize()
        # warm
        cs, ci, cn = g.ext().topk_scan_threshold_fp4x4(Q4, QS, X4, XS, theta, cap, 0)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(5):
            cs, ci, cn = g.ext().topk_scan_threshold_fp4x4(Q4, QS, X4, XS, theta, cap, 0)
        torch.cuda.synchronize()
        ms = (time.perf_counter() - t0) / 5 * 1000
        print(f"{tag}: {ms:.2f} ms/scan  counts sum={int(cn.sum())}")
        return cs.cpu(), ci.cpu(), cn.cpu(), ms

    # v1 only runs when the env var forces it; the binding latches the
    # choice in a static, so A/B needs two processes
    return once(os.environ.get("TAG", "run"))


if __name__ == "__main__":
    cs, ci, cn, ms = run()
    torch.save({"cs": cs, "ci": ci, "cn": cn, "ms": ms},
               f"gpurun_out/ab_{os.environ.get('TAG','run')}.pt")
