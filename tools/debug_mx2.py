"""Reproduce the failing threshold test shape and isolate scan vs wrapper."""

import torch

from vainplex_openclaw_amd.ops import gpu as g


def main():
    torch.manual_seed(11)
    nq, nx, D, k = 512, 32768, 1024, 16
    Q = torch.nn.functional.normalize(torch.randn(nq, D, device="cuda"), dim=1).bfloat16()
    X = torch.nn.functional.normalize(torch.randn(nx, D, device="cuda"), dim=1).bfloat16()
    X8 = g.to_fp8_bytes(X)
    Q8 = g.to_fp8_bytes(Q)
    ref = torch.topk(Q.float() @ X.float().T, k, dim=1)
    ref_ids = ref.indices.cpu().numpy()

    # raw scan with a realistic theta, both modes, same S as the wrapper (0)
    sample = torch.matmul(Q, X[:16384].T).float()
    theta = ((sample.mean(1) + 3.5 * sample.std(1)) * 64.0).contiguous()
    for mode in (False, True):
        cs, ci, n = g.ext().topk_scan_threshold(Q8, X8, theta, 1024, 0, True, mode)
        n = n.cpu()
        print(f"mode mx={mode}: counts min={int(n.min())} max={int(n.max())} "
              f"mean={float(n.float().mean()):.1f}")
        # check the scan captured the true top-k (scores are ~64x ref)
        missing = 0
        for q in range(0, nq, 37):
            got = {int(v) for v in ci[q, : min(int(n[q]), 1024)].cpu()}
            want = set(ref_ids[q])
            missing += len(want - got)
        print(f"  true top-16 ids missing from scan buffers (14 queries): {missing}")

    # wrapper end-to-end, count failing queries
    for mode in (False, True):
        scores, ids = g.topk_recall_threshold(Q, X, k, X8=X8, mx=mode)
        ids_np = ids.cpu().numpy()
        failq = sum(1 for q in range(nq)
                    if len(set(ids_np[q]) & set(ref_ids[q])) < k - 2)
        print(f"wrapper mx={mode}: failing queries {failq}/{nq}")


if __name__ == "__main__":
    main()
