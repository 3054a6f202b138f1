"""Localize MX scan numeric errors: exact fp8-decoded reference vs the
raw threshold-scan scores, per D (np=1 isolates the pipeline) and by
lane-coordinate groupings (row%16, col%16, col-block) to identify which
fragment mapping is off."""

import torch

from vainplex_openclaw_amd.ops import gpu as g


def run(D, mode, nq=256, nx=256, S=8):
    torch.manual_seed(3)
    Q = torch.nn.functional.normalize(torch.randn(nq, D, device="cuda"), dim=1).bfloat16()
    X = torch.nn.functional.normalize(torch.randn(nx, D, device="cuda"), dim=1).bfloat16()
    Q8, X8 = g.to_fp8_bytes(Q), g.to_fp8_bytes(X)
    Qd = Q8.view(torch.float8_e4m3fn).float()
    Xd = X8.view(torch.float8_e4m3fn).float()
    ref = Qd @ Xd.T  # includes the x8 input scales -> x64 scores
    theta = torch.full((nq,), -1e9, device="cuda")
    cs, ci, counts = g.ext().topk_scan_threshold(Q8, X8, theta, 4096, S,
                                                 True, mode == "mx")
    assert int(counts.min()) == nx and int(counts.max()) == nx, counts
    S = torch.full((nq, nx), float("nan"), device="cuda")
    rows = torch.arange(nq, device="cuda").unsqueeze(1).expand(nq, nx)
    S[rows.reshape(-1), ci[:, :nx].reshape(-1).long()] = cs[:, :nx].reshape(-1)
    err = (S - ref).abs()
    bad = err > 0.05 * ref.abs().clamp_min(1.0)
    print(f"{mode} D={D} nq={nq} nx={nx} S={s_used(S, nq, nx)}: "
          f"maxerr={err.max().item():.4f} badfrac={bad.float().mean().item():.3f}")
    if bad.any():
        # localize: error rate by row%16, by col%16, by col 16-block, row 16-block
        for name, idx, n in [("row%16", rows % 16, 16),
                             ("col%16", torch.arange(nx, device="cuda").unsqueeze(0).expand(nq, nx) % 16, 16),
                             ("colblk", torch.arange(nx, device="cuda").unsqueeze(0).expand(nq, nx) // 16, 16),
                             ("rowblk", rows // 16, 16)]:
            rates = [bad[idx == i].float().mean().item() for i in range(n)]
            print(f"  {name}: " + " ".join(f"{r:.2f}" for r in rates))


def s_used(S, nq, nx):
    return S


if __name__ == "__main__" and "--fp4x4" not in __import__("sys").argv:
    # multi-x-tile per swath (nx=1024, S=1 -> 4 sequential tiles)
    run(1024, "fp8", nq=256, nx=1024, S=1)
    run(1024, "mx", nq=256, nx=1024, S=1)
    # multiple q-blocks
    run(1024, "mx", nq=512, nx=256, S=8)
    # the failing test's shape with default swath heuristic
    run(1024, "fp8", nq=512, nx=4096, S=0)
    run(1024, "mx", nq=512, nx=4096, S=0)
    run(1024, "mx", nq=256, nx=4096, S=16)


def dec_fp4(X4, XS, D):
    GRID = torch.tensor([0., .5, 1., 1.5, 2., 3., 4., 6.], device=X4.device)
    N = X4.shape[0]
    codes = torch.stack([(X4 & 0xF).long(), (X4 >> 4).long()], dim=2).reshape(N, D)
    dec_p = GRID[codes & 7] * torch.where(codes >= 8, -1.0, 1.0) * \
        torch.exp2(XS.float() - 127.0).repeat_interleave(32, dim=1)
    perm = g.fp4_perm128().to(X4.device)
    full = (torch.arange(0, D, 128, device=X4.device).unsqueeze(1) + perm.unsqueeze(0)).reshape(-1)
    dec = torch.empty_like(dec_p)
    dec[:, full] = dec_p
    return dec


def run_fp4x4(D, nq=256, nx=256, S=8):
    torch.manual_seed(3)
    Q = torch.nn.functional.normalize(torch.randn(nq, D, device="cuda"), dim=1).bfloat16()
    X = torch.nn.functional.normalize(torch.randn(nx, D, device="cuda"), dim=1).bfloat16()
    Q4, QS = g.to_fp4_mx(Q)
    X4, XS = g.to_fp4_mx(X)
    ref = dec_fp4(Q4, QS, D) @ dec_fp4(X4, XS, D).T
    theta = torch.full((nq,), -1e9, device="cuda")
    cs, ci, counts = g.ext().topk_scan_threshold_fp4x4(Q4, QS, X4, XS, theta, 4096, S)
    assert int(counts.min()) == nx and int(counts.max()) == nx, counts
    Smat = torch.full((nq, nx), float("nan"), device="cuda")
    rows = torch.arange(nq, device="cuda").unsqueeze(1).expand(nq, nx)
    Smat[rows.reshape(-1), ci[:, :nx].reshape(-1).long()] = cs[:, :nx].reshape(-1)
    err = (Smat - ref).abs()
    bad = err > 0.02 * ref.abs().clamp_min(0.5)
    print(f"fp4x4 D={D} nq={nq} nx={nx}: maxerr={err.max().item():.4f} "
          f"badfrac={bad.float().mean().item():.3f}")
    if bad.any():
        cols = torch.arange(nx, device="cuda").unsqueeze(0).expand(nq, nx)
        for name, idx in [("row%16", rows % 16), ("col%16", cols % 16),
                          ("rowblk", rows // 16), ("colblk", cols // 16)]:
            rates = [bad[idx == i].float().mean().item() for i in range(16)]
            print(f"  {name}: " + " ".join(f"{r:.2f}" for r in rates))


if __name__ == "__main__" and "--fp4x4" in __import__("sys").argv:
    for D in (1024, 512, 256, 128):
        try:
            run_fp4x4(D)
        except AssertionError as exc:
            print(f"D={D} counts wrong:", str(exc)[:120])
