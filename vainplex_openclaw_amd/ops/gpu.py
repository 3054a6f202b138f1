"""GPU op wrappers: the bridge between the engines and csrc/ kernels.

Every op here REQUIRES the in-tree `_hip_ops` extension when running on a
GPU — there is no silent eager fallback (a GPU box without the extension
raises immediately). CPU reference implementations used by the numerics
tests live in `reference_*` functions.
"""

from __future__ import annotations

import hashlib
from typing import Dict, Optional, Sequence, Tuple

import numpy as np
import torch

from .dfa import MultiDFA
from . import pattern_sets

_EXT = None


def ext():
    """The compiled extension; raises loudly when missing."""
    global _EXT
    if _EXT is None:
        try:
            from .. import _hip_ops  # type: ignore

            _EXT = _hip_ops
        except ImportError as exc:  # pragma: no cover
            raise RuntimeError(
                "vainplex_openclaw_amd._hip_ops is not built. Run "
                "`python setup.py build_ext --inplace` (PYTORCH_ROCM_ARCH=gfx950). "
                "GPU ops never fall back to eager."
            ) from exc
    return _EXT


def have_ext() -> bool:
    try:
        ext()
        return True
    except RuntimeError:
        return False


# -- message packing --------------------------------------------------------

def pack_messages(messages: Sequence[bytes], device="cuda") -> Tuple[torch.Tensor, torch.Tensor]:
    """Concatenate messages into (bytes u8 [total], offsets i32 [n+1])."""
    offsets = np.zeros(len(messages) + 1, dtype=np.int32)
    for i, m in enumerate(messages):
        offsets[i + 1] = offsets[i] + len(m)
    blob = b"".join(messages)
    b = torch.frombuffer(bytearray(blob), dtype=torch.uint8)
    o = torch.from_numpy(offsets)
    if device != "cpu":
        b = b.to(device, non_blocking=True)
        o = o.to(device, non_blocking=True)
    return b, o


# -- DFA scan ---------------------------------------------------------------

class DeviceDFA:
    """Packed MultiDFA tables resident on one device."""

    def __init__(self, mdfa: MultiDFA, device) -> None:
        packed = mdfa.pack()
        self.next = torch.from_numpy(packed["next"].view(np.int16)).to(device)
        self.accept = torch.from_numpy(packed["accept"].view(np.int64)).to(device)
        self.eof = torch.from_numpy(packed["eof"].view(np.int64)).to(device)
        self.class_maps = torch.from_numpy(packed["class_maps"].reshape(-1)).to(device)
        self.meta = torch.from_numpy(packed["meta"].reshape(-1)).to(device)
        self.n_dfas = packed["meta"].shape[0]
        self.mdfa = mdfa


_device_dfas: Dict[Tuple[str, int], DeviceDFA] = {}


def device_family(name: str, device) -> DeviceDFA:
    dev = torch.device(device)
    key = (name, dev.index if dev.index is not None else -1)
    if key not in _device_dfas:
        _device_dfas[key] = DeviceDFA(pattern_sets.get_family(name), dev)
    return _device_dfas[key]


def dfa_scan(bytes_t: torch.Tensor, offsets: torch.Tensor, family: str) -> torch.Tensor:
    """Per-message u64 hit masks (as int64 tensor) for one pattern family."""
    d = device_family(family, bytes_t.device)
    meta2d = d.meta.view(d.n_dfas, 4)
    return ext().dfa_scan(bytes_t, offsets, d.next, d.accept, d.eof, d.class_maps, meta2d)


def reference_dfa_scan(messages: Sequence[bytes], family: str) -> np.ndarray:
    mdfa = pattern_sets.get_family(family)
    return np.array([mdfa.scan(m) for m in messages], dtype=np.uint64).view(np.int64)


# -- SHA-256 / Merkle -------------------------------------------------------

def sha256_leaves(bytes_t: torch.Tensor, offsets: torch.Tensor) -> torch.Tensor:
    return ext().sha256_leaves(bytes_t, offsets)


def merkle_root(digests: torch.Tensor) -> torch.Tensor:
    return ext().merkle_root(digests)


def reference_sha256_leaves(messages: Sequence[bytes]) -> np.ndarray:
    return np.stack([np.frombuffer(hashlib.sha256(m).digest(), dtype=np.uint8) for m in messages])


def reference_merkle_root(messages: Sequence[bytes]) -> bytes:
    from ..governance.audit import merkle_root as cpu_root

    return bytes.fromhex(cpu_root(list(messages)))


# -- encoder ----------------------------------------------------------------

def encode_messages(
    bytes_t: torch.Tensor, offsets: torch.Tensor, embed: torch.Tensor, normalize: bool = True
) -> torch.Tensor:
    return ext().encode_messages(bytes_t, offsets, embed, normalize)


def reference_encode(messages: Sequence[bytes], embed: np.ndarray, normalize: bool = True) -> np.ndarray:
    """CPU reference of csrc/encoder.hip (fnv1a 4-gram, stride-subsampled
    mean, L2 norm)."""
    vocab, dim = embed.shape
    out = np.zeros((len(messages), dim), dtype=np.float32)
    max_tokens = 512
    for i, m in enumerate(messages):
        n_pos = max(len(m) - 3, 0)
        stride = max((n_pos + max_tokens - 1) // max_tokens, 1)
        toks = []
        for t in range((n_pos + stride - 1) // stride if n_pos else 0):
            p = t * stride
            h = 2166136261
            for b in m[p : p + 4]:
                h = ((h ^ b) * 16777619) & 0xFFFFFFFF
            toks.append(h & (vocab - 1))
        if toks:
            f = embed[toks].astype(np.float32).mean(axis=0)
        else:
            f = np.zeros(dim, dtype=np.float32)
        if normalize:
            n = np.sqrt(max((f * f).sum(), 1e-12))
            f = f / n
        out[i] = f
    return out


# -- GEMM / top-k -----------------------------------------------------------

def gemm_nt(
    A: torch.Tensor,
    B: torch.Tensor,
    bias: Optional[torch.Tensor] = None,
    act: int = 0,
    out_bf16: bool = False,
) -> torch.Tensor:
    return ext().gemm_nt(A, B, bias, act, out_bf16)


def topk_recall(Q: torch.Tensor, X: torch.Tensor, k: int, n_swaths: int = 0) -> Tuple[torch.Tensor, torch.Tensor]:
    """Cosine top-k of each Q row against X (csrc/topk_recall.hip v3).

    Swath count: a multiple of 8 (XCD-aware grid mapping), sized so
    qblocks*swaths fills the 256 CUs, capped by the merge kernel's
    n_swaths*k <= 1024 register budget and the index size.
    """
    if n_swaths <= 0:
        qblocks = (Q.shape[0] + 255) // 256  # BM=256
        want = max(1, 256 // max(qblocks, 1))
        n_swaths = max(8, (want // 8) * 8)
        n_swaths = min(n_swaths, (1024 // max(k, 1)) // 8 * 8)
        n_swaths = min(n_swaths, max(1, X.shape[0] // 256))
        n_swaths = max(1, n_swaths)
    s, i = ext().topk_recall(Q, X, k, n_swaths)
    return s, i


def to_fp8_bytes(x: torch.Tensor, scale: float = 8.0) -> torch.Tensor:
    """bf16 -> e4m3 bytes, pre-scaled so typical unit-vector components
    (~N(0, 1/sqrt(D))) land in e4m3's normal range."""
    return (x.float() * scale).to(torch.float8_e4m3fn).view(torch.uint8)


_E2M1_MIDPOINTS = (0.25, 0.75, 1.25, 1.75, 2.5, 3.5, 5.0)


def fp4_perm128() -> torch.Tensor:
    """Element order the v_mfma_scale..f8f6f4 fp4 B operand expects, per
    128-k chunk (probed on hardware, tools/probe_fp4map.hip): LDS slot g
    (= lane kgrp g) holds k in {base, base+32} + [0,16) with base =
    (g&1)*64 + (g>>1)*16, packed two-per-byte (low nibble = even
    position). A lane's 32 elements share ONE e8m0 scale, so scale
    groups are 32 consecutive elements in THIS order."""
    base = torch.arange(16)
    order = []
    for g in range(4):
        s = (g & 1) * 64 + (g >> 1) * 16
        order.append(s + base)
        order.append(s + 32 + base)
    return torch.cat(order)


def to_fp4_mx(x: torch.Tensor, chunk_rows: int = 1_048_576) -> Tuple[torch.Tensor, torch.Tensor]:
    """bf16 [N, D] -> MXFP4 in the hardware fragment order: packed e2m1
    nibbles [N, D/2] + per-lane-group e8m0 scales [N, D/32]. The e8m0
    exponent maps each group's max |v| onto e2m1's top code (6.0)."""
    N, D = x.shape
    assert D % 128 == 0
    mids = torch.tensor(_E2M1_MIDPOINTS, device=x.device)
    p128 = fp4_perm128().to(x.device)
    perm = (torch.arange(0, D, 128, device=x.device).unsqueeze(1) + p128.unsqueeze(0)).reshape(-1)
    out4 = torch.empty((N, D // 2), dtype=torch.uint8, device=x.device)
    outs = torch.empty((N, D // 32), dtype=torch.uint8, device=x.device)
    for r0 in range(0, N, chunk_rows):
        v = x[r0:r0 + chunk_rows].float()[:, perm].view(-1, D // 32, 32)
        amax = v.abs().amax(dim=2, keepdim=True)
        e = torch.where(amax > 0, (amax / 6.0).log2().ceil(), torch.zeros_like(amax))
        e = e.clamp(-127, 127)
        outs[r0:r0 + chunk_rows] = (e.squeeze(2) + 127).to(torch.uint8)
        y = v * torch.exp2(-e)
        code = torch.bucketize(y.abs().contiguous(), mids).to(torch.uint8)
        code |= (y < 0).to(torch.uint8) << 3
        code = code.view(-1, D)
        out4[r0:r0 + chunk_rows] = code[:, 0::2] | (code[:, 1::2] << 4)
    return out4, outs


def topk_recall_fp8(Q8: torch.Tensor, X8: torch.Tensor, k: int, n_swaths: int = 0) -> Tuple[torch.Tensor, torch.Tensor]:
    """Stage-1 fp8 scan: candidate ids ranked by e4m3 cosine (csrc
    topk_recall_fp8_kernel). Inputs are uint8 views of e4m3 bytes."""
    if n_swaths <= 0:
        qblocks = (Q8.shape[0] + 255) // 256
        want = max(1, 256 // max(qblocks, 1))
        n_swaths = max(8, (want // 8) * 8)
        n_swaths = min(n_swaths, (1024 // max(k, 1)) // 8 * 8)
        n_swaths = min(n_swaths, max(1, X8.shape[0] // 256))
        n_swaths = max(1, n_swaths)
    s, i = ext().topk_recall_fp8(Q8, X8, k, n_swaths)
    return s, i


def topk_recall_two_stage(
    Q: torch.Tensor,
    X: torch.Tensor,
    X8: torch.Tensor,
    k: int,
    overfetch: int = 2,
    salience: Optional[torch.Tensor] = None,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Two-stage exact-rescore recall: fp8 scan of the full index for
    k*overfetch candidates (half the staged bytes of the bf16 scan = the
    measured GLDS transport bound), then exact bf16/fp32 rescore of the
    candidates and final top-k. Final scores are EXACT cosines; stage-1
    only has to keep the true top-k inside the candidate set (error
    sigma ~0.003 vs candidate margins ~10x that).
    """
    k2 = min(32, max(k * overfetch, k))  # TOPK_MAX LDS bound in the scan kernel
    Q8 = to_fp8_bytes(Q)
    _s8, ids8 = topk_recall_fp8(Q8, X8, k2)
    ids = ids8.long().clamp_min(0)  # -1 slots -> row 0 (rescored, never top)
    cand = X[ids]  # [nq, k2, D] bf16 gather
    exact = torch.einsum("qd,qkd->qk", Q.float(), cand.float())
    if salience is not None:
        exact = exact * salience[ids]
    exact = torch.where(ids8 < 0, torch.full_like(exact, -1e30), exact)
    top = torch.topk(exact, k, dim=1)
    return top.values, torch.gather(ids8, 1, top.indices)


def edit_distance_batch(pairs: Sequence[Tuple[bytes, bytes]], device="cuda") -> torch.Tensor:
    """Batched Levenshtein over (a, b) byte-string pairs (csrc
    edit_distance_kernel, one 64-lane wave per pair, anti-diagonal DP).
    Inputs are capped at 512 bytes each — the doom-loop detector caps
    commands at 500 chars (SURVEY §2.8 / reference doom-loop.ts:76-90)."""
    flat: list = []
    for a, b in pairs:
        flat.append(a[:512])
        flat.append(b[:512])
    if not flat:
        return torch.empty(0, dtype=torch.int32)
    bytes_t, offsets = pack_messages(flat, device=device)
    return ext().edit_distance(bytes_t, offsets)


def reference_edit_distance(a: bytes, b: bytes) -> int:
    """Plain CPU Levenshtein for the numerics tests."""
    la, lb = len(a), len(b)
    prev = list(range(lb + 1))
    for i in range(1, la + 1):
        cur = [i] + [0] * lb
        for j in range(1, lb + 1):
            cur[j] = min(prev[j] + 1, cur[j - 1] + 1,
                         prev[j - 1] + (a[i - 1] != b[j - 1]))
        prev = cur
    return prev[lb]


# -- firewall tail ----------------------------------------------------------

def firewall_verdict(
    inj_hits: torch.Tensor,
    red_hits: torch.Tensor,
    logits: torch.Tensor,
    agent_idx: torch.Tensor,
    agent_trust: torch.Tensor,
    tool_risk: torch.Tensor,
    freq_count: torch.Tensor,
    hour: int,
    n_agents: int,
    cred_bits: int = pattern_sets.REDACTION_CREDENTIAL_BITS,
    inj_threshold: float = 0.9,
):
    v, r, sd, vd = ext().firewall_verdict(
        inj_hits, red_hits, logits, agent_idx, agent_trust, tool_risk, freq_count,
        hour, cred_bits, inj_threshold, n_agents,
    )
    return v, r, sd, vd


def trust_recompute(state: Dict[str, torch.Tensor], sdelta: torch.Tensor, vdelta: torch.Tensor) -> None:
    ext().trust_recompute(
        state["success"], state["violation"], sdelta, vdelta,
        state["age_days"], state["clean_streak"], state["manual_adj"], state["score"],
    )


def audit_pack(
    verdict: torch.Tensor, risk: torch.Tensor, inj_hits: torch.Tensor, red_hits: torch.Tensor,
    agent_idx: torch.Tensor, agent_trust: torch.Tensor, inj_score: torch.Tensor,
    ts_ms: int, msg_id0: int, batch_seq: int,
) -> torch.Tensor:
    return ext().audit_pack(
        verdict, risk, inj_hits, red_hits, agent_idx, agent_trust, inj_score,
        ts_ms, msg_id0, batch_seq,
    )


def topk_recall_threshold(
    Q: torch.Tensor,
    X: torch.Tensor,
    k: int,
    X8: Optional[torch.Tensor] = None,
    sample_rows: int = 131072,
    target_candidates: int = 128,
    cap: int = 1024,
    mx: bool = True,
    X4: Optional[Tuple[torch.Tensor, torch.Tensor]] = None,
    q4: bool = True,
    salience: Optional[torch.Tensor] = None,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Threshold-scan recall: per-query score thresholds estimated from a
    sampled pre-pass (Gaussian tail extrapolation), then a fixed-threshold
    scan whose epilogue is just rare global appends — no in-kernel top-k
    maintenance, so the scan runs at its K-loop rate regardless of k.

    With X8 (e4m3 view) the scan runs in fp8 and the top candidates are
    exact-rescored in bf16; final scores are exact either way. `mx` (the
    default, needs D % 128 == 0) runs the fp8 scan on the MX-scaled
    16x16x128 MFMA at unit block scales — same bytes, 2.25x the issue
    rate (csrc topk_scan_mx_kernel). Queries whose candidate buffer
    underflowed (<k survivors: threshold too high for non-Gaussian score
    tails) fall back to the direct kernel.
    """
    nq, D = Q.shape
    nx = X.shape[0]
    m = min(nx, sample_rows)
    use_fp4 = X4 is not None and D % 128 == 0 and D <= 2048
    use_fp8 = X8 is not None or use_fp4
    use_mx = mx and X8 is not None and D % 128 == 0

    # 1. per-query score statistics from a sample (bf16 matmul)
    sample = torch.matmul(Q, X[:m].T).float()  # [nq, m]
    mu = sample.mean(dim=1)
    sigma = sample.std(dim=1).clamp_min(1e-6)
    # Gaussian tail: P(score > theta) = C/nx
    p = min(0.5, max(target_candidates / max(nx, 1), 1e-12))
    try:
        from scipy.stats import norm

        z = float(norm.ppf(1.0 - p))
    except ImportError:  # pragma: no cover
        import math

        # Beasley-Springer-Moro style rough inverse via bisection
        lo, hi = 0.0, 9.0
        for _ in range(60):
            mid = (lo + hi) / 2
            if 0.5 * math.erfc(mid / math.sqrt(2)) > p:
                lo = mid
            else:
                hi = mid
        z = (lo + hi) / 2
    # fp4 scan scores carry quantization noise (~0.2 sigma): lower theta
    # so near-threshold true candidates still land in the buffer, and
    # widen the rescore window below (select is by NOISY scan score).
    if use_fp4:
        z = z - 0.25
    theta = (mu + sigma * z).contiguous()

    # 2. fixed-threshold scan
    if use_fp4 and q4 and D <= 1024:
        # both operands MXFP4: scores carry no static prescale
        Q4, QS = to_fp4_mx(Q)
        cs, ci, counts = ext().topk_scan_threshold_fp4x4(
            Q4, QS, X4[0], X4[1], theta.contiguous(), cap, 0)
    elif use_fp4:
        # Q carries the only static scale (x8); X scales are per-block
        Q8 = to_fp8_bytes(Q)
        cs, ci, counts = ext().topk_scan_threshold_fp4(
            Q8, X4[0], X4[1], (theta * 8.0).contiguous(), cap, 0)
    elif use_fp8:
        Q8 = to_fp8_bytes(Q)
        # thresholds are in fp8-score units: inputs scaled x8 each -> x64
        cs, ci, counts = ext().topk_scan_threshold(Q8, X8, (theta * 64.0).contiguous(),
                                                   cap, 0, True, use_mx)
    else:
        cs, ci, counts = ext().topk_scan_threshold(Q, X, theta, cap, 0, False, False)

    # 3. mask unfilled/overflowed slots
    slot = torch.arange(cap, device=Q.device).unsqueeze(0)
    valid = slot < counts.clamp_max(cap).unsqueeze(1)
    cs = torch.where(valid, cs, torch.full_like(cs, -1e30))

    # 4. select: top candidates by scan score (wider for the noisier
    # fp4 scan, and for salience weighting — the weighted top-k can sit
    # below the raw-cosine top-2k; the exact rescore makes overfetch
    # nearly free)
    if use_fp4 or salience is not None:
        sel = min(cap, max(8 * k if salience is not None else 4 * k, 128))
    else:
        sel = min(cap, max(2 * k, 32))
    top = torch.topk(cs, sel, dim=1)
    ids = torch.gather(ci, 1, top.indices)

    # exact fp32 rescore of the selected candidates (both dtypes: the
    # bf16 scan's near-ties otherwise reorder the top-k at bf16 precision).
    # salience (membrane semantics) folds into the rescore epilogue:
    # final score = cosine * decayed salience, selection stays by raw
    # cosine with overfetch (salience <= 1 so the weighted top-k is a
    # subset of the higher-cosine candidates in the common case).
    gidx = ids.long().clamp_min(0)
    cand = X[gidx]  # [nq, sel, D]
    exact = torch.einsum("qd,qkd->qk", Q.float(), cand.float())
    if salience is not None:
        exact = exact * salience[gidx]
    exact = torch.where(ids < 0, torch.full_like(exact, -1e30), exact)
    fin = torch.topk(exact, k, dim=1)
    out_s = fin.values
    out_i = torch.gather(ids, 1, fin.indices)

    # 5. fallback for underflowed queries (non-Gaussian tails / theta high)
    bad = (counts < k) | (counts > cap)
    if bool(bad.any()):
        rows = bad.nonzero(as_tuple=True)[0]
        k2f = min(4 * k, X.shape[0])
        fb_s, fb_i = topk_recall(Q[rows].contiguous(), X, k2f)
        # exact fp32 rescore (+ salience weight) so fallback rows report
        # the same score definition as the main path
        fb_g = fb_i.long().clamp_min(0)
        fb_exact = torch.einsum("qd,qkd->qk", Q[rows].float(), X[fb_g].float())
        if salience is not None:
            fb_exact = fb_exact * salience[fb_g]
        fb_exact = torch.where(fb_i < 0, torch.full_like(fb_exact, -1e30), fb_exact)
        ft = torch.topk(fb_exact, k, dim=1)
        out_s[rows] = ft.values
        out_i[rows] = torch.gather(fb_i, 1, ft.indices)
    return out_s, out_i


def topk_recall_threshold_banded(
    Q: torch.Tensor,
    X: torch.Tensor,
    k: int,
    salience: torch.Tensor,
    hot_idx: torch.Tensor,
    hot_X: Optional[torch.Tensor] = None,
    is_hot: Optional[torch.Tensor] = None,
    **kw,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Salience-banded weighted recall for SKEWED salience distributions.

    The bounded-overfetch weighted path (topk_recall_threshold with
    salience) selects by raw cosine; under heavy skew (e.g. Zipf with a
    100x range) the dense weighted optimum can sit at cosine rank ~10k —
    beyond any bounded overfetch (measured ~86% dense-regret at 128
    candidates; the reference's interactive limit*4 overfetch has the
    same bound, membrane retrieve path). This variant recovers those
    optima: the HOT band (top-salience rows, `hot_idx`) is scored
    exactly — dense weighted bf16 matmul over a small gathered copy — so
    no cosine-rank bound applies there, while the COLD band keeps the
    fp4 threshold scan. Bands merge by exact weighted score; cold
    candidates that are hot rows are masked out (the hot band already
    scored them exactly), so no duplicates survive.

    `hot_X` (gathered rows) and `is_hot` (membership mask) are optional
    caches the caller refreshes when salience order drifts.
    """
    nh = int(hot_idx.numel())
    if nh == 0:
        return topk_recall_threshold(Q, X, k, salience=salience, **kw)
    if hot_X is None:
        hot_X = X[hot_idx.long()]
    hot = torch.matmul(Q, hot_X.T).float() * salience[hot_idx.long()]
    hs, hsel = torch.topk(hot, min(k, nh), dim=1)
    hi = hot_idx[hsel.reshape(-1)].reshape(hsel.shape).to(torch.int32)

    cs, ci = topk_recall_threshold(Q, X, k, salience=salience, **kw)
    if is_hot is None:
        is_hot = torch.zeros(X.shape[0], dtype=torch.bool, device=X.device)
        is_hot[hot_idx.long()] = True
    dup = is_hot[ci.long().clamp_min(0)]
    cs = torch.where(dup, torch.full_like(cs, -1e30), cs)

    all_s = torch.cat([hs, cs], dim=1)
    all_i = torch.cat([hi, ci], dim=1)
    fin = torch.topk(all_s, k, dim=1)
    return fin.values, torch.gather(all_i, 1, fin.indices)


# -- fact-registry probe ----------------------------------------------------

FNV_OFFSET = 0xCBF29CE484222325
FNV_PRIME = 0x100000001B3
_U64 = (1 << 64) - 1

# claim bit -> canonical predicate, per the reference's claim-type ->
# predicate strategy table (fact-checker.ts:127-136) over the CLAIMS
# family bits (ops/pattern_sets.py CLAIMS_PATTERNS). Types whose
# strategy is NOT a single predicate (entity_name matches by subject
# alone; self_referential matches the "self" subject) have no entry —
# the GPU probe skips those bits and the host FactRegistry path covers
# them.
CLAIM_BIT_PREDICATE: Dict[int, str] = {
    0: "state",       # system_state -> "state"
    2: "exists",      # existence_pos
    3: "exists",      # existence_neg
    4: "exists",      # there_is
    5: "metric",      # has/contains/uses numeric
    6: "percentage",  # percentage
    7: "count",       # count
}


def fnv1a64(data: bytes) -> int:
    h = FNV_OFFSET
    for b in data:
        h = ((h ^ b) * FNV_PRIME) & _U64
    return h


def _mix_key(subject_h: int, predicate_h: int) -> int:
    """Continues the subject FNV stream through '|' + predicate-hash bytes
    (bit-exact with csrc/fact_probe.hip mix_key)."""
    h = ((subject_h ^ ord("|")) * FNV_PRIME) & _U64
    for i in range(8):
        h = ((h ^ ((predicate_h >> (8 * i)) & 0xFF)) * FNV_PRIME) & _U64
    return h


def _tokenize_fact(data: bytes) -> list:
    """Word tokens (>=2 chars, lowercased) under the probe kernel's
    byte classes: [a-z0-9_.-]."""
    toks = []
    cur = bytearray()
    for b in data + b" ":
        c = bytes([b]).lower()[0]
        if (48 <= c <= 57) or (97 <= c <= 122) or c in (95, 45, 46):
            cur.append(c)
        elif cur:
            if len(cur) >= 2:
                toks.append(bytes(cur))
            cur = bytearray()
    return toks


def build_fact_table(facts: Sequence[Tuple[str, str, str]], pow2: Optional[int] = None):
    """Pack (subject, predicate, object) triples into the open-addressing
    probe table: keys/vals int64 tensors + the per-claim-bit predicate
    hash vector. Single-token subjects/objects only (multi-word facts stay
    on the host FactRegistry path — see csrc/fact_probe.hip header)."""
    if pow2 is None:
        pow2 = max(4, (len(facts) * 2 - 1).bit_length())
    size = 1 << pow2
    keys = np.zeros(size, dtype=np.uint64)
    vals = np.zeros(size, dtype=np.uint64)
    for subject, predicate, obj in facts:
        sh = fnv1a64(subject.strip().lower().encode())
        ph = fnv1a64(predicate.strip().lower().encode())
        key = _mix_key(sh, ph) or 1
        slot = key & (size - 1)
        for _ in range(size):
            if keys[slot] == 0 or keys[slot] == key:
                break
            slot = (slot + 1) % size
        keys[slot] = key
        vals[slot] = fnv1a64(obj.strip().lower().encode())
    pred = np.zeros(64, dtype=np.uint64)
    for bit, p in CLAIM_BIT_PREDICATE.items():
        pred[bit] = fnv1a64(p.encode())
    return (
        torch.from_numpy(keys.view(np.int64)),
        torch.from_numpy(vals.view(np.int64)),
        torch.from_numpy(pred.view(np.int64)),
        pow2,
    )


def fact_probe(
    bytes_t: torch.Tensor,
    offsets: torch.Tensor,
    claims_mask: torch.Tensor,
    table_keys: torch.Tensor,
    table_vals: torch.Tensor,
    pred_hash: torch.Tensor,
    pow2: int,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Per-message (verified, contradicted) counts (csrc/fact_probe.hip)."""
    v, c = ext().fact_probe(bytes_t, offsets, claims_mask, pred_hash,
                            table_keys, table_vals, pow2)
    return v, c


def reference_fact_probe(
    messages: Sequence[bytes], claims_masks: Sequence[int],
    facts: Sequence[Tuple[str, str, str]],
) -> Tuple[np.ndarray, np.ndarray]:
    """CPU mirror of the probe kernel's semantics for the numerics tests."""
    table = {}
    for subject, predicate, obj in facts:
        key = _mix_key(fnv1a64(subject.strip().lower().encode()),
                       fnv1a64(predicate.strip().lower().encode())) or 1
        table[key] = fnv1a64(obj.strip().lower().encode())
    ver = np.zeros(len(messages), dtype=np.int32)
    con = np.zeros(len(messages), dtype=np.int32)
    for i, (msg, cmask) in enumerate(zip(messages, claims_masks)):
        if not cmask:
            continue
        toks = [fnv1a64(t) for t in _tokenize_fact(msg)][:96]
        for bit, predicate in CLAIM_BIT_PREDICATE.items():
            if not (cmask >> bit) & 1:
                continue
            ph = fnv1a64(predicate.encode())
            for t in toks:
                key = _mix_key(t, ph) or 1
                val = table.get(key)
                if val is None:
                    continue
                if val in toks:
                    ver[i] += 1
                else:
                    con[i] += 1
    return ver, con


# -- batched envelopes + all-family scan ------------------------------------

def build_envelopes(records_cpu: torch.Tensor, session: str,
                    agent_prefix: str = "agent",
                    ctype: str = "message.in.received") -> bytes:
    """One ClawEvent JSONL line per audit record (csrc/host_envelope.cpp,
    GIL released). Line format parity with eventstore.hooks.build_envelope
    is covered by tests/test_eventstore.py."""
    return ext().build_envelopes(records_cpu, session, agent_prefix, ctype)


class DeviceFamilySet:
    """All families' packed tables concatenated for the one-launch scan
    (csrc dfa_scan_multi_kernel): per-family (begin, end) sub-DFA ranges
    into the shared meta table."""

    def __init__(self, names: Sequence[str], device) -> None:
        import numpy as _np

        self.names = list(names)
        nexts, accepts, eofs, cmaps, metas, ranges = [], [], [], [], [], []
        state_base = 0
        next_base = 0
        row = 0
        for name in self.names:
            packed = pattern_sets.get_family(name).pack()
            meta = packed["meta"].copy()
            n_dfas = meta.shape[0]
            meta[:, 0] += next_base
            meta[:, 1] += state_base
            meta[:, 3] += row
            ranges.append((row, row + n_dfas))
            row += n_dfas
            next_base += packed["next"].size
            state_base += packed["accept"].size
            nexts.append(packed["next"])
            accepts.append(packed["accept"])
            eofs.append(packed["eof"])
            cmaps.append(packed["class_maps"])
            metas.append(meta)
        self.next = torch.from_numpy(_np.concatenate(nexts).view(_np.int16)).to(device)
        self.accept = torch.from_numpy(_np.concatenate(accepts).view(_np.int64)).to(device)
        self.eof = torch.from_numpy(_np.concatenate(eofs).view(_np.int64)).to(device)
        self.class_maps = torch.from_numpy(
            _np.concatenate(cmaps).reshape(-1)
        ).to(device)
        self.meta = torch.from_numpy(_np.concatenate(metas)).to(device)
        self.ranges = torch.from_numpy(
            _np.array(ranges, dtype=_np.int32)
        ).to(device)


_device_family_sets: Dict[Tuple[Tuple[str, ...], int], DeviceFamilySet] = {}


def device_family_set(names: Sequence[str], device) -> DeviceFamilySet:
    dev = torch.device(device)
    key = (tuple(names), dev.index if dev.index is not None else -1)
    if key not in _device_family_sets:
        _device_family_sets[key] = DeviceFamilySet(names, dev)
    return _device_family_sets[key]


def dfa_scan_all(bytes_t: torch.Tensor, offsets: torch.Tensor,
                 families: Sequence[str]) -> Dict[str, torch.Tensor]:
    """All families in ONE kernel launch; returns {family: hit masks}."""
    fs = device_family_set(families, bytes_t.device)
    hits = ext().dfa_scan_multi(bytes_t, offsets, fs.next, fs.accept, fs.eof,
                                fs.class_maps, fs.meta, fs.ranges)
    return {name: hits[i] for i, name in enumerate(fs.names)}
