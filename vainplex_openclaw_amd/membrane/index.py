"""Membrane salience index: HBM-resident embedding matrix + cosine top-k.

GPU path (MI355X): embeddings from the 4-gram encoder kernel
(`csrc/encoder.hip`), recall through the LDS-tiled MFMA streaming top-k
kernel (`csrc/topk_recall.hip`) over a bf16 [capacity, dim] matrix kept
resident in HBM (288 GB/GPU = room for hundreds of millions of 1024-d
rows). Append doubles capacity in place; no rebuild.

CPU path (tests, no-GPU hosts): the same 4-gram feature hash computed in
numpy + fp32 matmul top-k — identical semantics, used as the numerics
reference.
"""

from __future__ import annotations

from typing import List, Optional, Sequence, Tuple

import numpy as np

try:
    import torch
except ImportError:  # pragma: no cover
    torch = None


class SalienceIndex:
    def __init__(
        self,
        dim: int = 1024,
        vocab: int = 65536,
        device: Optional[str] = None,
        capacity: int = 4096,
        seed: int = 1234,
    ):
        self.dim = dim
        self.vocab = vocab
        self.size = 0
        self.device = device
        self.use_gpu = bool(device and torch is not None and torch.cuda.is_available())
        rng = np.random.default_rng(seed)
        # shared random embedding table (encoder weights); bf16 on GPU.
        # Uniform, not gaussian: distribution shape is irrelevant for
        # 4-gram hash features and standard_normal is ~20x slower here.
        self._embed_np = (rng.random((vocab, dim), dtype=np.float32) - 0.5) * 0.1
        if self.use_gpu:
            self._embed = torch.from_numpy(self._embed_np).to(device).bfloat16()
            self._mat = torch.empty(capacity, dim, dtype=torch.bfloat16, device=device)
        else:
            self._mat_np = np.zeros((capacity, dim), dtype=np.float32)
        self.ids: List[str] = []  # row -> record id
        self.owners: List[str] = []  # row -> agent (isolation filter)

    @property
    def capacity(self) -> int:
        return self._mat.shape[0] if self.use_gpu else self._mat_np.shape[0]

    def _grow(self, need: int) -> None:
        cap = self.capacity
        new_cap = cap
        while new_cap < need:
            new_cap *= 2
        if new_cap == cap:
            return
        if self.use_gpu:
            m = torch.empty(new_cap, self.dim, dtype=torch.bfloat16, device=self.device)
            m[: self.size] = self._mat[: self.size]
            self._mat = m
        else:
            m = np.zeros((new_cap, self.dim), dtype=np.float32)
            m[: self.size] = self._mat_np[: self.size]
            self._mat_np = m

    # -- encoding ----------------------------------------------------------
    def encode(self, texts: Sequence[str]) -> "np.ndarray | torch.Tensor":
        msgs = [t.encode("utf-8", "replace") for t in texts]
        if self.use_gpu:
            from ..ops import gpu as g

            b, o = g.pack_messages(msgs, device=self.device)
            return g.encode_messages(b, o, self._embed, normalize=True)
        from ..ops.gpu import reference_encode

        return reference_encode(msgs, self._embed_np, normalize=True)

    # -- mutation ----------------------------------------------------------
    def add(self, agent: str, record_ids: Sequence[str], texts: Sequence[str]) -> None:
        n = len(texts)
        if n == 0:
            return
        self._grow(self.size + n)
        feats = self.encode(texts)
        if self.use_gpu:
            self._mat[self.size : self.size + n] = feats.to(torch.bfloat16)
        else:
            self._mat_np[self.size : self.size + n] = feats
        self.ids.extend(record_ids)
        self.owners.extend([agent] * n)
        self.size += n

    # -- search ------------------------------------------------------------
    def search(
        self, agent: str, query: str, k: int, overfetch: int = 4
    ) -> List[Tuple[str, float]]:
        """Top-k (record_id, cosine) for ONE agent. The matrix is shared
        across agents; isolation is enforced by over-fetching and masking
        non-owned rows (k*overfetch candidates, refill loop if needed)."""
        if self.size == 0:
            return []
        q = self.encode([query])
        want = min(self.size, max(k * overfetch, k))
        if self.use_gpu:
            from ..ops import gpu as g

            kk = min(32, want)
            scores, ids = g.topk_recall(q.to(torch.bfloat16), self._mat[: self.size], kk)
            pairs = [
                (int(i), float(s))
                for s, i in zip(scores[0].tolist(), ids[0].tolist())
                if i >= 0
            ]
        else:
            sims = self._mat_np[: self.size] @ np.asarray(q)[0]
            order = np.argsort(-sims)[:want]
            pairs = [(int(i), float(sims[i])) for i in order]
        out = [
            (self.ids[row], score)
            for row, score in pairs
            if self.owners[row] == agent
        ]
        return out[:k]
