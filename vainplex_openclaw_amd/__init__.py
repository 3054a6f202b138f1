"""vainplex_openclaw_amd — MI355X-native agent-guardrail and memory engine.

A from-scratch rebuild of the capability surface of the
alberthild/vainplex-openclaw plugin suite (Governance firewall, Cortex
conversation intelligence, Knowledge Engine, Membrane episodic recall,
EventStore, Leuko health), re-designed for AMD Instinct MI355X (gfx950):

- Host runtime: Python control plane with the openclaw plugin/hook API
  (reference: packages/openclaw-governance/src/types.ts:10-26).
- Per-message hot path: hand-written HIP/CDNA4 kernels (MFMA classifier
  heads, multi-pattern DFA scans, SHA-256 Merkle audit, LDS-tiled
  cosine-kNN recall) — see csrc/ and vainplex_openclaw_amd/ops.
- Scale-out: one process per GPU, torch.distributed over RCCL/xGMI
  (all-gather of recall queries, top-k merge, Merkle-root all-reduce).
"""

__version__ = "0.1.0"
