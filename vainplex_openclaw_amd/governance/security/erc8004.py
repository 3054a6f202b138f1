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
    """Reputation score -> governance trust tier (logging helper)."""
    if score >= 80:
        return "elevated"
    if score >= 60:
        return "trusted"
    if score >= 40:
        return "standard"
    if score >= 20:
        return "restricted"
    return "untrusted"


ZERO_ADDRESS = "0x" + "0" * 40


def encode_uint256_arg(value: int) -> str:
    """uint256 -> 64 hex chars (erc8004-client.ts encodeUint256)."""
    return format(max(0, int(value)), "x").rjust(64, "0")


def decode_address(hex_data: str) -> str:
    """Left-padded 32-byte slot -> 0x-address; short input -> zero addr."""
    data = str(hex_data).replace("0x", "")
    if len(data) < 64:
        return ZERO_ADDRESS
    return "0x" + data[24:64]


def decode_profile(hex_data: str) -> Dict[str, int]:
    """3-slot AgentProfile (score, feedbackCount, lastUpdated); short or
    empty responses decode to all-zero defaults."""
    data = str(hex_data).replace("0x", "")
    if len(data) < 192:
        return {"score": 0, "feedbackCount": 0, "lastUpdated": 0}
    return {
        "score": int(data[0:64], 16),
        "feedbackCount": int(data[64:128], 16),
        "lastUpdated": int(data[128:192], 16),
    }


def classify_reputation(registered: bool, feedback_count: int, score: int) -> str:
    """ERC-8004 reputation tier (erc8004-client.ts classifyTier):
    unregistered / none (no feedback) / high >= 70 / medium 30-69 / low."""
    if not registered:
        return "unregistered"
    if feedback_count <= 0:
        return "none"
    if score >= 70:
        return "high"
    if score >= 30:
        return "medium"
    return "low"


class LRUCache:
    """Capacity + TTL cache (erc8004-client.ts LRUCache). TTL is measured
    from insertion time; eviction is by least-recent ACCESS — get()
    refreshes recency (lastAccess) without extending the TTL, matching the
    reference's `entry.lastAccess` update on get."""

    def __init__(self, capacity: int = 100, ttl_s: float = 300.0, clock=time.time):
        self.capacity = capacity
        self.ttl_s = ttl_s
        self.clock = clock
        self._data: Dict[str, Dict[str, Any]] = {}

    def get(self, key: str) -> Optional[Any]:
        e = self._data.get(key)
        if e is None:
            return None
        if self.clock() - e["at"] >= self.ttl_s:
            del self._data[key]
            return None
        e["last_access"] = self.clock()
        return e["value"]

    def has(self, key: str) -> bool:
        return self.get(key) is not None

    def put(self, key: str, value: Any) -> None:
        if key not in self._data and len(self._data) >= self.capacity:
            coldest = min(self._data, key=lambda k: self._data[k]["last_access"])
            del self._data[coldest]
        now = self.clock()
        self._data[key] = {"at": now, "last_access": now, "value": value}

    def clear(self) -> None:
        self._data.clear()

    def __len__(self) -> int:
        return len(self._data)


class ERC8004Client:
    def __init__(
        self,
        rpc_call: Optional[Callable[[str, list], Any]] = None,
        registry: str = IDENTITY_REGISTRY,
        cache: Optional[LRUCache] = None,
    ):
        self.rpc_call = rpc_call
        self.registry = registry
        self.cache = cache or LRUCache()

    def get_cache(self) -> LRUCache:
        return self.cache

    def _call(self, data: str) -> Optional[str]:
        result = self.rpc_call("eth_call", [{"to": self.registry, "data": data}, "latest"])
        return None if result is None else str(result)

    def lookup_reputation(self, agent_address: str) -> Optional[Dict[str, Any]]:
        """Two-step lookup: ownerOf (registered?) then the profile.
        Fail-open: any transport/RPC error returns None."""
        if self.rpc_call is None:
            return None
        cached = self.cache.get(agent_address)
        if cached is not None:
            return cached
        try:
            owner_raw = self._call(encode_call("ownerOf(address)", agent_address))
            if owner_raw is None:
                return None  # transport failure: fail-open, not "unregistered"
            owner = decode_address(owner_raw)
            if owner == ZERO_ADDRESS:
                rep = {
                    "address": agent_address, "registered": False,
                    "score": 0, "feedbackCount": 0,
                    "tier": classify_reputation(False, 0, 0),
                }
                self.cache.put(agent_address, rep)
                return rep
            prof_raw = self._call(encode_call("getProfile(address)", agent_address))
            prof = decode_profile(prof_raw or "")
            score = min(100, max(0, prof["score"]))
            rep = {
                "address": agent_address, "registered": True,
                "score": score, "feedbackCount": prof["feedbackCount"],
                "tier": classify_reputation(True, prof["feedbackCount"], score),
            }
            self.cache.put(agent_address, rep)
            return rep
        except Exception:
            return None

    def submit_feedback(self, agent_address: str, score: int) -> None:
        """Phase 2 stub (erc8004-client.ts submitFeedback returns null)."""
        return None


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
