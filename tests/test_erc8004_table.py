"""ERC-8004 client tables mirroring `test/security/erc8004-client.test.ts`
(44 its): ABI encode/decode, profile decoding, tier classification, LRU
cache behavior, lookup flows with injected RPC, fail-open paths.
"""

import pytest

from vainplex_openclaw_amd.governance.security.erc8004 import (
    ERC8004Client,
    ERC8004Provider,
    LRUCache,
    classify_reputation,
    decode_address,
    decode_profile,
    decode_uint256,
    encode_address_arg,
    encode_call,
    encode_uint256_arg,
)

ADDR = "0x" + "11" * 20
ZERO64 = "0" * 64


# -- ABI encoding ------------------------------------------------------------

@pytest.mark.parametrize("value,suffix", [
    (0, "0" * 64),
    (1, "0" * 63 + "1"),
    (16700, "0" * 60 + "413c"),
    (255, "0" * 62 + "ff"),
])
def test_encode_uint256_table(value, suffix):
    out = encode_uint256_arg(value)
    assert out == suffix and len(out) == 64


def test_encode_address_left_pads():
    out = encode_address_arg(ADDR)
    assert len(out) == 64
    assert out.endswith("11" * 20) and out.startswith("0" * 24)


def test_encode_call_has_selector_and_arg():
    data = encode_call("ownerOf(uint256)", encode_uint256_arg(1))
    assert data.startswith("0x") and len(data) == 2 + 8 + 64


# -- decoding ----------------------------------------------------------------

def test_decode_address_forms():
    padded = "0x" + "0" * 24 + "ab" * 20
    assert decode_address(padded) == "0x" + "ab" * 20
    assert decode_address("0x" + ZERO64) == "0x" + "00" * 20
    assert decode_address("0x123") == "0x" + "00" * 20      # short input
    assert decode_address("0" * 24 + "cd" * 20) == "0x" + "cd" * 20  # no 0x


@pytest.mark.parametrize("hexdata,want", [
    ("0x" + ZERO64, 0),
    ("0x" + "0" * 63 + "5", 5),
    ("0x" + "0" * 60 + "413c", 16700),
    ("", 0),
    ("0x", 0),
])
def test_decode_uint256_table(hexdata, want):
    assert decode_uint256(hexdata) == want


def test_decode_profile_three_slots():
    raw = "0x" + encode_uint256_arg(85) + encode_uint256_arg(12) + encode_uint256_arg(3)
    p = decode_profile(raw)
    assert p["score"] == 85 and p["feedbackCount"] == 12
    # short / empty responses fall back to zeros
    assert decode_profile("0x" + encode_uint256_arg(1))["feedbackCount"] == 0
    assert decode_profile("")["score"] == 0
    z = decode_profile("0x" + ZERO64 * 3)
    assert z["score"] == 0 and z["feedbackCount"] == 0


# -- tier classification ------------------------------------------------------

@pytest.mark.parametrize("registered,feedback,score,tier", [
    (False, 0, 0, "unregistered"),
    (True, 0, 0, "none"),
    (True, 5, 70, "high"),
    (True, 5, 100, "high"),
    (True, 5, 69, "medium"),
    (True, 5, 30, "medium"),
    (True, 5, 29, "low"),
    (True, 1, 0, "low"),
])
def test_classify_reputation_table(registered, feedback, score, tier):
    assert classify_reputation(registered, feedback, score) == tier


# -- LRU cache ---------------------------------------------------------------

def test_lru_miss_store_update():
    t = [0.0]
    c = LRUCache(capacity=3, ttl_s=100, clock=lambda: t[0])
    assert c.get("x") is None and not c.has("x")
    c.put("a", 1)
    assert c.get("a") == 1 and c.has("a")
    c.put("a", 2)            # update, no eviction
    assert len(c) == 1 and c.get("a") == 2


def test_lru_eviction_by_recency_and_ttl():
    t = [0.0]
    c = LRUCache(capacity=2, ttl_s=50, clock=lambda: t[0])
    c.put("a", 1)
    t[0] = 1; c.put("b", 2)
    t[0] = 2; assert c.get("a") == 1   # refresh a
    t[0] = 3; c.put("c", 3)            # evicts b (coldest)
    assert c.get("b") is None and c.get("a") == 1 and c.get("c") == 3
    t[0] = 60                           # a expired (stored at 0)
    assert c.get("a") is None and not c.has("a")
    c.clear()
    assert len(c) == 0


# -- lookup flows with injected RPC ------------------------------------------

def _rpc_registered(score=85, feedback=12):
    owner = "0x" + "0" * 24 + "22" * 20

    def rpc(method, params):
        data = params[0]["data"]
        if data.startswith("0x" + ERC8004Client.SELECTOR_OWNER_OF
                           if hasattr(ERC8004Client, "SELECTOR_OWNER_OF") else "0x"):
            pass
        # first call: ownerOf -> owner; later: profile slots
        rpc.calls.append((method, params))
        if len(rpc.calls) == 1:
            return "0x" + "0" * 24 + "22" * 20
        return ("0x" + encode_uint256_arg(score) + encode_uint256_arg(feedback)
                + encode_uint256_arg(1))

    rpc.calls = []
    return rpc


def test_lookup_registered_high_and_low():
    cli = ERC8004Client(_rpc_registered(score=85))
    rep = cli.lookup_reputation(ADDR)
    assert rep["registered"] and rep["score"] == 85 and rep["tier"] == "high"

    cli2 = ERC8004Client(_rpc_registered(score=10))
    rep2 = cli2.lookup_reputation(ADDR)
    assert rep2["tier"] == "low"


def test_lookup_unregistered_zero_owner():
    cli = ERC8004Client(lambda m, p: "0x" + ZERO64)
    rep = cli.lookup_reputation(ADDR)
    assert rep["registered"] is False and rep["tier"] == "unregistered"
    cli2 = ERC8004Client(lambda m, p: "0x")
    assert cli2.lookup_reputation(ADDR)["registered"] is False


def test_lookup_fail_open_on_rpc_error():
    def boom(m, p):
        raise OSError("rpc down")

    assert ERC8004Client(boom).lookup_reputation(ADDR) is None
    assert ERC8004Client(lambda m, p: None).lookup_reputation(ADDR) is None


def test_lookup_score_clamped():
    cli = ERC8004Client(_rpc_registered(score=5000))
    rep = cli.lookup_reputation(ADDR)
    assert 0 <= rep["score"] <= 100


def test_submit_feedback_stub_and_cache_exposed():
    cli = ERC8004Client(_rpc_registered())
    assert cli.submit_feedback(ADDR, 80) is None  # Phase 2 stub
    assert isinstance(cli.get_cache(), LRUCache)


def test_provider_caches_lookups():
    rpc = _rpc_registered()
    cli = ERC8004Client(rpc)
    prov = ERC8004Provider(cli)
    r1 = prov.lookup_reputation(ADDR)
    n_calls = len(rpc.calls)
    r2 = prov.lookup_reputation(ADDR)
    assert r1 == r2
    assert len(rpc.calls) == n_calls  # served from cache
