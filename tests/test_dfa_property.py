"""Property-based fuzzing of the regex->DFA compiler against Python `re`.

The GPU pattern scan's correctness rests on this compiler (SURVEY.md §7
"hard parts": PCRE-class semantics compiled to DFAs). Hypothesis drives
random inputs through every production pattern family and asserts
bit-exact hit-mask parity with `re` on the same text.
"""

import re

import pytest
from hypothesis import given, settings, strategies as st

from vainplex_openclaw_amd.ops import pattern_sets as ps
from vainplex_openclaw_amd.ops.dfa import compile_patterns, scan_with_eof

FAMILIES = {
    "redaction": (ps.REDACTION_PATTERNS, ps.REDACTION_IGNORECASE),
    "injection": (ps.INJECTION_PATTERNS, getattr(ps, "INJECTION_IGNORECASE", set())),
    "claims": (ps.CLAIMS_PATTERNS, getattr(ps, "CLAIMS_IGNORECASE", set())),
    "entity": (ps.ENTITY_PATTERNS, getattr(ps, "ENTITY_IGNORECASE", set())),
}

_DFAS = {name: ps.get_family(name) for name in FAMILIES}


def ref_mask(patterns, ignore_case_ids, text: str) -> int:
    mask = 0
    for pat, bit, _name in patterns:
        # re.ASCII: the byte-level DFA (and the reference's JS regexes,
        # which use ASCII \w/\b without the u flag) treat word chars as
        # [A-Za-z0-9_]; Python's default Unicode \w would call '²' a
        # word char and diverge.
        flags = re.ASCII | (re.IGNORECASE if bit in ignore_case_ids else 0)
        if re.search(pat, text, flags):
            mask |= 1 << bit
    return mask


# seeds that look like near-misses of the sensitive patterns
SEED_FRAGMENTS = [
    "sk-", "sk-ant-", "AKIA", "AIza", "ghp_", "glpat-", "Bearer ",
    "password: ", "-----BEGIN ", "@example.com", "411 1-11", "ignore previous",
    "system prompt", "http://", "https://evil", " Inc.", "v2.5", "4111",
    "123-45-6789", "DE44", "+4915123", "is running", "definitely",
    # round-1 regression shapes: alternation where only one branch ends in \b
    # (pending lookahead must not leak across branches), and bounded loops
    # followed by a failing continuation
    "Bearer v2.5sk-", "do anything now", "DAN", "Claude III", " IIIx",
    "v2.5.1sk", "Claude 3X", "count is 42x", "there is no podX",
]

text_strategy = st.lists(
    st.one_of(
        st.sampled_from(SEED_FRAGMENTS),
        st.text(alphabet=st.characters(min_codepoint=32, max_codepoint=126), max_size=12),
        st.sampled_from(["a1B2" * 5, "f" * 36, "x" * 35, "0" * 16, "ABCDEFGHIJKLMNOP"]),
    ),
    max_size=8,
).map("".join)


@pytest.mark.parametrize("family", sorted(FAMILIES))
@settings(max_examples=2000, deadline=None)
@given(text=text_strategy)
def test_family_parity_fuzz(family, text):
    patterns, icase = FAMILIES[family]
    mdfa = _DFAS[family]
    got = mdfa.scan(text.encode("utf-8", "replace"))
    want = ref_mask(patterns, icase, text)
    assert got == want, f"{family}: {text!r}: got {got:#x} want {want:#x}"


# Minimized regressions from round 1's advisor findings (dfa.py alternation
# pending-lookahead leak): the shared alt end-state let one branch's trailing
# \b impose a next-byte-not-word requirement on every branch.
REGRESSION_CASES = [
    # (family, text) — parity with `re` is the assertion; these inputs
    # produced wrong masks before the frontier-based _build fix.
    ("entity", "Bearer v2.5sk-"),        # product_name bit 7 was dropped
    ("entity", "Bearer v2.5"),
    ("entity", "Claude IIIx"),           # roman branch \b must still bind
    ("entity", "Claude III done"),
    ("claims", "pod count is 42x"),
    ("injection", "do anything nowadays"),
    ("redaction", "Bearer abc.def-123x"),
]


@pytest.mark.parametrize("family,text", REGRESSION_CASES)
def test_family_parity_regressions(family, text):
    patterns, icase = FAMILIES[family]
    got = _DFAS[family].scan(text.encode())
    want = ref_mask(patterns, icase, text)
    assert got == want, f"{family}: {text!r}: got {got:#x} want {want:#x}"


def test_alt_branch_pending_isolation():
    """'\\bDAN\\b|do anything now': branch 2 has no trailing boundary, so
    'do anything nowX' must hit (advisor high-severity repro)."""
    dfa = compile_patterns([(r"\bDAN\b|do anything now", 0)])
    assert scan_with_eof(dfa, b"do anything nowX") == 1
    assert scan_with_eof(dfa, b"do anything now") == 1
    assert scan_with_eof(dfa, b" DAN ") == 1
    assert scan_with_eof(dfa, b"DANX") == 0
    assert scan_with_eof(dfa, b"DAN") == 1  # EOF matures the \b


@settings(max_examples=150, deadline=None)
@given(
    data=st.binary(max_size=64),
    prefix=st.sampled_from(SEED_FRAGMENTS),
)
def test_redaction_parity_binary_fuzz(data, prefix):
    """Raw bytes (including non-UTF8) through the redaction family."""
    patterns, icase = FAMILIES["redaction"]
    blob = prefix.encode() + data
    got = _DFAS["redaction"].scan(blob)
    want = ref_mask(patterns, icase, blob.decode("latin-1"))
    assert got == want, blob


@settings(max_examples=100, deadline=None)
@given(text=st.text(alphabet="ax0-9{}[]()*+?|.\\", max_size=10))
def test_adhoc_literal_compile_fuzz(text):
    """Literal-escaped arbitrary snippets must compile and match exactly
    where `re` does (no crashes on metacharacter-looking input)."""
    pat = re.escape(text) or "x"
    dfa = compile_patterns([(pat, 0)])
    hay = ("zz" + text + "yy").encode("utf-8", "replace")
    want = 1 if re.search(pat, hay.decode("utf-8", "replace")) else 0
    assert scan_with_eof(dfa, hay) == want


# -- cortex family fuzz -----------------------------------------------------
# The cortex family compiles with ignore_case + unicode_word (bytes >= 0x80
# count as \w). Parity reference: re.IGNORECASE with default Unicode
# semantics. The fuzz alphabet sticks to characters where both agree
# (ASCII + letters; no non-letter symbols >= U+0080, whose \w-ness
# diverges by design — WORD_U8 is hit-detection grade).

CORTEX_SEEDS = [
    "decided", "we decided to", "done", "it works", "fixed!", "✅",
    "waiting for", "blocked by", "let's talk about the schema",
    "regarding", "production", "deploy", "asap", "deadline",
    "awesome", "annoying", "maybe", "experiment", "shipped",
    "beschlossen", "erledigt", "funktioniert", "warte auf", "dringend",
    "decidimos", "resuelto", "esperando a", "urgente",
    "решено", "готово", "ждём",
]


@settings(max_examples=1000, deadline=None)
@given(text=st.lists(
    st.one_of(
        st.sampled_from(CORTEX_SEEDS),
        st.text(alphabet=st.characters(min_codepoint=32, max_codepoint=126), max_size=10),
        st.text(alphabet="äöüßéабвгдежз", max_size=4),
    ),
    max_size=6,
).map(" ".join))
def test_cortex_family_parity_fuzz(text):
    from vainplex_openclaw_amd.ops import pattern_sets as ps

    patterns, icase = ps._FAMILIES.get("cortex") or (None, None)
    if patterns is None:
        ps.get_family("cortex")
        patterns, icase = ps._FAMILIES["cortex"]
    mdfa = ps.get_family("cortex")
    got = mdfa.scan(text.encode("utf-8"))
    want = 0
    for pat, bit, _name in patterns:
        if re.search(pat, text, re.IGNORECASE):
            want |= 1 << bit
    if text.isascii():
        assert got == want, f"{text!r}: got {got:#x} want {want:#x}"
    else:
        # Non-ASCII: counted \w-reps count BYTES in the byte-level DFA but
        # CHARS in re, and char count >= n implies byte count >= n — so the
        # only permitted divergence is an OVERmatch on the topic bit (the
        # one family with a counted rep). Never a missed hit.
        t = 1 << ps.CORTEX_BIT_TOPIC
        assert got & ~t == want & ~t, f"{text!r}: got {got:#x} want {want:#x}"
        assert got & want & t == want & t, f"{text!r}: topic undermatch"
