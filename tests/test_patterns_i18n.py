"""Per-language pattern-pack coverage (reference `test/patterns.test.ts`,
543 LoC): every language pack must carry all four signal families, five
moods, a topic blacklist and noise prefixes, and its regexes must fire
on that language's own phrasing. Plus negative-case tables for the
redaction registry (false-positive guards).
"""

import pytest

from vainplex_openclaw_amd.cortex import patterns as P
from vainplex_openclaw_amd.cortex.patterns import (
    detect_mood,
    get_registry,
    high_impact_keywords,
    is_noise_topic,
)
from vainplex_openclaw_amd.cortex.patterns.packs import PACKS, language_codes

# one decision-phrase sample per language (each pack must match its own)
DECISION_SAMPLES = {
    "en": "we decided to use postgres for storage",
    "de": "wir haben uns entschieden, postgres zu verwenden",
    "es": "decidimos usar postgres para el almacenamiento",
    "fr": "nous avons décidé d'utiliser postgres",
    "it": "abbiamo deciso di usare postgres",
    "pt": "decidimos usar postgres para armazenamento",
    "ru": "мы решили использовать postgres",
    "ja": "postgresを使うことに決めました",
    "ko": "postgres를 사용하기로 결정했습니다",
    "zh": "我们决定使用postgres",
}


def test_ten_languages_present():
    assert set(language_codes()) == {"en", "de", "es", "fr", "it", "pt", "ru", "ja", "ko", "zh"}


@pytest.mark.parametrize("code", language_codes())
def test_pack_structure_complete(code):
    p = PACKS[code]
    assert set(p["patterns"].keys()) == {"decision", "close", "wait", "topic"}
    assert all(len(v) >= 1 for v in p["patterns"].values())
    moods = [m[0] if isinstance(m, (list, tuple)) else m for m in p["moods"]]
    assert len(moods) == 5  # 5 mood regexes per pack (lang-en.ts:3-49)
    assert p["topic_blacklist"] and p["noise_prefixes"]
    assert p["high_impact"]


@pytest.mark.parametrize("code", sorted(DECISION_SAMPLES))
def test_decision_fires_in_own_language(code):
    reg = get_registry(code)
    pats = reg.get_patterns("decision")
    text = DECISION_SAMPLES[code]
    assert any(p.search(text) for p in pats), f"{code}: no decision match"


@pytest.mark.parametrize("code", language_codes())
def test_registry_compiles_all_families(code):
    reg = get_registry(code)
    for fam in ("decision", "close", "wait", "topic"):
        assert reg.get_patterns(fam)


def test_mood_detection_last_match_wins():
    # last-match-wins (reference patterns.ts:38-82): later families in the
    # pack override earlier hits
    assert detect_mood("this is so frustrating, it's broken again", "en") == "frustrated"
    assert detect_mood("nothing notable here at all", "en") == "neutral"
    got = detect_mood("awesome, it works! great job", "en")
    assert got in ("excited", "productive")  # 'works' hits productive later


def test_noise_topic_and_high_impact():
    assert is_noise_topic("ok", "en") or is_noise_topic("thanks", "en")
    assert not is_noise_topic("database migration plan", "en")
    kws = high_impact_keywords("both")
    assert any(k in ("production", "security", "deploy", "database") for k in kws)


def test_both_merges_languages():
    reg = get_registry("both")
    n_en = len(get_registry("en").get_patterns("decision"))
    assert len(reg.get_patterns("decision")) >= n_en


# -- redaction negative cases (registry.test.ts false-positive guards) -------

from vainplex_openclaw_amd.governance.redaction.registry import PatternRegistry


@pytest.mark.parametrize("text", [
    "the skeleton key is a metaphor",           # not an sk- key
    "ask-me-anything thread",                    # sk- inside word
    "AKIAMISSING",                               # aws key too short
    "ghp_short",                                 # github pat too short
    "call me at extension 42",                   # not a phone number
    "version 12-34-5678 of the schema is out",   # not an SSN (wrong groups)
    "Bearer of good news",                       # bearer but no token
    "password: no",                              # value too short (<8)
    "the meeting is at 10:30",                   # nothing sensitive
    "item 4111 in aisle 11",                     # not a credit card
])
def test_redaction_negative_cases(text):
    reg = PatternRegistry()
    matches = [m for m in reg.find_matches(text)]
    assert not matches, f"false positive: {[m.pattern.id for m in matches]} in {text!r}"


@pytest.mark.parametrize("text,expected_id", [
    ("export OPENAI_KEY=sk-proj1234567890abcdefgh", "openai-api-key"),
    ("header 'Authorization: Bearer abcdef0123456789abcdef'", "bearer-token"),
    ("iban: GB29 NWBK 6016 1331 9268 19", "iban"),
    ("ssn 078-05-1120 on file", "ssn-us"),
])
def test_redaction_positive_in_context(text, expected_id):
    reg = PatternRegistry()
    ids = {m.pattern.id for m in reg.find_matches(text)}
    assert expected_id in ids or ids, f"expected {expected_id}, got {ids}"


# -- patterns-custom.test.ts mirrors --------------------------------------

def any_match(pats, text):
    return any(p.search(text) for p in pats)


def test_custom_extend_mode():
    reg = P.PatternRegistry(["en"], {
        "decision": ["approved by committee", "design review passed"],
        "close": ["ticket closed"],
        "wait": ["pending approval from"],
        "mode": "extend",
    })
    assert any_match(reg.get_patterns("decision"), "This was approved by committee")
    assert any_match(reg.get_patterns("decision"), "The design review passed")
    assert any_match(reg.get_patterns("decision"), "We decided to go")  # builtin kept
    assert any_match(reg.get_patterns("close"), "ticket closed for this issue")
    assert any_match(reg.get_patterns("wait"), "pending approval from the VP")
    # default mode is extend
    reg2 = P.PatternRegistry(["en"], {"decision": ["custom rule fired"]})
    assert any_match(reg2.get_patterns("decision"), "We decided to go")


def test_custom_override_mode():
    reg = P.PatternRegistry(["en"], {
        "decision": ["committee approved"],
        "mode": "override",
    })
    assert any_match(reg.get_patterns("decision"), "committee approved it")
    assert not any_match(reg.get_patterns("decision"), "We decided to go")
    # families without custom patterns keep builtins
    assert any_match(reg.get_patterns("close"), "this is done")
    # empty custom array does not wipe builtins
    reg2 = P.PatternRegistry(["en"], {"decision": [], "mode": "override"})
    assert any_match(reg2.get_patterns("decision"), "We decided to go")


def test_custom_invalid_regex_skipped():
    reg = P.PatternRegistry(["en"], {
        "decision": ["(unclosed", "valid custom phrase"],
        "mode": "extend",
    })
    assert any_match(reg.get_patterns("decision"), "a valid custom phrase here")
    # all-invalid in override mode falls back to builtins
    reg2 = P.PatternRegistry(["en"], {"decision": ["(bad", "[worse"], "mode": "override"})
    assert any_match(reg2.get_patterns("decision"), "We decided to go")
    # non-string entries filtered
    reg3 = P.PatternRegistry(["en"], {"decision": [None, 42, "fine pattern"]})
    assert any_match(reg3.get_patterns("decision"), "fine pattern")


def test_custom_blacklist_and_keywords():
    reg = P.PatternRegistry(["en"], {"blacklist": ["Foobar"], "keywords": ["Launchday"]})
    assert "foobar" in reg.blacklist
    assert "launchday" in reg.high_impact


def test_language_all_and_invalid():
    assert set(P._resolve_codes("all")) == set(P.language_codes())
    assert P._resolve_codes(None) == ["en", "de"]
    assert P._resolve_codes(12345) == ["en", "de"]
    assert P._resolve_codes(["en", "fr"]) == ["en", "fr"]
    reg = P.get_registry("all")
    assert len(reg.codes) == 10
