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


# ===========================================================================
# Full per-language signal tables (patterns.test.ts parity depth): one
# positive sample per family per language, asserted against THAT
# language's registry, plus cross-language negatives.
# ===========================================================================

CLOSE_SAMPLES = {
    "en": "ok that is done now",
    "de": "das ist erledigt und gut",
    "es": "está resuelto por fin",
    "fr": "c'est fait enfin",
    "it": "è fatto finalmente",
    "pt": "está feito afinal",
    "ru": "всё готово наконец",
    "ja": "タスクは完了です",
    "ko": "작업 완료 했습니다",
    "zh": "任务完成了",
}

WAIT_SAMPLES = {
    "en": "waiting for the review to land",
    "de": "wir warten auf die freigabe",
    "es": "esperando a la revisión",
    "fr": "en attente de la validation",
    "it": "in attesa di conferma",
    "pt": "aguardando a aprovação",
    "ru": "ждём подтверждения",
    "ja": "レビューを待っています",
    "ko": "승인 대기 중 입니다",
    "zh": "等待审核结果",
}

TOPIC_SAMPLES = {
    "en": "let's talk about the billing system",
    "de": "zurück zu dem deployment plan",
    "es": "hablemos de la nueva arquitectura",
    "fr": "parlons de la migration batch",
    "it": "parliamo di architettura nuova",
    "pt": "agora sobre o plano de backup",
    "ru": "давайте обсудим новую схему",
    "ja": "データベースについて 設計を考えます",
    "ko": "배포 에 대해 이야기해 봅시다 계획을",
    "zh": "关于 数据库迁移的事",
}

MOOD_SAMPLES = {
    ("en", "frustrated"): "this is broken again",
    ("en", "excited"): "awesome work team",
    ("en", "tense"): "careful, deadline is near",
    ("en", "productive"): "merged and shipped",
    ("en", "exploratory"): "what if we cache it",
    ("de", "frustrated"): "alles kaputt, mist",
    ("de", "excited"): "das ist genial",
    ("de", "tense"): "vorsicht, das ist riskant",
    ("de", "productive"): "behoben und fertig",
    ("de", "exploratory"): "vielleicht ein experiment",
    ("es", "frustrated"): "está roto otra vez",
    ("es", "excited"): "quedó genial",
    ("es", "tense"): "es urgente, cuidado",
    ("es", "productive"): "arreglado y desplegado",
    ("es", "exploratory"): "quizás un experimento",
    ("fr", "frustrated"): "c'est cassé encore, zut",
    ("fr", "excited"): "c'est génial",
    ("fr", "tense"): "attention c'est risqué",
    ("fr", "productive"): "déployé et réglé",
    ("fr", "exploratory"): "et si on essayait une idée",
    ("it", "frustrated"): "è rotto di nuovo, accidenti",
    ("it", "excited"): "è fantastico",
    ("it", "tense"): "attenzione, è rischioso",
    ("it", "productive"): "risolto e deployato",
    ("it", "exploratory"): "forse un esperimento",
    ("pt", "frustrated"): "quebrado de novo, droga",
    ("pt", "excited"): "ficou incrível",
    ("pt", "tense"): "cuidado, é arriscado",
    ("pt", "productive"): "resolvido e implantado",
    ("pt", "exploratory"): "talvez um experimento",
    ("ru", "frustrated"): "опять сломалось, блин",
    ("ru", "excited"): "получилось отлично",
    ("ru", "tense"): "осторожно, это срочно",
    ("ru", "productive"): "сделано и работает",
    ("ru", "exploratory"): "может быть эксперимент",
    ("ja", "frustrated"): "また壊れた、最悪",
    ("ja", "excited"): "すごい、最高です",
    ("ja", "tense"): "注意、締め切りが近い",
    ("ja", "productive"): "修正してデプロイ済み",
    ("ja", "exploratory"): "もし試してみたら",
    ("ko", "frustrated"): "또 고장났어, 짜증",
    ("ko", "excited"): "대박 멋지다",
    ("ko", "tense"): "조심해, 마감이 긴급해",
    ("ko", "productive"): "수정 후 배포됨",
    ("ko", "exploratory"): "아이디어 하나 해보자",
    ("zh", "frustrated"): "又坏了，真烦人",
    ("zh", "excited"): "太棒了",
    ("zh", "tense"): "小心，这个紧急",
    ("zh", "productive"): "修好了，已部署",
    ("zh", "exploratory"): "也许可以试试",
}

NEGATIVE_TEXT = "the quarterly numbers look flat across regions"


def _fires(reg, family, text):
    return any(rx.search(text) for rx in reg.get_patterns(family))


@pytest.mark.parametrize("code", language_codes())
def test_decision_fires_own_language(code):
    reg = get_registry(code)
    assert _fires(reg, "decision", DECISION_SAMPLES[code]), code


@pytest.mark.parametrize("code", language_codes())
def test_close_fires_own_language(code):
    reg = get_registry(code)
    assert _fires(reg, "close", CLOSE_SAMPLES[code]), code


@pytest.mark.parametrize("code", language_codes())
def test_wait_fires_own_language(code):
    reg = get_registry(code)
    assert _fires(reg, "wait", WAIT_SAMPLES[code]), code


@pytest.mark.parametrize("code", language_codes())
def test_topic_fires_own_language(code):
    reg = get_registry(code)
    assert _fires(reg, "topic", TOPIC_SAMPLES[code]), code


@pytest.mark.parametrize("code", language_codes())
def test_no_family_fires_on_neutral_text(code):
    reg = get_registry(code)
    for family in ("decision", "close", "wait", "topic"):
        assert not _fires(reg, family, NEGATIVE_TEXT), (code, family)


@pytest.mark.parametrize("code,mood", sorted(MOOD_SAMPLES))
def test_mood_detected_per_language(code, mood):
    got = detect_mood(MOOD_SAMPLES[(code, mood)], code)
    assert got == mood, (code, mood, got)


@pytest.mark.parametrize("code", language_codes())
def test_mood_neutral_on_neutral_text(code):
    assert detect_mood(NEGATIVE_TEXT, code) == "neutral"
    assert detect_mood("", code) == "neutral"


def test_close_checkmark_fires_in_every_language():
    for code in language_codes():
        assert _fires(get_registry(code), "close", "✅"), code


def test_both_is_en_plus_de():
    reg = get_registry("both")
    assert _fires(reg, "decision", DECISION_SAMPLES["en"])
    assert _fires(reg, "decision", DECISION_SAMPLES["de"])
    assert not _fires(reg, "decision", DECISION_SAMPLES["ru"])


def test_all_covers_every_language():
    reg = get_registry("all")
    for code in language_codes():
        assert _fires(reg, "decision", DECISION_SAMPLES[code]), code


def test_language_list_selection():
    reg = get_registry(["es", "ru"])
    assert _fires(reg, "decision", DECISION_SAMPLES["es"])
    assert _fires(reg, "decision", DECISION_SAMPLES["ru"])
    assert not _fires(reg, "wait", WAIT_SAMPLES["ja"])


@pytest.mark.parametrize("code", ["en", "de", "es", "fr", "it", "pt"])
def test_case_insensitive_decision(code):
    reg = get_registry(code)
    assert _fires(reg, "decision", DECISION_SAMPLES[code].upper()), code


def test_mood_last_match_wins_ordering():
    # 'broken again' (frustrated) appears AFTER 'awesome' (excited)
    assert detect_mood("awesome start but broken again", "en") == "frustrated"
    assert detect_mood("broken again but awesome now", "en") == "excited"
    # same rule across languages
    assert detect_mood("genial aber alles kaputt", "de") == "frustrated"


def test_undecided_partial_word_no_decision_fire():
    # patterns.test.ts: 'undecided' must not fire the decision family
    reg = get_registry("en")
    assert not _fires(reg, "decision", "the team is still undecided about it")


NOISE_TOPICS = [
    ("it", True), ("that", True), ("the", True), ("abc", True),
    ("i something", True), ("billing system", False),
    ("storage migration", False), ("x" * 61, True),
    ("two\nlines", True), ("today", True), ("deploy plan", False),
]


@pytest.mark.parametrize("topic,noisy", NOISE_TOPICS)
def test_noise_topic_table(topic, noisy):
    assert is_noise_topic(topic, "en") == noisy, topic


@pytest.mark.parametrize("code", language_codes())
def test_high_impact_keywords_nonempty_and_lower(code):
    kws = high_impact_keywords(code)
    assert len(kws) >= 8
    assert all(k == k.lower() for k in kws)


# ===========================================================================
# patterns-custom.test.ts depth: extend/override per family, invalid
# regex skip, custom blacklist/keywords, language resolution table
# ===========================================================================

from vainplex_openclaw_amd.cortex.patterns import _resolve_codes


@pytest.mark.parametrize("family,phrase", [
    ("decision", "XDECIDEX now"),
    ("close", "XCLOSEX done-ish"),
    ("wait", "XWAITX pending"),
    ("topic", "XTOPICX something"),
])
def test_custom_extend_appends_per_family(family, phrase):
    reg = get_registry("en", custom={family: [phrase.split()[0]]})
    assert any(rx.search(phrase) for rx in reg.get_patterns(family))
    # builtins still present in extend mode
    assert any(rx.search(DECISION_SAMPLES["en"])
               for rx in reg.get_patterns("decision"))


def test_custom_override_replaces_only_that_family():
    reg = get_registry("en", custom={"mode": "override",
                                     "decision": ["ONLYTHIS"]})
    pats = reg.get_patterns("decision")
    assert len(pats) == 1 and pats[0].search("ONLYTHIS here")
    assert not any(rx.search(DECISION_SAMPLES["en"]) for rx in pats)
    # close family untouched by a decision-only override
    assert any(rx.search(CLOSE_SAMPLES["en"]) for rx in reg.get_patterns("close"))


def test_custom_override_empty_array_keeps_builtins():
    reg = get_registry("en", custom={"mode": "override", "decision": []})
    assert any(rx.search(DECISION_SAMPLES["en"])
               for rx in reg.get_patterns("decision"))


def test_custom_invalid_regex_skipped_silently():
    reg = get_registry("en", custom={"decision": ["(unclosed", "VALIDPAT"]})
    pats = reg.get_patterns("decision")
    assert any(rx.search("VALIDPAT") for rx in pats)
    # all-invalid: builtins remain, nothing raises
    reg2 = get_registry("en", custom={"mode": "override",
                                      "decision": ["(bad", "[worse"]})
    assert any(rx.search(DECISION_SAMPLES["en"])
               for rx in reg2.get_patterns("decision"))


def test_custom_blacklist_and_keywords_merge():
    reg = get_registry("en", custom={"blacklist": ["Widget"],
                                     "keywords": ["Rollout"]})
    assert "widget" in reg.blacklist
    assert "rollout" in reg.high_impact
    assert "architecture" in reg.high_impact  # builtins kept


def test_custom_non_string_values_filtered():
    reg = get_registry("en", custom={"decision": [42, None, "OKPAT"],
                                     "blacklist": [7, "ok"],
                                     "keywords": [None, "kw"]})
    assert any(rx.search("OKPAT") for rx in reg.get_patterns("decision"))
    assert "ok" in reg.blacklist and 7 not in reg.blacklist
    assert "kw" in reg.high_impact


@pytest.mark.parametrize("arg,want", [
    ("en", ["en"]),
    ("de", ["de"]),
    ("both", ["en", "de"]),
    (["en", "fr"], ["en", "fr"]),
    ("ja", ["ja"]),
    (42, ["en", "de"]),
    (None, ["en", "de"]),
])
def test_resolve_codes_table(arg, want):
    assert _resolve_codes(arg) == want


def test_resolve_all_is_every_language():
    assert _resolve_codes("all") == list(language_codes())


def test_all_languages_merge_blacklist_and_keywords():
    reg = get_registry("all")
    # one word from each of several languages' blacklists
    for w in ("it", "das", "eso", "ça", "это"):
        assert w in reg.blacklist, w
    for kw in ("architecture", "sicherheit", "seguridad", "миграция", "安全"):
        assert kw in reg.high_impact, kw
