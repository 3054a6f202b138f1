"""SignalPatternRegistry tables mirroring cortex
`test/trace-analyzer/signals/lang/registry.test.ts` (19 its): load
semantics, merge+cache invalidation, universal patterns, runtime pack
registration (add + replace), CJK no-\b matching, builtin constant.
"""

import re

import pytest

from vainplex_openclaw_amd.cortex.trace.signals_lang import (
    BUILTIN_SIGNAL_LANGUAGES,
    SYNC_SIGNAL_LANGUAGES,
    SignalLanguagePack,
    SignalPatternRegistry,
    builtin_pack,
    default_registry,
)


def _any(rxs, text):
    return any(rx.search(text) for rx in rxs)


# -- loading ------------------------------------------------------------------

def test_load_en():
    reg = SignalPatternRegistry()
    reg.load(["en"])
    assert reg.loaded_languages() == ["en"]
    p = reg.get_patterns()
    assert len(p["correction"]["indicators"]) > 0
    assert len(p["dissatisfaction"]["indicators"]) > 0
    assert len(p["completion"]["claims"]) > 0


def test_load_de_falsch():
    reg = SignalPatternRegistry()
    reg.load(["de"])
    assert reg.loaded_languages() == ["de"]
    assert _any(reg.get_patterns()["correction"]["indicators"], "das ist falsch")


def test_load_merges_en_de():
    reg = SignalPatternRegistry()
    reg.load(["en", "de"])
    assert reg.loaded_languages() == ["en", "de"]
    p = reg.get_patterns()
    assert _any(p["correction"]["indicators"], "wrong")
    assert _any(p["correction"]["indicators"], "falsch")


def test_load_sync_honors_static_subset_only():
    # reference loadSync only loads statically-imported packs (en/de)
    reg = SignalPatternRegistry()
    reg.load_sync(["en", "fr"])
    assert reg.loaded_languages() == ["en"]
    assert SYNC_SIGNAL_LANGUAGES == ("en", "de")


def test_load_all_ten_languages():
    reg = SignalPatternRegistry()
    reg.load(BUILTIN_SIGNAL_LANGUAGES)
    assert reg.loaded_languages() == list(BUILTIN_SIGNAL_LANGUAGES)
    p = reg.get_patterns()
    assert len(p["correction"]["indicators"]) > 10
    assert len(p["dissatisfaction"]["indicators"]) > 10
    assert len(p["completion"]["claims"]) > 10


def test_load_unknown_code_ignored():
    reg = SignalPatternRegistry()
    assert reg.load(["en", "xx"]) == ["en"]
    assert reg.loaded_languages() == ["en"]


def test_fr_correction_and_es_dissatisfaction():
    reg = SignalPatternRegistry()
    reg.load(["fr"])
    assert _any(reg.get_patterns()["correction"]["indicators"], "c'est faux")
    reg2 = SignalPatternRegistry()
    reg2.load(["es"])
    assert _any(reg2.get_patterns()["dissatisfaction"]["indicators"], "olvídalo")


# -- universal patterns -------------------------------------------------------

def test_universal_question_mark():
    reg = SignalPatternRegistry()
    reg.load(["en"])
    assert _any(reg.get_patterns()["question"]["indicators"], "Is that right?")
    assert _any(reg.get_patterns()["question"]["indicators"], "真的吗？")


def test_universal_completion_emoji():
    reg = SignalPatternRegistry()
    reg.load(["en"])
    claims = reg.get_patterns()["completion"]["claims"]
    assert _any(claims, "✅") and _any(claims, "✓")


def test_universal_satisfaction_emoji():
    reg = SignalPatternRegistry()
    reg.load(["en"])
    assert _any(reg.get_patterns()["dissatisfaction"]["satisfactionOverrides"], "👍")


# -- caching ------------------------------------------------------------------

def test_get_patterns_cached_same_object():
    reg = SignalPatternRegistry()
    reg.load(["en", "de"])
    assert reg.get_patterns() is reg.get_patterns()


def test_cache_invalidated_on_load():
    reg = SignalPatternRegistry()
    reg.load(["en"])
    p1 = reg.get_patterns()
    reg.load(["en", "de"])
    p2 = reg.get_patterns()
    assert p1 is not p2
    assert len(p2["correction"]["indicators"]) > len(p1["correction"]["indicators"])


def test_cache_invalidated_on_register():
    reg = SignalPatternRegistry()
    reg.load(["en"])
    p1 = reg.get_patterns()
    reg.register_pack(builtin_pack("fr"))
    assert reg.get_patterns() is not p1


# -- runtime registration -----------------------------------------------------

def _custom_pack(code="xx"):
    return SignalLanguagePack(
        code=code, name="Test", name_en="Test Language",
        correction={"indicators": [re.compile(r"\btestfalsch\b", re.I)],
                    "shortNegatives": [re.compile(r"^\s*testnein\s*$", re.I)]},
        question={"indicators": [re.compile(r"\btestfrage\b", re.I)]},
        dissatisfaction={"indicators": [re.compile(r"\btestfrust\b", re.I)],
                         "satisfactionOverrides": [re.compile(r"\btestdanke\b", re.I)],
                         "resolutionIndicators": [re.compile(r"\btestsorry\b", re.I)]},
        completion={"claims": [re.compile(r"\btestfertig\b", re.I)]},
        system_state={"claims": [re.compile(r"\btestclaim\b", re.I)],
                      "opinionExclusions": [re.compile(r"\btestmaybe\b", re.I)]},
    )


def test_register_pack_adds_new():
    reg = SignalPatternRegistry()
    reg.load(["en"])
    reg.register_pack(_custom_pack())
    assert "xx" in reg.loaded_languages()
    p = reg.get_patterns()
    assert _any(p["correction"]["indicators"], "testfalsch")
    assert _any(p["completion"]["claims"], "testfertig")


def test_register_pack_replaces_same_code():
    reg = SignalPatternRegistry()
    reg.load(["en"])
    small = SignalLanguagePack(
        code="en", name="Custom EN", name_en="Custom English",
        correction={"indicators": [re.compile(r"\bcustomwrong\b", re.I)],
                    "shortNegatives": []},
        question={"indicators": []},
        dissatisfaction={"indicators": [], "satisfactionOverrides": [],
                         "resolutionIndicators": []},
        completion={"claims": []},
        system_state={"claims": [], "opinionExclusions": []},
    )
    reg.register_pack(small)
    p = reg.get_patterns()
    assert _any(p["correction"]["indicators"], "customwrong")
    assert not _any(p["correction"]["indicators"], "wrong is bad")


# -- CJK: substring matching, no word boundaries ------------------------------

@pytest.mark.parametrize("code,correction,completion,dissatisfied", [
    ("zh", "这个错了", "已经完成了", "算了吧"),
    ("ja", "それは違うよ", "完了しました", "もういいよ"),
    ("ko", "그건 틀렸어", "완료했습니다", "됐어 그만해"),
])
def test_cjk_no_word_boundaries(code, correction, completion, dissatisfied):
    reg = SignalPatternRegistry()
    reg.load([code])
    p = reg.get_patterns()
    assert _any(p["correction"]["indicators"], correction)
    assert _any(p["completion"]["claims"], completion)
    assert _any(p["dissatisfaction"]["indicators"], dissatisfied)


# -- constants / default registry --------------------------------------------

def test_builtin_languages_constant():
    assert len(BUILTIN_SIGNAL_LANGUAGES) == 10
    for code in ("en", "zh", "ko"):
        assert code in BUILTIN_SIGNAL_LANGUAGES


def test_default_registry_preloaded_singleton():
    a, b = default_registry(), default_registry()
    assert a is b
    assert a.loaded_languages() == list(BUILTIN_SIGNAL_LANGUAGES)


# -- detector wiring: registered packs feed detection -------------------------

def test_registry_extends_correction_detection():
    from vainplex_openclaw_amd.cortex.trace.events import NormalizedEvent
    from vainplex_openclaw_amd.cortex.trace.chains import reconstruct_chains
    from vainplex_openclaw_amd.cortex.trace.signals import detect_correction

    def ev(i, typ, content):
        return NormalizedEvent(id=f"e{i}", ts=1000.0 + i, agent="main",
                               session="s", type=typ,
                               payload={"content": content}, seq=i)

    events = [ev(1, "msg.in", "deploy it"), ev(2, "msg.out", "done"),
              ev(3, "msg.in", "testfalsch — not like that")]
    chain = reconstruct_chains(events)[0]
    assert detect_correction(chain) == []            # no builtin phrase hits
    reg = SignalPatternRegistry()
    reg.register_pack(_custom_pack())
    found = detect_correction(chain, registry=reg)
    assert len(found) == 1 and found[0].signal_type == "correction"


# -- per-language pack validation (signal-langs.test.ts: minimum pattern
#    requirements + representative phrases per language, CJK without \b) ------

_REP = {
    "en": {"correction": "that's wrong", "dissatisfaction": "this is not helpful",
           "completion": "all set", "systemState": "there are 3 errors"},
    "de": {"correction": "nicht was ich meinte", "dissatisfaction": "das hilft nicht",
           "completion": "erledigt", "systemState": "es gibt 5 fehler"},
    "es": {"correction": "eso está mal", "dissatisfaction": "sigues fallando",
           "completion": "ya funciona", "systemState": "hay 2 errores"},
    "fr": {"correction": "pas ce que j'ai demandé", "dissatisfaction": "laisse tomber",
           "completion": "ça marche maintenant", "systemState": "il y a 4 erreurs"},
    "it": {"correction": "è sbagliato", "dissatisfaction": "lascia perdere",
           "completion": "ora funziona", "systemState": "ci sono 2 errori"},
    "pt": {"correction": "está errado", "dissatisfaction": "você continua falhando",
           "completion": "funciona agora", "systemState": "há 3 erros"},
    "ru": {"correction": "это неверно", "dissatisfaction": "это не помогает",
           "completion": "теперь работает", "systemState": "есть 2 ошибки"},
    "ja": {"correction": "そうじゃない", "dissatisfaction": "役に立たない",
           "completion": "できました", "systemState": "エラーが3件"},
    "ko": {"correction": "그게 아니에요", "dissatisfaction": "도움이 안 돼요",
           "completion": "작동합니다", "systemState": "오류가 2개"},
    "zh": {"correction": "不是我要的", "dissatisfaction": "帮不上忙",
           "completion": "可以用了", "systemState": "有3个错误"},
}


@pytest.mark.parametrize("code", BUILTIN_SIGNAL_LANGUAGES)
def test_pack_minimum_pattern_requirements(code):
    pack = builtin_pack(code)
    assert pack is not None and pack.code == code
    assert pack.name and pack.name_en
    assert len(pack.correction["indicators"]) >= 4
    assert len(pack.correction["shortNegatives"]) >= 1
    assert len(pack.dissatisfaction["indicators"]) >= 4
    assert len(pack.dissatisfaction["satisfactionOverrides"]) >= 2
    assert len(pack.dissatisfaction["resolutionIndicators"]) >= 1
    assert len(pack.completion["claims"]) >= 4
    assert len(pack.system_state["claims"]) >= 1
    assert len(pack.system_state["opinionExclusions"]) >= 2
    assert len(pack.question["indicators"]) >= 2


@pytest.mark.parametrize("code", BUILTIN_SIGNAL_LANGUAGES)
def test_pack_matches_representative_phrases(code):
    pack = builtin_pack(code)
    rep = _REP[code]
    assert _any(pack.correction["indicators"], rep["correction"]), code
    assert _any(pack.dissatisfaction["indicators"], rep["dissatisfaction"]), code
    assert _any(pack.completion["claims"], rep["completion"]), code
    assert _any(pack.system_state["claims"], rep["systemState"]), code


@pytest.mark.parametrize("code", ["ja", "ko", "zh"])
def test_cjk_packs_have_no_word_boundaries(code):
    pack = builtin_pack(code)
    for cat in ("correction", "question", "dissatisfaction", "completion"):
        for rxs in pack.category(cat).values():
            for rx in rxs:
                assert r"\b" not in rx.pattern, (code, rx.pattern)
