"""Signal-language pattern registry: 10 builtin packs + runtime packs.

Parity target: cortex `src/trace-analyzer/signals/lang/` —
- `index.ts` SignalPatternRegistry: load languages, merge category
  patterns (correction / question / dissatisfaction / completion /
  systemState) with the universal patterns, cache the merged result and
  invalidate on any load or registration; `registerSignalLanguagePack`
  replaces a same-code pack at runtime.
- `types.ts` SignalLanguagePack shape; `signal-lang-*.ts` the 10 packs
  (BUILTIN_SIGNAL_LANGUAGES). CJK packs use substring regexes (no \b —
  CJK has no word boundaries); Latin/Cyrillic packs are \b-anchored.
The reference splits sync (en/de statically imported) from async
(dynamic import) loading; Python imports are eager so `load()` handles
every builtin and `load_sync` is an alias kept for parity of surface.
"""

from __future__ import annotations

import re
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Pattern, Sequence

BUILTIN_SIGNAL_LANGUAGES = ("en", "de", "es", "fr", "it", "pt", "ru", "ja", "ko", "zh")
# reference's statically-imported subset (index.ts loadSync)
SYNC_SIGNAL_LANGUAGES = ("en", "de")
_CJK = {"ja", "ko", "zh"}

CATEGORIES = {
    "correction": ("indicators", "shortNegatives"),
    "question": ("indicators",),
    "dissatisfaction": ("indicators", "satisfactionOverrides", "resolutionIndicators"),
    "completion": ("claims",),
    "systemState": ("claims", "opinionExclusions"),
}


@dataclass
class SignalLanguagePack:
    code: str
    name: str
    name_en: str
    correction: Dict[str, List[Pattern]] = field(default_factory=dict)
    question: Dict[str, List[Pattern]] = field(default_factory=dict)
    dissatisfaction: Dict[str, List[Pattern]] = field(default_factory=dict)
    completion: Dict[str, List[Pattern]] = field(default_factory=dict)
    system_state: Dict[str, List[Pattern]] = field(default_factory=dict)

    def category(self, name: str) -> Dict[str, List[Pattern]]:
        return getattr(self, "system_state" if name == "systemState" else name) or {}


def _rx(phrase: str, cjk: bool) -> Pattern:
    """Compile one phrase: \b-anchored where the edge is a word char
    (Latin/Cyrillic), plain substring for CJK (no word boundaries)."""
    if phrase.startswith("^") or phrase.endswith("$") or "\\" in phrase or "(" in phrase:
        return re.compile(phrase, re.I)  # already a regex
    esc = re.escape(phrase)
    if not cjk:
        if re.match(r"\w", phrase):
            esc = r"\b" + esc
        if re.search(r"\w$", phrase):
            esc = esc + r"\b"
    return re.compile(esc, re.I)


# phrase tables per language; compiled into packs below. Sources:
# signal-lang-<code>.ts category tables in the reference.
_SRC: Dict[str, Dict[str, Dict[str, Sequence[str]]]] = {
    "en": {
        "correction": {
            "indicators": ["no,", "that's wrong", "wrong", "not what i asked",
                           "incorrect", "you misunderstood", "try again",
                           "that is not right"],
            "shortNegatives": [r"^\s*(?:no|nope|nah)\s*[.!]?\s*$"],
        },
        "question": {"indicators": ["is that right", "are you sure", "really?"]},
        "dissatisfaction": {
            "indicators": ["useless", "this is not helpful", "you keep failing",
                           "frustrating", "give up", "terrible", "forget it"],
            "satisfactionOverrides": ["thanks", "thank you", "great", "perfect"],
            "resolutionIndicators": ["sorry", "my apologies", "apologize"],
        },
        "completion": {"claims": ["done", "completed", "finished", "fixed",
                                  "all set", "it works now"]},
        "systemState": {
            "claims": [r"\bthere (?:are|is)\s+\d+\s+error", r"\ball tests pass\b",
                       r"\bthe (?:server|service) is (?:up|running|down)\b"],
            "opinionExclusions": ["i think", "i believe", "maybe", "probably",
                                  "perhaps"],
        },
    },
    "de": {
        "correction": {
            "indicators": ["nein,", "das ist falsch", "falsch",
                           "nicht was ich meinte", "falsch verstanden",
                           "nochmal", "stimmt nicht"],
            "shortNegatives": [r"^\s*nein\s*[.!]?\s*$"],
        },
        "question": {"indicators": ["stimmt das", "bist du sicher", "wirklich?"]},
        "dissatisfaction": {
            "indicators": ["unbrauchbar", "das hilft nicht",
                           "du scheiterst ständig", "frustrierend",
                           "ich geb auf", "vergiss es"],
            "satisfactionOverrides": ["danke", "super", "perfekt"],
            "resolutionIndicators": ["entschuldigung", "tut mir leid"],
        },
        "completion": {"claims": ["fertig", "erledigt", "abgeschlossen",
                                  "funktioniert jetzt"]},
        "systemState": {
            "claims": [r"\bes gibt\s+\d+\s+fehler", r"\balle tests bestehen\b"],
            "opinionExclusions": ["ich glaube", "ich denke", "vielleicht",
                                  "wahrscheinlich"],
        },
    },
    "es": {
        "correction": {
            "indicators": ["no,", "eso está mal", "no es lo que pedí",
                           "incorrecto", "inténtalo de nuevo"],
            "shortNegatives": [r"^\s*no\s*[.!]?\s*$"],
        },
        "question": {"indicators": ["¿estás seguro", "es correcto"]},
        "dissatisfaction": {
            "indicators": ["inútil", "esto no ayuda", "sigues fallando",
                           "frustrante", "olvídalo", "déjalo"],
            "satisfactionOverrides": ["gracias", "perfecto", "genial"],
            "resolutionIndicators": ["lo siento", "perdón", "disculpa"],
        },
        "completion": {"claims": ["completado", "terminado", "arreglado",
                                  "listo", "ya funciona"]},
        "systemState": {
            "claims": [r"\bhay\s+\d+\s+errores?"],
            "opinionExclusions": ["creo que", "quizás", "tal vez"],
        },
    },
    "fr": {
        "correction": {
            "indicators": ["non,", "c'est faux", "pas ce que j'ai demandé",
                           "incorrect", "réessaie"],
            "shortNegatives": [r"^\s*non\s*[.!]?\s*$"],
        },
        "question": {"indicators": ["tu es sûr", "c'est correct"]},
        "dissatisfaction": {
            "indicators": ["inutile", "ça n'aide pas", "tu échoues encore",
                           "frustrant", "laisse tomber"],
            "satisfactionOverrides": ["merci", "parfait", "génial"],
            "resolutionIndicators": ["désolé", "pardon", "excuse"],
        },
        "completion": {"claims": ["terminé", "fini", "corrigé",
                                  "ça marche maintenant"]},
        "systemState": {
            "claims": [r"\bil y a\s+\d+\s+erreurs?"],
            "opinionExclusions": ["je crois", "je pense", "peut-être"],
        },
    },
    "it": {
        "correction": {
            "indicators": ["no,", "è sbagliato", "non è quello che ho chiesto",
                           "riprova"],
            "shortNegatives": [r"^\s*no\s*[.!]?\s*$"],
        },
        "question": {"indicators": ["sei sicuro", "è corretto"]},
        "dissatisfaction": {
            "indicators": ["inutile", "non aiuta", "continui a fallire",
                           "frustrante", "lascia perdere"],
            "satisfactionOverrides": ["grazie", "perfetto", "ottimo"],
            "resolutionIndicators": ["scusa", "mi dispiace"],
        },
        "completion": {"claims": ["completato", "finito", "risolto",
                                  "ora funziona"]},
        "systemState": {
            "claims": [r"\bci sono\s+\d+\s+errori"],
            "opinionExclusions": ["credo che", "forse"],
        },
    },
    "pt": {
        "correction": {
            "indicators": ["não,", "está errado", "não foi o que pedi",
                           "tente novamente"],
            "shortNegatives": [r"^\s*não\s*[.!]?\s*$"],
        },
        "question": {"indicators": ["tem certeza", "está correto"]},
        "dissatisfaction": {
            "indicators": ["inútil", "isso não ajuda", "você continua falhando",
                           "frustrante", "esquece"],
            "satisfactionOverrides": ["obrigado", "obrigada", "perfeito"],
            "resolutionIndicators": ["desculpa", "sinto muito"],
        },
        "completion": {"claims": ["concluído", "pronto", "consertado",
                                  "funciona agora"]},
        "systemState": {
            "claims": [r"\bhá\s+\d+\s+erros?"],
            "opinionExclusions": ["acho que", "talvez"],
        },
    },
    "ru": {
        "correction": {
            "indicators": ["нет,", "это неверно", "не то, что я просил",
                           "попробуй снова", "неправильно"],
            "shortNegatives": [r"^\s*нет\s*[.!]?\s*$"],
        },
        "question": {"indicators": ["ты уверен", "это точно"]},
        "dissatisfaction": {
            "indicators": ["бесполезно", "это не помогает", "ты снова ошибся",
                           "разочарование", "забудь"],
            "satisfactionOverrides": ["спасибо", "отлично", "идеально"],
            "resolutionIndicators": ["извини", "прости"],
        },
        "completion": {"claims": ["готово", "завершено", "исправлено",
                                  "теперь работает"]},
        "systemState": {
            "claims": [r"есть\s+\d+\s+ошиб"],
            "opinionExclusions": ["думаю", "наверное", "возможно"],
        },
    },
    "ja": {
        "correction": {
            "indicators": ["違います", "違う", "間違っています", "間違い",
                           "そうじゃない", "やり直して"],
            "shortNegatives": [r"^\s*いいえ\s*[。!]?\s*$"],
        },
        "question": {"indicators": ["本当に", "確かですか"]},
        "dissatisfaction": {
            "indicators": ["役に立たない", "助けになりません", "また失敗",
                           "もういい", "あきらめ"],
            "satisfactionOverrides": ["ありがとう", "完璧", "素晴らしい"],
            "resolutionIndicators": ["すみません", "ごめん", "申し訳"],
        },
        "completion": {"claims": ["完了", "完成した", "修正済み", "できました",
                                  "動きます"]},
        "systemState": {
            "claims": [r"エラーが\d+件"],
            "opinionExclusions": ["たぶん", "と思う", "かもしれない"],
        },
    },
    "ko": {
        "correction": {
            "indicators": ["아니요", "틀렸", "그게 아니에요", "다시 해보세요",
                           "잘못됐"],
            "shortNegatives": [r"^\s*아니\s*[.!]?\s*$"],
        },
        "question": {"indicators": ["확실해", "맞나요"]},
        "dissatisfaction": {
            "indicators": ["쓸모없", "도움이 안", "또 실패", "됐어", "포기"],
            "satisfactionOverrides": ["감사합니다", "고마워", "완벽"],
            "resolutionIndicators": ["죄송", "미안"],
        },
        "completion": {"claims": ["완료", "끝났", "수정됨", "됐습니다",
                                  "작동합니다"]},
        "systemState": {
            "claims": [r"오류가?\s*\d+개"],
            "opinionExclusions": ["아마", "같아요", "일지도"],
        },
    },
    "zh": {
        "correction": {
            "indicators": ["不对", "错了", "不是我要的", "再试一次", "搞错了"],
            "shortNegatives": [r"^\s*不\s*[。!]?\s*$"],
        },
        "question": {"indicators": ["确定吗", "真的吗"]},
        "dissatisfaction": {
            "indicators": ["没用", "帮不上忙", "又失败了", "算了", "放弃"],
            "satisfactionOverrides": ["谢谢", "完美", "太好了"],
            "resolutionIndicators": ["抱歉", "对不起"],
        },
        "completion": {"claims": ["完成", "搞定", "修好", "好了", "可以用了"]},
        "systemState": {
            "claims": [r"有\d+个错误"],
            "opinionExclusions": ["可能", "大概", "我觉得"],
        },
    },
}

_NAMES = {
    "en": ("English", "English"), "de": ("Deutsch", "German"),
    "es": ("Español", "Spanish"), "fr": ("Français", "French"),
    "it": ("Italiano", "Italian"), "pt": ("Português", "Portuguese"),
    "ru": ("Русский", "Russian"), "ja": ("日本語", "Japanese"),
    "ko": ("한국어", "Korean"), "zh": ("中文", "Chinese"),
}

# universal patterns merged into every getPatterns() result (index.ts):
# question marks, completion/ satisfaction emoji are language-free.
UNIVERSAL = {
    "question": {"indicators": [re.compile(r"[?？]\s*$")]},
    "completion": {"claims": [re.compile("✅"), re.compile("✓"), re.compile("☑")]},
    "dissatisfaction": {"satisfactionOverrides": [re.compile("👍"), re.compile("🙏"),
                                                  re.compile("❤️")]},
}


def builtin_pack(code: str) -> Optional[SignalLanguagePack]:
    src = _SRC.get(code)
    if src is None:
        return None
    cjk = code in _CJK
    cats = {
        cat: {key: [_rx(p, cjk) for p in src.get(cat, {}).get(key, [])]
              for key in keys}
        for cat, keys in CATEGORIES.items()
    }
    name, name_en = _NAMES[code]
    return SignalLanguagePack(
        code=code, name=name, name_en=name_en,
        correction=cats["correction"], question=cats["question"],
        dissatisfaction=cats["dissatisfaction"], completion=cats["completion"],
        system_state=cats["systemState"],
    )


class SignalPatternRegistry:
    """Loads language packs and serves merged+cached category patterns."""

    def __init__(self) -> None:
        self._packs: Dict[str, SignalLanguagePack] = {}
        self._cache: Optional[Dict[str, Dict[str, List[Pattern]]]] = None

    def load(self, codes: Sequence[str]) -> List[str]:
        """Load builtin packs by code (unknown codes ignored); returns the
        codes actually loaded this call. Invalidate the merge cache."""
        loaded = []
        for code in codes:
            pack = builtin_pack(code)
            if pack is not None:
                self._packs[code] = pack
                loaded.append(code)
        self._cache = None
        return loaded

    def load_sync(self, codes: Sequence[str]) -> List[str]:
        """Parity alias: the reference's loadSync only honors the
        statically-imported subset (en/de); other codes are skipped."""
        return self.load([c for c in codes if c in SYNC_SIGNAL_LANGUAGES])

    def register_pack(self, pack: SignalLanguagePack) -> None:
        """Add or REPLACE a pack at runtime (registerSignalLanguagePack)."""
        self._packs[pack.code] = pack
        self._cache = None

    def loaded_languages(self) -> List[str]:
        return list(self._packs.keys())

    def get_patterns(self) -> Dict[str, Dict[str, List[Pattern]]]:
        if self._cache is not None:
            return self._cache
        merged: Dict[str, Dict[str, List[Pattern]]] = {
            cat: {key: [] for key in keys} for cat, keys in CATEGORIES.items()
        }
        for pack in self._packs.values():
            for cat, keys in CATEGORIES.items():
                src = pack.category(cat)
                for key in keys:
                    merged[cat][key].extend(src.get(key, []))
        for cat, keys in UNIVERSAL.items():
            for key, rxs in keys.items():
                merged[cat][key].extend(rxs)
        self._cache = merged
        return merged


_default: Optional[SignalPatternRegistry] = None


def default_registry() -> SignalPatternRegistry:
    """Process-wide registry preloaded with all 10 builtin languages."""
    global _default
    if _default is None:
        _default = SignalPatternRegistry()
        _default.load(BUILTIN_SIGNAL_LANGUAGES)
    return _default
