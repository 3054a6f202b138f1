"""Pattern registry + shim API for cortex signal extraction.

Parity target: cortex `src/patterns/registry.ts` + the shim
`src/patterns.ts:38-82` — getPatterns (language or list or "both" = en+de,
"all" = every pack), detectMood (last match position wins), isNoiseTopic,
HIGH_IMPACT_KEYWORDS, and custom pattern packs (registry.ts loadSync:
extend appends to builtins, override replaces a family when it has at
least one VALID custom regex; invalid regexes are skipped silently;
custom blacklist words and high-impact keywords merge in).
"""

from __future__ import annotations

import re
from typing import Dict, Iterable, List, Optional, Set, Union

from .packs import PACKS, language_codes

Language = Union[str, List[str]]


def _resolve_codes(language: Language) -> List[str]:
    if language == "both":
        return ["en", "de"]
    if language == "all":
        return list(language_codes())
    if isinstance(language, str):
        return [language]
    if isinstance(language, (list, tuple)):
        return list(language)
    return ["en", "de"]


class PatternRegistry:
    def __init__(self, codes: Iterable[str], custom: Optional[Dict] = None):
        self.codes = [c for c in codes if c in PACKS]
        if not self.codes:
            self.codes = ["en"]
        custom = custom or {}
        mode = custom.get("mode", "extend")
        self._compiled: Dict[str, List["re.Pattern[str]"]] = {}
        for family in ("decision", "close", "wait", "topic"):
            pats = []
            for c in self.codes:
                for p in PACKS[c]["patterns"].get(family, []):
                    pats.append(re.compile(p, re.IGNORECASE))
            extra = []
            for p in custom.get(family, []) or []:
                if not isinstance(p, str):
                    continue
                try:
                    extra.append(re.compile(p, re.IGNORECASE))
                except re.error:
                    continue  # invalid custom regexes are skipped silently
            if extra and mode == "override":
                pats = extra
            else:
                pats.extend(extra)
            self._compiled[family] = pats
        self.moods: Dict[str, List["re.Pattern[str]"]] = {}
        for c in self.codes:
            for mood, p in PACKS[c]["moods"].items():
                self.moods.setdefault(mood, []).append(re.compile(p, re.IGNORECASE))
        self.blacklist: Set[str] = set()
        self.high_impact: List[str] = []
        self.noise_prefixes: Set[str] = set()
        for c in self.codes:
            self.blacklist.update(PACKS[c]["topic_blacklist"])
            for kw in PACKS[c]["high_impact"]:
                if kw not in self.high_impact:
                    self.high_impact.append(kw)
            self.noise_prefixes.update(PACKS[c]["noise_prefixes"])
        for w in custom.get("blacklist", []) or []:
            if isinstance(w, str):
                self.blacklist.add(w.lower())
        for kw in custom.get("keywords", []) or []:
            if isinstance(kw, str) and kw.lower() not in self.high_impact:
                self.high_impact.append(kw.lower())

    def get_patterns(self, family: str) -> List["re.Pattern[str]"]:
        return self._compiled.get(family, [])


_registries: Dict[str, PatternRegistry] = {}
_custom_global: Optional[Dict] = None


def set_custom_patterns(custom: Optional[Dict]) -> None:
    """Install plugin-level custom patterns (registry.ts loadSync is a
    module singleton in the reference too); clears the registry cache."""
    global _custom_global
    _custom_global = custom or None
    _registries.clear()


def get_registry(language: Language = "both",
                 custom: Optional[Dict] = None) -> PatternRegistry:
    if custom is None:
        custom = _custom_global
    if custom:
        return PatternRegistry(_resolve_codes(language), custom)
    key = ",".join(_resolve_codes(language))
    if key not in _registries:
        _registries[key] = PatternRegistry(_resolve_codes(language))
    return _registries[key]


def detect_mood(text: str, language: Language = "both") -> str:
    """Scan all mood patterns; the LAST match position wins
    (patterns.ts:47-66)."""
    if not text:
        return "neutral"
    reg = get_registry(language)
    last_mood, last_pos = "neutral", -1
    for mood, rxs in reg.moods.items():
        for rx in rxs:
            for m in rx.finditer(text):
                if m.start() > last_pos:
                    last_pos = m.start()
                    last_mood = mood
    return last_mood


def is_noise_topic(topic: str, language: Language = "both") -> bool:
    """Too short, blacklisted, pronoun-prefixed, multiline or overlong
    (patterns.ts isNoiseTopic)."""
    reg = get_registry(language)
    trimmed = topic.strip()
    if len(trimmed) < 4:
        return True
    words = trimmed.lower().split()
    if len(words) == 1 and words[0] in reg.blacklist:
        return True
    if words and all(w in reg.blacklist or len(w) < 3 for w in words):
        return True
    first = words[0] if words else ""
    if first in reg.noise_prefixes:
        return True
    if "\n" in trimmed or len(trimmed) > 60:
        return True
    return False


def high_impact_keywords(language: Language = "both") -> List[str]:
    return get_registry(language).high_impact


__all__ = [
    "set_custom_patterns",
    "PACKS",
    "language_codes",
    "PatternRegistry",
    "get_registry",
    "detect_mood",
    "is_noise_topic",
    "high_impact_keywords",
]
