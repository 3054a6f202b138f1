"""Knowledge-engine extraction regexes: 9 pattern families.

Parity target: reference `openclaw-knowledge-engine/src/patterns.ts` —
email, url, iso/common/german/english dates, proper_noun with a 60+ word
exclusion list, product_name (versions / Roman numerals / camelCase),
organization_suffix (Inc/LLC/Corp/GmbH/AG/Ltd).

The reference wraps its factories in a Proxy that returns a fresh RegExp
per property access to avoid JS `/g` lastIndex state-bleed
(`patterns.ts:70-90`); Python's `re` module is stateless across
`finditer` calls, so plain precompiled patterns are safe here.

The same families are compiled to a DFA for the batched GPU scan path
(`ops/pattern_sets.py` ENTITY family); this module is the per-message
host path with full-fidelity semantics.
"""

from __future__ import annotations

import re
from typing import Dict, Iterator, Tuple

# Words that look like proper nouns at sentence start but are not
# (patterns.ts EXCLUDED_WORDS).
EXCLUDED_WORDS = [
    "A", "An", "The", "Hello", "My", "This", "Contact", "He", "She",
    "It", "We", "They", "I", "You", "His", "Her", "Our", "Your",
    "Their", "Its", "That", "These", "Those", "What", "Which", "Who",
    "How", "When", "Where", "Why", "But", "And", "Or", "So", "Not",
    "No", "Yes", "Also", "Just", "For", "From", "With", "About",
    "After", "Before", "Between", "During", "Into", "Through",
    "Event", "Talk", "Project", "Multiple", "German",
    "Am", "Are", "Is", "Was", "Were", "Has", "Have",
    "Had", "Do", "Does", "Did", "Will", "Would", "Could", "Should",
    "May", "Might", "Must", "Can", "Shall", "If", "Then",
]

_EXCL = "|".join(w + r"\b" for w in EXCLUDED_WORDS)

# Capitalized word: O'Malley, McDonald's, acronyms like USS (patterns.ts CAP)
_CAP = r"(?:[A-Z][a-z']*(?:[A-Z][a-z']+)*|[A-Z]{2,})"

_DE_MONTHS = (
    "Januar|Februar|März|Mar|April|Mai|Juni|Juli|August|September|Oktober|"
    "November|Dezember"
)
_EN_MONTHS = (
    "January|February|March|April|May|June|July|August|September|October|"
    "November|December"
)

PATTERNS: Dict[str, re.Pattern] = {
    "email": re.compile(r"\b[a-zA-Z0-9._%+-]+@[a-zA-Z0-9.-]+\.[a-zA-Z]{2,}\b"),
    "url": re.compile(r"\bhttps?://[^\s/$.?#].[^\s]*\b"),
    "iso_date": re.compile(r"\b\d{4}-\d{2}-\d{2}(T\d{2}:\d{2}:\d{2}(\.\d+)?Z?)?\b"),
    "common_date": re.compile(r"\b(?:\d{1,2}/\d{1,2}/\d{2,4})|(?:\d{1,2}\.\d{1,2}\.\d{2,4})\b"),
    "german_date": re.compile(
        r"\b\d{1,2}\.\s(?:%s)\s+\d{4}\b" % _DE_MONTHS, re.IGNORECASE
    ),
    "english_date": re.compile(
        r"\b(?:%s)\s+\d{1,2}(?:st|nd|rd|th)?,\s+\d{4}\b" % _EN_MONTHS, re.IGNORECASE
    ),
    "proper_noun": re.compile(
        r"\b(?!%s)%s(?:(?:-|\s)(?!%s)%s)*\b" % (_EXCL, _CAP, _EXCL, _CAP)
    ),
    "product_name": re.compile(
        r"\b(?:(?!%s)[A-Z][a-zA-Z0-9]{2,}(?:\s[a-zA-Z]+)*\s[IVXLCDM]+"
        r"|[a-zA-Z][a-zA-Z0-9-]{2,}[\s-]v?\d+(?:\.\d+)?"
        r"|[a-zA-Z][a-zA-Z0-9]+[IVXLCDM]+)\b" % _EXCL
    ),
    "organization_suffix": re.compile(
        r"\b(?:[A-Z][A-Za-z0-9]+(?:\s[A-Z][A-Za-z0-9]+)*),?\s?"
        r"(?:Inc\.|LLC|Corp\.|GmbH|AG|Ltd\.)"
    ),
}

# Pattern family -> entity type (entity-extractor.ts PATTERN_TYPE_MAP)
PATTERN_TYPE_MAP: Dict[str, str] = {
    "email": "email",
    "url": "url",
    "iso_date": "date",
    "common_date": "date",
    "german_date": "date",
    "english_date": "date",
    "proper_noun": "unknown",
    "product_name": "product",
    "organization_suffix": "organization",
}


def iter_matches(text: str) -> Iterator[Tuple[str, str, str]]:
    """Yield (family, entity_type, matched_text) over all families."""
    for key, pat in PATTERNS.items():
        etype = PATTERN_TYPE_MAP.get(key, "unknown")
        for m in pat.finditer(text):
            value = m.group(0).strip()
            if value:
                yield key, etype, value
