"""Redaction pattern registry: ~17 pre-compiled builtin patterns.

Parity target: governance `src/redaction/registry.ts:31-150` — OpenAI /
Anthropic / AWS / Google keys, GitHub/GitLab tokens, private-key headers,
Bearer/Basic auth, key=value credentials, email, phone, SSN, credit card,
IBAN. Category evaluation order credential -> financial -> pii -> custom
(`:17-22`); builtins non-overridable; overlap resolution longest-match then
category priority (`:286-320`); custom patterns get a ReDoS sanity check.

The same builtin set is compiled to the GPU multi-pattern DFA
(`ops/dfa.py` -> `csrc/pattern_scan.hip`); this module is the host-side
semantic reference the kernel is tested against.
"""

from __future__ import annotations

import re
import time
from dataclasses import dataclass
from typing import Any, Dict, List, Optional

from ...core.api import PluginLogger, NullLogger

CATEGORY_ORDER = ("credential", "financial", "pii", "custom")


@dataclass(frozen=True)
class RedactionPattern:
    id: str
    category: str
    regex: "re.Pattern[str]"
    replacement_type: str
    builtin: bool = True


def _p(pid: str, category: str, rx: str, rtype: str, flags: int = 0) -> RedactionPattern:
    return RedactionPattern(pid, category, re.compile(rx, flags), rtype, True)


BUILTIN_PATTERNS: List[RedactionPattern] = [
    _p("openai-api-key", "credential", r"sk-[a-zA-Z0-9]{20,}", "api_key"),
    _p("anthropic-api-key", "credential", r"sk-ant-[a-zA-Z0-9-]{80,}", "api_key"),
    _p("aws-key", "credential", r"(?<![A-Z0-9])AKIA[0-9A-Z]{16}(?![A-Z0-9])", "api_key"),
    _p("generic-api-key", "credential", r"sk-[a-zA-Z0-9_-]{20,}", "api_key"),
    _p("google-api-key", "credential", r"AIza[0-9A-Za-z_-]{35}", "api_key"),
    _p("github-pat", "credential", r"ghp_[a-zA-Z0-9]{36}", "token"),
    _p("github-server-token", "credential", r"ghs_[a-zA-Z0-9]{36}", "token"),
    _p("gitlab-pat", "credential", r"glpat-[a-zA-Z0-9_-]{20,}", "token"),
    _p("private-key-header", "credential", r"-----BEGIN (?:RSA |EC |OPENSSH )?PRIVATE KEY-----", "private_key"),
    _p("bearer-token", "credential", r"Bearer [a-zA-Z0-9_./-]{20,}", "bearer"),
    _p("basic-auth", "credential", r"Basic [A-Za-z0-9+/]{16,}={0,2}", "basic_auth"),
    _p(
        "key-value-credential",
        "credential",
        r"(?:password|passwd|pwd|secret|token|api_key|apikey)\s*[:=]\s*['\"]?[^\s'\"]{8,64}",
        "credential",
        re.IGNORECASE,
    ),
    _p("email-address", "pii", r"\b[a-zA-Z0-9._%+-]+@[a-zA-Z0-9.-]+\.[a-zA-Z]{2,}\b", "email"),
    _p("phone-number", "pii", r"(?<!\d)\+?[1-9]\d{6,14}(?!\d)", "phone"),
    _p("ssn-us", "pii", r"\b\d{3}-\d{2}-\d{4}\b", "ssn"),
    _p("credit-card", "financial", r"\b[45]\d{3}[\s-]?\d{4}[\s-]?\d{4}[\s-]?\d{4}\b", "credit_card"),
    _p("iban", "financial", r"\b[A-Z]{2}\d{2}\s?[A-Z0-9]{4}\s?(?:\d{4}\s?){2,7}\d{1,4}\b", "iban"),
]


@dataclass
class PatternMatch:
    pattern: RedactionPattern
    match: str
    start: int
    end: int


class PatternRegistry:
    def __init__(
        self,
        enabled_categories: Optional[List[str]] = None,
        custom_patterns: Optional[List[Dict[str, Any]]] = None,
        logger: Optional[PluginLogger] = None,
    ):
        self.logger = logger or NullLogger()
        enabled = set(enabled_categories if enabled_categories is not None else CATEGORY_ORDER)
        self.patterns: List[RedactionPattern] = [p for p in BUILTIN_PATTERNS if p.category in enabled]
        for cp in custom_patterns or []:
            compiled = self._compile_custom(cp)
            if compiled:
                self.patterns.append(compiled)

    def get_patterns(self) -> List[RedactionPattern]:
        return list(self.patterns)

    def get_by_category(self, category: str) -> List[RedactionPattern]:
        return [p for p in self.patterns if p.category == category]

    @staticmethod
    def is_credential_category(category: str) -> bool:
        return category == "credential"

    def _compile_custom(self, cfg: Dict[str, Any]) -> Optional[RedactionPattern]:
        try:
            rx = re.compile(cfg["regex"])
        except (re.error, KeyError) as exc:
            self.logger.warn("[redaction] Custom pattern %r failed to compile: %s", cfg.get("name"), exc)
            return None
        # ReDoS sanity check against adversarial input (registry.ts:250-262)
        start = time.perf_counter()
        rx.search("a" * 1000)
        if (time.perf_counter() - start) * 1000 > 10:
            self.logger.warn("[redaction] Custom pattern %r rejected: ReDoS risk", cfg.get("name"))
            return None
        return RedactionPattern(
            id=f"custom-{cfg.get('name')}",
            category=cfg.get("category", "custom"),
            regex=rx,
            replacement_type=str(cfg.get("name")),
            builtin=False,
        )

    def find_matches(self, text: str) -> List[PatternMatch]:
        """All matches, overlaps resolved longest-first then category
        priority (registry.ts:210-240, 286-320)."""
        all_matches: List[PatternMatch] = []
        for category in CATEGORY_ORDER:
            for pattern in self.patterns:
                if pattern.category != category:
                    continue
                for m in pattern.regex.finditer(text):
                    if m.group(0):
                        all_matches.append(PatternMatch(pattern, m.group(0), m.start(), m.end()))
        return self._resolve_overlaps(all_matches)

    @staticmethod
    def _resolve_overlaps(matches: List[PatternMatch]) -> List[PatternMatch]:
        if len(matches) <= 1:
            return matches
        matches.sort(
            key=lambda m: (m.start, -(m.end - m.start), CATEGORY_ORDER.index(m.pattern.category))
        )
        resolved: List[PatternMatch] = []
        last_end = -1
        for m in matches:
            if m.start >= last_end:
                resolved.append(m)
                last_end = m.end
        return resolved


def get_builtin_patterns() -> List[RedactionPattern]:
    return list(BUILTIN_PATTERNS)
