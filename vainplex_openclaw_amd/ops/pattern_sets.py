"""GPU pattern families: the fixed pattern sets compiled to DFAs at startup.

Families and their reference sources:
- REDACTION: the 17 builtin credential/PII/financial patterns
  (governance/redaction/registry.py <- reference registry.ts:31-150).
- INJECTION: local prompt-injection / URL-threat heuristics replacing the
  reference's remote ShieldAPI scan ("208 patterns",
  openclaw-governance/README.md:196-228) — scanned locally on-GPU here.
- CLAIMS: the 5 claim-detector families (claim-detector.ts) reduced to
  hit-detection form.
- ENTITY: knowledge-engine extraction families (patterns.ts).

Each family is a list of (pattern, bit, name). Bits are per-family (each
family gets its own DFA + u64 hit mask).
"""

from __future__ import annotations

from typing import Dict, List, Set, Tuple

import hashlib
import os

import numpy as np

from .dfa import DFA, DFA_COMPILER_VERSION, MultiDFA, compile_multi

# -- redaction (bit order = registry order; category via bit ranges) --------
REDACTION_PATTERNS: List[Tuple[str, int, str]] = [
    (r"sk-[a-zA-Z0-9]{20}", 0, "openai-api-key"),
    (r"sk-ant-[a-zA-Z0-9-]{80}", 1, "anthropic-api-key"),
    (r"(?<![A-Z0-9])AKIA[0-9A-Z]{16}(?![A-Z0-9])", 2, "aws-key"),
    (r"sk-[a-zA-Z0-9_-]{20}", 3, "generic-api-key"),
    (r"AIza[0-9A-Za-z_-]{35}", 4, "google-api-key"),
    (r"ghp_[a-zA-Z0-9]{36}", 5, "github-pat"),
    (r"ghs_[a-zA-Z0-9]{36}", 6, "github-server-token"),
    (r"glpat-[a-zA-Z0-9_-]{20}", 7, "gitlab-pat"),
    (r"-----BEGIN (?:RSA |EC |OPENSSH )?PRIVATE KEY-----", 8, "private-key-header"),
    (r"Bearer [a-zA-Z0-9_./-]{20}", 9, "bearer-token"),
    (r"Basic [A-Za-z0-9+/]{16}", 10, "basic-auth"),
    (r"(?:password|passwd|pwd|secret|token|api_key|apikey)\s*[:=]\s*['\"]?[^\s'\"]{8}", 11, "key-value-credential"),
    (r"\b[a-zA-Z0-9._%+-]+@[a-zA-Z0-9.-]+\.[a-zA-Z]{2,}\b", 12, "email-address"),
    (r"(?<!\d)\+?[1-9]\d{6,14}(?!\d)", 13, "phone-number"),
    (r"\b\d{3}-\d{2}-\d{4}\b", 14, "ssn-us"),
    (r"\b[45]\d{3}[\s-]?\d{4}[\s-]?\d{4}[\s-]?\d{4}\b", 15, "credit-card"),
    (r"\b[A-Z]{2}\d{2}\s?[A-Z0-9]{4}\s?(?:\d{4}\s?){2,7}\d{1,4}\b", 16, "iban"),
]
REDACTION_IGNORECASE: Set[int] = {11}
REDACTION_CREDENTIAL_BITS = 0xFFF  # bits 0..11
REDACTION_PII_BITS = (1 << 12) | (1 << 13) | (1 << 14)
REDACTION_FINANCIAL_BITS = (1 << 15) | (1 << 16)

# -- prompt-injection / URL-threat heuristics (local ShieldAPI replacement) -
INJECTION_PATTERNS: List[Tuple[str, int, str]] = [
    (r"ignore (?:all |any )?(?:previous|prior|above) (?:instructions|prompts|rules)", 0, "override-instructions"),
    (r"disregard (?:all |any )?(?:previous|prior|your) (?:instructions|rules|guidelines)", 1, "disregard"),
    (r"forget (?:everything|all|your) (?:previous|prior|instructions)", 2, "forget"),
    (r"you are now (?:a|an|in) ", 3, "role-override"),
    (r"pretend (?:to be|you are)", 4, "pretend"),
    (r"act as (?:if|though|a|an) ", 5, "act-as"),
    (r"(?:reveal|print|show|repeat|output) (?:your|the) (?:system|initial|hidden) prompt", 6, "prompt-exfil"),
    (r"system prompt", 7, "system-prompt-mention"),
    (r"\bDAN\b|do anything now", 8, "dan-jailbreak"),
    (r"jailbreak", 9, "jailbreak"),
    (r"developer mode", 10, "developer-mode"),
    (r"no (?:restrictions|limitations|filters|guardrails)", 11, "no-restrictions"),
    (r"bypass (?:your |the )?(?:safety|security|filter|guardrail)", 12, "bypass-safety"),
    (r"<\s*/?system\s*>", 13, "fake-system-tag"),
    (r"\[\s*system\s*\]", 14, "fake-system-bracket"),
    (r"(?:^|\n)\s*(?:system|assistant)\s*:", 15, "role-spoof"),
    (r"BEGIN (?:ADMIN|SYSTEM|ROOT) ", 16, "fake-admin-block"),
    (r"exfiltrat", 17, "exfiltration"),
    (r"(?:send|post|upload) (?:all |the )?(?:conversation|chat|history|secrets|credentials)", 18, "data-exfil"),
    (r"curl\s+(?:-[a-zA-Z]+\s+)*https?://", 19, "curl-url"),
    (r"wget\s+https?://", 20, "wget-url"),
    (r"https?://(?:bit\.ly|tinyurl\.com|t\.co|goo\.gl|is\.gd|ow\.ly)/", 21, "url-shortener"),
    (r"https?://\d{1,3}\.\d{1,3}\.\d{1,3}\.\d{1,3}", 22, "raw-ip-url"),
    (r"data:text/html", 23, "data-url"),
    (r"javascript:", 24, "javascript-url"),
    (r"base64\s*(?:-d|--decode)", 25, "base64-decode"),
    (r"eval\s*\(", 26, "eval-call"),
    (r"exec\s*\(", 27, "exec-call"),
    (r"rm\s+-rf\s+[/~]", 28, "rm-rf"),
    (r"chmod\s+777", 29, "chmod-777"),
    (r"(?:nc|ncat|netcat)\s+(?:-[a-zA-Z]+\s+)*\d{1,3}\.\d{1,3}", 30, "netcat"),
    (r"/etc/passwd|/etc/shadow", 31, "sensitive-path"),
    (r"\.ssh/(?:id_rsa|authorized_keys)", 32, "ssh-keys"),
    (r"(?:drop|truncate)\s+table", 33, "sql-drop"),
    (r";\s*--\s|'\s*or\s+'?1'?\s*=\s*'?1", 34, "sql-injection"),
    (r"<script[\s>]", 35, "xss-script"),
    (r"onerror\s*=", 36, "xss-onerror"),
    (r"\{\{.*\}\}", 37, "template-injection"),
    (r"\$\(\s*(?:curl|wget|cat|id|whoami)", 38, "cmd-substitution"),
    (r"(?:^|\s)sudo\s+(?:su|bash|sh|rm|chmod|chown)", 39, "sudo-escalation"),
]
INJECTION_IGNORECASE: Set[int] = set(range(19)) | {25, 33, 34, 35, 36, 39}

# -- claim families (hit-detection form of claim-detector.ts) ---------------
CLAIMS_PATTERNS: List[Tuple[str, int, str]] = [
    (r"[\w][\w.:-]{0,60}\s+(?:is|are)\s+(?:running|stopped|online|offline|active|inactive|enabled|disabled|up|down|started|paused|healthy|unhealthy)\b", 0, "system_state"),
    (r"\bthe\s+(?:agent|service|server|container|process|pod|node|instance|database|cluster|daemon|plugin|module)\s+(?:named|called|known as|labelled|labeled)?\s*[\"`']?[\w][\w.:-]{0,60}", 1, "entity_name"),
    (r"[\w][\w.:-]{0,60}\s+(?:exists|is available|is present|is configured|is installed|is deployed|is registered)\b", 2, "existence_pos"),
    (r"[\w][\w.:-]{0,60}\s+(?:does(?:n't| not) exist|is not available|is not present|is not configured|is not installed|is not deployed|is not registered|doesn't exist)\b", 3, "existence_neg"),
    (r"\bthere\s+(?:is|are)\s+(?:no\s+)?[\w][\w.:-]{0,60}\b", 4, "there_is"),
    (r"[\w][\w.:-]{0,60}\s+(?:has|contains|uses|consumes|shows|reports)\s+\d[\d,.]*", 5, "metric"),
    (r"[\w][\w.:-]{0,60}\s+is\s+at\s+\d[\d,.]*\s*%", 6, "percentage"),
    (r"[\w][\w.:-]{0,60}\s+count\s+is\s+\d[\d,.]*\b", 7, "count"),
    (r"\bI\s+am\s+[\w][\w\s.:-]{0,60}?\s*[.,!?\n]", 8, "self_identity"),
    (r"\bmy\s+name\s+is\s+[\w][\w\s.:-]{0,60}?\s*[.,!?\n]", 9, "my_name"),
    (r"\bI\s+(?:have|possess|contain)\s+[\w][\w\s.:-]{0,60}?\s*[.,!?\n]", 10, "i_have"),
]
CLAIMS_IGNORECASE: Set[int] = set(range(11))

# -- entity families (knowledge-engine patterns.ts hit form) ----------------
ENTITY_PATTERNS: List[Tuple[str, int, str]] = [
    (r"\b[a-zA-Z0-9._%+-]+@[a-zA-Z0-9.-]+\.[a-zA-Z]{2,}\b", 0, "email"),
    (r"https?://[^\s<>\"]+", 1, "url"),
    (r"\b\d{4}-\d{2}-\d{2}\b", 2, "iso_date"),
    (r"\b\d{1,2}/\d{1,2}/\d{2,4}\b", 3, "common_date"),
    (r"\b\d{1,2}\.\d{1,2}\.\d{2,4}\b", 4, "german_date"),
    (r"\b(?:January|February|March|April|May|June|July|August|September|October|November|December)\s+\d{1,2}(?:st|nd|rd|th)?,?\s+\d{4}\b", 5, "english_date"),
    (r"\b[A-Z][a-z]+(?:\s+[A-Z][a-z]+){1,3}\b", 6, "proper_noun"),
    (r"\b[A-Z][a-zA-Z]*\s+(?:v?\d+(?:\.\d+)+|[IVX]{1,5}\b)", 7, "product_name"),
    # trailing \.? dropped: optional suffix is irrelevant for hit detection
    (r"\b[A-Z][\w&.-]*(?:\s+[A-Z][\w&.-]*){0,3}\s+(?:Inc|LLC|Ltd|GmbH|AG|Corp|Co|SA|SE|KG|PLC)\b", 8, "organization_suffix"),
]
ENTITY_IGNORECASE: Set[int] = set()


# -- cortex signal family (thread-tracker.ts:42-82 + patterns.ts:47-66) ----
# Built from the 10-language packs (cortex/patterns/packs.py): each signal
# bit ORs every language's patterns for that signal, so one GPU scan
# covers all enabled languages — matching the reference's extractSignals
# loop over language packs. Bits:
CORTEX_BIT_DECISION = 0
CORTEX_BIT_CLOSE = 1
CORTEX_BIT_WAIT = 2
CORTEX_BIT_TOPIC = 3
CORTEX_BIT_HIGH_IMPACT = 4   # any high-impact keyword (impact inference)
CORTEX_MOOD_BIT0 = 8         # bits 8..12: frustrated, excited, tense,
CORTEX_MOODS = ("frustrated", "excited", "tense", "productive", "exploratory")


def _build_cortex_patterns() -> List[Tuple[str, int, str]]:
    import re as _re

    from ..cortex.patterns.packs import PACKS

    out: List[Tuple[str, int, str]] = []
    sig_bits = {"decision": CORTEX_BIT_DECISION, "close": CORTEX_BIT_CLOSE,
                "wait": CORTEX_BIT_WAIT, "topic": CORTEX_BIT_TOPIC}
    for code, pack in PACKS.items():
        for sig, bit in sig_bits.items():
            for i, pat in enumerate(pack["patterns"].get(sig, ())):
                # hit-detection equivalence: a TRAILING bounded repeat
                # {m,n} matches somewhere iff {m} does (the first m copies
                # of a longer run) — truncating keeps the DFA small
                pat = pat.replace("{3,40})", "{3})")
                out.append((pat, bit, f"{code}-{sig}-{i}"))
        for kw in pack.get("high_impact", ()):
            out.append((_re.escape(kw), CORTEX_BIT_HIGH_IMPACT, f"{code}-impact-{kw}"))
        for mi, mood in enumerate(CORTEX_MOODS):
            pat = pack.get("moods", {}).get(mood)
            if pat:
                out.append((pat, CORTEX_MOOD_BIT0 + mi, f"{code}-mood-{mood}"))
    return out


_cache: Dict[str, MultiDFA] = {}

_FAMILIES = {
    "redaction": (REDACTION_PATTERNS, REDACTION_IGNORECASE),
    "injection": (INJECTION_PATTERNS, INJECTION_IGNORECASE),
    "claims": (CLAIMS_PATTERNS, CLAIMS_IGNORECASE),
    "entity": (ENTITY_PATTERNS, ENTITY_IGNORECASE),
}

# families compiled with unicode_word (\w, \b include bytes >= 0x80)
_UNICODE_WORD_FAMILIES = {"cortex"}


def _ensure_family_registered(name: str) -> None:
    if name == "cortex" and "cortex" not in _FAMILIES:
        pats = _build_cortex_patterns()
        _FAMILIES["cortex"] = (pats, set(p[1] for p in pats))  # all icase

_CACHE_DIR = os.path.join(os.path.dirname(os.path.abspath(__file__)), "_dfa_cache")


def _family_hash(name: str) -> str:
    patterns, icase = _FAMILIES[name]
    blob = repr((DFA_COMPILER_VERSION, patterns, sorted(icase),
                 name in _UNICODE_WORD_FAMILIES)).encode()
    return hashlib.sha256(blob).hexdigest()[:16]


def get_family(name: str) -> MultiDFA:
    """Compile (and cache, incl. on disk) one of the named families."""
    if name in _cache:
        return _cache[name]
    _ensure_family_registered(name)
    patterns, icase = _FAMILIES[name]
    cache_path = os.path.join(_CACHE_DIR, f"{name}-{_family_hash(name)}.npz")
    if os.path.isfile(cache_path):
        try:
            data = np.load(cache_path)
            n = int(data["n_dfas"])
            dfas = []
            for i in range(n):
                d = DFA(data[f"next_{i}"], data[f"accept_{i}"], data[f"class_{i}"])
                d.eof_mask = data[f"eof_{i}"]
                dfas.append(d)
            mdfa = MultiDFA(dfas)
            _cache[name] = mdfa
            return mdfa
        except Exception:
            pass
    mdfa = compile_multi([(p, bit) for p, bit, _ in patterns], ignore_case_ids=icase,
                         unicode_word=name in _UNICODE_WORD_FAMILIES)
    try:
        os.makedirs(_CACHE_DIR, exist_ok=True)
        payload = {"n_dfas": np.int32(len(mdfa.dfas))}
        for i, d in enumerate(mdfa.dfas):
            payload[f"next_{i}"] = d.next_state
            payload[f"accept_{i}"] = d.accept_mask
            payload[f"class_{i}"] = d.byte_class
            payload[f"eof_{i}"] = getattr(d, "eof_mask", np.zeros(d.n_states, dtype=np.uint64))
        np.savez(cache_path, **payload)
    except Exception:
        pass
    _cache[name] = mdfa
    return mdfa


def family_names() -> List[str]:
    return ["redaction", "injection", "claims", "entity", "cortex"]
