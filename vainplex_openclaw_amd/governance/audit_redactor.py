"""Audit-context redaction before write.

Parity target: governance `src/audit-redactor.ts` — applies configured
regex redaction patterns to every string field of the audit context
(recursively), replacing matches with `[REDACTED]`.
"""

from __future__ import annotations

import re
from typing import Any, Callable, Dict, List, Optional

DEFAULT_PATTERNS = [
    r"(?i)(api[_-]?key|token|secret|password)\s*[=:]\s*\S+",
    r"sk-[A-Za-z0-9_\-]{16,}",
    r"Bearer\s+[A-Za-z0-9\-._~+/]+=*",
]

REPLACEMENT = "[REDACTED]"


def create_redactor(patterns: Optional[List[str]] = None) -> Callable[[Dict[str, Any]], Dict[str, Any]]:
    compiled = []
    for p in patterns if patterns is not None else DEFAULT_PATTERNS:
        try:
            compiled.append(re.compile(p))
        except re.error:
            continue

    def redact_value(v: Any) -> Any:
        if isinstance(v, str):
            for rx in compiled:
                v = rx.sub(REPLACEMENT, v)
            return v
        if isinstance(v, dict):
            return {k: redact_value(x) for k, x in v.items()}
        if isinstance(v, list):
            return [redact_value(x) for x in v]
        return v

    def redact(ctx: Dict[str, Any]) -> Dict[str, Any]:
        return redact_value(dict(ctx))

    return redact
