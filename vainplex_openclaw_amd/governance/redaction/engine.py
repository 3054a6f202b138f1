"""Redaction scanning engine: recursive deep-scan + placeholder replace.

Parity target: governance `src/redaction/engine.ts` — deep object
traversal (max depth 20), JSON-within-string detection with recursive
scanning, circular-reference protection, string-level pattern replace via
the registry + vault.
"""

from __future__ import annotations

import json
import time
from typing import Any, Dict, Set

from .registry import PatternRegistry
from .vault import RedactionVault

MAX_DEPTH = 20
MAX_JSON_PARSE_LENGTH = 1_000_000


class RedactionEngine:
    def __init__(self, registry: PatternRegistry, vault: RedactionVault):
        self.registry = registry
        self.vault = vault

    def scan(self, value: Any) -> Dict[str, Any]:
        start = time.perf_counter()
        state = {"count": 0, "categories": set()}  # type: Dict[str, Any]
        output = self._scan_value(value, set(), 0, state)
        return {
            "output": output,
            "redactionCount": state["count"],
            "categories": state["categories"],
            "elapsedMs": (time.perf_counter() - start) * 1000,
        }

    def scan_string(self, text: str) -> Dict[str, Any]:
        state = {"count": 0, "categories": set()}  # type: Dict[str, Any]
        output = self._redact_string(text, state)
        return {"output": output, "redactionCount": state["count"], "categories": state["categories"]}

    def _scan_value(self, value: Any, seen: Set[int], depth: int, state: Dict[str, Any]) -> Any:
        if depth > MAX_DEPTH or value is None:
            return value
        if isinstance(value, str):
            return self._scan_string_value(value, seen, depth, state)
        if isinstance(value, dict):
            oid = id(value)
            if oid in seen:
                return value
            seen.add(oid)
            return {k: self._scan_value(v, seen, depth + 1, state) for k, v in value.items()}
        if isinstance(value, (list, tuple)):
            oid = id(value)
            if oid in seen:
                return value
            seen.add(oid)
            out = [self._scan_value(v, seen, depth + 1, state) for v in value]
            return out if isinstance(value, list) else tuple(out)
        return value

    def _scan_string_value(self, text: str, seen: Set[int], depth: int, state: Dict[str, Any]) -> str:
        # JSON-within-string: parse, scan structure, re-serialize
        stripped = text.strip()
        if (
            len(text) <= MAX_JSON_PARSE_LENGTH
            and len(stripped) > 1
            and stripped[0] in "{["
            and stripped[-1] in "}]"
        ):
            try:
                parsed = json.loads(text)
            except (json.JSONDecodeError, RecursionError):
                parsed = None
            if isinstance(parsed, (dict, list)):
                before = state["count"]
                scanned = self._scan_value(parsed, seen, depth + 1, state)
                if state["count"] > before:
                    return json.dumps(scanned, ensure_ascii=False)
                return text
        return self._redact_string(text, state)

    def _redact_string(self, text: str, state: Dict[str, Any]) -> str:
        matches = self.registry.find_matches(text)
        if not matches:
            return text
        out = []
        last = 0
        for m in matches:
            out.append(text[last : m.start])
            out.append(self.vault.store(m.match, m.pattern.category))
            state["count"] += 1
            state["categories"].add(m.pattern.category)
            last = m.end
        out.append(text[last:])
        return "".join(out)
