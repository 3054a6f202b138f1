"""Redaction vault: placeholder <-> secret mapping, in-memory only.

Parity target: governance `src/redaction/vault.ts` — SHA-256 placeholder
`[REDACTED:<category>:<hash8|hash12>]` (`:33-38`), hash8 -> hash12 on
collision (`:86-104`), 1 h TTL, 5-min cleanup; never persisted, never
logged (`:8-12`).
"""

from __future__ import annotations

import hashlib
import re
import time
from typing import Dict, List, Optional

DEFAULT_EXPIRY_SECONDS = 3600

PLACEHOLDER_RX = re.compile(r"\[REDACTED:(?:credential|pii|financial|custom):([a-f0-9]{8,12})\]")


def _sha256(data: str) -> str:
    return hashlib.sha256(data.encode("utf-8")).hexdigest()


def format_placeholder(category: str, hash_slice: str) -> str:
    return f"[REDACTED:{category}:{hash_slice}]"


class RedactionVault:
    def __init__(self, expiry_seconds: float = DEFAULT_EXPIRY_SECONDS, clock=time.time):
        self.expiry_seconds = expiry_seconds
        self.clock = clock
        self._entries: Dict[str, Dict] = {}  # full hash -> entry
        self._hash_index: Dict[str, List[str]] = {}  # hash8 -> [full hashes]
        self._by_placeholder: Dict[str, str] = {}  # placeholder -> full hash

    def store(self, original: str, category: str) -> str:
        full = _sha256(original)
        h8 = full[:8]
        now_ms = self.clock() * 1000

        existing = self._entries.get(full)
        if existing and existing["expiresAt"] > now_ms:
            return existing["placeholder"]

        slice_ = h8
        for other in self._hash_index.get(h8, []):
            entry = self._entries.get(other)
            if entry and other != full and entry["expiresAt"] > now_ms:
                slice_ = full[:12]
                break

        placeholder = format_placeholder(category, slice_)
        self._entries[full] = {
            "original": original,
            "category": category,
            "placeholder": placeholder,
            "hash": full,
            "createdAt": now_ms,
            "expiresAt": now_ms + self.expiry_seconds * 1000,
        }
        bucket = self._hash_index.setdefault(h8, [])
        if full not in bucket:
            bucket.append(full)
        self._by_placeholder[placeholder] = full
        return placeholder

    def resolve(self, text: str) -> str:
        """Replace placeholders with originals (for outbound restore)."""
        now_ms = self.clock() * 1000

        def sub(m: "re.Match[str]") -> str:
            full = self._by_placeholder.get(m.group(0))
            entry = self._entries.get(full) if full else None
            if entry and entry["expiresAt"] > now_ms:
                return entry["original"]
            return m.group(0)

        return PLACEHOLDER_RX.sub(sub, text)

    def lookup(self, placeholder: str) -> Optional[str]:
        now_ms = self.clock() * 1000
        full = self._by_placeholder.get(placeholder)
        entry = self._entries.get(full) if full else None
        if entry and entry["expiresAt"] > now_ms:
            return entry["original"]
        return None

    def evict_expired(self) -> int:
        now_ms = self.clock() * 1000
        expired = [h for h, e in self._entries.items() if e["expiresAt"] <= now_ms]
        for h in expired:
            e = self._entries.pop(h)
            bucket = self._hash_index.get(e["hash"][:8])
            if bucket and h in bucket:
                bucket.remove(h)
            self._by_placeholder.pop(e["placeholder"], None)
        return len(expired)

    @property
    def size(self) -> int:
        """Non-expired entries only (vault.ts size semantics)."""
        now_ms = self.clock() * 1000
        return sum(1 for e in self._entries.values() if e["expiresAt"] > now_ms)

    def clear(self) -> None:
        self._entries.clear()
        self._hash_index.clear()
        self._by_placeholder.clear()
