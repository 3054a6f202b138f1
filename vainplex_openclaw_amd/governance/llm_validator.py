"""Stage-3 LLM semantic hallucination check for external communications.

Parity target: reference `openclaw-governance/src/llm-validator.ts` —
djb2-hash keyed cache with 5-min TTL (`:37-49`), callLlm
dependency-injected (`:24-29`), default external channels
twitter/linkedin/email and external commands `bird tweet` (`:52-58`),
JSON verdict prompt over the fact registry, fail-open on LLM errors.
"""

from __future__ import annotations

import json
import time
from typing import Any, Callable, Dict, List, Optional

DEFAULT_EXTERNAL_CHANNELS = ("twitter", "linkedin", "email")
DEFAULT_EXTERNAL_COMMANDS = ("bird tweet",)
CACHE_TTL_S = 300.0  # 5-minute TTL (llm-validator.ts:37-49)


def djb2(text: str) -> int:
    h = 5381
    for ch in text:
        h = ((h * 33) + ord(ch)) & 0xFFFFFFFF
    return h


def is_external_comm(
    channel: Optional[str] = None,
    tool_name: Optional[str] = None,
    command: Optional[str] = None,
    channels=DEFAULT_EXTERNAL_CHANNELS,
    commands=DEFAULT_EXTERNAL_COMMANDS,
) -> bool:
    """External = one of the configured channels or command prefixes."""
    if channel and any(c in channel.lower() for c in channels):
        return True
    if tool_name and any(c in tool_name.lower() for c in channels):
        return True
    if command and any(command.lower().startswith(c) for c in commands):
        return True
    return False


class LlmValidator:
    def __init__(
        self,
        call_llm: Optional[Callable[[str], str]] = None,
        external_channels=DEFAULT_EXTERNAL_CHANNELS,
        external_commands=DEFAULT_EXTERNAL_COMMANDS,
        cache_ttl_s: float = CACHE_TTL_S,
        logger=None,
        clock=time.time,
    ):
        self._call = call_llm
        self.external_channels = tuple(external_channels)
        self.external_commands = tuple(external_commands)
        self.cache_ttl_s = cache_ttl_s
        self._log = logger
        self._clock = clock
        self._cache: Dict[int, Dict[str, Any]] = {}  # djb2(text) -> {ts, result}
        self.stats = {"calls": 0, "cache_hits": 0, "errors": 0}

    @property
    def enabled(self) -> bool:
        return self._call is not None

    def is_external(self, channel=None, tool_name=None, command=None) -> bool:
        return is_external_comm(
            channel, tool_name, command, self.external_channels, self.external_commands
        )

    def _prompt(self, text: str, facts: List[Dict[str, Any]]) -> str:
        fact_lines = "\n".join(
            f"- {f.get('subject')} {f.get('predicate')}: {f.get('value', f.get('object'))}"
            for f in facts[:50]
        )
        return "\n".join(
            [
                "You are a factual-accuracy validator for outbound agent messages.",
                "Known facts:",
                fact_lines or "(none)",
                "",
                "Message to validate:",
                "---",
                text,
                "---",
                "",
                'Respond with JSON: {"verdict": "pass"|"flag"|"block", "reason": "..."}.',
                "Block only for clear contradictions of known facts; flag for",
                "unverifiable strong claims; pass otherwise.",
            ]
        )

    def validate(self, text: str, facts: List[Dict[str, Any]], is_external: bool) -> Dict[str, Any]:
        """Returns {verdict, reason, cached}. Fail-open (pass) on errors;
        non-external text is never sent to the LLM."""
        if not self.enabled or not is_external or not text.strip():
            return {"verdict": "pass", "reason": "", "cached": False}
        key = djb2(text)
        now = self._clock()
        hit = self._cache.get(key)
        if hit is not None and now - hit["ts"] < self.cache_ttl_s:
            self.stats["cache_hits"] += 1
            return {**hit["result"], "cached": True}
        self.stats["calls"] += 1
        try:
            raw = self._call(self._prompt(text, facts))
            data = json.loads(raw)
            verdict = data.get("verdict", "pass")
            if verdict not in ("pass", "flag", "block"):
                verdict = "pass"
            result = {"verdict": verdict, "reason": str(data.get("reason", ""))}
        except Exception as exc:
            self.stats["errors"] += 1
            if self._log is not None:
                self._log.warn("LLM validation failed (fail-open): %s", exc)
            result = {"verdict": "pass", "reason": f"llm-error: {exc}"}
        self._cache[key] = {"ts": now, "result": result}
        # opportunistic TTL sweep to bound the cache
        if len(self._cache) > 1024:
            self._cache = {
                k: v for k, v in self._cache.items() if now - v["ts"] < self.cache_ttl_s
            }
        return {**result, "cached": False}
