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


_SEVERITY_VERDICT = {"critical": "block", "high": "flag", "medium": "flag",
                     "low": "pass"}


def parse_llm_response(raw: str) -> Dict[str, Any]:
    """Parse an LLM validation reply (llm-validator.ts parseResponse):
    accepts either {"verdict", "reason"} or {"issues": [{severity,...}]};
    strips markdown code fences; malformed JSON / missing / non-array
    issues parse as pass; malformed issue objects are skipped; unknown
    severities default to medium (-> flag); the worst severity wins."""
    text = raw.strip()
    if text.startswith("```"):
        # ```json ... ``` (or bare ```)
        lines = text.splitlines()
        if lines and lines[0].startswith("```"):
            lines = lines[1:]
        if lines and lines[-1].strip().startswith("```"):
            lines = lines[:-1]
        text = "\n".join(lines).strip()
    try:
        data = json.loads(text)
    except (ValueError, TypeError):
        return {"verdict": "pass", "reason": "unparseable LLM reply"}
    if not isinstance(data, dict):
        return {"verdict": "pass", "reason": "unparseable LLM reply"}
    if "verdict" in data:
        verdict = data.get("verdict")
        if verdict not in ("pass", "flag", "block"):
            verdict = "pass"
        return {"verdict": verdict, "reason": str(data.get("reason", ""))}
    issues = data.get("issues")
    if not isinstance(issues, list) or not issues:
        return {"verdict": "pass", "reason": ""}
    worst = "pass"
    reasons = []
    order = {"pass": 0, "flag": 1, "block": 2}
    for issue in issues:
        if not isinstance(issue, dict):
            continue
        sev = str(issue.get("severity", "medium")).lower()
        v = _SEVERITY_VERDICT.get(sev, "flag")  # unknown -> medium -> flag
        if order[v] > order[worst]:
            worst = v
        if issue.get("description"):
            reasons.append(str(issue["description"]))
    return {"verdict": worst, "reason": "; ".join(reasons[:3])}


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
        fail_mode: str = "open",
        max_attempts: int = 2,
    ):
        self._call = call_llm
        self.external_channels = tuple(external_channels)
        self.external_commands = tuple(external_commands)
        self.cache_ttl_s = cache_ttl_s
        self._log = logger
        self._clock = clock
        self.fail_mode = fail_mode if fail_mode in ("open", "closed") else "open"
        self.max_attempts = max(1, int(max_attempts))
        self._cache: Dict[int, Dict[str, Any]] = {}  # djb2(text) -> {ts, result}
        self.stats = {"calls": 0, "cache_hits": 0, "errors": 0, "retries": 0}

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
        result = None
        last_exc: Optional[Exception] = None
        for attempt in range(self.max_attempts):
            try:
                raw = self._call(self._prompt(text, facts))
                result = parse_llm_response(raw)
                break
            except Exception as exc:
                last_exc = exc
                if attempt + 1 < self.max_attempts:
                    self.stats["retries"] += 1
        if result is None:
            self.stats["errors"] += 1
            if self._log is not None:
                self._log.warn("LLM validation failed (fail-%s): %s",
                               self.fail_mode, last_exc)
            # fail-open passes; fail-closed flags (llm-validator.ts
            # failMode 'closed' escalates instead of silently passing)
            verdict = "pass" if self.fail_mode == "open" else "flag"
            return {"verdict": verdict, "reason": f"llm-error: {last_exc}",
                    "cached": False}
        self._cache[key] = {"ts": now, "result": result}
        # opportunistic TTL sweep to bound the cache
        if len(self._cache) > 1024:
            self._cache = {
                k: v for k, v in self._cache.items() if now - v["ts"] < self.cache_ttl_s
            }
        return {**result, "cached": False}
