"""Optional batched LLM enhancement of tracker analysis.

Parity target: cortex `src/llm-enhance.ts` — OpenAI-compatible batch
analysis (batchSize 3, default Ollama http://localhost:11434 with
mistral:7b, `:28-35`); JSON-schema prompt asking for
threads/decisions/closures/mood (`:52-67`); results merged via
ThreadTracker.apply_llm_analysis. The HTTP transport is injected
(`call_llm`) — offline by default.
"""

from __future__ import annotations

import json
from typing import Any, Callable, Dict, List, Optional

DEFAULT_CONFIG = {
    "enabled": False,
    "baseUrl": "http://localhost:11434",
    "model": "mistral:7b",
    "batchSize": 3,
    "timeoutMs": 30000,
}

ANALYSIS_PROMPT = """Analyze the following conversation messages. Respond with ONLY a JSON object:
{"threads": [{"title": "...", "status": "open"|"closed", "summary": "..."}],
 "decisions": ["..."], "closures": ["..."], "mood": "neutral|frustrated|excited|tense|productive|exploratory"}

Messages:
"""


class LlmEnhancer:
    def __init__(
        self,
        config: Optional[Dict[str, Any]] = None,
        call_llm: Optional[Callable[[str], str]] = None,
    ):
        cfg = dict(DEFAULT_CONFIG)
        cfg.update(config or {})
        self.config = cfg
        self.call_llm = call_llm
        self._batch: List[str] = []

    @property
    def enabled(self) -> bool:
        return bool(self.config.get("enabled")) and self.call_llm is not None

    def add_message(self, content: str) -> Optional[Dict[str, Any]]:
        """Buffer messages; when batchSize reached, run analysis."""
        if not self.enabled or not content:
            return None
        self._batch.append(content[:600])
        if len(self._batch) < int(self.config.get("batchSize", 3)):
            return None
        return self.flush()

    def flush(self) -> Optional[Dict[str, Any]]:
        if not self.enabled or not self._batch:
            return None
        batch, self._batch = self._batch, []
        prompt = ANALYSIS_PROMPT + "\n".join(f"- {m}" for m in batch)
        try:
            raw = self.call_llm(prompt)
        except Exception:
            return None
        return parse_analysis(raw)


def parse_analysis(raw: str) -> Optional[Dict[str, Any]]:
    """Tolerant JSON extraction from an LLM reply."""
    if not raw:
        return None
    start = raw.find("{")
    end = raw.rfind("}")
    if start < 0 or end <= start:
        return None
    try:
        data = json.loads(raw[start : end + 1])
    except json.JSONDecodeError:
        return None
    if not isinstance(data, dict):
        return None
    return {
        "threads": [t for t in data.get("threads", []) if isinstance(t, dict) and t.get("title")],
        "decisions": [d for d in data.get("decisions", []) if isinstance(d, str)],
        "closures": [c for c in data.get("closures", []) if isinstance(c, str)],
        "mood": data.get("mood", "neutral"),
    }
