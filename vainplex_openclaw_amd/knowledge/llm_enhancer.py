"""Batched LLM entity/fact extraction (optional enhancement layer).

Parity target: reference `openclaw-knowledge-engine/src/llm-enhancer.ts` —
batch of {id, text} items, immediate send at batchSize, cooldown-timer
send otherwise, Ollama-style JSON-mode prompt, tolerant response parsing
(outer `response` string or whole object), entity/fact transforms
(predicate lowercased + hyphenated, importance clamped to [0,1] with 0.7
default).

`call_llm` is dependency-injected (host wires an OpenAI-compatible or
Ollama endpoint; tests inject fakes) — there is no network in this
environment, so the default is None and the enhancer is a no-op.
"""

from __future__ import annotations

import json
import re
import threading
import time
from typing import Callable, Dict, List, Optional


def construct_prompt(texts: List[str]) -> str:
    conversation = "\n".join(texts)
    return "\n".join(
        [
            "Analyze the following conversation and extract key entities and facts.",
            'Respond with a single JSON object containing "entities" and "facts".',
            "",
            'For "entities", provide objects with "type", "value", and "importance".',
            'Valid types: "person", "location", "organization", "product", "concept".',
            "",
            'For "facts", provide triples (subject, predicate, object).',
            "",
            "Conversation:",
            "---",
            conversation,
            "---",
            "",
            "JSON Response:",
        ]
    )


def parse_llm_response(response_json: str) -> Dict[str, list]:
    """Tolerates Ollama envelopes ({'response': '<json string>'}) and bare
    JSON objects (llm-enhancer.ts parseLlmResponse)."""
    outer = json.loads(response_json)
    if isinstance(outer, dict) and isinstance(outer.get("response"), str):
        data = json.loads(outer["response"])
    else:
        data = outer
    if not isinstance(data, dict):
        raise ValueError("LLM response is not a valid object.")
    return {
        "entities": data.get("entities") if isinstance(data.get("entities"), list) else [],
        "facts": data.get("facts") if isinstance(data.get("facts"), list) else [],
    }


def transform_entities(raw_entities: list, clock=time.time) -> List[Dict]:
    out = []
    now = time.strftime("%Y-%m-%dT%H:%M:%S", time.gmtime(clock())) + "Z"
    for r in raw_entities:
        if not isinstance(r, dict):
            continue
        value, etype = r.get("value"), r.get("type")
        if not isinstance(value, str) or not isinstance(etype, str):
            continue
        value = value.strip()
        etype = etype.lower()
        imp = r.get("importance")
        imp = max(0.0, min(1.0, imp)) if isinstance(imp, (int, float)) else 0.7
        out.append(
            {
                "id": f"{etype}:{re.sub(chr(92) + 's+', '-', value.lower())}",
                "value": value,
                "type": etype,
                "mentions": [value],
                "count": 1,
                "importance": imp,
                "lastSeen": now,
                "source": ["llm"],
            }
        )
    return out


def transform_facts(raw_facts: list) -> List[Dict]:
    out = []
    for r in raw_facts:
        if not isinstance(r, dict):
            continue
        s, p, o = r.get("subject"), r.get("predicate"), r.get("object")
        if not (isinstance(s, str) and isinstance(p, str) and isinstance(o, str)):
            continue
        out.append(
            {
                "subject": s.strip(),
                "predicate": re.sub(r"\s+", "-", p.strip().lower()),
                "object": o.strip(),
                "source": "extracted-llm",
            }
        )
    return out


class LlmEnhancer:
    def __init__(
        self,
        call_llm: Optional[Callable[[str], str]] = None,
        batch_size: int = 3,
        cooldown_ms: int = 5000,
        logger=None,
        clock=time.time,
    ):
        self._call = call_llm
        self.batch_size = batch_size
        self.cooldown_ms = cooldown_ms
        self._log = logger
        self._clock = clock
        self.batch: List[Dict] = []
        self._timer: Optional[threading.Timer] = None
        self._on_result: Optional[Callable[[Dict], None]] = None

    @property
    def enabled(self) -> bool:
        return self._call is not None

    def set_result_handler(self, fn: Callable[[Dict], None]) -> None:
        self._on_result = fn

    def add_to_batch(self, item_id: str, text: str) -> Optional[Dict]:
        """Returns the batch result when the add triggered an immediate
        send, else None (cooldown timer queues a later send)."""
        if not self.enabled:
            return None
        self.batch.append({"id": item_id, "text": text})
        if len(self.batch) >= self.batch_size:
            return self.send_batch()
        self._reset_cooldown()
        return None

    def _reset_cooldown(self) -> None:
        self.clear_timers()
        self._timer = threading.Timer(self.cooldown_ms / 1000.0, self._cooldown_fire)
        self._timer.daemon = True
        self._timer.start()

    def _cooldown_fire(self) -> None:
        result = self.send_batch()
        if result is not None and self._on_result is not None:
            self._on_result(result)

    def clear_timers(self) -> None:
        if self._timer is not None:
            self._timer.cancel()
            self._timer = None

    def send_batch(self) -> Optional[Dict]:
        self.clear_timers()
        if not self.batch or self._call is None:
            return None
        current, self.batch = self.batch, []
        prompt = construct_prompt([i["text"] for i in current])
        try:
            raw = self._call(prompt)
            parsed = parse_llm_response(raw)
            return {
                "entities": transform_entities(parsed["entities"], self._clock),
                "facts": transform_facts(parsed["facts"]),
            }
        except Exception as exc:
            if self._log is not None:
                self._log.error("Failed to send or process LLM batch: %s", exc)
            return None
