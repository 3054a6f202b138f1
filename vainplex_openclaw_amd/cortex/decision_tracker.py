"""Decision extraction and persistence.

Parity target: cortex `src/decision-tracker.ts` — decision regex scan with
context windows (what: -50/+100, why: -100/+200, `:33-44`), impact
inference from high-impact keywords (`:22-28`), 24 h dedupe window on the
'what' text, maxDecisions cap, `decisions.json`.
"""

from __future__ import annotations

import datetime as _dt
import os
import time
import uuid
from dataclasses import dataclass
from typing import Any, Dict, List, Optional

from . import patterns as P
from .storage import ensure_reboot_dir, load_json, reboot_dir, save_json


@dataclass
class DecisionTrackerConfig:
    enabled: bool = True
    max_decisions: int = 200
    dedupe_window_hours: float = 24.0


def infer_impact(text: str, language: P.Language = "both") -> str:
    lower = text.lower()
    for kw in P.high_impact_keywords(language):
        if kw in lower:
            return "high"
    return "medium"


def extract_context(text: str, start: int, length: int) -> Dict[str, str]:
    what = text[max(0, start - 50): min(len(text), start + length + 100)].strip()
    why = text[max(0, start - 100): min(len(text), start + length + 200)].strip()
    return {"what": what, "why": why}


class DecisionTracker:
    def __init__(
        self,
        workspace: str,
        config: Optional[DecisionTrackerConfig] = None,
        language: P.Language = "both",
        clock=time.time,
    ):
        self.config = config or DecisionTrackerConfig()
        self.language = language
        self.clock = clock
        self.file_path = os.path.join(reboot_dir(workspace), "decisions.json")
        self.writeable = ensure_reboot_dir(workspace)
        data = load_json(self.file_path)
        self.decisions: List[Dict[str, Any]] = data.get("decisions", []) if isinstance(data.get("decisions"), list) else []

    def _now_dt(self) -> _dt.datetime:
        return _dt.datetime.fromtimestamp(self.clock(), _dt.timezone.utc)

    def _is_duplicate(self, what: str, now: _dt.datetime) -> bool:
        window = _dt.timedelta(hours=self.config.dedupe_window_hours)
        for d in self.decisions:
            if d.get("what") != what:
                continue
            try:
                ts = _dt.datetime.fromisoformat(str(d.get("timestamp", "")).replace("Z", "+00:00"))
            except ValueError:
                continue
            if now - ts < window:
                return True
        return False

    def process_message(self, content: str, sender: str = "user") -> int:
        if not content or not self.config.enabled:
            return 0
        reg = P.get_registry(self.language)
        now = self._now_dt()
        added = 0
        for rx in reg.get_patterns("decision"):
            for m in rx.finditer(content):
                ctx = extract_context(content, m.start(), len(m.group(0)))
                if self._is_duplicate(ctx["what"], now):
                    continue
                self.decisions.append({
                    "id": str(uuid.uuid4()),
                    "what": ctx["what"],
                    "why": ctx["why"],
                    "date": now.date().isoformat(),
                    "timestamp": now.isoformat().replace("+00:00", "Z"),
                    "impact": infer_impact(ctx["what"], self.language),
                    "sender": sender,
                })
                added += 1
        if added:
            if len(self.decisions) > self.config.max_decisions:
                self.decisions = self.decisions[-self.config.max_decisions:]
            self.persist()
        return added

    def recent(self, n: int = 10) -> List[Dict[str, Any]]:
        return self.decisions[-n:]

    def recent_within(self, days: float = 7.0, limit: int = 10) -> List[Dict[str, Any]]:
        """Decisions from the last `days`, newest last, capped at `limit`
        (decision-tracker.ts getRecentDecisions)."""
        cutoff = (self._now_dt() - _dt.timedelta(days=days)).date().isoformat()
        recent = [d for d in self.decisions if str(d.get("date", "")) >= cutoff]
        return recent[-limit:]

    def persist(self) -> None:
        if not self.writeable:
            return
        data = {
            "version": 1,
            "updated": self._now_dt().isoformat().replace("+00:00", "Z"),
            "decisions": self.decisions,
        }
        if not save_json(self.file_path, data):
            self.writeable = False

    def flush(self) -> bool:
        self.persist()
        return self.writeable
