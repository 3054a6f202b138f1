"""Commitment (promise) tracking.

Parity target: cortex `src/commitment-tracker.ts` + `commitment-patterns.ts`
— multi-language promise detection ("I'll ...", "ich mach ...", etc.),
overdue after 7 days (`:7,32-41`), debounced saves (15 s in the reference;
synchronous flush always available), `commitments.json`.
"""

from __future__ import annotations

import datetime as _dt
import os
import re
import time
import uuid
from dataclasses import dataclass
from typing import Any, Dict, List, Optional

from ..utils.storage import DebouncedSaver
from .storage import ensure_reboot_dir, load_json, reboot_dir, save_json

OVERDUE_DAYS = 7.0

# (pattern, language); group 1 (when present) is the committed action text
COMMITMENT_PATTERNS: List = [
    (re.compile(r"\b(?:I'll|I will|I'm going to)\b\s+(.{5,80})", re.I), "en"),
    (re.compile(r"\b(?:let me|allow me to)\b\s+(.{5,80})", re.I), "en"),
    (re.compile(r"\b(?:I can do that|I'll handle|I'll take care)\b", re.I), "en"),
    (re.compile(r"\b(?:I promise|I commit to|I guarantee)\b\s*(.{5,80})?", re.I), "en"),
    (re.compile(r"\b(?:consider it done|I'm on it)\b", re.I), "en"),
    (re.compile(r"\b(?:ich werde|ich mach|ich kümmere mich um)\b\s*(.{5,80})?", re.I), "de"),
    (re.compile(r"\b(?:mach ich|wird gemacht|klar mach ich)\b", re.I), "de"),
    (re.compile(r"\b(?:versprochen|abgemacht|geht klar)\b", re.I), "de"),
    (re.compile(r"\b(?:je vais|je ferai|je m'en occupe)\b\s*(.{5,80})?", re.I), "fr"),
    (re.compile(r"\b(?:c'est noté|je m'engage à)\b", re.I), "fr"),
    (re.compile(r"\b(?:lo haré|me encargo|yo me ocupo)\b", re.I), "es"),
    (re.compile(r"\b(?:eu vou|eu farei|fico responsável)\b", re.I), "pt"),
    (re.compile(r"\b(?:pode deixar)\b", re.I), "pt"),
    (re.compile(r"\b(?:lo farò|me ne occupo|ci penso io)\b", re.I), "it"),
    (re.compile(r"(?:我会|我来|我负责|包在我身上)"), "zh"),
    (re.compile(r"(?:やります|対応します|承知しました)"), "ja"),
    (re.compile(r"(?:제가 할게요|처리하겠습니다)"), "ko"),
    (re.compile(r"(?:я сделаю|я займусь|обещаю)", re.I), "ru"),
]


def detect_commitments(text: str) -> List[Dict[str, str]]:
    out = []
    for rx, lang in COMMITMENT_PATTERNS:
        m = rx.search(text)
        if m:
            action = (m.group(1) if m.groups() and m.group(1) else m.group(0)).strip()
            out.append({"action": action, "language": lang, "source": m.group(0).strip()})
    return out


@dataclass
class CommitmentTrackerConfig:
    enabled: bool = True
    max_commitments: int = 100
    debounce_seconds: float = 15.0


class CommitmentTracker:
    def __init__(
        self,
        workspace: str,
        config: Optional[CommitmentTrackerConfig] = None,
        clock=time.time,
    ):
        self.config = config or CommitmentTrackerConfig()
        self.clock = clock
        self.file_path = os.path.join(reboot_dir(workspace), "commitments.json")
        self.writeable = ensure_reboot_dir(workspace)
        data = load_json(self.file_path)
        self.commitments: List[Dict[str, Any]] = (
            data.get("commitments", []) if isinstance(data.get("commitments"), list) else []
        )
        self.saver = DebouncedSaver(self._persist, self.config.debounce_seconds)

    def _now_iso(self) -> str:
        return _dt.datetime.fromtimestamp(self.clock(), _dt.timezone.utc).isoformat().replace("+00:00", "Z")

    def process_message(self, content: str, sender: str = "agent") -> int:
        if not content or not self.config.enabled:
            return 0
        found = detect_commitments(content)
        for f in found:
            self.commitments.append({
                "id": str(uuid.uuid4()),
                "action": f["action"][:120],
                "language": f["language"],
                "by": sender,
                "status": "open",
                "created": self._now_iso(),
                "completed": None,
            })
        if found:
            if len(self.commitments) > self.config.max_commitments:
                self.commitments = self.commitments[-self.config.max_commitments:]
            self.saver.mark_dirty()
        return len(found)

    def complete(self, commitment_id: str) -> bool:
        for c in self.commitments:
            if c["id"] == commitment_id and c["status"] == "open":
                c["status"] = "done"
                c["completed"] = self._now_iso()
                self.saver.mark_dirty()
                return True
        return False

    def overdue(self) -> List[Dict[str, Any]]:
        """Open commitments older than 7 days (commitment-tracker.ts:7)."""
        cutoff = self.clock() - OVERDUE_DAYS * 86400
        out = []
        for c in self.commitments:
            if c["status"] != "open":
                continue
            try:
                created = _dt.datetime.fromisoformat(c["created"].replace("Z", "+00:00")).timestamp()
            except ValueError:
                continue
            if created < cutoff:
                out.append(c)
        return out

    def open_commitments(self) -> List[Dict[str, Any]]:
        return [c for c in self.commitments if c["status"] == "open"]

    def _persist(self) -> None:
        if not self.writeable:
            return
        data = {"version": 1, "updated": self._now_iso(), "commitments": self.commitments}
        if not save_json(self.file_path, data):
            self.writeable = False

    def flush(self) -> None:
        self.saver.flush()
