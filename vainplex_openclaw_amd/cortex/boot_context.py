"""BOOT-CONTEXT.md generation on session start.

Parity target: cortex `src/boot-context.ts` — execution mode by hour
(`:18-24`), open threads sorted priority (critical first) then recency
(newest first) (`:43-56`), integrity staleness warnings at 2 h / 8 h
(`:61-84`), hot snapshot younger than 1 h (`:87-94`), recent decisions
within `decision_recency_days` (`:99-112`), narrative younger than 36 h
(`:115-120`), session mood with emoji (`types.ts:283`), footer stats,
maxChars budget, `should_generate` gate and `write()` to the workspace
root BOOTSTRAP.md (`:140-260`).
"""

from __future__ import annotations

import datetime as _dt
import os
import time
from dataclasses import dataclass
from typing import Any, Dict, List, Optional

from .storage import is_file_older_than, load_json, load_text, reboot_dir, save_text

PRIORITY_ORDER = {"critical": 0, "high": 1, "medium": 2, "low": 3}
PRIORITY_EMOJI = {"critical": "🔴", "high": "🟠", "medium": "🟡", "low": "🔵"}
MOOD_EMOJI = {
    "neutral": "",
    "frustrated": "😤",
    "excited": "🔥",
    "tense": "⚡",
    "productive": "🔧",
    "exploratory": "🔬",
}
IMPACT_EMOJI = {"critical": "🔴", "high": "🟠", "medium": "🟡", "low": "🔵"}


@dataclass
class BootContextConfig:
    enabled: bool = True
    on_session_start: bool = True
    max_chars: int = 6000
    max_threads: int = 10
    max_decisions: int = 5
    decision_recency_days: int = 7


def execution_mode(hour: int) -> str:
    """Mode by hour of day (boot-context.ts:18-24)."""
    if 8 <= hour < 18:
        return "business-hours"
    if 18 <= hour < 23:
        return "evening"
    return "night-watch"


def get_open_threads(threads_data: Dict[str, Any], limit: int) -> List[Dict[str, Any]]:
    """Open threads, critical first, newest within a priority
    (boot-context.ts:43-56)."""
    threads = [t for t in threads_data.get("threads", []) if t.get("status") == "open"]
    threads.sort(key=lambda t: str(t.get("last_activity", "")), reverse=True)
    threads.sort(key=lambda t: PRIORITY_ORDER.get(t.get("priority", "medium"), 3))
    return threads[:limit]


def integrity_warning(threads_data: Dict[str, Any], now_s: float) -> str:
    """Staleness warning from the threads.json integrity block
    (boot-context.ts:61-84)."""
    integrity = threads_data.get("integrity") or {}
    last_ts = integrity.get("last_event_timestamp")
    if not last_ts:
        return "⚠️ No integrity data — thread tracker may not have run yet."
    try:
        last = _dt.datetime.fromisoformat(str(last_ts).replace("Z", "+00:00"))
        if last.tzinfo is None:
            last = last.replace(tzinfo=_dt.timezone.utc)
        age_min = (now_s - last.timestamp()) / 60
    except ValueError:
        return "⚠️ Could not parse integrity timestamp."
    if age_min > 480:
        return f"🚨 STALE DATA: Thread data is {round(age_min / 60)}h old."
    if age_min > 120:
        return f"⚠️ Data staleness: Thread data is {round(age_min / 60)}h old."
    return ""


class BootContextGenerator:
    def __init__(self, workspace: str, config: Optional[BootContextConfig] = None, clock=time.time):
        self.workspace = workspace
        self.config = config or BootContextConfig()
        self.clock = clock
        self.dir = reboot_dir(workspace)

    def _path(self, name: str) -> str:
        return os.path.join(self.dir, name)

    def should_generate(self) -> bool:
        return bool(self.config.enabled and self.config.on_session_start)

    def _load_threads_data(self) -> Dict[str, Any]:
        """threads.json; tolerates the legacy v1 bare-array format
        (boot-context.ts:29-38)."""
        import json

        try:
            with open(self._path("threads.json"), "r", encoding="utf-8") as fh:
                data = json.load(fh)
        except (OSError, ValueError):
            return {}
        if isinstance(data, list):
            return {"threads": data}
        return data if isinstance(data, dict) else {}

    def _recent_decisions(self) -> List[Dict[str, Any]]:
        cfg = self.config
        decisions = load_json(self._path("decisions.json")).get("decisions", [])
        cutoff = _dt.datetime.fromtimestamp(
            self.clock() - cfg.decision_recency_days * 86400
        ).strftime("%Y-%m-%d")
        recent = [d for d in decisions if str(d.get("date", "")) >= cutoff]
        return recent[-cfg.max_decisions:][::-1]

    def generate(self) -> str:
        cfg = self.config
        now = _dt.datetime.fromtimestamp(self.clock())
        lines: List[str] = []
        lines.append("# BOOT CONTEXT")
        lines.append(f"Generated: {now.isoformat(timespec='seconds')}")
        lines.append(f"Execution mode: **{execution_mode(now.hour)}**")

        threads_data = self._load_threads_data()

        mood = str(threads_data.get("session_mood", "neutral"))
        if mood != "neutral":
            lines.append(f"Last session mood: {mood} {MOOD_EMOJI.get(mood, '')}".rstrip())

        warning = integrity_warning(threads_data, self.clock())
        if warning:
            lines.append("")
            lines.append(warning)
        lines.append("")

        threads = get_open_threads(threads_data, cfg.max_threads)
        if threads:
            lines.append("## Open threads")
            for t in threads:
                wait = f" (waiting: {t['waiting_for']})" if t.get("waiting_for") else ""
                emoji = PRIORITY_EMOJI.get(t.get("priority", "medium"), "⚪")
                lines.append(f"- {emoji} [{t.get('priority', 'medium')}] {t.get('title')}{wait}")
            lines.append("")

        # hot snapshot (< 1 h)
        snap_path = self._path("hot-snapshot.md")
        if not is_file_older_than(snap_path, 1, now=self.clock()):
            snap = load_text(snap_path).strip()[:1000]
            if snap:
                lines.append("## Hot snapshot (last hour)")
                lines.append(snap)
                lines.append("")

        decisions = self._recent_decisions()
        if decisions:
            lines.append("## Recent decisions")
            for d in decisions:
                emoji = IMPACT_EMOJI.get(str(d.get("impact", "")), "⚪")
                lines.append(f"- {emoji} {d.get('date', '')}: {str(d.get('what', ''))[:100]}")
                if d.get("why"):
                    lines.append(f"  Why: {str(d['why'])[:100]}")
            lines.append("")

        commitments = load_json(self._path("commitments.json")).get("commitments", [])
        open_c = [c for c in commitments if c.get("status") == "open"]
        if open_c:
            lines.append("## Open commitments")
            for c in open_c[:5]:
                lines.append(f"- {c.get('action')} (by {c.get('by')})")
            lines.append("")

        # narrative (< 36 h)
        narr_path = self._path("narrative.md")
        if not is_file_older_than(narr_path, 36, now=self.clock()):
            narr = load_text(narr_path).strip()
            if narr:
                lines.append("## Narrative")
                lines.append(narr)
                lines.append("")

        lines.append("---")
        lines.append(
            f"_Boot context | {len(threads)} active threads | "
            f"{len(decisions)} recent decisions_"
        )

        out = "\n".join(lines)
        if len(out) > cfg.max_chars:
            out = out[: cfg.max_chars] + "\n\n_[truncated to token budget]_"
        save_text(self._path("BOOT-CONTEXT.md"), out)
        return out

    def write(self) -> bool:
        """Generate and write BOOTSTRAP.md at the workspace root
        (boot-context.ts:248-258)."""
        try:
            content = self.generate()
            save_text(os.path.join(self.workspace, "BOOTSTRAP.md"), content)
            return True
        except OSError:
            return False
