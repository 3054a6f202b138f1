"""BOOT-CONTEXT.md generation on session start.

Parity target: cortex `src/boot-context.ts` — execution mode by hour
(`:18-24`), open threads sorted priority+recency (`:43-56`), staleness
warnings at 2 h / 8 h (`:61-84`), hot snapshot younger than 1 h
(`:87-94`), recent decisions (`:99-112`), narrative younger than 36 h
(`:115-120`), maxChars budget.
"""

from __future__ import annotations

import datetime as _dt
import os
import time
from dataclasses import dataclass
from typing import List, Optional

from .storage import is_file_older_than, load_json, load_text, reboot_dir, save_text

PRIORITY_ORDER = {"high": 0, "medium": 1, "low": 2}


@dataclass
class BootContextConfig:
    enabled: bool = True
    max_chars: int = 6000
    max_threads: int = 10
    max_decisions: int = 5


def execution_mode(hour: int) -> str:
    """Mode by hour of day (boot-context.ts:18-24)."""
    if 8 <= hour < 18:
        return "business-hours"
    if 18 <= hour < 23:
        return "evening"
    return "night-watch"


class BootContextGenerator:
    def __init__(self, workspace: str, config: Optional[BootContextConfig] = None, clock=time.time):
        self.workspace = workspace
        self.config = config or BootContextConfig()
        self.clock = clock
        self.dir = reboot_dir(workspace)

    def _path(self, name: str) -> str:
        return os.path.join(self.dir, name)

    def generate(self) -> str:
        cfg = self.config
        now = _dt.datetime.fromtimestamp(self.clock())
        lines: List[str] = []
        lines.append("# BOOT CONTEXT")
        lines.append(f"Generated: {now.isoformat(timespec='seconds')}")
        lines.append(f"Execution mode: **{execution_mode(now.hour)}**")
        lines.append("")

        threads_data = load_json(self._path("threads.json"))
        threads = [t for t in threads_data.get("threads", []) if t.get("status") == "open"]
        threads.sort(
            key=lambda t: (
                PRIORITY_ORDER.get(t.get("priority", "medium"), 1),
                t.get("last_activity", ""),
            )
        )
        # staleness warnings (2 h soft, 8 h hard)
        updated = threads_data.get("updated")
        if updated:
            try:
                age_h = (self.clock() - _dt.datetime.fromisoformat(
                    str(updated).replace("Z", "+00:00")).timestamp()) / 3600
                if age_h > 8:
                    lines.append(f"⚠️ Thread state is {age_h:.0f}h old — treat as unreliable.")
                elif age_h > 2:
                    lines.append(f"Note: thread state is {age_h:.1f}h old.")
                lines.append("")
            except ValueError:
                pass

        if threads:
            lines.append("## Open threads")
            for t in threads[: cfg.max_threads]:
                wait = f" (waiting: {t['waiting_for']})" if t.get("waiting_for") else ""
                lines.append(f"- [{t.get('priority', 'medium')}] {t.get('title')}{wait}")
            lines.append("")

        # hot snapshot (< 1 h)
        snap_path = self._path("hot-snapshot.md")
        if not is_file_older_than(snap_path, 1, now=self.clock()):
            snap = load_text(snap_path).strip()
            if snap:
                lines.append("## Hot snapshot (last hour)")
                lines.append(snap)
                lines.append("")

        decisions_data = load_json(self._path("decisions.json"))
        decisions = decisions_data.get("decisions", [])
        if decisions:
            lines.append("## Recent decisions")
            for d in decisions[-cfg.max_decisions:][::-1]:
                lines.append(f"- {d.get('date', '')}: {str(d.get('what', ''))[:100]}")
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

        out = "\n".join(lines)
        if len(out) > cfg.max_chars:
            out = out[: cfg.max_chars - 20] + "\n…(truncated)"
        save_text(self._path("BOOT-CONTEXT.md"), out)
        return out
