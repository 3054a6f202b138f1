"""Pre-compaction snapshot + narrative generation.

Parity target: cortex `src/pre-compaction.ts` (hot-snapshot.md of the last
N messages on before_compaction, force-flush of all trackers, narrative +
boot context regeneration) and `src/narrative-generator.ts` (narrative.md
summary of threads/decisions/mood).
"""

from __future__ import annotations

import datetime as _dt
import os
import time
from collections import deque
from typing import Any, Deque, Dict, List, Tuple

from .boot_context import BootContextGenerator
from .storage import reboot_dir, save_text


class NarrativeGenerator:
    """Writes narrative.md: a short prose state-of-the-session summary."""

    def __init__(self, workspace: str, clock=time.time):
        self.workspace = workspace
        self.clock = clock

    def generate(self, threads: List[Dict[str, Any]], decisions: List[Dict[str, Any]], mood: str) -> str:
        now = _dt.datetime.fromtimestamp(self.clock())
        open_t = [t for t in threads if t.get("status") == "open"]
        closed_t = [t for t in threads if t.get("status") == "closed"]
        parts = [f"As of {now.isoformat(timespec='minutes')}, the session mood is {mood}."]
        if open_t:
            titles = ", ".join(t.get("title", "?") for t in open_t[:5])
            parts.append(f"{len(open_t)} thread(s) remain open: {titles}.")
        if closed_t:
            parts.append(f"{len(closed_t)} thread(s) were closed.")
        if decisions:
            last = decisions[-1]
            parts.append(f"Most recent decision: {str(last.get('what', ''))[:120]}")
        waiting = [t for t in open_t if t.get("waiting_for")]
        if waiting:
            parts.append(
                "Blocked on: " + "; ".join(f"{t['title']} ({t['waiting_for']})" for t in waiting[:3])
            )
        text = "\n\n".join(parts) + "\n"
        save_text(os.path.join(reboot_dir(self.workspace), "narrative.md"), text)
        return text


class PreCompaction:
    """Keeps a rolling window of recent messages; on before_compaction,
    writes hot-snapshot.md, flushes trackers, regenerates narrative +
    boot context."""

    def __init__(self, workspace: str, window: int = 20, clock=time.time):
        self.workspace = workspace
        self.window = window
        self.clock = clock
        self.recent: Deque[Tuple[str, str]] = deque(maxlen=window)
        self.narrative = NarrativeGenerator(workspace, clock)
        self.boot = BootContextGenerator(workspace, clock=clock)

    def observe(self, sender: str, content: str) -> None:
        if content:
            self.recent.append((sender, content[:400]))

    def run(self, thread_tracker=None, decision_tracker=None, commitment_tracker=None) -> str:
        now = _dt.datetime.fromtimestamp(self.clock())
        lines = [f"# Hot snapshot — {now.isoformat(timespec='seconds')}", ""]
        for sender, content in self.recent:
            lines.append(f"**{sender}**: {content}")
        snapshot = "\n\n".join(lines) + "\n"
        save_text(os.path.join(reboot_dir(self.workspace), "hot-snapshot.md"), snapshot)

        threads: List[Dict[str, Any]] = []
        decisions: List[Dict[str, Any]] = []
        mood = "neutral"
        if thread_tracker is not None:
            thread_tracker.flush()
            threads = thread_tracker.get_threads()
            mood = thread_tracker.session_mood
        if decision_tracker is not None:
            decision_tracker.flush()
            decisions = decision_tracker.decisions
        if commitment_tracker is not None:
            commitment_tracker.flush()
        self.narrative.generate(threads, decisions, mood)
        self.boot.generate()
        return snapshot
