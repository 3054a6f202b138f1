"""Conversation thread tracking.

Parity target: cortex `src/thread-tracker.ts` — per message: regex signal
extraction with context windows (decision -50/+100, wait +80, `:42-82`),
word-overlap thread matching (>=2 shared words of >2 chars, `:24-37`),
mood detection (last match wins), topic -> thread creation with noise
filter (`:130-145`), closure/decision/wait/mood application (`:193-239`),
prune closed > pruneDays + maxThreads cap keeping open threads
(`:269-289`), persists `threads.json` v2 with integrity block
(`:308-320`).
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
class ThreadTrackerConfig:
    enabled: bool = True
    prune_days: float = 14.0
    max_threads: int = 50


def matches_thread(thread: Dict[str, Any], text: str, min_overlap: int = 2) -> bool:
    """Word overlap >= min_overlap, words > 2 chars (thread-tracker.ts:24-37)."""
    thread_words = {w for w in str(thread.get("title", "")).lower().split() if len(w) > 2}
    text_words = {w for w in text.lower().split() if len(w) > 2}
    return len(thread_words & text_words) >= min_overlap


def extract_signals(text: str, language: P.Language = "both") -> Dict[str, List]:
    """Decision/close/wait/topic signals with the reference's context
    windows (thread-tracker.ts:42-82)."""
    reg = P.get_registry(language)
    signals: Dict[str, List] = {"decisions": [], "closures": [], "waits": [], "topics": []}
    for rx in reg.get_patterns("decision"):
        for m in rx.finditer(text):
            start = max(0, m.start() - 50)
            end = min(len(text), m.end() + 100)
            signals["decisions"].append(text[start:end].strip())
    for rx in reg.get_patterns("close"):
        if rx.search(text):
            signals["closures"].append(True)
    for rx in reg.get_patterns("wait"):
        for m in rx.finditer(text):
            end = min(len(text), m.end() + 80)
            signals["waits"].append(text[m.start():end].strip())
    for rx in reg.get_patterns("topic"):
        for m in rx.finditer(text):
            if m.groups() and m.group(1):
                signals["topics"].append(m.group(1).strip())
    return signals


def infer_priority(text: str, language: P.Language = "both") -> str:
    lower = text.lower()
    for kw in P.high_impact_keywords(language):
        if kw in lower:
            return "high"
    return "medium"


class ThreadTracker:
    def __init__(
        self,
        workspace: str,
        config: Optional[ThreadTrackerConfig] = None,
        language: P.Language = "both",
        clock=time.time,
    ):
        self.config = config or ThreadTrackerConfig()
        self.language = language
        self.clock = clock
        self.file_path = os.path.join(reboot_dir(workspace), "threads.json")
        self.writeable = ensure_reboot_dir(workspace)
        data = load_json(self.file_path)
        self.threads: List[Dict[str, Any]] = data.get("threads", []) if isinstance(data.get("threads"), list) else []
        self.session_mood: str = data.get("session_mood", "neutral")
        self.events_processed = 0
        self.last_event_timestamp = ""
        self.dirty = False

    def _now(self) -> str:
        return _dt.datetime.fromtimestamp(self.clock(), _dt.timezone.utc).isoformat().replace("+00:00", "Z")

    # -- thread mutations --------------------------------------------------
    def _create_from_topics(self, topics: List[str], sender: str, mood: str, now: str) -> None:
        for topic in topics:
            if P.is_noise_topic(topic, self.language):
                continue
            exists = any(
                t.get("title", "").lower() == topic.lower() or matches_thread(t, topic)
                for t in self.threads
            )
            if not exists:
                self.threads.append({
                    "id": str(uuid.uuid4()),
                    "title": topic,
                    "status": "open",
                    "priority": infer_priority(topic, self.language),
                    "summary": f"Topic detected from {sender}",
                    "decisions": [],
                    "waiting_for": None,
                    "mood": mood,
                    "last_activity": now,
                    "created": now,
                })

    def _close_matching(self, content: str, closures: List[bool], now: str) -> None:
        if not closures:
            return
        for t in self.threads:
            if t["status"] == "open" and matches_thread(t, content):
                t["status"] = "closed"
                t["last_activity"] = now

    def _apply_decisions(self, decisions: List[str], now: str) -> None:
        for ctx in decisions:
            for t in self.threads:
                if t["status"] == "open" and matches_thread(t, ctx):
                    short = ctx[:100]
                    if short not in t["decisions"]:
                        t["decisions"].append(short)
                        t["last_activity"] = now

    def _apply_waits(self, waits: List[str], content: str, now: str) -> None:
        for wait_ctx in waits:
            for t in self.threads:
                if t["status"] == "open" and matches_thread(t, content):
                    t["waiting_for"] = wait_ctx[:100]
                    t["last_activity"] = now

    def _apply_mood(self, mood: str, content: str) -> None:
        if mood == "neutral":
            return
        for t in self.threads:
            if t["status"] == "open" and matches_thread(t, content):
                t["mood"] = mood

    # -- API ---------------------------------------------------------------
    def process_message(self, content: str, sender: str = "user") -> None:
        if not content or not self.config.enabled:
            return
        signals = extract_signals(content, self.language)
        mood = P.detect_mood(content, self.language)
        now = self._now()
        self.events_processed += 1
        self.last_event_timestamp = now
        if mood != "neutral":
            self.session_mood = mood
        self._create_from_topics(signals["topics"], sender, mood, now)
        self._close_matching(content, signals["closures"], now)
        self._apply_decisions(signals["decisions"], now)
        self._apply_waits(signals["waits"], content, now)
        self._apply_mood(mood, content)
        self.dirty = True
        self._prune_and_cap()
        self.persist()

    def apply_llm_analysis(self, analysis: Dict[str, Any]) -> None:
        """Merge LLM enhancement results (thread-tracker.ts:151-191)."""
        now = self._now()
        for lt in analysis.get("threads", []):
            title = lt.get("title", "")
            if not title or P.is_noise_topic(title, self.language):
                continue
            exists = any(
                t.get("title", "").lower() == title.lower() or matches_thread(t, title)
                for t in self.threads
            )
            if not exists:
                self.threads.append({
                    "id": str(uuid.uuid4()),
                    "title": title,
                    "status": lt.get("status", "open"),
                    "priority": infer_priority(title, self.language),
                    "summary": lt.get("summary", "LLM-detected"),
                    "decisions": [],
                    "waiting_for": None,
                    "mood": analysis.get("mood", "neutral"),
                    "last_activity": now,
                    "created": now,
                })
        for closure in analysis.get("closures", []):
            for t in self.threads:
                if t["status"] == "open" and matches_thread(t, closure):
                    t["status"] = "closed"
                    t["last_activity"] = now
        if analysis.get("mood") and analysis["mood"] != "neutral":
            self.session_mood = analysis["mood"]
        self.dirty = True
        self.persist()

    def _prune_and_cap(self) -> None:
        cutoff = (
            _dt.datetime.fromtimestamp(
                self.clock() - self.config.prune_days * 86400, _dt.timezone.utc
            )
            .isoformat()
            .replace("+00:00", "Z")
        )
        self.threads = [
            t for t in self.threads
            if not (t["status"] == "closed" and t.get("last_activity", "") < cutoff)
        ]
        if len(self.threads) > self.config.max_threads:
            open_t = [t for t in self.threads if t["status"] == "open"]
            closed = sorted(
                (t for t in self.threads if t["status"] == "closed"),
                key=lambda t: t.get("last_activity", ""),
            )
            budget = self.config.max_threads - len(open_t)
            self.threads = open_t + closed[max(0, len(closed) - budget):]

    def _build_data(self) -> Dict[str, Any]:
        return {
            "version": 2,
            "updated": self._now(),
            "threads": self.threads,
            "integrity": {
                "last_event_timestamp": self.last_event_timestamp or self._now(),
                "events_processed": self.events_processed,
                "source": "hooks",
            },
            "session_mood": self.session_mood,
        }

    def persist(self) -> None:
        if not self.writeable:
            return
        if save_json(self.file_path, self._build_data()):
            self.dirty = False
        else:
            self.writeable = False

    def flush(self) -> bool:
        if not self.dirty:
            return True
        return save_json(self.file_path, self._build_data())

    def get_threads(self) -> List[Dict[str, Any]]:
        return list(self.threads)
