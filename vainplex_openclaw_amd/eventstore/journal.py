"""Embedded event journal: the MI355X-native JetStream equivalent.

The reference's only distributed element is NATS JetStream over TCP
(`openclaw-nats-eventstore/src/nats-client.ts:53-91`). This environment
has no network, so the event backbone is an embedded append-only journal
with the same semantics the suite depends on (SURVEY.md §5 "Distributed
communication backend"):

- stream of subjects `<prefix>.>` with limits-based retention
  (maxMessages / maxBytes / maxAgeHours; -1/-1/0 = unlimited, matching
  `config.ts:18-33`),
- monotonic per-stream sequence numbers,
- durable JSONL segments (one file per UTC day, like the audit trail),
- replay from a timestamp or sequence (the trace-analyzer's incremental
  `ProcessingState` consumer), subject-filtered consume,
- fire-and-forget publish from the hot path: the writer thread owns file
  I/O so `publish()` never blocks (`hooks.ts (nats):161-181` semantics),
- status counters {connected, stream, disconnectCount, publishFailures}
  (`nats-client.ts:18-23`).
"""

from __future__ import annotations

import json
import os
import threading
import time
from typing import Dict, Iterator, List, Optional, Tuple


class EventBlock:
    """A batch of pre-serialized envelopes published as one append (the
    high-rate GPU pipeline leg: one block per step, one ENVELOPE per
    message). Sequence numbers are per event: the block owns
    [first_seq, first_seq + count). Lines are parsed lazily on replay, so
    the hot-path publish cost is one lock + one list append."""

    __slots__ = ("blob", "count")

    def __init__(self, blob: bytes, count: int):
        self.blob = blob
        self.count = count

    def envelopes(self) -> Iterator[Dict]:
        for line in self.blob.splitlines():
            if line.strip():
                yield json.loads(line)


class EventJournal:
    def __init__(
        self,
        directory: Optional[str] = None,
        stream: str = "openclaw-events",
        subject_prefix: str = "openclaw.events",
        max_messages: int = -1,
        max_bytes: int = -1,
        max_age_hours: float = 0,
        durable: bool = True,
        clock=time.time,
    ):
        self.directory = directory
        self.stream = stream
        self.subject_prefix = subject_prefix
        self.max_messages = max_messages
        self.max_bytes = max_bytes
        self.max_age_hours = max_age_hours
        self.durable = durable and directory is not None
        self._clock = clock
        # in-memory ring: (seq, ts, subject, envelope | EventBlock)
        self._events: List[Tuple[int, float, str, object]] = []
        self._bytes = 0
        self._count = 0
        self._seq = 0
        self._lock = threading.Lock()
        self._cv = threading.Condition(self._lock)
        self._pending: List[Tuple[int, float, str, Dict]] = []
        self._stop = False
        self._writer: Optional[threading.Thread] = None
        self.publish_failures = 0
        self.disconnect_count = 0
        if self.durable:
            os.makedirs(directory, exist_ok=True)
            self._load_existing()
            self._writer = threading.Thread(target=self._writer_run, daemon=True)
            self._writer.start()

    # -- publish (hot path, never blocks on I/O) ---------------------------
    def publish(self, subject: str, envelope: Dict) -> int:
        """Append; returns the assigned sequence number."""
        ts = float(envelope.get("ts", self._clock() * 1000)) / 1000.0
        with self._lock:
            self._seq += 1
            seq = self._seq
            item = (seq, ts, subject, envelope)
            self._events.append(item)
            self._bytes += len(json.dumps(envelope))
            self._count += 1
            self._apply_retention_locked()
            if self.durable:
                self._pending.append(item)
                self._cv.notify()
        return seq

    def publish_block(self, subject: str, jsonl_blob: bytes, count: int,
                      ts_ms: Optional[float] = None) -> int:
        """Append `count` pre-serialized envelopes (newline-separated JSON)
        as one block; returns the FIRST assigned sequence number. Every
        envelope gets its own seq; replay expands the block lazily."""
        ts = float(ts_ms if ts_ms is not None else self._clock() * 1000) / 1000.0
        block = EventBlock(jsonl_blob, count)
        with self._lock:
            first = self._seq + 1
            self._seq += count
            item = (first, ts, subject, block)
            self._events.append(item)
            self._bytes += len(jsonl_blob)
            self._count += count
            self._apply_retention_locked()
            if self.durable:
                self._pending.append(item)
                self._cv.notify()
        return first

    def _drop_first_locked(self) -> None:
        dropped = self._events.pop(0)
        body = dropped[3]
        if isinstance(body, EventBlock):
            self._bytes -= len(body.blob)
            self._count -= body.count
        else:
            self._bytes -= len(json.dumps(body))
            self._count -= 1

    def _apply_retention_locked(self) -> None:
        if self.max_messages > 0:
            while self._count > self.max_messages and self._events:
                self._drop_first_locked()
        if self.max_bytes > 0:
            while self._events and self._bytes > self.max_bytes:
                self._drop_first_locked()
        if self.max_age_hours > 0:
            cutoff = self._clock() - self.max_age_hours * 3600.0
            while self._events and self._events[0][1] < cutoff:
                self._drop_first_locked()

    # -- read side ---------------------------------------------------------
    def replay(
        self,
        since_ts: float = 0,
        since_seq: int = 0,
        subject_filter: Optional[str] = None,
        limit: int = 0,
    ) -> Iterator[Tuple[int, Dict]]:
        """Yield (seq, envelope) in order. `subject_filter` supports the
        NATS `>` tail wildcard (`openclaw.events.main.>`)."""
        with self._lock:
            snapshot = list(self._events)
        n = 0
        for seq, ts, subject, env in snapshot:
            if ts < since_ts:
                continue
            if subject_filter and not _subject_match(subject, subject_filter):
                continue
            if isinstance(env, EventBlock):
                for i, e in enumerate(env.envelopes()):
                    s_i = seq + i
                    if s_i <= since_seq:
                        continue
                    yield s_i, e
                    n += 1
                    if limit and n >= limit:
                        return
                continue
            if seq <= since_seq:
                continue
            yield seq, env
            n += 1
            if limit and n >= limit:
                return

    def fetch_range(self, start_ts: float, end_ts: float) -> List[Dict]:
        return [
            env
            for _seq, env in self.replay(since_ts=start_ts)
            if float(env.get("ts", 0)) / 1000.0 <= end_ts
        ]

    def __len__(self) -> int:
        with self._lock:
            return self._count

    @property
    def last_seq(self) -> int:
        with self._lock:
            return self._seq

    def status(self) -> Dict:
        """Counter block parity (nats-client.ts:18-23)."""
        return {
            "connected": True,
            "stream": self.stream,
            "disconnectCount": self.disconnect_count,
            "publishFailures": self.publish_failures,
            "messages": len(self),
            "lastSeq": self.last_seq,
        }

    # -- durability --------------------------------------------------------
    def _segment_path(self, ts: float) -> str:
        day = time.strftime("%Y-%m-%d", time.gmtime(ts))
        return os.path.join(self.directory, f"{self.stream}-{day}.jsonl")

    def _writer_run(self) -> None:
        while True:
            with self._cv:
                while not self._pending and not self._stop:
                    self._cv.wait(0.2)
                if self._stop and not self._pending:
                    return
                items, self._pending = self._pending, []
            try:
                by_path: Dict[str, List[str]] = {}
                for seq, ts, subject, env in items:
                    if isinstance(env, EventBlock):
                        for i, e in enumerate(env.envelopes()):
                            line = json.dumps(
                                {"seq": seq + i, "subject": subject, "event": e},
                                ensure_ascii=False,
                            )
                            by_path.setdefault(self._segment_path(ts), []).append(line)
                        continue
                    line = json.dumps(
                        {"seq": seq, "subject": subject, "event": env}, ensure_ascii=False
                    )
                    by_path.setdefault(self._segment_path(ts), []).append(line)
                for path, lines in by_path.items():
                    with open(path, "a", encoding="utf-8") as fh:
                        fh.write("\n".join(lines) + "\n")
            except OSError:
                with self._lock:
                    self.publish_failures += len(items)

    def _load_existing(self) -> None:
        """Resume seq + ring from existing segments (replay across restarts)."""
        try:
            segs = sorted(
                f for f in os.listdir(self.directory)
                if f.startswith(self.stream + "-") and f.endswith(".jsonl")
            )
        except OSError:
            return
        for seg in segs:
            try:
                with open(os.path.join(self.directory, seg), "r", encoding="utf-8") as fh:
                    for line in fh:
                        line = line.strip()
                        if not line:
                            continue
                        try:
                            rec = json.loads(line)
                        except json.JSONDecodeError:
                            continue
                        seq = int(rec.get("seq", 0))
                        env = rec.get("event", {})
                        ts = float(env.get("ts", 0)) / 1000.0
                        self._events.append((seq, ts, rec.get("subject", ""), env))
                        self._bytes += len(json.dumps(env))
                        self._count += 1
                        self._seq = max(self._seq, seq)
            except OSError:
                continue
        self._apply_retention_locked()

    def drain(self) -> None:
        """Flush pending writes (drain() parity)."""
        if not self.durable:
            return
        deadline = time.time() + 5.0
        while time.time() < deadline:
            with self._lock:
                if not self._pending:
                    return
            time.sleep(0.01)

    def close(self) -> None:
        self.drain()
        with self._cv:
            self._stop = True
            self._cv.notify()
        if self._writer is not None:
            self._writer.join(timeout=5.0)


def _subject_match(subject: str, pattern: str) -> bool:
    """NATS-style matching: `*` = one token, `>` = rest."""
    st = subject.split(".")
    pt = pattern.split(".")
    i = 0
    for i, p in enumerate(pt):
        if p == ">":
            return True
        if i >= len(st):
            return False
        if p != "*" and p != st[i]:
            return False
    return len(st) == len(pt)
