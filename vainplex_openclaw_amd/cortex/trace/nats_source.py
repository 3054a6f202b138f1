"""NATS JetStream trace source for the trace analyzer.

Parity target: cortex `src/trace-analyzer/nats-trace-source.ts` —
reads the eventstore's JetStream stream via $JS.API message GETs:
binary-search of the start sequence by event timestamp (`:204-228`),
sequential scan with a consecutive-miss cutoff of 50 (`:176-189`),
time-range / event-type / agent filters with past-end early stop
(`:192-199`), fetchByAgent delegation, last-sequence / event-count via
STREAM.INFO state, graceful-null creation when no client can connect
(`:103-115` — the reference gates on the optional `nats` npm package;
this build gates on the wire connection itself, eventstore/nats_client).
"""

from __future__ import annotations

import json
from typing import Any, Dict, Iterator, List, Optional

from ...eventstore.nats_client import JetStreamClient
from .events import NormalizedEvent, normalize_event

MAX_CONSECUTIVE_MISSES = 50


class NatsTraceSource:
    """TraceSource over a connected JetStreamClient (injectable for tests)."""

    def __init__(self, js: JetStreamClient, logger=None):
        self.js = js
        self.logger = logger
        self._closed = False

    # -- stream state -------------------------------------------------------
    def _state(self) -> Optional[Dict[str, Any]]:
        info = self.js.stream_info()
        state = (info or {}).get("state")
        return state if isinstance(state, dict) else None

    def get_last_sequence(self) -> int:
        state = self._state()
        return int(state.get("last_seq", 0)) if state else 0

    def get_event_count(self) -> int:
        state = self._state()
        return int(state.get("messages", 0)) if state else 0

    # -- message decode -----------------------------------------------------
    def _raw(self, seq: int) -> Optional[Dict[str, Any]]:
        msg = self.js.get_message(seq)
        if msg is None:
            return None
        try:
            data = json.loads(msg["data"].decode("utf-8"))
        except (ValueError, UnicodeDecodeError):
            return None
        return data if isinstance(data, dict) else None

    def _ts_at(self, seq: int) -> Optional[float]:
        raw = self._raw(seq)
        if raw is None:
            return None
        ts = raw.get("ts", raw.get("timestamp"))
        return float(ts) if isinstance(ts, (int, float)) else None

    def _find_start_sequence(self, first: int, last: int, target_ms: float) -> int:
        """First seq with ts >= target_ms. An unreadable probe (retained-
        out seq or junk payload) is resolved by scanning a few seqs
        forward for a readable timestamp before deciding the direction —
        slightly more robust than the reference's null-means-before-start
        rule, which skips valid events past an in-range junk record."""
        lo, hi = first, last
        while lo < hi:
            mid = (lo + hi) // 2
            ts = None
            for probe in range(mid, min(mid + 8, hi) + 1):
                ts = self._ts_at(probe)
                if ts is not None:
                    break
            if ts is None or ts < target_ms:
                lo = mid + 1
            else:
                hi = mid   # a readable seq at/after mid qualifies; mid may too
        return lo

    # -- fetch ----------------------------------------------------------------
    def fetch_by_time_range(
        self,
        start_ms: float,
        end_ms: float = float("inf"),
        event_types: Optional[List[str]] = None,
        agents: Optional[List[str]] = None,
        max_events: Optional[int] = None,
    ) -> Iterator[NormalizedEvent]:
        state = self._state()
        if not state:
            return
        first = int(state.get("first_seq", 1))
        last = int(state.get("last_seq", 0))
        if last < first:
            return
        start_seq = self._find_start_sequence(first, last, start_ms)
        if self.logger:
            self.logger.info(
                f"[trace-analyzer] Scanning seq {start_seq}-{last} "
                f"(skipped {start_seq - first} of {last - first + 1} events)")
        yielded = 0
        misses = 0
        for seq in range(start_seq, last + 1):
            if max_events is not None and yielded >= max_events:
                break
            raw = self._raw(seq)
            if raw is None:
                misses += 1
                if misses > MAX_CONSECUTIVE_MISSES:
                    break
                continue
            misses = 0
            ev = normalize_event(raw, seq)
            if ev is None:
                continue
            if ev.ts < start_ms:
                continue
            if ev.ts > end_ms:
                break                       # stream is time-ordered: past end
            if event_types is not None and ev.type not in event_types:
                continue
            if agents is not None and ev.agent not in agents:
                continue
            yielded += 1
            yield ev

    def fetch_by_agent(self, agent: str, start_ms: float,
                       end_ms: float = float("inf"), **kw) -> Iterator[NormalizedEvent]:
        return self.fetch_by_time_range(start_ms, end_ms, agents=[agent], **kw)

    def fetch(self, since_ts: float = 0) -> List[NormalizedEvent]:
        """TraceAnalyzer source protocol (analyzer.py run())."""
        return list(self.fetch_by_time_range(since_ts))

    def close(self) -> None:
        if self._closed:
            return
        self._closed = True
        try:
            self.js.close()
        except Exception:
            pass


def create_nats_trace_source(nats_config: Dict[str, Any], logger=None,
                             transport=None) -> Optional[NatsTraceSource]:
    """Connect a JetStreamClient and wrap it; None when the connection
    fails (graceful degradation, nats-trace-source.ts:103-115). The
    trace-analyzer nats config {url, stream, subjectPrefix, user,
    password} maps onto the eventstore client's config keys."""
    cfg = {
        "natsUrl": nats_config.get("url", "nats://localhost:4222"),
        "streamName": nats_config.get("stream", "openclaw-events"),
        "subjectPrefix": nats_config.get("subjectPrefix", "openclaw.events"),
        "connectTimeoutMs": nats_config.get("connectTimeoutMs", 10000),
        "publishTimeoutMs": nats_config.get("publishTimeoutMs", 5000),
    }
    if nats_config.get("user"):
        cfg["natsUrl"] = cfg["natsUrl"].replace(
            "nats://", f"nats://{nats_config['user']}:{nats_config.get('password', '')}@", 1)
    js = JetStreamClient(cfg, logger=logger, transport=transport)
    try:
        js.connect()
    except Exception as exc:
        if logger:
            logger.warn(f"[trace-analyzer] NATS connection failed: {exc}")
        return None
    if logger:
        logger.info(f"[trace-analyzer] Connected to NATS at {cfg['natsUrl']}")
    return NatsTraceSource(js, logger=logger)
