"""Conversation chain reconstruction.

Parity target: cortex `src/trace-analyzer/chain-reconstructor.ts` —
bucket by (session, agent) (`:47-59`), split on lifecycle events /
30-minute gaps (configurable) / run.end->run.start gaps > 5 min /
1000-event cap (`:33-45`), dedupe by event id, drop chains with fewer
than 2 events, chain id = sha256(session:agent:firstTs)[:16]
(`:14-21`); each chain records its boundary type (lifecycle vs gap)
and per-type counts.
"""

from __future__ import annotations

import hashlib
from dataclasses import dataclass, field
from typing import Dict, List

from .events import NormalizedEvent

GAP_MS = 30 * 60 * 1000
RUN_GAP_MS = 5 * 60 * 1000
MAX_CHAIN_EVENTS = 1000
LIFECYCLE_SPLITS = {"session.start", "session.end"}
RUN_END_TYPES = {"run.end", "run.ended"}
RUN_START_TYPES = {"run.start", "run.started"}


@dataclass
class ConversationChain:
    id: str
    session: str
    agent: str
    events: List[NormalizedEvent] = field(default_factory=list)
    boundary_type: str = "gap"

    @property
    def start_ts(self) -> float:
        return self.events[0].ts if self.events else 0

    @property
    def end_ts(self) -> float:
        return self.events[-1].ts if self.events else 0

    @property
    def type_counts(self) -> Dict[str, int]:
        counts: Dict[str, int] = {}
        for e in self.events:
            counts[e.type] = counts.get(e.type, 0) + 1
        return counts


def chain_id(session: str, agent: str, first_ts: float) -> str:
    return hashlib.sha256(f"{session}:{agent}:{int(first_ts)}".encode()).hexdigest()[:16]


def reconstruct_chains(
    events: List[NormalizedEvent],
    gap_minutes: float = 30.0,
    max_events: int = MAX_CHAIN_EVENTS,
    min_events: int = 2,
) -> List[ConversationChain]:
    # dedupe by id (keep first), bucket by (session, agent)
    gap_ms = gap_minutes * 60 * 1000
    seen = set()
    buckets: Dict[tuple, List[NormalizedEvent]] = {}
    for ev in sorted(events, key=lambda e: (e.ts, e.seq)):
        if ev.id and ev.id in seen:
            continue
        if ev.id:
            seen.add(ev.id)
        buckets.setdefault((ev.session, ev.agent), []).append(ev)

    chains: List[ConversationChain] = []
    for (session, agent), evs in buckets.items():
        current: List[NormalizedEvent] = []

        def flush():
            nonlocal current
            if len(current) >= min_events:
                boundary = (
                    "lifecycle"
                    if any(e.type in LIFECYCLE_SPLITS for e in current)
                    else "gap"
                )
                chains.append(
                    ConversationChain(
                        id=chain_id(session, agent, current[0].ts),
                        session=session,
                        agent=agent,
                        events=current,
                        boundary_type=boundary,
                    )
                )
            current = []

        prev_ts = None
        prev_type = None
        for ev in evs:
            split = False
            if ev.type in LIFECYCLE_SPLITS and current:
                split = True
            elif prev_ts is not None and ev.ts - prev_ts > gap_ms:
                split = True
            elif (
                prev_type in RUN_END_TYPES
                and ev.type in RUN_START_TYPES
                and prev_ts is not None
                and ev.ts - prev_ts > RUN_GAP_MS
            ):
                split = True
            elif len(current) >= max_events:
                split = True
            if split:
                flush()
            current.append(ev)
            prev_ts = ev.ts
            prev_type = ev.type
        flush()
    chains.sort(key=lambda c: c.start_ts)
    return chains
