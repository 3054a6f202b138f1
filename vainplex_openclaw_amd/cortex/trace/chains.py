"""Conversation chain reconstruction.

Parity target: cortex `src/trace-analyzer/chain-reconstructor.ts` —
bucket by (session, agent) (`:47-59`), split on lifecycle events /
30-minute gaps / 1000-event cap (`:33-45`), dedupe by event id,
chain id = sha256(session:agent:firstTs)[:16] (`:14-21`).
"""

from __future__ import annotations

import hashlib
from dataclasses import dataclass, field
from typing import Dict, List

from .events import NormalizedEvent

GAP_MS = 30 * 60 * 1000
MAX_CHAIN_EVENTS = 1000
LIFECYCLE_SPLITS = {"session.start", "session.end"}


@dataclass
class ConversationChain:
    id: str
    session: str
    agent: str
    events: List[NormalizedEvent] = field(default_factory=list)

    @property
    def start_ts(self) -> float:
        return self.events[0].ts if self.events else 0

    @property
    def end_ts(self) -> float:
        return self.events[-1].ts if self.events else 0


def chain_id(session: str, agent: str, first_ts: float) -> str:
    return hashlib.sha256(f"{session}:{agent}:{int(first_ts)}".encode()).hexdigest()[:16]


def reconstruct_chains(events: List[NormalizedEvent]) -> List[ConversationChain]:
    # dedupe by id (keep first), bucket by (session, agent)
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
            if current:
                chains.append(
                    ConversationChain(
                        id=chain_id(session, agent, current[0].ts),
                        session=session,
                        agent=agent,
                        events=current,
                    )
                )
                current = []

        prev_ts = None
        for ev in evs:
            split = False
            if ev.type in LIFECYCLE_SPLITS and current:
                split = True
            elif prev_ts is not None and ev.ts - prev_ts > GAP_MS:
                split = True
            elif len(current) >= MAX_CHAIN_EVENTS:
                split = True
            if split:
                flush()
            current.append(ev)
            prev_ts = ev.ts
        flush()
    chains.sort(key=lambda c: c.start_ts)
    return chains
