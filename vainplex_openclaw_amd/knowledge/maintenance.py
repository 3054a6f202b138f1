"""Background maintenance: fact relevance decay + embeddings sync timers.

Parity target: reference `openclaw-knowledge-engine/src/maintenance.ts` —
decay every `decay.intervalHours` at `decay.rate`, embeddings sync every
`embeddings.syncIntervalMinutes` over unembedded facts, timers unref'd
(daemon threads here), run_* methods callable directly for tests.
"""

from __future__ import annotations

import threading
from typing import Optional


class Maintenance:
    def __init__(self, config: dict, fact_store, embeddings=None, logger=None):
        self.config = config
        self.fact_store = fact_store
        self.embeddings = embeddings
        self._log = logger
        self._decay_timer: Optional[threading.Timer] = None
        self._emb_timer: Optional[threading.Timer] = None

    def start(self) -> None:
        self.stop()
        decay = self.config.get("decay", {})
        if decay.get("enabled"):
            self._schedule_decay(decay.get("intervalHours", 24) * 3600.0)
        emb = self.config.get("embeddings", {})
        if self.embeddings is not None and self.embeddings.is_enabled():
            self._schedule_emb(emb.get("syncIntervalMinutes", 30) * 60.0)

    def stop(self) -> None:
        for t in (self._decay_timer, self._emb_timer):
            if t is not None:
                t.cancel()
        self._decay_timer = None
        self._emb_timer = None

    def _schedule_decay(self, interval_s: float) -> None:
        def fire() -> None:
            self.run_decay()
            self._schedule_decay(interval_s)

        self._decay_timer = threading.Timer(interval_s, fire)
        self._decay_timer.daemon = True
        self._decay_timer.start()

    def _schedule_emb(self, interval_s: float) -> None:
        def fire() -> None:
            self.run_embeddings_sync()
            self._schedule_emb(interval_s)

        self._emb_timer = threading.Timer(interval_s, fire)
        self._emb_timer.daemon = True
        self._emb_timer.start()

    def run_decay(self) -> int:
        """Decay pass; returns decayed count (maintenance.ts runDecay)."""
        try:
            rate = self.config.get("decay", {}).get("rate", 0.05)
            return self.fact_store.decay_facts(rate)
        except Exception as exc:
            if self._log is not None:
                self._log.error("Error during fact decay: %s", exc)
            return 0

    def run_embeddings_sync(self) -> int:
        """Sync unembedded facts; returns synced count."""
        if self.embeddings is None or not self.embeddings.is_enabled():
            return 0
        try:
            unembedded = self.fact_store.unembedded_facts()
            if not unembedded:
                return 0
            synced = self.embeddings.sync(unembedded)
            if synced > 0:
                self.fact_store.mark_embedded([f["id"] for f in unembedded[:synced]])
            return synced
        except Exception as exc:
            if self._log is not None:
                self._log.error("Error during embeddings sync: %s", exc)
            return 0
