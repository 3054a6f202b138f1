"""Append-only audit trail: per-day JSONL + Merkle hash chain.

Parity target: governance `src/audit-trail.ts` — record shape
{id, timestamp, timestampIso, verdict, reason, context(redacted), trust,
risk, matchedPolicies, evaluationUs, controls[]} (`:76-110`), one file per
UTC day (`:151-179`), ISO-27001 control derivation incl. forced
A.5.24/A.5.28 on deny (`:25-41`), buffered flush @100 records or 1 s timer
(`:105,181-193`), retention cleanup (`:210-230`), query filters (`:112-149`).

The Merkle layer implements what the governance README promises
(`README.md:178`): every flushed batch gets a SHA-256 Merkle root over its
record lines, chained to the previous root, persisted as
`audit/YYYY-MM-DD.merkle.jsonl`. The same tree shape (pairwise, odd leaf
duplicated) is computed by the GPU kernel in `csrc/sha256_merkle.hip` for
the batched firewall path; `merkle_root()` here is its CPU reference.
"""

from __future__ import annotations

import datetime as _dt
import hashlib
import json
import os
import threading
import time
import uuid
from typing import Any, Dict, List, Optional, Sequence

from ..core.api import PluginLogger, NullLogger
from .audit_redactor import create_redactor


def sha256_hex(data: bytes) -> str:
    return hashlib.sha256(data).hexdigest()


def merkle_root(leaves: Sequence[bytes]) -> str:
    """SHA-256 Merkle root: leaf = sha256(data); inner = sha256(l || r);
    odd node duplicated. Returns hex; empty input -> sha256(b"")."""
    if not leaves:
        return sha256_hex(b"")
    level = [hashlib.sha256(x).digest() for x in leaves]
    while len(level) > 1:
        nxt = []
        for i in range(0, len(level), 2):
            left = level[i]
            right = level[i + 1] if i + 1 < len(level) else left
            nxt.append(hashlib.sha256(left + right).digest())
        level = nxt
    return level[0].hex()


def derive_controls(matched_policies: List[Dict[str, Any]], verdict: str) -> List[str]:
    """Union of matched policies' controls; denials force A.5.24 + A.5.28
    (audit-trail.ts:25-41)."""
    controls = set()
    for mp in matched_policies:
        for c in mp.get("controls", []):
            controls.add(c)
    if verdict == "deny":
        controls.add("A.5.24")
        controls.add("A.5.28")
    return sorted(controls)


def _date_str(ts_ms: float) -> str:
    return _dt.datetime.fromtimestamp(ts_ms / 1000, _dt.timezone.utc).strftime("%Y-%m-%d")


class AuditTrail:
    FLUSH_THRESHOLD = 100

    def __init__(
        self,
        config: Optional[Dict[str, Any]] = None,
        workspace: str = ".",
        logger: Optional[PluginLogger] = None,
        clock=time.time,
    ):
        config = config or {}
        self.config = config
        self.audit_dir = os.path.join(workspace, "governance", "audit")
        self.logger = logger or NullLogger()
        self.clock = clock
        self.redact = create_redactor(config.get("redactPatterns"))
        self.retention_days = float(config.get("retentionDays", 90))
        self.merkle_enabled = bool(config.get("merkle", True))
        self.buffer: List[Dict[str, Any]] = []
        self._lock = threading.Lock()
        self._flush_timer: Optional[threading.Timer] = None
        self.today_record_count = 0
        self._prev_root: Optional[str] = None

    # -- lifecycle ---------------------------------------------------------
    def load(self) -> None:
        os.makedirs(self.audit_dir, exist_ok=True)
        self._clean_old_files()
        self._count_today_records()
        self._prev_root = self._load_last_root()

    def start_auto_flush(self) -> None:
        self._schedule_flush()

    def stop_auto_flush(self) -> None:
        with self._lock:
            if self._flush_timer is not None:
                self._flush_timer.cancel()
                self._flush_timer = None
        self.flush()

    def _schedule_flush(self) -> None:
        if self._flush_timer is not None:
            return
        self._flush_timer = threading.Timer(1.0, self._timer_flush)
        self._flush_timer.daemon = True
        self._flush_timer.start()

    def _timer_flush(self) -> None:
        with self._lock:
            self._flush_timer = None
        self.flush()
        with self._lock:
            if self.buffer:
                self._schedule_flush()

    # -- recording ---------------------------------------------------------
    def record(
        self,
        verdict: str,
        reason: str,
        context: Dict[str, Any],
        trust: Dict[str, Any],
        risk: Dict[str, Any],
        matched_policies: Optional[List[Dict[str, Any]]] = None,
        evaluation_us: int = 0,
    ) -> Dict[str, Any]:
        matched_policies = matched_policies or []
        now_ms = self.clock() * 1000
        rec = {
            "id": str(uuid.uuid4()),
            "timestamp": int(now_ms),
            "timestampIso": _dt.datetime.fromtimestamp(now_ms / 1000, _dt.timezone.utc)
            .isoformat()
            .replace("+00:00", "Z"),
            "verdict": verdict,
            "reason": reason,
            "context": self.redact(context),
            "trust": trust,
            "risk": risk,
            "matchedPolicies": matched_policies,
            "evaluationUs": evaluation_us,
            "controls": derive_controls(matched_policies, verdict),
        }
        do_flush = False
        with self._lock:
            self.buffer.append(rec)
            self.today_record_count += 1
            if len(self.buffer) >= self.FLUSH_THRESHOLD:
                do_flush = True
        if do_flush:
            self.flush()
        return rec

    def flush(self) -> None:
        with self._lock:
            batch = self.buffer
            self.buffer = []
        if not batch:
            return
        os.makedirs(self.audit_dir, exist_ok=True)
        groups: Dict[str, List[Dict[str, Any]]] = {}
        for rec in batch:
            groups.setdefault(_date_str(rec["timestamp"]), []).append(rec)
        for day, records in groups.items():
            lines = [json.dumps(r, separators=(",", ":"), ensure_ascii=False) for r in records]
            path = os.path.join(self.audit_dir, f"{day}.jsonl")
            with open(path, "a", encoding="utf-8") as fh:
                fh.write("\n".join(lines) + "\n")
            if self.merkle_enabled:
                self._append_merkle(day, lines)

    def _append_merkle(self, day: str, lines: List[str]) -> None:
        leaves = [ln.encode("utf-8") for ln in lines]
        root = merkle_root(leaves)
        chained = sha256_hex(((self._prev_root or "") + root).encode("ascii"))
        entry = {
            "ts": int(self.clock() * 1000),
            "count": len(leaves),
            "root": root,
            "prevRoot": self._prev_root,
            "chained": chained,
        }
        self._prev_root = chained
        path = os.path.join(self.audit_dir, f"{day}.merkle.jsonl")
        with open(path, "a", encoding="utf-8") as fh:
            fh.write(json.dumps(entry, separators=(",", ":")) + "\n")

    def _load_last_root(self) -> Optional[str]:
        if not os.path.isdir(self.audit_dir):
            return None
        files = sorted(f for f in os.listdir(self.audit_dir) if f.endswith(".merkle.jsonl"))
        if not files:
            return None
        last = None
        with open(os.path.join(self.audit_dir, files[-1]), "r", encoding="utf-8") as fh:
            for line in fh:
                line = line.strip()
                if line:
                    last = line
        if last is None:
            return None
        try:
            return json.loads(last).get("chained")
        except json.JSONDecodeError:
            return None

    def verify_merkle(self, day: str) -> bool:
        """Recompute batch roots for a day's JSONL against its merkle file."""
        jsonl = os.path.join(self.audit_dir, f"{day}.jsonl")
        mpath = os.path.join(self.audit_dir, f"{day}.merkle.jsonl")
        if not (os.path.isfile(jsonl) and os.path.isfile(mpath)):
            return False
        with open(jsonl, "r", encoding="utf-8") as fh:
            lines = [ln.rstrip("\n") for ln in fh if ln.strip()]
        with open(mpath, "r", encoding="utf-8") as fh:
            entries = [json.loads(ln) for ln in fh if ln.strip()]
        pos = 0
        for entry in entries:
            n = entry["count"]
            batch = lines[pos : pos + n]
            if len(batch) != n:
                return False
            if merkle_root([b.encode("utf-8") for b in batch]) != entry["root"]:
                return False
            pos += n
        return pos == len(lines)

    # -- querying ----------------------------------------------------------
    def query(self, filt: Optional[Dict[str, Any]] = None) -> List[Dict[str, Any]]:
        filt = filt or {}
        limit = int(filt.get("limit", 100))
        results: List[Dict[str, Any]] = []

        def matches(rec: Dict[str, Any]) -> bool:
            if filt.get("agentId") and (rec.get("context") or {}).get("agentId") != filt["agentId"]:
                return False
            if filt.get("verdict") and rec.get("verdict") != filt["verdict"]:
                return False
            if filt.get("after") and rec.get("timestamp", 0) < filt["after"]:
                return False
            if filt.get("before") and rec.get("timestamp", 0) > filt["before"]:
                return False
            return True

        files: List[str] = []
        if os.path.isdir(self.audit_dir):
            files = sorted(
                (f for f in os.listdir(self.audit_dir) if f.endswith(".jsonl") and ".merkle." not in f),
                reverse=True,
            )
        for fname in files:
            with open(os.path.join(self.audit_dir, fname), "r", encoding="utf-8") as fh:
                recs = [json.loads(ln) for ln in fh if ln.strip()]
            for rec in reversed(recs):
                if matches(rec):
                    results.append(rec)
                    if len(results) >= limit:
                        return results
        with self._lock:
            buffered = list(self.buffer)
        for rec in reversed(buffered):
            if matches(rec):
                results.append(rec)
                if len(results) >= limit:
                    return results
        return results

    def get_stats(self) -> Dict[str, Any]:
        files = []
        if os.path.isdir(self.audit_dir):
            files = sorted(
                f for f in os.listdir(self.audit_dir) if f.endswith(".jsonl") and ".merkle." not in f
            )
        return {
            "totalRecords": self.today_record_count,
            "todayRecords": self.today_record_count,
            "oldestRecord": files[0].replace(".jsonl", "") if files else None,
            "newestRecord": files[-1].replace(".jsonl", "") if files else None,
        }

    # -- maintenance -------------------------------------------------------
    def _clean_old_files(self) -> None:
        if not os.path.isdir(self.audit_dir):
            return
        cutoff = self.clock() * 1000 - self.retention_days * 86_400_000
        for fname in os.listdir(self.audit_dir):
            if not fname.endswith(".jsonl"):
                continue
            day = fname.split(".")[0]
            try:
                file_ms = (
                    _dt.datetime.strptime(day, "%Y-%m-%d").replace(tzinfo=_dt.timezone.utc).timestamp() * 1000
                )
            except ValueError:
                continue
            if file_ms < cutoff:
                try:
                    os.unlink(os.path.join(self.audit_dir, fname))
                except OSError:
                    pass

    def _count_today_records(self) -> None:
        today = _date_str(self.clock() * 1000)
        path = os.path.join(self.audit_dir, f"{today}.jsonl")
        if os.path.isfile(path):
            with open(path, "r", encoding="utf-8") as fh:
                self.today_record_count = sum(1 for ln in fh if ln.strip())
