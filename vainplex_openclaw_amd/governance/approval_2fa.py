"""TOTP-based human-in-the-loop 2FA approval.

Parity target: governance `src/approval-2fa.ts` — RFC-6238 TOTP
(SHA1 / 6 digits / 30 s, `:71-78`); a `2fa` verdict creates a pending
request batch (3 s debounce window); a notifier (Matrix in the reference,
pluggable here) is told; a valid 6-digit reply resolves the whole batch
(`:1-11`). Replay protection: each TOTP step is accepted once (`:53`);
per-session cooldown; 10-min session auto-approval after a success
(`:30-37`). TOTP itself is implemented with stdlib hmac — no external
otpauth dependency.
"""

from __future__ import annotations

import base64
import hashlib
import hmac
import secrets
import struct
import threading
import time
import uuid
from typing import Any, Callable, Dict, List, Optional

TOTP_PERIOD = 30
TOTP_DIGITS = 6
BATCH_DEBOUNCE_S = 3.0
SESSION_AUTO_APPROVE_S = 600.0
DEFAULT_COOLDOWN_S = 5.0
DEFAULT_TIMEOUT_S = 300.0


def generate_secret(nbytes: int = 20) -> str:
    """Base32 secret for authenticator apps."""
    return base64.b32encode(secrets.token_bytes(nbytes)).decode("ascii").rstrip("=")


def _b32decode(secret: str) -> bytes:
    pad = "=" * (-len(secret) % 8)
    return base64.b32decode(secret.upper() + pad)


def totp_at(secret: str, ts: float, period: int = TOTP_PERIOD, digits: int = TOTP_DIGITS) -> str:
    """RFC-6238 TOTP, HMAC-SHA1."""
    counter = int(ts // period)
    mac = hmac.new(_b32decode(secret), struct.pack(">Q", counter), hashlib.sha1).digest()
    offset = mac[-1] & 0x0F
    code = (struct.unpack(">I", mac[offset : offset + 4])[0] & 0x7FFFFFFF) % (10 ** digits)
    return str(code).zfill(digits)


def verify_totp(secret: str, code: str, ts: float, window: int = 1) -> Optional[int]:
    """Accept codes from the current step +/- `window` steps; returns the
    matched counter (for replay protection) or None."""
    if not code or not code.isdigit() or len(code) != TOTP_DIGITS:
        return None
    base_counter = int(ts // TOTP_PERIOD)
    for delta in range(-window, window + 1):
        counter = base_counter + delta
        if counter < 0:
            continue
        if hmac.compare_digest(totp_at(secret, counter * TOTP_PERIOD), code):
            return counter
    return None


class Approval2FA:
    """Pending-approval batches resolved by a TOTP code.

    `request()` returns a request dict whose "status" becomes
    approved/denied/expired; callers poll or pass a callback. Notification
    is via an injected `notify(batch)` callable (Matrix poller wires in)."""

    def __init__(
        self,
        secret: Optional[str] = None,
        notify: Optional[Callable[[Dict[str, Any]], None]] = None,
        clock=time.time,
        timeout_s: float = DEFAULT_TIMEOUT_S,
        cooldown_s: float = DEFAULT_COOLDOWN_S,
        max_attempts: int = 5,
    ):
        self.secret = secret or generate_secret()
        self.notify = notify
        self.clock = clock
        self.timeout_s = timeout_s
        self.cooldown_s = cooldown_s
        self.max_attempts = max_attempts
        self._failed_attempts = 0
        self._attempt_cooldown_until = 0.0
        self._lock = threading.Lock()
        self._pending: Dict[str, Dict[str, Any]] = {}  # batchId -> batch
        self._used_counters: set = set()
        self._session_approved_until: Dict[str, float] = {}
        self._last_denied_at: Dict[str, float] = {}

    # -- request side ------------------------------------------------------
    def request(
        self,
        session_key: str,
        agent_id: str,
        reason: str,
        details: Optional[Dict[str, Any]] = None,
    ) -> Dict[str, Any]:
        now = self.clock()
        # 10-min session auto-approval after a successful code
        if self._session_approved_until.get(session_key, 0) > now:
            return {
                "id": str(uuid.uuid4()),
                "status": "approved",
                "reason": reason,
                "auto": True,
            }
        # cooldown after a denial
        if now - self._last_denied_at.get(session_key, -1e18) < self.cooldown_s:
            return {"id": str(uuid.uuid4()), "status": "denied", "reason": "2FA cooldown active"}

        req = {
            "id": str(uuid.uuid4()),
            "sessionKey": session_key,
            "agentId": agent_id,
            "reason": reason,
            "details": details or {},
            "createdAt": now,
            "status": "pending",
        }
        with self._lock:
            # batch create is fully synchronous (reference comment,
            # approval-2fa.ts:~88-91): find an open batch in its debounce
            # window or open a new one.
            batch = None
            for b in self._pending.values():
                if b["sessionKey"] == session_key and now - b["createdAt"] <= BATCH_DEBOUNCE_S and b["open"]:
                    batch = b
                    break
            if batch is None:
                batch = {
                    "id": str(uuid.uuid4()),
                    "sessionKey": session_key,
                    "createdAt": now,
                    "requests": [],
                    "open": True,
                    "notified": False,
                }
                self._pending[batch["id"]] = batch
            batch["requests"].append(req)
        self._maybe_notify(batch)
        return req

    def _maybe_notify(self, batch: Dict[str, Any]) -> None:
        if self.notify is None or batch.get("notified"):
            return
        batch["notified"] = True
        try:
            self.notify(batch)
        except Exception:
            pass

    # -- resolution side ---------------------------------------------------
    def try_resolve_any(self, code: str, approve: bool = True) -> List[Dict[str, Any]]:
        """A valid TOTP code resolves ALL pending batches' requests."""
        now = self.clock()
        # rate limiting: too many bad codes trips a cooldown
        if now < self._attempt_cooldown_until:
            return []
        counter = verify_totp(self.secret, code, now)
        if counter is None:
            self._failed_attempts += 1
            if self._failed_attempts >= self.max_attempts:
                self._attempt_cooldown_until = now + self.cooldown_s
                self._failed_attempts = 0
            return []
        self._failed_attempts = 0
        with self._lock:
            if counter in self._used_counters:  # replay protection
                return []
            self._used_counters.add(counter)
            resolved: List[Dict[str, Any]] = []
            for batch in list(self._pending.values()):
                for req in batch["requests"]:
                    if req["status"] == "pending":
                        req["status"] = "approved" if approve else "denied"
                        resolved.append(req)
                if approve:
                    self._session_approved_until[batch["sessionKey"]] = now + SESSION_AUTO_APPROVE_S
                else:
                    self._last_denied_at[batch["sessionKey"]] = now
                del self._pending[batch["id"]]
        return resolved

    def expire_stale(self) -> List[Dict[str, Any]]:
        now = self.clock()
        expired: List[Dict[str, Any]] = []
        with self._lock:
            for bid, batch in list(self._pending.items()):
                if now - batch["createdAt"] > self.timeout_s:
                    for req in batch["requests"]:
                        if req["status"] == "pending":
                            req["status"] = "expired"
                            expired.append(req)
                    del self._pending[bid]
        return expired

    def in_attempt_cooldown(self) -> bool:
        return self.clock() < self._attempt_cooldown_until

    def has_pending_batch(self, session_key: Optional[str] = None) -> bool:
        with self._lock:
            return any(
                (session_key is None or b["sessionKey"] == session_key)
                and any(r["status"] == "pending" for r in b["requests"])
                for b in self._pending.values()
            )

    def pending_requests(self) -> List[Dict[str, Any]]:
        with self._lock:
            return [r for b in self._pending.values() for r in b["requests"] if r["status"] == "pending"]

    def provisioning_uri(self, account: str = "openclaw", issuer: str = "governance") -> str:
        return f"otpauth://totp/{issuer}:{account}?secret={self.secret}&issuer={issuer}&algorithm=SHA1&digits=6&period=30"
