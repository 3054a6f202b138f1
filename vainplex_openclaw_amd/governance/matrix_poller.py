"""Matrix room poller for TOTP approval codes.

Parity target: governance `src/matrix-poller.ts` — polls a Matrix room via
the Client-Server API (`/sync` + `/rooms/<id>/messages`) for 6-digit codes
independent of any host Matrix integration, and feeds them to
`Approval2FA.try_resolve_any`. Network access is injected (`http_get`) so
the poller is fully testable offline; in production pass a real HTTP
client.
"""

from __future__ import annotations

import json
import re
import threading
import time
from typing import Any, Callable, Dict, List, Optional

from ..core.api import PluginLogger, NullLogger
from .approval_2fa import Approval2FA

CODE_RX = re.compile(r"\b(\d{6})\b")


class MatrixPoller:
    def __init__(
        self,
        approval: Approval2FA,
        homeserver: str = "https://matrix.org",
        room_id: str = "",
        access_token: str = "",
        poll_interval: float = 2.0,
        http_get: Optional[Callable[[str, Dict[str, str]], Dict[str, Any]]] = None,
        logger: Optional[PluginLogger] = None,
    ):
        self.approval = approval
        self.homeserver = homeserver.rstrip("/")
        self.room_id = room_id
        self.access_token = access_token
        self.poll_interval = poll_interval
        self.http_get = http_get
        self.logger = logger or NullLogger()
        self._since: Optional[str] = None
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None
        self.codes_seen = 0

    # -- core --------------------------------------------------------------
    def poll_once(self) -> List[str]:
        """Fetch new room messages, extract 6-digit codes, resolve pending
        2FA batches. Returns the codes found."""
        if self.http_get is None:
            return []
        url = (
            f"{self.homeserver}/_matrix/client/v3/rooms/{self.room_id}/messages"
            f"?dir=b&limit=10"
        )
        if self._since:
            url += f"&from={self._since}"
        try:
            resp = self.http_get(url, {"Authorization": f"Bearer {self.access_token}"})
        except Exception as exc:
            self.logger.warn("[2fa] Matrix poll failed: %s", exc)
            return []
        codes: List[str] = []
        for ev in (resp or {}).get("chunk", []):
            if ev.get("type") != "m.room.message":
                continue
            body = str((ev.get("content") or {}).get("body", ""))
            for m in CODE_RX.finditer(body):
                codes.append(m.group(1))
        token = (resp or {}).get("end")
        if token:
            self._since = token
        for code in codes:
            self.codes_seen += 1
            self.approval.try_resolve_any(code)
        return codes

    # -- thread ------------------------------------------------------------
    def start(self) -> None:
        if self._thread is not None or self.http_get is None:
            return
        self._stop.clear()

        def run() -> None:
            while not self._stop.wait(self.poll_interval):
                try:
                    self.poll_once()
                    self.approval.expire_stale()
                except Exception:
                    pass

        self._thread = threading.Thread(target=run, daemon=True)
        self._thread.start()

    def stop(self) -> None:
        self._stop.set()
        if self._thread is not None:
            self._thread.join(timeout=2.0)
            self._thread = None


def notify_via_matrix(
    homeserver: str,
    room_id: str,
    access_token: str,
    http_post: Optional[Callable[[str, Dict[str, str], Dict[str, Any]], Dict[str, Any]]] = None,
    logger: Optional[PluginLogger] = None,
) -> Callable[[Dict[str, Any]], None]:
    """Build an Approval2FA notifier that posts the batch summary to a
    Matrix room (hooks.ts:777-874 Matrix notify)."""
    log = logger or NullLogger()

    def notify(batch: Dict[str, Any]) -> None:
        if http_post is None:
            return
        lines = [f"2FA approval needed for session {batch['sessionKey']}:"]
        for req in batch["requests"]:
            lines.append(f"- {req['agentId']}: {req['reason']}")
        lines.append("Reply with your 6-digit code to approve.")
        url = (
            f"{homeserver.rstrip('/')}/_matrix/client/v3/rooms/{room_id}"
            f"/send/m.room.message"
        )
        try:
            http_post(
                url,
                {"Authorization": f"Bearer {access_token}"},
                {"msgtype": "m.text", "body": "\n".join(lines)},
            )
        except Exception as exc:
            log.warn("[2fa] Matrix notify failed: %s", exc)

    return notify
