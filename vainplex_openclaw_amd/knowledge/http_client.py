"""Minimal JSON POST helper.

Parity target: reference `openclaw-knowledge-engine/src/http-client.ts`
(74 LoC) — a dependency-free POST wrapper with a timeout, JSON body and
response text; used by the LLM enhancer and the ChromaDB embeddings
sync. Injectable everywhere so tests (and this offline environment)
never touch the network.
"""

from __future__ import annotations

import json
import urllib.error
import urllib.request
from typing import Any, Dict, Optional


class HttpError(RuntimeError):
    def __init__(self, status: int, body: str):
        super().__init__(f"HTTP {status}: {body[:200]}")
        self.status = status
        self.body = body


def http_post(
    url: str,
    body: Dict[str, Any],
    headers: Optional[Dict[str, str]] = None,
    timeout_s: float = 30.0,
) -> str:
    """POST JSON; returns the response text; raises HttpError on non-2xx."""
    data = json.dumps(body).encode("utf-8")
    req = urllib.request.Request(
        url, data=data,
        headers={"Content-Type": "application/json", **(headers or {})},
        method="POST",
    )
    try:
        with urllib.request.urlopen(req, timeout=timeout_s) as resp:
            return resp.read().decode("utf-8", "replace")
    except urllib.error.HTTPError as exc:
        raise HttpError(exc.code, exc.read().decode("utf-8", "replace")) from exc
