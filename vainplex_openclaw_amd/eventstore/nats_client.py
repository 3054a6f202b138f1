"""NATS / JetStream wire-protocol client over an injectable transport.

Parity target: `openclaw-nats-eventstore/src/nats-client.ts:53-206` —
connect, ensure the JetStream stream exists (subjects `<prefix>.>`,
limits retention, max_age in NANOSECONDS), fire-and-forget publish that
never blocks agent operations (failure counters reset on success, warn
on the 1st and every 10th consecutive failure), status block, drain with
timeout falling back to force-close — plus `parseNatsUrl` credential
extraction with a safe log URL.

Unlike the reference (which rides the `nats` npm package), this speaks
the actual NATS protocol: INFO/CONNECT/PING/PONG, PUB/SUB/MSG parsing,
and the JetStream `$JS.API.STREAM.*` request-reply endpoints. The
transport is injected so the loopback tests drive a scripted in-memory
server; production wiring uses TcpTransport. There is no egress in this
environment, so TcpTransport is exercised only by its connect-failure
path; the protocol state machine is fully covered by the loopback tests
(tests/test_nats_client.py).
"""

from __future__ import annotations

import json
import threading
import time
from typing import Callable, Dict, Optional, Tuple
from urllib.parse import urlsplit

MAX_PUBLISH_FAILURES_BEFORE_WARN = 10


def parse_nats_url(url: str) -> Dict[str, Optional[str]]:
    """nats://user:pass@host:4222 -> servers/user/pass/safe_url
    (nats-client.ts:32-51)."""
    try:
        p = urlsplit(url)
        if not p.hostname:
            raise ValueError("no host")
        port = p.port or 4222
        return {
            "servers": f"{p.hostname}:{port}",
            "user": p.username or None,
            "pass": p.password or None,
            "safe_url": f"{p.scheme}://{p.hostname}:{port}",
        }
    except (ValueError, AttributeError):
        return {"servers": url, "user": None, "pass": None, "safe_url": url}


class TcpTransport:
    """Real-socket transport (production path; needs egress)."""

    def __init__(self):
        import socket

        self._socket_mod = socket
        self.sock = None

    def connect(self, host: str, port: int, timeout_s: float) -> None:
        self.sock = self._socket_mod.create_connection((host, port), timeout=timeout_s)
        self.sock.settimeout(timeout_s)

    def send(self, data: bytes) -> None:
        self.sock.sendall(data)

    def recv(self, n: int = 65536) -> bytes:
        return self.sock.recv(n)

    def close(self) -> None:
        if self.sock is not None:
            try:
                self.sock.close()
            finally:
                self.sock = None


class NatsError(Exception):
    pass


class NatsConnection:
    """Core protocol client: one reader thread, sid-keyed callbacks,
    request/reply over `_INBOX.` subjects."""

    def __init__(self, transport, logger=None, connect_timeout_s: float = 5.0,
                 name: str = "vainplex-openclaw-amd"):
        self.t = transport
        self.logger = logger
        self.connect_timeout_s = connect_timeout_s
        self.name = name
        self.server_info: Dict = {}
        self._buf = b""
        self._sid = 0
        self._subs: Dict[int, Callable[[str, Optional[str], bytes], None]] = {}
        self._lock = threading.Lock()
        self._pong = threading.Event()
        self._closed = False
        self._reader: Optional[threading.Thread] = None
        self.disconnects = 0

    # -- connection handshake ----------------------------------------------
    def connect(self, host: str = "127.0.0.1", port: int = 4222,
                user: Optional[str] = None, password: Optional[str] = None) -> None:
        self.t.connect(host, port, self.connect_timeout_s)
        line = self._read_line_blocking()
        if not line.startswith(b"INFO "):
            raise NatsError(f"expected INFO, got {line[:40]!r}")
        self.server_info = json.loads(line[5:].decode())
        opts = {
            "verbose": False, "pedantic": False, "lang": "py",
            "name": self.name, "version": "0.2.0", "protocol": 1,
            "headers": False,
        }
        if user:
            opts["user"] = user
            opts["pass"] = password or ""
        self.t.send(b"CONNECT " + json.dumps(opts).encode() + b"\r\nPING\r\n")
        self._reader = threading.Thread(target=self._read_loop, daemon=True)
        self._reader.start()
        if not self._pong.wait(self.connect_timeout_s):
            raise NatsError("no PONG after CONNECT")

    # -- protocol ops -------------------------------------------------------
    def publish(self, subject: str, payload: bytes, reply: Optional[str] = None) -> None:
        if self._closed:
            raise NatsError("connection closed")
        head = f"PUB {subject} {reply + ' ' if reply else ''}{len(payload)}\r\n"
        with self._lock:
            self.t.send(head.encode() + payload + b"\r\n")

    def subscribe(self, subject: str,
                  cb: Callable[[str, Optional[str], bytes], None]) -> int:
        self._sid += 1
        sid = self._sid
        self._subs[sid] = cb
        with self._lock:
            self.t.send(f"SUB {subject} {sid}\r\n".encode())
        return sid

    def unsubscribe(self, sid: int) -> None:
        self._subs.pop(sid, None)
        with self._lock:
            try:
                self.t.send(f"UNSUB {sid}\r\n".encode())
            except Exception:
                pass

    def request(self, subject: str, payload: bytes,
                timeout_s: float = 5.0) -> bytes:
        inbox = f"_INBOX.{id(self) & 0xFFFFFF}.{self._sid + 1}.{time.monotonic_ns() & 0xFFFFFF}"
        got = {}
        ev = threading.Event()

        def _cb(_subj, _reply, data):
            got["data"] = data
            ev.set()

        sid = self.subscribe(inbox, _cb)
        try:
            self.publish(subject, payload, reply=inbox)
            if not ev.wait(timeout_s):
                raise NatsError(f"request timeout on {subject}")
            return got["data"]
        finally:
            self.unsubscribe(sid)

    def flush(self, timeout_s: float = 5.0) -> None:
        """PING/PONG round trip: everything sent before is processed."""
        self._pong.clear()
        with self._lock:
            self.t.send(b"PING\r\n")
        if not self._pong.wait(timeout_s):
            raise NatsError("flush: no PONG")

    def close(self) -> None:
        self._closed = True
        try:
            self.t.close()
        except Exception:
            pass

    @property
    def is_closed(self) -> bool:
        return self._closed

    # -- reader -------------------------------------------------------------
    def _read_line_blocking(self) -> bytes:
        while b"\r\n" not in self._buf:
            chunk = self.t.recv()
            if not chunk:
                raise NatsError("connection closed during handshake")
            self._buf += chunk
        line, self._buf = self._buf.split(b"\r\n", 1)
        return line

    def _read_loop(self) -> None:
        try:
            while not self._closed:
                if b"\r\n" not in self._buf:
                    chunk = self.t.recv()
                    if not chunk:
                        break
                    self._buf += chunk
                    continue
                line, self._buf = self._buf.split(b"\r\n", 1)
                self._dispatch(line)
        except Exception:
            pass
        if not self._closed:
            self.disconnects += 1
            self._closed = True
            if self.logger:
                self.logger.warn(
                    f"[nats-eventstore] Disconnected ({self.disconnects} total)")

    def _dispatch(self, line: bytes) -> None:
        if line == b"PONG":
            self._pong.set()
            return
        if line == b"PING":
            with self._lock:
                self.t.send(b"PONG\r\n")
            return
        if line.startswith(b"MSG "):
            parts = line[4:].decode().split(" ")
            if len(parts) == 3:
                subject, sid_s, n_s = parts
                reply = None
            else:
                subject, sid_s, reply, n_s = parts
            n = int(n_s)
            while len(self._buf) < n + 2:
                chunk = self.t.recv()
                if not chunk:
                    raise NatsError("short read in MSG")
                self._buf += chunk
            payload, self._buf = self._buf[:n], self._buf[n + 2:]
            cb = self._subs.get(int(sid_s))
            if cb is not None:
                cb(subject, reply, payload)
            return
        if line.startswith(b"-ERR"):
            if self.logger:
                self.logger.error(f"[nats-eventstore] {line.decode(errors='replace')}")
            return
        # +OK / INFO updates: ignore


class JetStreamClient:
    """JetStream layer mirroring nats-client.ts createNatsClient."""

    def __init__(self, config: Dict, logger=None, transport=None,
                 transport_factory=None, clock=time.time):
        self.config = config
        self.logger = logger
        self.clock = clock
        # transport_factory produces a FRESH transport per (re)connect;
        # a plain injected transport is reused (loopback tests)
        self.transport_factory = transport_factory
        self.transport = transport if transport is not None else (
            transport_factory() if transport_factory is not None else TcpTransport())
        self.nc: Optional[NatsConnection] = None
        self.publish_failures = 0
        self.reconnect_attempts = 0
        self.max_reconnect_attempts = int(config.get("maxReconnectAttempts", 10))
        self.stream = config.get("streamName", "openclaw-events")
        self.subject_prefix = config.get("subjectPrefix", "openclaw.events")
        self.publish_timeout_s = float(config.get("publishTimeoutMs", 5000)) / 1000.0
        self.drain_timeout_s = float(config.get("drainTimeoutMs", 5000)) / 1000.0

    # -- lifecycle ----------------------------------------------------------
    def connect(self) -> None:
        parsed = parse_nats_url(self.config.get("natsUrl", "nats://localhost:4222"))
        host, _, port = parsed["servers"].partition(":")
        self.nc = NatsConnection(
            self.transport, logger=self.logger,
            connect_timeout_s=float(self.config.get("connectTimeoutMs", 5000)) / 1000.0,
        )
        self.nc.connect(host or "127.0.0.1", int(port or 4222),
                        user=parsed["user"], password=parsed["pass"])
        if self.logger:
            self.logger.info(f"[nats-eventstore] Connected to {parsed['safe_url']}")
        self.ensure_stream()

    def ensure_stream(self) -> None:
        """STREAM.INFO; on 404 STREAM.CREATE with subjects <prefix>.>,
        limits retention, max_age in ns (nats-client.ts:53-91)."""
        assert self.nc is not None
        info = self._js_request(f"STREAM.INFO.{self.stream}", b"")
        err = (info or {}).get("error")
        if not err:
            if self.logger:
                self.logger.debug(f'[nats-eventstore] Stream "{self.stream}" exists')
            return
        code = err.get("code")
        not_found = code in (404, "404") or "not found" in str(err.get("description", ""))
        if not not_found:
            raise NatsError(f"STREAM.INFO failed: {err}")
        retention = self.config.get("retention", {})
        max_age_h = float(retention.get("maxAgeHours", 0))
        stream_cfg = {
            "name": self.stream,
            "subjects": [f"{self.subject_prefix}.>"],
            "retention": "limits",
            "max_msgs": int(retention.get("maxMessages", -1)),
            "max_bytes": int(retention.get("maxBytes", -1)),
            "max_age": int(max_age_h * 3600 * 1_000_000_000) if max_age_h > 0 else 0,
        }
        created = self._js_request(f"STREAM.CREATE.{self.stream}",
                                   json.dumps(stream_cfg).encode())
        if (created or {}).get("error"):
            raise NatsError(f"STREAM.CREATE failed: {created['error']}")
        if self.logger:
            self.logger.info(f'[nats-eventstore] Created stream "{self.stream}"')

    def stream_info(self) -> Optional[Dict]:
        """$JS.API.STREAM.INFO — returns the info response (with its
        `state` block: first_seq/last_seq/messages) or None on error.
        Used by the trace-analyzer NATS source (nats-trace-source.ts)."""
        try:
            info = self._js_request(f"STREAM.INFO.{self.stream}", b"")
        except NatsError:
            return None
        return None if (info or {}).get("error") else info

    def get_message(self, seq: int) -> Optional[Dict]:
        """Stored message by stream sequence ($JS.API.STREAM.MSG.GET):
        {"seq", "subject", "data": bytes} or None (missing seq / error)."""
        try:
            resp = self._js_request(f"STREAM.MSG.GET.{self.stream}",
                                    json.dumps({"seq": int(seq)}).encode())
        except NatsError:
            return None
        msg = (resp or {}).get("message")
        if not isinstance(msg, dict):
            return None
        import base64

        try:
            data = base64.b64decode(msg.get("data") or "")
        except Exception:
            return None
        return {"seq": int(msg.get("seq", seq)),
                "subject": msg.get("subject", ""), "data": data}

    def _js_request(self, op: str, payload: bytes) -> Optional[Dict]:
        try:
            raw = self.nc.request(f"$JS.API.{op}", payload,
                                  timeout_s=self.publish_timeout_s)
            return json.loads(raw.decode())
        except NatsError:
            raise
        except Exception as exc:
            raise NatsError(f"bad $JS.API.{op} reply: {exc}")

    # -- publish (never blocks agent operations) ---------------------------
    def _try_reconnect(self) -> bool:
        """Bounded auto-reconnect (the reference connects with
        reconnect: true / maxReconnectAttempts: 10 — nats.js handles it
        in-library; this client re-dials on a dead connection). The
        attempt counter resets on a successful publish."""
        if self.reconnect_attempts >= self.max_reconnect_attempts:
            return False
        self.reconnect_attempts += 1
        try:
            if self.transport_factory is not None:
                self.transport = self.transport_factory()
            elif isinstance(self.transport, TcpTransport):
                self.transport = TcpTransport()
            self.connect()
            if self.logger:
                self.logger.info(
                    f"[nats-eventstore] Reconnected "
                    f"(attempt {self.reconnect_attempts})")
            return True
        except Exception as exc:
            if self.logger:
                self.logger.warn(
                    f"[nats-eventstore] Reconnect attempt "
                    f"{self.reconnect_attempts} failed: {exc}")
            return False

    def publish(self, subject: str, data: str) -> bool:
        """JetStream publish with ack; failures are swallowed, counted,
        and warn on the 1st + every 10th consecutive failure; the counter
        resets on success (nats-client.ts:147-177). A dead connection
        triggers one bounded reconnect attempt before the publish."""
        try:
            if self.nc is None or self.nc.is_closed:
                if not self._try_reconnect():
                    raise NatsError("not connected")
            ack_raw = self.nc.request(subject, data.encode(),
                                      timeout_s=self.publish_timeout_s)
            ack = json.loads(ack_raw.decode())
            if ack.get("error"):
                raise NatsError(f"publish nack: {ack['error']}")
            self.publish_failures = 0
            self.reconnect_attempts = 0
            return True
        except Exception as exc:
            self.publish_failures += 1
            if self.publish_failures == 1 or \
                    self.publish_failures % MAX_PUBLISH_FAILURES_BEFORE_WARN == 0:
                if self.logger:
                    self.logger.warn(
                        f"[nats-eventstore] Publish failed "
                        f"({self.publish_failures} consecutive): {exc}")
            return False

    # -- status / drain / close --------------------------------------------
    def is_connected(self) -> bool:
        return self.nc is not None and not self.nc.is_closed

    def get_status(self) -> Dict:
        return {
            "connected": self.is_connected(),
            "stream": self.stream,
            "disconnectCount": self.nc.disconnects if self.nc else 0,
            "publishFailures": self.publish_failures,
        }

    def drain(self) -> None:
        """Flush in-flight protocol; on timeout force-close
        (nats-client.ts:186-200)."""
        if self.nc is None:
            return
        try:
            self.nc.flush(timeout_s=self.drain_timeout_s)
        except NatsError:
            if self.logger:
                self.logger.warn("[nats-eventstore] Drain timed out, forcing close")
        finally:
            self.nc.close()

    def close(self) -> None:
        if self.nc is not None:
            self.nc.close()


class NatsPublishAdapter:
    """Journal-shaped facade over JetStreamClient so EventPublisher and
    the plugin status surface work identically against a live NATS
    server or the embedded journal (hooks.EventPublisher duck-type)."""

    def __init__(self, client: JetStreamClient):
        self.client = client

    def publish(self, subject: str, envelope: Dict) -> int:
        self.client.publish(subject, json.dumps(envelope, separators=(",", ":")))
        return 0

    def status(self) -> Dict:
        st = self.client.get_status()
        st.setdefault("messages", None)
        st.setdefault("lastSeq", None)
        return st

    def drain(self) -> None:
        self.client.drain()

    def close(self) -> None:
        self.client.close()
