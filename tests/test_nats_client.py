"""Loopback tests for the NATS wire-protocol client (no network).

A scripted in-memory NATS/JetStream server speaks the real wire
protocol (INFO/CONNECT/PING/PONG/PUB/SUB/MSG + $JS.API request-reply)
over an injectable duplex transport, covering the behaviors
nats-client.ts implements: ensure-stream (exists vs 404-create with
ns max_age), fire-and-forget publish with consecutive-failure counters
and warn cadence, status block, drain-with-timeout force close.
"""

import json
import os
import queue
import threading

import pytest

from vainplex_openclaw_amd.eventstore.nats_client import (
    JetStreamClient,
    NatsConnection,
    NatsError,
    parse_nats_url,
)


class DuplexTransport:
    """Client-side transport over two byte queues."""

    def __init__(self, c2s: "queue.Queue", s2c: "queue.Queue"):
        self.c2s = c2s
        self.s2c = s2c
        self.closed = False

    def connect(self, host, port, timeout_s):
        pass

    def send(self, data: bytes) -> None:
        self.c2s.put(data)

    def recv(self, n: int = 65536) -> bytes:
        try:
            return self.s2c.get(timeout=2.0)
        except queue.Empty:
            return b""

    def close(self) -> None:
        self.closed = True
        self.c2s.put(b"")  # wake the server


def _match(subject: str, pattern: str) -> bool:
    st, pt = subject.split("."), pattern.split(".")
    i = 0
    for j, p in enumerate(pt):
        if p == ">":
            return True
        if i >= len(st):
            return False
        if p != "*" and p != st[i]:
            return False
        i += 1
    return i == len(st)


class FakeNatsServer:
    """Protocol-complete-enough JetStream server for loopback tests."""

    def __init__(self, existing_streams=None, nack_publishes=False,
                 mute_acks=False):
        self.c2s = queue.Queue()
        self.s2c = queue.Queue()
        self.streams = dict(existing_streams or {})
        self.nack_publishes = nack_publishes
        self.mute_acks = mute_acks
        self.messages = []           # (subject, payload)
        self.stored = {}             # seq -> (subject, payload): stream storage
        self.subs = {}               # sid -> subject pattern
        self.created_configs = []
        self.connect_opts = None
        self._buf = b""
        self._seq = 0
        self.thread = threading.Thread(target=self._run, daemon=True)
        self.thread.start()

    def transport(self) -> DuplexTransport:
        return DuplexTransport(self.c2s, self.s2c)

    def _send(self, data: bytes) -> None:
        self.s2c.put(data)

    def _read_line(self):
        while b"\r\n" not in self._buf:
            chunk = self.c2s.get(timeout=5.0)
            if not chunk:
                return None
            self._buf += chunk
        line, self._buf = self._buf.split(b"\r\n", 1)
        return line

    def _read_n(self, n):
        while len(self._buf) < n + 2:
            chunk = self.c2s.get(timeout=5.0)
            if not chunk:
                return None
            self._buf += chunk
        payload, self._buf = self._buf[:n], self._buf[n + 2:]
        return payload

    def _run(self):
        self._send(b'INFO {"server_id":"fake","version":"2.12.0","jetstream":true,"max_payload":1048576}\r\n')
        try:
            while True:
                line = self._read_line()
                if line is None:
                    return
                if line.startswith(b"CONNECT "):
                    self.connect_opts = json.loads(line[8:].decode())
                elif line == b"PING":
                    self._send(b"PONG\r\n")
                elif line.startswith(b"SUB "):
                    parts = line[4:].decode().split(" ")
                    self.subs[int(parts[-1])] = parts[0]
                elif line.startswith(b"UNSUB "):
                    self.subs.pop(int(line[6:].decode().split(" ")[0]), None)
                elif line.startswith(b"PUB "):
                    parts = line[4:].decode().split(" ")
                    if len(parts) == 2:
                        subject, reply, n = parts[0], None, int(parts[1])
                    else:
                        subject, reply, n = parts[0], parts[1], int(parts[2])
                    payload = self._read_n(n)
                    if payload is None:
                        return
                    self._handle_pub(subject, reply, payload)
        except queue.Empty:
            return

    def _deliver(self, subject, payload, reply=None):
        for sid, pat in list(self.subs.items()):
            if _match(subject, pat):
                head = f"MSG {subject} {sid} {reply + ' ' if reply else ''}{len(payload)}\r\n"
                self._send(head.encode() + payload + b"\r\n")

    def _handle_pub(self, subject, reply, payload):
        if subject.startswith("$JS.API."):
            op = subject[len("$JS.API."):]
            if op.startswith("STREAM.INFO."):
                name = op[len("STREAM.INFO."):]
                if name in self.streams:
                    resp = {"type": "io.nats.jetstream.api.v1.stream_info_response",
                            "config": self.streams[name],
                            "state": {"first_seq": min(self.stored) if self.stored else 0,
                                      "last_seq": self._seq,
                                      "messages": len(self.stored)}}
                else:
                    resp = {"error": {"code": 404, "description": "stream not found"}}
            elif op.startswith("STREAM.MSG.GET."):
                import base64 as _b64
                q = json.loads(payload.decode())
                seq = int(q.get("seq", 0))
                if seq in self.stored:
                    subj, pl = self.stored[seq]
                    resp = {"message": {"seq": seq, "subject": subj,
                                        "data": _b64.b64encode(pl).decode()}}
                else:
                    resp = {"error": {"code": 404, "description": "no message found"}}
            elif op.startswith("STREAM.CREATE."):
                cfg = json.loads(payload.decode())
                self.streams[cfg["name"]] = cfg
                self.created_configs.append(cfg)
                resp = {"type": "io.nats.jetstream.api.v1.stream_create_response",
                        "config": cfg}
            else:
                resp = {"error": {"code": 400, "description": f"unknown op {op}"}}
            if reply:
                self._deliver(reply, json.dumps(resp).encode())
            return
        # regular JetStream publish
        self.messages.append((subject, payload))
        self._seq += 1
        self.stored[self._seq] = (subject, payload)
        self._deliver(subject, payload, reply)
        if reply and not self.mute_acks:
            if self.nack_publishes:
                ack = {"error": {"code": 503, "description": "no responders"}}
            else:
                ack = {"stream": "openclaw-events", "seq": self._seq}
            self._deliver(reply, json.dumps(ack).encode())


class _Log:
    def __init__(self):
        self.lines = []

    def _rec(self, lvl, msg):
        self.lines.append((lvl, msg))

    def info(self, m):
        self._rec("info", m)

    def warn(self, m):
        self._rec("warn", m)

    def error(self, m):
        self._rec("error", m)

    def debug(self, m):
        self._rec("debug", m)


CFG = {
    "natsUrl": "nats://localhost:4222",
    "streamName": "openclaw-events",
    "subjectPrefix": "openclaw.events",
    "retention": {"maxMessages": -1, "maxBytes": -1, "maxAgeHours": 2},
    "connectTimeoutMs": 2000,
    "publishTimeoutMs": 1000,
    "drainTimeoutMs": 1000,
}


def test_parse_nats_url():
    p = parse_nats_url("nats://user:secret@nats.example:5222")
    assert p == {"servers": "nats.example:5222", "user": "user",
                 "pass": "secret", "safe_url": "nats://nats.example:5222"}
    p2 = parse_nats_url("nats://host")
    assert p2["servers"] == "host:4222" and p2["user"] is None
    assert "secret" not in p["safe_url"]
    p3 = parse_nats_url("::notaurl::")
    assert p3["servers"] == "::notaurl::"


def test_connect_handshake_and_connect_opts():
    srv = FakeNatsServer()
    js = JetStreamClient(CFG, logger=_Log(), transport=srv.transport())
    js.connect()
    assert js.is_connected()
    assert srv.connect_opts["lang"] == "py"
    assert srv.connect_opts["protocol"] == 1
    assert js.nc.server_info["jetstream"] is True
    js.close()


def test_ensure_stream_creates_on_404_with_ns_max_age():
    srv = FakeNatsServer()
    log = _Log()
    js = JetStreamClient(CFG, logger=log, transport=srv.transport())
    js.connect()
    assert len(srv.created_configs) == 1
    cfg = srv.created_configs[0]
    assert cfg["name"] == "openclaw-events"
    assert cfg["subjects"] == ["openclaw.events.>"]
    assert cfg["retention"] == "limits"
    assert cfg["max_age"] == 2 * 3600 * 1_000_000_000  # hours -> ns
    assert any("Created stream" in m for _l, m in log.lines)
    js.close()


def test_ensure_stream_skips_existing():
    srv = FakeNatsServer(existing_streams={"openclaw-events": {"name": "openclaw-events"}})
    log = _Log()
    js = JetStreamClient(CFG, logger=log, transport=srv.transport())
    js.connect()
    assert srv.created_configs == []
    js.close()


def test_publish_ack_and_counter_reset():
    srv = FakeNatsServer()
    js = JetStreamClient(CFG, logger=_Log(), transport=srv.transport())
    js.connect()
    assert js.publish("openclaw.events.main.msg_in", '{"id":"evt-1"}')
    assert js.publish("openclaw.events.main.msg_in", '{"id":"evt-2"}')
    assert js.publish_failures == 0
    assert [s for s, _ in srv.messages] == ["openclaw.events.main.msg_in"] * 2
    assert json.loads(srv.messages[1][1].decode())["id"] == "evt-2"
    st = js.get_status()
    assert st["connected"] and st["stream"] == "openclaw-events"
    js.close()


def test_publish_failure_counting_and_warn_cadence():
    srv = FakeNatsServer(nack_publishes=True)
    log = _Log()
    js = JetStreamClient(CFG, logger=log, transport=srv.transport())
    js.connect()
    for _ in range(10):
        assert not js.publish("openclaw.events.main.msg_in", "{}")
    assert js.publish_failures == 10
    warns = [m for lvl, m in log.lines if lvl == "warn" and "Publish failed" in m]
    # warn on the 1st and the 10th consecutive failure only
    assert len(warns) == 2
    assert "(1 consecutive)" in warns[0] and "(10 consecutive)" in warns[1]
    # success resets
    srv.nack_publishes = False
    assert js.publish("openclaw.events.main.msg_in", "{}")
    assert js.publish_failures == 0
    js.close()


def test_publish_timeout_counts_as_failure():
    srv = FakeNatsServer(mute_acks=True)
    cfg = dict(CFG, publishTimeoutMs=150)
    js = JetStreamClient(cfg, logger=_Log(), transport=srv.transport())
    js.connect()
    assert not js.publish("openclaw.events.main.msg_in", "{}")
    assert js.publish_failures == 1
    js.close()


def test_subscribe_roundtrip_and_server_ping():
    srv = FakeNatsServer(existing_streams={"openclaw-events": {}})
    js = JetStreamClient(CFG, transport=srv.transport())
    js.connect()
    got = []
    ev = threading.Event()
    js.nc.subscribe("openclaw.events.*.msg_in",
                    lambda s, r, p: (got.append((s, p)), ev.set()))
    js.nc.flush()
    js.publish("openclaw.events.main.msg_in", '{"x":1}')
    assert ev.wait(2.0)
    assert got[0][0] == "openclaw.events.main.msg_in"
    assert json.loads(got[0][1].decode()) == {"x": 1}
    # server-initiated PING must be answered (keepalive)
    srv._send(b"PING\r\n")
    js.nc.flush()
    js.close()


def test_drain_flushes_then_closes():
    srv = FakeNatsServer(existing_streams={"openclaw-events": {}})
    js = JetStreamClient(CFG, logger=_Log(), transport=srv.transport())
    js.connect()
    js.drain()
    assert not js.is_connected()


def test_request_timeout_raises():
    srv = FakeNatsServer(existing_streams={"openclaw-events": {}})
    js = JetStreamClient(CFG, transport=srv.transport())
    js.connect()
    # an unknown $JS.API op still gets a (error-body) reply — no raise
    resp = json.loads(js.nc.request("$JS.API.NOPE.x", b"", timeout_s=1.0).decode())
    assert resp["error"]["code"] == 400
    js.close()
    # with acks muted, a request times out and raises
    srv2 = FakeNatsServer(existing_streams={"openclaw-events": {}}, mute_acks=True)
    js2 = JetStreamClient(CFG, transport=srv2.transport())
    js2.connect()
    with pytest.raises(NatsError):
        js2.nc.request("no.responder.here", b"", timeout_s=0.2)
    js2.close()


def test_tcp_transport_connect_failure_is_clean():
    from vainplex_openclaw_amd.eventstore.nats_client import TcpTransport

    t = TcpTransport()
    with pytest.raises(OSError):
        t.connect("127.0.0.1", 59999, 0.2)  # nothing listens here
    t.close()


def test_eventstore_plugin_over_live_nats_backend():
    """Full integration: hook event -> EventPublisher envelope -> NATS
    wire publish -> fake JetStream server stores it; /eventstatus and
    eventstore.status reflect the live client."""
    from vainplex_openclaw_amd.core.api import HookBus, NullLogger, PluginApi
    from vainplex_openclaw_amd.eventstore.plugin import EventStorePlugin

    srv = FakeNatsServer()
    bus = HookBus()
    api = PluginApi(id="nats-eventstore", plugin_config={"enabled": True},
                    logger=NullLogger(), config={}, bus=bus)
    plugin = EventStorePlugin(nats_transport=srv.transport())
    plugin.register(api)
    # the stream was ensured on the wire
    assert "openclaw-events" in srv.streams
    bus.emit("message_received", {"content": "hello", "ctx": {
        "agentId": "main", "sessionKey": "s1", "messageId": "m-1"}})
    deadline = threading.Event()
    for _ in range(100):
        if srv.messages:
            break
        deadline.wait(0.02)
    assert srv.messages, "no envelope reached the fake server"
    subject, payload = srv.messages[0]
    # agent resolution follows util.extract_agent_id (sessionKey-derived)
    assert subject.startswith("openclaw.events.") and subject.endswith(".msg_in")
    env = json.loads(payload.decode())
    assert env["schemaVersion"] == 1
    assert env["id"].startswith("evt-")
    st = api.gateway_methods["eventstore.status"]()
    assert st["connected"] is True and st["stream"] == "openclaw-events"


@pytest.mark.skipif(not os.environ.get("NATS_URL"),
                    reason="needs a live NATS server (set NATS_URL)")
def test_live_jetstream_round_trip():
    """Env-gated live-infrastructure test (the reference's only
    infra-gated suite: nats-eventstore integration.test.ts skipIf
    !NATS_URL): connect over real TCP, ensure the stream, publish an
    envelope, read it back by sequence."""
    import json as _json

    from vainplex_openclaw_amd.eventstore.nats_client import JetStreamClient

    cfg = dict(CFG)
    cfg["natsUrl"] = os.environ["NATS_URL"]
    cfg["streamName"] = "openclaw-events-test"
    js = JetStreamClient(cfg)
    js.connect()
    try:
        assert js.publish("openclaw.events.agent.live",
                          _json.dumps({"id": "live-1", "ts": 1.0}))
        info = js.stream_info()
        assert info and info["state"]["messages"] >= 1
        msg = js.get_message(int(info["state"]["last_seq"]))
        assert msg and _json.loads(msg["data"].decode())["id"] == "live-1"
    finally:
        js.close()


def test_auto_reconnect_on_dead_connection():
    """A dead connection triggers a bounded reconnect before publish;
    the attempt counter resets on success (reference maxReconnectAttempts)."""
    servers = []

    def factory():
        srv = FakeNatsServer(existing_streams={"openclaw-events":
                                               {"name": "openclaw-events"}})
        servers.append(srv)
        return srv.transport()

    js = JetStreamClient(CFG, transport_factory=factory)
    js.connect()
    assert js.publish("openclaw.events.a", "{}")
    js.nc.close()                                    # kill the connection
    assert js.publish("openclaw.events.b", "{}")     # reconnected + published
    assert len(servers) == 2
    assert servers[1].messages                       # landed on the NEW server
    assert js.reconnect_attempts == 0                # reset on success


def test_reconnect_attempts_bounded():
    cfg = dict(CFG)
    cfg["maxReconnectAttempts"] = 2

    calls = []

    def bad_factory():
        calls.append(1)
        raise ConnectionRefusedError("down")

    srv = FakeNatsServer(existing_streams={"openclaw-events":
                                           {"name": "openclaw-events"}})
    js = JetStreamClient(cfg, transport=srv.transport())
    js.connect()
    js.transport_factory = bad_factory
    js.nc.close()
    for _ in range(5):
        assert js.publish("openclaw.events.x", "{}") is False
    assert len(calls) == 2                           # capped at 2 attempts
    assert js.reconnect_attempts == 2
