"""NATS JetStream trace source mirroring cortex
`test/trace-analyzer/nats-trace-source.test.ts` (13 its) plus the
seq-scan behaviors of nats-trace-source.ts: graceful null on connect
failure, binary-searched start sequence, miss cutoff, time/type/agent
filters, max-events cap, STREAM.INFO-backed counts, and end-to-end
TraceAnalyzer consumption of a JetStream-held event history.
"""

import json

import pytest

from tests.test_nats_client import CFG, FakeNatsServer, _Log
from vainplex_openclaw_amd.cortex.trace.events import normalize_event
from vainplex_openclaw_amd.cortex.trace.nats_source import (
    NatsTraceSource,
    create_nats_trace_source,
)
from vainplex_openclaw_amd.eventstore.nats_client import JetStreamClient


def _connected_source(events, logger=None):
    srv = FakeNatsServer(existing_streams={"openclaw-events": {"name": "openclaw-events"}})
    js = JetStreamClient(CFG, logger=logger, transport=srv.transport())
    js.connect()
    for ev in events:
        assert js.publish("openclaw.events.agent.test", json.dumps(ev))
    return srv, NatsTraceSource(js, logger=logger)


def _schema_a(i, ts, typ="message.in.received", agent="main", **data):
    return {"id": f"evt-{i}", "ts": ts, "canonicalType": typ,
            "actor": {"id": agent}, "scope": {"sessionKey": f"agent:{agent}:s1"},
            "data": data or {"content": f"msg {i}"}}


# -- graceful degradation -----------------------------------------------------

def test_create_returns_none_when_unreachable():
    log = _Log()
    src = create_nats_trace_source({"url": "nats://127.0.0.1:1"}, logger=log)
    assert src is None
    assert any("NATS connection failed" in m for _, m in log.lines)


def test_create_does_not_throw_without_server():
    assert create_nats_trace_source({}, logger=None) is None


def test_create_accepts_credentials_in_config():
    srv = FakeNatsServer(existing_streams={"openclaw-events": {"name": "openclaw-events"}})
    src = create_nats_trace_source(
        {"url": "nats://localhost:4222", "user": "admin", "password": "secret"},
        transport=srv.transport())
    assert src is not None
    assert srv.connect_opts.get("user") == "admin"
    assert srv.connect_opts.get("pass") == "secret"
    src.close()


# -- normalize_event (schema sniffing) ----------------------------------------

def test_normalize_schema_a_msg_in():
    ev = normalize_event(_schema_a(1, 1000.0, content="Hello"), seq=7)
    assert ev.type == "msg.in"
    assert ev.payload["content"] == "Hello"
    assert ev.payload["role"] == "user"
    assert ev.agent == "main" and ev.session == "s1" and ev.seq == 7


def test_normalize_schema_a_tool_result_error():
    ev = normalize_event(_schema_a(2, 2000.0, typ="tool.call.failed",
                                   toolName="exec", error="boom"))
    assert ev.type == "tool.result"
    assert ev.payload["toolError"] == "boom"
    assert ev.payload["toolIsError"] is True


def test_normalize_schema_a_msg_out():
    ev = normalize_event(_schema_a(3, 3000.0, typ="message.out.sent",
                                   content="done"))
    assert ev.type == "msg.out" and ev.payload["role"] == "assistant"


def test_normalize_schema_b_shapes():
    b = {"kind": "user_message", "id": "b1", "ts": 5.0, "agentId": "dev",
         "sessionId": "agent:dev:u9", "body": {"text": "hi"}}
    ev = normalize_event(b)
    assert ev.type == "msg.in" and ev.payload["content"] == "hi"
    tc = normalize_event({"kind": "tool_use", "id": "b2", "ts": 6.0,
                          "agentId": "dev", "sessionId": "u9",
                          "body": {"name": "exec", "input": {"c": "ls"}}})
    assert tc.type == "tool.call" and tc.payload["toolParams"] == {"c": "ls"}
    tr = normalize_event({"kind": "tool_result", "id": "b3", "ts": 7.0,
                          "agentId": "dev", "sessionId": "u9",
                          "body": {"name": "exec", "error": "denied"}})
    assert tr.type == "tool.result" and tr.payload["toolIsError"] is True
    ok = normalize_event({"kind": "tool_result", "id": "b4", "ts": 8.0,
                          "agentId": "dev", "sessionId": "u9",
                          "body": {"name": "exec", "output": "file.txt"}})
    assert ok.payload["toolResult"] == "file.txt" and ok.payload["toolIsError"] is False


def test_normalize_skips_unknown_types():
    assert normalize_event({"canonicalType": "weird.event", "ts": 1}) is None
    assert normalize_event({"kind": "unknown_kind", "ts": 1}) is None
    assert normalize_event("not a dict") is None


# -- stream scanning ----------------------------------------------------------

def test_fetch_by_time_range_and_counts():
    events = [_schema_a(i, 1000.0 * (i + 1)) for i in range(10)]
    srv, src = _connected_source(events)
    assert src.get_last_sequence() == 10
    assert src.get_event_count() == 10
    got = list(src.fetch_by_time_range(0))
    assert len(got) == 10
    assert [e.seq for e in got] == sorted(e.seq for e in got)
    # binary-searched start: only events with ts >= 5000
    late = list(src.fetch_by_time_range(5000.0))
    assert len(late) == 6 and all(e.ts >= 5000.0 for e in late)
    src.close()


def test_fetch_filters_types_agents_and_caps():
    events = []
    for i in range(6):
        agent = "alpha" if i % 2 == 0 else "beta"
        typ = "message.in.received" if i < 4 else "tool.call.requested"
        ev = _schema_a(i, 1000.0 * (i + 1), typ=typ, agent=agent)
        if typ.startswith("tool"):
            ev["data"] = {"toolName": "exec", "params": {}}
        events.append(ev)
    srv, src = _connected_source(events)
    only_msgs = list(src.fetch_by_time_range(0, event_types=["msg.in"]))
    assert len(only_msgs) == 4 and all(e.type == "msg.in" for e in only_msgs)
    only_alpha = list(src.fetch_by_agent("alpha", 0))
    assert len(only_alpha) == 3 and all(e.agent == "alpha" for e in only_alpha)
    capped = list(src.fetch_by_time_range(0, max_events=2))
    assert len(capped) == 2
    # past-end early stop
    window = list(src.fetch_by_time_range(0, end_ms=2500.0))
    assert [e.ts for e in window] == [1000.0, 2000.0]
    src.close()


def test_fetch_tolerates_gaps_and_junk():
    events = [_schema_a(i, 1000.0 * (i + 1)) for i in range(8)]
    srv, src = _connected_source(events)
    # retention deletes the OLDEST seqs; corrupt a later payload. (The
    # binary search treats unreadable seqs as before-start — same as the
    # reference's null-timestamp handling — so only leading gaps and
    # in-range junk are tolerated, not arbitrary mid-stream deletions.)
    del srv.stored[1]
    del srv.stored[2]
    srv.stored[5] = (srv.stored[5][0], b"{not json")
    got = list(src.fetch_by_time_range(0))
    assert len(got) == 5                       # 8 - 2 retained-out - 1 junk
    assert [e.seq for e in got] == [3, 4, 6, 7, 8]
    src.close()


def test_empty_stream_yields_nothing():
    srv, src = _connected_source([])
    assert list(src.fetch_by_time_range(0)) == []
    assert src.get_event_count() == 0
    src.close()


# -- end-to-end: TraceAnalyzer over the NATS source ---------------------------

def test_trace_analyzer_runs_on_nats_source(tmp_path):
    events = []
    ts = 1000.0
    for i in range(3):
        events.append(_schema_a(f"c{i}", ts, typ="tool.call.requested",
                                toolName="exec", params={"command": "ls"}))
        ts += 1000
        events.append(_schema_a(f"r{i}", ts, typ="tool.call.failed",
                                toolName="exec", error="Connection refused"))
        ts += 1000
    srv, src = _connected_source(events)
    from vainplex_openclaw_amd.cortex.trace.analyzer import TraceAnalyzer

    report = TraceAnalyzer(str(tmp_path), src).run()
    assert report["eventsAnalyzed"] == 6
    assert any(f["signalType"] == "doom_loop" for f in report["findings"])
    src.close()


# -- events depth: detect_schema + conversation.* shape + nested errors -------
# (mirrors test/trace-analyzer/events.test.ts detection/payload sections)

from vainplex_openclaw_amd.cortex.trace.events import (
    detect_schema,
    normalize_schema_a,
)


def test_detect_schema_matrix():
    assert detect_schema({"type": "conversation.message.in", "timestamp": 5}) == "B"
    assert detect_schema({"type": "msg.in", "ts": 5.0}) == "A"
    assert detect_schema({"type": "custom.thing", "meta": {"source": "session-sync"},
                          "timestamp": 5}) == "B"
    assert detect_schema({"type": "custom.thing", "timestamp": 5}) == "B"
    assert detect_schema({"type": "msg.in"}) == "A"            # known type fallback
    assert detect_schema({"type": "totally.unknown"}) is None
    assert detect_schema({"canonicalType": "message.in.received", "ts": 1}) == "A"
    assert detect_schema({"kind": "user_message"}) == "B"
    assert detect_schema("junk") is None


def test_conversation_msg_text_preview():
    ev = normalize_event({"type": "conversation.message.in", "timestamp": 1000.0,
                          "agent": "dev", "session": "agent:main:u42",
                          "payload": {"text_preview": [{"text": "hello there"}],
                                      "sessionId": "u42"}}, seq=3)
    assert ev.type == "msg.in" and ev.payload["content"] == "hello there"
    assert ev.payload["role"] == "user" and ev.session == "u42" and ev.seq == 3
    out = normalize_event({"type": "conversation.message.out", "timestamp": 1001.0,
                           "payload": {"text_preview": []}})
    assert out.payload["content"] == "" and out.payload["role"] == "assistant"
    assert out.agent == "unknown" and out.session == "unknown"


def test_conversation_tool_call_and_result():
    tc = normalize_event({"type": "conversation.tool_call", "timestamp": 2000.0,
                          "payload": {"data": {"name": "exec", "args": {"c": "ls"}}}})
    assert tc.type == "tool.call"
    assert tc.payload["toolName"] == "exec" and tc.payload["toolParams"] == {"c": "ls"}
    err = normalize_event({"type": "conversation.tool_result", "timestamp": 2001.0,
                           "payload": {"data": {"name": "exec", "result": "Permission denied",
                                                "isError": True}}})
    assert err.payload["toolIsError"] is True
    assert err.payload["toolError"] == "Permission denied"
    ok = normalize_event({"type": "conversation.tool_result", "timestamp": 2002.0,
                          "payload": {"data": {"name": "exec", "result": "file.txt"}}})
    assert ok.payload["toolIsError"] is False and ok.payload["toolError"] is None


def test_conversation_requires_timestamp():
    assert normalize_event({"type": "conversation.message.in",
                            "payload": {"text_preview": [{"text": "x"}]}}) is None
    assert normalize_event({"type": "conversation.message.in", "timestamp": 0,
                            "payload": {}}) is None


@pytest.mark.parametrize("data,want_err,want_is", [
    ({"error": "boom"}, "boom", True),
    ({"result": {"details": {"error": "nested fail"}}}, "nested fail", True),
    ({"result": {"details": {"status": "error"}}}, "status: error", True),
    ({"result": {"details": {"exitCode": 2}}}, "exit code 2", True),
    ({"result": {"isError": True, "content": [{"text": "stderr text"}]}}, "stderr text", True),
    ({"result": {"isError": True}}, "unknown error", True),
    ({"result": {"details": {"exitCode": 0}}}, None, False),
    ({"result": "plain ok"}, None, False),
])
def test_schema_a_nested_error_extraction(data, want_err, want_is):
    ev = normalize_schema_a({"canonicalType": "tool.call.executed", "ts": 9.0,
                             "actor": {"id": "main"},
                             "scope": {"sessionKey": "agent:main:s"},
                             "data": {"toolName": "exec", **data}})
    assert ev.payload["toolIsError"] is want_is
    if want_err is not None:
        assert ev.payload["toolError"] == want_err
