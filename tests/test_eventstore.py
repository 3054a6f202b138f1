"""EventStore tests: subjects, envelope, deterministic ids, mappings,
journal (retention/replay/durability), hook registration, plugin surface.

Reference behaviors cited per test (openclaw-nats-eventstore).
"""

import json
import time

import pytest

from vainplex_openclaw_amd.core.api import HookBus, NullLogger, PluginApi
from vainplex_openclaw_amd.eventstore import (
    CANONICAL_EVENT_TYPES,
    LEGACY_EVENT_TYPES,
    EventJournal,
    EventPublisher,
    EventStorePlugin,
    build_envelope,
    build_subject,
    derive_event_id,
    extract_agent_id,
)
from vainplex_openclaw_amd.eventstore.config import resolve_config
from vainplex_openclaw_amd.eventstore.hooks import should_publish
from vainplex_openclaw_amd.eventstore.journal import _subject_match
from vainplex_openclaw_amd.eventstore.mappings import EXTRA_EMITTERS, HOOK_MAPPINGS


# -- util (util.ts) ----------------------------------------------------------

def test_extract_agent_id_priority():
    # agentId -> sessionKey first segment -> "main" (util.ts:1-15)
    assert extract_agent_id({"agentId": "viola"}) == "viola"
    assert extract_agent_id({"agentId": "main", "sessionKey": "viola:telegram:1"}) == "viola"
    assert extract_agent_id({"sessionKey": "main"}) == "main"
    assert extract_agent_id({}) == "main"


def test_build_subject_dots_to_underscores():
    assert build_subject("openclaw.events", "main", "msg.in") == "openclaw.events.main.msg_in"


# -- envelope (hooks.ts) -----------------------------------------------------

def test_derive_event_id_deterministic_with_stable_source():
    # evt- + sha256(session:type:stableSourceId)[:16] (hooks.ts:67-98)
    a = derive_event_id("run.started", "s1", {}, {"runId": "r-42"})
    b = derive_event_id("run.started", "s1", {}, {"runId": "r-42"})
    assert a == b and a.startswith("evt-") and len(a) == 4 + 16
    c = derive_event_id("run.started", "s2", {}, {"runId": "r-42"})
    assert c != a


def test_derive_event_id_uuid_without_stable_source():
    a = derive_event_id("gateway.started", "system", {}, {})
    b = derive_event_id("gateway.started", "system", {}, {})
    assert a != b and not a.startswith("evt-")


def test_envelope_schema_v1_fields():
    env = build_envelope(
        "message.in.received", "main", "main:tg:1",
        {"content": "hi", "messageId": "m1"},
        legacy_type="msg.in", visibility="confidential",
        ctx={"sessionKey": "main:tg:1", "senderId": "u9", "channelId": "tg"},
        clock=lambda: 1000.0,
    )
    assert env["schemaVersion"] == 1
    assert env["type"] == "msg.in" and env["canonicalType"] == "message.in.received"
    assert env["ts"] == 1_000_000
    assert env["source"] == {"plugin": "nats-eventstore"}
    assert env["actor"]["userId"] == "u9" and env["actor"]["channel"] == "tg"
    assert env["scope"]["messageId"] == "m1"
    assert env["trace"]["correlationId"] == "main:tg:1"
    assert env["visibility"] == "confidential"


def test_envelope_redaction_block_passthrough():
    env = build_envelope(
        "model.input.observed", "main", "s", {}, redaction={"applied": True, "omittedFields": ["prompt"]}
    )
    assert env["redaction"]["applied"] is True


# -- mappings (hook-mappings.ts) ---------------------------------------------

def test_all_16_hooks_mapped_and_types_canonical():
    hooks = [m.hook_name for m in HOOK_MAPPINGS]
    assert len(hooks) == 16 and len(set(hooks)) == 16
    for m in HOOK_MAPPINGS:
        t = m.resolve_type({}, {})
        assert t in CANONICAL_EVENT_TYPES
        if m.legacy_type:
            assert m.legacy_type in LEGACY_EVENT_TYPES


def test_after_tool_call_error_switches_type():
    m = next(m for m in HOOK_MAPPINGS if m.hook_name == "after_tool_call")
    assert m.resolve_type({"error": "boom"}, {}) == "tool.call.failed"
    assert m.resolve_type({}, {}) == "tool.call.executed"


def test_llm_mappers_redact_to_lengths():
    m = next(m for m in HOOK_MAPPINGS if m.hook_name == "llm_input")
    p = m.mapper({"systemPrompt": "abc", "prompt": "defg", "historyMessages": [1, 2]}, {})
    assert p["systemPromptLength"] == 3 and p["promptLength"] == 4
    assert p["historyMessageCount"] == 2 and "prompt" not in p
    mo = next(m for m in HOOK_MAPPINGS if m.hook_name == "llm_output")
    po = mo.mapper({"assistantTexts": ["ab", "c", 7]}, {})
    assert po["assistantTextCount"] == 3 and po["assistantTextTotalLength"] == 3


def test_run_failed_extra_emitter_condition():
    extra = EXTRA_EMITTERS[0]
    assert extra.hook_name == "agent_end"
    assert extra.condition({"success": False}) and not extra.condition({"success": True})


def test_should_publish_include_exclude():
    # include wins; exclude otherwise; default all (hooks.ts:42-50)
    assert should_publish("a", ["a"], ["a"]) is True
    assert should_publish("b", ["a"], []) is False
    assert should_publish("b", [], ["b"]) is False
    assert should_publish("b", [], []) is True


# -- journal -----------------------------------------------------------------

def test_subject_match_wildcards():
    assert _subject_match("p.e.main.msg_in", "p.e.>")
    assert _subject_match("p.e.main.msg_in", "p.e.*.msg_in")
    assert not _subject_match("p.e.main.msg_in", "p.e.other.msg_in")
    assert not _subject_match("p.e.main", "p.e.main.msg_in")


def test_journal_seq_replay_and_filter():
    j = EventJournal(durable=False)
    s1 = j.publish("p.main.msg_in", {"ts": 1000, "id": "a"})
    s2 = j.publish("p.viola.msg_in", {"ts": 2000, "id": "b"})
    assert (s1, s2) == (1, 2)
    got = list(j.replay())
    assert [e["id"] for _s, e in got] == ["a", "b"]
    assert [e["id"] for _s, e in j.replay(since_seq=1)] == ["b"]
    assert [e["id"] for _s, e in j.replay(subject_filter="p.viola.>")] == ["b"]
    assert [e["id"] for _s, e in j.replay(since_ts=1.5)] == ["b"]


def test_journal_retention_max_messages():
    j = EventJournal(durable=False, max_messages=2)
    for i in range(5):
        j.publish("s", {"ts": i * 1000, "id": str(i)})
    assert [e["id"] for _s, e in j.replay()] == ["3", "4"]
    assert j.last_seq == 5  # seq keeps counting past retention


def test_journal_durable_round_trip(tmp_path):
    d = str(tmp_path / "journal")
    j = EventJournal(directory=d, stream="openclaw-events")
    j.publish("p.main.msg_in", {"ts": int(time.time() * 1000), "id": "x"})
    j.close()
    j2 = EventJournal(directory=d, stream="openclaw-events")
    assert [e["id"] for _s, e in j2.replay()] == ["x"]
    assert j2.last_seq == 1
    j2.close()


def test_journal_status_counters():
    j = EventJournal(durable=False)
    j.publish("s", {"ts": 0})
    st = j.status()
    assert st["connected"] and st["stream"] == "openclaw-events"
    assert st["messages"] == 1 and st["publishFailures"] == 0


# -- hook registration end-to-end --------------------------------------------

def _api(bus):
    return PluginApi(id="nats-eventstore", plugin_config={}, logger=NullLogger(),
                     config={}, bus=bus)


def test_publisher_hook_to_journal_flow():
    bus = HookBus()
    api = _api(bus)
    j = EventJournal(durable=False)
    pub = EventPublisher(j, resolve_config({}), logger=NullLogger())
    pub.register(api)
    bus.emit("message_received", {
        "from": "u", "content": "hello",
        "ctx": {"agentId": "viola", "sessionKey": "viola:tg:1", "messageId": "m7"},
    })
    events = [e for _s, e in j.replay()]
    assert len(events) == 1
    env = events[0]
    assert env["canonicalType"] == "message.in.received"
    assert env["agent"] == "viola" and env["session"] == "viola:tg:1"
    assert env["payload"]["content"] == "hello"
    assert env["id"].startswith("evt-")  # messageId is a stable source id
    # subject scheme
    got = list(j.replay(subject_filter="openclaw.events.viola.msg_in"))
    assert len(got) == 1


def test_publisher_agent_end_failure_emits_run_failed():
    bus = HookBus()
    api = _api(bus)
    j = EventJournal(durable=False)
    EventPublisher(j, resolve_config({}), logger=NullLogger()).register(api)
    bus.emit("agent_end", {"success": False, "error": "x",
                           "ctx": {"sessionKey": "main", "runId": "r1"}})
    types = [e["canonicalType"] for _s, e in j.replay()]
    assert types == ["run.ended", "run.failed"]


def test_publisher_gateway_hooks_are_system_events():
    bus = HookBus()
    api = _api(bus)
    j = EventJournal(durable=False)
    EventPublisher(j, resolve_config({}), logger=NullLogger()).register(api)
    bus.emit("gateway_start", {"port": 8080})
    env = next(e for _s, e in j.replay())
    assert env["agent"] == "system" and env["session"] == "system"
    assert env["actor"]["agentId"] is None


def test_publisher_exclude_hooks():
    bus = HookBus()
    api = _api(bus)
    j = EventJournal(durable=False)
    cfg = resolve_config({"excludeHooks": ["message_received"]})
    EventPublisher(j, cfg, logger=NullLogger()).register(api)
    bus.emit("message_received", {"content": "x", "ctx": {"sessionKey": "s"}})
    assert len(j) == 0


# -- plugin surface ----------------------------------------------------------

def test_plugin_registers_commands_and_status(tmp_path, monkeypatch):
    monkeypatch.setenv("HOME", str(tmp_path))  # avoid real ~/.openclaw configs
    bus = HookBus()
    api = _api(bus)
    p = EventStorePlugin(journal=EventJournal(durable=False))
    p.register(api)
    assert "eventstatus" in api.commands
    assert "eventstore.status" in api.gateway_methods
    st = api.gateway_methods["eventstore.status"]()
    assert st["connected"] is True
    out = api.commands["eventstatus"]()
    assert "NATS Event Store" in out["text"]


def test_config_defaults_and_overrides():
    cfg = resolve_config({"streamName": "s2", "retention": {"maxMessages": 10}})
    assert cfg["streamName"] == "s2"
    assert cfg["retention"]["maxMessages"] == 10
    assert cfg["retention"]["maxBytes"] == -1
    assert cfg["natsUrl"] == "nats://localhost:4222"
    assert cfg["subjectPrefix"] == "openclaw.events"


# -- trace-analyzer integration ----------------------------------------------

def test_journal_feeds_trace_analyzer_source():
    from vainplex_openclaw_amd.cortex.trace.analyzer import JournalTraceSource

    bus = HookBus()
    api = _api(bus)
    j = EventJournal(durable=False)
    EventPublisher(j, resolve_config({}), logger=NullLogger()).register(api)
    bus.emit("before_tool_call", {"toolName": "exec", "params": {"cmd": "ls"},
                                  "ctx": {"sessionKey": "main", "toolCallId": "t1"}})
    bus.emit("after_tool_call", {"toolName": "exec", "error": "denied",
                                 "ctx": {"sessionKey": "main", "toolCallId": "t1"}})
    src = JournalTraceSource(j)
    events = src.fetch()
    assert len(events) >= 2
    assert any(e.type == "tool_result" or "tool" in e.type for e in events)


def test_journal_retention_bytes_and_age():
    """maxBytes + maxAgeHours retention (config.ts limits semantics)."""
    t = [1000.0]
    j = EventJournal(durable=False, max_bytes=400, clock=lambda: t[0])
    for i in range(20):
        j.publish("s", {"ts": t[0] * 1000, "pad": "x" * 20, "i": i})
    assert len(j) < 20  # byte cap evicted oldest
    kept = [e["i"] for _s, e in j.replay()]
    assert kept == sorted(kept) and kept[-1] == 19

    j2 = EventJournal(durable=False, max_age_hours=1, clock=lambda: t[0])
    j2.publish("s", {"ts": (t[0] - 7200) * 1000, "i": "old"})
    j2.publish("s", {"ts": t[0] * 1000, "i": "new"})
    j2.publish("s", {"ts": t[0] * 1000, "i": "new2"})  # triggers eviction pass
    ids = [e["i"] for _s, e in j2.replay()]
    assert "old" not in ids and "new" in ids


def test_journal_replay_limit_and_fetch_range():
    j = EventJournal(durable=False)
    for i in range(10):
        j.publish("s", {"ts": i * 1000.0, "i": i})
    assert len(list(j.replay(limit=3))) == 3
    got = j.fetch_range(2.0, 5.0)
    assert [e["i"] for e in got] == [2, 3, 4, 5]


def test_envelope_visibility_default_and_values():
    env = build_envelope("run.started", "a", "s", {})
    assert env["visibility"] == "internal"
    from vainplex_openclaw_amd.eventstore.events import VISIBILITIES

    assert VISIBILITIES == ("public", "internal", "confidential", "secret")


# -- hook-mappings.test.ts per-hook table ---------------------------------

HOOK_TABLE = [
    # (hook, event, canonical type, legacy type, payload key checks)
    ("message_received", {"content": "hi", "channel": "slack"},
     "message.in.received", "msg.in", {}),
    ("message_sending", {"content": "out"}, "message.out.sending", "msg.sending", {}),
    ("message_sent", {"content": "sent"}, "message.out.sent", "msg.out", {}),
    ("before_tool_call", {"toolName": "exec", "params": {"c": "ls"}},
     "tool.call.requested", "tool.call", {"toolName": "exec"}),
    ("after_tool_call", {"toolName": "exec", "result": "ok", "durationMs": 5},
     "tool.call.executed", "tool.result", {"durationMs": 5}),
    ("after_tool_call", {"toolName": "exec", "error": "boom"},
     "tool.call.failed", "tool.result", {"error": "boom"}),
    ("before_agent_start", {"prompt": "go"}, "run.started", "run.start",
     {"prompt": "go"}),
    ("agent_end", {"success": True, "durationMs": 9, "messages": [1, 2]},
     "run.ended", "run.end", {"messageCount": 2, "success": True}),
    ("llm_input", {"systemPrompt": "spsp", "prompt": "pp", "historyMessages": [1]},
     "model.input.observed", "llm.input", {}),
    ("llm_output", {"assistantTexts": ["aaaa", "bb"]},
     "model.output.observed", "llm.output", {}),
    ("before_compaction", {"messageCount": 10, "compactingCount": 4, "tokenCount": 99},
     "session.compaction.started", "session.compaction_start", {"messageCount": 10}),
    ("after_compaction", {"messageCount": 6, "compactedCount": 4, "tokenCount": 50},
     "session.compaction.ended", "session.compaction_end", {"compactedCount": 4}),
    ("before_reset", {"reason": "manual"}, "session.reset", None, {"reason": "manual"}),
    ("session_start", {"sessionId": "s1"}, "session.started", "session.start",
     {"sessionId": "s1"}),
    ("session_end", {"sessionId": "s1", "messageCount": 3, "durationMs": 7},
     "session.ended", "session.end", {"messageCount": 3}),
    ("gateway_start", {"port": 8080}, "gateway.started", "gateway.start", {"port": 8080}),
    ("gateway_stop", {"reason": "shutdown"}, "gateway.stopped", "gateway.stop", {}),
]


@pytest.mark.parametrize("hook,event,ctype,legacy,payload_checks", HOOK_TABLE,
                         ids=[f"{h}-{t}" for h, _, t, _, _ in HOOK_TABLE])
def test_hook_mapping_table(hook, event, ctype, legacy, payload_checks):
    mapping = next(m for m in HOOK_MAPPINGS
                   if m.hook_name == hook and m.resolve_type(event, {}) == ctype)
    assert mapping.legacy_type == legacy
    payload = mapping.mapper(event, {})
    for key, want in payload_checks.items():
        assert payload.get(key) == want, (key, payload)


def test_llm_mappings_privacy_lengths_only():
    m_in = next(m for m in HOOK_MAPPINGS if m.hook_name == "llm_input")
    p = m_in.mapper({"systemPrompt": "abcd", "prompt": "xy",
                      "historyMessages": [1, 2, 3]}, {})
    dumped = json.dumps(p)
    assert "abcd" not in dumped and "xy" not in dumped
    assert p.get("systemPromptLength") == 4 or p.get("systemPromptChars") == 4
    m_out = next(m for m in HOOK_MAPPINGS if m.hook_name == "llm_output")
    p2 = m_out.mapper({"assistantTexts": ["hello", "hi!"]}, {})
    assert "hello" not in json.dumps(p2)
    # graceful on missing assistantTexts
    p3 = m_out.mapper({}, {})
    assert isinstance(p3, dict)
    p4 = m_out.mapper({"assistantTexts": None}, {})
    assert isinstance(p4, dict)


def test_gateway_events_marked_system():
    for hook in ("gateway_start", "gateway_stop"):
        m = next(m for m in HOOK_MAPPINGS if m.hook_name == hook)
        assert m.system_event is True


def test_run_failed_only_on_failure():
    ee = EXTRA_EMITTERS[0]
    assert ee.hook_name == "agent_end" and ee.legacy_type == "run.error"
    assert ee.condition({"success": False})
    assert not ee.condition({"success": True})


# -- EventBlock batched publish (GPU-pipeline per-message envelope leg) -----

def test_journal_publish_block_replay_and_counts():
    import json

    from vainplex_openclaw_amd.eventstore import EventJournal

    j = EventJournal(durable=False)
    envs = [{"id": f"evt-{i}", "ts": 1000 + i, "type": "message.in.received",
             "payload": {"i": i}} for i in range(5)]
    blob = ("\n".join(json.dumps(e, separators=(",", ":")) for e in envs) + "\n").encode()
    first = j.publish_block("openclaw.events.bench.msg_in", blob, 5, ts_ms=1000)
    assert first == 1
    assert len(j) == 5
    assert j.last_seq == 5
    # a plain publish continues the sequence after the block
    seq = j.publish("openclaw.events.bench.other", {"id": "evt-x", "ts": 2000})
    assert seq == 6
    out = list(j.replay())
    assert [s for s, _ in out] == [1, 2, 3, 4, 5, 6]
    assert out[2][1]["payload"]["i"] == 2
    # since_seq threads into block interiors
    tail = list(j.replay(since_seq=3))
    assert [s for s, _ in tail] == [4, 5, 6]
    # subject filter applies to the whole block
    only = list(j.replay(subject_filter="openclaw.events.bench.msg_in"))
    assert len(only) == 5


def test_journal_block_retention_by_messages():
    import json

    from vainplex_openclaw_amd.eventstore import EventJournal

    j = EventJournal(durable=False, max_messages=7)
    for b in range(3):
        envs = [{"id": f"e{b}-{i}", "ts": b * 10 + i} for i in range(5)]
        blob = ("\n".join(json.dumps(e) for e in envs) + "\n").encode()
        j.publish_block("s.x", blob, 5)
    # 15 events with a 7-message cap -> oldest blocks dropped whole
    assert len(j) == 5
    assert [s for s, _ in j.replay()] == [11, 12, 13, 14, 15]


def test_journal_block_durable_segments(workspace):
    import json
    import os

    from vainplex_openclaw_amd.eventstore import EventJournal

    d = os.path.join(workspace, "jr")
    j = EventJournal(directory=d, durable=True)
    envs = [{"id": f"evt-{i}", "ts": 1757000000000 + i} for i in range(3)]
    blob = ("\n".join(json.dumps(e) for e in envs) + "\n").encode()
    j.publish_block("s.block", blob, 3, ts_ms=1757000000000)
    j.close()
    j2 = EventJournal(directory=d, durable=True)
    out = list(j2.replay())
    assert [s for s, _ in out] == [1, 2, 3]
    assert out[0][1]["id"] == "evt-0"
    j2.close()


def test_cpp_envelope_builder_parity():
    """csrc/host_envelope.cpp vs eventstore.hooks.build_envelope: same id
    derivation, same structural fields, for the audit-record inputs."""
    import json

    import numpy as np
    import torch

    from vainplex_openclaw_amd.eventstore.hooks import build_envelope
    from vainplex_openclaw_amd.ops import gpu as g

    if not g.have_ext():
        import pytest

        pytest.skip("extension not built")

    dt = np.dtype([("msg_id", "<u8"), ("inj", "<u8"), ("red", "<u8"),
                   ("risk", "<f4"), ("trust", "<f4"), ("agent", "<i4"),
                   ("verdict", "u1"), ("resv", "3u1"), ("ts", "<i8"),
                   ("injs", "<f4"), ("bseq", "<u4"), ("pad", "<u8")])
    assert dt.itemsize == 64
    rec = np.zeros(4, dtype=dt)
    rec["msg_id"] = [100, 101, 102, 103]
    rec["agent"] = [0, 3, 7, 63]
    rec["verdict"] = [0, 1, 2, 3]
    rec["risk"] = [5.25, 42.5, 77.0, 99.75]
    rec["trust"] = [40.0, 61.5, 20.25, 80.0]
    rec["ts"] = 1757000000123
    rec["injs"] = [0.0, 0.25, 0.5, 0.9375]
    rec["bseq"] = 9
    raw = torch.from_numpy(rec.view(np.uint8).reshape(4, 64).copy())
    blob = g.build_envelopes(raw, "sess-1", "agent", "message.in.received")
    lines = blob.decode().strip().split("\n")
    assert len(lines) == 4
    for i, line in enumerate(lines):
        got = json.loads(line)
        want = build_envelope(
            "message.in.received", f"agent{rec['agent'][i]}", "sess-1",
            payload={}, ctx={"messageId": f"msg-{rec['msg_id'][i]}",
                             "sessionKey": "sess-1"},
            clock=lambda: 0,
        )
        # deterministic id must match the Python derivation exactly
        assert got["id"] == want["id"], (got["id"], want["id"])
        assert got["schemaVersion"] == want["schemaVersion"] == 1
        assert got["source"] == want["source"]
        assert got["actor"]["agentId"] == f"agent{rec['agent'][i]}"
        assert got["scope"]["messageId"] == f"msg-{rec['msg_id'][i]}"
        assert got["visibility"] == "internal"
        assert set(got["trace"].keys()) == set(want["trace"].keys())
        assert got["payload"]["verdict"] == int(rec["verdict"][i])
        assert abs(got["payload"]["risk"] - float(rec["risk"][i])) < 1e-3
        assert got["ts"] == 1757000000123
