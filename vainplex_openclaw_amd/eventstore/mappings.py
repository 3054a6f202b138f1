"""Data-driven hook -> event mapping table.

Parity target: reference `openclaw-nats-eventstore/src/hook-mappings.ts:33-209`
— 16 hook mappings + the run.failed extra emitter; after_tool_call maps to
tool.call.failed when the event carries an error; gateway hooks are system
events; llm_input/llm_output payloads are redacted to lengths/counts only.
"""

from __future__ import annotations

from typing import Callable, Dict, List, Optional


class HookMapping:
    def __init__(
        self,
        hook_name: str,
        event_type,  # str or callable(event, ctx) -> str
        mapper: Callable[[Dict, Dict], Dict],
        legacy_type: Optional[str] = None,
        visibility: Optional[str] = None,
        redaction: Optional[Dict] = None,
        system_event: bool = False,
    ):
        self.hook_name = hook_name
        self.event_type = event_type
        self.mapper = mapper
        self.legacy_type = legacy_type
        self.visibility = visibility
        self.redaction = redaction
        self.system_event = system_event

    def resolve_type(self, event: Dict, ctx: Dict) -> str:
        if callable(self.event_type):
            return self.event_type(event, ctx)
        return self.event_type


class ExtraEmitter(HookMapping):
    def __init__(self, *a, condition: Callable[[Dict], bool] = lambda e: True, **kw):
        super().__init__(*a, **kw)
        self.condition = condition


def _msg_received(e: Dict, c: Dict) -> Dict:
    return {
        "from": e.get("from"),
        "content": e.get("content"),
        "timestamp": e.get("timestamp"),
        "channel": c.get("channelId"),
        "metadata": e.get("metadata"),
    }


def _msg_sending(e: Dict, c: Dict) -> Dict:
    return {"to": e.get("to"), "content": e.get("content"), "channel": c.get("channelId")}


def _msg_sent(e: Dict, c: Dict) -> Dict:
    return {
        "to": e.get("to"),
        "content": e.get("content"),
        "success": e.get("success"),
        "error": e.get("error"),
        "channel": c.get("channelId"),
    }


def _llm_input(e: Dict, c: Dict) -> Dict:
    sp, p = e.get("systemPrompt"), e.get("prompt")
    hist = e.get("historyMessages")
    return {
        "runId": e.get("runId"),
        "sessionId": e.get("sessionId"),
        "provider": e.get("provider"),
        "model": e.get("model"),
        "systemPromptLength": len(sp) if isinstance(sp, str) else 0,
        "promptLength": len(p) if isinstance(p, str) else 0,
        "historyMessageCount": len(hist) if isinstance(hist, list) else 0,
        "imagesCount": e.get("imagesCount", 0),
    }


def _llm_output(e: Dict, c: Dict) -> Dict:
    texts = e.get("assistantTexts") if isinstance(e.get("assistantTexts"), list) else []
    return {
        "runId": e.get("runId"),
        "sessionId": e.get("sessionId"),
        "provider": e.get("provider"),
        "model": e.get("model"),
        "assistantTextCount": len(texts),
        "assistantTextTotalLength": sum(len(t) for t in texts if isinstance(t, str)),
        "usage": e.get("usage"),
    }


HOOK_MAPPINGS: List[HookMapping] = [
    HookMapping("message_received", "message.in.received", _msg_received,
                legacy_type="msg.in", visibility="confidential"),
    HookMapping("message_sending", "message.out.sending", _msg_sending,
                legacy_type="msg.sending", visibility="confidential"),
    HookMapping("message_sent", "message.out.sent", _msg_sent,
                legacy_type="msg.out", visibility="confidential"),
    HookMapping(
        "before_tool_call", "tool.call.requested",
        lambda e, c: {"toolName": e.get("toolName"), "params": e.get("params")},
        legacy_type="tool.call", visibility="confidential",
    ),
    HookMapping(
        "after_tool_call",
        lambda e, c: "tool.call.failed" if e.get("error") else "tool.call.executed",
        lambda e, c: {
            "toolName": e.get("toolName"),
            "params": e.get("params"),
            "result": e.get("result"),
            "error": e.get("error"),
            "durationMs": e.get("durationMs"),
        },
        legacy_type="tool.result", visibility="confidential",
    ),
    HookMapping(
        "before_agent_start", "run.started",
        lambda e, c: {"prompt": e.get("prompt")},
        legacy_type="run.start", visibility="confidential",
    ),
    HookMapping(
        "agent_end", "run.ended",
        lambda e, c: {
            "success": e.get("success"),
            "error": e.get("error"),
            "durationMs": e.get("durationMs"),
            "messageCount": len(e.get("messages")) if isinstance(e.get("messages"), list) else 0,
        },
        legacy_type="run.end",
    ),
    HookMapping(
        "llm_input", "model.input.observed", _llm_input, legacy_type="llm.input",
        redaction={"applied": True, "omittedFields": ["systemPrompt", "prompt", "historyMessages"]},
    ),
    HookMapping(
        "llm_output", "model.output.observed", _llm_output, legacy_type="llm.output",
        redaction={"applied": True, "omittedFields": ["assistantTexts"]},
    ),
    HookMapping(
        "before_compaction", "session.compaction.started",
        lambda e, c: {
            "messageCount": e.get("messageCount"),
            "compactingCount": e.get("compactingCount"),
            "tokenCount": e.get("tokenCount"),
        },
        legacy_type="session.compaction_start",
    ),
    HookMapping(
        "after_compaction", "session.compaction.ended",
        lambda e, c: {
            "messageCount": e.get("messageCount"),
            "compactedCount": e.get("compactedCount"),
            "tokenCount": e.get("tokenCount"),
        },
        legacy_type="session.compaction_end",
    ),
    HookMapping("before_reset", "session.reset", lambda e, c: {"reason": e.get("reason")}),
    HookMapping(
        "session_start", "session.started",
        lambda e, c: {"sessionId": e.get("sessionId"), "resumedFrom": e.get("resumedFrom")},
        legacy_type="session.start",
    ),
    HookMapping(
        "session_end", "session.ended",
        lambda e, c: {
            "sessionId": e.get("sessionId"),
            "messageCount": e.get("messageCount"),
            "durationMs": e.get("durationMs"),
        },
        legacy_type="session.end",
    ),
    HookMapping("gateway_start", "gateway.started", lambda e, c: {"port": e.get("port")},
                legacy_type="gateway.start", system_event=True),
    HookMapping("gateway_stop", "gateway.stopped", lambda e, c: {"reason": e.get("reason")},
                legacy_type="gateway.stop", system_event=True),
]

EXTRA_EMITTERS: List[ExtraEmitter] = [
    ExtraEmitter(
        "agent_end", "run.failed",
        lambda e, c: {"success": False, "error": e.get("error"), "durationMs": e.get("durationMs")},
        legacy_type="run.error",
        condition=lambda e: not e.get("success"),
    ),
]
