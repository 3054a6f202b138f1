"""Governance hook wiring: the enforcement glue between bus and engine.

Parity target: governance `src/hooks.ts` — registration map with
priorities (`:883-919`: 1000 enforcement, 900 trust feedback, 5 context
injection); before_tool_call pipeline (`:166-243`); message_sending
(`:245-290`); before_message_write sync gate (`:297-389`); after_tool_call
trust feedback + toolCallLog + sub-agent spawn detection (`:391-440`);
external-comm detection (`:96-146`); /governance and /trust commands
(`:566-672`); gateway methods governance.status / governance.trust
(index.ts:104-115).
"""

from __future__ import annotations

from typing import Any, Dict, List, Optional

from ..core.api import PluginApi, PluginLogger, NullLogger
from .approval_2fa import Approval2FA
from .engine import GovernanceEngine
from .response_gate import ResponseGate
from .util import resolve_agent_id

TOOL_CALL_LOG_LIMIT = 50


def _extract_text_param(params: Dict[str, Any]) -> Optional[str]:
    for key in ("message", "text", "content", "body"):
        v = params.get(key)
        if isinstance(v, str) and v:
            return v
    return None


def detect_external_comm(ev: Dict[str, Any], config: Dict[str, Any]) -> Optional[str]:
    """Return the outbound text if this tool call is an external
    communication (hooks.ts:96-146)."""
    llm_cfg = ((config.get("outputValidation") or {}).get("llmValidator")) or {}
    if not llm_cfg.get("enabled"):
        return None
    channels = [c.lower() for c in llm_cfg.get("externalChannels", ["twitter", "linkedin", "email"])]
    commands = llm_cfg.get("externalCommands", ["bird tweet"])
    params = ev.get("params") or ev.get("toolParams") or {}
    tool = ev.get("toolName")
    if tool == "message":
        channel = params.get("channel")
        if isinstance(channel, str) and channel.lower() in channels:
            return _extract_text_param(params)
        target = params.get("target") or params.get("to")
        if params.get("action") == "send" and isinstance(target, str):
            if any(ch in target.lower() for ch in channels):
                return _extract_text_param(params)
    if tool == "exec" and isinstance(params.get("command"), str):
        cmd = params["command"]
        if any(pattern in cmd for pattern in commands):
            return cmd
    if tool == "sessions_send":
        message = params.get("message")
        if isinstance(message, str) and message:
            label = str(params.get("label") or params.get("sessionKey") or "")
            if any(ch in label.lower() for ch in channels):
                return message
    return None


class GovernanceHooks:
    """Holds the per-session toolCallLog and the handler closures."""

    def __init__(
        self,
        engine: GovernanceEngine,
        config: Dict[str, Any],
        approval: Optional[Approval2FA] = None,
        logger: Optional[PluginLogger] = None,
    ):
        self.engine = engine
        self.config = config
        self.approval = approval
        self.logger = logger or NullLogger()
        self.response_gate = ResponseGate(config.get("responseGate"))
        self.tool_call_log: Dict[str, List[Dict[str, str]]] = {}
        self._recent_agent_ctx: Dict[str, Dict[str, str]] = {}
        # ERC-8004 config lives top-level or nested under agentFirewall
        # (reference config.ts:205,330-334)
        self.erc8004_config = resolve_erc8004_config(config)
        self.erc8004_provider = None
        if self.erc8004_config.get("enabled"):
            from .security.erc8004 import ERC8004Provider

            self.erc8004_provider = ERC8004Provider()

    # -- handlers ----------------------------------------------------------
    def before_tool_call(self, ev: Dict[str, Any]) -> Optional[Dict[str, Any]]:
        try:
            agent_id = resolve_agent_id(ev)
            session = ev.get("sessionKey") or f"agent:{agent_id}"
            tool = ev.get("toolName")
            params = ev.get("params") or ev.get("toolParams") or {}
            self._recent_agent_ctx[str(tool)] = {"agentId": agent_id, "sessionId": session}
            ctx = self.engine.build_context(
                "before_tool_call", agent_id, session, tool_name=tool, tool_params=params,
                message_to=ev.get("messageTo"), channel=ev.get("channel"),
            )
            verdict = self.engine.evaluate(ctx)
            if verdict["action"] == "deny":
                return {"block": True, "blockReason": verdict["reason"], "verdict": verdict}
            if verdict["action"] == "2fa":
                if self.approval is None:
                    return {"block": True, "blockReason": "2FA required but no approver configured"}
                req = self.approval.request(session, agent_id, verdict["reason"], {"toolName": tool})
                if req["status"] == "approved":
                    return {"verdict": verdict, "approval": req}
                if req["status"] == "denied":
                    return {"block": True, "blockReason": req.get("reason", "2FA denied"), "approval": req}
                return {"block": True, "blockReason": "Awaiting 2FA approval", "pending2fa": req, "verdict": verdict}
            # external comm -> output validation (Stage 1+2 sync)
            external_text = detect_external_comm(ev, self.config)
            if external_text:
                result = self.engine.output_validator.validate(
                    external_text, self.engine.trust_manager.score(agent_id), is_external=True
                )
                if result["verdict"] == "block":
                    return {"block": True, "blockReason": result["reason"], "outputValidation": result}
                return {"verdict": verdict, "outputValidation": result}
            return {"verdict": verdict}
        except Exception as exc:
            self.logger.error("[governance] before_tool_call hook error: %s", exc)
            if self.config.get("failMode") == "closed":
                return {"block": True, "blockReason": "Governance hook error (fail-closed)"}
            return None

    def message_sending(self, ev: Dict[str, Any]) -> Optional[Dict[str, Any]]:
        agent_id = resolve_agent_id(ev)
        session = ev.get("sessionKey") or f"agent:{agent_id}"
        ctx = self.engine.build_context(
            "message_sending", agent_id, session,
            message_content=ev.get("content"), message_to=ev.get("to"), channel=ev.get("channel"),
        )
        verdict = self.engine.evaluate(ctx)
        if verdict["action"] == "deny":
            return {"block": True, "blockReason": verdict["reason"], "verdict": verdict}
        return {"verdict": verdict}

    def before_message_write(self, ev: Dict[str, Any]) -> Optional[Dict[str, Any]]:
        """Synchronous gate: output validation Stage 1+2 + Response Gate
        (hooks.ts:297-389)."""
        agent_id = resolve_agent_id(ev)
        session = ev.get("sessionKey") or f"agent:{agent_id}"
        content = str(ev.get("content") or "")
        if not content:
            return None
        log = self.tool_call_log.get(session, [])
        gate = self.response_gate.validate(content, agent_id, log)
        if not gate["passed"]:
            out = {"block": True, "blockReason": "; ".join(gate["reasons"]), "responseGate": gate}
            if gate.get("fallbackMessage"):
                out["fallbackMessage"] = gate["fallbackMessage"]
            return out
        result = self.engine.output_validator.validate(content, self.engine.trust_manager.score(agent_id))
        if result["verdict"] == "block":
            return {"block": True, "blockReason": result["reason"], "outputValidation": result}
        if result["verdict"] == "flag":
            return {"flagged": True, "outputValidation": result}
        return {"outputValidation": result}

    def after_tool_call(self, ev: Dict[str, Any]) -> None:
        try:
            agent_id = resolve_agent_id(ev)
            session = ev.get("sessionKey") or ev.get("sessionId") or f"agent:{agent_id}"
            tool = str(ev.get("toolName"))
            if agent_id == "unresolved":
                cached = self._recent_agent_ctx.get(tool)
                if cached:
                    agent_id, session = cached["agentId"], cached["sessionId"]
            success = not ev.get("error")
            self.engine.record_outcome(agent_id, session, success)
            result = ev.get("result")
            if success and result is not None:
                output = result if isinstance(result, str) else repr(result)
                log = self.tool_call_log.setdefault(session, [])
                log.append({"toolName": tool, "output": output})
                if len(log) > TOOL_CALL_LOG_LIMIT:
                    del log[: len(log) - TOOL_CALL_LOG_LIMIT]
            # sub-agent spawn detection (hooks.ts:424-435)
            if tool == "sessions_spawn" and success and isinstance(result, dict):
                child = result.get("sessionId") or result.get("sessionKey")
                if isinstance(child, str) and session:
                    self.engine.cross_agent.register_relationship(session, child)
        except Exception:
            pass

    def before_agent_start(self, ev: Dict[str, Any]) -> Optional[Dict[str, Any]]:
        """Context injection + ERC-8004 reputation log (hooks.ts:445-498)."""
        try:
            ctx = ev.get("ctx") or {}
            agent_id = ctx.get("agentId") or ev.get("agentId") or "main"
            session = ctx.get("sessionKey") or ctx.get("sessionId") or f"agent:{agent_id}"
            agent_rec = self.engine.trust_manager.get(agent_id)
            sess_rec = self.engine.session_trust.get(session, agent_id)
            cfg = self.erc8004_config
            if cfg.get("enabled") and self.erc8004_provider is not None:
                on_chain = cfg.get("agentMapping", {}).get(agent_id)
                if isinstance(on_chain, int):
                    try:
                        rep = self.erc8004_provider.lookup_reputation(on_chain)
                        if rep:
                            self.logger.info(
                                "[firewall] ERC-8004 reputation for agent %s: %s (score=%s)",
                                agent_id, rep.get("tier"), rep.get("reputationScore"),
                            )
                    except Exception as exc:  # fail-open
                        self.logger.warn("[firewall] ERC-8004 lookup failed: %s", exc)
            status = self.engine.status()
            context = (
                f"\n[Governance] Agent: {agent_id} "
                f"({agent_rec.get('score')}/{agent_rec.get('tier')}) | "
                f"Session: {sess_rec.get('score')}/{sess_rec.get('tier')} | "
                f"Policies: {len(status.get('policies', []))}"
            )
            return {"prependContext": context}
        except Exception as exc:
            self.logger.error("[governance] Error in before_agent_start: %s", exc)
            return None

    def session_start(self, ev: Dict[str, Any]) -> None:
        agent_id = resolve_agent_id(ev)
        session = ev.get("sessionKey") or ev.get("sessionId") or f"agent:{agent_id}"
        self.engine.session_trust.initialize(session, agent_id)

    def session_end(self, ev: Dict[str, Any]) -> None:
        session = ev.get("sessionKey") or ev.get("sessionId") or ""
        self.engine.session_trust.destroy(session)
        self.tool_call_log.pop(session, None)

    def totp_intercept(self, ev: Dict[str, Any]) -> Optional[Dict[str, Any]]:
        """message_received: a bare 6-digit message resolves pending 2FA
        (hooks.ts:677-731)."""
        if self.approval is None:
            return None
        content = str(ev.get("content") or "").strip()
        if len(content) == 6 and content.isdigit():
            resolved = self.approval.try_resolve_any(content)
            if resolved:
                return {"handled": True, "resolved2fa": [r["id"] for r in resolved]}
        return None

    # -- commands ----------------------------------------------------------
    def governance_command(self, *args: Any) -> Dict[str, Any]:
        return self.engine.status()

    def trust_command(self, agent_id: Optional[str] = None, *args: Any) -> Dict[str, Any]:
        if agent_id:
            return {agent_id: self.engine.trust_manager.get(agent_id)}
        return self.engine.trust_manager.snapshot()


def resolve_erc8004_config(config: Dict[str, Any]) -> Dict[str, Any]:
    """Top-level `erc8004` or nested `agentFirewall.erc8004`
    (reference config.ts:204-217, 330-334)."""
    af = config.get("agentFirewall")
    raw = None
    if isinstance(af, dict) and isinstance(af.get("erc8004"), dict):
        raw = af["erc8004"]
    elif isinstance(config.get("erc8004"), dict):
        raw = config["erc8004"]
    if not isinstance(raw, dict):
        return {"enabled": False, "agentMapping": {}}
    return {
        "enabled": bool(raw.get("enabled", False)),
        "rpcUrl": raw.get("rpcUrl"),
        "identityRegistryAddress": raw.get("identityRegistryAddress"),
        "agentMapping": raw.get("agentMapping") if isinstance(raw.get("agentMapping"), dict) else {},
    }


def register_governance_hooks(
    api: PluginApi,
    engine: GovernanceEngine,
    config: Dict[str, Any],
    approval: Optional[Approval2FA] = None,
) -> GovernanceHooks:
    h = GovernanceHooks(engine, config, approval, api.logger)
    api.on("before_tool_call", h.before_tool_call, priority=1000)
    api.on("message_sending", h.message_sending, priority=1000)
    api.on("before_message_write", h.before_message_write, priority=1000)
    api.on("after_tool_call", h.after_tool_call, priority=900)
    api.on("message_received", h.totp_intercept, priority=1000)
    api.on("before_agent_start", h.before_agent_start, priority=5)
    api.on("session_start", h.session_start, priority=1)
    api.on("session_end", h.session_end, priority=999)
    api.on("gateway_start", lambda ev: None, priority=1)
    api.on("gateway_stop", lambda ev: engine.audit_trail.flush(), priority=999)
    api.register_command("governance", h.governance_command)
    api.register_command("trust", h.trust_command)
    api.register_gateway_method("governance.status", h.governance_command)
    api.register_gateway_method("governance.trust", h.trust_command)
    return h
