"""GovernanceEngine: the per-call evaluation pipeline orchestrator.

Parity target: governance `src/engine.ts:31-545` — composes
PolicyEvaluator, RiskAssessor, TrustManager, SessionTrustManager,
CrossAgentManager, AuditTrail, OutputValidator, FrequencyTracker.
Pipeline (`:189-267`): enrich ctx -> record frequency -> assess risk ->
resolve effective policies -> evaluate -> trust learning on deny
(skipping `builtin-night-mode` denials, `:248-263`) -> audit.
Fail-open/closed on error per config.failMode (`:301-350`).
Stats with running average evaluation microseconds (`:535-544`).
"""

from __future__ import annotations

import datetime as _dt
import time
from typing import Any, Dict, List, Optional

from ..core.api import PluginLogger, NullLogger
from .audit import AuditTrail
from .conditions import ConditionDeps
from .cross_agent import CrossAgentManager
from .evaluator import PolicyEvaluator
from .frequency import FrequencyTracker
from .policies import PolicyIndex, build_policy_index
from .risk import RiskAssessor
from .session_trust import SessionTrustConfig, SessionTrustManager
from .trust import TrustConfig, TrustManager
from .output_validator import OutputValidator


def _now_us() -> int:
    return int(time.perf_counter() * 1_000_000)


class GovernanceEngine:
    def __init__(
        self,
        config: Optional[Dict[str, Any]] = None,
        workspace: str = ".",
        logger: Optional[PluginLogger] = None,
        clock=time.time,
    ):
        self.config = config or {}
        self.workspace = workspace
        self.logger = logger or NullLogger()
        self.clock = clock

        self.fail_mode = self.config.get("failMode", "open")
        self.trust_enabled = bool((self.config.get("trust") or {}).get("enabled", True))
        self.audit_enabled = bool((self.config.get("audit") or {}).get("enabled", True))

        self.trust_manager = TrustManager(
            TrustConfig.from_dict(self.config.get("trust")), workspace, self.logger, clock=clock
        )
        # sessionTrust nests under trust (config.ts resolveTrust); accept
        # the top-level spelling as a fallback
        st_cfg = (self.config.get("trust") or {}).get("sessionTrust") or self.config.get("sessionTrust")
        self.session_trust = SessionTrustManager(
            SessionTrustConfig.from_dict(st_cfg), self.trust_manager, clock=clock
        )
        self.cross_agent = CrossAgentManager(self.trust_manager, clock=clock)
        self.frequency = FrequencyTracker(clock=clock)
        self.risk_assessor = RiskAssessor((self.config.get("risk") or {}).get("toolRiskOverrides"))
        self.policy_index: PolicyIndex = build_policy_index(
            self.config, self.config.get("policyDir")
        )
        self.evaluator = PolicyEvaluator()
        self.audit_trail = AuditTrail(self.config.get("audit"), workspace, self.logger, clock=clock)
        self.output_validator = OutputValidator(self.config.get("outputValidation"), logger=self.logger)

        self.known_agents: List[str] = []
        self.stats = {
            "evaluations": 0,
            "allows": 0,
            "denies": 0,
            "twofas": 0,
            "errorCount": 0,
            "avgEvaluationUs": 0.0,
        }

    # -- lifecycle ---------------------------------------------------------
    def start(self) -> None:
        self.trust_manager.start()
        self.audit_trail.load()
        self.audit_trail.start_auto_flush()

    def stop(self) -> None:
        self.audit_trail.stop_auto_flush()
        self.trust_manager.stop()

    def set_known_agents(self, agents: List[str]) -> None:
        self.known_agents = list(agents)
        for a in agents:
            self.trust_manager.get(a)

    # -- helpers -----------------------------------------------------------
    def _time_ctx(self) -> Dict[str, Any]:
        now = _dt.datetime.fromtimestamp(self.clock())
        return {
            "hour": now.hour,
            "minute": now.minute,
            "dayOfWeek": now.strftime("%a"),
        }

    def build_context(
        self,
        hook: str,
        agent_id: str,
        session_key: str = "",
        tool_name: Optional[str] = None,
        tool_params: Optional[Dict[str, Any]] = None,
        message_content: Optional[str] = None,
        message_to: Optional[str] = None,
        channel: Optional[str] = None,
        **extra: Any,
    ) -> Dict[str, Any]:
        agent = self.trust_manager.get(agent_id)
        session = self.session_trust.get(session_key or f"agent:{agent_id}", agent_id)
        ctx: Dict[str, Any] = {
            "hook": hook,
            "agentId": agent_id,
            "sessionKey": session_key or f"agent:{agent_id}",
            "toolName": tool_name,
            "toolParams": tool_params,
            "messageContent": message_content,
            "messageTo": message_to,
            "channel": channel,
            "time": self._time_ctx(),
            "trust": {
                "agent": {"score": agent["score"], "tier": agent["tier"]},
                "session": {"score": session["score"], "tier": session["tier"]},
            },
        }
        ctx.update(extra)
        return ctx

    def _deps(self, risk: Dict[str, Any]) -> ConditionDeps:
        return ConditionDeps(
            regex_cache=self.policy_index.regex_cache,
            time_windows=self.config.get("timeWindows") or {},
            risk=risk,
            frequency_tracker=self.frequency,
        )

    # -- evaluation --------------------------------------------------------
    def evaluate(self, ctx: Dict[str, Any]) -> Dict[str, Any]:
        start_us = _now_us()
        try:
            verdict = self._run_pipeline(ctx, start_us)
            self._update_stats(verdict["action"], verdict["evaluationUs"])
            return verdict
        except Exception as exc:
            return self._handle_eval_error(exc, ctx, start_us)

    def _run_pipeline(self, ctx: Dict[str, Any], start_us: int) -> Dict[str, Any]:
        enriched = self.cross_agent.enrich_context(ctx)
        self.frequency.record(
            enriched.get("agentId", ""), enriched.get("sessionKey", ""), enriched.get("toolName") or ""
        )
        risk = self.risk_assessor.assess(enriched, self.frequency)
        policies = self.cross_agent.resolve_effective_policies(enriched, self.policy_index)
        result = self.evaluator.evaluate(enriched, policies, risk, self._deps(risk))

        elapsed = _now_us() - start_us
        session_trust = (enriched.get("trust") or {}).get("session") or {}
        verdict = {
            "action": result["action"],
            "reason": result["reason"],
            "risk": risk,
            "matchedPolicies": result["matches"],
            "trust": {"score": session_trust.get("score", 0), "tier": session_trust.get("tier", "untrusted")},
            "evaluationUs": elapsed,
        }

        # Trust learning on denial — but never for time-based night-mode
        # denials (engine.ts:248-263: avoids the trust death spiral for
        # timer-triggered agents).
        if verdict["action"] == "deny" and self.trust_enabled:
            time_based = any(m.get("policyId") == "builtin-night-mode" for m in result["matches"])
            if not time_based:
                self.trust_manager.record_violation(
                    enriched.get("agentId", ""), f"Policy denial: {verdict['reason']}"
                )
                self.session_trust.apply_signal(
                    enriched.get("sessionKey", ""), enriched.get("agentId", ""), "policyBlock"
                )

        self._record_audit(enriched, verdict, risk, elapsed)
        return verdict

    def _record_audit(
        self, ctx: Dict[str, Any], verdict: Dict[str, Any], risk: Dict[str, Any], elapsed_us: int
    ) -> None:
        if not self.audit_enabled:
            return
        audit_ctx = {
            "hook": ctx.get("hook"),
            "agentId": ctx.get("agentId"),
            "sessionKey": ctx.get("sessionKey"),
            "channel": ctx.get("channel"),
            "toolName": ctx.get("toolName"),
            "toolParams": ctx.get("toolParams"),
            "messageContent": ctx.get("messageContent"),
            "messageTo": ctx.get("messageTo"),
            "crossAgent": ctx.get("crossAgent"),
        }
        self.audit_trail.record(
            verdict["action"],
            verdict["reason"],
            audit_ctx,
            dict(verdict["trust"]),
            {"level": risk["level"], "score": risk["score"]},
            verdict["matchedPolicies"],
            elapsed_us,
        )

    def _handle_eval_error(self, exc: Exception, ctx: Dict[str, Any], start_us: int) -> Dict[str, Any]:
        elapsed = _now_us() - start_us
        self.stats["errorCount"] += 1
        self.logger.error("[governance] Evaluation error: %s", exc)
        fallback = "deny" if self.fail_mode == "closed" else "allow"
        reason = (
            "Governance engine error (fail-closed)"
            if fallback == "deny"
            else "Governance engine error (fail-open)"
        )
        session_trust = (ctx.get("trust") or {}).get("session") or {}
        trust = {"score": session_trust.get("score", 0), "tier": session_trust.get("tier", "untrusted")}
        if self.audit_enabled:
            self.audit_trail.record(
                "error_fallback",
                reason,
                {
                    "hook": ctx.get("hook"),
                    "agentId": ctx.get("agentId"),
                    "sessionKey": ctx.get("sessionKey"),
                    "toolName": ctx.get("toolName"),
                },
                trust,
                {"level": "critical", "score": 100},
                [],
                elapsed,
            )
        return {
            "action": fallback,
            "reason": reason,
            "risk": {"level": "critical", "score": 100, "factors": []},
            "matchedPolicies": [],
            "trust": trust,
            "evaluationUs": elapsed,
        }

    def _update_stats(self, action: str, eval_us: int) -> None:
        s = self.stats
        s["evaluations"] += 1
        key = {"allow": "allows", "deny": "denies", "2fa": "twofas"}.get(action)
        if key:
            s[key] += 1
        n = s["evaluations"]
        s["avgEvaluationUs"] += (eval_us - s["avgEvaluationUs"]) / n

    # -- outcome / output validation entry points --------------------------
    def record_outcome(self, agent_id: str, session_key: str, success: bool) -> None:
        if not self.trust_enabled:
            return
        if success:
            self.trust_manager.record_success(agent_id)
            self.session_trust.apply_signal(session_key, agent_id, "success")
        else:
            self.session_trust.apply_signal(session_key, agent_id, "validationFailure")

    def validate_output(
        self,
        content: str,
        agent_id: str,
        channel: Optional[str] = None,
        command: Optional[str] = None,
    ) -> Dict[str, Any]:
        trust_score = self.trust_manager.score(agent_id)
        return self.output_validator.validate(content, trust_score, channel=channel, command=command)

    def status(self) -> Dict[str, Any]:
        return {
            "stats": dict(self.stats),
            "policies": [p.get("id") for p in self.policy_index.policies],
            "agents": self.trust_manager.snapshot(),
            "audit": self.audit_trail.get_stats(),
            "crossAgent": self.cross_agent.graph_summary(),
        }
