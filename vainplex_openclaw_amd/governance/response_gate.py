"""Response Gate: synchronous pre-response enforcement.

Parity target: governance `src/response-gate.ts` — per-agent rules with
validators `requiredTools` (checked against the session toolCallLog),
`mustMatch`, `mustNotMatch`; invalid regex => fail-closed block
(`:105-147,177-188`); fallback message templating with
{reasons}/{validators}/{agent} (`:153-169`).
"""

from __future__ import annotations

import re
from typing import Any, Dict, List, Optional


def resolve_response_gate(raw: Any) -> Dict[str, Any]:
    r = raw if isinstance(raw, dict) else {}
    return {
        "enabled": bool(r.get("enabled", False)),
        "rules": r.get("rules") if isinstance(r.get("rules"), list) else [],
        "fallbackMessage": r.get("fallbackMessage") if isinstance(r.get("fallbackMessage"), str) else None,
        "fallbackTemplate": r.get("fallbackTemplate") if isinstance(r.get("fallbackTemplate"), str) else None,
    }


class ResponseGate:
    def __init__(self, config: Optional[Dict[str, Any]] = None):
        self.config = resolve_response_gate(config)
        self._regex_cache: Dict[str, Optional["re.Pattern[str]"]] = {}

    def validate(self, content: str, agent_id: str, tool_call_log: List[Dict[str, str]]) -> Dict[str, Any]:
        if not self.config["enabled"]:
            return {"passed": True, "failedValidators": [], "reasons": []}
        failed: List[str] = []
        reasons: List[str] = []
        for rule in self.config["rules"]:
            if not self._rule_for_agent(rule, agent_id):
                continue
            for validator in rule.get("validators", []):
                res = self._run_validator(validator, content, tool_call_log)
                if not res["passed"]:
                    vtype = validator.get("type")
                    if vtype == "requiredTools":
                        failed.append(f"requiredTools:{','.join(validator.get('tools', []))}")
                    else:
                        failed.append(f"{vtype}:{validator.get('pattern')}")
                    reasons.append(res["reason"])
        passed = not failed
        result: Dict[str, Any] = {"passed": passed, "failedValidators": failed, "reasons": reasons}
        if not passed:
            fb = self._render_fallback(agent_id, failed, reasons)
            if fb is not None:
                result["fallbackMessage"] = fb
        return result

    def _run_validator(self, validator: Dict[str, Any], content: str, log: List[Dict[str, str]]) -> Dict[str, Any]:
        vtype = validator.get("type")
        if vtype == "requiredTools":
            called = {e.get("toolName") for e in log}
            missing = [t for t in validator.get("tools", []) if t not in called]
            if missing:
                return {
                    "passed": False,
                    "reason": validator.get("message")
                    or f"Response Gate: required tool(s) not called: {', '.join(missing)}",
                }
            return {"passed": True}
        if vtype in ("mustMatch", "mustNotMatch"):
            pattern = validator.get("pattern", "")
            rx = self._get_regex(pattern)
            if rx is None:
                return {
                    "passed": False,
                    "reason": f"Response Gate: invalid regex pattern /{pattern}/ — blocked (fail-closed)",
                }
            hit = bool(rx.search(content))
            if vtype == "mustMatch" and not hit:
                return {
                    "passed": False,
                    "reason": validator.get("message")
                    or f"Response Gate: content does not match required pattern /{pattern}/",
                }
            if vtype == "mustNotMatch" and hit:
                return {
                    "passed": False,
                    "reason": validator.get("message")
                    or f"Response Gate: content matches forbidden pattern /{pattern}/",
                }
            return {"passed": True}
        return {"passed": True}

    def _render_fallback(self, agent_id: str, failed: List[str], reasons: List[str]) -> Optional[str]:
        # fallbackMessage is STATIC and takes precedence; fallbackTemplate
        # renders {reasons}/{validators}/{agent} (response-gate.ts:153-169)
        static = self.config.get("fallbackMessage")
        if static:
            return static
        template = self.config.get("fallbackTemplate")
        if not template:
            return None
        return (
            template.replace("{reasons}", "; ".join(reasons))
            .replace("{validators}", ", ".join(failed))
            .replace("{agent}", agent_id)
        )

    @staticmethod
    def _rule_for_agent(rule: Dict[str, Any], agent_id: str) -> bool:
        rid = rule.get("agentId")
        if not rid:
            return True
        if isinstance(rid, list):
            return agent_id in rid
        return rid == agent_id

    def _get_regex(self, pattern: str) -> Optional["re.Pattern[str]"]:
        if pattern not in self._regex_cache:
            try:
                self._regex_cache[pattern] = re.compile(pattern)
            except re.error:
                self._regex_cache[pattern] = None
        return self._regex_cache[pattern]
