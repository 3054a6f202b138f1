"""Policy evaluation: scope filter -> sort -> rule matching -> aggregation.

Parity target: governance `src/policy-evaluator.ts` — scope filter
(`:18-26`), priority + specificity sort (`:28-42`), per-rule AND of
conditions with minTrust/maxTrust tier guards on the SESSION tier
(`:134-139`), verdict aggregation deny > 2fa > audit > allow (`:44-78`).
"""

from __future__ import annotations

from typing import Any, Dict, List, Optional

from .conditions import ConditionDeps, evaluate_conditions
from .util import tier_ordinal


def matches_scope(policy: Dict[str, Any], ctx: Dict[str, Any]) -> bool:
    scope = policy.get("scope") or {}
    exclude = scope.get("excludeAgents")
    if exclude and ctx.get("agentId") in exclude:
        return False
    agents = scope.get("agents")
    if agents and ctx.get("agentId") not in agents:
        return False
    channels = scope.get("channels")
    if channels:
        if not ctx.get("channel") or ctx["channel"] not in channels:
            return False
    return True


def policy_specificity(policy: Dict[str, Any]) -> int:
    scope = policy.get("scope") or {}
    score = 0
    if scope.get("agents"):
        score += 10
    if scope.get("channels"):
        score += 5
    if scope.get("hooks"):
        score += 3
    return score


def sort_policies(policies: List[Dict[str, Any]]) -> List[Dict[str, Any]]:
    return sorted(
        policies,
        key=lambda p: (-(p.get("priority") or 0), -policy_specificity(p)),
    )


def aggregate_matches(matches: List[Dict[str, Any]]) -> Dict[str, Any]:
    """deny > 2fa > audit > allow (policy-evaluator.ts:44-78)."""
    deny_reason = ""
    twofa_reason = ""
    has_deny = has_2fa = has_audit = False
    for m in matches:
        action = (m.get("effect") or {}).get("action")
        if action == "deny":
            has_deny = True
            if not deny_reason:
                deny_reason = (m.get("effect") or {}).get("reason", "")
        elif action == "2fa":
            has_2fa = True
            if not twofa_reason:
                twofa_reason = (m.get("effect") or {}).get("reason", "") or ""
        elif action == "audit":
            has_audit = True
    if has_deny:
        return {"action": "deny", "reason": deny_reason or "Denied by governance policy", "matches": matches}
    if has_2fa:
        return {"action": "2fa", "reason": twofa_reason or "Requires 2FA approval", "matches": matches}
    if has_audit:
        return {"action": "allow", "reason": "Allowed with audit logging", "matches": matches}
    return {
        "action": "allow",
        "reason": "Allowed by governance policy" if matches else "No matching policies",
        "matches": matches,
    }


class PolicyEvaluator:
    def evaluate(
        self,
        ctx: Dict[str, Any],
        policies: List[Dict[str, Any]],
        risk: Optional[Dict[str, Any]] = None,
        deps: Optional[ConditionDeps] = None,
    ) -> Dict[str, Any]:
        deps = deps or ConditionDeps(risk=risk)
        if risk is not None:
            deps.risk = risk
        applicable = sort_policies([p for p in policies if matches_scope(p, ctx)])
        matches: List[Dict[str, Any]] = []
        for policy in applicable:
            m = self._match_policy(policy, ctx, deps)
            if m:
                matches.append(m)
        return aggregate_matches(matches)

    def _match_policy(
        self, policy: Dict[str, Any], ctx: Dict[str, Any], deps: ConditionDeps
    ) -> Optional[Dict[str, Any]]:
        session_tier = str(((ctx.get("trust") or {}).get("session") or {}).get("tier", "untrusted"))
        for rule in policy.get("rules", []):
            min_trust = rule.get("minTrust")
            if min_trust and tier_ordinal(session_tier) < tier_ordinal(min_trust):
                continue
            max_trust = rule.get("maxTrust")
            if max_trust and tier_ordinal(session_tier) > tier_ordinal(max_trust):
                continue
            if evaluate_conditions(rule.get("conditions", []), ctx, deps):
                return {
                    "policyId": policy.get("id"),
                    "ruleId": rule.get("id"),
                    "effect": rule.get("effect") or {"action": "allow"},
                    "controls": policy.get("controls") or [],
                }
        return None
