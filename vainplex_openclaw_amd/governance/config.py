"""Full GovernanceConfig resolution.

Parity target: reference `openclaw-governance/src/config.ts` (337 LoC) —
per-field typed resolution with defaults for trust (incl. nested
sessionTrust signals and decay), audit, output validation (claim
detectors + llmValidator), performance, ERC-8004 (top-level or
`agentFirewall.erc8004` nesting, `:204-217,330-334`), approval2fa (only
when enabled AND a TOTP secret is present, `:272-291`), builtin-policy
toggles, failMode and timezone.

The engine/components accept plain dicts and apply their own local
defaults too; this resolver is the single place that produces the
complete, documented shape (what `config-loader.ts` returns).
"""

from __future__ import annotations

from typing import Any, Dict, Optional

from .hooks import resolve_erc8004_config

ALL_DETECTOR_IDS = (
    "system_state", "entity_name", "existence",
    "operational_status", "self_referential",
)


def _rec(v: Any) -> Dict[str, Any]:
    return v if isinstance(v, dict) else {}


def _num(v: Any, d: float):
    return v if isinstance(v, (int, float)) and not isinstance(v, bool) else d


def _boolv(v: Any, d: bool) -> bool:
    return v if isinstance(v, bool) else d


def _str(v: Any, d: str) -> str:
    return v if isinstance(v, str) else d


def resolve_session_trust(raw: Any) -> Dict[str, Any]:
    r = _rec(raw)
    s = _rec(r.get("signals"))
    return {
        "enabled": _boolv(r.get("enabled"), True),
        "seedFactor": _num(r.get("seedFactor"), 0.7),
        "ceilingFactor": _num(r.get("ceilingFactor"), 1.2),
        "signals": {
            "success": _num(s.get("success"), 1),
            "policyBlock": _num(s.get("policyBlock"), -2),
            "credentialViolation": _num(s.get("credentialViolation"), -10),
            "cleanStreakBonus": _num(s.get("cleanStreakBonus"), 3),
            "cleanStreakThreshold": _num(s.get("cleanStreakThreshold"), 10),
        },
    }


def resolve_trust(raw: Any) -> Dict[str, Any]:
    r = _rec(raw)
    decay = _rec(r.get("decay"))
    defaults = {
        k: v for k, v in _rec(r.get("defaults")).items()
        if isinstance(v, (int, float)) and not isinstance(v, bool)
    } or {"main": 60, "*": 10}
    return {
        "enabled": _boolv(r.get("enabled"), True),
        "defaults": defaults,
        "persistIntervalSeconds": _num(r.get("persistIntervalSeconds"), 60),
        "decay": {
            "enabled": _boolv(decay.get("enabled"), True),
            "inactivityDays": _num(decay.get("inactivityDays"), 30),
            "rate": _num(decay.get("rate"), 0.95),
        },
        "weights": _rec(r.get("weights")) or None,
        "maxHistoryPerAgent": _num(r.get("maxHistoryPerAgent"), 100),
        "sessionTrust": resolve_session_trust(r.get("sessionTrust")),
    }


def resolve_audit(raw: Any) -> Dict[str, Any]:
    r = _rec(raw)
    level = r.get("level")
    return {
        "enabled": _boolv(r.get("enabled"), True),
        "retentionDays": _num(r.get("retentionDays"), 90),
        "redactPatterns": [p for p in r.get("redactPatterns", []) if isinstance(p, str)]
        if isinstance(r.get("redactPatterns"), list) else [],
        "level": level if level in ("minimal", "standard", "verbose") else "standard",
    }


def _str_list(v: Any, default: list) -> list:
    if not isinstance(v, list):
        return list(default)
    return [x for x in v if isinstance(x, str)]


def resolve_llm_validator(raw: Any) -> Dict[str, Any]:
    r = _rec(raw)
    return {
        "enabled": _boolv(r.get("enabled"), False),
        "externalChannels": _str_list(r.get("externalChannels"),
                                      ["twitter", "linkedin", "email"]),
        "externalCommands": _str_list(r.get("externalCommands"), ["bird tweet"]),
        "cacheTtlSeconds": _num(r.get("cacheTtlSeconds"), 300),
        "model": _str(r.get("model"), "mistral:7b"),
        "endpoint": _str(r.get("endpoint"), "http://localhost:11434/api/generate"),
    }


def _policy3(v: Any) -> str:
    return v if v in ("ignore", "flag", "block") else "ignore"


def resolve_output_validation(raw: Any) -> Dict[str, Any]:
    r = _rec(raw)
    # accept both the reference's `enabledDetectors`/`claimDetectors`
    # spellings and the legacy `detectors`
    detectors = (r.get("enabledDetectors") if isinstance(r.get("enabledDetectors"), list)
                 else r.get("claimDetectors") if isinstance(r.get("claimDetectors"), list)
                 else r.get("detectors"))
    return {
        "enabled": _boolv(r.get("enabled"), True),
        "detectors": [d for d in detectors if d in ALL_DETECTOR_IDS]
        if isinstance(detectors, list) else list(ALL_DETECTOR_IDS),
        "facts": r.get("facts") if isinstance(r.get("facts"), list) else [],
        "factsFile": r.get("factsFile") if isinstance(r.get("factsFile"), str) else None,
        "unverifiedClaimPolicy": _policy3(r.get("unverifiedClaimPolicy")),
        "selfReferentialPolicy": _policy3(r.get("selfReferentialPolicy")),
        "llmValidator": resolve_llm_validator(r.get("llmValidator")),
    }


def resolve_performance(raw: Any) -> Dict[str, Any]:
    r = _rec(raw)
    return {
        "trackEvaluationTime": _boolv(r.get("trackEvaluationTime"), True),
        "slowEvaluationUs": _num(r.get("slowEvaluationUs"), 10_000),
    }


def resolve_approval_2fa(raw: Any) -> Optional[Dict[str, Any]]:
    """Only when enabled AND a TOTP secret is present (config.ts:272-291)."""
    r = _rec(raw)
    if not r.get("enabled") or not isinstance(r.get("totpSecret"), str) or not r["totpSecret"]:
        return None
    return {
        "enabled": True,
        "totpSecret": r["totpSecret"],
        "totpIssuer": _str(r.get("totpIssuer"), "Vainplex Governance"),
        "sessionApprovalMinutes": _num(r.get("sessionApprovalMinutes"), 10),
        "batchWindowSeconds": _num(r.get("batchWindowSeconds"), 3),
        "timeoutSeconds": _num(r.get("timeoutSeconds"), 300),
        "matrix": _rec(r.get("matrix")) or None,
    }


def resolve_config(raw: Optional[Dict[str, Any]]) -> Dict[str, Any]:
    r = _rec(raw)
    bp = _rec(r.get("builtinPolicies"))
    return {
        "enabled": _boolv(r.get("enabled"), True),
        "timezone": _str(r.get("timezone"), "UTC"),
        "failMode": r.get("failMode") if r.get("failMode") in ("open", "closed") else "open",
        "workspace": r.get("workspace") if isinstance(r.get("workspace"), str) else None,
        "trust": resolve_trust(r.get("trust")),
        "audit": resolve_audit(r.get("audit")),
        "outputValidation": resolve_output_validation(r.get("outputValidation")),
        "performance": resolve_performance(r.get("performance")),
        "builtinPolicies": {
            "nightMode": _boolv(bp.get("nightMode"), False),
            "credentialGuard": _boolv(bp.get("credentialGuard"), False),
            "productionSafeguard": _boolv(bp.get("productionSafeguard"), False),
            "rateLimiter": _boolv(bp.get("rateLimiter"), False),
        },
        "policies": r.get("policies") if isinstance(r.get("policies"), list) else [],
        "responseGate": _rec(r.get("responseGate")) or {"enabled": False},
        "redaction": _rec(r.get("redaction")) or {},
        "erc8004": resolve_erc8004_config(r),
        "approval2fa": resolve_approval_2fa(r.get("approval2fa")),
    }
