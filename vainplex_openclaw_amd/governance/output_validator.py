"""Output validation pipeline: claims -> fact-check -> optional LLM stage.

Parity target: governance `src/output-validator.ts` — Stage 1 claim
detection + Stage 2 fact-check synchronous; Stage 3 LLM only for
external comms (`:100-143`). Trust-proportional contradiction verdict
(`:243-275`): block if trust < blockBelow (40), pass if trust >= flagAbove
(60), else flag. Most-restrictive-wins merge with the LLM verdict
(`:159,288-290`). Defaults: unverifiedClaimPolicy="ignore",
selfReferentialPolicy="ignore" (`:36-52`).
"""

from __future__ import annotations

import time
from typing import Any, Dict, List, Optional

from ..core.api import PluginLogger, NullLogger
from .claims import detect_claims
from .facts import FactRegistry, check_claims

VERDICT_SEVERITY = {"pass": 0, "flag": 1, "block": 2}

DEFAULT_CONFIG: Dict[str, Any] = {
    "enabled": True,
    "enabledDetectors": None,  # None = all
    "factRegistries": [],
    "unverifiedClaimPolicy": "ignore",
    "selfReferentialPolicy": "ignore",
    "contradictionThresholds": {"flagAbove": 60, "blockBelow": 40},
    "llmValidator": {"enabled": False},
}


def more_restrictive(a: str, b: str) -> str:
    return a if VERDICT_SEVERITY.get(a, 0) >= VERDICT_SEVERITY.get(b, 0) else b


def _now_us() -> int:
    return int(time.perf_counter() * 1_000_000)


class OutputValidator:
    def __init__(self, config: Optional[Dict[str, Any]] = None, logger: Optional[PluginLogger] = None):
        cfg = dict(DEFAULT_CONFIG)
        cfg.update(config or {})
        thr = dict(DEFAULT_CONFIG["contradictionThresholds"])
        thr.update((config or {}).get("contradictionThresholds") or {})
        cfg["contradictionThresholds"] = thr
        self.config = cfg
        self.logger = logger or NullLogger()
        self.fact_registry = FactRegistry(cfg.get("factRegistries"))
        self.llm_validator: Optional[Any] = None

    def set_llm_validator(self, validator: Any) -> None:
        self.llm_validator = validator

    def get_fact_count(self) -> int:
        return self.fact_registry.size

    def validate(
        self,
        text: str,
        trust_score: float,
        is_external: bool = False,
        channel: Optional[str] = None,
        command: Optional[str] = None,
    ) -> Dict[str, Any]:
        start = _now_us()
        if not self.config.get("enabled") or not text:
            return self._pass_result(start)

        claims = detect_claims(text, self.config.get("enabledDetectors"))
        if not claims and not is_external:
            return self._pass_result(start, reason="No claims detected")

        results = check_claims(claims, self.fact_registry) if claims else []
        contradictions = [r for r in results if r["status"] == "contradicted"]
        unverified = [r for r in results if r["status"] == "unverified"]
        verdict = self._determine_verdict(contradictions, unverified, trust_score)

        llm_cfg = self.config.get("llmValidator") or {}
        if is_external and self.llm_validator is not None and llm_cfg.get("enabled"):
            try:
                llm = self.llm_validator.validate(text, self.fact_registry.get_all_facts(), True)
                final = more_restrictive(verdict["action"], llm.get("verdict", "pass"))
                reasons = [verdict["reason"]]
                if llm.get("reason"):
                    reasons.append(f"LLM: {llm['reason']}")
                return {
                    "verdict": final,
                    "claims": claims,
                    "factCheckResults": results,
                    "contradictions": contradictions,
                    "reason": "; ".join(r for r in reasons if r),
                    "llmResult": llm,
                    "evaluationUs": _now_us() - start,
                }
            except Exception as exc:
                self.logger.warn("[governance] LLM validation failed: %s", exc)

        return {
            "verdict": verdict["action"],
            "claims": claims,
            "factCheckResults": results,
            "contradictions": contradictions,
            "reason": verdict["reason"],
            "evaluationUs": _now_us() - start,
        }

    def _determine_verdict(
        self, contradictions: List[Dict[str, Any]], unverified: List[Dict[str, Any]], trust: float
    ) -> Dict[str, str]:
        if contradictions:
            return self._verdict_for_contradiction(contradictions, trust)
        if unverified and self.config.get("unverifiedClaimPolicy") != "ignore":
            self_ref = [r for r in unverified if r["claim"].get("type") == "self_referential"]
            other = [r for r in unverified if r["claim"].get("type") != "self_referential"]
            if self_ref and self.config.get("selfReferentialPolicy") != "ignore":
                action = "block" if self.config["selfReferentialPolicy"] == "block" else "flag"
                plural = "s" if len(self_ref) > 1 else ""
                srcs = ", ".join(f'"{r["claim"]["source"]}"' for r in self_ref)
                return {"action": action, "reason": f"Self-referential claim{plural} detected: {srcs}"}
            if other:
                action = "block" if self.config["unverifiedClaimPolicy"] == "block" else "flag"
                plural = "s" if len(other) > 1 else ""
                srcs = ", ".join(f'"{r["claim"]["source"]}"' for r in other)
                return {"action": action, "reason": f"Unverified claim{plural}: {srcs}"}
        return {"action": "pass", "reason": "All claims verified or no contradictions found"}

    def _verdict_for_contradiction(self, contradictions: List[Dict[str, Any]], trust: float) -> Dict[str, str]:
        thr = self.config["contradictionThresholds"]
        block_below, flag_above = thr["blockBelow"], thr["flagAbove"]
        detail = "; ".join(
            f'{c["claim"]["subject"]}: claimed "{c["claim"]["value"]}", '
            f'actual "{(c.get("fact") or {}).get("value", "unknown")}"'
            for c in contradictions
        )
        if trust < block_below:
            return {"action": "block", "reason": f"Contradiction detected (trust {trust} < {block_below}): {detail}"}
        if trust >= flag_above:
            return {
                "action": "pass",
                "reason": f"Contradiction detected but trusted (trust {trust} >= {flag_above}): {detail}",
            }
        return {"action": "flag", "reason": f"Contradiction detected (trust {trust}): {detail}"}

    @staticmethod
    def _pass_result(start_us: int, reason: str = "Output validation disabled or empty text") -> Dict[str, Any]:
        return {
            "verdict": "pass",
            "claims": [],
            "factCheckResults": [],
            "contradictions": [],
            "reason": reason,
            "evaluationUs": _now_us() - start_us,
        }
