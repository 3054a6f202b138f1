"""Trace analyzer: orchestrator, classifier, redactor, outputs, report,
sources.

Parity target: cortex `src/trace-analyzer/` —
- analyzer.ts run(): connect source -> incremental fetch (lastProcessedTs
  minus context window, `:171-194`) -> reconstruct -> detect -> classify
  -> outputs -> report -> persist state (`:124-155`).
- classifier.ts: optional triage LLM (keep? severity?) then analysis LLM,
  chains redacted first (`:30-60`).
- redactor.ts: strips credentials/PII from chains before any LLM call
  (reuses the governance redaction registry).
- output-generator.ts: group findings by normalized actionText ->
  soul_rule / governance_policy / cortex_pattern with confidence
  (`:13-27,36-50`).
- report.ts: AnalysisReport + ProcessingState for incremental resume.
- trace-source.ts / nats-trace-source.ts: source protocol; the NATS
  source degrades to None gracefully when no client is available
  (nats-trace-source.ts:1-12); a journal source reads the eventstore's
  embedded journal.
"""

from __future__ import annotations

import json
import os
import time
import uuid
from dataclasses import dataclass, field
from typing import Any, Callable, Dict, List, Optional

from ...governance.redaction.engine import RedactionEngine
from ...governance.redaction.registry import PatternRegistry
from ...governance.redaction.vault import RedactionVault
from ..storage import load_json, save_json
from .chains import ConversationChain, reconstruct_chains
from .events import NormalizedEvent, normalize_schema_a, normalize_schema_b
from .signals import Finding, detect_all_signals


# -- sources ----------------------------------------------------------------

def resolve_analyzer_llm_config(top_level, override):
    """Merge the top-level cortex LLM config with the trace-analyzer
    override (classifier.ts resolveAnalyzerLlmConfig): the override's
    `enabled` gates the result; any field it omits falls back to the
    top-level value."""
    top = dict(top_level or {})
    ov = dict(override or {})
    if not ov.get("enabled", False):
        return {"enabled": False}
    merged = {
        "enabled": True,
        "endpoint": top.get("endpoint", ""),
        "model": top.get("model", ""),
        "apiKey": top.get("apiKey", ""),
        "timeoutMs": top.get("timeoutMs", 15000),
    }
    for key in ("endpoint", "model", "apiKey", "timeoutMs"):
        if key in ov:
            merged[key] = ov[key]
    return merged


class MockTraceSource:
    """Deterministic in-memory source (test helper parity:
    openclaw-cortex/test/trace-analyzer/helpers.ts)."""

    def __init__(self, events: Optional[List[NormalizedEvent]] = None):
        self.events = list(events or [])

    def fetch(self, since_ts: float = 0) -> List[NormalizedEvent]:
        return [e for e in self.events if e.ts >= since_ts]


class JournalTraceSource:
    """Reads the embedded eventstore journal (Schema A envelopes)."""

    def __init__(self, journal) -> None:
        self.journal = journal  # eventstore.journal.EventJournal

    def fetch(self, since_ts: float = 0) -> List[NormalizedEvent]:
        out = []
        for seq, env in self.journal.replay(since_ts=since_ts):
            ne = normalize_schema_a(env, seq)
            if ne:
                out.append(ne)
        return out


def create_nats_source(url: Optional[str] = None, transport=None, logger=None):
    """NATS JetStream source — delegates to trace.nats_source (real
    wire-protocol client over eventstore/nats_client); returns None
    gracefully when the connection fails (nats-trace-source.ts:103-115),
    in which case the journal source is the production path."""
    from .nats_source import create_nats_trace_source

    cfg = {"url": url} if url else {}
    return create_nats_trace_source(cfg, logger=logger, transport=transport)


# -- redactor ---------------------------------------------------------------

# trace-specific credential shapes beyond the 17 registry builtins
# (redactor.ts: Stripe keys, raw JWTs, url-embedded passwords, env
# assignments) — chains quote raw tool output, so the surface is wider
_TRACE_EXTRA_PATTERNS = [
    {"name": "stripe-key", "category": "custom",
     "regex": r"\b[ps]k_(?:live|test)_[A-Za-z0-9]{16,}"},
    {"name": "jwt", "category": "custom",
     "regex": r"\beyJ[A-Za-z0-9_-]{8,}\.[A-Za-z0-9_-]{8,}\.[A-Za-z0-9_-]{8,}"},
    {"name": "url-password", "category": "custom",
     "regex": r"://[^/\s:]+:[^@\s]{4,}@"},
    {"name": "env-secret", "category": "custom",
     "regex": r"\b(?:PASSWORD|SECRET|TOKEN|API_KEY)=\S{4,}"},
]


class ChainRedactor:
    def __init__(self, custom_patterns: Optional[List[Dict[str, Any]]] = None) -> None:
        patterns = list(_TRACE_EXTRA_PATTERNS)
        for p in custom_patterns or []:
            if isinstance(p, dict) and isinstance(p.get("regex"), str):
                patterns.append({"name": p.get("name", "custom"),
                                 "category": "custom", "regex": p["regex"]})
        # PatternRegistry drops invalid regexes silently (redactor.ts:141)
        self.engine = RedactionEngine(PatternRegistry(None, patterns), RedactionVault())

    def redact_chain(self, chain: ConversationChain) -> List[Dict[str, Any]]:
        """Returns redacted copies; the chain itself is never mutated."""
        out = []
        for ev in chain.events:
            payload = dict(ev.payload)
            for key in ("content", "toolError"):
                if isinstance(payload.get(key), str):
                    payload[key] = self.engine.scan_string(payload[key])["output"]
            for key in ("toolParams", "toolResult"):
                if isinstance(payload.get(key), (dict, list)):
                    payload[key] = self.engine.scan(payload[key])["output"]
                elif isinstance(payload.get(key), str):
                    payload[key] = self.engine.scan_string(payload[key])["output"]
            out.append({"type": ev.type, "ts": ev.ts, "payload": payload})
        return out


# -- classifier -------------------------------------------------------------

TRIAGE_PROMPT = (
    "You triage agent-failure findings. Reply ONLY JSON: "
    '{"keep": true|false, "severity": "low"|"medium"|"high"}\nFinding: '
)
ANALYSIS_PROMPT = (
    "You analyze an agent failure. Reply ONLY JSON: "
    '{"rootCause": "...", "actionType": "soul_rule"|"governance_policy"|'
    '"cortex_pattern"|"manual_review", "actionText": "...", "confidence": 0.0}\n'
)

# classifier.ts: unknown actionType values default to manual_review
ACTION_TYPES = ("soul_rule", "governance_policy", "cortex_pattern", "manual_review")


class FindingClassifier:
    """Stage 2: optional triage + analysis LLM per finding (classifier.ts)."""

    def __init__(
        self,
        call_llm: Optional[Callable[[str], str]] = None,
        triage_enabled: bool = True,
        model: str = "",
    ):
        self.call_llm = call_llm
        self.triage_enabled = triage_enabled
        self.model = model
        self.redactor = ChainRedactor()

    @staticmethod
    def _parse(raw: str) -> Optional[Dict[str, Any]]:
        start, end = raw.find("{"), raw.rfind("}")
        if start < 0 or end <= start:
            return None
        try:
            out = json.loads(raw[start : end + 1])
            return out if isinstance(out, dict) else None
        except json.JSONDecodeError:
            return None

    def classify(self, findings: List[Finding], chains: Dict[str, ConversationChain]) -> List[Dict[str, Any]]:
        out = []
        for f in findings:
            d = f.to_dict()
            if self.call_llm is None:
                d["actionText"] = default_action_text(f)
                out.append(d)
                continue
            chain = chains.get(f.chain_id)
            redacted = self.redactor.redact_chain(chain) if chain else []
            context = json.dumps({"finding": d, "chain": redacted[-10:]}, default=str)[:4000]
            try:
                if self.triage_enabled:
                    triage = self._parse(self.call_llm(TRIAGE_PROMPT + context)) or {}
                    if triage.get("keep") is False:
                        continue
                    if triage.get("severity") in ("low", "medium", "high"):
                        d["severity"] = triage["severity"]
                analysis = self._parse(self.call_llm(ANALYSIS_PROMPT + context))
                if analysis is None:
                    # invalid JSON / HTTP error / timeout: classification
                    # stays null (classifier.ts), default action text kept
                    d["actionText"] = default_action_text(f)
                    out.append(d)
                    continue
                d["rootCause"] = analysis.get("rootCause", "")
                d["actionText"] = analysis.get("actionText") or default_action_text(f)
                if isinstance(analysis.get("confidence"), (int, float)):
                    d["confidence"] = float(analysis["confidence"])
                atype = analysis.get("actionType")
                d["classification"] = {
                    "rootCause": d["rootCause"],
                    "actionType": atype if atype in ACTION_TYPES else "manual_review",
                    "actionText": d["actionText"],
                    "confidence": float(analysis["confidence"])
                    if isinstance(analysis.get("confidence"), (int, float)) else 0.5,
                    "model": self.model,
                }
            except Exception:
                d["actionText"] = default_action_text(f)
            out.append(d)
        return out


def default_action_text(f: Finding) -> str:
    table = {
        "doom_loop": f"Stop retrying {f.evidence.get('toolName', 'the tool')} after 2 consecutive identical failures; change approach instead",
        "correction": "Re-read the user's original request before answering; confirm understanding on ambiguity",
        "tool_fail": "Check tool preconditions before calling; validate inputs",
        "dissatisfied": "Acknowledge the problem and summarize a concrete recovery plan",
        "repeat_fail": f"Investigate the root cause of repeated {f.evidence.get('toolName', 'tool')} errors before retrying",
        "hallucination": "Verify system state with a tool call before asserting it",
        "unverified_claim": "Qualify claims or verify them with a tool call first",
    }
    return table.get(f.signal_type, "Review this failure pattern")


# -- output generator -------------------------------------------------------

def _norm_action(text: str) -> str:
    return " ".join(text.lower().split())[:120]


def generate_outputs(classified: List[Dict[str, Any]]) -> List[Dict[str, Any]]:
    """Group by normalized actionText -> soul_rule / governance_policy /
    cortex_pattern (output-generator.ts:13-50)."""
    groups: Dict[str, List[Dict[str, Any]]] = {}
    for c in classified:
        groups.setdefault(_norm_action(c.get("actionText", "")), []).append(c)
    outputs = []
    for action, items in groups.items():
        if not action:
            continue
        sig_types = {i["signalType"] for i in items}
        confidence = min(0.95, max(i.get("confidence", 0.5) for i in items) + 0.1 * (len(items) - 1))
        if sig_types & {"doom_loop", "repeat_fail", "tool_fail"}:
            kind = "governance_policy"
        elif sig_types & {"hallucination", "unverified_claim"}:
            kind = "soul_rule"
        else:
            kind = "cortex_pattern"
        outputs.append({
            "id": f"out-{uuid.uuid4().hex[:10]}",
            "kind": kind,
            "actionText": items[0].get("actionText", ""),
            "occurrences": len(items),
            "agents": sorted({i["agent"] for i in items}),
            "signalTypes": sorted(sig_types),
            "confidence": round(confidence, 3),
            "findingIds": [i["id"] for i in items],
        })
    outputs.sort(key=lambda o: -o["confidence"])
    return outputs


# classification-driven generator (output-generator.ts R-021/22/23):
# consumes the nested LLM classification; unclassified and manual_review
# findings produce no output.

_HOOKS_BY_SIGNAL = {
    "doom_loop": ["before_tool_call"],
    "tool_fail": ["before_tool_call"],
    "hallucination": ["message_sending"],
    "unverified_claim": ["message_sending"],
}


def _classified(findings: List[Dict[str, Any]], atype: str) -> List[Dict[str, Any]]:
    return [f for f in findings
            if (f.get("classification") or {}).get("actionType") == atype]


def generate_classified_outputs(findings: List[Dict[str, Any]]) -> List[Dict[str, Any]]:
    outputs: List[Dict[str, Any]] = []

    # soul rules: grouped by normalized actionText (first 80 chars),
    # content carries the observation count + finding-id refs
    groups: Dict[str, List[Dict[str, Any]]] = {}
    for f in _classified(findings, "soul_rule"):
        key = " ".join(f["classification"]["actionText"].lower().split())[:80]
        groups.setdefault(key, []).append(f)
    for items in groups.values():
        action = items[0]["classification"]["actionText"]
        ids = [i["id"] for i in items]
        id_ref = ", ".join(i[:8] for i in ids[:3])
        outputs.append({
            "id": f"out-{uuid.uuid4().hex[:12]}",
            "type": "soul_rule",
            "content": f"{action} [{len(items)}× beobachtet in Traces, Findings: {id_ref}]",
            "sourceFindings": ids,
            "observationCount": len(items),
            "confidence": sum(i["classification"]["confidence"] for i in items) / len(items),
        })

    # governance policies: one audit-effect policy JSON per finding
    for f in _classified(findings, "governance_policy"):
        c = f["classification"]
        sig = str(f.get("signalType", "unknown"))
        policy = {
            "id": f"trace-gen-{sig.replace('_', '-')}-{f['id'][:8]}",
            "name": f"Auto: {str(f.get('summary', ''))[:60]}",
            "version": "1.0.0",
            "description": f"Auto-generated from trace finding {f['id']}. "
                           f"Root cause: {c.get('rootCause', '')}",
            "scope": {"hooks": _HOOKS_BY_SIGNAL.get(sig, ["message_sent"])},
            "rules": [{
                "id": f"rule-{f['id'][:8]}",
                "description": c["actionText"],
                "conditions": {"signal": sig, "severity": f.get("severity")},
                "effect": {"action": "audit", "reason": c["actionText"]},
            }],
        }
        outputs.append({
            "id": f"out-{uuid.uuid4().hex[:12]}",
            "type": "governance_policy",
            "content": json.dumps(policy, indent=2),
            "sourceFindings": [f["id"]],
            "observationCount": 1,
            "confidence": c["confidence"],
        })

    # cortex patterns: the actionText IS the regex
    for f in _classified(findings, "cortex_pattern"):
        outputs.append({
            "id": f"out-{uuid.uuid4().hex[:12]}",
            "type": "cortex_pattern",
            "content": f["classification"]["actionText"],
            "sourceFindings": [f["id"]],
            "observationCount": 1,
            "confidence": f["classification"]["confidence"],
        })
    return outputs


# -- report + orchestrator --------------------------------------------------

@dataclass
class AnalyzerConfig:
    enabled: bool = True
    detectors: Optional[List[str]] = None
    incremental_context_window_ms: float = 30 * 60 * 1000
    min_confidence: float = 0.0
    chain_gap_minutes: float = 30.0
    redact_patterns: List[str] = field(default_factory=list)
    max_findings: int = 1000


class TraceAnalyzer:
    def __init__(
        self,
        workspace: str,
        source,
        config: Optional[AnalyzerConfig] = None,
        call_llm: Optional[Callable[[str], str]] = None,
        clock=time.time,
        signal_registry=None,
    ):
        self.workspace = workspace
        self.source = source
        self.config = config or AnalyzerConfig()
        self.classifier = FindingClassifier(call_llm)
        self.clock = clock
        self.signal_registry = signal_registry
        if self.config.redact_patterns:
            self.classifier.redactor = ChainRedactor(
                [{"name": f"cfg-{i}", "regex": p}
                 for i, p in enumerate(self.config.redact_patterns)])
        self.state_path = os.path.join(workspace, "memory", "reboot", "trace-analyzer-state.json")
        self.report_path = os.path.join(workspace, "memory", "reboot", "trace-analysis-report.json")
        self.state: Dict[str, Any] = load_json(self.state_path) or {}

    @classmethod
    def from_config(cls, workspace: str, source, resolved: Dict[str, Any],
                    call_llm=None, clock=time.time, signal_registry=None):
        """Build from a resolve_trace_analyzer_config() dict: per-signal
        enables -> detectors, chainGapMinutes, redactPatterns, output
        caps (config.ts -> analyzer.ts wiring)."""
        from .config import enabled_detectors

        cfg = AnalyzerConfig(
            enabled=bool(resolved.get("enabled")),
            detectors=enabled_detectors(resolved),
            chain_gap_minutes=float(resolved["chainGapMinutes"]),
            redact_patterns=list(resolved.get("redactPatterns", [])),
            max_findings=int(resolved["output"]["maxFindings"]),
        )
        ta = cls(workspace, source, cfg, call_llm=call_llm, clock=clock,
                 signal_registry=signal_registry)
        if resolved["output"].get("reportPath"):
            ta.report_path = resolved["output"]["reportPath"]
        return ta

    def run(self) -> Dict[str, Any]:
        """Full pipeline: fetch -> chains -> detect -> classify -> outputs
        -> report -> persist state (analyzer.ts:124-155)."""
        last_ts = float(self.state.get("lastProcessedTs", 0))
        since = max(0.0, last_ts - self.config.incremental_context_window_ms)
        events = self.source.fetch(since_ts=since)
        chains = reconstruct_chains(events, gap_minutes=self.config.chain_gap_minutes)
        chain_map = {c.id: c for c in chains}
        findings = detect_all_signals(chains, self.config.detectors,
                                      registry=self.signal_registry)
        classified = self.classifier.classify(findings, chain_map)
        classified = [c for c in classified if c.get("confidence", 0) >= self.config.min_confidence]
        classified = classified[: self.config.max_findings]
        outputs = generate_outputs(classified)
        classified_outputs = generate_classified_outputs(classified)
        now_ms = self.clock() * 1000
        # aggregates (report.ts assembleReport): per-signal stats, top
        # agents per signal, chain time range, classified-finding count
        signal_stats: Dict[str, int] = {}
        agents_by_signal: Dict[str, Dict[str, int]] = {}
        for f in classified:
            sig = f.get("signalType", "unknown")
            signal_stats[sig] = signal_stats.get(sig, 0) + 1
            agent = f.get("agent", "unknown")
            agents_by_signal.setdefault(sig, {})[agent] = (
                agents_by_signal.get(sig, {}).get(agent, 0) + 1
            )
        top_agents = {
            sig: [
                {"agent": a, "count": n}
                for a, n in sorted(counts.items(), key=lambda kv: -kv[1])[:5]
            ]
            for sig, counts in agents_by_signal.items()
        }
        time_range = (
            {"start": min(c.start_ts for c in chains),
             "end": max(c.end_ts for c in chains)}
            if chains else None
        )
        report = {
            "version": 1,
            "generatedAt": int(now_ms),
            "eventsAnalyzed": len(events),
            "chains": len(chains),
            "stats": {
                "events": len(events),
                "chains": len(chains),
                "findings": len(classified),
                "findingsClassified": sum(
                    1 for f in classified if f.get("classification") is not None
                ),
            },
            "signalStats": signal_stats,
            "topAgents": top_agents,
            "timeRange": time_range,
            "findings": classified,
            "outputs": outputs,
            "classifiedOutputs": classified_outputs,
        }
        self.state = {
            "lastProcessedTs": max([e.ts for e in events], default=last_ts),
            "lastProcessedSeq": max([e.seq for e in events], default=self.state.get("lastProcessedSeq", 0)),
            "lastRunAt": int(now_ms),
            "runsCompleted": int(self.state.get("runsCompleted", 0)) + 1,
            "totalEventsAnalyzed": int(self.state.get("totalEventsAnalyzed", 0)) + len(events),
            "totalFindings": int(self.state.get("totalFindings", 0)) + len(classified),
        }
        save_json(self.report_path, report)
        save_json(self.state_path, self.state)
        return report
