"""The 7 failure-signal detectors + localized phrase packs.

Parity target: cortex `src/trace-analyzer/signals/` — detectors:
correction, tool-fail, doom-loop (3+ consecutive similar failing tool
calls, Jaccard on params, Levenshtein <=500 chars for exec commands,
doom-loop.ts:49-90), dissatisfied, repeat-fail, hallucination,
unverified-claim; registry runs enabled detectors per chain
(signals/index.ts:1-50); localized correction/dissatisfaction phrases per
language (signals/lang/signal-lang-*.ts, 10 packs).
"""

from __future__ import annotations

import json
import re
import uuid
from dataclasses import dataclass, field
from typing import Any, Callable, Dict, List, Optional

from .chains import ConversationChain
from .events import NormalizedEvent

# -- localized phrase packs (10 languages) ---------------------------------
SIGNAL_LANG: Dict[str, Dict[str, List[str]]] = {
    "en": {
        "correction": ["no,", "that's wrong", "not what i asked", "incorrect", "you misunderstood", "try again", "that is not right"],
        "dissatisfied": ["useless", "this is not helpful", "you keep failing", "frustrating", "give up", "terrible"],
    },
    "de": {
        "correction": ["nein,", "das ist falsch", "nicht was ich meinte", "falsch verstanden", "nochmal", "stimmt nicht"],
        "dissatisfied": ["unbrauchbar", "das hilft nicht", "du scheiterst ständig", "frustrierend", "ich geb auf"],
    },
    "es": {
        "correction": ["no,", "eso está mal", "no es lo que pedí", "incorrecto", "inténtalo de nuevo"],
        "dissatisfied": ["inútil", "esto no ayuda", "sigues fallando", "frustrante"],
    },
    "fr": {
        "correction": ["non,", "c'est faux", "pas ce que j'ai demandé", "incorrect", "réessaie"],
        "dissatisfied": ["inutile", "ça n'aide pas", "tu échoues encore", "frustrant"],
    },
    "it": {
        "correction": ["no,", "è sbagliato", "non è quello che ho chiesto", "riprova"],
        "dissatisfied": ["inutile", "non aiuta", "continui a fallire", "frustrante"],
    },
    "pt": {
        "correction": ["não,", "está errado", "não foi o que pedi", "tente novamente"],
        "dissatisfied": ["inútil", "isso não ajuda", "você continua falhando", "frustrante"],
    },
    "ru": {
        "correction": ["нет,", "это неверно", "не то, что я просил", "попробуй снова"],
        "dissatisfied": ["бесполезно", "это не помогает", "ты снова ошибся", "разочарование"],
    },
    "ja": {
        "correction": ["違います", "間違っています", "そうじゃない", "やり直して"],
        "dissatisfied": ["役に立たない", "助けになりません", "また失敗"],
    },
    "ko": {
        "correction": ["아니요", "틀렸습니다", "그게 아니에요", "다시 해보세요"],
        "dissatisfied": ["쓸모없", "도움이 안", "또 실패"],
    },
    "zh": {
        "correction": ["不对", "错了", "不是我要的", "再试一次"],
        "dissatisfied": ["没用", "帮不上忙", "又失败了"],
    },
}


def _phrases(kind: str) -> List[str]:
    out: List[str] = []
    for pack in SIGNAL_LANG.values():
        out.extend(pack.get(kind, []))
    return out


@dataclass
class Finding:
    id: str
    signal_type: str
    chain_id: str
    agent: str
    session: str
    severity: str  # low / medium / high
    summary: str
    evidence: Dict[str, Any] = field(default_factory=dict)
    confidence: float = 0.5

    def to_dict(self) -> Dict[str, Any]:
        return {
            "id": self.id,
            "signalType": self.signal_type,
            "chainId": self.chain_id,
            "agent": self.agent,
            "session": self.session,
            "severity": self.severity,
            "summary": self.summary,
            "evidence": self.evidence,
            "confidence": self.confidence,
        }


def _mk(chain: ConversationChain, sig: str, severity: str, summary: str, evidence: Dict[str, Any], conf: float) -> Finding:
    return Finding(
        id=f"find-{uuid.uuid4().hex[:12]}",
        signal_type=sig,
        chain_id=chain.id,
        agent=chain.agent,
        session=chain.session,
        severity=severity,
        summary=summary,
        evidence=evidence,
        confidence=conf,
    )


# -- helpers ----------------------------------------------------------------

def jaccard_similarity(a: Dict[str, Any], b: Dict[str, Any]) -> float:
    """Jaccard on stringified key=value pairs, ignoring volatile fields
    (doom-loop.ts:49-74)."""
    volatile = {"timeout", "timestamp", "ts"}
    sa = {f"{k}={json.dumps(v, sort_keys=True, default=str)}" for k, v in a.items() if k not in volatile}
    sb = {f"{k}={json.dumps(v, sort_keys=True, default=str)}" for k, v in b.items() if k not in volatile}
    union = sa | sb
    if not union:
        return 1.0
    return len(sa & sb) / len(union)


def levenshtein(a: str, b: str, cap: int = 500) -> int:
    """Edit distance, operands capped at 500 chars (doom-loop.ts:76-90).
    This is the CPU reference of the batched wavefront-DP GPU kernel
    (csrc/edit_distance.hip, batch path)."""
    sa, sb = a[:cap], b[:cap]
    if sa == sb:
        return 0
    if not sa:
        return len(sb)
    if not sb:
        return len(sa)
    prev = list(range(len(sa) + 1))
    for j, cb in enumerate(sb, 1):
        cur = [j]
        for i, ca in enumerate(sa, 1):
            cur.append(min(prev[i] + 1, cur[i - 1] + 1, prev[i - 1] + (ca != cb)))
        prev = cur
    return prev[len(sa)]


def _tool_attempts(chain: ConversationChain) -> List[Dict[str, Any]]:
    """(tool.call, tool.result) adjacent pairs (doom-loop.ts extractAttempts)."""
    out = []
    evs = chain.events
    for i in range(len(evs) - 1):
        if evs[i].type == "tool.call" and evs[i + 1].type == "tool.result":
            res = evs[i + 1].payload
            out.append({
                "idx": i,
                "toolName": evs[i].payload.get("toolName") or "",
                "params": evs[i].payload.get("toolParams") or {},
                "error": res.get("toolError") or "",
                "isError": bool(res.get("toolError")) or res.get("toolIsError") is True,
            })
    return out


def _similar_attempts(a: Dict[str, Any], b: Dict[str, Any]) -> bool:
    if a["toolName"] != b["toolName"]:
        return False
    if a["toolName"] == "exec":
        ca = str(a["params"].get("command", ""))
        cb = str(b["params"].get("command", ""))
        longest = max(len(ca), len(cb), 1)
        return levenshtein(ca, cb) / longest < 0.3
    return jaccard_similarity(a["params"], b["params"]) >= 0.6


# -- detectors --------------------------------------------------------------

def detect_correction(chain: ConversationChain, registry=None) -> List[Finding]:
    """User messages that correct the agent right after an agent message.
    A SignalPatternRegistry (signals_lang) extends the builtin phrase
    tables with its merged correction indicators — runtime-registered
    language packs feed detection (signals/lang/index.ts)."""
    findings = []
    phrases = _phrases("correction")
    extra = registry.get_patterns()["correction"]["indicators"] if registry else []
    for i, ev in enumerate(chain.events):
        if ev.type != "msg.in":
            continue
        content = str(ev.payload.get("content") or "").lower()
        if any(p in content for p in phrases) or any(rx.search(content) for rx in extra):
            prior_agent = any(e.type == "msg.out" for e in chain.events[:i])
            if prior_agent:
                findings.append(_mk(chain, "correction", "medium",
                                    "User corrected the agent",
                                    {"message": content[:200]}, 0.6))
    return findings


def detect_tool_fail(chain: ConversationChain) -> List[Finding]:
    attempts = _tool_attempts(chain)
    fails = [a for a in attempts if a["isError"]]
    if not fails:
        return []
    rate = len(fails) / len(attempts)
    if len(fails) >= 3 or rate > 0.5:
        return [_mk(chain, "tool_fail", "medium" if rate <= 0.5 else "high",
                    f"{len(fails)}/{len(attempts)} tool calls failed",
                    {"failures": [f["toolName"] for f in fails][:10], "rate": round(rate, 2)}, 0.7)]
    return []


def detect_doom_loop(chain: ConversationChain) -> List[Finding]:
    """3+ consecutive similar failing tool calls (doom-loop.ts)."""
    attempts = _tool_attempts(chain)
    findings = []
    run: List[Dict[str, Any]] = []
    for a in attempts:
        if not a["isError"]:
            if len(run) >= 3:
                findings.append(_mk(chain, "doom_loop", "high",
                                    f"Doom loop: {len(run)} consecutive similar failing "
                                    f"calls to {run[0]['toolName']}",
                                    {"toolName": run[0]["toolName"], "count": len(run),
                                     "lastError": str(run[-1]["error"])[:200]}, 0.85))
            run = []
            continue
        if run and _similar_attempts(run[-1], a):
            run.append(a)
        else:
            if len(run) >= 3:
                findings.append(_mk(chain, "doom_loop", "high",
                                    f"Doom loop: {len(run)} consecutive similar failing "
                                    f"calls to {run[0]['toolName']}",
                                    {"toolName": run[0]["toolName"], "count": len(run),
                                     "lastError": str(run[-1]["error"])[:200]}, 0.85))
            run = [a]
    if len(run) >= 3:
        findings.append(_mk(chain, "doom_loop", "high",
                            f"Doom loop: {len(run)} consecutive similar failing calls to {run[0]['toolName']}",
                            {"toolName": run[0]["toolName"], "count": len(run),
                             "lastError": str(run[-1]["error"])[:200]}, 0.85))
    return findings


def detect_dissatisfied(chain: ConversationChain, registry=None) -> List[Finding]:
    phrases = _phrases("dissatisfied")
    pats = registry.get_patterns()["dissatisfaction"] if registry else None
    for ev in chain.events:
        if ev.type != "msg.in":
            continue
        content = str(ev.payload.get("content") or "").lower()
        hit = any(p in content for p in phrases) or (
            pats and any(rx.search(content) for rx in pats["indicators"]))
        if hit and pats and any(rx.search(content) for rx in pats["satisfactionOverrides"]):
            continue  # "thanks, forget it" reads as satisfied (lang packs)
        if hit:
            return [_mk(chain, "dissatisfied", "high", "User expressed dissatisfaction",
                        {"message": content[:200]}, 0.7)]
    return []


def detect_repeat_fail(chain: ConversationChain) -> List[Finding]:
    """Same tool failing with the same error class, non-consecutively."""
    attempts = _tool_attempts(chain)
    by_key: Dict[tuple, int] = {}
    for a in attempts:
        if a["isError"]:
            key = (a["toolName"], str(a["error"])[:80])
            by_key[key] = by_key.get(key, 0) + 1
    findings = []
    for (tool, err), n in by_key.items():
        if n >= 2:
            findings.append(_mk(chain, "repeat_fail", "medium",
                                f"Tool {tool} failed {n} times with the same error",
                                {"toolName": tool, "error": err, "count": n}, 0.6))
    return findings


_ABS_CLAIM_RX = re.compile(
    r"\b(?:definitely|certainly|guaranteed|always|never fails|100%|no doubt)\b", re.I
)

# localized completion-claim words (multilang hallucination detection:
# signal-lang packs in the reference's multilang-detectors suite)
_COMPLETION_RX = re.compile(
    "|".join([
        r"\b(?:done|completed|finished|fixed)\b",
        r"\b(?:fertig|erledigt|abgeschlossen)\b",
        r"\b(?:terminé|fini|corrigé)\b",
        r"\b(?:completado|terminado|arreglado)\b",
        r"\b(?:completato|finito|risolto)\b",
        r"\b(?:concluído|pronto|consertado)\b",
        r"готово|завершено|исправлено",
        "完了|完成した|修正済み",
        "완료|끝났|수정됨",
        "完成|搞定|修好",
    ]),
    re.I,
)

# localized system-state claim shapes + opinion exclusions
# ("es gibt 5 fehler" counts; "je crois qu'il y a..." does not)
_STATE_CLAIM_RX = re.compile(
    "|".join([
        r"\bthere (?:are|is)\s+\d+\s+error",
        r"\bes gibt\s+\d+\s+fehler",
        r"\bil y a\s+\d+\s+erreurs?",
        r"\bhay\s+\d+\s+errores?",
        r"\bci sono\s+\d+\s+errori",
        r"\bhá\s+\d+\s+erros?",
        r"есть\s+\d+\s+ошиб",
    ]),
    re.I,
)
_OPINION_RX = re.compile(
    "|".join([
        r"\b(?:i think|i believe|maybe|probably|perhaps)\b",
        r"\b(?:ich glaube|ich denke|vielleicht|wahrscheinlich)\b",
        r"\b(?:je crois|je pense|peut-être)\b",
        r"\b(?:creo que|quizás|tal vez)\b",
        r"\b(?:credo che|forse)\b",
        r"\b(?:acho que|talvez)\b",
        r"думаю|наверное|возможно",
        "たぶん|と思う",
        "아마|같아요",
        "可能|大概|我觉得",
    ]),
    re.I,
)


def detect_hallucination(chain: ConversationChain) -> List[Finding]:
    """Agent asserts a fact that the next tool result contradicts
    (simplified form of hallucination.ts: claim followed by error)."""
    findings = []
    evs = chain.events
    for i, ev in enumerate(evs):
        if ev.type != "msg.out":
            continue
        content = str(ev.payload.get("content") or "")
        claims = re.findall(r"([\w.-]{2,40})\s+(?:is|are)\s+(?:running|done|deployed|fixed|available)", content, re.I)
        completion = _COMPLETION_RX.search(content) is not None
        if not claims and not completion:
            continue
        # next tool result errors -> contradiction (subject-matched for
        # the English state claims; any error for localized completion
        # claims, the multilang-detectors contract)
        for nxt in evs[i + 1 : i + 5]:
            if nxt.type == "tool.result" and (nxt.payload.get("toolError") or nxt.payload.get("toolIsError")):
                err = str(nxt.payload.get("toolError") or "")
                subj = next((c for c in claims if c.lower() in err.lower()), None)
                if subj:
                    findings.append(_mk(chain, "hallucination", "high",
                                        f"Agent claimed '{subj}' OK but tool errored",
                                        {"subject": subj, "predicate": "state",
                                         "value": "error", "claim": content[:150],
                                         "error": err[:150]}, 0.65))
                elif completion:
                    findings.append(_mk(chain, "hallucination", "medium",
                                        "Completion claim followed by tool error",
                                        {"claim": content[:150],
                                         "error": err[:150]}, 0.55))
                break
    return findings


def detect_unverified_claim(chain: ConversationChain) -> List[Finding]:
    """Absolute claims in agent output with no tool call backing them."""
    findings = []
    evs = chain.events
    for i, ev in enumerate(evs):
        if ev.type != "msg.out":
            continue
        content = str(ev.payload.get("content") or "")
        state_claim = (_STATE_CLAIM_RX.search(content) is not None
                       and _OPINION_RX.search(content) is None)
        if _ABS_CLAIM_RX.search(content) or state_claim:
            # only a SUCCESSFUL tool result counts as verification
            tool_before = any(
                e.type == "tool.result"
                and not (e.payload.get("toolError") or e.payload.get("toolIsError"))
                for e in evs[max(0, i - 4):i]
            )
            if not tool_before:
                findings.append(_mk(chain, "unverified_claim", "low",
                                    "Absolute claim with no verifying tool call",
                                    {"claim": content[:200]}, 0.5))
    return findings


DETECTORS: Dict[str, Callable[[ConversationChain], List[Finding]]] = {
    "correction": detect_correction,
    "tool_fail": detect_tool_fail,
    "doom_loop": detect_doom_loop,
    "dissatisfied": detect_dissatisfied,
    "repeat_fail": detect_repeat_fail,
    "hallucination": detect_hallucination,
    "unverified_claim": detect_unverified_claim,
}


# detectors that consult the signal-language registry's merged patterns
_REGISTRY_AWARE = {"correction", "dissatisfied"}


def detect_all_signals(
    chains: List[ConversationChain],
    enabled: Optional[List[str]] = None,
    registry=None,
) -> List[Finding]:
    names = enabled if enabled is not None else list(DETECTORS.keys())
    findings: List[Finding] = []
    for chain in chains:
        for name in names:
            fn = DETECTORS.get(name)
            if fn is None:
                continue
            if name in _REGISTRY_AWARE:
                findings.extend(fn(chain, registry=registry))
            else:
                findings.extend(fn(chain))
    return findings
