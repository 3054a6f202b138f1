"""The `/brainplex` dashboard (brainplex README "Dashboard": trust
scores, governance stats, notable events, shield score — one command).

Registers a plugin whose `/brainplex` command aggregates whatever suite
plugins are live on the same gateway: `governance.status` +
`governance.trust` gateway methods, `eventstore.status`, cortex status,
Leuko health, Membrane stats. Absent modules render as "not installed"
instead of failing — the dashboard is read-only glue.
"""

from __future__ import annotations

from typing import Any, Callable, Dict, List, Optional

from ..core.api import PluginApi

PLUGIN_ID = "brainplex"


def _shield_score(gov_status: Optional[Dict[str, Any]],
                  events_status: Optional[Dict[str, Any]]) -> int:
    """0-100 composite: governance activity quality + event-store health.
    (The reference surfaces a 'shield score' on the dashboard; the
    composition here is: start at 100, lose points for denials-heavy
    traffic, publish failures and disconnects.)"""
    score = 100.0
    if gov_status:
        ev = gov_status.get("evaluations") or 0
        denies = gov_status.get("denied") or 0
        if ev:
            score -= min(40.0, 100.0 * denies / max(ev, 1) * 2.0)
    if events_status:
        score -= min(20.0, 2.0 * float(events_status.get("publishFailures") or 0))
        score -= min(10.0, 2.0 * float(events_status.get("disconnectCount") or 0))
        if not events_status.get("connected", True):
            score -= 10.0
    return max(0, min(100, round(score)))


def _norm_gov(raw):
    """governance.status returns {stats:{evaluations,denies,avgEvaluationUs},
    policies,...} (engine.status()); also accept a flat shape."""
    if not isinstance(raw, dict):
        return None
    stats = raw.get("stats") if isinstance(raw.get("stats"), dict) else raw
    return {
        "evaluations": stats.get("evaluations", 0),
        "denied": stats.get("denies", stats.get("denied", 0)),
        "avgEvaluationUs": stats.get("avgEvaluationUs", stats.get("avg_us", 0)),
        "policies": raw.get("policies"),
    }


def _norm_trust(raw):
    """governance.trust returns {agentId: {score, tier, ...}} (trust
    snapshot); also accept {"agents": [...]}."""
    if not isinstance(raw, dict):
        return None
    if isinstance(raw.get("agents"), list):
        return raw
    agents = [
        {"agentId": aid, "score": rec.get("score", 0), "tier": rec.get("tier", "?")}
        for aid, rec in raw.items() if isinstance(rec, dict) and "score" in rec
    ]
    return {"agents": agents}


def _norm_cortex(raw):
    """cortex.status returns {threads: {open, closed, mood, ...},
    decisions, commitments} (CortexWorkspace.status())."""
    if not isinstance(raw, dict):
        return None
    th = raw.get("threads")
    if isinstance(th, dict):
        return {
            "openThreads": th.get("open", 0),
            "decisions": raw.get("decisions", 0),
            "sessionMood": th.get("mood", "neutral"),
        }
    return raw


def _norm_leuko(raw):
    """leuko.report returns {health: {overall, details}, items: [...]}."""
    if not isinstance(raw, dict):
        return None
    health = raw.get("health")
    if isinstance(health, dict):
        notable = [i.get("summary", i.get("message", str(i)))
                   for i in (raw.get("items") or [])
                   if isinstance(i, dict) and i.get("severity") in ("warn", "critical")]
        return {"status": health.get("overall", "unknown"), "notable": notable}
    return raw


class BrainplexDashboard:
    """Renders the dashboard from the gateway method registry."""

    def __init__(self, gateway_methods: Dict[str, Callable], logger=None):
        self.gateway = gateway_methods
        self.logger = logger

    def _call(self, name: str, *args, **kw) -> Optional[Any]:
        fn = self.gateway.get(name)
        if fn is None:
            return None
        try:
            return fn(*args, **kw)
        except Exception as exc:
            if self.logger:
                self.logger.warn("[brainplex] %s failed: %s", name, exc)
            return None

    def render(self) -> str:
        gov = _norm_gov(self._call("governance.status"))
        trust = _norm_trust(self._call("governance.trust"))
        events = self._call("eventstore.status")
        cortex = _norm_cortex(self._call("cortex.status"))
        leuko = _norm_leuko(self._call("leuko.health") or self._call("leuko.report"))
        membrane = self._call("membrane.stats")

        lines: List[str] = ["# 🧠 Brainplex Dashboard", ""]

        lines.append(f"**Shield score:** {_shield_score(gov, events)}/100")
        lines.append("")

        lines.append("## 🔒 Governance")
        if gov:
            lines.append(
                f"- evaluations: {gov.get('evaluations', 0)} | "
                f"denied: {gov.get('denied', 0)} | "
                f"avg latency: {gov.get('avgEvaluationUs', gov.get('avg_us', 0))} µs"
            )
            pols = gov.get("policies")
            if pols is not None:
                lines.append(f"- policies: {len(pols) if isinstance(pols, list) else pols}")
        else:
            lines.append("- not installed")

        lines.append("")
        lines.append("## 🤝 Trust")
        agents = (trust or {}).get("agents") if isinstance(trust, dict) else None
        if agents:
            for a in sorted(agents, key=lambda x: -float(x.get("score", 0)))[:8]:
                lines.append(
                    f"- {a.get('agentId', a.get('id', '?'))}: "
                    f"{round(float(a.get('score', 0)), 1)} ({a.get('tier', '?')})"
                )
        else:
            lines.append("- no agents tracked")

        lines.append("")
        lines.append("## 📨 Event store")
        if events:
            lines.append(
                f"- connected: {'yes' if events.get('connected') else 'no'} | "
                f"stream: {events.get('stream') or 'n/a'} | "
                f"messages: {events.get('messages', 'n/a')} | "
                f"publish failures: {events.get('publishFailures', 0)}"
            )
        else:
            lines.append("- not installed")

        lines.append("")
        lines.append("## 🧠 Cortex")
        if cortex:
            lines.append(
                f"- open threads: {cortex.get('openThreads', cortex.get('open_threads', 0))} | "
                f"decisions: {cortex.get('decisions', 0)} | "
                f"mood: {cortex.get('sessionMood', cortex.get('mood', 'neutral'))}"
            )
        else:
            lines.append("- not installed")

        if membrane:
            lines.append("")
            lines.append("## 💾 Membrane")
            lines.append(
                f"- memories: {membrane.get('records', membrane.get('memories', 0))} | "
                f"retrievals: {membrane.get('retrievals', 0)}"
            )

        lines.append("")
        lines.append("## 🩺 Health (Leuko)")
        if leuko:
            status = leuko.get("status", leuko.get("health", "unknown"))
            lines.append(f"- status: {status}")
            for n in (leuko.get("notable") or leuko.get("anomalies") or [])[:5]:
                lines.append(f"- ⚠️ {n if isinstance(n, str) else n.get('message', n)}")
        else:
            lines.append("- not installed")

        return "\n".join(lines)


class BrainplexPlugin:
    id = PLUGIN_ID
    name = "Brainplex"
    description = "Suite dashboard: trust, governance, events, health in one command"
    version = "0.1.0"

    def register(self, api: PluginApi) -> None:
        dash = BrainplexDashboard(api.gateway_methods, logger=api.logger)

        def cmd(*a, **kw) -> Dict[str, Any]:
            return {"text": dash.render()}

        api.register_command("brainplex", cmd)
        api.register_gateway_method("brainplex.dashboard", lambda *a, **kw: dash.render())


def create_plugin() -> BrainplexPlugin:
    return BrainplexPlugin()
