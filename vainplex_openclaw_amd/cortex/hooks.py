"""Cortex hook wiring + plugin entry + agent tools.

Parity target: cortex `src/hooks.ts` — message_received / message_sent /
agent_end fallback (`:113-164`), session_start boot context (`:167-182`),
before/after_compaction (`:185-213`), per-workspace tracker map
(`:57-68`), hook diagnostics counters (`:31-77`); `index.ts` — tool +
/cortexstatus registration (`:32-84`); `src/tools/*` — agent-callable
threads/decisions/commitments/search/status tools.
"""

from __future__ import annotations

import time
from typing import Any, Dict, List, Optional

from ..core.api import PluginApi, PluginLogger, NullLogger
from ..core.config import load_raw_layered
from .boot_context import BootContextConfig, BootContextGenerator
from .commitment_tracker import CommitmentTracker, CommitmentTrackerConfig
from .decision_tracker import DecisionTracker, DecisionTrackerConfig
from .llm_enhance import LlmEnhancer
from .pre_compaction import PreCompaction
from .thread_tracker import ThreadTracker, ThreadTrackerConfig


class CortexWorkspace:
    """All trackers for one workspace (hooks.ts per-workspace map)."""

    def __init__(self, workspace: str, config: Dict[str, Any], clock=time.time):
        language = config.get("language", "both")
        if config.get("customPatterns"):
            from . import patterns as _P

            _P.set_custom_patterns(config["customPatterns"])
        self.workspace = workspace
        self.threads = ThreadTracker(
            workspace,
            ThreadTrackerConfig(
                enabled=config.get("threadTracker", {}).get("enabled", True),
                prune_days=config.get("threadTracker", {}).get("pruneDays", 14),
                max_threads=config.get("threadTracker", {}).get("maxThreads", 50),
            ),
            language,
            clock=clock,
        )
        self.decisions = DecisionTracker(workspace, DecisionTrackerConfig(), language, clock=clock)
        self.commitments = CommitmentTracker(workspace, CommitmentTrackerConfig(), clock=clock)
        self.pre_compaction = PreCompaction(workspace, clock=clock)
        self.boot = BootContextGenerator(workspace, BootContextConfig(), clock=clock)
        self.enhancer = LlmEnhancer(config.get("llm"))

    def process_message(self, content: str, sender: str) -> None:
        """The "always runs — zero cost" regex path (hooks.ts:80-110)."""
        if not content:
            return
        self.threads.process_message(content, sender)
        self.decisions.process_message(content, sender)
        self.commitments.process_message(content, sender)
        self.pre_compaction.observe(sender, content)
        analysis = self.enhancer.add_message(content)
        if analysis:
            self.threads.apply_llm_analysis(analysis)

    def flush(self) -> None:
        self.threads.flush()
        self.decisions.flush()
        self.commitments.flush()

    def status(self) -> Dict[str, Any]:
        threads = self.threads.get_threads()
        return {
            "workspace": self.workspace,
            "threads": {
                "open": sum(1 for t in threads if t["status"] == "open"),
                "closed": sum(1 for t in threads if t["status"] == "closed"),
                "mood": self.threads.session_mood,
                "eventsProcessed": self.threads.events_processed,
            },
            "decisions": len(self.decisions.decisions),
            "commitments": {
                "open": len(self.commitments.open_commitments()),
                "overdue": len(self.commitments.overdue()),
            },
        }


class CortexHooks:
    def __init__(self, config: Dict[str, Any], workspace: str, logger: Optional[PluginLogger] = None, clock=time.time):
        self.config = config
        self.default_workspace = workspace
        self.logger = logger or NullLogger()
        self.clock = clock
        self.workspaces: Dict[str, CortexWorkspace] = {}
        self.diagnostics: Dict[str, Dict[str, Any]] = {}

    def ws(self, workspace: Optional[str] = None) -> CortexWorkspace:
        key = workspace or self.default_workspace
        if key not in self.workspaces:
            self.workspaces[key] = CortexWorkspace(key, self.config, clock=self.clock)
        return self.workspaces[key]

    def _diag(self, hook: str) -> None:
        d = self.diagnostics.setdefault(hook, {"fires": 0, "errors": 0, "lastFired": 0})
        d["fires"] += 1
        d["lastFired"] = self.clock()

    # -- handlers ----------------------------------------------------------
    def on_message_received(self, ev: Dict[str, Any]) -> None:
        self._diag("message_received")
        self.ws(ev.get("workspace")).process_message(str(ev.get("content") or ""), ev.get("sender", "user"))

    def on_message_sent(self, ev: Dict[str, Any]) -> None:
        self._diag("message_sent")
        self.ws(ev.get("workspace")).process_message(str(ev.get("content") or ""), ev.get("sender", "agent"))

    def on_agent_end(self, ev: Dict[str, Any]) -> None:
        """Fallback when message_sent isn't delivered (hooks.ts:141-164)."""
        self._diag("agent_end")
        msgs = ev.get("messages") or []
        if msgs:
            last = msgs[-1]
            content = last.get("content") if isinstance(last, dict) else str(last)
            self.ws(ev.get("workspace")).process_message(str(content or ""), "agent")

    def on_session_start(self, ev: Dict[str, Any]) -> Dict[str, Any]:
        self._diag("session_start")
        ctx = self.ws(ev.get("workspace")).boot.generate()
        return {"bootContext": ctx}

    def on_before_compaction(self, ev: Dict[str, Any]) -> None:
        self._diag("before_compaction")
        w = self.ws(ev.get("workspace"))
        w.pre_compaction.run(w.threads, w.decisions, w.commitments)

    def on_after_compaction(self, ev: Dict[str, Any]) -> None:
        self._diag("after_compaction")
        self.ws(ev.get("workspace")).boot.generate()

    def on_gateway_stop(self, ev: Dict[str, Any]) -> None:
        for w in self.workspaces.values():
            w.flush()

    # -- tools (src/tools/*) ----------------------------------------------
    def tool_threads(self, workspace: Optional[str] = None, status: Optional[str] = None) -> List[Dict[str, Any]]:
        threads = self.ws(workspace).threads.get_threads()
        if status:
            threads = [t for t in threads if t.get("status") == status]
        return threads

    def tool_decisions(self, workspace: Optional[str] = None, n: int = 10) -> List[Dict[str, Any]]:
        return self.ws(workspace).decisions.recent(n)

    def tool_commitments(self, workspace: Optional[str] = None) -> List[Dict[str, Any]]:
        return self.ws(workspace).commitments.open_commitments()

    def tool_search(self, query: str, workspace: Optional[str] = None) -> Dict[str, Any]:
        w = self.ws(workspace)
        q = query.lower()
        return {
            "threads": [t for t in w.threads.get_threads() if q in t.get("title", "").lower()],
            "decisions": [d for d in w.decisions.decisions if q in str(d.get("what", "")).lower()],
            "commitments": [c for c in w.commitments.commitments if q in str(c.get("action", "")).lower()],
        }

    def tool_status(self, workspace: Optional[str] = None) -> Dict[str, Any]:
        return self.ws(workspace).status()


class CortexPlugin:
    id = "openclaw-cortex"
    name = "Cortex"
    description = "Conversation intelligence: thread/decision/commitment tracking, boot context"
    version = "0.1.0"

    def __init__(self, workspace: Optional[str] = None, journal=None, call_llm=None):
        self.workspace = workspace
        self.hooks: Optional[CortexHooks] = None
        self.journal = journal
        self.call_llm = call_llm
        self.analyzer = None
        self._analyzer_timer = None

    def register(self, api: PluginApi) -> None:
        config = load_raw_layered(self.id, api.plugin_config)
        workspace = self.workspace or config.get("workspace") or "."
        h = CortexHooks(config, workspace, api.logger)
        self.hooks = h
        api.on("message_received", h.on_message_received, priority=100)
        api.on("message_sent", h.on_message_sent, priority=100)
        api.on("agent_end", h.on_agent_end, priority=100)
        api.on("session_start", h.on_session_start, priority=50)
        api.on("before_compaction", h.on_before_compaction, priority=100)
        api.on("after_compaction", h.on_after_compaction, priority=100)
        api.on("gateway_stop", h.on_gateway_stop, priority=100)
        api.register_command("cortexstatus", h.tool_status)
        api.register_gateway_method("cortex.status", lambda *a, **kw: h.tool_status())
        api.register_command("cortex.threads", h.tool_threads)
        api.register_command("cortex.decisions", h.tool_decisions)
        api.register_command("cortex.commitments", h.tool_commitments)
        api.register_command("cortex.search", h.tool_search)
        self._register_trace_analyzer(api, config, workspace)

    def _register_trace_analyzer(self, api: PluginApi, config, workspace: str) -> None:
        """Conditional trace-analyzer registration: analyze + status
        commands, signal-registry preload, optional schedule timer
        (reference cortex hooks.ts:242-253, trace-analyzer/hooks.ts)."""
        ta_cfg = config.get("traceAnalyzer") or {}
        if not ta_cfg.get("enabled", self.journal is not None):
            return
        from .trace.analyzer import AnalyzerConfig, JournalTraceSource, TraceAnalyzer, create_nats_source
        from .trace.signals_lang import default_registry

        source = None
        if self.journal is not None:
            source = JournalTraceSource(self.journal)
        else:
            nats_cfg = ta_cfg.get("nats") or {}
            source = create_nats_source(
                nats_cfg.get("url") or ta_cfg.get("natsUrl"), logger=api.logger)
        if source is None:
            api.logger.info("[cortex] trace analyzer: no event source available")
            return
        registry = default_registry()
        api.logger.info("[trace-analyzer] Loaded signal patterns for: "
                        + ", ".join(registry.loaded_languages()))
        if "signals" in ta_cfg:   # resolved trace-analyzer config shape
            from .trace.config import resolve_trace_analyzer_config

            self.analyzer = TraceAnalyzer.from_config(
                workspace, source, resolve_trace_analyzer_config(ta_cfg),
                call_llm=self.call_llm, signal_registry=registry)
        else:                     # ad-hoc keys (detectors/minConfidence)
            self.analyzer = TraceAnalyzer(
                workspace, source,
                AnalyzerConfig(
                    detectors=ta_cfg.get("detectors"),
                    min_confidence=ta_cfg.get("minConfidence", 0.0),
                ),
                call_llm=self.call_llm,
                signal_registry=registry,
            )
        api.register_command("cortexanalyze", lambda *a, **kw: self.analyzer.run())
        # reference command names (trace-analyzer/hooks.ts:176,182)
        api.register_command("trace-analyze", lambda *a, **kw: self.analyzer.run())
        api.register_gateway_method("cortex.analyze", lambda *a, **kw: self.analyzer.run())
        api.register_command("cortextracestatus", lambda *a, **kw: self.trace_status())
        api.register_command("trace-status", lambda *a, **kw: self.trace_status())
        api.register_gateway_method("cortex.trace.status", lambda *a, **kw: self.trace_status())

        schedule_cfg = ta_cfg.get("schedule") or {}
        interval_min = float(ta_cfg.get("intervalMinutes", 0))
        if schedule_cfg.get("enabled"):
            interval_min = float(schedule_cfg.get("intervalHours", 24)) * 60.0
        if interval_min > 0:
            import threading

            def schedule():
                def fire():
                    try:
                        api.logger.info("[trace-analyzer] Running scheduled analysis...")
                        self.analyzer.run()
                    except Exception as exc:
                        api.logger.warn(f"[trace-analyzer] Scheduled analysis failed: {exc}")
                    schedule()

                self._analyzer_timer = threading.Timer(interval_min * 60.0, fire)
                self._analyzer_timer.daemon = True
                self._analyzer_timer.start()

            schedule()
            api.logger.info(
                f"[trace-analyzer] Scheduled analysis every {interval_min / 60.0:g}h")
            api.on("gateway_stop",
                   lambda ev: self.cleanup_trace_analyzer(),
                   priority=999)

    def trace_status(self) -> Dict[str, Any]:
        """/trace-status: analyzer state summary (trace-analyzer/hooks.ts
        handleTraceStatus)."""
        if self.analyzer is None:
            return {"enabled": False}
        st = self.analyzer.state or {}
        return {
            "enabled": True,
            "scheduled": getattr(self, "_analyzer_timer", None) is not None,
            "runsCompleted": int(st.get("runsCompleted", 0)),
            "lastRunAt": st.get("lastRunAt"),
            "totalEventsAnalyzed": int(st.get("totalEventsAnalyzed", 0)),
            "totalFindings": int(st.get("totalFindings", 0)),
            "reportPath": self.analyzer.report_path,
        }

    def cleanup_trace_analyzer(self) -> None:
        """Cancel the schedule timer and drop the analyzer; safe to call
        repeatedly (trace-analyzer/hooks.ts cleanupTraceAnalyzerHooks)."""
        timer = getattr(self, "_analyzer_timer", None)
        if timer is not None:
            timer.cancel()
            self._analyzer_timer = None
        self.analyzer = None


def create_plugin(workspace: Optional[str] = None, journal=None, call_llm=None) -> CortexPlugin:
    return CortexPlugin(workspace, journal=journal, call_llm=call_llm)
