"""Leuko plugin: periodic sitrep service + /sitrep command.

Parity target: reference `openclaw-sitrep/src/{service,hooks,config}.ts`
— interval service (default 120 min), /sitrep command rendering the
report, config defaults with per-collector merge; plus the Leuko anomaly
layer (anomaly.py) registered as an extra collector.
"""

from __future__ import annotations

import os
import threading
import time
from typing import Any, Dict, Optional

from ..core.api import PluginApi
from ..core.config import load_raw_layered
from .aggregator import generate_sitrep, write_sitrep
from .anomaly import AnomalyDetector, MetricHistory

PLUGIN_ID = "openclaw-leuko"


def default_config(home: Optional[str] = None) -> Dict[str, Any]:
    home = home or os.environ.get("HOME", "/tmp")
    base = os.path.join(home, ".openclaw", "sitrep")
    return {
        "enabled": True,
        "outputPath": os.path.join(base, "sitrep.json"),
        "previousPath": os.path.join(base, "sitrep-previous.json"),
        "intervalMinutes": 120,
        "collectors": {
            "systemd_timers": {"enabled": True},
            "nats": {"enabled": False},
            "goals": {"enabled": False, "goalsPath": ""},
            "goal_quality": {"enabled": False, "goalsPath": "", "maxOpenGoals": 7},
            "threads": {"enabled": False, "threadsPath": ""},
            "errors": {"enabled": False, "patternsPath": ""},
            "calendar": {"enabled": False, "command": ""},
            "gpu_health": {"enabled": True},
        },
        "customCollectors": [],
        "anomaly": {
            "enabled": True,
            "watchDirs": [],
            "dirGrowthWarnMbPerHour": 100.0,
            "failureWindowS": 600.0,
            "failureThreshold": 3,
        },
        "scoring": {
            "criticalWeight": 100,
            "warnWeight": 50,
            "infoWeight": 10,
            "staleThresholdHours": 6,
        },
        "summaryMaxChars": 2000,
    }


def resolve_config(raw: Optional[Dict[str, Any]], home: Optional[str] = None) -> Dict[str, Any]:
    cfg = default_config(home)
    raw = raw or {}
    for key in ("enabled", "outputPath", "previousPath", "intervalMinutes",
                "customCollectors", "summaryMaxChars"):
        if key in raw:
            cfg[key] = raw[key]
    for name, val in (raw.get("collectors") or {}).items():
        if isinstance(val, dict):
            base = cfg["collectors"].get(name, {"enabled": False})
            cfg["collectors"][name] = {**base, **val}
    for section in ("anomaly", "scoring"):
        if isinstance(raw.get(section), dict):
            cfg[section] = {**cfg[section], **raw[section]}
    return cfg


class LeukoPlugin:
    id = PLUGIN_ID
    name = "Leuko"
    description = "Cognitive immune system: health checks, anomaly detection, sitrep"
    version = "0.1.0"

    def __init__(self, workspace: Optional[str] = None, journal=None, clock=time.time,
                 call_llm=None):
        self.workspace = workspace or "."
        self.journal = journal
        self._clock = clock
        self.call_llm = call_llm
        self.config: Dict[str, Any] = {}
        self.detector: Optional[AnomalyDetector] = None
        self._timer: Optional[threading.Timer] = None
        self.last_report: Optional[Dict] = None

    def run_once(self) -> Dict:
        extra = {}
        acfg = self.config.get("anomaly", {})
        if acfg.get("enabled") and self.detector is not None:
            extra["leuko_anomaly"] = lambda: self.detector.run_all(
                acfg, journal=self.journal, workspace=self.workspace
            )
        report = generate_sitrep(self.config, extra=extra, clock=self._clock)
        if self.call_llm is not None and self.config.get("llmRecommendations", True):
            from .quality import llm_recommendations

            report["recommendations"] = llm_recommendations(report, self.call_llm)
        write_sitrep(report, self.config["outputPath"], self.config["previousPath"])
        self.last_report = report
        return report

    def _schedule(self) -> None:
        mins = float(self.config.get("intervalMinutes", 120))

        def fire() -> None:
            try:
                self.run_once()
            except Exception:
                pass
            self._schedule()

        self._timer = threading.Timer(mins * 60.0, fire)
        self._timer.daemon = True
        self._timer.start()

    def stop(self) -> None:
        if self._timer is not None:
            self._timer.cancel()
            self._timer = None

    def register(self, api: PluginApi) -> None:
        self.config = resolve_config(load_raw_layered(PLUGIN_ID, api.plugin_config))
        if not self.config["enabled"]:
            return
        hist_path = os.path.join(self.workspace, "memory", "leuko", "metrics.jsonl")
        self.detector = AnomalyDetector(MetricHistory(hist_path, clock=self._clock),
                                        clock=self._clock)
        if self.journal is not None:
            # in-process journal injected: the event-backbone collector is
            # live by construction (sitrep's `nats` collector needed a CLI)
            ncfg = self.config["collectors"].setdefault("nats", {})
            ncfg["journal"] = self.journal
            ncfg["enabled"] = True

        api.register_service({
            "id": self.id,
            "start": lambda *a: self._schedule(),
            "stop": lambda *a: self.stop(),
        })

        def sitrep_cmd(*a, **kw) -> Dict[str, str]:
            report = self.run_once()
            lines = [
                f"**Sitrep** — {report['health']['overall'].upper()}",
                report["summary"],
            ]
            for it in report["items"][:10]:
                lines.append(f"- [{it['severity']}] {it['title']}")
            return {"text": "\n".join(lines)}

        api.register_command("sitrep", sitrep_cmd)
        api.register_gateway_method("leuko.report", lambda *a, **kw: self.last_report or self.run_once())


def create_plugin(workspace: Optional[str] = None, journal=None) -> LeukoPlugin:
    return LeukoPlugin(workspace, journal)
