"""Leuko anomaly detection: trends over time-series health metrics.

Capability target: the external Leuko plugin (reference `README.md:19`,
brainplex README §Leuko): anomaly detection over directory growth,
declining metrics, trend analysis; bootstrap integrity checks; pipeline
failure correlation. Sitrep (§2.5 of SURVEY.md) is the in-repo ancestor;
this module adds the Leuko-only layer on top of its collectors.

History persists as JSONL under the workspace; detection uses
least-squares slope over a sliding window — no external deps.
"""

from __future__ import annotations

import json
import os
import time
from typing import Dict, List, Optional

from .collectors import item, result


class MetricHistory:
    """Append-only time series per metric name (JSONL, one per line)."""

    def __init__(self, path: str, max_points: int = 500, clock=time.time):
        self.path = path
        self.max_points = max_points
        self._clock = clock
        self._series: Dict[str, List[tuple]] = {}
        self._load()

    def _load(self) -> None:
        if not os.path.isfile(self.path):
            return
        try:
            with open(self.path, "r", encoding="utf-8") as fh:
                for line in fh:
                    try:
                        rec = json.loads(line)
                        self._series.setdefault(rec["name"], []).append(
                            (float(rec["ts"]), float(rec["value"]))
                        )
                    except (json.JSONDecodeError, KeyError, ValueError):
                        continue
        except OSError:
            return
        for name in self._series:
            self._series[name] = self._series[name][-self.max_points :]

    def record(self, name: str, value: float, ts: Optional[float] = None) -> None:
        ts = self._clock() if ts is None else ts
        self._series.setdefault(name, []).append((ts, value))
        self._series[name] = self._series[name][-self.max_points :]
        os.makedirs(os.path.dirname(os.path.abspath(self.path)), exist_ok=True)
        try:
            with open(self.path, "a", encoding="utf-8") as fh:
                fh.write(json.dumps({"ts": ts, "name": name, "value": value}) + "\n")
        except OSError:
            pass

    def series(self, name: str) -> List[tuple]:
        return list(self._series.get(name, ()))

    def slope_per_hour(self, name: str, window: int = 20) -> Optional[float]:
        """Least-squares slope (units/hour) over the last `window` points."""
        pts = self._series.get(name, ())[-window:]
        if len(pts) < 3:
            return None
        xs = [(t - pts[0][0]) / 3600.0 for t, _ in pts]
        ys = [v for _, v in pts]
        n = len(pts)
        mx, my = sum(xs) / n, sum(ys) / n
        den = sum((x - mx) ** 2 for x in xs)
        if den == 0:
            return None
        return sum((x - mx) * (y - my) for x, y in zip(xs, ys)) / den


def dir_size_bytes(path: str) -> int:
    total = 0
    for root, _dirs, files in os.walk(path):
        for f in files:
            try:
                total += os.path.getsize(os.path.join(root, f))
            except OSError:
                continue
    return total


class AnomalyDetector:
    """Leuko detections: directory growth, declining metrics, failure
    correlation, bootstrap integrity."""

    def __init__(self, history: MetricHistory, clock=time.time):
        self.history = history
        self._clock = clock

    # -- directory growth ---------------------------------------------------
    def check_directory_growth(
        self, path: str, warn_mb_per_hour: float = 100.0
    ) -> Optional[Dict]:
        name = f"dirsize:{path}"
        self.history.record(name, float(dir_size_bytes(path)))
        slope = self.history.slope_per_hour(name)
        if slope is None:
            return None
        mb_h = slope / 1e6
        if mb_h > warn_mb_per_hour:
            return item(
                f"leuko-growth-{os.path.basename(path)}", "leuko", "warn", "needs_owner",
                f"Directory {path} growing {mb_h:.0f} MB/h (> {warn_mb_per_hour:.0f})",
                65, detail=f"slope={slope:.0f} B/h over {len(self.history.series(name))} samples",
            )
        return None

    # -- declining metric ---------------------------------------------------
    def check_declining_metric(
        self, name: str, value: float, warn_decline_per_hour: float
    ) -> Optional[Dict]:
        """Record `value`; warn when it declines faster than the threshold
        (e.g. messages/sec throughput, trust scores)."""
        self.history.record(name, value)
        slope = self.history.slope_per_hour(name)
        if slope is not None and slope < -abs(warn_decline_per_hour):
            return item(
                f"leuko-decline-{name}", "leuko", "warn", "needs_owner",
                f"Metric {name} declining {abs(slope):.2f}/h", 60,
                detail=f"current={value}",
            )
        return None

    # -- pipeline failure correlation ---------------------------------------
    def correlate_failures(self, journal, window_s: float = 600.0,
                           threshold: int = 3) -> Optional[Dict]:
        """N+ tool/run failures across agents inside one window -> one
        correlated incident instead of N alerts."""
        if journal is None:
            return None
        now = self._clock()
        fails = []
        for _seq, env in journal.replay(since_ts=now - window_s):
            t = env.get("canonicalType") or env.get("type", "")
            if t in ("tool.call.failed", "run.failed") or (
                t == "tool.result" and (env.get("payload") or {}).get("error")
            ):
                fails.append(env)
        if len(fails) < threshold:
            return None
        agents = sorted({e.get("agent", "?") for e in fails})
        return item(
            "leuko-failure-correlation", "leuko", "critical", "needs_owner",
            f"{len(fails)} failures across {len(agents)} agent(s) in {int(window_s / 60)} min",
            95, detail=f"agents: {', '.join(agents)}",
        )

    # -- bootstrap integrity -------------------------------------------------
    def check_bootstrap_integrity(self, workspace: str,
                                  required: Optional[List[str]] = None) -> List[Dict]:
        """Verify the suite's persistence roots exist and parse."""
        required = required or [
            "governance/trust.json",
            "memory/reboot/threads.json",
            "facts.json",
        ]
        items = []
        for rel in required:
            p = os.path.join(workspace, rel)
            if not os.path.isfile(p):
                continue  # not yet created = fine (first boot)
            try:
                with open(p, "r", encoding="utf-8") as fh:
                    json.load(fh)
            except (OSError, json.JSONDecodeError) as exc:
                items.append(item(
                    f"leuko-integrity-{rel.replace('/', '-')}", "leuko", "critical",
                    "needs_owner", f"State file corrupt: {rel}", 100, detail=str(exc),
                ))
        return items

    def run_all(self, config: Dict, journal=None, workspace: str = ".") -> Dict:
        """Collector-shaped entry so the aggregator can schedule it."""
        items: List[Dict] = []
        for d in config.get("watchDirs", []):
            it = self.check_directory_growth(
                d, config.get("dirGrowthWarnMbPerHour", 100.0)
            )
            if it:
                items.append(it)
        it = self.correlate_failures(
            journal,
            config.get("failureWindowS", 600.0),
            config.get("failureThreshold", 3),
        )
        if it:
            items.append(it)
        items.extend(self.check_bootstrap_integrity(workspace, config.get("requiredFiles")))
        status = "critical" if any(i["severity"] == "critical" for i in items) else (
            "warn" if items else "ok"
        )
        return result(status, items, f"{len(items)} anomaly(ies)" if items else "no anomalies")
