"""Leuko collectors: health checks feeding the situation report.

Parity target: reference `openclaw-sitrep/src/collectors/*` +
`collector.ts` (safeCollect wrapper, 10 s shell timeout, readJsonSafe) —
systemd_timers (stale/no-next-trigger + failed units), nats (stream
message count / last-event age; here served by the embedded
eventstore journal), goals (red-zone approved, stale proposals), threads
(stale open threads from cortex threads.json), errors (recent
critical/high patterns), calendar (command lines -> info items), custom
shell collectors with warn/critical thresholds and output flags.

MI355X addition: `gpu_health` — per-device HBM usage via torch and (when
present) rocm-smi temperature/utilization; the Leuko layer in
anomaly.py builds trend detection on top of these.
"""

from __future__ import annotations

import base64
import json
import os
import re
import subprocess
import time
from typing import Callable, Dict, List, Optional


def shell(command: str, timeout: float = 10.0) -> str:
    out = subprocess.run(
        command, shell=True, capture_output=True, text=True, timeout=timeout
    )
    if out.returncode != 0:
        raise RuntimeError(out.stderr.strip() or f"exit {out.returncode}")
    return out.stdout.strip()


def read_json_safe(path: str):
    try:
        if not os.path.isfile(path):
            return None
        with open(path, "r", encoding="utf-8") as fh:
            return json.load(fh)
    except (OSError, json.JSONDecodeError):
        return None


def item(id_: str, source: str, severity: str, category: str, title: str,
         score: float, detail: Optional[str] = None) -> Dict:
    d = {"id": id_, "source": source, "severity": severity,
         "category": category, "title": title, "score": score}
    if detail is not None:
        d["detail"] = detail
    return d


def result(status: str, items: List[Dict], summary: str, error: Optional[str] = None) -> Dict:
    r = {"status": status, "items": items, "summary": summary, "duration_ms": 0}
    if error is not None:
        r["error"] = error
    return r


def _status_from_items(items: List[Dict]) -> str:
    if any(i["severity"] == "critical" for i in items):
        return "critical"
    return "warn" if items else "ok"


def safe_collect(name: str, fn: Callable, config: Dict, logger=None, clock=time.time) -> Dict:
    """Disabled -> ok/disabled; errors -> standardized error result."""
    if not config.get("enabled"):
        return result("ok", [], "disabled")
    start = clock()
    try:
        r = fn(config)
        r["duration_ms"] = int((clock() - start) * 1000)
        return r
    except Exception as exc:
        if logger is not None:
            logger.warn("[leuko] Collector %s failed: %s", name, exc)
        r = result("error", [], f"error: {exc}", error=str(exc))
        r["duration_ms"] = int((clock() - start) * 1000)
        return r


# -- builtins ----------------------------------------------------------------

def collect_systemd_timers(config: Dict) -> Dict:
    items: List[Dict] = []
    try:
        raw = shell("systemctl --user list-timers --all --no-pager --no-legend 2>/dev/null || true")
    except Exception:
        raw = ""
    for line in (l for l in raw.split("\n") if l.strip()):
        parts = re.split(r"\s{2,}", line.strip())
        unit = (parts[4] if len(parts) > 4 else parts[3] if len(parts) > 3 else "unknown")
        timer = unit.replace(".timer", "")
        left = (parts[1] if len(parts) > 1 else "").strip()
        if left in ("n/a", ""):
            items.append(item(f"timer-{timer}-not-scheduled", "systemd_timers", "warn",
                              "auto_fixable", f"Timer {timer} is not scheduled (no next trigger)",
                              50, detail=f"Raw: {line.strip()}"))
    try:
        failed = shell("systemctl --user list-units --state=failed --no-pager --no-legend 2>/dev/null || true")
    except Exception:
        failed = ""
    for fl in (l for l in failed.split("\n") if ".timer" in l or ".service" in l):
        unit = re.sub(r"\.(timer|service)$", "", fl.strip().split()[0]) if fl.strip() else "unknown"
        items.append(item(f"timer-{unit}-failed", "systemd_timers", "critical",
                          "auto_fixable", f"Unit {unit} is in failed state", 100,
                          detail=fl.strip()))
    return result(_status_from_items(items), items,
                  f"{len(items)} timer issue(s)" if items else "all timers healthy")


def collect_journal(config: Dict) -> Dict:
    """Event-backbone health (sitrep's `nats` collector; the embedded
    journal replaces JetStream here — `journal` key injects it)."""
    journal = config.get("journal")
    if journal is None:
        return result("ok", [], "no journal configured")
    st = journal.status()
    items: List[Dict] = []
    max_age_mins = config.get("maxAgeMins", 60)
    last = None
    for _seq, env in journal.replay(since_seq=max(0, journal.last_seq - 1)):
        last = env
    if last is not None:
        age_mins = (time.time() - float(last.get("ts", 0)) / 1000.0) / 60.0
        if age_mins > max_age_mins:
            items.append(item("journal-stale", "nats", "warn", "needs_owner",
                              f"No events for {int(age_mins)} min (threshold {max_age_mins})",
                              60))
    if st["publishFailures"] > 0:
        items.append(item("journal-publish-failures", "nats", "warn", "needs_owner",
                          f"{st['publishFailures']} publish failure(s)", 50))
    return result(_status_from_items(items), items,
                  f"{st['messages']} message(s), seq {st['lastSeq']}")


def collect_goals(config: Dict) -> Dict:
    path = config.get("goalsPath")
    if not path:
        return result("ok", [], "no goalsPath configured")
    data = read_json_safe(path)
    if data is None:
        return result("ok", [], "goals file not found")
    goals = data.get("goals", data) if isinstance(data, dict) else data
    if not isinstance(goals, list):
        return result("ok", [], "no goals")
    items = []
    stale_hours = config.get("staleHours", 48)
    now = time.time()
    for g in goals:
        gid = g.get("id", "unknown")
        title = g.get("title", "Untitled goal")
        if g.get("zone") == "red" and g.get("status") == "approved":
            items.append(item(f"goal-{gid}-red-approved", "goals", "warn", "needs_owner",
                              f"Red-zone goal awaiting manual execution: {title}", 70,
                              detail="Status: approved, Zone: red"))
            continue
        proposed = g.get("proposed_at")
        if proposed and g.get("status") == "proposed":
            try:
                ts = time.mktime(time.strptime(proposed[:19], "%Y-%m-%dT%H:%M:%S"))
            except ValueError:
                continue
            age_h = (now - ts) / 3600.0
            if age_h > stale_hours:
                items.append(item(f"goal-{gid}-stale", "goals", "info", "needs_owner",
                                  f"Goal proposed {round(age_h)}h ago, not yet approved: {title}", 20))
    return result(_status_from_items(items), items,
                  f"{len(items)} goal issue(s)" if items else "goals nominal")


def collect_threads(config: Dict) -> Dict:
    path = config.get("threadsPath")
    if not path:
        return result("ok", [], "no threadsPath configured")
    raw = read_json_safe(path)
    if raw is None:
        return result("ok", [], "threads file not found")
    threads = raw.get("threads", []) if isinstance(raw, dict) else []
    if isinstance(threads, dict):
        threads = list(threads.values())
    items = []
    stale_days = config.get("staleDays", 7)
    now = time.time()
    for t in threads:
        status = t.get("status")
        if status not in ("open", None):
            continue
        last = t.get("last_activity") or t.get("lastActivity")
        ts = 0.0
        if isinstance(last, (int, float)):
            ts = float(last) / (1000.0 if last > 1e11 else 1.0)
        elif isinstance(last, str):
            try:
                ts = time.mktime(time.strptime(last[:19], "%Y-%m-%dT%H:%M:%S"))
            except ValueError:
                ts = 0.0
        age_days = (now - ts) / 86400.0 if ts else 0.0
        if age_days > stale_days:
            items.append(item(f"thread-{t.get('id', 'unknown')}-stale", "threads", "info",
                              "informational",
                              f"Thread stale {round(age_days)}d: {t.get('topic', '?')}", 15))
        elif t.get("priority") == "high":
            items.append(item(f"thread-{t.get('id', 'unknown')}-high", "threads", "info",
                              "delegatable",
                              f"High-priority open thread: {t.get('topic', '?')}", 30))
    return result(_status_from_items(items), items,
                  f"{len(items)} thread issue(s)" if items else "threads nominal")


def collect_errors(config: Dict) -> Dict:
    path = config.get("patternsPath")
    if not path:
        return result("ok", [], "no patternsPath configured")
    data = read_json_safe(path)
    if not isinstance(data, list):
        return result("ok", [], "error patterns file not found")
    recent_hours = config.get("recentHours", 24)
    cutoff = time.time() - recent_hours * 3600.0
    items = []
    for p in data:
        last_seen = p.get("last_seen")
        try:
            ts = time.mktime(time.strptime(str(last_seen)[:19], "%Y-%m-%dT%H:%M:%S")) if last_seen else 0
        except ValueError:
            ts = 0
        if ts <= cutoff or p.get("severity") not in ("critical", "high"):
            continue
        sev = "critical" if p.get("severity") == "critical" else "warn"
        items.append(item(
            f"error-{p.get('id') or str(p.get('pattern', 'unknown'))[:20]}", "errors", sev,
            "needs_owner", f"{str(p.get('severity', '')).upper()}: {p.get('pattern', 'unknown pattern')}",
            100 if sev == "critical" else 60,
            detail=f"Type: {p.get('type', '?')}, Count: {p.get('count', '?')}, Last: {last_seen}",
        ))
        if len(items) >= 10:
            break
    return result(_status_from_items(items), items,
                  f"{len(items)} recent error(s)" if items else "no recent errors")


def collect_calendar(config: Dict) -> Dict:
    cmd = config.get("command")
    if not cmd:
        return result("ok", [], "no calendar command configured")
    try:
        out = shell(cmd)
    except Exception:
        return result("warn", [item("calendar-command-failed", "calendar", "warn",
                                    "informational", "Calendar command failed", 20,
                                    detail=f"Command: {cmd}")],
                      "calendar command failed")
    lines = [l for l in out.split("\n") if l.strip()]
    if not lines:
        return result("ok", [], "no upcoming events")
    items = [
        item("calendar-" + base64.urlsafe_b64encode(l[:50].encode()).decode()[:16],
             "calendar", "info", "informational", l.strip(), 10)
        for l in lines[:10]
    ]
    return result("ok", items, f"{len(lines)} upcoming event(s)")


def collect_gpu_health(config: Dict) -> Dict:
    """MI355X health: HBM usage per device + rocm-smi when present."""
    items: List[Dict] = []
    summary = "no GPU visible"
    try:
        import torch

        if torch.cuda.is_available():
            n = torch.cuda.device_count()
            frac_warn = config.get("memWarnFraction", 0.92)
            used_parts = []
            for d in range(n):
                free, total = torch.cuda.mem_get_info(d)
                used = 1.0 - free / total
                used_parts.append(f"gpu{d}:{used:.0%}")
                if used > frac_warn:
                    items.append(item(f"gpu-{d}-hbm", "gpu_health", "warn", "needs_owner",
                                      f"GPU {d} HBM {used:.0%} used (> {frac_warn:.0%})",
                                      60))
            summary = f"{n} GPU(s): " + ", ".join(used_parts)
    except Exception:
        pass
    return result(_status_from_items(items), items, summary)


def run_custom_collector(defn: Dict) -> Dict:
    """Custom shell command + threshold/output-flag checks (custom.ts)."""
    cid = defn.get("id", "unnamed")
    try:
        out = shell(defn["command"])
    except Exception as exc:
        return result("error", [], f"error: {exc}", error=str(exc))
    items = []
    warn_t = defn.get("warnThreshold")
    if warn_t is not None:
        num = re.sub(r"[^0-9.]", "", out)
        thr = re.sub(r"[^0-9.]", "", str(warn_t))
        try:
            numv, thrv = float(num), float(thr)
        except ValueError:
            numv = thrv = None
        if numv is not None and numv >= thrv:
            crit_t = defn.get("criticalThreshold")
            is_crit = False
            if crit_t is not None:
                try:
                    is_crit = numv >= float(re.sub(r"[^0-9.]", "", str(crit_t)))
                except ValueError:
                    pass
            items.append(item(f"custom-{cid}-threshold", f"custom:{cid}",
                              "critical" if is_crit else "warn",
                              "needs_owner" if is_crit else "auto_fixable",
                              f'Custom check "{cid}": {out.strip()} (threshold: {warn_t})',
                              90 if is_crit else 50))
    if defn.get("warnIfOutput") and out.strip():
        items.append(item(f"custom-{cid}-output", f"custom:{cid}", "warn", "informational",
                          f'Custom check "{cid}" produced output', 40, detail=out[:500]))
    if defn.get("warnIfNoOutput") and not out.strip():
        items.append(item(f"custom-{cid}-no-output", f"custom:{cid}", "warn", "informational",
                          f'Custom check "{cid}" produced no output (expected some)', 30))
    return result(_status_from_items(items), items,
                  f"{len(items)} issue(s)" if items else "ok")


def _goal_quality(config: Dict) -> Dict:
    from .quality import check_goal_quality

    return check_goal_quality(config)


BUILT_IN_COLLECTORS: Dict[str, Callable[[Dict], Dict]] = {
    "systemd_timers": collect_systemd_timers,
    "nats": collect_journal,
    "goals": collect_goals,
    "threads": collect_threads,
    "errors": collect_errors,
    "calendar": collect_calendar,
    "gpu_health": collect_gpu_health,
    "goal_quality": _goal_quality,
}
