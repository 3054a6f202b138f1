"""Situation-report aggregator.

Parity target: reference `openclaw-sitrep/src/aggregator.ts:20-165` —
run builtin + custom collectors, flatten + score-sort items, categorize
(needs_owner / auto_fixable / delegatable / informational), overall
health, delta vs previous report (new/resolved item ids), summary string
capped at summaryMaxChars, per-collector metadata; report schema v1
written to sitrep.json.
"""

from __future__ import annotations

import time
from typing import Dict, List, Optional

from ..utils.storage import atomic_write_json, read_json
from .collectors import BUILT_IN_COLLECTORS, run_custom_collector, safe_collect


def _iso_now(clock=time.time) -> str:
    return time.strftime("%Y-%m-%dT%H:%M:%S", time.gmtime(clock())) + "Z"


def run_all_collectors(config: Dict, logger=None, extra: Optional[Dict] = None,
                       clock=time.time) -> Dict[str, Dict]:
    results: Dict[str, Dict] = {}
    coll_cfg = config.get("collectors", {})
    for name, fn in BUILT_IN_COLLECTORS.items():
        results[name] = safe_collect(
            name, fn, coll_cfg.get(name, {"enabled": False}), logger, clock
        )
    for defn in config.get("customCollectors", []):
        start = clock()
        try:
            r = run_custom_collector(defn)
        except Exception as exc:
            from .collectors import result as mk

            r = mk("error", [], f"error: {exc}", error=str(exc))
        r["duration_ms"] = int((clock() - start) * 1000)
        results[f"custom:{defn.get('id', 'unnamed')}"] = r
    for name, fn in (extra or {}).items():
        start = clock()
        try:
            r = fn()
        except Exception as exc:
            from .collectors import result as mk

            r = mk("error", [], f"error: {exc}", error=str(exc))
        r["duration_ms"] = int((clock() - start) * 1000)
        results[name] = r
    return results


def compute_health(items: List[Dict], results: Dict[str, Dict]) -> Dict:
    details = {}
    for name, r in results.items():
        details[name] = "disabled" if r.get("summary") == "disabled" else r["status"]
    overall = (
        "critical" if any(i["severity"] == "critical" for i in items)
        else "warn" if any(i["severity"] == "warn" for i in items)
        else "ok"
    )
    return {"overall": overall, "details": details}


def compute_delta(current: List[Dict], previous_path: str) -> Dict:
    prev = read_json(previous_path)
    prev_ids = {i["id"] for i in (prev or {}).get("items", [])} if isinstance(prev, dict) else set()
    cur_ids = {i["id"] for i in current}
    return {
        "new_items": len(cur_ids - prev_ids),
        "resolved_items": len(prev_ids - cur_ids),
        "previous_generated": (prev or {}).get("generated") if isinstance(prev, dict) else None,
    }


def generate_summary(categories: Dict, results: Dict[str, Dict], max_chars: int) -> str:
    parts = []
    if categories["needs_owner"]:
        parts.append(f"{len(categories['needs_owner'])} item(s) need owner attention")
    if categories["auto_fixable"]:
        parts.append(f"{len(categories['auto_fixable'])} auto-fixable")
    for name, r in results.items():
        if r["status"] != "ok" and r["summary"] != "disabled":
            parts.append(f"{name}: {r['summary']}")
    if not parts:
        parts.append("All systems nominal")
    return (". ".join(parts) + ".")[:max_chars]


def generate_sitrep(config: Dict, logger=None, extra: Optional[Dict] = None,
                    clock=time.time) -> Dict:
    results = run_all_collectors(config, logger, extra, clock)
    all_items = sorted(
        (i for r in results.values() for i in r["items"]),
        key=lambda i: -i["score"],
    )
    categories = {
        "needs_owner": [i for i in all_items if i["category"] == "needs_owner"],
        "auto_fixable": [i for i in all_items if i["category"] == "auto_fixable"],
        "delegatable": [i for i in all_items if i["category"] == "delegatable"],
        "informational": [i for i in all_items if i["category"] == "informational"],
    }
    collectors = {}
    for name, r in results.items():
        meta = {"status": r["status"], "duration_ms": r["duration_ms"]}
        if r.get("error"):
            meta["error"] = r["error"]
        collectors[name] = meta
    return {
        "version": 1,
        "generated": _iso_now(clock),
        "summary": generate_summary(categories, results, config.get("summaryMaxChars", 2000)),
        "health": compute_health(all_items, results),
        "items": all_items,
        "categories": categories,
        "delta": compute_delta(all_items, config.get("previousPath", "")),
        "collectors": collectors,
    }


def write_sitrep(report: Dict, output_path: str, previous_path: str) -> None:
    """Rotate current -> previous, write new report (service.ts flow)."""
    prev = read_json(output_path)
    if prev is not None and previous_path:
        atomic_write_json(previous_path, prev)
    atomic_write_json(output_path, report)
