"""Goal quality checks + LLM recommendations.

Parity target: the external Leuko's remaining capability surface
("LLM recommendations, goal quality checks"; brainplex
README §Leuko). Goal quality inspects the same goals file the goals
collector reads and flags structural problems — vague titles, missing
success criteria / deadlines, red-zone goals without an owner, open-goal
overload. LLM recommendations feed the finished sitrep summary to an
injectable call_llm and attach a short action list; any failure degrades
to no recommendations (the sitrep must never break on LLM trouble).
"""

from __future__ import annotations

import json
import re
import time
from typing import Any, Callable, Dict, List, Optional

from .collectors import item, read_json_safe, result

_VAGUE_RX = re.compile(
    r"^(?:improve|fix|handle|look into|investigate|better|optimi[sz]e|"
    r"clean ?up|update)\b[\w\s]*$", re.I)
# measurable outcome: a number, percentage, date, or explicit criterion
_MEASURABLE_RX = re.compile(
    r"\d|%|\bby\s+\w+day\b|\buntil\b|\bbefore\b|\bdeadline\b", re.I)


def check_goal_quality(config: Dict) -> Dict:
    """Collector-shaped: structural quality of each goal."""
    path = config.get("goalsPath")
    if not path:
        return result("ok", [], "no goalsPath configured")
    data = read_json_safe(path)
    if data is None:
        return result("ok", [], "goals file not found")
    goals = data.get("goals", data) if isinstance(data, dict) else data
    if not isinstance(goals, list):
        return result("ok", [], "no goals")

    items: List[Dict[str, Any]] = []
    max_open = int(config.get("maxOpenGoals", 7))
    open_goals = [g for g in goals if isinstance(g, dict)
                  and g.get("status") in ("proposed", "approved", "open", None)]
    for g in goals:
        if not isinstance(g, dict):
            continue
        gid = g.get("id", "unknown")
        title = str(g.get("title", "")).strip()
        if not title:
            items.append(item(f"goalq-{gid}-untitled", "goal_quality", "warn",
                              "needs_owner", f"Goal {gid} has no title", 40))
            continue
        if len(title) < 8 or _VAGUE_RX.match(title):
            items.append(item(f"goalq-{gid}-vague", "goal_quality", "info",
                              "needs_owner",
                              f"Vague goal title (no concrete outcome): {title}", 25))
        criteria = g.get("success_criteria") or g.get("successCriteria")
        if not criteria and not _MEASURABLE_RX.search(title):
            items.append(item(f"goalq-{gid}-unmeasurable", "goal_quality", "info",
                              "needs_owner",
                              f"Goal has no success criteria or measurable outcome: {title}",
                              20))
        if g.get("zone") == "red" and not (g.get("owner") or g.get("assignee")):
            items.append(item(f"goalq-{gid}-red-unowned", "goal_quality", "warn",
                              "needs_owner",
                              f"Red-zone goal has no owner: {title}", 60))
    if len(open_goals) > max_open:
        items.append(item("goalq-overload", "goal_quality", "warn", "delegatable",
                          f"{len(open_goals)} goals open (cap {max_open}) — "
                          "focus or delegate", 45))
    return result("warn" if any(i["severity"] != "info" for i in items) else "ok",
                  items, f"{len(items)} quality issue(s)" if items else
                  "goal quality nominal")


RECOMMEND_PROMPT = (
    "You are an operations advisor. Given this situation report, reply "
    'ONLY JSON: {"recommendations": [{"action": "...", "priority": '
    '"high"|"medium"|"low", "reason": "..."}]} with at most {n} items.\n'
    "Report:\n"
)


def llm_recommendations(report: Dict, call_llm: Optional[Callable[[str], str]],
                        max_items: int = 5) -> List[Dict[str, Any]]:
    """Attach LLM-generated action recommendations; [] on any failure."""
    if call_llm is None:
        return []
    context = json.dumps({
        "summary": report.get("summary", ""),
        "health": report.get("health"),
        "top_items": [
            {"title": i.get("title"), "severity": i.get("severity"),
             "category": i.get("category")}
            for i in (report.get("items") or [])[:10]
        ],
    })[:4000]
    try:
        raw = call_llm(RECOMMEND_PROMPT.replace("{n}", str(max_items)) + context)
    except Exception:
        return []
    start, end = raw.find("{"), raw.rfind("}")
    if start < 0 or end <= start:
        return []
    try:
        parsed = json.loads(raw[start:end + 1])
    except json.JSONDecodeError:
        return []
    recs = parsed.get("recommendations")
    if not isinstance(recs, list):
        return []
    out = []
    for r in recs[:max_items]:
        if isinstance(r, dict) and isinstance(r.get("action"), str):
            out.append({
                "action": r["action"][:200],
                "priority": r.get("priority")
                if r.get("priority") in ("high", "medium", "low") else "medium",
                "reason": str(r.get("reason", ""))[:200],
            })
    return out
