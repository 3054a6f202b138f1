"""Standalone suite runner: `python -m vainplex_openclaw_amd [--demo]`.

Boots the Gateway with all seven plugins on a workspace and feeds it
messages — from a scripted demo conversation (`--demo`) or stdin (REPL).
The reference relies on an external OpenClaw host; this makes the
MI355X-native suite runnable end-to-end on its own.
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import tempfile

from .core.api import NullLogger, PluginLogger
from .core.gateway import Gateway
from .cortex.hooks import create_plugin as create_cortex
from .eventstore import EventJournal
from .eventstore.plugin import create_plugin as create_eventstore
from .governance.plugin import create_plugin as create_governance
from .knowledge.hooks import create_plugin as create_knowledge
from .leuko.plugin import create_plugin as create_leuko
from .membrane.hooks import create_plugin as create_membrane

DEMO_SCRIPT = [
    ("user", "We decided to use postgres for the storage layer."),
    ("user", "I will prepare the migration plan by friday."),
    ("user", "Contact ada@example.org at Acme Corp. about the rollout."),
    ("user", "what did we decide about storage?"),
    ("assistant", "You decided to use postgres for the storage layer."),
]


def build_suite(workspace: str, logger=None, device=None):
    gw = Gateway(config={"agents": [{"id": "main"}]}, logger=logger or NullLogger())
    journal = EventJournal(durable=False)
    gw.load(create_governance(workspace=workspace), {})
    gw.load(create_cortex(workspace=workspace), {})
    gw.load(create_knowledge(workspace=workspace), {})
    gw.load(create_membrane(workspace=workspace, device=device), {})
    gw.load(create_eventstore(journal=journal), {})
    gw.load(create_leuko(workspace=workspace, journal=journal), {})
    return gw, journal


def process_message(gw: Gateway, content: str, role: str = "user") -> dict:
    hook = "message_received" if role == "user" else "message_sent"
    ev = gw.bus.emit(hook, {
        "content": content, "from": role,
        "ctx": {"agentId": "main", "sessionKey": "main:cli:1"},
    })
    return ev


def main(argv=None) -> int:
    ap = argparse.ArgumentParser(prog="vainplex_openclaw_amd")
    ap.add_argument("--demo", action="store_true", help="run the scripted demo conversation")
    ap.add_argument("--workspace", default=None)
    ap.add_argument("--gpu", action="store_true", help="use cuda:0 for the Membrane index")
    args = ap.parse_args(argv)

    ws = args.workspace or tempfile.mkdtemp(prefix="openclaw-amd-")
    device = "cuda:0" if args.gpu else None
    gw, journal = build_suite(ws, device=device)
    gw.bus.emit("gateway_start", {"port": 0})
    gw.bus.emit("session_start", {"sessionId": "cli", "ctx": {"sessionKey": "main:cli:1"}})
    print(f"suite up: {len(gw.plugins)} plugins, workspace {ws}", file=sys.stderr)

    def handle(role, content):
        ev = process_message(gw, content, role)
        out = {"role": role, "content": content}
        if ev.get("membrane_context"):
            out["membrane_context"] = ev["membrane_context"]
        if ev.get("block"):
            out["blocked"] = ev.get("blockReason", True)
        print(json.dumps(out, ensure_ascii=False))

    if args.demo:
        for role, content in DEMO_SCRIPT:
            handle(role, content)
    else:
        for line in sys.stdin:
            line = line.rstrip("\n")
            if not line.strip():
                continue
            handle("user", line)

    gw.bus.emit("gateway_stop", {"reason": "cli exit"})
    # closing status from the suite's own surfaces
    leuko = gw.plugins["openclaw-leuko"]
    report = leuko.run_once()
    print(json.dumps({
        "events": len(journal),
        "health": report["health"]["overall"],
        "membrane": gw.apis["openclaw-membrane"].gateway_methods["membrane.stats"](),
    }), file=sys.stderr)
    return 0


if __name__ == "__main__":
    sys.exit(main())
