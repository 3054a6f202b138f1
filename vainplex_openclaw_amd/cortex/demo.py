"""Cortex demo: drives the real trackers through a scripted bilingual
conversation in a temp workspace (reference `demo/demo.ts:1-30`; BASELINE
config #1 — CPU plumbing smoke, no GPU).

Run: python -m vainplex_openclaw_amd.cortex.demo
"""

from __future__ import annotations

import json
import shutil
import tempfile

from .hooks import CortexWorkspace

SCRIPT = [
    ("user", "Let's talk about the database migration for the billing service"),
    ("agent", "I'll prepare the migration plan and check the production schema first"),
    ("user", "We decided to use a blue-green deployment for this"),
    ("agent", "Noted. Waiting for the staging credentials from ops before I can test"),
    ("user", "Ist das Backup erledigt? Das Thema Sicherheit ist kritisch"),
    ("agent", "Ja, das Backup ist fertig. Ich kümmere mich um die Verschlüsselung"),
    ("user", "Great, the migration plan looks done. It works!"),
    ("agent", "Deployed and shipped. I am MigrationBot, happy to help."),
]


def run_demo(workspace: str = "") -> dict:
    tmp = workspace or tempfile.mkdtemp(prefix="cortex-demo-")
    try:
        ws = CortexWorkspace(tmp, {"language": "both"})
        for sender, msg in SCRIPT:
            ws.process_message(msg, sender)
        ws.pre_compaction.run(ws.threads, ws.decisions, ws.commitments)
        status = ws.status()
        boot = ws.boot.generate()
        return {"status": status, "boot_context": boot, "workspace": tmp}
    finally:
        if not workspace:
            shutil.rmtree(tmp, ignore_errors=True)


if __name__ == "__main__":
    out = run_demo()
    print(json.dumps(out["status"], indent=2))
    print()
    print(out["boot_context"])
