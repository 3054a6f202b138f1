"""Governance plugin entry: register(api).

Parity target: `openclaw-governance/index.ts:14-116` — Ollama callLlm
factory, load config (file first, pluginConfig fallback), build engine,
set known agents from host config, register service start/stop, wire
hooks + gateway methods.
"""

from __future__ import annotations

from typing import Any, Dict, List, Optional

from ..core.api import PluginApi
from ..core.config import load_raw_layered
from .approval_2fa import Approval2FA
from .engine import GovernanceEngine
from .hooks import register_governance_hooks
from .redaction.hooks import register_redaction_hooks


def make_call_llm(llm_cfg: Dict[str, Any], http_post=None):
    """callLlm factory (index.ts:14-62): a prompt -> text callable over
    an OpenAI/Ollama-compatible chat endpoint. Transport injectable (no
    network in tests); returns None when disabled/unconfigured."""
    llm_cfg = llm_cfg or {}
    if not llm_cfg.get("enabled") or not llm_cfg.get("endpoint"):
        return None
    if http_post is None:
        from ..knowledge.http_client import http_post as _hp

        http_post = _hp
    endpoint = str(llm_cfg["endpoint"]).rstrip("/") + "/chat/completions"
    model = llm_cfg.get("model", "mistral:7b")
    timeout_s = float(llm_cfg.get("timeoutMs", 15000)) / 1000.0
    headers = {}
    if llm_cfg.get("apiKey"):
        headers["Authorization"] = f"Bearer {llm_cfg['apiKey']}"

    def call_llm(prompt: str) -> str:
        import json as _json

        raw = http_post(endpoint, {
            "model": model,
            "messages": [{"role": "user", "content": prompt}],
            "temperature": 0,
        }, headers=headers, timeout_s=timeout_s)
        data = _json.loads(raw)
        return data["choices"][0]["message"]["content"]

    return call_llm


def extract_agent_ids(host_config: Dict[str, Any]) -> List[str]:
    """Agent ids from the host openclaw.json (index.ts:85-88); tolerant of
    the 4 config shapes brainplex scans (scanner.ts:58-91)."""
    agents: List[str] = []
    cfg = host_config or {}
    raw = cfg.get("agents")
    if isinstance(raw, list):
        for a in raw:
            if isinstance(a, str):
                agents.append(a)
            elif isinstance(a, dict) and a.get("id"):
                agents.append(str(a["id"]))
            elif isinstance(a, dict) and a.get("name"):
                agents.append(str(a["name"]))
    elif isinstance(raw, dict):
        agents.extend(str(k) for k in raw.keys())
    for key in ("agent", "defaultAgent"):
        v = cfg.get(key)
        if isinstance(v, str):
            agents.append(v)
        elif isinstance(v, dict) and v.get("id"):
            agents.append(str(v["id"]))
    seen = set()
    out = []
    for a in agents:
        if a not in seen:
            seen.add(a)
            out.append(a)
    return out


class GovernancePlugin:
    id = "openclaw-governance"
    name = "Governance"
    description = "Agent firewall: policy engine, trust scoring, 2FA, output validation, redaction, audit"
    version = "0.1.0"

    def __init__(self, workspace: Optional[str] = None):
        self.workspace = workspace
        self.engine: Optional[GovernanceEngine] = None
        self.hooks = None
        self.redaction = None
        self.approval: Optional[Approval2FA] = None

    def register(self, api: PluginApi) -> None:
        from .config import resolve_config

        config = resolve_config(load_raw_layered(self.id, api.plugin_config))
        workspace = self.workspace or config.get("workspace") or "."
        engine = GovernanceEngine(config, workspace, api.logger)
        engine.set_known_agents(extract_agent_ids(api.config))
        self.engine = engine
        twofa_cfg = config.get("approval2fa") or {}
        if twofa_cfg.get("enabled"):
            self.approval = Approval2FA(secret=twofa_cfg.get("secret"))
        # Stage-3 LLM validator wiring (index.ts:14-62): only when the
        # config enables an endpoint; fail-open inside the validator
        call_llm = make_call_llm(config.get("llm") or
                                 (config.get("outputValidation") or {}).get("llm"))
        if call_llm is not None:
            from .llm_validator import LlmValidator

            engine.output_validator.set_llm_validator(LlmValidator(call_llm))
        api.register_service({"id": self.id, "start": engine.start, "stop": engine.stop})
        self.hooks = register_governance_hooks(api, engine, config, self.approval)
        self.redaction = register_redaction_hooks(api, config.get("redaction"))


def create_plugin(workspace: Optional[str] = None) -> GovernancePlugin:
    return GovernancePlugin(workspace)
