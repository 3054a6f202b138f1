"""Parent <-> child agent session graph, trust ceilings, policy cascade.

Parity target: governance `src/cross-agent.ts` — relationships registered
from `sessions_spawn` results (hooks.ts:424-435) or inferred from
`:subagent:` session keys; `enrich_context` caps child session AND agent
trust at the parent's agent score; `resolve_effective_policies` merges the
child's and parent's (1-level) policies by id, filtered by hook.
"""

from __future__ import annotations

import time
from typing import Any, Dict, List, Optional

from .policies import PolicyIndex
from .trust import TrustManager
from .util import extract_agent_id, is_sub_agent, parent_session_key, score_to_tier


class CrossAgentManager:
    def __init__(self, trust: TrustManager, clock=time.time):
        self.trust = trust
        self.clock = clock
        self._relationships: Dict[str, Dict[str, Any]] = {}

    def register_relationship(self, parent_session_key_: str, child_session_key: str) -> None:
        self._relationships[child_session_key] = {
            "parentAgentId": extract_agent_id(parent_session_key_),
            "parentSessionKey": parent_session_key_,
            "childAgentId": extract_agent_id(child_session_key),
            "childSessionKey": child_session_key,
            "createdAt": self.clock() * 1000,
        }

    def remove_relationship(self, child_session_key: str) -> None:
        self._relationships.pop(child_session_key, None)

    def get_parent(self, child_session_key: str) -> Optional[Dict[str, Any]]:
        explicit = self._relationships.get(child_session_key)
        if explicit:
            return explicit
        if not is_sub_agent(child_session_key):
            return None
        pkey = parent_session_key(child_session_key)
        if not pkey:
            return None
        return {
            "parentAgentId": extract_agent_id(pkey),
            "parentSessionKey": pkey,
            "childAgentId": extract_agent_id(child_session_key),
            "childSessionKey": child_session_key,
            "createdAt": 0,
        }

    def get_children(self, parent_session_key_: str) -> List[Dict[str, Any]]:
        return [r for r in self._relationships.values() if r["parentSessionKey"] == parent_session_key_]

    def compute_trust_ceiling(self, session_key: str) -> float:
        parent = self.get_parent(session_key)
        if not parent:
            return float("inf")
        return self.trust.score(parent["parentAgentId"])

    def enrich_context(self, ctx: Dict[str, Any]) -> Dict[str, Any]:
        parent = self.get_parent(ctx.get("sessionKey", ""))
        if not parent:
            return ctx
        ceiling = self.compute_trust_ceiling(ctx["sessionKey"])
        trust = ctx.get("trust") or {"agent": {}, "session": {}}
        agent_t = dict(trust.get("agent") or {})
        session_t = dict(trust.get("session") or {})
        session_t["score"] = min(float(session_t.get("score", 0)), ceiling)
        session_t["tier"] = score_to_tier(session_t["score"])
        agent_t["score"] = min(float(agent_t.get("score", 0)), ceiling)
        agent_t["tier"] = score_to_tier(agent_t["score"])
        out = dict(ctx)
        out["trust"] = {"agent": agent_t, "session": session_t}
        out["crossAgent"] = {
            "parentAgentId": parent["parentAgentId"],
            "parentSessionKey": parent["parentSessionKey"],
            "inheritedPolicyIds": [f"inherited-from:{parent['parentAgentId']}"],
            "trustCeiling": ceiling,
        }
        return out

    def resolve_effective_policies(self, ctx: Dict[str, Any], index: PolicyIndex) -> List[Dict[str, Any]]:
        own = self._collect(ctx.get("agentId", ""), ctx.get("hook", ""), index)
        parent = self.get_parent(ctx.get("sessionKey", ""))
        if not parent:
            return own
        parents = self._collect(parent["parentAgentId"], ctx.get("hook", ""), index)
        merged = list(own)
        seen = {p.get("id") for p in own}
        for p in parents:
            if p.get("id") not in seen:
                seen.add(p.get("id"))
                merged.append(p)
        return merged

    def graph_summary(self) -> Dict[str, Any]:
        return {
            "agentCount": len(self._relationships),
            "relationships": list(self._relationships.values()),
        }

    def _collect(self, agent_id: str, hook: str, index: PolicyIndex) -> List[Dict[str, Any]]:
        result: List[Dict[str, Any]] = []
        seen = set()
        for p in index.by_agent.get(agent_id, []) + index.by_agent.get("*", []):
            if p.get("id") not in seen:
                seen.add(p.get("id"))
                result.append(p)
        hook_policies = index.by_hook.get(hook)
        if hook_policies is not None:
            hook_ids = {p.get("id") for p in hook_policies}
            return [p for p in result if p.get("id") in hook_ids]
        return result
