"""Fact embeddings sync.

Parity target: reference `openclaw-knowledge-engine/src/embeddings.ts` —
ChromaDB v2 upsert payload: document = "subject predicate object." with
hyphens in the predicate turned into spaces (`:63-81`), string-only
metadata, endpoint URL with `{name}` collection substitution and
double-slash collapse (protocol preserved).

MI355X-native addition: `LocalGpuEmbedder` routes facts into the
in-process Membrane index (the HBM-resident shard searched by
`csrc/topk_recall.hip`) instead of an external vector DB, using the same
4-gram encoder the firewall pipeline uses — no network, no duplicate
embedding model. The remote ChromaDB client is kept for drop-in parity
(`http_post` injectable; no egress in this environment).
"""

from __future__ import annotations

import re
from typing import Callable, Dict, List, Optional


def build_endpoint_url(endpoint: str, collection: str) -> str:
    """`{name}` substitution + non-protocol double-slash collapse
    (embeddings.ts buildEndpointUrl)."""
    url = endpoint.replace("{name}", collection)
    return re.sub(r"([^:])//", r"\1/", url)


def fact_document(fact: Dict) -> str:
    return f"{fact['subject']} {fact['predicate'].replace('-', ' ')} {fact['object']}."


def construct_chroma_payload(facts: List[Dict]) -> Dict[str, list]:
    """ChromaDB v2 payload; metadata values all strings (v2 requirement)."""
    payload: Dict[str, list] = {"ids": [], "documents": [], "metadatas": []}
    for fact in facts:
        payload["ids"].append(fact["id"])
        payload["documents"].append(fact_document(fact))
        payload["metadatas"].append(
            {
                "subject": fact["subject"],
                "predicate": fact["predicate"],
                "object": fact["object"],
                "source": fact["source"],
                "createdAt": fact["createdAt"],
            }
        )
    return payload


class Embeddings:
    """ChromaDB-compatible sync (embeddings.ts)."""

    def __init__(
        self,
        endpoint: str,
        collection: str,
        enabled: bool = True,
        http_post: Optional[Callable[[str, Dict], object]] = None,
        logger=None,
    ):
        self.endpoint = endpoint
        self.collection = collection
        self._enabled = enabled
        self._post = http_post
        self._log = logger

    def is_enabled(self) -> bool:
        return self._enabled and self._post is not None

    def sync(self, facts: List[Dict]) -> int:
        """Returns the number of successfully synced facts (0 on error)."""
        if not self.is_enabled() or not facts:
            return 0
        payload = construct_chroma_payload(facts)
        url = build_endpoint_url(self.endpoint, self.collection)
        try:
            self._post(url, payload)
            return len(facts)
        except Exception as exc:
            if self._log is not None:
                self._log.error("Failed to sync embeddings: %s", exc)
            return 0


class LocalGpuEmbedder:
    """Embeds fact documents with the pipeline's 4-gram encoder and
    appends them to a Membrane index shard on-GPU (MI355X-native
    replacement for the external ChromaDB dependency)."""

    def __init__(self, membrane, agent_id: str = "knowledge"):
        self.membrane = membrane
        self.agent_id = agent_id

    def is_enabled(self) -> bool:
        return True

    def sync(self, facts: List[Dict]) -> int:
        if not facts:
            return 0
        docs = [fact_document(f) for f in facts]
        metas = [{"fact_id": f["id"], "subject": f["subject"]} for f in facts]
        self.membrane.ingest(self.agent_id, docs, metas)
        return len(facts)
