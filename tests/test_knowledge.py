"""Knowledge-engine tests: patterns, entity extraction, fact store,
maintenance, LLM enhancer, hooks.

Mirrors the reference test strategy: table-style unit
tests for pure functions, tmp-dir workspaces with real file round-trips,
fake plugin API driving captured handlers.
Reference behaviors cited per test (openclaw-knowledge-engine).
"""

import json
import time

import pytest

from vainplex_openclaw_amd.core.api import HookBus, NullLogger, PluginApi
from vainplex_openclaw_amd.knowledge.embeddings import (
    Embeddings,
    build_endpoint_url,
    construct_chroma_payload,
)
from vainplex_openclaw_amd.knowledge.entity_extractor import (
    EntityExtractor,
    canonicalize,
    entity_id,
    initial_importance,
    merge_entities,
)
from vainplex_openclaw_amd.knowledge.fact_store import FactStore, boost_relevance
from vainplex_openclaw_amd.knowledge.hooks import HookManager
from vainplex_openclaw_amd.knowledge.llm_enhancer import (
    LlmEnhancer,
    parse_llm_response,
    transform_entities,
    transform_facts,
)
from vainplex_openclaw_amd.knowledge.maintenance import Maintenance
from vainplex_openclaw_amd.knowledge.patterns import PATTERNS


# -- patterns (patterns.ts) --------------------------------------------------

@pytest.mark.parametrize(
    "family,text,expect",
    [
        ("email", "mail me at jo.doe+x@corp.example.org now", "jo.doe+x@corp.example.org"),
        ("url", "see https://docs.example.com/a/b?q=1 please", "https://docs.example.com/a/b?q=1"),
        ("iso_date", "due 2026-03-14T12:30:00Z ok", "2026-03-14T12:30:00Z"),
        ("common_date", "on 3/14/2026 we ship", "3/14/2026"),
        ("common_date", "am 14.3.2026 liefern wir", "14.3.2026"),
        ("german_date", "am 14. März 2026 liefern wir", "14. März 2026"),
        ("english_date", "by March 14th, 2026 at latest", "March 14th, 2026"),
        ("organization_suffix", "works at Acme Corp. now", "Acme Corp."),
        ("product_name", "upgrade to ClawEngine v2.5 today", "ClawEngine v2.5"),
    ],
)
def test_pattern_families(family, text, expect):
    matches = [m.group(0) for m in PATTERNS[family].finditer(text)]
    assert expect in matches


def test_proper_noun_exclusions():
    # EXCLUDED_WORDS filtered per word (patterns.ts EXCL)
    text = "The quick Marie Curie met Hello world in Paris"
    got = {m.group(0) for m in PATTERNS["proper_noun"].finditer(text)}
    assert "Marie Curie" in got
    assert "Paris" in got
    assert "The" not in got and "Hello" not in got


# -- entity extractor (entity-extractor.ts) ----------------------------------

def test_canonicalize_org_suffix_and_punct():
    assert canonicalize("Acme, Inc.", "organization") == "Acme"
    assert canonicalize("Acme GmbH", "organization") == "Acme"
    assert canonicalize("Paris.", "unknown") == "Paris"


def test_initial_importance_table():
    # org .8, product .6, date/email/url .4, multiword unknown .5, else .3
    assert initial_importance("organization", "Acme") == 0.8
    assert initial_importance("product", "ClawEngine v2") == 0.6
    assert initial_importance("email", "a@b.co") == 0.4
    assert initial_importance("unknown", "Marie Curie") == 0.5
    assert initial_importance("unknown", "Paris") == 0.3


def test_extract_dedupes_by_id_and_counts_mentions():
    ex = EntityExtractor(NullLogger())
    ents = ex.extract("Email a@b.example and again a@b.example plus Acme Corp.")
    by_id = {e.id: e for e in ents}
    em = by_id[entity_id("email", "a@b.example")]
    assert em.count == 2 and em.mentions == ["a@b.example"]
    org = by_id[entity_id("organization", "Acme")]
    assert org.value == "Acme" and org.importance == 0.8
    assert org.source == ["regex"]


def test_merge_entities_unions_and_maxes():
    ex = EntityExtractor(NullLogger())
    a = ex.extract("Acme Corp. ships")
    b = ex.extract("Acme Corp. again from Acme Corp.")
    merged = merge_entities(a, b)
    org = [e for e in merged if e.type == "organization"][0]
    assert org.count == 1 + 2
    assert org.importance == 0.8


# -- fact store (fact-store.ts) ----------------------------------------------

def test_fact_store_requires_load(tmp_path):
    fs = FactStore(str(tmp_path))
    with pytest.raises(RuntimeError):
        fs.add_fact("a", "is-a", "b")


def test_fact_dedupe_boosts_relevance(tmp_path):
    fs = FactStore(str(tmp_path))
    fs.load()
    f1 = fs.add_fact("claude", "works-at", "anthropic")
    assert f1["relevance"] == 1.0
    f1["relevance"] = 0.5  # simulate decay
    f2 = fs.add_fact("claude", "works-at", "anthropic")
    assert f2["id"] == f1["id"]
    assert f2["relevance"] == boost_relevance(0.5) == 0.75
    assert len(fs.facts) == 1


def test_fact_decay_floor_and_count(tmp_path):
    fs = FactStore(str(tmp_path))
    fs.load()
    fs.add_fact("a", "p", "b")
    fs.facts[list(fs.facts)[0]]["relevance"] = 0.11
    assert fs.decay_facts(0.5) == 1
    assert list(fs.facts.values())[0]["relevance"] == 0.1  # floor


def test_fact_prune_drops_least_relevant(tmp_path):
    fs = FactStore(str(tmp_path), max_facts=2)
    fs.load()
    fs.add_fact("a", "p", "1")
    fs.add_fact("b", "p", "2")
    for i, f in enumerate(fs.facts.values()):
        f["relevance"] = 0.2 + i * 0.1
    fs.add_fact("c", "p", "3")  # relevance 1.0; store capped at 2
    subjects = {f["subject"] for f in fs.facts.values()}
    assert subjects == {"b", "c"}


def test_fact_query_sorted_by_relevance(tmp_path):
    fs = FactStore(str(tmp_path))
    fs.load()
    fs.add_fact("a", "p", "1")
    fs.add_fact("a", "p", "2")
    fs.add_fact("b", "p", "3")
    vals = list(fs.facts.values())
    vals[0]["relevance"] = 0.3
    got = fs.query(subject="a")
    assert len(got) == 2
    assert got[0]["relevance"] >= got[1]["relevance"]


def test_fact_persist_round_trip(tmp_path):
    fs = FactStore(str(tmp_path))
    fs.load()
    fs.add_fact("claude", "is-a", "agent", source="ingested")
    fs.flush()
    data = json.loads((tmp_path / "facts.json").read_text())
    assert data["facts"][0]["subject"] == "claude"
    assert "updated" in data
    fs2 = FactStore(str(tmp_path))
    fs2.load()
    assert len(fs2.facts) == 1


def test_unembedded_tracking(tmp_path):
    fs = FactStore(str(tmp_path))
    fs.load()
    f = fs.add_fact("a", "p", "b")
    assert fs.unembedded_facts() == [f]
    fs.mark_embedded([f["id"]])
    assert fs.unembedded_facts() == []
    assert f["embedded"]


# -- embeddings (embeddings.ts) ----------------------------------------------

def test_chroma_payload_and_url():
    fact = {
        "id": "x",
        "subject": "claude",
        "predicate": "works-at",
        "object": "anthropic",
        "source": "ingested",
        "createdAt": "2026-01-01T00:00:00Z",
    }
    p = construct_chroma_payload([fact])
    assert p["documents"] == ["claude works at anthropic."]
    assert p["metadatas"][0]["predicate"] == "works-at"
    url = build_endpoint_url("http://h:8000/api/v2/collections/{name}//upsert", "facts")
    assert url == "http://h:8000/api/v2/collections/facts/upsert"


def test_embeddings_sync_counts_and_fails_soft():
    calls = []
    emb = Embeddings("http://e/{name}", "c", http_post=lambda u, p: calls.append((u, p)))
    facts = [
        {"id": "1", "subject": "a", "predicate": "p", "object": "b", "source": "ingested", "createdAt": "t"}
    ]
    assert emb.sync(facts) == 1 and len(calls) == 1

    def boom(u, p):
        raise OSError("down")

    emb2 = Embeddings("http://e/{name}", "c", http_post=boom, logger=NullLogger())
    assert emb2.sync(facts) == 0


# -- maintenance (maintenance.ts) --------------------------------------------

def test_maintenance_decay_and_sync(tmp_path):
    fs = FactStore(str(tmp_path))
    fs.load()
    fs.add_fact("a", "p", "b")
    synced = []
    emb = Embeddings("http://e/{name}", "c", http_post=lambda u, p: synced.append(p))
    cfg = {"decay": {"enabled": True, "intervalHours": 1, "rate": 0.5},
           "embeddings": {"syncIntervalMinutes": 1}}
    m = Maintenance(cfg, fs, emb)
    assert m.run_decay() == 1
    assert m.run_embeddings_sync() == 1
    assert fs.unembedded_facts() == []
    assert m.run_embeddings_sync() == 0  # nothing left


# -- llm enhancer (llm-enhancer.ts) ------------------------------------------

def test_parse_llm_response_ollama_envelope_and_bare():
    inner = {"entities": [{"type": "person", "value": "Ada"}], "facts": []}
    assert parse_llm_response(json.dumps({"response": json.dumps(inner)}))["entities"]
    assert parse_llm_response(json.dumps(inner))["entities"]


def test_transforms_normalize_predicate_and_clamp_importance():
    ents = transform_entities([{"type": "Person", "value": " Ada ", "importance": 3.0},
                               {"type": "person"}])  # second invalid
    assert len(ents) == 1
    assert ents[0]["id"] == "person:ada" and ents[0]["importance"] == 1.0
    facts = transform_facts([{"subject": "Ada", "predicate": "Works At", "object": "X"}])
    assert facts[0]["predicate"] == "works-at"


def test_enhancer_batch_size_triggers_send():
    prompts = []

    def call(prompt):
        prompts.append(prompt)
        return json.dumps({"entities": [], "facts": [
            {"subject": "a", "predicate": "p", "object": "b"}]})

    enh = LlmEnhancer(call, batch_size=2, cooldown_ms=60000)
    assert enh.add_to_batch("1", "hello") is None
    out = enh.add_to_batch("2", "world")
    assert out is not None and len(out["facts"]) == 1
    assert "hello\nworld" in prompts[0]
    assert enh.batch == []
    enh.clear_timers()


def test_enhancer_error_returns_none():
    def boom(prompt):
        raise OSError("no llm")

    enh = LlmEnhancer(boom, batch_size=1, logger=NullLogger())
    assert enh.add_to_batch("1", "x") is None


# -- hooks (hooks.ts) --------------------------------------------------------

def _mk_api(bus=None):
    return PluginApi(
        id="openclaw-knowledge-engine",
        plugin_config={},
        logger=NullLogger(),
        config={},
        bus=bus or HookBus(),
    )


def _cfg(ws, llm_enabled=False):
    return {
        "enabled": True,
        "extraction": {
            "regex": {"enabled": True},
            "llm": {"enabled": llm_enabled, "batchSize": 1, "cooldownMs": 60000},
        },
        "decay": {"enabled": False},
        "embeddings": {"enabled": False},
        "storage": {"maxEntities": 100, "maxFacts": 100, "writeDebounceMs": 10},
    }


def test_hook_manager_message_flow(tmp_path):
    bus = HookBus()
    api = _mk_api(bus)
    hm = HookManager(_cfg(str(tmp_path)), str(tmp_path), logger=NullLogger())
    hm.register(api)
    bus.emit("session_start", {})
    bus.emit("message_received", {"content": "Contact ada@example.org at Acme Corp."})
    assert any(e.type == "email" for e in hm.entities.values())
    assert any(e.type == "organization" for e in hm.entities.values())
    bus.emit("gateway_stop", {})
    assert (tmp_path / "facts.json").exists()


def test_hook_manager_llm_facts_stored(tmp_path):
    def call(prompt):
        return json.dumps({"entities": [], "facts": [
            {"subject": "ada", "predicate": "works at", "object": "acme"}]})

    bus = HookBus()
    api = _mk_api(bus)
    hm = HookManager(
        _cfg(str(tmp_path), llm_enabled=True), str(tmp_path),
        logger=NullLogger(), call_llm=call,
    )
    hm.register(api)
    bus.emit("session_start", {})
    bus.emit("message_sent", {"text": "Ada works at Acme"})
    got = hm.fact_store.query(subject="ada")
    assert got and got[0]["predicate"] == "works-at"


def test_hook_manager_ignores_blank_and_disabled(tmp_path):
    cfg = _cfg(str(tmp_path))
    cfg["extraction"]["regex"]["enabled"] = False
    hm = HookManager(cfg, str(tmp_path), logger=NullLogger())
    hm.on_message({"content": "Contact ada@example.org"})
    assert hm.entities == {}
    hm.on_message({"content": "   "})
    assert hm.entities == {}


def test_http_client_post_loopback():
    """http-client.ts parity: JSON POST round trip against a loopback
    server (no egress)."""
    import http.server
    import threading

    from vainplex_openclaw_amd.knowledge.http_client import HttpError, http_post

    received = {}

    class H(http.server.BaseHTTPRequestHandler):
        def do_POST(self):
            n = int(self.headers.get("Content-Length", 0))
            received["body"] = json.loads(self.rfile.read(n))
            received["ct"] = self.headers.get("Content-Type")
            if self.path == "/fail":
                self.send_response(500)
                self.end_headers()
                self.wfile.write(b"boom")
                return
            self.send_response(200)
            self.end_headers()
            self.wfile.write(b'{"ok": true}')

        def log_message(self, *a):
            pass

    srv = http.server.HTTPServer(("127.0.0.1", 0), H)
    t = threading.Thread(target=srv.serve_forever, daemon=True)
    t.start()
    try:
        port = srv.server_address[1]
        out = http_post(f"http://127.0.0.1:{port}/x", {"a": 1})
        assert json.loads(out) == {"ok": True}
        assert received["body"] == {"a": 1} and received["ct"] == "application/json"
        with pytest.raises(HttpError) as ei:
            http_post(f"http://127.0.0.1:{port}/fail", {})
        assert ei.value.status == 500
    finally:
        srv.shutdown()


def test_fact_get_boosts_on_access_and_multi_filter(tmp_path):
    fs = FactStore(str(tmp_path))
    fs.load()
    f = fs.add_fact("svc", "runs-on", "k8s")
    f["relevance"] = 0.4
    before_access = f["lastAccessed"]
    got = fs.get_fact(f["id"])
    assert got["relevance"] == boost_relevance(0.4)
    assert got["lastAccessed"] >= before_access
    assert fs.get_fact("nope") is None
    # multiple filters AND together; empty query returns everything
    fs.add_fact("svc", "runs-on", "vm")
    fs.add_fact("db", "runs-on", "k8s")
    assert len(fs.query(subject="svc", obj="k8s")) == 1
    assert len(fs.query()) == 3


# ===========================================================================
# fact-store.test.ts depth: load-gate, access boost, query matrix, decay
# floor, embedding lifecycle, prune order
# ===========================================================================

def _store(workspace, **kw):
    from vainplex_openclaw_amd.knowledge.fact_store import FactStore

    t = [1_700_000_000.0]
    fs = FactStore(workspace, clock=lambda: t[0], **kw)
    fs.load()
    return fs, t


def test_fact_store_add_before_load_raises(workspace):
    from vainplex_openclaw_amd.knowledge.fact_store import FactStore

    fs = FactStore(workspace)
    with pytest.raises(RuntimeError):
        fs.add_fact("a", "b", "c")


def test_fact_store_get_boosts_and_touches(workspace):
    # relevance starts at 1.0 and boost moves 50% closer to 1.0, so the
    # boost is only visible after decay (fact-store.ts boostRelevance)
    fs, t = _store(workspace)
    f = fs.add_fact("svc", "status", "running")
    fs.decay_facts(0.5)
    assert f["relevance"] == 0.5
    t[0] += 100
    got = fs.get_fact(f["id"])
    assert got["relevance"] == 0.75  # 0.5 + (1-0.5)*0.5
    assert got["lastAccessed"] != f["createdAt"]
    assert fs.get_fact("nope") is None


def test_fact_store_query_matrix(workspace):
    fs, _ = _store(workspace)
    fs.add_fact("nginx", "status", "running")
    fs.add_fact("nginx", "port", "443")
    fs.add_fact("redis", "status", "running")
    assert len(fs.query(subject="nginx")) == 2
    assert len(fs.query(predicate="status")) == 2
    assert len(fs.query(obj="running")) == 2
    assert len(fs.query(subject="nginx", predicate="status")) == 1
    assert len(fs.query()) == 3
    assert fs.query(subject="ghost") == []


def test_fact_store_query_sorted_by_relevance(workspace):
    fs, _ = _store(workspace)
    fs.add_fact("a", "p", "1")
    fs.add_fact("b", "p", "2")
    fs.decay_facts(0.5)          # both at 0.5
    fs.add_fact("b", "p", "2")   # dedupe boost lifts b to 0.75
    out = fs.query(predicate="p")
    assert out[0]["subject"] == "b" and out[1]["subject"] == "a"
    assert out[0]["relevance"] > out[1]["relevance"]


def test_fact_store_decay_floor_and_counts(workspace):
    fs, _ = _store(workspace)
    fs.add_fact("a", "p", "1")
    assert fs.decay_facts(0.5) == 1
    for _ in range(10):
        fs.decay_facts(0.5)
    assert all(f["relevance"] >= 0.1 for f in fs.query())
    empty, _t = _store(workspace + "/sub2")
    assert empty.decay_facts(0.5) == 0


def test_fact_store_embedding_lifecycle(workspace):
    fs, _ = _store(workspace)
    a = fs.add_fact("a", "p", "1")
    b = fs.add_fact("b", "p", "2")
    assert {f["id"] for f in fs.unembedded_facts()} == {a["id"], b["id"]}
    fs.mark_embedded([a["id"], "ghost-id"])
    left = fs.unembedded_facts()
    assert [f["id"] for f in left] == [b["id"]]
    assert fs.facts[a["id"]]["embedded"]
    fs.mark_embedded([b["id"]])
    assert fs.unembedded_facts() == []


def test_fact_store_prune_least_relevant_first(workspace):
    fs, t = _store(workspace, max_facts=3)
    keep = fs.add_fact("hot", "p", "v")
    cold = fs.add_fact("cold0", "p", "v0")
    fs.decay_facts(0.5)
    fs.add_fact("hot", "p", "v")     # boost the keeper back to 0.75
    for i in range(1, 3):
        t[0] += 1
        fs.add_fact(f"cold{i}", "p", f"v{i}")
    assert len(fs.facts) == 3
    assert keep["id"] in fs.facts     # boosted fact survived
    assert cold["id"] not in fs.facts  # least relevant evicted


# ===========================================================================
# KE llm-enhancer depth: prompt shape, Ollama envelope parsing,
# transforms (type/importance normalization), batch trigger
# ===========================================================================

def test_enhancer_prompt_shape():
    from vainplex_openclaw_amd.knowledge.llm_enhancer import construct_prompt

    p = construct_prompt(["Alice met Bob.", "Initech shipped v2."])
    assert "Alice met Bob." in p and "Initech shipped v2." in p
    assert '"entities"' in p and '"facts"' in p
    assert "JSON" in p


def test_enhancer_parse_envelope_and_bare():
    import json as _json

    from vainplex_openclaw_amd.knowledge.llm_enhancer import parse_llm_response

    inner = {"entities": [{"type": "person", "value": "Alice"}], "facts": []}
    # Ollama envelope
    out = parse_llm_response(_json.dumps({"response": _json.dumps(inner)}))
    assert out["entities"][0]["value"] == "Alice"
    # bare object
    out2 = parse_llm_response(_json.dumps(inner))
    assert out2["entities"] and out2["facts"] == []
    # non-list fields degrade to []
    out3 = parse_llm_response(_json.dumps({"entities": "no", "facts": 4}))
    assert out3 == {"entities": [], "facts": []}
    import pytest as _pt

    with _pt.raises(Exception):
        parse_llm_response("[1,2]")


def test_enhancer_transform_entities_normalization():
    from vainplex_openclaw_amd.knowledge.llm_enhancer import transform_entities

    raw = [
        {"type": "Person", "value": "  Alice Johnson "},
        {"type": "org", "value": "Initech", "importance": 3.5},   # clamped
        {"type": "concept", "value": "caching", "importance": -1},
        {"value": "no-type"}, "junk", {"type": "person", "value": 42},
    ]
    out = transform_entities(raw, clock=lambda: 1_700_000_000.0)
    assert len(out) == 3
    alice = out[0]
    assert alice["id"] == "person:alice-johnson"
    assert alice["type"] == "person" and alice["value"] == "Alice Johnson"
    assert out[1]["importance"] == 1.0 and out[2]["importance"] == 0.0
    assert all(e["source"] == ["llm"] and e["count"] == 1 for e in out)


def test_enhancer_transform_facts_normalization():
    from vainplex_openclaw_amd.knowledge.llm_enhancer import transform_facts

    raw = [
        {"subject": " Alice ", "predicate": "Works At", "object": " Initech "},
        {"subject": "x", "predicate": "p"},          # incomplete
        {"subject": 1, "predicate": "p", "object": "o"},
        "junk",
    ]
    out = transform_facts(raw)
    assert out == [{"subject": "Alice", "predicate": "works-at",
                    "object": "Initech", "source": "extracted-llm"}]


def test_enhancer_batch_triggers_on_size():
    import json as _json

    from vainplex_openclaw_amd.knowledge.llm_enhancer import LlmEnhancer

    calls = []

    def llm(prompt):
        calls.append(prompt)
        return _json.dumps({"entities": [{"type": "person", "value": "A"}],
                            "facts": []})

    results = []
    enh = LlmEnhancer(call_llm=llm, batch_size=2, clock=lambda: 0.0)
    enh.set_result_handler(results.append)
    assert enh.add_to_batch("m1", "first text") is None
    out = enh.add_to_batch("m2", "second text")
    assert out is not None and len(calls) == 1
    assert "first text" in calls[0] and "second text" in calls[0]
    # disabled enhancer is inert
    assert not LlmEnhancer(None).enabled
    assert LlmEnhancer(None).add_to_batch("x", "y") is None
