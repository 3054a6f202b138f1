"""Membrane tests: store (isolation/decay/reinforce/prune/persistence),
index (CPU reference path), engine (buffering, salience+sensitivity
filters, reinforcement), hooks (inject + ingest flow), and a GPU parity
test for the kernel-backed index.
"""

import time

import pytest

from vainplex_openclaw_amd.core.api import HookBus, NullLogger, PluginApi
from vainplex_openclaw_amd.membrane import (
    MembraneEngine,
    MembranePlugin,
    MemoryRecord,
    MemoryStore,
    SalienceIndex,
)
from vainplex_openclaw_amd.membrane.store import DECAY_HALF_LIFE_H


class FakeClock:
    def __init__(self, t=1_000_000.0):
        self.t = t

    def __call__(self):
        return self.t

    def advance_h(self, hours):
        self.t += hours * 3600.0


# -- store -------------------------------------------------------------------

def test_record_validation():
    with pytest.raises(ValueError):
        MemoryRecord.make("a", "x", kind="nope")
    with pytest.raises(ValueError):
        MemoryRecord.make("a", "x", sensitivity="nope")


def test_store_agent_isolation(tmp_path):
    st = MemoryStore(str(tmp_path))
    st.add(MemoryRecord.make("alice", "alice secret"))
    st.add(MemoryRecord.make("bob", "bob note"))
    assert [r["text"] for r in st.all("alice")] == ["alice secret"]
    assert [r["text"] for r in st.all("bob")] == ["bob note"]


def test_store_organic_decay_half_life(tmp_path):
    clock = FakeClock()
    st = MemoryStore(str(tmp_path), clock=clock)
    rec = st.add(MemoryRecord.make("a", "x", salience=0.8, ts=clock()))
    assert st.decayed_salience(rec) == pytest.approx(0.8)
    clock.advance_h(DECAY_HALF_LIFE_H)
    assert st.decayed_salience(rec) == pytest.approx(0.4, rel=1e-6)


def test_store_reinforce_boosts_and_resets_decay(tmp_path):
    clock = FakeClock()
    st = MemoryStore(str(tmp_path), clock=clock)
    rec = st.add(MemoryRecord.make("a", "x", salience=0.8, ts=clock()))
    clock.advance_h(DECAY_HALF_LIFE_H)
    st.reinforce("a", rec["id"])
    # committed decayed 0.4, boosted 25% toward 1.0 -> 0.55
    assert rec["salience"] == pytest.approx(0.55, rel=1e-6)
    assert rec["recalls"] == 1
    assert st.decayed_salience(rec) == pytest.approx(0.55, rel=1e-6)  # decay restarted


def test_store_prune_decayed_and_cap(tmp_path):
    clock = FakeClock()
    st = MemoryStore(str(tmp_path), clock=clock)
    old = st.add(MemoryRecord.make("a", "old", salience=0.05, ts=clock()))
    st.add(MemoryRecord.make("a", "fresh", salience=1.0, ts=clock()))
    clock.advance_h(DECAY_HALF_LIFE_H * 3)
    st.add(MemoryRecord.make("a", "new", salience=1.0, ts=clock()))
    n = st.prune("a")
    assert n >= 1 and st.get("a", old["id"]) is None
    st.prune("a", max_records=1)
    assert st.count("a") == 1


def test_store_persistence_round_trip(tmp_path):
    st = MemoryStore(str(tmp_path))
    rec = st.add(MemoryRecord.make("ag-1", "remember me", kind="semantic"))
    st.flush()
    st2 = MemoryStore(str(tmp_path))
    got = st2.get("ag-1", rec["id"])
    assert got is not None and got["text"] == "remember me" and got["kind"] == "semantic"


# -- index (CPU reference path) ----------------------------------------------

def test_index_add_search_and_isolation():
    idx = SalienceIndex(dim=128, capacity=4)
    idx.add("a", ["r1", "r2"], ["the quick brown fox", "pay the invoice tomorrow"])
    idx.add("b", ["r3"], ["the quick brown fox"])
    got = idx.search("a", "quick brown fox jumps", k=2)
    assert got and got[0][0] == "r1"  # best match, owned by a
    assert all(rid != "r3" for rid, _ in got)  # isolation
    assert idx.size == 3


def test_index_growth():
    idx = SalienceIndex(dim=64, capacity=2)
    idx.add("a", [f"r{i}" for i in range(10)], [f"text number {i}" for i in range(10)])
    assert idx.size == 10 and idx.capacity >= 10
    assert idx.search("a", "text number 3", k=1)


# -- engine ------------------------------------------------------------------

def _engine(tmp_path, clock=None, **cfg):
    return MembraneEngine(str(tmp_path), config=cfg or None, dim=128,
                          clock=clock or time.time)


def test_engine_retrieve_filters_and_reinforces(tmp_path):
    clock = FakeClock()
    eng = _engine(tmp_path, clock=clock)
    eng.remember("a", "deploy the service on friday", sensitivity="low")
    eng.remember("a", "the root password is hunter2", sensitivity="high")
    out = eng.retrieve("a", "when do we deploy the service", limit=2)
    texts = [r["record"]["text"] for r in out]
    assert any("deploy" in t for t in texts)
    assert all("password" not in t for t in texts)  # high > medium filtered
    rec = out[0]["record"]
    assert rec["recalls"] == 1  # reinforced


def test_engine_min_salience_filter(tmp_path):
    clock = FakeClock()
    eng = _engine(tmp_path, clock=clock)
    eng.remember("a", "ancient memory about deploys", salience=0.2)
    clock.advance_h(DECAY_HALF_LIFE_H * 2)  # 0.2 -> 0.05 < 0.1
    eng.remember("a", "unrelated topic entirely")
    out = eng.retrieve("a", "memory about deploys")
    assert all(r["record"]["text"] != "ancient memory about deploys" for r in out)


def test_engine_buffering_flush_on_size(tmp_path):
    eng = _engine(tmp_path, buffer_size=3)
    eng.remember("a", "one")
    eng.remember("a", "two")
    assert eng.index.size == 0  # buffered
    eng.remember("a", "three")
    assert eng.index.size == 3  # flushed at buffer_size


def test_engine_retrieve_sees_buffered(tmp_path):
    eng = _engine(tmp_path, buffer_size=100)
    eng.remember("a", "fresh fact about gpus")
    out = eng.retrieve("a", "fact about gpus", limit=1)
    assert out and out[0]["record"]["text"] == "fresh fact about gpus"


def test_engine_format_context(tmp_path):
    eng = _engine(tmp_path)
    eng.remember("a", "ship v2 on monday", kind="working")
    out = eng.retrieve("a", "when does v2 ship", limit=1)
    ctx = eng.format_context(out)
    assert "## Relevant memories" in ctx and "ship v2 on monday" in ctx
    assert eng.format_context([]) == ""


# -- hooks -------------------------------------------------------------------

def test_plugin_inject_and_ingest_flow(tmp_path, monkeypatch):
    monkeypatch.setenv("HOME", str(tmp_path))
    bus = HookBus()
    api = PluginApi(id="openclaw-membrane", plugin_config={"workspace": str(tmp_path)},
                    logger=NullLogger(), config={}, bus=bus)
    p = MembranePlugin(workspace=str(tmp_path))
    p.register(api)
    bus.emit("message_received", {"content": "we chose postgres for storage",
                                  "ctx": {"agentId": "dev"}})
    ev = bus.emit("message_received", {"content": "what storage did we choose",
                                       "ctx": {"agentId": "dev"}})
    assert "postgres" in ev.get("membrane_context", "")
    # other agent sees nothing (isolation)
    ev2 = bus.emit("message_received", {"content": "what storage did we choose",
                                        "ctx": {"agentId": "intern"}})
    assert "postgres" not in ev2.get("membrane_context", "")
    bus.emit("gateway_stop", {})
    assert "membranestatus" in api.commands
    assert api.gateway_methods["membrane.stats"]()["ingested"] >= 2


# -- GPU parity ---------------------------------------------------------------

@pytest.mark.gpu
def test_index_gpu_matches_cpu_reference():
    import numpy as np

    texts = [f"memory item number {i} about topic {i % 7}" for i in range(300)]
    ids = [f"r{i}" for i in range(300)]
    # encoder kernel requires dim % 1024 == 0
    cpu = SalienceIndex(dim=1024, capacity=512)
    gpu = SalienceIndex(dim=1024, capacity=512, device="cuda:0")
    cpu.add("a", ids, texts)
    gpu.add("a", ids, texts)
    for q in ("memory about topic 3", "item number 250"):
        got_c = [r for r, _ in cpu.search("a", q, k=5)]
        got_g = [r for r, _ in gpu.search("a", q, k=5)]
        assert len(set(got_c) & set(got_g)) >= 3  # bf16 vs fp32 tie tolerance


# ===========================================================================
# Membrane depth: sensitivity ceilings, retrieve limits, decay pass,
# configurator-key parity, stats counters
# ===========================================================================

def _eng(tmp_path, **cfg):
    from vainplex_openclaw_amd.membrane.engine import MembraneEngine

    t = [1_700_000_000.0]
    e = MembraneEngine(str(tmp_path), config=cfg or None, clock=lambda: t[0])
    return e, t


def test_engine_sensitivity_ceiling(tmp_path):
    e, _ = _eng(tmp_path, retrieve_max_sensitivity="medium")
    e.remember("a", "public fact about storage", sensitivity="low")
    e.remember("a", "medium fact about storage", sensitivity="medium")
    e.remember("a", "secret fact about storage", sensitivity="high")
    out = e.retrieve("a", "fact about storage", limit=10, min_salience=0.0)
    texts = [r["record"]["text"] for r in out]
    assert any("public" in t for t in texts)
    assert any("medium" in t for t in texts)
    assert not any("secret" in t for t in texts)
    # per-call override widens the ceiling
    out2 = e.retrieve("a", "fact about storage", limit=10, min_salience=0.0,
                      max_sensitivity="high")
    assert any("secret" in r["record"]["text"] for r in out2)


def test_engine_retrieve_limit_and_score_order(tmp_path):
    e, _ = _eng(tmp_path)
    for i in range(6):
        e.remember("a", f"note number {i} about the deploy pipeline")
    out = e.retrieve("a", "deploy pipeline", limit=3, min_salience=0.0)
    assert len(out) == 3
    scores = [r["score"] for r in out]
    assert scores == sorted(scores, reverse=True)
    for r in out:
        assert abs(r["score"] - r["cosine"] * r["salience"]) < 1e-9


def test_engine_decay_pass_prunes(tmp_path):
    e, t = _eng(tmp_path)
    e.remember("a", "ephemeral memory one")
    e.flush_buffer()
    t[0] += 400 * 24 * 3600       # far past many half-lives
    n = e.decay_pass("a")
    assert n >= 1
    assert e.store.count("a") == 0


def test_engine_stats_counters(tmp_path):
    e, _ = _eng(tmp_path)
    e.remember("a", "stat memory about things")
    e.retrieve("a", "memory about things", limit=1, min_salience=0.0)
    st = dict(e.stats)
    assert st.get("remembered", st.get("records", 1)) >= 1
    assert st["retrievals"] == 1 and st["retrieved"] >= 0


def test_engine_agent_isolation_in_retrieve(tmp_path):
    e, _ = _eng(tmp_path)
    e.remember("alice", "alice private plan for launch")
    e.remember("bob", "bob different note entirely")
    out = e.retrieve("bob", "private plan for launch", limit=5, min_salience=0.0)
    assert all("alice" not in r["record"]["text"] for r in out)


def test_configurator_membrane_keys_match_defaults():
    """brainplex configurator's Membrane keys (configurator.ts:137-148)
    line up with the engine's accepted config."""
    from vainplex_openclaw_amd.brainplex.configurator import generate_configs
    from vainplex_openclaw_amd.membrane.engine import MembraneEngine

    cfgs = generate_configs(["main"], "UTC", full=True)
    mem = next(c for c in cfgs if "membrane" in c["pluginId"])
    keys = set(mem["config"])
    assert {"buffer_size", "default_sensitivity", "retrieve_limit",
            "retrieve_min_salience", "retrieve_max_sensitivity"} <= keys
    # the engine accepts exactly these keys without error
    import tempfile

    MembraneEngine(tempfile.mkdtemp(), config=mem["config"])
