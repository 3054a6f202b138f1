"""Concurrency stress battery (SURVEY §5 race-detection row).

The reference is single-threaded Node and handles concurrency "by
construction"; this build has real threads (journal writer, audit
writer, 2FA resolution vs request, service micro-batcher, TTL caches).
There is no compute-sanitizer equivalent on ROCm and TSAN cannot wrap a
torch extension practically, so the race posture is: Python-side
invariants hammered from many threads — no lost updates, no duplicate
sequence numbers, no deadlocks, counters that add up — run in CI with
the normal suite.
"""

import threading
import time

import pytest


def _hammer(n_threads, fn):
    errs = []
    barrier = threading.Barrier(n_threads)

    def run(i):
        try:
            barrier.wait(timeout=10)
            fn(i)
        except Exception as exc:  # pragma: no cover
            errs.append(f"t{i}: {type(exc).__name__}: {exc}")

    ts = [threading.Thread(target=run, args=(i,)) for i in range(n_threads)]
    for t in ts:
        t.start()
    for t in ts:
        t.join(timeout=30)
    assert not errs, errs
    assert not any(t.is_alive() for t in ts), "deadlocked thread"


def test_journal_concurrent_publish_unique_seqs():
    from vainplex_openclaw_amd.eventstore import EventJournal

    j = EventJournal(durable=False)
    N, T = 200, 8
    seqs = [[] for _ in range(T)]

    def pub(i):
        for k in range(N):
            if k % 10 == 0:
                blob = b'{"id": "b"}\n' * 3
                seqs[i].append(j.publish_block(f"s.t{i}", blob, 3))
            else:
                seqs[i].append(j.publish(f"s.t{i}", {"ts": k, "id": f"e{i}-{k}"}))

    _hammer(T, pub)
    flat = [s for lane in seqs for s in lane]
    assert len(flat) == len(set(flat))           # no duplicate seq starts
    total = T * (180 + 20 * 3)                   # singles + blocks
    assert len(j) == total
    replayed = [s for s, _ in j.replay()]
    assert len(replayed) == total
    assert replayed == sorted(replayed)
    assert len(set(replayed)) == total           # per-event seqs unique


def test_journal_concurrent_publish_with_retention_and_replay():
    from vainplex_openclaw_amd.eventstore import EventJournal

    j = EventJournal(durable=False, max_messages=500)
    stop = threading.Event()
    replay_counts = []

    def reader(i):
        while not stop.is_set():
            replay_counts.append(sum(1 for _ in j.replay()))

    def writer(i):
        for k in range(400):
            j.publish(f"s.w{i}", {"ts": k, "id": f"w{i}-{k}"})
        stop.set()

    errs = []

    def guarded(fn, i):
        try:
            fn(i)
        except Exception as exc:
            errs.append(str(exc))

    ts = [threading.Thread(target=guarded, args=(writer, i)) for i in range(4)]
    ts += [threading.Thread(target=guarded, args=(reader, 9))]
    for t in ts:
        t.start()
    for t in ts:
        t.join(timeout=30)
    assert not errs
    assert len(j) <= 500                          # retention held under load
    assert all(c <= 1600 for c in replay_counts)


def test_vault_concurrent_store_resolve():
    from vainplex_openclaw_amd.governance.redaction.vault import RedactionVault

    v = RedactionVault()
    out = [dict() for _ in range(8)]

    def work(i):
        for k in range(300):
            ph = v.store(f"secret-{k % 50}", "credential")
            out[i][f"secret-{k % 50}"] = ph

    _hammer(8, work)
    # same value always produced the same placeholder across threads
    merged = {}
    for d in out:
        for val, ph in d.items():
            assert merged.setdefault(val, ph) == ph
            assert v.lookup(ph) == val


def test_2fa_concurrent_requests_and_resolution():
    from vainplex_openclaw_amd.governance.approval_2fa import (
        Approval2FA,
        generate_secret,
        totp_at,
    )

    t = [1_700_000_000.0]
    fa = Approval2FA(secret=generate_secret(), clock=lambda: t[0])
    reqs = [[] for _ in range(6)]

    def request(i):
        for k in range(50):
            reqs[i].append(fa.request(f"sess-{i}", f"a{i}", f"cmd-{k}"))

    _hammer(6, request)
    all_reqs = [r for lane in reqs for r in lane]
    assert len(all_reqs) == 300
    resolved = fa.try_resolve_any(totp_at(fa.secret, t[0]))
    # one code resolves every pending request exactly once
    assert len(resolved) == sum(1 for r in all_reqs if r["status"] == "approved")
    assert all(r["status"] == "approved" for r in all_reqs)
    assert not fa.has_pending_batch()


def test_frequency_tracker_concurrent_records():
    from vainplex_openclaw_amd.governance.frequency import FrequencyTracker

    ft = FrequencyTracker(capacity=5000)
    t0 = time.time()

    def rec(i):
        for k in range(500):
            ft.record(f"a{i}", f"s{i}", "tool")

    _hammer(8, rec)
    assert ft.count(3600, scope="global") == 4000     # no lost updates
    assert ft.count(3600, scope="agent", agent_id="a3") == 500


def test_service_concurrent_submitters_cpu():
    from vainplex_openclaw_amd.pipeline.service import FirewallService

    class FakePipe:
        cfg = type("C", (), {"batch": 64})()

        def step(self, batch, staged=None):
            import torch

            n = len(batch.messages)
            z = torch.zeros(n, dtype=torch.int64)
            return {
                "verdict": torch.zeros(n, dtype=torch.int8),
                "risk": torch.zeros(n),
                "hits": {"injection": z, "redaction": z,
                         "claims": z, "entity": z, "cortex": z},
                "logits": torch.zeros(n, 8),
                "recall_scores": torch.zeros(n, 4),
                "recall_ids": torch.zeros(n, 4, dtype=torch.int32),
                "merkle_root": torch.zeros(32, dtype=torch.uint8),
                "trust_scores": torch.zeros(8),
            }

        def stage(self, batch):
            return None

    svc = FirewallService(FakePipe(), max_batch=64, max_wait_ms=2.0)
    results = [[] for _ in range(8)]
    try:
        def submit(i):
            futs = [svc.submit(f"msg {i}-{k}".encode()) for k in range(40)]
            results[i] = [f.result(timeout=20) for f in futs]

        _hammer(8, submit)
    finally:
        svc.close()
    assert all(len(lane) == 40 for lane in results)
    assert all(o["verdict"] == "allow" for lane in results for o in lane)
