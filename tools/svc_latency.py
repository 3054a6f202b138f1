"""Serving-latency probe: p50/p95/p99 of FirewallService.check() against
the real pipeline at a small index (sync fast-path under load)."""
import sys
import threading
import time

sys.path.insert(0, ".")


def main(index_rows=8_000_000, n_checks=300, bg_rate_hz=200.0):
    from vainplex_openclaw_amd.pipeline.engine import FirewallPipeline, PipelineConfig
    from vainplex_openclaw_amd.pipeline.service import FirewallService
    from vainplex_openclaw_amd.pipeline.synth import synthetic_batch

    cfg = PipelineConfig(batch=1024, index_size=index_rows)
    pipe = FirewallPipeline(cfg, device="cuda:0")
    svc = FirewallService(pipe, max_batch=1024, max_wait_ms=5.0)
    msgs = synthetic_batch(512, seed=7, n_agents=cfg.n_agents).messages

    stop = threading.Event()

    def background():
        i = 0
        while not stop.is_set():
            svc.submit(msgs[i % len(msgs)])
            i += 1
            time.sleep(1.0 / bg_rate_hz)

    bg = threading.Thread(target=background, daemon=True)
    bg.start()
    # warmup
    for i in range(20):
        svc.check(msgs[i])
    lats = []
    for i in range(n_checks):
        t0 = time.perf_counter()
        out = svc.check(msgs[i % len(msgs)])
        lats.append((time.perf_counter() - t0) * 1000)
        assert out["verdict"] in ("allow", "audit", "flag", "deny")
    stop.set()
    svc.close()
    lats.sort()

    def pct(p):
        return lats[min(len(lats) - 1, int(p / 100 * len(lats)))]

    print(f"sync-check latency over {n_checks} checks @ {index_rows} rows, "
          f"{bg_rate_hz:.0f} msg/s background: "
          f"p50={pct(50):.1f}ms p95={pct(95):.1f}ms p99={pct(99):.1f}ms "
          f"max={lats[-1]:.1f}ms")


if __name__ == "__main__":
    main()
