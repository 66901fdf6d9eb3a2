"""End-to-end throughput/latency bench of the scoring daemon.

Starts `factorvae_amd.serve` in-process on a background thread (uvicorn),
then drives it over real HTTP on loopback with CSI300-shaped
cross-sections (N=300, T=20, C=158, random fp32) and reports, per
endpoint:

  /score_raw  binary float32 body      (production fast path)
  /score      JSON                      (debug/interactive path)

Metrics: requests/s, cross-sections/s (= requests/s here: one day per
request), and p50/p95 latency. Run on a GPU box:

  python scripts/bench_serve.py [--n 300] [--reqs 200] [--port 8441]
"""
import argparse
import os
import sys
import threading
import time

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--n", type=int, default=300)
    p.add_argument("--t", type=int, default=20)
    p.add_argument("--c", type=int, default=158)
    p.add_argument("--reqs", type=int, default=200)
    p.add_argument("--warmup", type=int, default=20)
    p.add_argument("--port", type=int, default=8441)
    p.add_argument("--skip-json", action="store_true",
                   help="only bench /score_raw (JSON is ~300x slower at "
                        "large N; skips a multi-minute wait)")
    args = p.parse_args()

    from factorvae_amd.serve import ScoringEngine, build_app
    import uvicorn

    eng = ScoringEngine(None, num_latent=args.c, hidden_size=64,
                        num_portfolio=128, num_factor=96,
                        seq_length=args.t)
    print(f"engine={eng.engine} device={eng.device}")
    app = build_app(eng)
    cfg = uvicorn.Config(app, host="127.0.0.1", port=args.port,
                         log_level="error")
    server = uvicorn.Server(cfg)
    th = threading.Thread(target=server.run, daemon=True)
    th.start()
    for _ in range(100):
        if server.started:
            break
        time.sleep(0.1)

    import httpx

    rng = np.random.default_rng(0)
    x = rng.standard_normal((args.n, args.t, args.c)).astype("<f4")
    base = f"http://127.0.0.1:{args.port}"
    with httpx.Client(base_url=base, timeout=60.0) as client:
        r = client.get("/health")
        print("health:", r.json())

        def run(name, fn):
            for _ in range(args.warmup):
                fn()
            lat = []
            t0 = time.perf_counter()
            for _ in range(args.reqs):
                s = time.perf_counter()
                r = fn()
                lat.append(time.perf_counter() - s)
                assert r.status_code == 200, r.text
            el = time.perf_counter() - t0
            lat = np.sort(np.array(lat))
            print(f"{name}: {args.reqs/el:8.1f} req/s "
                  f"({args.n*args.reqs/el:9.0f} stock-scores/s)  "
                  f"p50 {lat[len(lat)//2]*1e3:6.2f} ms  "
                  f"p95 {lat[int(len(lat)*0.95)]*1e3:6.2f} ms",
                  flush=True)

        body = x.tobytes()
        run("/score_raw (binary)",
            lambda: client.post("/score_raw", content=body))
        if not args.skip_json:
            payload = {"x": x.tolist()}
            run("/score     (json)  ",
                lambda: client.post("/score", json=payload))
    server.should_exit = True
    th.join(timeout=5)


if __name__ == "__main__":
    main()
