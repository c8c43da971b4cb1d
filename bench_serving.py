#!/usr/bin/env python3
"""HTTP serving throughput (reference: TSBS query throughput @N clients,
v0.9.1: single-groupby-1-1-1 1,511 qps @50 clients on Ryzen 7).

Starts the real standalone HTTP server (uvicorn) in-process, loads a
scale=4000 TSBS fixture, then hammers /v1/sql with N client threads.

  python bench_serving.py --clients 50 --duration 10
"""

from __future__ import annotations

import argparse
import json
import threading
import time

import numpy as np
import torch


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--clients", type=int, default=50)
    ap.add_argument("--duration", type=float, default=10.0)
    ap.add_argument("--scale", type=int, default=4000)
    ap.add_argument("--hours", type=int, default=24)
    ap.add_argument("--port", type=int, default=14123)
    args = ap.parse_args()

    import tempfile

    import httpx
    import uvicorn

    from greptimedb_amd.engine.engine import EngineConfig, MitoEngine
    from greptimedb_amd.models.tsbs_fixture import START_TS_S, load_cpu_fixture
    from greptimedb_amd.servers.http import ServerContext, build_app

    device = "cuda:0" if torch.cuda.is_available() else "cpu"
    # concurrent queries each spawn torch intra-op threads — cap to avoid
    # oversubscription under N in-flight queries
    torch.set_num_threads(2)
    eng = MitoEngine(EngineConfig(data_dir=tempfile.mkdtemp(prefix="gdb_serve_"),
                                  device=device, background_flush=False))
    n = load_cpu_fixture(eng, scale=args.scale, hours=args.hours)
    print(f"# fixture: {n} rows on {device}", flush=True)
    app = build_app(ServerContext(eng))
    server = uvicorn.Server(uvicorn.Config(app, host="127.0.0.1", port=args.port,
                                           log_level="error"))
    th = threading.Thread(target=server.run, daemon=True)
    th.start()
    base = f"http://127.0.0.1:{args.port}"
    for _ in range(100):
        try:
            httpx.get(base + "/health", timeout=1)
            break
        except Exception:
            time.sleep(0.1)

    rng = np.random.RandomState(9)
    t0_ms = START_TS_S * 1000
    span = args.hours * 3600_000 - 3600_000

    def make_query():
        h = rng.randint(args.scale)
        lo = t0_ms + int(rng.randint(max(span, 1)))
        return (f"SELECT date_trunc('minute', ts) AS minute, max(usage_user) "
                f"FROM cpu WHERE hostname = 'host_{h}' AND ts >= {lo} "
                f"AND ts < {lo + 3600_000} GROUP BY minute ORDER BY minute")

    queries = [make_query() for _ in range(512)]
    stop = time.perf_counter() + args.duration
    counts = [0] * args.clients
    lats: list[list[float]] = [[] for _ in range(args.clients)]

    clients = [httpx.Client(base_url=base, timeout=60) for _ in range(args.clients)]

    def client(ci):
        c = clients[ci]
        i = ci
        while time.perf_counter() < stop:
            q = queries[i % len(queries)]
            i += args.clients
            t0 = time.perf_counter()
            r = c.get("/v1/sql", params={"sql": q})
            lats[ci].append((time.perf_counter() - t0) * 1000)
            assert r.status_code == 200
            counts[ci] += 1

    # warmup (also primes each client's connection)
    for ci, c in enumerate(clients):
        c.get("/v1/sql", params={"sql": queries[ci % len(queries)]})
    t_start = time.perf_counter()
    stop = t_start + args.duration
    threads = [threading.Thread(target=client, args=(i,))
               for i in range(args.clients)]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    wall = time.perf_counter() - t_start
    total = sum(counts)
    all_lats = np.concatenate([np.asarray(l) for l in lats if l])
    print(json.dumps({
        "bench": "http-serving", "query": "single-groupby-1-1-1",
        "clients": args.clients, "qps": round(total / wall, 1),
        "mean_ms": round(float(all_lats.mean()), 2),
        "p50_ms": round(float(np.percentile(all_lats, 50)), 2),
        "p99_ms": round(float(np.percentile(all_lats, 99)), 2),
        "queries": total, "device": device,
        "ref_qps_50_clients": 1511.74,
    }), flush=True)
    server.should_exit = True
    eng.close()


if __name__ == "__main__":
    main()
