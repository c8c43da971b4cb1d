#!/usr/bin/env python3
"""HTTP serving throughput (reference: TSBS query throughput @N clients,
v0.9.1: single-groupby-1-1-1 1,511 qps @50 clients on Ryzen 7).

Serving model: N server PROCESSES on one GPU, each with its own engine
over the same flushed data directory (read replicas — the same shape as
meta/replication followers, here sharing the SST files). One Python
process is GIL-bound at ~600 qps; replicas scale it like the reference's
worker threads scale its tokio runtime. Clients run in separate processes
(in-process clients would steal the server's GIL and corrupt the
measurement).

  python bench_serving.py --clients 50 --duration 10 --server-procs 4
"""

from __future__ import annotations

import argparse
import json
import multiprocessing as mp
import os
import shutil
import tempfile
import time


def serve_proc(port: int, data_dir: str, device: str):
    import torch
    import uvicorn

    from greptimedb_amd.engine.engine import EngineConfig, MitoEngine
    from greptimedb_amd.servers.http import ServerContext, build_app

    torch.set_num_threads(2)
    eng = MitoEngine(EngineConfig(data_dir=data_dir, device=device,
                                  background_flush=False))
    app = build_app(ServerContext(eng))
    uvicorn.run(app, host="127.0.0.1", port=port, log_level="error")


def client_proc(ports: list, queries: list, t_start: float, duration: float,
                nthreads: int, out_q):
    import threading

    import httpx

    counts = [0] * nthreads
    lats: list[list[float]] = [[] for _ in range(nthreads)]
    clients = [httpx.Client(
        base_url=f"http://127.0.0.1:{ports[i % len(ports)]}", timeout=60)
        for i in range(nthreads)]
    for ci, c in enumerate(clients):          # prime connections
        c.get("/v1/sql", params={"sql": queries[ci % len(queries)]})
    while time.time() < t_start:
        time.sleep(0.005)
    stop = t_start + duration

    def run(ci):
        c = clients[ci]
        i = ci
        while time.time() < stop:
            q = queries[i % len(queries)]
            i += nthreads
            t0 = time.perf_counter()
            r = c.get("/v1/sql", params={"sql": q})
            lats[ci].append((time.perf_counter() - t0) * 1000)
            assert r.status_code == 200, r.text[:200]
            counts[ci] += 1

    threads = [threading.Thread(target=run, args=(i,)) for i in range(nthreads)]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    out_q.put((sum(counts), [x for l in lats for x in l]))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--clients", type=int, default=50)
    ap.add_argument("--duration", type=float, default=10.0)
    ap.add_argument("--scale", type=int, default=4000)
    ap.add_argument("--hours", type=int, default=12)
    ap.add_argument("--port", type=int, default=14123)
    ap.add_argument("--server-procs", type=int, default=4)
    ap.add_argument("--client-procs", type=int, default=4)
    args = ap.parse_args()

    import numpy as np
    import torch

    from greptimedb_amd.engine.engine import EngineConfig, MitoEngine
    from greptimedb_amd.models.tsbs_fixture import START_TS_S, load_cpu_fixture

    device = "cuda:0" if torch.cuda.is_available() else "cpu"
    data_dir = tempfile.mkdtemp(prefix="gdb_serve_")
    eng = MitoEngine(EngineConfig(data_dir=data_dir, device=device,
                                  background_flush=False))
    n = load_cpu_fixture(eng, scale=args.scale, hours=args.hours)
    eng.flush_all()          # replicas open the SSTs
    eng.close()
    del eng
    if device.startswith("cuda"):
        torch.cuda.empty_cache()
    print(f"# fixture: {n} rows flushed ({device})", flush=True)

    ctx = mp.get_context("spawn")
    ports = [args.port + i for i in range(args.server_procs)]
    servers = [ctx.Process(target=serve_proc, args=(p, data_dir, device),
                           daemon=True) for p in ports]
    for s in servers:
        s.start()
    import httpx
    t_up = time.time()
    for p in ports:
        for _ in range(900):
            try:
                httpx.get(f"http://127.0.0.1:{p}/health", timeout=1)
                break
            except Exception:
                time.sleep(0.2)
    print(f"# {len(ports)} replicas up in {time.time() - t_up:.0f}s", flush=True)

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
    per_proc = max(args.clients // args.client_procs, 1)
    t_start = time.time() + 3.0          # clients prime, then start together
    out_q = ctx.Queue()
    cps = [ctx.Process(target=client_proc,
                       args=(ports, queries, t_start, args.duration,
                             per_proc, out_q))
           for _ in range(args.client_procs)]
    for c in cps:
        c.start()
    total = 0
    all_lats: list = []
    for _ in cps:
        cnt, lat = out_q.get()
        total += cnt
        all_lats.extend(lat)
    for c in cps:
        c.join()
    lat_a = np.asarray(all_lats)
    print(json.dumps({
        "bench": "http-serving", "query": "single-groupby-1-1-1",
        "clients": per_proc * args.client_procs,
        "server_procs": args.server_procs,
        "qps": round(total / args.duration, 1),
        "mean_ms": round(float(lat_a.mean()), 2),
        "p50_ms": round(float(np.percentile(lat_a, 50)), 2),
        "p99_ms": round(float(np.percentile(lat_a, 99)), 2),
        "queries": total, "device": device,
        "ref_qps_50_clients": 1511.74,
    }), flush=True)
    for s in servers:
        s.terminate()
    shutil.rmtree(data_dir, ignore_errors=True)


if __name__ == "__main__":
    main()
