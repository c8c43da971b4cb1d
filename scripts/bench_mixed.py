#!/usr/bin/env python3
"""Mixed workload: sustained ingest INTO the query table while the TSBS
query suite runs against it — the production serving shape (reference:
tsbs_load + tsbs_run_queries against one live cluster).

Phases:
  1. load the scale-4000 cpu fixture (reference benchmark config)
  2. idle query baseline: p50 per query, no ingest running
  3. mixed: N ingest workers replay line-protocol batches into the SAME
     `cpu` table (WAL group commit + background flush live) while a query
     thread loops the suite; report ingest rows/s under query load and
     query p50/p95 under ingest load.

Run on the GPU box:
  python scripts/bench_mixed.py [--hours 72] [--mixed-s 25]
"""
import argparse
import json
import os
import sys
import tempfile
import threading
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch  # noqa: E402  (loads libc10 before our extensions)
import numpy as np  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--scale", type=int, default=4000)
    ap.add_argument("--hours", type=int, default=72)
    ap.add_argument("--workers", type=int, default=4)
    ap.add_argument("--mixed-s", type=float, default=25.0)
    ap.add_argument("--reps", type=int, default=7)
    ap.add_argument("--pool-rows", type=int, default=1_200_000)
    args = ap.parse_args()

    from greptimedb_amd.engine.engine import EngineConfig, MitoEngine
    from greptimedb_amd.engine.ingest import Ingestor
    from greptimedb_amd.models.tsbs import CpuWorkload
    from greptimedb_amd.models.tsbs_fixture import load_cpu_fixture, tsbs_queries
    from greptimedb_amd.query.executor import Executor

    have_gpu = torch.cuda.is_available()
    device = "cuda:0" if have_gpu else "cpu"
    if not have_gpu:
        args.scale, args.hours = 100, 12
        args.pool_rows = 60_000

    d = tempfile.mkdtemp(prefix="gdb_mixed_")
    eng = MitoEngine(EngineConfig(data_dir=d, device=device,
                                  background_flush=True, wal_shards=4,
                                  flush_bytes=64 << 20))
    t0 = time.perf_counter()
    rows = load_cpu_fixture(eng, scale=args.scale, hours=args.hours)
    if have_gpu:
        torch.cuda.synchronize()
    print(f"# fixture: {rows} rows in {time.perf_counter()-t0:.1f}s", flush=True)

    ex = Executor(eng)
    suite = tsbs_queries(args.scale, args.hours)
    names = ["single-groupby-1-1-1", "single-groupby-1-8-1",
             "double-groupby-1", "lastpoint", "high-cpu-1"]
    names = [n for n in names if n in suite]

    def run_q(n):
        qt = time.perf_counter()
        ex.execute(suite[n])
        if have_gpu:
            torch.cuda.synchronize()
        return (time.perf_counter() - qt) * 1000

    idle = {}
    for n in names:
        ts = [run_q(n) for _ in range(args.reps)]
        idle[n] = round(float(np.median(ts)), 2)
    print(f"# idle p50 ms: {idle}", flush=True)

    # ---- pre-generate ingest pool: same `cpu` measurement, new hosts ----
    pools = []
    for wi in range(args.workers):
        w = CpuWorkload(scale=200, seed=100 + wi)
        w.tagsets = [t.replace(b"host_", b"ihost_%d_" % wi) for t in w.tagsets]
        nb = max(args.pool_rows // (args.workers * 3000), 1)
        pools.append([w.next_batch(3000) for _ in range(nb)])

    stop = [False]
    ingested = [0] * args.workers

    def ingest_worker(wi):
        ing = Ingestor(eng, default_regions=4, append_mode=True, durable=True)
        i = 0
        bs = pools[wi]
        while not stop[0]:
            ing.ingest_lines(bs[i % len(bs)])
            i += 1
            ingested[wi] = i * 3000

    ths = [threading.Thread(target=ingest_worker, args=(wi,))
           for wi in range(args.workers)]
    m0 = time.perf_counter()
    for t in ths:
        t.start()
    time.sleep(1.0)  # let ingest reach steady state
    mixed: dict[str, list[float]] = {n: [] for n in names}
    while time.perf_counter() - m0 < args.mixed_s:
        for n in names:
            mixed[n].append(run_q(n))
    stop[0] = True
    for t in ths:
        t.join()
    m1 = time.perf_counter()
    total_ing = sum(ingested)
    ing_rps = total_ing / (m1 - m0)

    out = {
        "bench": "mixed-ingest+query", "device": device,
        "fixture_rows": rows, "scale": args.scale, "hours": args.hours,
        "ingest_workers": args.workers,
        "ingest_rows_per_s_under_query_load": round(ing_rps, 1),
        "mixed_window_s": round(m1 - m0, 1),
        "queries": {n: {"idle_p50_ms": idle[n],
                        "mixed_p50_ms": round(float(np.median(mixed[n])), 2),
                        "mixed_p95_ms": round(float(np.percentile(mixed[n], 95)), 2),
                        "reps_under_load": len(mixed[n])}
                    for n in names},
    }
    print(json.dumps(out), flush=True)
    os.makedirs("gpurun_out", exist_ok=True)
    with open("gpurun_out/bench_mixed.json", "w") as f:
        json.dump(out, f, indent=1)
    # long soaks fill the scratch tmpfs with WAL+SST bytes; skip python
    # teardown (which can hit allocator failures near the host-memory
    # ceiling) — results are already flushed to disk above
    try:
        eng.close()
    except Exception:
        pass
    os._exit(0)


if __name__ == "__main__":
    main()
