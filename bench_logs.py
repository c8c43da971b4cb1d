#!/usr/bin/env python3
"""Log fulltext MATCH benchmark (BASELINE config 4 shape).

Builds an N-event log fixture (template-pool messages + numeric fields)
directly as device-resident segments with GPU posting lists, then times
fulltext MATCHES queries (rare term, common term, AND of terms, match +
field predicate, row materialization).

  python bench_logs.py --events 20000000
"""

from __future__ import annotations

import argparse
import json
import os
import tempfile
import time

import numpy as np
import torch

START_MS = 1451606400000

WORDS = ("error timeout connection refused retry disk volume user login request "
         "response status latency upstream backend cache miss hit eviction "
         "thread pool exhausted queue overflow packet dropped route gateway "
         "dns resolve failure auth token expired session invalid payload "
         "parse checksum mismatch replica sync lag elect leader follower "
         "snapshot compact flush merge segment index probe scan commit abort "
         "transaction lock contention deadlock stall throttle quota limit").split()


def make_templates(n_templates: int, rng) -> np.ndarray:
    out = []
    for _ in range(n_templates):
        k = rng.randint(4, 9)
        out.append(" ".join(rng.choice(WORDS, size=k)))
    return np.array(out, dtype=object)


def load_log_fixture(engine, n_events: int, rank=0, world=1, seed=5):
    from greptimedb_amd.engine import sst as sst_mod
    from greptimedb_amd.models.schema import (ColumnSchema, DataType,
                                              SemanticType, TableSchema)
    rng = np.random.RandomState(seed + rank)
    schema = TableSchema(
        name="applogs",
        columns=[
            ColumnSchema("service", DataType.STRING, SemanticType.TAG, 0),
            ColumnSchema("ts", DataType.TIMESTAMP_MS, SemanticType.TIMESTAMP, 1),
            ColumnSchema("latency", DataType.FLOAT64, SemanticType.FIELD, 2),
            ColumnSchema("message", DataType.STRING, SemanticType.FIELD, 3, fulltext=True),
        ],
        primary_key=["service"], options={"append_mode": "true"})
    st = engine.create_table(schema, append_mode=True, if_not_exists=True)
    device = engine.config.device
    templates = make_templates(10_000, rng)
    # one rare marker template
    templates[0] = "xenon isotope anomaly detected in reactor four"
    n_my = n_events // world
    per_region = n_my // len(st.regions)
    services = [f"svc_{i}" for i in range(32)]
    total = 0
    for ridx, region in enumerate(st.regions):
        codes = region.register_series_bulk([(s,) for s in services])
        n = per_region
        tidx = rng.zipf(1.3, size=n) % len(templates)
        suffix = rng.randint(0, 1000, size=n)
        msgs = np.char.add(np.char.add(templates[tidx].astype(str), " req"),
                           suffix.astype(str)).astype(object)
        se = np.sort(codes[rng.randint(0, len(codes), size=n)]).astype(np.int32)
        ts = START_MS + np.arange(n, dtype=np.int64) * 100  # 10 ev/ms… synthetic
        lat = rng.exponential(10.0, size=n)
        ft = region.text_cols["message"]
        seg = ft.build_segment(list(msgs), device)
        batch = sst_mod.SstBatch(
            torch.as_tensor(ts).to(device),
            torch.as_tensor(se).to(device),
            torch.as_tensor(lat[None, :].copy()).to(device), None,
            int(ts[0]), int(ts[-1]), ["latency"])
        batch.str_cols["message"] = msgs
        batch.text_index["message"] = seg
        region.sst_cache[f"logfix_{ridx}"] = batch
        total += n
    return st, total


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--events", type=int, default=20_000_000)
    ap.add_argument("--iters", type=int, default=5)
    ap.add_argument("--warmup", type=int, default=2)
    args = ap.parse_args()

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    have_gpu = torch.cuda.is_available()
    device = f"cuda:{local_rank}" if have_gpu else "cpu"
    if have_gpu:
        torch.cuda.set_device(local_rank)
    if world > 1:
        import torch.distributed as dist
        dist.init_process_group("nccl" if have_gpu else "gloo")

    from greptimedb_amd.engine.engine import EngineConfig, MitoEngine
    from greptimedb_amd.parallel.dist import DistContext
    from greptimedb_amd.query.executor import Executor

    base = tempfile.mkdtemp(prefix="gdb_logs_")
    eng = MitoEngine(EngineConfig(data_dir=os.path.join(base, f"r{rank}"),
                                  device=device, background_flush=False))
    t0 = time.perf_counter()
    st, n_local = load_log_fixture(eng, args.events, rank, world)
    dist_ctx = DistContext(device=device) if world > 1 else None
    total = n_local if dist_ctx is None else int(dist_ctx.all_sum(n_local))
    if rank == 0:
        print(f"# log fixture: {total} events in {time.perf_counter()-t0:.1f}s",
              flush=True)

    ex = Executor(eng, dist=dist_ctx)
    queries = {
        "match-rare": "SELECT count(*) FROM applogs WHERE matches(message, 'xenon isotope')",
        "match-common": "SELECT count(*) FROM applogs WHERE matches(message, 'timeout')",
        "match-and": "SELECT count(*) FROM applogs WHERE matches(message, 'error timeout')",
        "match-field": ("SELECT count(*) FROM applogs WHERE "
                        "matches(message, 'deadlock') AND latency > 20"),
        "match-rows": ("SELECT ts, service, message FROM applogs WHERE "
                       "matches(message, 'xenon') ORDER BY ts LIMIT 100"),
    }
    results = {}
    for name, q in queries.items():
        times = []
        val = None
        for i in range(args.warmup + args.iters):
            if dist_ctx:
                dist_ctx.barrier()
            if have_gpu:
                torch.cuda.synchronize()
            qt0 = time.perf_counter()
            r = ex.execute(q)
            if have_gpu:
                torch.cuda.synchronize()
            dt = (time.perf_counter() - qt0) * 1000
            if i >= args.warmup:
                times.append(dt)
            val = r.columns[0][0] if len(r) and name != "match-rows" else len(r)
        results[name] = {"p50_ms": round(float(np.median(times)), 2),
                         "result": int(val) if val is not None else 0}
        if rank == 0:
            print(f"# {name}: p50 {np.median(times):.1f} ms → {val}", flush=True)
    if rank == 0:
        print(json.dumps({"bench": "log-fulltext", "events": total,
                          "n_gpus": world, "queries": results}), flush=True)
    eng.close()
    if world > 1:
        import torch.distributed as dist
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
