#!/usr/bin/env python3
"""TSBS DevOps query-latency benchmark on the direct-load fixture.

Mirrors the reference's published query table (BASELINE.md, v0.12.0:
scale=4000, 3 days @10s ≈ 104M rows): bulk-loads the same-shape synthetic
fixture into device memory and times each TSBS query. Distributed: launch
via torch.distributed.run; hosts shard across ranks, queries all-reduce.

  python bench_queries.py [--scale 4000] [--hours 72] [--iters 7]
"""

from __future__ import annotations

import argparse
import json
import os
import tempfile
import time

import numpy as np
import torch

REFERENCE_MS = {  # docs/benchmarks/tsbs/v0.12.0.md (EC2 c5d.2xlarge)
    "cpu-max-all-1": 12.46, "cpu-max-all-8": 24.20,
    "double-groupby-1": 673.08, "double-groupby-5": 963.99,
    "double-groupby-all": 1330.05, "groupby-orderby-limit": 952.46,
    "high-cpu-1": 5.08, "high-cpu-all": 4638.57, "lastpoint": 591.02,
    "single-groupby-1-1-1": 4.06, "single-groupby-1-1-12": 4.73,
    "single-groupby-1-8-1": 8.23, "single-groupby-5-1-1": 4.61,
    "single-groupby-5-1-12": 5.61, "single-groupby-5-8-1": 9.74,
}


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--scale", type=int, default=4000)
    ap.add_argument("--hours", type=int, default=72)
    ap.add_argument("--iters", type=int, default=7)
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
    from greptimedb_amd.models.tsbs_fixture import load_cpu_fixture, tsbs_queries
    from greptimedb_amd.parallel.dist import DistContext
    from greptimedb_amd.query.executor import Executor

    base = tempfile.mkdtemp(prefix="gdb_qbench_")
    eng = MitoEngine(EngineConfig(data_dir=os.path.join(base, f"r{rank}"),
                                  device=device, background_flush=False))
    t0 = time.perf_counter()
    n = load_cpu_fixture(eng, scale=args.scale, hours=args.hours,
                         rank=rank, world=world)
    load_s = time.perf_counter() - t0
    total_rows = n
    dist_ctx = None
    if world > 1:
        import torch.distributed as dist
        dist_ctx = DistContext(device=device)
        total_rows = int(dist_ctx.all_sum(n))
    if rank == 0:
        print(f"# fixture: {total_rows} rows loaded in {load_s:.1f}s "
              f"({args.scale} hosts x {args.hours}h, world={world})", flush=True)

    ex = Executor(eng, dist=dist_ctx)
    queries = tsbs_queries(args.scale, args.hours)
    results = {}
    for name, sql in queries.items():
        times = []
        rows = 0
        for i in range(args.warmup + args.iters):
            if dist_ctx:
                dist_ctx.barrier()
            if have_gpu:
                torch.cuda.synchronize()
            qt0 = time.perf_counter()
            r = ex.execute(sql)
            if have_gpu:
                torch.cuda.synchronize()
            dt = (time.perf_counter() - qt0) * 1000
            if i >= args.warmup:
                times.append(dt)
            rows = len(r)
        p50 = float(np.median(times))
        ref = REFERENCE_MS.get(name)
        results[name] = {
            "p50_ms": round(p50, 3),
            "mean_ms": round(float(np.mean(times)), 3),
            "rows": rows,
            "ref_ms": ref,
            "speedup_vs_ref": round(ref / p50, 1) if ref else None,
        }
        if rank == 0:
            print(f"# {name}: p50 {p50:.2f} ms (ref {ref} ms) rows={rows}", flush=True)

    if rank == 0:
        print(json.dumps({
            "bench": "tsbs-query-suite",
            "scale": args.scale, "hours": args.hours, "rows": total_rows,
            "n_gpus": world, "device": device,
            "queries": results,
        }), flush=True)
    eng.close()
    if world > 1:
        import torch.distributed as dist
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
