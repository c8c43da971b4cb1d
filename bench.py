#!/usr/bin/env python3
"""Flagship benchmark: TSBS DevOps cpu-only ingest (+ query latencies).

Measures the BASELINE.json headline — TSBS ingest rows/sec (whole node) on
synthetic TSBS cpu-only data (scale=100 hosts per GPU, influx line protocol,
batch size 3000 like the reference's TSBS runs) — through the full ingest
path: C++ line-protocol parse → routing → WAL group commit → GPU memtable
append. Query latencies for the TSBS single-groupby-1-1-1 shape are reported
in config.queries (not part of the timed ingest region).

Contract (driver):
  python bench.py --gpus N --steps K --warmup W
N>1 is launched by the driver via torch.distributed.run (one rank per GPU,
RCCL). Weak scaling: each rank ingests its own disjoint 100-host shard.
Rank 0 prints ONE JSON line.
"""

from __future__ import annotations

import argparse
import json
import os
import shutil
import tempfile
import time

import numpy as np
import torch

ROWS_PER_BATCH = 3000          # TSBS --batch-size=3000
BATCHES_PER_STEP = 20          # 60k rows per step per rank
SCALE_PER_RANK = 100           # BASELINE config: scale=100 per MI355X
BASELINE_INGEST = 326_839.28   # rows/s, reference v0.12.0


def log(msg):
    if int(os.environ.get("RANK", "0")) == 0:
        print(msg, flush=True)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=10)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--scale", type=int, default=SCALE_PER_RANK)
    ap.add_argument("--workers", type=int, default=6,
                    help="ingest worker threads per rank (6 = TSBS's own "
                         "client worker count; measured best on MI355X)")
    ap.add_argument("--data-dir", default=None)
    ap.add_argument("--durable", action="store_true", default=True)
    ap.add_argument("--flush-mb", type=int, default=1024,
                    help="memtable flush threshold (small values exercise "
                         "flush+compaction during the timed region)")
    args = ap.parse_args()

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    have_gpu = torch.cuda.is_available()
    device = f"cuda:{local_rank}" if have_gpu else "cpu"
    if have_gpu:
        torch.cuda.set_device(local_rank)

    dist_on = world > 1
    if dist_on:
        import torch.distributed as dist
        backend = "nccl" if have_gpu else "gloo"
        dist.init_process_group(backend)

    from greptimedb_amd.engine.engine import EngineConfig, MitoEngine
    from greptimedb_amd.engine.ingest import Ingestor
    from greptimedb_amd.models.tsbs import CpuWorkload
    from greptimedb_amd.query.executor import Executor
    from greptimedb_amd.ops import hip_ops_available
    from greptimedb_amd.utils.errors import NativeExtensionMissing

    if have_gpu and not hip_ops_available():
        raise NativeExtensionMissing("HIP ops extension not built — bench refuses eager fallback")

    base = args.data_dir or tempfile.mkdtemp(prefix="gdb_bench_")
    data_dir = os.path.join(base, f"rank{rank}")
    shutil.rmtree(data_dir, ignore_errors=True)

    eng = MitoEngine(EngineConfig(
        data_dir=data_dir, device=device,
        background_flush=True, wal_sync=False,
        flush_bytes=args.flush_mb << 20))

    # ---------------- pre-generate all line batches (untimed) ----------------
    # Each worker thread owns a disjoint host shard (its own Ingestor/parser,
    # like the reference's 6 TSBS client workers; P5 write-worker axis) —
    # parse (C++, GIL released) and H2D copies overlap across workers.
    import threading

    n_workers = max(args.workers, 1)
    total_steps = args.warmup + args.steps
    gen_t0 = time.perf_counter()
    per_worker_scale = max(args.scale // n_workers, 1)
    workers = []
    for wi in range(n_workers):
        w = CpuWorkload(scale=per_worker_scale, seed=7 + rank * 100 + wi)
        w.tagsets = [t.replace(b"host_", b"host_%d_%d_" % (rank, wi))
                     for t in w.tagsets]
        batches = [
            [w.next_batch(ROWS_PER_BATCH)
             for _ in range(BATCHES_PER_STEP // n_workers)]
            for _ in range(total_steps)
        ]
        ing = Ingestor(eng, default_regions=4, append_mode=True,
                       durable=args.durable)
        workers.append((ing, batches))
    rows_per_step = n_workers * (BATCHES_PER_STEP // n_workers) * ROWS_PER_BATCH
    log(f"# generated {total_steps * rows_per_step} rows "
        f"in {time.perf_counter() - gen_t0:.1f}s ({n_workers} workers)")

    def barrier_sync():
        if dist_on:
            import torch.distributed as dist
            dist.barrier()
        if have_gpu:
            torch.cuda.synchronize()

    def run_steps(lo, hi):
        def worker_fn(ing, batches):
            for i in range(lo, hi):
                for b in batches[i]:
                    ing.ingest_lines(b)
        threads = [threading.Thread(target=worker_fn, args=wk) for wk in workers]
        for t in threads:
            t.start()
        for t in threads:
            t.join()

    # ---------------- warmup ----------------
    run_steps(0, args.warmup)
    barrier_sync()

    # ---------------- timed ----------------
    t0 = time.perf_counter()
    run_steps(args.warmup, total_steps)
    barrier_sync()
    elapsed = time.perf_counter() - t0

    # max over ranks
    if dist_on:
        import torch.distributed as dist
        t = torch.tensor([elapsed], dtype=torch.float64,
                         device=device if torch.distributed.get_backend() == "nccl" else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    rows_per_rank = args.steps * rows_per_step
    total_rows = rows_per_rank * world
    rows_per_s = total_rows / elapsed
    ms_per_step = elapsed / args.steps * 1000

    # ---------------- query latencies (informational, untimed region) -------
    from greptimedb_amd.parallel.dist import DistContext
    ex = Executor(eng, dist=DistContext(device=device) if dist_on else None)
    host = f"host_{rank}_0_0"
    t_lo = 1451606400000
    t_hi = t_lo + 3600_000
    q_single = (f"SELECT date_trunc('minute', ts) AS minute, max(usage_user) FROM cpu "
                f"WHERE hostname = '{host}' AND ts >= {t_lo} AND ts < {t_hi} "
                f"GROUP BY minute ORDER BY minute")
    q_double = ("SELECT date_trunc('hour', ts) AS hour, hostname, avg(usage_user) "
                "FROM cpu GROUP BY hour, hostname ORDER BY hour, hostname")
    queries = {}
    for name, q in [("single-groupby-1-1-1", q_single), ("double-groupby-1", q_double)]:
        times = []
        for _ in range(5):
            barrier_sync()
            qt0 = time.perf_counter()
            r = ex.execute(q)
            if have_gpu:
                torch.cuda.synchronize()
            times.append((time.perf_counter() - qt0) * 1000)
        queries[name + "_p50_ms"] = round(float(np.median(times)), 3)

    eng.close()

    result = {
        "metric": "TSBS DevOps ingest rows/sec (whole node)",
        "value": round(rows_per_s, 1),
        "unit": "rows/s",
        "n_gpus": world,
        "steps": args.steps,
        "warmup": args.warmup,
        "ms_per_step": round(ms_per_step, 2),
        "higher_is_better": True,
        "scaling": "weak",
        "vs_baseline": round(rows_per_s / BASELINE_INGEST, 3),
        "dtype": "f64",
        "data": "synthetic TSBS cpu-only (influx line protocol, random-walk values)",
        "config": {
            "model": "tsbs-devops-cpu-only",
            "global_batch": ROWS_PER_BATCH,
            "seq_len": 0,
            "parallelism": f"region-shard dp{world}",
            "scale_per_gpu": args.scale,
            "workers": n_workers,
            "batches_per_step": BATCHES_PER_STEP,
            "wal": "group-commit, no fsync",
            "flush_mb": args.flush_mb,
            "queries": queries,
            "device": device,
        },
    }
    if rank == 0:
        print(json.dumps(result), flush=True)
    if dist_on:
        import torch.distributed as dist
        dist.destroy_process_group()
    if args.data_dir is None:
        shutil.rmtree(base, ignore_errors=True)


if __name__ == "__main__":
    main()
