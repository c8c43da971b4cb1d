#!/usr/bin/env python3
"""Flagship benchmark: TSBS DevOps cpu-only — sustained ingest + query suite.

Measures the BASELINE.json headline metric ("TSBS DevOps: ingest rows/sec
(whole node) + p50 query latency"):

  1. SUSTAINED INGEST (the timed region): each step ingests a fixed number
     of rows (default 10M per rank) through the full path — C++ influx
     line-protocol parse → routing → WAL group commit → GPU memtable
     append — with background flush + TWCS compaction LIVE during the
     timed region (--flush-mb 64 default). Line batches are pre-generated
     once and replayed (exactly TSBS methodology: tsbs_load replays a
     pre-generated data file); every replayed row is fully re-parsed,
     re-WAL-written and re-appended.
  2. QUERY SUITE (after the timed region, reported in config.queries):
     the 16-query TSBS DevOps suite on a scale=4000 / 72h ≈ 104M-row
     device-resident fixture (reference's own benchmark config,
     docs/benchmarks/tsbs/v0.12.0.md) — p50 over --query-reps runs.

Contract (driver):
  python bench.py --gpus N --steps K --warmup W
N>1 is launched by the driver via torch.distributed.run (one rank per GPU,
RCCL). Weak scaling: each rank ingests its own disjoint host shard and
holds a hash shard of the scale-4000 query fixture. Rank 0 prints ONE JSON
line.
"""

from __future__ import annotations

import argparse
import json
import os
import shutil
import tempfile
import threading
import time

import numpy as np
import torch

ROWS_PER_BATCH = 3000          # TSBS --batch-size=3000
BASELINE_INGEST = 326_839.28   # rows/s, reference v0.12.0


def log(msg):
    if int(os.environ.get("RANK", "0")) == 0:
        print(msg, flush=True)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=10)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--step-rows", type=int, default=None,
                    help="rows ingested per step per rank (default 10M on "
                         "GPU — ≥30s sustained over 20 steps — 60k on CPU)")
    ap.add_argument("--scale", type=int, default=100,
                    help="TSBS scale (hosts) per ingest worker shard")
    ap.add_argument("--workers", type=int, default=6,
                    help="ingest worker threads per rank (tsbs_load "
                         "--workers analog; 6 vs 8 within run-to-run "
                         "variance on MI355X — see profiles worker sweep)")
    ap.add_argument("--pool-rows", type=int, default=None,
                    help="pre-generated line pool size per rank (replayed "
                         "cyclically; default 2.4M on GPU, 120k on CPU)")
    ap.add_argument("--data-dir", default=None)
    ap.add_argument("--wal", choices=["buffered", "fsync", "off"],
                    default="buffered",
                    help="WAL durability for ingest: buffered = group "
                         "commit, no fsync (reference raft-engine default "
                         "sync=false); fsync = fdatasync per group commit; "
                         "off = no WAL")
    ap.add_argument("--wal-shards", type=int, default=4,
                    help="parallel WAL writers (region-sharded segments, "
                         "merged by seq at replay)")
    ap.add_argument("--flush-mb", type=int, default=64,
                    help="per-region memtable flush threshold — 64MB keeps "
                         "flush+compaction live during the timed region")
    ap.add_argument("--query-scale", type=int, default=None,
                    help="TSBS scale for the query-suite fixture "
                         "(default 4000 on GPU — the reference's published "
                         "config — 100 on CPU)")
    ap.add_argument("--query-hours", type=int, default=None,
                    help="fixture time span (default 72h on GPU = 104M rows)")
    ap.add_argument("--query-reps", type=int, default=7)
    ap.add_argument("--skip-queries", action="store_true")
    ap.add_argument("--no-route", action="store_true",
                    help="diagnostic: pre-sharded ingest (no cross-rank "
                         "fan-out); NOT the default because any-rank "
                         "routing is the honest distributed path")
    args = ap.parse_args()

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    have_gpu = torch.cuda.is_available()
    device = f"cuda:{local_rank}" if have_gpu else "cpu"
    if have_gpu:
        torch.cuda.set_device(local_rank)

    step_rows = args.step_rows or (10_000_000 if have_gpu else 60_000)
    pool_rows = args.pool_rows or (2_400_000 if have_gpu else 120_000)
    q_scale = args.query_scale or (4000 if have_gpu else 100)
    q_hours = args.query_hours or (72 if have_gpu else 12)

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
        background_flush=True, wal_sync=(args.wal == "fsync"),
        wal_shards=args.wal_shards,
        flush_bytes=args.flush_mb << 20))

    # -------- cross-rank write fan-out (any rank accepts any write) ---------
    # Each rank's clients generate their own host universe (weak scaling),
    # but OWNERSHIP is global: series hash to any rank, and rows are shipped
    # to the owner (parallel/write_fanout.py; reference insert.rs:389-496).
    exchange = None
    if world > 1 and not args.no_route:
        from greptimedb_amd.parallel.write_fanout import WriteExchange

        def _mk_handler():
            # fresh receive pipeline per peer connection (parallel applies)
            ing = Ingestor(eng, default_regions=4, append_mode=True,
                           durable=(args.wal != "off"), rank=rank, world=world)
            return ing.handle_remote

        exchange = WriteExchange(rank, world, handler_factory=_mk_handler)
        import torch.distributed as dist
        dist.barrier()  # every rank's exchange is listening

    # -------- pre-generate the line pool (untimed; replayed cyclically) -----
    # Each worker thread owns a disjoint host shard with its own Ingestor /
    # C++ parser (like the reference's 6 TSBS client workers; P5 axis).
    n_workers = max(args.workers, 1)
    gen_t0 = time.perf_counter()
    pool_batches_per_worker = max(pool_rows // (n_workers * ROWS_PER_BATCH), 1)
    workers = []

    def gen_worker(wi):
        w = CpuWorkload(scale=args.scale, seed=7 + rank * 100 + wi)
        w.tagsets = [t.replace(b"host_", b"host_%d_%d_" % (rank, wi))
                     for t in w.tagsets]
        batches = [w.next_batch(ROWS_PER_BATCH)
                   for _ in range(pool_batches_per_worker)]
        ing = Ingestor(eng, default_regions=4, append_mode=True,
                       durable=(args.wal != "off"),
                       rank=rank, world=world, exchange=exchange)
        workers.append((ing, batches))

    gen_threads = [threading.Thread(target=gen_worker, args=(wi,))
                   for wi in range(n_workers)]
    for t in gen_threads:
        t.start()
    for t in gen_threads:
        t.join()
    pool_total = n_workers * pool_batches_per_worker * ROWS_PER_BATCH
    batches_per_step_w = max(step_rows // (n_workers * ROWS_PER_BATCH), 1)
    rows_per_step = n_workers * batches_per_step_w * ROWS_PER_BATCH
    log(f"# line pool: {pool_total} rows in "
        f"{time.perf_counter() - gen_t0:.1f}s ({n_workers} workers); "
        f"{rows_per_step} rows/step")

    def barrier_sync():
        if dist_on:
            import torch.distributed as dist
            dist.barrier()
        if have_gpu:
            torch.cuda.synchronize()

    def run_steps(lo, hi):
        def worker_fn(ing, batches):
            k = lo * batches_per_step_w
            npool = len(batches)
            for _ in range(lo, hi):
                for _ in range(batches_per_step_w):
                    ing.ingest_lines(batches[k % npool])
                    k += 1
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
    run_steps(args.warmup, args.warmup + args.steps)
    barrier_sync()
    elapsed = time.perf_counter() - t0

    # max over ranks
    if dist_on:
        import torch.distributed as dist
        t = torch.tensor([elapsed], dtype=torch.float64,
                         device=device if dist.get_backend() == "nccl" else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    total_rows = args.steps * rows_per_step * world
    rows_per_s = total_rows / elapsed
    ms_per_step = elapsed / args.steps * 1000
    rows_shipped = sum(wk[0].rows_shipped for wk in workers)
    ingest_rows_stored = sum(r.num_rows for st in eng.tables.values()
                             for r in st.regions)
    n_ssts = sum(len(r.manifest.files) for st in eng.tables.values()
                 for r in st.regions)
    log(f"# ingest: {total_rows} rows in {elapsed:.1f}s = "
        f"{rows_per_s:,.0f} rows/s (stored {ingest_rows_stored}, "
        f"{n_ssts} SSTs live)")
    # drain background flush/compaction before tearing the engine down so
    # rmtree doesn't race the flusher thread
    drain0 = time.perf_counter()
    while not eng._flush_q.empty() and time.perf_counter() - drain0 < 120:
        time.sleep(0.2)
    if exchange is not None:
        import torch.distributed as dist
        dist.barrier()  # all remote applies acked everywhere
        exchange.close()
    eng.close()
    shutil.rmtree(data_dir, ignore_errors=True)

    # -------- query suite on the reference benchmark config (scale=4000) ----
    queries = {}
    q_meta = {}
    if not args.skip_queries:
        from greptimedb_amd.models.tsbs_fixture import load_cpu_fixture, tsbs_queries
        from greptimedb_amd.parallel.dist import DistContext
        qdir = os.path.join(base, f"qrank{rank}")
        qeng = MitoEngine(EngineConfig(data_dir=qdir, device=device,
                                       background_flush=False))
        ld0 = time.perf_counter()
        local_rows = load_cpu_fixture(qeng, scale=q_scale, hours=q_hours,
                                      rank=rank, world=world)
        if have_gpu:
            torch.cuda.synchronize()
        load_s = time.perf_counter() - ld0
        total_fixture = local_rows
        if dist_on:
            import torch.distributed as dist
            tr = torch.tensor([local_rows], dtype=torch.float64,
                              device=device if dist.get_backend() == "nccl" else "cpu")
            dist.all_reduce(tr)
            total_fixture = int(tr.item())
        log(f"# query fixture: scale={q_scale} hours={q_hours} "
            f"rows={total_fixture} loaded in {load_s:.1f}s")
        ex = Executor(qeng, dist=DistContext(device=device) if dist_on else None)
        for name, q in tsbs_queries(q_scale, q_hours).items():
            times = []
            for _ in range(args.query_reps):
                barrier_sync()
                qt0 = time.perf_counter()
                ex.execute(q)
                if have_gpu:
                    torch.cuda.synchronize()
                times.append((time.perf_counter() - qt0) * 1000)
            queries[name + "_p50_ms"] = round(float(np.median(times)), 3)
            log(f"#   {name}: p50 {queries[name + '_p50_ms']} ms")
        q_meta = {"query_fixture_scale": q_scale, "query_fixture_hours": q_hours,
                  "query_fixture_rows": total_fixture, "query_reps": args.query_reps}
        qeng.close()

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
        "data": "synthetic TSBS cpu-only (influx line protocol, random-walk "
                "values; pre-generated pool replayed, TSBS tsbs_load style)",
        "config": {
            "model": "tsbs-devops-cpu-only",
            "global_batch": ROWS_PER_BATCH,
            "seq_len": 0,
            "parallelism": f"region-shard dp{world}",
            "scale_per_worker": args.scale,
            "workers": n_workers,
            "rows_per_step": rows_per_step,
            "pool_rows": pool_total,
            "rows_shipped_cross_rank": rows_shipped,
            "ingest_elapsed_s": round(elapsed, 2),
            "wal": {"buffered": "group-commit, no fsync",
                    "fsync": "group-commit + fdatasync",
                    "off": "disabled"}[args.wal],
            "wal_shards": args.wal_shards,
            "flush_mb": args.flush_mb,
            "ssts_written": n_ssts,
            "queries": queries,
            **q_meta,
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
