#!/usr/bin/env python3
"""Measure region reopen with vs without persisted fulltext sidecars.

VERDICT r1 #4 done-criterion: reopen of a large log fixture in seconds,
postings loaded from the per-SST sidecar instead of re-tokenizing.

  python scripts/bench_ftindex_reopen.py --events 10000000
"""

import argparse
import glob
import json
import os
import sys
import tempfile
import time

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--events", type=int, default=10_000_000)
    ap.add_argument("--chunk", type=int, default=500_000)
    args = ap.parse_args()

    import torch
    from greptimedb_amd.engine.engine import EngineConfig, MitoEngine
    from greptimedb_amd.query.executor import Executor
    from bench_logs import make_templates

    device = "cuda:0" if torch.cuda.is_available() else "cpu"
    base = tempfile.mkdtemp(prefix="gdb_ftidx_")
    eng = MitoEngine(EngineConfig(data_dir=base, device=device,
                                  background_flush=False, default_regions=2))
    ex = Executor(eng)
    ex.execute("CREATE TABLE applogs (service STRING, ts TIMESTAMP TIME INDEX,"
               " latency DOUBLE, message STRING FULLTEXT INDEX,"
               " PRIMARY KEY (service)) WITH ('append_mode'='true')")
    rng = np.random.RandomState(7)
    templates = make_templates(10_000, rng)
    templates[0] = "xenon isotope anomaly detected in reactor four"
    st = eng.table("applogs")
    gen0 = time.perf_counter()
    written = 0
    import pyarrow as pa
    from greptimedb_amd.engine.bulk import bulk_insert_arrow
    while written < args.events:
        n = min(args.chunk, args.events - written)
        tidx = rng.zipf(1.3, size=n) % len(templates)
        msgs = np.char.add(templates[tidx].astype(str),
                           rng.randint(0, 1000, size=n).astype(str)).tolist()
        batch = pa.table({
            "service": pa.array([f"svc_{i % 32}" for i in range(n)]),
            "ts": pa.array(np.arange(written, written + n, dtype=np.int64) * 10
                           + 1451606400000, type=pa.int64()).cast(pa.timestamp("ms")),
            "latency": pa.array(rng.exponential(10.0, size=n)),
            "message": pa.array(msgs, type=pa.string()),
        })
        bulk_insert_arrow(eng, "applogs", batch, durable=False)
        written += n
        eng.flush_all()   # one SST (+sidecar) per chunk per region
    gen_s = time.perf_counter() - gen0
    n_sidecars = len(glob.glob(f"{base}/region/*/sst/*.ftidx"))
    eng.close()

    def reopen_time():
        t0 = time.perf_counter()
        e2 = MitoEngine(EngineConfig(data_dir=base, device=device,
                                     background_flush=False))
        if device.startswith("cuda"):
            torch.cuda.synchronize()
        dt = time.perf_counter() - t0
        r = Executor(e2).execute(
            "SELECT count(*) FROM applogs WHERE matches(message, 'xenon reactor')")
        hits = int(r.columns[0][0])
        e2.close()
        return dt, hits

    reopen_time()   # warm-up (imports, page cache) — not measured
    with_s, hits1 = reopen_time()
    sidecars = glob.glob(f"{base}/region/*/sst/*.ftidx")
    saved = {f: open(f, "rb").read() for f in sidecars}
    for f in sidecars:
        os.unlink(f)
    without_s, hits2 = reopen_time()
    for f, blob in saved.items():
        open(f, "wb").write(blob)
    assert hits1 == hits2, (hits1, hits2)
    print(json.dumps({
        "bench": "ftindex-sidecar-reopen", "events": written,
        "device": device, "ingest_flush_s": round(gen_s, 1),
        "sidecars": n_sidecars,
        "reopen_with_sidecar_s": round(with_s, 2),
        "reopen_rebuild_s": round(without_s, 2),
        "speedup": round(without_s / max(with_s, 1e-9), 1),
        "xenon_hits": hits1,
    }))
    import shutil
    shutil.rmtree(base, ignore_errors=True)


if __name__ == "__main__":
    main()
