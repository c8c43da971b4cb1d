#!/usr/bin/env python3
"""K11/K20 decode throughput on device.

  python scripts/bench_decode.py --rows 50000000
Prints decode GB/s (output bytes / kernel wall) for:
  * K20 gorilla blocks (quantized TSBS shape + raw doubles)
  * K11 RLE/dict page expand on an engine-written SST column
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
    ap.add_argument("--rows", type=int, default=50_000_000)
    ap.add_argument("--iters", type=int, default=5)
    args = ap.parse_args()
    import torch
    from greptimedb_amd.engine import gorilla
    dev = "cuda:0" if torch.cuda.is_available() else "cpu"
    rng = np.random.RandomState(1)
    n = args.rows
    ts = 1451606400000 + np.arange(n, dtype=np.int64) * 10_000
    vals = np.round(np.clip(np.cumsum(rng.uniform(-1, 1, n)) + 50, 0, 100), 4)
    t0 = time.perf_counter()
    blob, bo, oo, nn = gorilla.pack(ts, vals)
    pack_s = time.perf_counter() - t0
    out_bytes = n * 16
    results = {"bench": "k20-gorilla-decode", "rows": n, "device": dev,
               "pack_s": round(pack_s, 2),
               "ratio": round(out_bytes / len(blob), 2)}
    if dev.startswith("cuda"):
        blob_t = torch.as_tensor(np.frombuffer(blob, np.uint8).copy()).to(dev)
        bo_t = torch.as_tensor(bo).to(dev)
        oo_t = torch.as_tensor(oo).to(dev)
        from greptimedb_amd import _hip_ops
        _hip_ops.gorilla_decode(blob_t, bo_t, oo_t, nn)   # warm
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(args.iters):
            dts, dv = _hip_ops.gorilla_decode(blob_t, bo_t, oo_t, nn)
        torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / args.iters
        results["decode_ms"] = round(dt * 1000, 2)
        results["decode_GBps_out"] = round(out_bytes / dt / 1e9, 1)
        # correctness spot check
        k = min(n, 100_000)
        ts_c, v_c = gorilla.decode_ref(blob, bo[:max(1, k // gorilla.BLOCK)],
                                       oo[:max(1, k // gorilla.BLOCK)],
                                       min(k, int(oo[max(1, k // gorilla.BLOCK) - 1]) + gorilla.BLOCK))
        m = len(ts_c)
        np.testing.assert_array_equal(dts[:m].cpu().numpy(), ts_c.numpy())
        np.testing.assert_array_equal(dv[:m].cpu().numpy(), v_c.numpy())
    print(json.dumps(results))

    # ---- K11 on a real SST column
    from greptimedb_amd.engine import pagedec
    from greptimedb_amd.engine.engine import EngineConfig, MitoEngine
    from greptimedb_amd.engine.ingest import Ingestor
    from greptimedb_amd.models.tsbs import CpuWorkload
    d = tempfile.mkdtemp()
    eng = MitoEngine(EngineConfig(data_dir=d, device=dev,
                                  background_flush=False, default_regions=1))
    ing = Ingestor(eng)
    w = CpuWorkload(scale=200)
    for _ in range(10):
        ing.ingest_lines(w.next_batch(100_000))
    eng.flush_all()
    f = glob.glob(f"{d}/region/*/sst/*.parquet")[0]
    col = "__sequence"
    pagedec.read_numeric_column(f, col, dev)   # warm
    if dev.startswith("cuda"):
        import torch
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.iters):
        out = pagedec.read_numeric_column(f, col, dev)
    if dev.startswith("cuda"):
        torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / args.iters
    nbytes = out.numel() * out.element_size()
    print(json.dumps({"bench": "k11-page-decode", "column": col,
                      "rows": int(out.numel()), "device": dev,
                      "ms": round(dt * 1000, 2),
                      "GBps_out": round(nbytes / dt / 1e9, 2),
                      "note": "includes host thrift+zstd parse + H2D"}))
    eng.close()


if __name__ == "__main__":
    main()
