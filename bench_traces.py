#!/usr/bin/env python3
"""OTLP trace ingest benchmark (BASELINE config 5 shape).

Pre-generates OTLP protobuf ExportTraceServiceRequest batches (synthetic
spans, 20 services × 50 operations) untimed, then times the full ingest
path: native protobuf decode → routing → WAL → GPU memtable, with
background flush (HBM → parquet spill) enabled.

  python bench_traces.py --spans-per-batch 10000 --steps 10
"""

from __future__ import annotations

import argparse
import json
import os
import struct
import tempfile
import time

import numpy as np
import torch


def _v(x):
    out = b""
    while True:
        b7 = x & 0x7F
        x >>= 7
        out += bytes([b7 | (0x80 if x else 0)])
        if not x:
            return out


def _ld(f, payload):
    return _v((f << 3) | 2) + _v(len(payload)) + payload


def _s(f, s):
    return _ld(f, s.encode() if isinstance(s, str) else s)


def _fixed64(f, x):
    return _v((f << 3) | 1) + struct.pack("<Q", x)


def gen_batch(rng, n_spans, services, ops, base_ns):
    by_service: dict[int, list[bytes]] = {}
    tid = rng.bytes(16 * n_spans)
    sid = rng.bytes(8 * n_spans)
    svc_idx = rng.randint(0, len(services), n_spans)
    op_idx = rng.randint(0, len(ops), n_spans)
    start = base_ns + rng.randint(0, 10_000_000_000, n_spans)
    dur = rng.randint(100_000, 500_000_000, n_spans)
    for i in range(n_spans):
        body = (_s(1, tid[16 * i:16 * i + 16]) + _s(2, sid[8 * i:8 * i + 8]) +
                _s(5, ops[op_idx[i]]) +
                _fixed64(7, int(start[i])) + _fixed64(8, int(start[i] + dur[i])))
        by_service.setdefault(svc_idx[i], []).append(_ld(2, body))
    req = b""
    for s_i, spans in by_service.items():
        resource = _ld(1, _ld(1, _s(1, "service.name") +
                              _ld(2, _s(1, services[s_i]))))
        req += _ld(1, resource + _ld(2, b"".join(spans)))
    return req


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--spans-per-batch", type=int, default=10_000)
    ap.add_argument("--batches-per-step", type=int, default=10)
    ap.add_argument("--steps", type=int, default=10)
    ap.add_argument("--warmup", type=int, default=2)
    args = ap.parse_args()

    have_gpu = torch.cuda.is_available()
    device = "cuda:0" if have_gpu else "cpu"

    from greptimedb_amd.engine.engine import EngineConfig, MitoEngine
    from greptimedb_amd.engine.tracestore import TraceStore

    base = tempfile.mkdtemp(prefix="gdb_traces_")
    eng = MitoEngine(EngineConfig(data_dir=base, device=device,
                                  background_flush=True, flush_bytes=1 << 29))
    store = TraceStore(eng)

    rng = np.random.RandomState(3)
    services = [f"svc_{i}" for i in range(20)]
    ops = [f"op_{i}" for i in range(50)]
    total_steps = args.warmup + args.steps
    t0 = time.perf_counter()
    step_batches = [
        [gen_batch(rng, args.spans_per_batch, services, ops,
                   1_451_606_400_000_000_000 + s * 10_000_000_000)
         for _ in range(args.batches_per_step)]
        for s in range(total_steps)
    ]
    print(f"# generated {total_steps * args.batches_per_step * args.spans_per_batch} "
          f"spans in {time.perf_counter()-t0:.1f}s", flush=True)

    for i in range(args.warmup):
        for b in step_batches[i]:
            store.write(b)
    if have_gpu:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for i in range(args.warmup, total_steps):
        for b in step_batches[i]:
            store.write(b)
    if have_gpu:
        torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    spans = args.steps * args.batches_per_step * args.spans_per_batch
    rate = spans / dt
    print(json.dumps({
        "bench": "otlp-trace-ingest", "spans_per_s": round(rate, 1),
        "spans": spans, "elapsed_s": round(dt, 2), "device": device,
        "n_gpus": 1,
    }), flush=True)
    eng.close()


if __name__ == "__main__":
    main()
