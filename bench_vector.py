#!/usr/bin/env python3
"""Vector-search benchmark: brute-force GPU kNN vs IVF-flat probe.

Measures the device kernel path (build + query) at configurable scale —
N vectors × D dims, top-k by L2. The SQL surface (`vec_l2sq_distance …
ORDER BY d LIMIT k`, ADMIN build_vector_index) drives the same tensors;
this bench isolates the search cost from SQL overhead.
"""

from __future__ import annotations

import argparse
import json
import time

import numpy as np
import torch


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--n", type=int, default=1_000_000)
    ap.add_argument("--dim", type=int, default=128)
    ap.add_argument("--k", type=int, default=10)
    ap.add_argument("--nlist", type=int, default=0, help="0 = 4*sqrt(n)")
    ap.add_argument("--nprobe", type=int, default=32)
    ap.add_argument("--queries", type=int, default=100)
    args = ap.parse_args()

    from greptimedb_amd.vector import build_ivf, ivf_candidates

    device = "cuda:0" if torch.cuda.is_available() else "cpu"
    g = torch.Generator().manual_seed(7)
    # clustered data (mixture of gaussians) — embeddings cluster; uniform
    # gaussian noise would make every IVF list equidistant and recall
    # meaningless as a proxy for real workloads
    n_centers = max(64, args.n // 1000)
    centers = torch.randn(n_centers, args.dim, generator=g)
    assign = torch.randint(0, n_centers, (args.n,), generator=g)
    x = (centers[assign] + 0.25 * torch.randn(args.n, args.dim, generator=g)
         ).to(device)
    qi_rows = torch.randint(0, args.n, (args.queries,), generator=g)
    qs = (x[qi_rows.to(device)] +
          0.1 * torch.randn(args.queries, args.dim, generator=g).to(device))
    nlist = args.nlist or max(8, int(4 * args.n ** 0.5))

    def sync():
        if device.startswith("cuda"):
            torch.cuda.synchronize()

    # brute force, per query (the latency a single SQL kNN pays)
    sync()
    t0 = time.perf_counter()
    exact_ids = []
    for qi in range(args.queries):
        d = ((x - qs[qi][None, :]) ** 2).sum(dim=1)
        exact_ids.append(torch.topk(d, args.k, largest=False).indices.cpu())
    sync()
    brute_ms = (time.perf_counter() - t0) * 1000 / args.queries

    # brute force, batched (throughput mode: one GEMM for all queries)
    sync()
    t0 = time.perf_counter()
    d_all = torch.cdist(qs, x)               # [Q, N]
    torch.topk(d_all, args.k, dim=1, largest=False)
    sync()
    brute_batch_ms = (time.perf_counter() - t0) * 1000 / args.queries
    del d_all

    sync()
    t0 = time.perf_counter()
    ivf = build_ivf(x, nlist)
    sync()
    build_s = time.perf_counter() - t0

    sync()
    t0 = time.perf_counter()
    hits = 0
    for qi in range(args.queries):
        rows = ivf_candidates(ivf, qs[qi], args.nprobe)
        d = ((x[rows] - qs[qi][None, :]) ** 2).sum(dim=1)
        kk = min(args.k, d.numel())
        ids = rows[torch.topk(d, kk, largest=False).indices].cpu()
        hits += len(set(ids.tolist()) & set(exact_ids[qi].tolist()))
    sync()
    ivf_ms = (time.perf_counter() - t0) * 1000 / args.queries
    recall = hits / (args.queries * args.k)

    print(json.dumps({
        "bench": "vector-knn",
        "n": args.n, "dim": args.dim, "k": args.k,
        "nlist": nlist, "nprobe": args.nprobe,
        "brute_ms_per_query": round(brute_ms, 3),
        "brute_batched_ms_per_query": round(brute_batch_ms, 3),
        "ivf_ms_per_query": round(ivf_ms, 3),
        "ivf_build_s": round(build_s, 2),
        "speedup": round(brute_ms / ivf_ms, 1),
        "recall_at_k": round(recall, 4),
        "device": device,
    }), flush=True)


if __name__ == "__main__":
    main()
