"""IVF-flat ANN index for VECTOR columns.

Reference parity: src/index vector index (the reference builds usearch
HNSW files per SST). MI355X redesign: graph traversal is pointer-chasing —
hostile to a 64-wide-wavefront machine — so the index is IVF-flat instead:
a k-means codebook partitions the vectors, a probe touches nprobe
clusters, and the candidate scan inside each cluster is dense GEMM-shaped
work on HBM3E, exactly what the hardware is fastest at. The lists are
stored as a cluster-sorted row permutation + CSR offsets, so a probe is a
contiguous gather. Exactness: sources without an index (e.g. the live
memtable) stay brute-force, so freshness never loses rows; only indexed
SST sources trade recall for speed.
"""

from __future__ import annotations

import torch


@torch.no_grad()
def kmeans(x: torch.Tensor, k: int, iters: int = 10,
           seed: int = 0) -> torch.Tensor:
    """Plain Lloyd k-means on device → [k, D] centroids (f32)."""
    n = x.shape[0]
    k = min(k, n)
    g = torch.Generator(device="cpu").manual_seed(seed)
    init = torch.randperm(n, generator=g)[:k].to(x.device)
    c = x[init].clone()
    for _ in range(iters):
        # chunk the assignment so [n, k] distance blocks stay in HBM budget
        assign = torch.empty(n, dtype=torch.int64, device=x.device)
        step = max(1, (64 << 20) // max(k * 4, 1))
        for s in range(0, n, step):
            d = torch.cdist(x[s:s + step], c)
            assign[s:s + step] = d.argmin(dim=1)
        sums = torch.zeros_like(c)
        sums.index_add_(0, assign, x)
        cnt = torch.bincount(assign, minlength=k).clamp_min(1)
        c = sums / cnt[:, None].float()
    return c


@torch.no_grad()
def build_ivf(vt: torch.Tensor, nlist: int, iters: int = 10):
    """→ dict(centroids [nlist, D], perm [n] cluster-sorted row ids,
    offsets [nlist+1] CSR bounds). vt is the [n, D] f32 source tensor
    (NaN rows allowed — they land in a cluster but score NaN→inf later)."""
    x = torch.nan_to_num(vt, nan=0.0).float()
    c = kmeans(x, nlist, iters)
    nlist = c.shape[0]
    assign = torch.empty(x.shape[0], dtype=torch.int64, device=x.device)
    step = max(1, (64 << 20) // max(nlist * 4, 1))
    for s in range(0, x.shape[0], step):
        assign[s:s + step] = torch.cdist(x[s:s + step], c).argmin(dim=1)
    perm = torch.argsort(assign, stable=True)
    counts = torch.bincount(assign, minlength=nlist)
    offsets = torch.zeros(nlist + 1, dtype=torch.int64, device=x.device)
    offsets[1:] = torch.cumsum(counts, 0)
    return {"centroids": c, "perm": perm, "offsets": offsets}


@torch.no_grad()
def ivf_candidates(ivf: dict, q: torch.Tensor, nprobe: int) -> torch.Tensor:
    """Row ids (into the original [n, D] tensor) of the nprobe nearest
    clusters — a contiguous CSR gather per probed list."""
    c = ivf["centroids"]
    nprobe = min(nprobe, c.shape[0])
    dc = ((c - q.float()[None, :]) ** 2).sum(dim=1)
    lists = torch.topk(dc, nprobe, largest=False).indices
    off = ivf["offsets"]
    segs = [ivf["perm"][off[li]: off[li + 1]] for li in lists.tolist()]
    if not segs:
        return torch.zeros(0, dtype=torch.int64, device=q.device)
    return torch.cat(segs)
