"""Distributed query/ingest context: RCCL (or gloo) collectives.

Reference parity: the scatter-gather in src/query/src/dist_plan
(MergeScanExec fans out per-region streams, partial aggregates pushed down
via commutativity analysis and combined at the frontend). MI355X redesign:
one process per GPU (torch.distributed, backend "nccl" == RCCL over xGMI);
each rank owns a shard of every table's regions; partial aggregates are
combined with all-reduce (sum/count: SUM, min: MIN, max: MAX) after group
keys are unified — small tensors, so a single fused all-reduce per query
beats streaming Arrow batches through a frontend by orders of magnitude.

Works on CPU with the gloo backend (multi-process tests, no GPU needed).
"""

from __future__ import annotations

import numpy as np
import torch
import torch.distributed as dist


class DistContext:
    def __init__(self, device: str = "cpu"):
        assert dist.is_initialized(), "torch.distributed must be initialized"
        self.rank = dist.get_rank()
        self.world = dist.get_world_size()
        self.device = device
        # collectives run on the compute device for nccl, cpu for gloo
        self.coll_device = device if dist.get_backend() == "nccl" else "cpu"

    # ------------------------------------------------------------ scalars

    def minmax_ts(self, lo: int | None, hi: int | None):
        t = torch.tensor([lo if lo is not None else (1 << 62),
                          -(hi if hi is not None else -(1 << 62))],
                         dtype=torch.int64, device=self.coll_device)
        dist.all_reduce(t, op=dist.ReduceOp.MIN)
        lo2 = int(t[0].item())
        hi2 = -int(t[1].item())
        if lo2 == (1 << 62):
            return None, None
        return lo2, hi2

    def all_sum(self, value: float) -> float:
        t = torch.tensor([value], dtype=torch.float64, device=self.coll_device)
        dist.all_reduce(t, op=dist.ReduceOp.SUM)
        return float(t.item())

    def all_max(self, value: float) -> float:
        t = torch.tensor([value], dtype=torch.float64, device=self.coll_device)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        return float(t.item())

    def barrier(self):
        dist.barrier()

    # ------------------------------------------------------------ groups

    def merge_groups(self, group_keys: dict, sums, cnts, mins, maxs, rowcnt):
        """Unify group keys across ranks and all-reduce the partial
        aggregate planes. Arrays are numpy [nf, n_slots, n_buckets] (+rowcnt
        [n_slots, n_buckets]); returns (global_keys_dict, arrays)."""
        all_keys: list = [None] * self.world
        dist.all_gather_object(all_keys, list(group_keys.keys()))
        merged = sorted({k for ks in all_keys for k in ks})
        gmap = {k: i for i, k in enumerate(merged)}
        ng = max(len(merged), 1)
        nf, _, nb = sums.shape

        g_sums = np.zeros((nf, ng, nb))
        g_cnts = np.zeros((nf, ng, nb), dtype=np.int64)
        g_mins = np.full((nf, ng, nb), np.inf)
        g_maxs = np.full((nf, ng, nb), -np.inf)
        g_rows = np.zeros((ng, nb), dtype=np.int64)
        if group_keys:
            local_slots = np.array([gmap[k] for k in group_keys.keys()])
            src_slots = np.array(list(group_keys.values()))
            g_sums[:, local_slots] = sums[:, src_slots]
            g_cnts[:, local_slots] = cnts[:, src_slots]
            g_mins[:, local_slots] = np.nan_to_num(mins[:, src_slots], nan=np.inf)
            g_maxs[:, local_slots] = np.nan_to_num(maxs[:, src_slots], nan=-np.inf)
            g_rows[local_slots] = rowcnt[src_slots]

        dev = self.coll_device
        t_sum = torch.as_tensor(g_sums, device=dev)
        t_cnt = torch.as_tensor(g_cnts, device=dev)
        t_min = torch.as_tensor(g_mins, device=dev)
        t_max = torch.as_tensor(g_maxs, device=dev)
        t_row = torch.as_tensor(g_rows, device=dev)
        dist.all_reduce(t_sum, op=dist.ReduceOp.SUM)
        dist.all_reduce(t_cnt, op=dist.ReduceOp.SUM)
        dist.all_reduce(t_min, op=dist.ReduceOp.MIN)
        dist.all_reduce(t_max, op=dist.ReduceOp.MAX)
        dist.all_reduce(t_row, op=dist.ReduceOp.SUM)
        g_sums = t_sum.cpu().numpy()
        g_cnts = t_cnt.cpu().numpy()
        g_mins = t_min.cpu().numpy()
        g_maxs = t_max.cpu().numpy()
        g_rows = t_row.cpu().numpy()
        # restore NaN for empty cells
        g_mins[g_cnts == 0] = np.nan
        g_maxs[g_cnts == 0] = np.nan
        return gmap, (g_sums, g_cnts, g_mins, g_maxs, g_rows)

    def merge_lastpoint(self, group_keys: dict, best_ts, best_val):
        """Combine per-rank lastpoint partials: per group, the row with the
        globally newest ts wins (ties: max value)."""
        all_keys: list = [None] * self.world
        dist.all_gather_object(all_keys, list(group_keys.keys()))
        merged = sorted({k for ks in all_keys for k in ks})
        gmap = {k: i for i, k in enumerate(merged)}
        ng = max(len(merged), 1)
        nf = best_val.shape[0]
        g_ts = np.full(ng, -(1 << 62), dtype=np.int64)
        g_val = np.full((nf, ng), -np.inf)
        if group_keys:
            l = np.array([gmap[k] for k in group_keys.keys()])
            s = np.array(list(group_keys.values()))
            g_ts[l] = best_ts[s]
            g_val[:, l] = np.nan_to_num(best_val[:, s], nan=-np.inf)
        dev = self.coll_device
        t_ts = torch.as_tensor(g_ts, device=dev)
        dist.all_reduce(t_ts, op=dist.ReduceOp.MAX)
        g_ts_glob = t_ts.cpu().numpy()
        mine = g_ts == g_ts_glob
        g_val[:, ~mine] = -np.inf
        t_val = torch.as_tensor(g_val, device=dev)
        dist.all_reduce(t_val, op=dist.ReduceOp.MAX)
        g_val = t_val.cpu().numpy()
        g_val[g_val == -np.inf] = np.nan
        return gmap, g_ts_glob, g_val

    def merge_prom_planes(self, keys: list, cnt, s, sq, mn, mx):
        """Merge PromQL aggregation partial planes [G, T] across ranks.
        keys are hashable label tuples; cnt/s/sq all-reduce SUM, mn MIN,
        mx MAX. Returns (global_keys, cnt, s, sq, mn, mx)."""
        all_keys: list = [None] * self.world
        dist.all_gather_object(all_keys, keys)
        merged = sorted({k for ks in all_keys for k in ks})
        gmap = {k: i for i, k in enumerate(merged)}
        G = max(len(merged), 1)
        T = cnt.shape[1]
        dev = self.coll_device

        def scatter(t, fill=0.0):
            if t is None:
                return None
            out = torch.full((G, T), fill, dtype=torch.float64, device=dev)
            if keys:
                idx = torch.as_tensor([gmap[k] for k in keys], device=dev)
                out[idx] = t.to(dev)
            return out

        cnt2 = scatter(cnt)
        s2 = scatter(s)
        sq2 = scatter(sq)
        mn2 = scatter(mn, float("inf"))
        mx2 = scatter(mx, float("-inf"))
        dist.all_reduce(cnt2, op=dist.ReduceOp.SUM)
        dist.all_reduce(s2, op=dist.ReduceOp.SUM)
        if sq2 is not None:
            dist.all_reduce(sq2, op=dist.ReduceOp.SUM)
        if mn2 is not None:
            dist.all_reduce(mn2, op=dist.ReduceOp.MIN)
        if mx2 is not None:
            dist.all_reduce(mx2, op=dist.ReduceOp.MAX)
        return merged, cnt2, s2, sq2, mn2, mx2

    def merge_planes(self, keys: list, planes: list):
        """Generic partial-plane merge (RANGE queries): `planes` is a list
        of ([G, T] tensor, op) with op ∈ {sum, min, max}. Group keys are
        unified across ranks (all_gather_object), each plane scattered into
        the global slot order and all-reduced. Returns (merged_keys, outs)."""
        all_keys: list = [None] * self.world
        dist.all_gather_object(all_keys, keys)
        merged = sorted({k for ks in all_keys for k in ks})
        gmap = {k: i for i, k in enumerate(merged)}
        G = max(len(merged), 1)
        T = next((p.shape[1] for p, _o in planes if p is not None), 1)
        dev = self.coll_device
        fills = {"sum": 0.0, "min": float("inf"), "max": float("-inf")}
        ops = {"sum": dist.ReduceOp.SUM, "min": dist.ReduceOp.MIN,
               "max": dist.ReduceOp.MAX}
        outs = []
        idx = torch.as_tensor([gmap[k] for k in keys], device=dev) if keys \
            else None
        for p, op in planes:
            out = torch.full((G, T), fills[op], dtype=torch.float64, device=dev)
            if p is not None and idx is not None:
                out[idx] = p.to(dev)
            dist.all_reduce(out, op=ops[op])
            outs.append(out)
        return merged, outs

    def gather_matrix(self, labels: list, values):
        """Gather per-rank series matrices (labels + [S, T]) on all ranks."""
        payload = (labels, values.cpu().numpy())
        all_p: list = [None] * self.world
        dist.all_gather_object(all_p, payload)
        out_labels = []
        mats = []
        for ls, m in all_p:
            out_labels.extend(ls)
            mats.append(m)
        import numpy as _np
        vals = _np.concatenate(mats) if mats else _np.zeros((0, values.shape[1]))
        return out_labels, torch.as_tensor(vals)

    # ------------------------------------------------------------ raw rows

    def gather_columns(self, col_data: dict[str, np.ndarray]) -> dict[str, np.ndarray]:
        """Concatenate per-rank row columns (raw scans) on every rank."""
        gathered: list = [None] * self.world
        dist.all_gather_object(gathered, {k: np.asarray(v) for k, v in col_data.items()})
        keys = col_data.keys()
        return {k: np.concatenate([g[k] for g in gathered]) for k in keys}


    def argmax_combine(self, keys: list, order_plane, value_plane):
        """Cross-rank 'newest wins' merge (RANGE last_value): the value
        whose order key (sample ts) is globally maximal is kept per cell.
        Two reductions: MAX on the order plane, then MAX over values masked
        to cells where the local order equals the global winner (a series
        lives on one rank, so ties are degenerate)."""
        all_keys: list = [None] * self.world
        dist.all_gather_object(all_keys, keys)
        merged = sorted({k for ks in all_keys for k in ks})
        gmap = {k: i for i, k in enumerate(merged)}
        G = max(len(merged), 1)
        T = order_plane.shape[1] if order_plane is not None and \
            order_plane.ndim == 2 else 1
        dev = self.coll_device
        neg = float("-inf")
        g_ord = torch.full((G, T), neg, dtype=torch.float64, device=dev)
        l_ord = torch.full((G, T), neg, dtype=torch.float64, device=dev)
        l_val = torch.full((G, T), neg, dtype=torch.float64, device=dev)
        if keys:
            idx = torch.as_tensor([gmap[k] for k in keys], device=dev)
            l_ord[idx] = torch.nan_to_num(order_plane.to(dev), nan=neg)
            l_val[idx] = torch.nan_to_num(value_plane.to(dev), nan=neg)
        g_ord.copy_(l_ord)
        dist.all_reduce(g_ord, op=dist.ReduceOp.MAX)
        win = (l_ord == g_ord) & torch.isfinite(g_ord)
        cand = torch.where(win, l_val, torch.full_like(l_val, neg))
        dist.all_reduce(cand, op=dist.ReduceOp.MAX)
        out = torch.where(torch.isfinite(g_ord), cand,
                          torch.full_like(cand, float("nan")))
        return merged, out
