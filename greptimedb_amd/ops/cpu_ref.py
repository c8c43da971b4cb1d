"""CPU reference implementations of the HIP kernels (plain PyTorch, fp64).

These define the semantics the gfx950 kernels in csrc/kernels.hip must match;
GPU numerics tests (tests/test_ops_gpu.py) compare the two. They also serve
as the execution path on CPU-only hosts (tests, CI).
"""

from __future__ import annotations

import torch


def ts_bucket_agg(ts, series, fields, field_idx, slot_lut, ts_lo, ts_hi,
                  origin, bucket_ms, n_slots, n_buckets):
    """Fused filter + time-bucket aggregate.

    ts: i64[n] ms, series: i32[n], fields: f64[nf_total, >=n],
    field_idx: i32[nf], slot_lut: i32[lut]. Returns (sum, count, min, max,
    rows) — the first four [nf, n_slots, n_buckets] (min/max NaN where
    count==0), rows [n_slots, n_buckets] = matching row count per cell
    (for count(*)/group existence). NaN field values are nulls, skipped.
    """
    n = ts.numel()
    keep = (ts >= ts_lo) & (ts < ts_hi)
    s = series.long().clamp(0, max(slot_lut.numel() - 1, 0))
    in_lut = (series >= 0) & (series < slot_lut.numel())
    slot = torch.where(in_lut, slot_lut[s].long(), torch.full_like(s, -1))
    keep &= slot >= 0
    bucket = torch.div(ts - origin, bucket_ms, rounding_mode="floor")
    keep &= (bucket >= 0) & (bucket < n_buckets)

    nf = field_idx.numel()
    out_sum = torch.zeros(nf, n_slots, n_buckets, dtype=torch.float64)
    out_cnt = torch.zeros(nf, n_slots, n_buckets, dtype=torch.int64)
    out_min = torch.full((nf, n_slots, n_buckets), float("nan"), dtype=torch.float64)
    out_max = torch.full((nf, n_slots, n_buckets), float("nan"), dtype=torch.float64)
    out_rows = torch.zeros(n_slots, n_buckets, dtype=torch.int64)

    idx = keep.nonzero(as_tuple=True)[0]
    if idx.numel() == 0:
        return out_sum, out_cnt, out_min, out_max, out_rows
    cell = slot[idx] * n_buckets + bucket[idx]
    out_rows.view(-1).index_add_(0, cell, torch.ones_like(cell))
    for f in range(nf):
        v = fields[int(field_idx[f])][:n][idx]
        ok = ~torch.isnan(v)
        c = cell[ok]
        vv = v[ok]
        out_sum[f].view(-1).index_add_(0, c, vv)
        out_cnt[f].view(-1).index_add_(0, c, torch.ones_like(c))
        mn = out_min[f].view(-1)
        mx = out_max[f].view(-1)
        mn.index_reduce_(0, c, vv, "amin", include_self=False)
        mx.index_reduce_(0, c, vv, "amax", include_self=False)
    # index_reduce with include_self=False leaves untouched cells at init NaN
    return out_sum, out_cnt, out_min, out_max, out_rows


def filter_series_time(ts, series, slot_lut, ts_lo, ts_hi):
    keep = (ts >= ts_lo) & (ts < ts_hi)
    if slot_lut is not None and slot_lut.numel() > 0:
        in_lut = (series >= 0) & (series < slot_lut.numel())
        s = series.long().clamp(0, slot_lut.numel() - 1)
        keep &= in_lut & (slot_lut[s] >= 0)
    return keep


def series_last(sources, slot_lut, ts_lo, ts_hi, n_slots):
    """Lastpoint primitive: over sources [(ts, series), ...] (in recency
    order), find per slot the newest ts and its (source, row); later sources
    / later rows win ties. Returns (best_ts i64[n_slots], best_src i64,
    best_row i64), src/row = -1 where empty."""
    best_ts = torch.full((n_slots,), -(1 << 63) + 1, dtype=torch.int64)
    best_src = torch.full((n_slots,), -1, dtype=torch.int64)
    best_row = torch.full((n_slots,), -1, dtype=torch.int64)
    lut = slot_lut.long()
    for ts, series in sources:
        keep = (ts >= ts_lo) & (ts < ts_hi) & (series >= 0) & (series < lut.numel())
        idx = keep.nonzero(as_tuple=True)[0]
        if idx.numel() == 0:
            continue
        slots = lut[series.long()[idx]]
        ok = slots >= 0
        idx, slots = idx[ok], slots[ok]
        t = ts[idx]
        mx = torch.full((n_slots,), -(1 << 63) + 1, dtype=torch.int64)
        mx.index_reduce_(0, slots, t, "amax", include_self=True)
        best_ts = torch.maximum(best_ts, mx)
    best_pack = torch.full((n_slots,), -1, dtype=torch.int64)
    for si, (ts, series) in enumerate(sources):
        keep = (ts >= ts_lo) & (ts < ts_hi) & (series >= 0) & (series < lut.numel())
        idx = keep.nonzero(as_tuple=True)[0]
        if idx.numel() == 0:
            continue
        slots = lut[series.long()[idx]]
        ok = slots >= 0
        idx, slots = idx[ok], slots[ok]
        at = best_ts[slots] == ts[idx]
        ridx = at.nonzero(as_tuple=True)[0]
        # deterministic tie-break: max (source, row) wins (LastRow recency)
        pack = (si << 40) | idx[ridx]
        mx2 = torch.full((n_slots,), -1, dtype=torch.int64)
        mx2.index_reduce_(0, slots[ridx], pack, "amax", include_self=True)
        best_pack = torch.maximum(best_pack, mx2)
    found = best_pack >= 0
    best_src = torch.where(found, best_pack >> 40, best_src)
    best_row = torch.where(found, best_pack & ((1 << 40) - 1), best_row)
    return best_ts, best_src, best_row


PROM_MODES = {
    "instant": 0, "rate": 1, "increase": 2, "delta": 3,
    "avg_over_time": 4, "sum_over_time": 5, "min_over_time": 6,
    "max_over_time": 7, "count_over_time": 8, "last_over_time": 9,
    "idelta": 10, "irate": 11, "deriv": 12, "predict_linear": 13,
    "resets": 14, "changes": 15, "stddev_over_time": 16, "stdvar_over_time": 17,
    "absent_over_time": 18, "quantile_over_time": 19, "last_ts": 20,
}


def prom_range_eval(ts, vals, seg_lo, seg_hi, T, t0, step_ms, range_ms,
                    offset_ms, param, mode):
    """Reference implementation of the PromQL window evaluator (see
    csrc/kernels.hip prom_range_eval_kernel; semantics follow Prometheus'
    extrapolatedRate & friends)."""
    import numpy as np
    ts_h = ts.numpy() if torch.is_tensor(ts) else ts
    v_h = vals.numpy() if torch.is_tensor(vals) else vals
    lo_h = seg_lo.numpy() if torch.is_tensor(seg_lo) else seg_lo
    hi_h = seg_hi.numpy() if torch.is_tensor(seg_hi) else seg_hi
    S = len(lo_h)
    out = np.full((S, T), np.nan)
    for s in range(S):
        a0, b0 = int(lo_h[s]), int(hi_h[s])
        tseg = ts_h[a0:b0]
        vseg = v_h[a0:b0]
        for t in range(T):
            te = t0 + t * step_ms - offset_ms
            tb = te - range_ms
            w_lo = int(np.searchsorted(tseg, tb, "right"))
            w_hi = int(np.searchsorted(tseg, te, "right"))
            cnt = w_hi - w_lo
            w = vseg[w_lo:w_hi]
            wt = tseg[w_lo:w_hi]
            r = np.nan
            if mode in (0, 9):
                if cnt:
                    r = w[-1]
            elif mode == 20:       # last sample ts (dist last_value merge)
                if cnt:
                    r = float(wt[-1])
            elif mode == 8:
                if cnt:
                    r = float(cnt)
            elif mode == 18:
                r = np.nan if cnt else 1.0
            elif mode in (4, 5, 6, 7, 16, 17) and cnt:
                if mode == 5:
                    r = w.sum()
                elif mode == 4:
                    r = w.mean()
                elif mode == 6:
                    r = w.min()
                elif mode == 7:
                    r = w.max()
                else:
                    var = ((w - w.mean()) ** 2).mean()
                    r = var if mode == 17 else np.sqrt(var)
            elif mode in (1, 2, 3) and cnt >= 2:
                is_counter = mode != 3
                total = w[-1] - w[0]
                if is_counter:
                    drops = np.diff(w)
                    if (drops < 0).any():
                        total += w[:-1][drops < 0].sum()
                sampled = (wt[-1] - wt[0]) / 1000.0
                range_s = range_ms / 1000.0
                avg_dur = sampled / (cnt - 1)
                dur_start = (wt[0] - tb) / 1000.0
                dur_end = (te - wt[-1]) / 1000.0
                if is_counter and total > 0 and w[0] >= 0:
                    dz = sampled * (w[0] / total)
                    dur_start = min(dur_start, dz)
                thresh = avg_dur * 1.1
                ext = sampled
                ext += dur_start if dur_start < thresh else avg_dur / 2
                ext += dur_end if dur_end < thresh else avg_dur / 2
                factor = ext / sampled if sampled > 0 else 1.0
                r = total * factor
                if mode == 1:
                    r /= range_s
            elif mode in (10, 11) and cnt >= 2:
                dv = w[-1] - w[-2]
                dt = (wt[-1] - wt[-2]) / 1000.0
                if mode == 10:
                    r = dv
                else:
                    d = w[-1] if w[-1] < w[-2] else dv
                    r = d / dt if dt > 0 else np.nan
            elif mode in (12, 13) and cnt >= 2:
                x = (wt - te) / 1000.0
                n = float(cnt)
                sx, sy = x.sum(), w.sum()
                sxx, sxy = (x * x).sum(), (x * w).sum()
                den = n * sxx - sx * sx
                if den != 0:
                    slope = (n * sxy - sx * sy) / den
                    intercept = (sy - slope * sx) / n
                    r = slope if mode == 12 else intercept + slope * param
            elif mode in (14, 15) and cnt >= 1:
                d = np.diff(w)
                r = float((d < 0).sum()) if mode == 14 else float((d != 0).sum())
            elif mode == 19 and cnt:
                # Prometheus quantile: sorted linear interpolation; q outside
                # [0,1] yields ∓Inf (promql/quantile.go)
                if param < 0:
                    r = -np.inf
                elif param > 1:
                    r = np.inf
                else:
                    sw = np.sort(w)
                    rank = param * (cnt - 1)
                    lo_i = int(np.floor(rank))
                    hi_i = min(lo_i + 1, cnt - 1)
                    r = sw[lo_i] + (sw[hi_i] - sw[lo_i]) * (rank - lo_i)
            out[s, t] = r
    return torch.as_tensor(out)


def dedup_mark_last(series, ts):
    """keep[i] ⇔ row i is the last of its (series, ts) group (sorted input)."""
    n = ts.numel()
    if n == 0:
        return torch.zeros(0, dtype=torch.bool)
    keep = torch.ones(n, dtype=torch.bool)
    keep[:-1] = (series[:-1] != series[1:]) | (ts[:-1] != ts[1:])
    return keep


def scatter_append(ts, series, fields, region_of, dst_off,
                   dst_ts, dst_se, dst_fields):
    """Oracle for kernels.scatter_append (K16): routed bulk memtable write."""
    for r in range(len(dst_ts)):
        m = region_of == r
        if not bool(m.any()):
            continue
        o = dst_off[m]
        dst_ts[r][o] = ts[m]
        dst_se[r][o] = series[m]
        dst_fields[r][:, o] = fields[:, m]
