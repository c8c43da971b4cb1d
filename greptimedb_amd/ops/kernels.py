"""Dispatch layer: GPU tensors → _hip_ops (gfx950 kernels), CPU → cpu_ref."""

from __future__ import annotations

import torch

from greptimedb_amd.ops import cpu_ref
from greptimedb_amd.utils.errors import NativeExtensionMissing

try:
    from greptimedb_amd import _hip_ops  # built by setup.py (in-tree .so)

    _HIP_OPS = _hip_ops
    _HIP_IMPORT_ERROR = None
except Exception as e:  # pragma: no cover - exercised only when build broken
    _HIP_OPS = None
    _HIP_IMPORT_ERROR = e


def hip_ops_available() -> bool:
    return _HIP_OPS is not None


def _require_hip():
    if _HIP_OPS is None:
        raise NativeExtensionMissing(
            "greptimedb_amd._hip_ops is not built; refusing to run the GPU path "
            "on eager fallback. Run `python setup.py build_ext --inplace`. "
            f"(import error: {_HIP_IMPORT_ERROR})"
        )
    return _HIP_OPS


def ts_bucket_agg_acc(ts, series, fields, field_idx, slot_lut, ts_lo, ts_hi,
                      origin, bucket_ms, n_slots, n_buckets, acc=None):
    """Accumulating form for multi-source scans: pass the returned opaque
    `acc` into the next call, then ts_bucket_agg_finish(acc). On GPU the
    kernel atomics accumulate into shared buffers (no per-source combine)."""
    if ts.is_cuda:
        ops = _require_hip()
        return ("gpu", ops.ts_bucket_agg(
            ts, series, fields, field_idx, slot_lut,
            int(ts_lo), int(ts_hi), int(origin), int(bucket_ms),
            int(n_slots), int(n_buckets),
            acc[1] if acc is not None else []))
    out = cpu_ref.ts_bucket_agg(ts, series, fields, field_idx, slot_lut,
                                ts_lo, ts_hi, origin, bucket_ms,
                                n_slots, n_buckets)
    if acc is None:
        return ("cpu", list(out))
    prev = acc[1]
    prev[0] += out[0]
    prev[1] += out[1]
    prev[2] = torch.fmin(prev[2], out[2])
    prev[3] = torch.fmax(prev[3], out[3])
    prev[4] += out[4]
    return ("cpu", prev)


def ts_bucket_agg_finish(acc):
    """(sum, cnt, min, max, rows) from an accumulator handle."""
    kind, t = acc
    if kind == "gpu":
        return tuple(_require_hip().ts_bucket_agg_finish(t))
    return tuple(t)


def ts_bucket_agg(ts, series, fields, field_idx, slot_lut, ts_lo, ts_hi,
                  origin, bucket_ms, n_slots, n_buckets):
    return ts_bucket_agg_finish(ts_bucket_agg_acc(
        ts, series, fields, field_idx, slot_lut, ts_lo, ts_hi, origin,
        bucket_ms, n_slots, n_buckets))


def filter_series_time(ts, series, slot_lut, ts_lo, ts_hi):
    if ts.is_cuda:
        ops = _require_hip()
        lut = slot_lut if slot_lut is not None else torch.empty(0, dtype=torch.int32, device=ts.device)
        return ops.filter_series_time(ts, series, lut, int(ts_lo), int(ts_hi))
    return cpu_ref.filter_series_time(ts, series, slot_lut, ts_lo, ts_hi)


def dedup_mark_last(series, ts):
    if ts.is_cuda:
        return _require_hip().dedup_mark_last(series, ts)
    return cpu_ref.dedup_mark_last(series, ts)


def prom_range_eval(ts, vals, seg_lo, seg_hi, T, t0, step_ms, range_ms,
                    offset_ms, param, mode):
    """PromQL window evaluator over (slot, ts)-sorted samples → [S, T]."""
    if torch.is_tensor(ts) and ts.is_cuda:
        return _require_hip().prom_range_eval(
            ts.contiguous(), vals.contiguous(), seg_lo.contiguous(),
            seg_hi.contiguous(), int(T), int(t0), int(step_ms), int(range_ms),
            int(offset_ms), float(param), int(mode))
    return cpu_ref.prom_range_eval(ts, vals, seg_lo, seg_hi, T, t0, step_ms,
                                   range_ms, offset_ms, param, mode)


def series_last(sources, slot_lut, ts_lo, ts_hi, n_slots):
    """See cpu_ref.series_last. Sources may be (ts, series) or
    (ts, series, sorted) triples. GPU: (series, ts)-sorted sources resolve
    per-code last rows from segment boundaries (no full scan — rocprof
    showed the atomic scan at 715µs/source); unsorted sources use the
    two-pass atomic-max kernels."""
    pairs = [(s[0], s[1], (s[2] if len(s) > 2 else False)) for s in sources]
    if not pairs or not pairs[0][0].is_cuda:
        return cpu_ref.series_last([(t, se) for t, se, _ in pairs],
                                   slot_lut, ts_lo, ts_hi, n_slots)
    dev = pairs[0][0].device
    lut = slot_lut.long()
    ncodes = lut.numel()
    NEG = -(1 << 62)
    best_ts = torch.full((n_slots,), NEG, dtype=torch.int64, device=dev)
    best_src = torch.full((n_slots,), -1, dtype=torch.int64, device=dev)
    best_row = torch.full((n_slots,), -1, dtype=torch.int64, device=dev)
    unsorted = []
    code_range = torch.arange(ncodes, dtype=torch.int64, device=dev)
    slots_of_code = lut
    for si, (ts, series, srt) in enumerate(pairs):
        if not srt:
            unsorted.append((si, ts, series))
            continue
        se = series.long()
        lo_b = torch.searchsorted(se, code_range)
        hi_b = torch.searchsorted(se, code_range + 1)
        has = hi_b > lo_b
        last_idx = (hi_b - 1).clamp_min(0)
        cand_ts = ts[last_idx]
        # honor the [ts_lo, ts_hi) window: if the last row is outside it,
        # binary-search the last row <= ts_hi-1 within the segment
        if ts_hi < (1 << 62) or ts_lo > -(1 << 62):
            # a segment-last row outside [ts_lo, ts_hi) needs an in-segment
            # search — defer that source to the kernel path (rare: lastpoint
            # is normally unbounded)
            outside = has & ((cand_ts >= ts_hi) | (cand_ts < ts_lo))
            if bool(outside.any()):
                unsorted.append((si, ts, series))
                continue
        valid = has & (slots_of_code >= 0)
        if not bool(valid.any()):
            continue
        sl = slots_of_code[valid]
        c_ts = cand_ts[valid]
        c_row = last_idx[valid]
        tmp = torch.full((n_slots,), NEG, dtype=torch.int64, device=dev)
        tmp.scatter_reduce_(0, sl, c_ts, "amax", include_self=True)
        win = c_ts == tmp[sl]
        rowtmp = torch.full((n_slots,), -1, dtype=torch.int64, device=dev)
        rowtmp.scatter_reduce_(0, sl[win], c_row[win], "amax", include_self=True)
        upd = (tmp > NEG) & (tmp >= best_ts)
        best_ts = torch.where(upd, tmp, best_ts)
        best_src = torch.where(upd, torch.full_like(best_src, si), best_src)
        best_row = torch.where(upd, rowtmp, best_row)
    if unsorted:
        ops = _require_hip()
        best_key = torch.zeros(n_slots, dtype=torch.int64, device=dev)
        for _si, ts, series in unsorted:
            ops.series_last_ts(ts.contiguous(), series.contiguous(), slot_lut,
                               int(ts_lo), int(ts_hi), best_key)
        best_pack = torch.zeros(n_slots, dtype=torch.int64, device=dev)
        for si, ts, series in unsorted:
            ops.series_last_row(ts.contiguous(), series.contiguous(), slot_lut,
                                int(ts_lo), int(ts_hi), best_key, si, best_pack)
        k_ts = torch.bitwise_xor(best_key, torch.tensor(-(1 << 63), device=dev))
        found = best_pack != 0
        k_src = torch.where(found, best_pack >> 40, torch.full_like(best_pack, -1))
        k_row = torch.where(found, (best_pack & ((1 << 40) - 1)) - 1,
                            torch.full_like(best_pack, -1))
        # kernel sources appear after sorted ones in recency order when they
        # include the memtable; ties go to the kernel result iff its source
        # index is later
        upd = found & ((k_ts > best_ts) |
                       ((k_ts == best_ts) & (k_src >= best_src)))
        best_ts = torch.where(upd, k_ts, best_ts)
        best_src = torch.where(upd, k_src, best_src)
        best_row = torch.where(upd, k_row, best_row)
    none = best_src < 0
    best_ts = torch.where(none, torch.full_like(best_ts, -(1 << 63) + 1), best_ts)
    return best_ts.cpu(), best_src.cpu(), best_row.cpu()


def scatter_append(ts, series, fields, region_of, dst_off,
                   dst_ts, dst_se, dst_fields):
    """K16 bulk memtable append: write row i of (ts, series, fields[nf, n])
    into region region_of[i] at row dst_off[i]. `dst_*` are per-region
    destination tensors (fields [nf, cap], stride = cap). One kernel replaces
    the per-region narrow+copy_ chain on the ingest hot path."""
    if torch.is_tensor(ts) and ts.is_cuda:
        _require_hip().scatter_append(
            ts.contiguous(), series.contiguous(), fields.contiguous(),
            region_of.contiguous(), dst_off.contiguous(),
            list(dst_ts), list(dst_se), list(dst_fields))
        return
    cpu_ref.scatter_append(ts, series, fields, region_of, dst_off,
                           dst_ts, dst_se, dst_fields)
