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


def ts_bucket_agg(ts, series, fields, field_idx, slot_lut, ts_lo, ts_hi,
                  origin, bucket_ms, n_slots, n_buckets):
    if ts.is_cuda:
        ops = _require_hip()
        return tuple(ops.ts_bucket_agg(
            ts, series, fields, field_idx, slot_lut,
            int(ts_lo), int(ts_hi), int(origin), int(bucket_ms),
            int(n_slots), int(n_buckets)))
    return cpu_ref.ts_bucket_agg(ts, series, fields, field_idx, slot_lut,
                                 ts_lo, ts_hi, origin, bucket_ms, n_slots, n_buckets)


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
    """See cpu_ref.series_last. GPU path: two atomic-max passes per source."""
    if not sources or not sources[0][0].is_cuda:
        return cpu_ref.series_last(sources, slot_lut, ts_lo, ts_hi, n_slots)
    ops = _require_hip()
    dev = sources[0][0].device
    best_key = torch.zeros(n_slots, dtype=torch.int64, device=dev)
    for ts, series in sources:
        ops.series_last_ts(ts.contiguous(), series.contiguous(), slot_lut,
                           int(ts_lo), int(ts_hi), best_key)
    best_pack = torch.zeros(n_slots, dtype=torch.int64, device=dev)
    for si, (ts, series) in enumerate(sources):
        ops.series_last_row(ts.contiguous(), series.contiguous(), slot_lut,
                            int(ts_lo), int(ts_hi), best_key, si, best_pack)
    best_ts = torch.bitwise_xor(best_key, torch.tensor(-(1 << 63), device=dev))
    none = best_pack == 0
    best_src = torch.where(none, torch.full_like(best_pack, -1), best_pack >> 40)
    best_row = torch.where(none, torch.full_like(best_pack, -1),
                           (best_pack & ((1 << 40) - 1)) - 1)
    best_ts = torch.where(none, torch.full_like(best_ts, -(1 << 63) + 1), best_ts)
    return best_ts.cpu(), best_src.cpu(), best_row.cpu()
