"""CPU reference op semantics vs brute-force numpy (the oracle the GPU
kernels must match — see tests/test_ops_gpu.py for the device comparison)."""

import numpy as np
import torch

from greptimedb_amd.ops import cpu_ref


def _brute_force(ts, series, fields, field_idx, slot_lut, ts_lo, ts_hi,
                 origin, bucket_ms, n_slots, n_buckets):
    nf = len(field_idx)
    sums = np.zeros((nf, n_slots, n_buckets))
    cnts = np.zeros((nf, n_slots, n_buckets), dtype=np.int64)
    mins = np.full((nf, n_slots, n_buckets), np.nan)
    maxs = np.full((nf, n_slots, n_buckets), np.nan)
    rows = np.zeros((n_slots, n_buckets), dtype=np.int64)
    for i in range(len(ts)):
        t = ts[i]
        if not (ts_lo <= t < ts_hi):
            continue
        s = series[i]
        if not (0 <= s < len(slot_lut)):
            continue
        slot = slot_lut[s]
        if slot < 0:
            continue
        b = (t - origin) // bucket_ms
        if not (0 <= b < n_buckets):
            continue
        rows[slot, b] += 1
        for f in range(nf):
            v = fields[field_idx[f]][i]
            if np.isnan(v):
                continue
            sums[f, slot, b] += v
            cnts[f, slot, b] += 1
            mins[f, slot, b] = v if np.isnan(mins[f, slot, b]) else min(mins[f, slot, b], v)
            maxs[f, slot, b] = v if np.isnan(maxs[f, slot, b]) else max(maxs[f, slot, b], v)
    return sums, cnts, mins, maxs, rows


def make_case(seed=0, n=5000, n_series=37, nf=4, with_nan=True):
    rng = np.random.RandomState(seed)
    ts = rng.randint(0, 1_000_000, n).astype(np.int64)
    series = rng.randint(-1, n_series + 3, n).astype(np.int32)
    fields = rng.uniform(-1e6, 1e6, size=(nf, n))
    if with_nan:
        fields[rng.uniform(size=(nf, n)) < 0.1] = np.nan
    slot_lut = rng.randint(-1, 5, n_series).astype(np.int32)
    return ts, series, fields, slot_lut


def test_ts_bucket_agg_matches_brute_force():
    ts, series, fields, slot_lut = make_case()
    field_idx = np.array([0, 2, 3], dtype=np.int32)
    args = (100_000, 900_000, 100_000, 50_000, 5, 16)
    exp = _brute_force(ts, series, fields, field_idx, slot_lut, *args)
    got = cpu_ref.ts_bucket_agg(
        torch.as_tensor(ts), torch.as_tensor(series), torch.as_tensor(fields),
        torch.as_tensor(field_idx), torch.as_tensor(slot_lut), *args)
    for e, g, name in zip(exp, got, ["sum", "cnt", "min", "max", "rows"]):
        np.testing.assert_allclose(e, g.numpy(), rtol=1e-12, equal_nan=True,
                                   err_msg=name)


def test_filter_series_time():
    ts, series, fields, slot_lut = make_case(seed=1)
    got = cpu_ref.filter_series_time(
        torch.as_tensor(ts), torch.as_tensor(series), torch.as_tensor(slot_lut),
        200_000, 700_000).numpy()
    exp = np.zeros(len(ts), dtype=bool)
    for i in range(len(ts)):
        ok = 200_000 <= ts[i] < 700_000
        if ok:
            ok = 0 <= series[i] < len(slot_lut) and slot_lut[series[i]] >= 0
        exp[i] = ok
    np.testing.assert_array_equal(exp, got)


def test_filter_no_lut():
    ts = torch.tensor([1, 5, 9], dtype=torch.int64)
    se = torch.tensor([0, 1, 2], dtype=torch.int32)
    got = cpu_ref.filter_series_time(ts, se, None, 2, 9)
    assert got.tolist() == [False, True, False]


def test_dedup_mark_last():
    series = torch.tensor([1, 1, 1, 2, 2, 3], dtype=torch.int32)
    ts = torch.tensor([10, 10, 20, 20, 20, 20], dtype=torch.int64)
    got = cpu_ref.dedup_mark_last(series, ts)
    assert got.tolist() == [False, True, True, False, True, True]
    assert cpu_ref.dedup_mark_last(torch.zeros(0, dtype=torch.int32),
                                   torch.zeros(0, dtype=torch.int64)).numel() == 0


def test_series_last():
    ts1 = torch.tensor([10, 20, 30, 5], dtype=torch.int64)
    se1 = torch.tensor([0, 0, 1, 2], dtype=torch.int32)
    ts2 = torch.tensor([20, 25], dtype=torch.int64)
    se2 = torch.tensor([0, 1], dtype=torch.int32)
    lut = torch.tensor([0, 1, -1], dtype=torch.int32)
    b_ts, b_src, b_row = cpu_ref.series_last(
        [(ts1, se1), (ts2, se2)], lut, 0, 100, 2)
    # slot0 (series 0): max ts 20 appears in src0 row1 and src1 row0 → src1 wins tie
    assert b_ts.tolist() == [20, 30]
    assert b_src.tolist() == [1, 0]
    assert b_row.tolist() == [0, 2]
    # series 2 excluded by lut
