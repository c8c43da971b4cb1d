"""GPU-resident columnar memtable (K16 target).

Reference parity: mito2 memtable/time_series.rs (TimeSeriesMemtable) — but
redesigned for the GPU: instead of a BTreeMap of per-series builders, one
flat append-only column set lives in HBM (ts i64, series-code i32, fields
f64[nf, cap]); row index == arrival (sequence) order, so "last wins" dedup
is a stable sort away. 288 GB HBM3E makes a flat layout with generous
preallocation the right call — no per-series pointer chasing on device.

Appends are H2D copies at a row offset (torch.narrow().copy_), so ingest is
one pinned-staging copy per column per batch. Sorting/dedup happen at scan
or flush time on device.
"""

from __future__ import annotations

import numpy as np
import torch


class Memtable:
    def __init__(self, n_fields: int, device: str = "cpu", cap: int = 1 << 16):
        self.device = device
        self.nf = n_fields
        self.cap = cap
        self.len = 0
        self.ts = torch.empty(cap, dtype=torch.int64, device=device)
        self.series = torch.empty(cap, dtype=torch.int32, device=device)
        self.fields = torch.empty((n_fields, cap), dtype=torch.float64, device=device)
        self.str_cols: dict[str, list] = {}   # string fields stay host-side
        self.min_ts: int | None = None
        self.max_ts: int | None = None

    @property
    def bytes_used(self) -> int:
        return self.len * (8 + 4 + 8 * self.nf)

    def _grow(self, need: int):
        new_cap = self.cap
        while new_cap < need:
            new_cap *= 2
        for name in ("ts", "series"):
            t = getattr(self, name)
            nt = torch.empty(new_cap, dtype=t.dtype, device=self.device)
            nt[: self.len] = t[: self.len]
            setattr(self, name, nt)
        nf_t = torch.empty((self.nf, new_cap), dtype=torch.float64, device=self.device)
        nf_t[:, : self.len] = self.fields[:, : self.len]
        self.fields = nf_t
        self.cap = new_cap

    def append(self, series: np.ndarray, ts_ms: np.ndarray, fields: np.ndarray,
               str_fields: dict[str, list] | None = None):
        """series i32[n], ts_ms i64[n], fields f64[nf, n] (host arrays or
        tensors); str_fields: {name: list[str|None] of length n}."""
        n = len(ts_ms)
        if n == 0:
            return
        if self.len + n > self.cap:
            self._grow(self.len + n)
        sf = str_fields or {}
        for name in set(self.str_cols) | set(sf):
            col = self.str_cols.get(name)
            if col is None:
                col = [None] * self.len
                self.str_cols[name] = col
            vals = sf.get(name)
            col.extend(vals if vals is not None else [None] * n)
        s = torch.as_tensor(series)
        t = torch.as_tensor(ts_ms)
        f = torch.as_tensor(fields)
        # concurrent auto-ALTER: a writer may carry more (or fewer) field
        # columns than this memtable has seen — reconcile under the region lock
        if f.shape[0] > self.nf:
            self.add_fields(f.shape[0] - self.nf)
        lo, hi = self.len, self.len + n
        self.ts[lo:hi].copy_(t, non_blocking=True)
        self.series[lo:hi].copy_(s, non_blocking=True)
        if f.shape[0] < self.nf:
            self.fields[: f.shape[0], lo:hi].copy_(f, non_blocking=True)
            self.fields[f.shape[0]:, lo:hi] = float("nan")
        else:
            self.fields[:, lo:hi].copy_(f, non_blocking=True)
        mn, mx = int(t.min()), int(t.max())
        self.min_ts = mn if self.min_ts is None else min(self.min_ts, mn)
        self.max_ts = mx if self.max_ts is None else max(self.max_ts, mx)
        self.len = hi

    def add_fields(self, k: int):
        """Grow the field dimension by k (auto-ALTER add-column); new rows NaN."""
        nf2 = self.nf + k
        nt = torch.full((nf2, self.cap), float("nan"), dtype=torch.float64,
                        device=self.device)
        nt[: self.nf, : self.len] = self.fields[:, : self.len]
        self.fields = nt
        self.nf = nf2

    def active(self):
        """(ts, series, fields) views over the filled rows; fields keeps the
        full stride (kernel takes stride separately)."""
        return self.ts[: self.len], self.series[: self.len], self.fields

    def sorted_view(self):
        """Return (ts, series, fields[nf, n], perm) sorted by (series, ts,
        arrival) — stable, so equal (series, ts) keep arrival order and the
        last row of each group is the newest (LastRow dedup semantics)."""
        n = self.len
        ts = self.ts[:n]
        se = self.series[:n]
        ord1 = torch.argsort(ts, stable=True)
        ord2 = torch.argsort(se[ord1], stable=True)
        perm = ord1[ord2]
        return ts[perm], se[perm], self.fields[:, :n][:, perm], perm
