"""PromStore: Prometheus remote-write ingestion via the metric engine.

Reference parity: src/servers/src/prom_store.rs (+ prom_row_builder.rs) and
src/metric-engine — thousands of logical metrics multiplexed onto ONE
physical table's regions, series identified by a sparse label-set primary
key with __name__ as an ordinary label (reference row_modifier.rs adds
__table_id/__tsid; we carry __name__ in the sparse pk and dict-encode it in
the series index, which plays the same multiplexing role).

The native PromWriteParser (csrc/native.cpp) does snappy + protobuf decode
and interns label sets to dense series refs, so the steady-state hot path is
LUT routing + GPU memtable appends, same as influx ingest.
"""

from __future__ import annotations

import numpy as np

from greptimedb_amd import _native
from greptimedb_amd.engine import pk_codec
from greptimedb_amd.engine.engine import MitoEngine
from greptimedb_amd.engine.series import tsid_hash
from greptimedb_amd.models.schema import ColumnSchema, DataType, SemanticType, TableSchema

PHYSICAL_TABLE = "greptime_metrics"
VALUE_FIELD = "greptime_value"


def physical_schema() -> TableSchema:
    return TableSchema(
        name=PHYSICAL_TABLE,
        columns=[
            ColumnSchema("ts", DataType.TIMESTAMP_MS, SemanticType.TIMESTAMP, 0),
            ColumnSchema(VALUE_FIELD, DataType.FLOAT64, SemanticType.FIELD, 1),
        ],
        primary_key=[],
        options={"metric_engine": "true"},
    )


class PromStore:
    def __init__(self, engine: MitoEngine, n_regions: int | None = None,
                 durable: bool = True):
        self.engine = engine
        self.durable = durable
        self.parser = _native.PromWriteParser()
        self.table = engine.create_table(physical_schema(),
                                         n_regions=n_regions,
                                         append_mode=True, if_not_exists=True)
        self._cap = 1024
        self.sid_region = np.full(self._cap, -1, dtype=np.int32)
        self.sid_local = np.full(self._cap, -1, dtype=np.int32)
        self.rows_ingested = 0
        self.metrics: set[str] = set()

    def _grow(self, need):
        cap = self._cap
        while cap <= need:
            cap *= 2
        for name in ("sid_region", "sid_local"):
            a = getattr(self, name)
            na = np.full(cap, -1, dtype=np.int32)
            na[: len(a)] = a
            setattr(self, name, na)
        self._cap = cap

    def write_points(self, points) -> int:
        """Direct datapoint ingestion: [(metric, {tag: value}, ts_ms, value)]
        (OpenTSDB put / OTLP metric datapoints land here)."""
        if not points:
            return 0
        st = self.table
        n = len(points)
        series = np.empty(n, dtype=np.int32)
        regions = np.empty(n, dtype=np.int32)
        ts = np.empty(n, dtype=np.int64)
        vals = np.empty(n, dtype=np.float64)
        for i, (metric, tags, t, v) in enumerate(points):
            lab = {k: str(tv) for k, tv in (tags or {}).items()}
            lab["__name__"] = metric
            self.metrics.add(metric)
            pk = pk_codec.encode_sparse(lab)
            ridx = tsid_hash(pk) % len(st.regions)
            series[i] = st.regions[ridx].register_series_labels(lab)
            regions[i] = ridx
            ts[i] = int(t)
            vals[i] = float(v)
        order = np.argsort(regions, kind="stable")
        rs = regions[order]
        bounds = np.flatnonzero(np.diff(rs)) + 1
        for s, e in zip(np.concatenate(([0], bounds)),
                        np.concatenate((bounds, [n]))):
            ridx = int(rs[s])
            rows = order[s:e]
            self.engine.write_region(st, ridx, series[rows], ts[rows],
                                     vals[rows][None, :], [],
                                     durable=self.durable)
        if self.durable:
            self.engine.commit_wal()
        self.rows_ingested += n
        return n

    def write(self, body: bytes, snappy: bool = True) -> int:
        """Ingest one remote-write request body. Returns sample count."""
        series, ts, vals, new_series = self.parser.parse(body, snappy)
        n = len(series)
        if n == 0:
            return 0
        st = self.table
        for sid, metric, labels in new_series:
            lab = dict(labels)
            if metric:
                lab["__name__"] = metric
                self.metrics.add(metric)
            pk = pk_codec.encode_sparse(lab)
            ridx = tsid_hash(pk) % len(st.regions)
            local = st.regions[ridx].register_series_labels(lab)
            if sid >= self._cap:
                self._grow(sid)
            self.sid_region[sid] = ridx
            self.sid_local[sid] = local

        region_of = self.sid_region[series]
        local = self.sid_local[series]
        order = np.argsort(region_of, kind="stable")
        rs = region_of[order]
        bounds = np.flatnonzero(np.diff(rs)) + 1
        starts = np.concatenate(([0], bounds))
        ends = np.concatenate((bounds, [n]))
        for s, e in zip(starts, ends):
            ridx = int(rs[s])
            rows = order[s:e]
            self.engine.write_region(
                st, ridx, local[rows].astype(np.int32), ts[rows],
                vals[rows][None, :], [], durable=self.durable)
        if self.durable:
            self.engine.commit_wal()
        self.engine.maybe_flush()
        self.rows_ingested += n
        return n
