"""OTLP trace ingestion (greptime_trace_v1-style table model).

Reference parity: src/servers/src/otlp/trace/ — OTLP spans land in an
`opentelemetry_traces` table: tags (service_name, span_name), time index =
span start, duration_ms field, id/attribute string columns. The protobuf
decode + (service, span) interning is native (csrc OtlpTraceParser) so the
span hot path is LUT routing + GPU appends like the other stores.
"""

from __future__ import annotations

import numpy as np

from greptimedb_amd import _native
from greptimedb_amd.engine import pk_codec
from greptimedb_amd.engine.engine import MitoEngine
from greptimedb_amd.engine.series import tsid_hash
from greptimedb_amd.models.schema import ColumnSchema, DataType, SemanticType, TableSchema

TRACE_TABLE = "opentelemetry_traces"


def trace_schema() -> TableSchema:
    return TableSchema(
        name=TRACE_TABLE,
        columns=[
            ColumnSchema("service_name", DataType.STRING, SemanticType.TAG, 0),
            ColumnSchema("span_name", DataType.STRING, SemanticType.TAG, 1),
            ColumnSchema("ts", DataType.TIMESTAMP_MS, SemanticType.TIMESTAMP, 2),
            ColumnSchema("duration_ms", DataType.FLOAT64, SemanticType.FIELD, 3),
            ColumnSchema("status_code", DataType.FLOAT64, SemanticType.FIELD, 4),
            # id columns are point-looked-up, not fulltext-searched: no index
            ColumnSchema("trace_id", DataType.STRING, SemanticType.FIELD, 5,
                         fulltext=False),
            ColumnSchema("span_id", DataType.STRING, SemanticType.FIELD, 6,
                         fulltext=False),
            ColumnSchema("parent_span_id", DataType.STRING, SemanticType.FIELD, 7,
                         fulltext=False),
            ColumnSchema("span_attributes", DataType.JSON, SemanticType.FIELD, 8,
                         fulltext=False),
        ],
        primary_key=["service_name", "span_name"],
        options={"append_mode": "true"},
    )


class TraceStore:
    def __init__(self, engine: MitoEngine, n_regions: int | None = None,
                 durable: bool = True):
        self.engine = engine
        self.durable = durable
        self.parser = _native.OtlpTraceParser()
        self.table = engine.create_table(trace_schema(), n_regions=n_regions,
                                         append_mode=True, if_not_exists=True)
        self._cap = 1024
        self.sid_region = np.full(self._cap, -1, dtype=np.int32)
        self.sid_local = np.full(self._cap, -1, dtype=np.int32)
        self.spans_ingested = 0

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

    def write(self, body: bytes) -> int:
        """Ingest one OTLP ExportTraceServiceRequest. Returns span count."""
        (series, start_ns, dur_ms, status, trace_ids, span_ids, parent_ids,
         attrs, new_series) = self.parser.parse(body)
        n = len(series)
        if n == 0:
            return 0
        st = self.table
        for sid, service, name in new_series:
            tags = (service, name)
            pk = pk_codec.encode_pk(tags)
            ridx = tsid_hash(pk) % len(st.regions)
            local = st.regions[ridx].register_series(tags)
            if sid >= self._cap:
                self._grow(sid)
            self.sid_region[sid] = ridx
            self.sid_local[sid] = local
        ts_ms = start_ns // 1_000_000
        region_of = self.sid_region[series]
        local = self.sid_local[series]
        order = np.argsort(region_of, kind="stable")
        rs = region_of[order]
        bounds = np.flatnonzero(np.diff(rs)) + 1
        starts = np.concatenate(([0], bounds))
        ends = np.concatenate((bounds, [n]))
        tid_a = np.asarray(trace_ids, dtype=object)
        sid_a = np.asarray(span_ids, dtype=object)
        pid_a = np.asarray(parent_ids, dtype=object)
        att_a = np.asarray(attrs, dtype=object)
        for s, e in zip(starts, ends):
            ridx = int(rs[s])
            rows = order[s:e]
            fmat = np.stack([dur_ms[rows], status[rows].astype(np.float64)])
            strs = {
                "trace_id": list(tid_a[rows]),
                "span_id": list(sid_a[rows]),
                "parent_span_id": list(pid_a[rows]),
                "span_attributes": list(att_a[rows]),
            }
            self.engine.write_region(st, ridx, local[rows].astype(np.int32),
                                     ts_ms[rows], fmat, [],
                                     durable=self.durable, str_fields=strs)
        if self.durable:
            self.engine.commit_wal()
        self.engine.maybe_flush()
        self.spans_ingested += n
        return n
