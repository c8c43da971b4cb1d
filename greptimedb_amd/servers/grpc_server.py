"""gRPC surface: greptime.v1.GreptimeDatabase (+ health).

Reference parity: src/servers/src/grpc/builder.rs:138-172 (service
assembly) and src/servers/src/grpc/database.rs (GreptimeDatabase.Handle:
row inserts / ddl; unary + streaming). Queries over gRPC travel via Arrow
Flight (servers/flight.py), as in the reference where FlightService is the
query data plane and GreptimeDatabase.Handle returns affected-rows only.

Message layout follows greptime-proto v1 (greptime/v1/database.proto,
common.proto, row.proto). grpcio-tools is not installed in this image, so
the messages are hand-encoded with utils/pb.py; field numbers below are
the greptime-proto v1 layout (re-verify against the proto when vendoring
becomes possible — the self-consistency of client+server is covered by
tests/test_grpc.py round trips).

  GreptimeRequest { RequestHeader header=1;
                    oneof { InsertRequests inserts=2; QueryRequest query=3;
                            DdlRequest ddl=4; DeleteRequests deletes=5;
                            RowInsertRequests row_inserts=6; } }
  RequestHeader   { catalog=1; schema=2; authorization=3; dbname=4; }
  QueryRequest    { oneof { sql=1; logical_plan=2; } }
  RowInsertRequests { repeated RowInsertRequest inserts=1 }
  RowInsertRequest  { table_name=1; Rows rows=2 }
  Rows { repeated ColumnSchema schema=1; repeated Row rows=2 }
  ColumnSchema { column_name=1; datatype=2; semantic_type=3 }
  Row  { repeated Value values=1 }
  Value oneof: i8=1 i16=2 i32=3 i64=4 u8=5 u16=6 u32=7 u64=8 f32=9 f64=10
               bool=11 binary=12 string=13 date=14 datetime=15
               ts_second=16 ts_ms=17 ts_us=18 ts_ns=19
  GreptimeResponse { ResponseHeader header=1; AffectedRows affected_rows=2 }
  ResponseHeader { Status status=1 };  Status { status_code=1; err_msg=2 }
"""

from __future__ import annotations

from concurrent import futures

import grpc
import numpy as np

from greptimedb_amd.utils import pb

# SemanticType enum (greptime-proto common.proto) == models.schema order
SEM_TAG, SEM_FIELD, SEM_TIMESTAMP = 0, 1, 2

# ColumnDataType enum (subset)
DT_BOOLEAN, DT_INT8, DT_INT16, DT_INT32, DT_INT64 = 1, 2, 3, 4, 5
DT_UINT8, DT_UINT16, DT_UINT32, DT_UINT64 = 6, 7, 8, 9
DT_FLOAT32, DT_FLOAT64, DT_BINARY, DT_STRING = 10, 11, 12, 14
DT_DATE, DT_DATETIME = 15, 16
DT_TS_SECOND, DT_TS_MILLI, DT_TS_MICRO, DT_TS_NANO = 17, 18, 19, 20

_VALUE_TS_SCALE = {16: 1000, 17: 1, 18: 1 / 1000, 19: 1 / 1_000_000}

METHOD_HANDLE = "/greptime.v1.GreptimeDatabase/Handle"
METHOD_HEALTH = "/grpc.health.v1.Health/Check"


# ------------------------------------------------------------------ encode

def encode_value(kind: int, v) -> bytes:
    w = pb.Writer()
    if v is None:
        return w.build()
    if kind in (9,):
        w.f32(kind, v)
    elif kind == 10:
        w.f64(kind, v)
    elif kind in (12, 13):
        w.bytes(kind, v.encode() if isinstance(v, str) else v)
    else:
        w.varint(kind, int(v))
    return w.build()


def decode_value(buf: bytes):
    """One Value message → (value_kind, python value) or (None, None)."""
    for field, wire, v in pb.fields(buf):
        if wire == 1:
            return field, pb.as_f64(v)
        if wire == 5:
            return field, pb.as_f32(v)
        if wire == 2:
            return field, v
        return field, pb.as_i64(v)
    return None, None


def encode_row_insert(table: str, schema: list[tuple[str, int, int]],
                      rows: list[list]) -> pb.Writer:
    """RowInsertRequest writer. schema: [(name, datatype, semantic)]."""
    rw = pb.Writer()
    rw.string(1, table)
    rows_w = pb.Writer()
    value_kind = []
    for name, dt, sem in schema:
        cs = pb.Writer().string(1, name).varint(2, dt).varint(3, sem)
        rows_w.msg(1, cs)
        value_kind.append(_dt_to_value_kind(dt))
    for row in rows:
        r = pb.Writer()
        for kind, v in zip(value_kind, row):
            r.bytes(1, encode_value(kind, v))
        rows_w.msg(2, r)
    rw.msg(2, rows_w)
    return rw


def _dt_to_value_kind(dt: int) -> int:
    return {
        DT_BOOLEAN: 11, DT_INT8: 1, DT_INT16: 2, DT_INT32: 3, DT_INT64: 4,
        DT_UINT8: 5, DT_UINT16: 6, DT_UINT32: 7, DT_UINT64: 8,
        DT_FLOAT32: 9, DT_FLOAT64: 10, DT_BINARY: 12, DT_STRING: 13,
        DT_DATE: 14, DT_DATETIME: 15, DT_TS_SECOND: 16, DT_TS_MILLI: 17,
        DT_TS_MICRO: 18, DT_TS_NANO: 19,
    }[dt]


def encode_request(row_inserts: list[pb.Writer] | None = None,
                   sql: str | None = None, dbname: str = "public") -> bytes:
    req = pb.Writer()
    req.msg(1, pb.Writer().string(4, dbname))
    if row_inserts is not None:
        ri = pb.Writer()
        for r in row_inserts:
            ri.msg(1, r)
        req.msg(6, ri)
    elif sql is not None:
        req.msg(3, pb.Writer().string(1, sql))
    return req.build()


def encode_response(affected: int, code: int = 0, err: str = "") -> bytes:
    resp = pb.Writer()
    status = pb.Writer().varint(1, code)
    if err:
        status.string(2, err)
    resp.msg(1, pb.Writer().msg(1, status))
    resp.msg(2, pb.Writer().varint(1, affected))
    return resp.build()


def decode_response(buf: bytes) -> tuple[int, int, str]:
    """→ (affected_rows, status_code, err_msg)."""
    affected, code, err = 0, 0, ""
    for field, _w, v in pb.fields(buf):
        if field == 1:
            for f2, _w2, v2 in pb.fields(v):
                if f2 == 1:
                    for f3, _w3, v3 in pb.fields(v2):
                        if f3 == 1:
                            code = v3
                        elif f3 == 2:
                            err = v3.decode()
        elif field == 2:
            for f2, _w2, v2 in pb.fields(v):
                if f2 == 1:
                    affected = v2
    return affected, code, err


# ------------------------------------------------------------------ server

class GreptimeGrpcServer:
    """GreptimeDatabase.Handle (row inserts, SQL ddl/inserts) + health."""

    def __init__(self, engine, executor=None, host: str = "127.0.0.1",
                 port: int = 0, max_workers: int = 8):
        self.engine = engine
        if executor is None:
            from greptimedb_amd.query.executor import Executor
            executor = Executor(engine)
        self.executor = executor
        self._server = grpc.server(
            futures.ThreadPoolExecutor(max_workers=max_workers))
        self._server.add_generic_rpc_handlers((_Handlers(self),))
        self.port = self._server.add_insecure_port(f"{host}:{port}")
        self._server.start()

    def shutdown(self):
        self._server.stop(grace=1)

    # ---------------------------------------------------------- dispatch
    def handle(self, request: bytes, context) -> bytes:
        try:
            affected = 0
            for field, _w, v in pb.fields(request):
                if field == 6:      # row_inserts
                    affected += self._handle_row_inserts(v)
                elif field == 3:    # query (sql only; results go via Flight)
                    affected += self._handle_query(v)
                elif field == 4:    # ddl — not decoded; SQL DDL via query
                    raise ValueError("DdlRequest not supported; send SQL DDL")
            return encode_response(affected)
        except Exception as e:
            return encode_response(0, code=3000, err=f"{type(e).__name__}: {e}")

    def _handle_query(self, buf: bytes) -> int:
        sql = None
        for field, _w, v in pb.fields(buf):
            if field == 1:
                sql = v.decode()
        if sql is None:
            raise ValueError("only QueryRequest.sql supported")
        res = self.executor.execute(sql)
        return len(res) if res.columns else 0

    def _handle_row_inserts(self, buf: bytes) -> int:
        total = 0
        for field, _w, v in pb.fields(buf):
            if field == 1:
                total += self._one_insert(v)
        return total

    def _one_insert(self, buf: bytes) -> int:
        import pyarrow as pa
        table_name = None
        schema: list[tuple[str, int, int]] = []
        rows: list[list] = []
        for field, _w, v in pb.fields(buf):
            if field == 1:
                table_name = v.decode()
            elif field == 2:
                for f2, _w2, v2 in pb.fields(v):
                    if f2 == 1:   # ColumnSchema
                        name, dt, sem = None, DT_FLOAT64, SEM_FIELD
                        for f3, _w3, v3 in pb.fields(v2):
                            if f3 == 1:
                                name = v3.decode()
                            elif f3 == 2:
                                dt = v3
                            elif f3 == 3:
                                sem = v3
                        schema.append((name, dt, sem))
                    elif f2 == 2:  # Row
                        row = []
                        for f3, _w3, v3 in pb.fields(v2):
                            if f3 == 1:
                                row.append(decode_value(v3))
                        rows.append(row)
        if table_name is None or not schema:
            raise ValueError("RowInsertRequest missing table_name/schema")
        self._ensure_table(table_name, schema)
        # row-major Values → columns
        ncol = len(schema)
        cols: list[list] = [[] for _ in range(ncol)]
        for row in rows:
            for i in range(ncol):
                kind, v = row[i] if i < len(row) else (None, None)
                if kind in (12, 13) and isinstance(v, bytes) and kind == 13:
                    v = v.decode()
                if kind in _VALUE_TS_SCALE and v is not None:
                    v = int(v * _VALUE_TS_SCALE[kind])
                cols[i].append(v)
        arrays, names = [], []
        for (name, dt, sem), col in zip(schema, cols):
            names.append(name)
            if sem == SEM_TIMESTAMP or dt in (DT_TS_SECOND, DT_TS_MILLI,
                                              DT_TS_MICRO, DT_TS_NANO):
                arrays.append(pa.array(np.asarray(col, dtype=np.int64))
                              .cast(pa.timestamp("ms")))
            elif sem == SEM_TAG or dt in (DT_STRING,):
                arrays.append(pa.array(
                    [None if c is None
                     else (c.decode() if isinstance(c, bytes) else str(c))
                     for c in col], type=pa.string()))
            elif dt == DT_BINARY:
                arrays.append(pa.array(col, type=pa.binary()))
            else:
                arrays.append(pa.array(
                    [None if c is None else float(c) for c in col],
                    type=pa.float64()))
        batch = pa.table(dict(zip(names, arrays)))
        from greptimedb_amd.engine.bulk import bulk_insert_arrow
        return bulk_insert_arrow(self.engine, table_name, batch)

    def _ensure_table(self, name: str, schema: list[tuple[str, int, int]]):
        """Auto-create from the proto schema's semantic types (reference:
        grpc-expr insert→create inference)."""
        if name in self.engine.tables:
            return
        from greptimedb_amd.models.schema import (ColumnSchema, DataType,
                                                  SemanticType, TableSchema)
        cols, pk = [], []
        for cid, (cname, dt, sem) in enumerate(schema):
            if sem == SEM_TIMESTAMP:
                cols.append(ColumnSchema(cname, DataType.TIMESTAMP_MS,
                                         SemanticType.TIMESTAMP, cid))
            elif sem == SEM_TAG:
                cols.append(ColumnSchema(cname, DataType.STRING,
                                         SemanticType.TAG, cid))
                pk.append(cname)
            elif dt in (DT_STRING, DT_BINARY):
                cols.append(ColumnSchema(cname, DataType.STRING,
                                         SemanticType.FIELD, cid,
                                         fulltext=True))
            else:
                cols.append(ColumnSchema(cname, DataType.FLOAT64,
                                         SemanticType.FIELD, cid))
        self.engine.create_table(TableSchema(name=name, columns=cols,
                                             primary_key=pk),
                                 if_not_exists=True)


class _Handlers(grpc.GenericRpcHandler):
    def __init__(self, server: GreptimeGrpcServer):
        self.server = server

    def service(self, details):
        if details.method == METHOD_HANDLE:
            return grpc.unary_unary_rpc_method_handler(
                lambda req, ctx: self.server.handle(req, ctx))
        if details.method == METHOD_HEALTH:
            # HealthCheckResponse { status = 1 } ; SERVING = 1
            return grpc.unary_unary_rpc_method_handler(
                lambda req, ctx: pb.Writer().varint(1, 1).build())
        return None


# ------------------------------------------------------------------ client

class GreptimeGrpcClient:
    """Minimal client for tests/SDK parity: insert rows + run SQL."""

    def __init__(self, host: str, port: int, dbname: str = "public"):
        self.channel = grpc.insecure_channel(f"{host}:{port}")
        self.dbname = dbname
        self._handle = self.channel.unary_unary(METHOD_HANDLE)
        self._health = self.channel.unary_unary(METHOD_HEALTH)

    def insert_rows(self, table: str, schema: list[tuple[str, int, int]],
                    rows: list[list]) -> int:
        req = encode_request(
            row_inserts=[encode_row_insert(table, schema, rows)],
            dbname=self.dbname)
        affected, code, err = decode_response(self._handle(req))
        if code:
            raise RuntimeError(err)
        return affected

    def sql(self, q: str) -> int:
        affected, code, err = decode_response(
            self._handle(encode_request(sql=q, dbname=self.dbname)))
        if code:
            raise RuntimeError(err)
        return affected

    def health(self) -> bool:
        resp = self._health(b"")
        return any(f == 1 and v == 1 for f, _w, v in pb.fields(resp))

    def close(self):
        self.channel.close()
