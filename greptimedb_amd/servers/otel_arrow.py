"""OTel-Arrow (OTAP) gRPC service: streaming Arrow-encoded metrics.

Reference parity: src/servers/src/otel_arrow.rs — ArrowMetricsService
(opentelemetry.proto.experimental.arrow.v1): the client streams
BatchArrowRecords {batch_id, arrow_payloads[{schema_id, type, record}]}
where each record is an Arrow IPC stream; the server acks every batch with
BatchStatus {batch_id, status_code}. The reference feeds batches through
the otel-arrow Consumer into the metric engine; here the service decodes
the IPC payloads with pyarrow and ingests rows into the PromStore (metric
engine multiplexing).

Message layout (hand-encoded with utils/pb.py; grpcio-tools absent):
  BatchArrowRecords { int64 batch_id=1; repeated ArrowPayload
                      arrow_payloads=2; bytes headers=3 }
  ArrowPayload { string schema_id=1; ArrowPayloadType type=2;
                 bytes record=3 }
  BatchStatus  { int64 batch_id=1; int32 status_code=2;
                 string status_message=3 }

Payload subset: flattened univariate metrics — an IPC batch with columns
`metric` (utf8), a timestamp column (`ts`/`time_unix_nano`), `value`
(float), and any further utf8 columns as labels. (The full OTAP
multi-payload layout normalizes resources/scopes into side tables; the
flattened form carries the same information for the metric engine.)
"""

from __future__ import annotations

from concurrent import futures

import grpc
import numpy as np

from greptimedb_amd.utils import pb

METHOD_ARROW_METRICS = ("/opentelemetry.proto.experimental.arrow.v1."
                        "ArrowMetricsService/ArrowMetrics")


def decode_batch_arrow_records(buf: bytes):
    batch_id = 0
    payloads = []
    for field, _w, v in pb.fields(buf):
        if field == 1:
            batch_id = pb.as_i64(v)
        elif field == 2:
            schema_id, ptype, record = "", 0, b""
            for f2, _w2, v2 in pb.fields(v):
                if f2 == 1:
                    schema_id = v2.decode()
                elif f2 == 2:
                    ptype = v2
                elif f2 == 3:
                    record = v2
            payloads.append((schema_id, ptype, record))
    return batch_id, payloads


def encode_batch_arrow_records(batch_id: int, payloads) -> bytes:
    w = pb.Writer().varint(1, batch_id)
    for schema_id, ptype, record in payloads:
        p = pb.Writer().string(1, schema_id).varint(2, ptype).bytes(3, record)
        w.msg(2, p)
    return w.build()


def encode_batch_status(batch_id: int, code: int = 0, msg: str = "") -> bytes:
    w = pb.Writer().varint(1, batch_id).varint(2, code)
    if msg:
        w.string(3, msg)
    return w.build()


def decode_batch_status(buf: bytes):
    bid, code, msg = 0, 0, ""
    for field, _w, v in pb.fields(buf):
        if field == 1:
            bid = pb.as_i64(v)
        elif field == 2:
            code = v
        elif field == 3:
            msg = v.decode()
    return bid, code, msg


class OtelArrowServer:
    def __init__(self, engine, host: str = "127.0.0.1", port: int = 0):
        from greptimedb_amd.engine.promstore import PromStore
        self.engine = engine
        self.store = PromStore(engine)
        self._server = grpc.server(futures.ThreadPoolExecutor(max_workers=4))
        self._server.add_generic_rpc_handlers((_Handlers(self),))
        self.port = self._server.add_insecure_port(f"{host}:{port}")
        self._server.start()

    def shutdown(self):
        self._server.stop(grace=1)

    def _ingest_payload(self, record: bytes) -> int:
        import pyarrow as pa
        import pyarrow.ipc as ipc
        reader = ipc.open_stream(pa.py_buffer(record))
        table = reader.read_all()
        names = table.column_names
        ts_col = next((c for c in ("ts", "time_unix_nano", "timestamp")
                       if c in names), None)
        if ts_col is None or "metric" not in names or "value" not in names:
            raise ValueError("payload needs metric/ts/value columns")
        col = table.column(ts_col)
        if pa.types.is_timestamp(col.type):
            ts_ms = col.cast(pa.timestamp("ms")).cast(pa.int64()) \
                .to_numpy(zero_copy_only=False)
        else:
            ts = col.cast(pa.int64()).to_numpy(zero_copy_only=False)
            ts_ms = ts // 1_000_000 if ts_col == "time_unix_nano" else ts
        metric = table.column("metric").to_pylist()
        value = table.column("value").cast(pa.float64()) \
            .to_numpy(zero_copy_only=False)
        label_cols = {c: table.column(c).to_pylist() for c in names
                      if c not in (ts_col, "metric", "value") and
                      pa.types.is_string(table.schema.field(c).type)}
        n = len(ts_ms)
        points = []
        for i in range(n):
            tags = {ln: vals[i] for ln, vals in label_cols.items()
                    if vals[i] is not None}
            points.append((metric[i], tags, int(ts_ms[i]), float(value[i])))
        self.store.write_points(points)
        return n

    def handle_stream(self, request_iter, context):
        for req in request_iter:
            batch_id, payloads = decode_batch_arrow_records(req)
            try:
                total = 0
                for _sid, _ptype, record in payloads:
                    if record:
                        total += self._ingest_payload(record)
                yield encode_batch_status(batch_id, 0, f"rows={total}")
            except Exception as e:
                yield encode_batch_status(batch_id, 13, f"{type(e).__name__}: {e}")


class _Handlers(grpc.GenericRpcHandler):
    def __init__(self, server: OtelArrowServer):
        self.server = server

    def service(self, details):
        if details.method == METHOD_ARROW_METRICS:
            return grpc.stream_stream_rpc_method_handler(
                lambda it, ctx: self.server.handle_stream(it, ctx))
        return None
