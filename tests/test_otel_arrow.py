"""OTel-Arrow (OTAP) streaming metrics service.

Reference parity: src/servers/src/otel_arrow.rs (ArrowMetricsService over
BatchArrowRecords / BatchStatus).
"""

import grpc
import numpy as np
import pyarrow as pa
import pyarrow.ipc as ipc
import pytest

from greptimedb_amd.servers.otel_arrow import (METHOD_ARROW_METRICS,
                                               OtelArrowServer,
                                               decode_batch_status,
                                               encode_batch_arrow_records)


def _ipc_bytes(table: pa.Table) -> bytes:
    sink = pa.BufferOutputStream()
    with ipc.new_stream(sink, table.schema) as w:
        w.write_table(table)
    return sink.getvalue().to_pybytes()


def test_otel_arrow_stream_ingest(tmp_engine):
    srv = OtelArrowServer(tmp_engine)
    chan = grpc.insecure_channel(f"127.0.0.1:{srv.port}")
    stream = chan.stream_stream(METHOD_ARROW_METRICS)

    t0 = 1_600_000_000_000
    tbl = pa.table({
        "metric": pa.array(["http_requests_total"] * 4 + ["cpu_usage"] * 2),
        "ts": pa.array(np.arange(6, dtype=np.int64) * 1000 + t0,
                       type=pa.int64()).cast(pa.timestamp("ms")),
        "value": pa.array([1.0, 2.0, 3.0, 4.0, 0.5, 0.7]),
        "job": pa.array(["api", "api", "db", "db", "api", "api"]),
    })
    reqs = [encode_batch_arrow_records(7, [("s0", 9, _ipc_bytes(tbl))]),
            encode_batch_arrow_records(8, [("s0", 9, _ipc_bytes(tbl.slice(0, 2)))])]
    statuses = [decode_batch_status(resp) for resp in stream(iter(reqs))]
    assert [s[0] for s in statuses] == [7, 8]
    assert all(s[1] == 0 for s in statuses), statuses
    # data queryable through PromQL (metric engine multiplexing)
    from greptimedb_amd.query.promql.eval import PromEvaluator
    ev = PromEvaluator(tmp_engine)
    m = ev.query_range('sum(http_requests_total)', t0 / 1000 + 3, t0 / 1000 + 3, 1)
    assert float(m.values[0][-1]) > 0
    m2 = ev.query_range('cpu_usage{job="api"}', t0 / 1000 + 5, t0 / 1000 + 5, 1)
    assert m2.S == 1
    chan.close()
    srv.shutdown()


def test_otel_arrow_bad_payload_status(tmp_engine):
    srv = OtelArrowServer(tmp_engine)
    chan = grpc.insecure_channel(f"127.0.0.1:{srv.port}")
    stream = chan.stream_stream(METHOD_ARROW_METRICS)
    bad = pa.table({"nope": pa.array([1.0])})
    reqs = [encode_batch_arrow_records(3, [("s0", 9, _ipc_bytes(bad))])]
    statuses = [decode_batch_status(r) for r in stream(iter(reqs))]
    assert statuses[0][0] == 3 and statuses[0][1] != 0
    chan.close()
    srv.shutdown()
