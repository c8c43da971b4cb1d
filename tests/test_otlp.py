"""OTLP trace ingestion: native protobuf parse → trace table → SQL."""

import struct

import numpy as np
import pytest
import torch  # noqa: F401

from greptimedb_amd import _native
from greptimedb_amd.engine.tracestore import TraceStore
from greptimedb_amd.query.executor import Executor


def _v(x):
    out = b""
    while True:
        b7 = x & 0x7F
        x >>= 7
        out += bytes([b7 | (0x80 if x else 0)])
        if not x:
            return out


def _ld(f, payload):
    return _v((f << 3) | 2) + _v(len(payload)) + payload


def _s(f, s):
    return _ld(f, s.encode() if isinstance(s, str) else s)


def _fixed64(f, x):
    return _v((f << 3) | 1) + struct.pack("<Q", x)


def _anyvalue_str(s):
    return _s(1, s)


def _kv(k, v_payload):
    return _ld(1, _s(1, k) + _ld(2, v_payload))


def make_span(trace_id, span_id, name, start_ns, end_ns, attrs=(), status=0,
              parent=b""):
    body = _s(1, trace_id) + _s(2, span_id)
    if parent:
        body += _s(4, parent)
    body += _s(5, name)
    body += _fixed64(7, start_ns) + _fixed64(8, end_ns)
    for k, v in attrs:
        body += _v((9 << 3) | 2) + _v(len(_s(1, k) + _ld(2, _anyvalue_str(v)))) + \
            _s(1, k) + _ld(2, _anyvalue_str(v))
    if status:
        body += _ld(15, _v(2 << 3) + _v(status))
    return _ld(2, body)  # Span is field 2 of ScopeSpans


def make_request(service, spans):
    resource = _ld(1, _kv("service.name", _anyvalue_str(service)))
    scope_spans = _ld(2, b"".join(spans))
    return _ld(1, resource + scope_spans)  # ResourceSpans is field 1 of the request


def test_otlp_parse():
    p = _native.OtlpTraceParser()
    req = make_request("svcA", [
        make_span(b"\x01" * 16, b"\x02" * 8, "GET /x", 1_000_000_000,
                  1_500_000_000, attrs=[("http.method", "GET")]),
        make_span(b"\x01" * 16, b"\x03" * 8, "db.query", 1_100_000_000,
                  1_200_000_000, status=2, parent=b"\x02" * 8),
    ])
    s, st, dur, status, tids, sids, pids, attrs, new = p.parse(req)
    assert len(s) == 2 and len(new) == 2
    assert new[0][1] == "svcA" and new[0][2] == "GET /x"
    assert tids[0] == "01" * 16 and sids[1] == "03" * 8
    assert pids[1] == "02" * 8
    assert dur[0] == 500.0 and dur[1] == 100.0
    assert status[1] == 2
    assert '"http.method":"GET"' in attrs[0]


def test_trace_store_end_to_end(tmp_engine):
    store = TraceStore(tmp_engine)
    req = make_request("svcA", [
        make_span(b"\xaa" * 16, b"\x01" * 8, "GET /users", 10_000_000_000,
                  10_250_000_000),
        make_span(b"\xaa" * 16, b"\x02" * 8, "SELECT", 10_050_000_000,
                  10_100_000_000, parent=b"\x01" * 8),
    ]) + make_request("svcB", [
        make_span(b"\xbb" * 16, b"\x03" * 8, "POST /o", 11_000_000_000,
                  11_400_000_000, status=2),
    ])
    n = store.write(req)
    assert n == 3
    ex = Executor(tmp_engine)
    r = ex.execute("SELECT service_name, span_name, duration_ms, trace_id "
                   "FROM opentelemetry_traces ORDER BY ts")
    assert len(r) == 3
    assert r.rows()[0][:3] == ("svcA", "GET /users", 250.0)
    assert r.rows()[0][3] == "aa" * 16
    r = ex.execute("SELECT count(*) FROM opentelemetry_traces WHERE status_code >= 2")
    assert r.columns[0][0] == 1
    # spans of one trace
    r = ex.execute(f"SELECT span_name FROM opentelemetry_traces "
                   f"WHERE trace_id = '{'aa'*16}' ORDER BY ts")
    assert len(r) == 2


def test_otlp_http_endpoint(tmp_engine):
    pytest.importorskip("fastapi")
    from fastapi.testclient import TestClient
    from greptimedb_amd.servers.http import ServerContext, build_app
    client = TestClient(build_app(ServerContext(tmp_engine)))
    req = make_request("svcZ", [make_span(b"\x05" * 16, b"\x06" * 8, "op",
                                          1_000_000_000, 2_000_000_000)])
    r = client.post("/v1/otlp/v1/traces", content=req)
    assert r.status_code == 200
    r = client.get("/v1/sql", params={"sql":
        "SELECT count(*) FROM opentelemetry_traces"})
    assert r.json()["output"][0]["records"]["rows"][0][0] == 1


def _anyvalue_double(d):
    return _v((4 << 3) | 1) + struct.pack("<d", d)


def test_otlp_metrics_endpoint(tmp_engine):
    pytest.importorskip("fastapi")
    from fastapi.testclient import TestClient
    from greptimedb_amd.servers.http import ServerContext, build_app
    client = TestClient(build_app(ServerContext(tmp_engine)))
    # Metric{name="req_total", sum{data_points=[{attrs{job=a}, t, as_double}]}}
    dp = (_ld(7, _s(1, "job") + _ld(2, _anyvalue_str("a"))) +
          _v((3 << 3) | 1) + struct.pack("<Q", 2_000_000_000) +
          _v((4 << 3) | 1) + struct.pack("<d", 42.0))
    metric = _s(1, "req_total") + _ld(7, _ld(1, dp))
    scope_metrics = _ld(2, _ld(2, metric))
    resource = _ld(1, _kv("service.name", _anyvalue_str("svcM")))
    req = _ld(1, resource + scope_metrics)
    r = client.post("/v1/otlp/v1/metrics", content=req)
    assert r.status_code == 200
    r = client.get("/v1/prometheus/api/v1/query",
                   params={"query": 'req_total{job="a"}', "time": "3"})
    res = r.json()["data"]["result"]
    assert len(res) == 1 and res[0]["value"][1] == "42.0"
    assert res[0]["metric"]["service.name"] == "svcM"


def test_otlp_logs_endpoint(tmp_engine):
    pytest.importorskip("fastapi")
    from fastapi.testclient import TestClient
    from greptimedb_amd.servers.http import ServerContext, build_app
    client = TestClient(build_app(ServerContext(tmp_engine)))
    # LogRecord{time, severity_text="ERROR", body="db conn failed", attrs{k8s.pod=p1}}
    rec = (_v((1 << 3) | 1) + struct.pack("<Q", 5_000_000_000) +
           _s(3, "ERROR") + _ld(5, _anyvalue_str("db conn failed")) +
           _ld(6, _s(1, "k8s.pod") + _ld(2, _anyvalue_str("p1"))))
    # ResourceLogs{scope_logs=2 → ScopeLogs{log_records=2}}
    req = _ld(1, _ld(2, _ld(2, rec)))
    r = client.post("/v1/otlp/v1/logs", content=req)
    assert r.status_code == 200
    r = client.get("/v1/sql", params={"sql":
        "SELECT severity, message FROM opentelemetry_logs WHERE matches(message, 'failed')"})
    rows = r.json()["output"][0]["records"]["rows"]
    assert rows == [["ERROR", "db conn failed"]]
