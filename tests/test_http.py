"""HTTP server endpoints (fastapi TestClient; reference: servers/src/http.rs)."""

import struct

import pytest

pytest.importorskip("fastapi")
from fastapi.testclient import TestClient  # noqa: E402

from greptimedb_amd.servers.http import ServerContext, build_app  # noqa: E402


@pytest.fixture
def client(tmp_engine):
    ctx = ServerContext(tmp_engine)
    return TestClient(build_app(ctx))


def _pb_varint(v):
    out = b""
    while True:
        b7 = v & 0x7F
        v >>= 7
        out += bytes([b7 | (0x80 if v else 0)])
        if not v:
            return out


def _pb_str(fnum, s):
    s = s.encode() if isinstance(s, str) else s
    return _pb_varint((fnum << 3) | 2) + _pb_varint(len(s)) + s


def _label(n, v):
    return _pb_str(1, _pb_str(1, n) + _pb_str(2, v))


def _sample(val, ts):
    return _pb_str(2, _pb_varint((1 << 3) | 1) + struct.pack("<d", val) +
                   _pb_varint(2 << 3) + _pb_varint(ts))


def _ts_msg(labels, samples):
    return _pb_str(1, b"".join(_label(n, v) for n, v in labels) +
                   b"".join(_sample(v, t) for v, t in samples))


def test_health(client):
    assert client.get("/health").status_code == 200


def test_sql_roundtrip(client):
    r = client.post("/v1/sql", params={"sql":
        "CREATE TABLE t (h STRING, ts TIMESTAMP TIME INDEX, v DOUBLE, PRIMARY KEY (h))"})
    assert r.status_code == 200 and "output" in r.json()
    client.post("/v1/sql", params={"sql":
        "INSERT INTO t (h, ts, v) VALUES ('a', 1000, 1.5)"})
    r = client.get("/v1/sql", params={"sql": "SELECT h, ts, v FROM t"})
    body = r.json()
    rows = body["output"][0]["records"]["rows"]
    assert rows == [["a", 1000, 1.5]]
    cols = [c["name"] for c in body["output"][0]["records"]["schema"]["column_schemas"]]
    assert cols == ["h", "ts", "v"]


def test_sql_error(client):
    r = client.get("/v1/sql", params={"sql": "SELEKT 1"})
    assert "error" in r.json()


def test_influx_write_and_query(client):
    lines = b"weather,city=sf temp=13.5 1000000000\nweather,city=la temp=22.0 1000000000\n"
    r = client.post("/v1/influxdb/write", content=lines)
    assert r.status_code == 204
    r = client.get("/v1/sql", params={"sql": "SELECT city, temp FROM weather ORDER BY city"})
    rows = r.json()["output"][0]["records"]["rows"]
    assert rows == [["la", 22.0], ["sf", 13.5]]


def test_influx_precision_ms(client):
    client.post("/v1/influxdb/write", params={"precision": "ms"},
                content=b"m1,h=x v=1 1500\n")
    r = client.get("/v1/sql", params={"sql": "SELECT ts FROM m1"})
    assert r.json()["output"][0]["records"]["rows"] == [[1500]]


def test_remote_write_and_promql(client):
    req = _ts_msg([("__name__", "up"), ("job", "api")], [(1.0, 1000), (1.0, 61000)]) + \
          _ts_msg([("__name__", "up"), ("job", "db")], [(0.0, 61000)])
    r = client.post("/v1/prometheus/write", content=req,
                    headers={"content-encoding": "identity"})
    assert r.status_code == 204
    r = client.get("/v1/prometheus/api/v1/query",
                   params={"query": "up", "time": "61"})
    data = r.json()
    assert data["status"] == "success"
    assert len(data["data"]["result"]) == 2
    r = client.get("/v1/prometheus/api/v1/query_range",
                   params={"query": "sum(up)", "start": "0", "end": "120", "step": "60"})
    res = r.json()["data"]["result"]
    assert res and res[0]["metric"] == {}
    r = client.get("/v1/prometheus/api/v1/label/__name__/values")
    assert "up" in r.json()["data"]
    r = client.get("/v1/prometheus/api/v1/labels")
    assert "job" in r.json()["data"]
    r = client.get("/v1/prometheus/api/v1/series", params={"match[]": 'up{job="api"}'})
    assert r.json()["data"] == [{"job": "api", "__name__": "up"}]
    r = client.get("/v1/prometheus/api/v1/format_query",
                   params={"query": "sum( up )"})
    assert r.json() == {"status": "success", "data": "sum( up )"}
    r = client.get("/v1/prometheus/api/v1/parse_query",
                   params={"query": "rate(up[5m])"})
    d = r.json()
    assert d["status"] == "success" and d["data"]["type"] == "call"
    r = client.get("/v1/prometheus/api/v1/parse_query",
                   params={"query": "rate(up[5m"})
    assert r.json()["status"] == "error"


def test_metrics_endpoint(client):
    client.get("/v1/sql", params={"sql": "SELECT 1"})
    r = client.get("/metrics")
    assert "greptime_http_sql_requests" in r.text


def test_status(client):
    r = client.get("/status")
    assert "tables" in r.json()


def test_remote_read_roundtrip(client):
    # write two series via remote write, read back via remote read
    req = _ts_msg([("__name__", "rr_metric"), ("job", "j1")],
                  [(1.5, 1000), (2.5, 2000)]) + \
          _ts_msg([("__name__", "rr_metric"), ("job", "j2")], [(9.0, 1500)])
    r = client.post("/v1/prometheus/write", content=req,
                    headers={"content-encoding": "identity"})
    assert r.status_code == 204
    # ReadRequest: Query{start=0,end=5000,matchers=[EQ __name__ rr_metric]}
    matcher = _pb_str(2, "__name__") + _pb_str(3, "rr_metric")  # type EQ=0 default
    qbody = _pb_varint(1 << 3) + _pb_varint(0) + _pb_varint(2 << 3) + _pb_varint(5000) + \
        _pb_str(3, matcher)
    read_req = _pb_str(1, qbody)
    r = client.post("/v1/prometheus/read", content=read_req,
                    headers={"content-encoding": "identity"})
    assert r.status_code == 200
    from greptimedb_amd import _native
    body = _native.snappy_uncompress(bytes(r.content))
    # decode: results → timeseries count + sample values present
    assert body.count(b"rr_metric") == 2  # two series carry the name label
    import struct as _s
    assert _s.pack("<d", 9.0) in body and _s.pack("<d", 1.5) in body


def test_pipeline_crud_and_ingest(client):
    yml = """
version: 2
processors:
  - dissect:
      fields: [message]
      patterns:
        - '%{level} %{msg}'
  - letter: {fields: [level], method: lower}
transform:
  - fields: [level]
    type: string
    index: tag
  - fields: [msg]
    type: string
    index: fulltext
"""
    r = client.post("/v1/events/pipelines/applog", content=yml)
    assert r.status_code == 200 and r.json()["status"] == "created"
    r = client.get("/v1/events/pipelines/applog")
    assert r.json()["version"] == 2
    # dryrun shows transformed rows without writing
    r = client.post("/v1/events/pipelines/_dryrun?pipeline_name=applog",
                    json={"data": [{"message": "ERROR disk full"}]})
    row = r.json()["rows"][0]["row"]
    assert row["level"] == "error" and row["msg"] == "disk full"
    # ingest through the pipeline
    r = client.post("/v1/events/logs?table=app&pipeline_name=applog",
                    json=[{"message": "WARN low memory", "timestamp": 1000},
                          {"message": "ERROR disk full", "timestamp": 2000}])
    assert r.json()["rows"] == 2
    r = client.post("/v1/sql", params={"sql":
                    "SELECT level, msg FROM app ORDER BY ts"})
    rows = r.json()["output"][0]["records"]["rows"]
    assert rows == [["warn", "low memory"], ["error", "disk full"]]
    r = client.delete("/v1/events/pipelines/applog")
    assert r.json()["status"] == "deleted"


def test_probe_endpoints(client):
    r = client.get("/v1/prometheus/api/v1/status/buildinfo")
    assert r.json()["status"] == "success"
    assert client.get("/v1/influxdb/ping").status_code == 204
    assert client.get("/v1/influxdb/health").status_code == 204


def test_readonly_user_permission(tmp_path):
    """`user=pw:ro` users are denied DML/DDL but can query (reference:
    src/auth permission checker)."""
    import base64

    from fastapi.testclient import TestClient

    from greptimedb_amd.engine.engine import EngineConfig, MitoEngine
    from greptimedb_amd.engine.ingest import Ingestor
    from greptimedb_amd.query.executor import Executor
    from greptimedb_amd.servers.auth import StaticUserProvider
    from greptimedb_amd.servers.http import ServerContext, build_app

    upath = tmp_path / "users"
    upath.write_text("admin=secret\nviewer=pw:ro\n")
    prov = StaticUserProvider.from_file(str(upath))
    assert prov.mode("viewer") == "ro" and prov.mode("admin") == "rw"
    assert prov.allow("viewer", "pw")

    eng = MitoEngine(EngineConfig(data_dir=str(tmp_path / "d"), device="cpu",
                                  background_flush=False))
    ctx = ServerContext(eng, user_provider=prov)
    app = build_app(ctx)
    c = TestClient(app)

    def hdr(u, p):
        return {"authorization":
                "Basic " + base64.b64encode(f"{u}:{p}".encode()).decode()}

    # admin can create + insert
    r = c.post("/v1/sql", params={"sql": "CREATE TABLE pt (ts TIMESTAMP "
               "TIME INDEX, h STRING PRIMARY KEY, v DOUBLE)"},
               headers=hdr("admin", "secret"))
    assert r.status_code == 200 and "error" not in r.json()
    r = c.post("/v1/sql", params={
        "sql": "INSERT INTO pt VALUES (1000,'a',1)"},
        headers=hdr("admin", "secret"))
    assert r.status_code == 200
    # read-only user: SELECT ok, INSERT/DROP denied
    r = c.post("/v1/sql", params={"sql": "SELECT count(*) FROM pt"},
               headers=hdr("viewer", "pw"))
    assert r.status_code == 200 and "error" not in r.json()
    for bad in ("INSERT INTO pt VALUES (2000,'b',2)", "DROP TABLE pt",
                "TRUNCATE TABLE pt"):
        r = c.post("/v1/sql", params={"sql": bad}, headers=hdr("viewer", "pw"))
        assert r.status_code == 403, bad
    # data unchanged
    r = c.post("/v1/sql", params={"sql": "SELECT count(*) FROM pt"},
               headers=hdr("admin", "secret"))
    assert r.json()["output"][0]["records"]["rows"][0][0] == 1
    eng.close()


def test_dyn_log_level(client):
    import logging
    prev = logging.getLogger().level
    try:
        r = client.post("/debug/log_level", params={"level": "debug"})
        assert r.json()["level"] == "DEBUG"
        r = client.get("/debug/log_level")
        assert r.json()["level"] == "DEBUG"
        assert client.post("/debug/log_level",
                           params={"level": "nope"}).status_code == 400
    finally:
        logging.getLogger().setLevel(prev)
