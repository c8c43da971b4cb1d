"""OpenTSDB / Elasticsearch bulk / Splunk HEC / Jaeger endpoints."""

import pytest

pytest.importorskip("fastapi")
from fastapi.testclient import TestClient

from greptimedb_amd.servers.http import ServerContext, build_app
from tests.test_otlp import make_request, make_span


@pytest.fixture
def client(tmp_engine):
    return TestClient(build_app(ServerContext(tmp_engine)))


def test_opentsdb_put_and_promql(client):
    r = client.post("/v1/opentsdb/api/put", json=[
        {"metric": "sys.cpu.user", "timestamp": 1451606400,
         "value": 42.5, "tags": {"host": "web01"}},
        {"metric": "sys.cpu.user", "timestamp": 1451606410,
         "value": 43.0, "tags": {"host": "web02"}},
    ])
    assert r.status_code == 204
    # dotted OpenTSDB names are addressable via the __name__ matcher
    r = client.get("/v1/prometheus/api/v1/query",
                   params={"query": '{__name__="sys.cpu.user", host="web01"}',
                           "time": str(1451606420)})
    res = r.json()["data"]["result"]
    assert len(res) == 1 and res[0]["value"][1] == "42.5"


def test_es_bulk(client):
    body = (b'{"index":{"_index":"weblogs"}}\n'
            b'{"@timestamp": 1000, "message": "GET / ok", "status": 200}\n'
            b'{"index":{}}\n'
            b'{"@timestamp": 2000, "message": "GET /x failed badly", "status": 500}\n')
    r = client.post("/v1/elasticsearch/weblogs/_bulk", content=body)
    assert not r.json()["errors"]
    r = client.get("/v1/sql", params={"sql":
        "SELECT count(*) FROM weblogs WHERE matches(message, 'failed')"})
    assert r.json()["output"][0]["records"]["rows"][0][0] == 1


def test_splunk_hec(client):
    body = ('{"time": 1.5, "host": "h1", "event": {"message": "disk full on sda"}}'
            '{"time": 2.5, "host": "h2", "event": "plain text event"}')
    r = client.post("/v1/splunk/services/collector", content=body.encode())
    assert r.json()["code"] == 0
    r = client.get("/v1/sql", params={"sql":
        "SELECT count(*) FROM splunk_logs WHERE matches(message, 'disk full')"})
    assert r.json()["output"][0]["records"]["rows"][0][0] == 1


def test_jaeger_api(client):
    req = make_request("svcJ", [
        make_span(b"\xcc" * 16, b"\x01" * 8, "root-op", 5_000_000_000,
                  5_300_000_000),
        make_span(b"\xcc" * 16, b"\x02" * 8, "child-op", 5_050_000_000,
                  5_100_000_000, parent=b"\x01" * 8),
    ])
    assert client.post("/v1/otlp/v1/traces", content=req).status_code == 200
    assert "svcJ" in client.get("/v1/jaeger/api/services").json()["data"]
    ops = client.get("/v1/jaeger/api/services/svcJ/operations").json()["data"]
    assert set(ops) == {"root-op", "child-op"}
    tr = client.get(f"/v1/jaeger/api/traces/{'cc'*16}").json()
    assert tr["total"] == 1
    spans = tr["data"][0]["spans"]
    assert len(spans) == 2
    child = [s for s in spans if s["operationName"] == "child-op"][0]
    assert child["references"][0]["spanID"] == "01" * 8
    lst = client.get("/v1/jaeger/api/traces", params={"service": "svcJ"}).json()
    assert lst["total"] == 1
