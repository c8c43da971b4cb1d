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


def test_auth_challenge_response():
    """mysql_native_password + pg md5 verification math."""
    import hashlib
    from greptimedb_amd.servers.auth import (StaticUserProvider,
                                             mysql_native_check, password_of,
                                             pg_md5_check)
    prov = StaticUserProvider({"kim": "s3cret"})
    assert password_of(prov, "kim") == "s3cret"
    assert password_of(prov, "nope") is None
    scr = b"12345678123456789012"
    h1 = hashlib.sha1(b"s3cret").digest()
    h2 = hashlib.sha1(h1).digest()
    token = bytes(a ^ b for a, b in zip(h1, hashlib.sha1(scr + h2).digest()))
    assert mysql_native_check("s3cret", scr, token)
    assert not mysql_native_check("wrong", scr, token)
    salt = b"\x9a\x17\x2e\x41"
    inner = hashlib.md5(b"s3cretkim").hexdigest()
    resp = "md5" + hashlib.md5(inner.encode() + salt).hexdigest()
    assert pg_md5_check("s3cret", "kim", salt, resp)
    assert not pg_md5_check("other", "kim", salt, resp)


def test_http_basic_auth(tmp_engine):
    import base64
    import pytest as _pytest
    _pytest.importorskip("fastapi")
    from fastapi.testclient import TestClient
    from greptimedb_amd.servers.auth import StaticUserProvider
    from greptimedb_amd.servers.http import ServerContext, build_app
    ctx = ServerContext(tmp_engine,
                        user_provider=StaticUserProvider({"kim": "pw"}))
    client = TestClient(build_app(ctx))
    assert client.get("/health").status_code == 200        # probes open
    r = client.get("/v1/sql", params={"sql": "SELECT 1"})
    assert r.status_code == 401
    hdr = {"Authorization": "Basic " + base64.b64encode(b"kim:pw").decode()}
    r = client.get("/v1/sql", params={"sql": "SELECT 1"}, headers=hdr)
    assert r.status_code == 200
    bad = {"Authorization": "Basic " + base64.b64encode(b"kim:no").decode()}
    assert client.get("/v1/sql", params={"sql": "SELECT 1"},
                      headers=bad).status_code == 401
