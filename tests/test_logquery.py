"""Log query DSL (/v1/logs) — reference src/log-query/src/log_query.rs."""

import pytest

from greptimedb_amd.query.executor import Executor
from greptimedb_amd.query.logquery import logquery_to_sql


@pytest.fixture
def log_table(tmp_engine):
    ex = Executor(tmp_engine)
    ex.execute("CREATE TABLE applogs (svc STRING, ts TIMESTAMP TIME INDEX,"
               " msg STRING, level STRING, latency DOUBLE, PRIMARY KEY (svc))"
               " WITH ('append_mode'='true')")
    ex.execute("""INSERT INTO applogs (svc, ts, msg, level, latency) VALUES
        ('api', 1000, 'request failed with timeout', 'error', 30.0),
        ('api', 2000, 'request ok', 'info', 1.0),
        ('db', 3000, 'disk error detected', 'error', 0.0),
        ('db', 4000, 'compaction done', 'info', 5.0)""")
    return tmp_engine, ex


def test_basic_time_and_contains(log_table):
    eng, ex = log_table
    sql = logquery_to_sql({
        "table": "applogs",
        "time_filter": {"start": "1970-01-01T00:00:01Z",
                        "end": "1970-01-01T00:00:04Z"},
        "filters": [{"expr": {"NamedIdent": "msg"},
                     "filters": [{"Contains": "error"}]}],
        "columns": [{"NamedIdent": "ts"}, {"NamedIdent": "msg"}],
        "limit": {"fetch": 100},
    })
    r = ex.execute(sql)
    assert list(r.columns[0]) == [3000]


def test_exact_uses_matches(log_table):
    eng, ex = log_table
    sql = logquery_to_sql({
        "table": "applogs",
        "time_filter": {},
        "filters": [{"expr": {"NamedIdent": "msg"},
                     "filters": [{"Exact": "timeout"}]}],
        "limit": {"fetch": 10},
    })
    assert "matches(msg, 'timeout')" in sql
    r = ex.execute(sql)
    assert len(r) == 1


def test_equal_prefix_limit_skip(log_table):
    eng, ex = log_table
    sql = logquery_to_sql({
        "table": "applogs",
        "time_filter": {},
        "filters": [{"expr": {"NamedIdent": "level"},
                     "filters": [{"Equal": "info"}]},
                    {"expr": {"NamedIdent": "msg"},
                     "filters": [{"Prefix": "request"}]}],
        "limit": {"fetch": 5, "skip": 0},
    })
    r = ex.execute(sql)
    assert len(r) == 1


def test_http_endpoint(log_table):
    eng, ex = log_table
    from fastapi.testclient import TestClient
    from greptimedb_amd.servers.http import ServerContext, build_app
    app = build_app(ServerContext(eng))
    cli = TestClient(app)
    resp = cli.post("/v1/logs", json={
        "table": "applogs",
        "time_filter": {},
        "filters": [{"expr": {"NamedIdent": "msg"},
                     "filters": [{"Contains": "request"}]}],
        "columns": [{"NamedIdent": "ts"}, {"NamedIdent": "svc"}],
        "limit": {"fetch": 10},
    })
    assert resp.status_code == 200
    body = resp.json()
    assert "sql" in body
    rows = body["output"][0]["records"]["rows"]
    assert len(rows) == 2


def test_prof_endpoints(log_table):
    eng, ex = log_table
    from fastapi.testclient import TestClient
    from greptimedb_amd.servers.http import ServerContext, build_app
    app = build_app(ServerContext(eng))
    cli = TestClient(app)
    r = cli.get("/debug/prof/cpu?seconds=1&frequency=50")
    assert r.status_code == 200
    r2 = cli.get("/debug/prof/mem")
    assert r2.status_code == 200
