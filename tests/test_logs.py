"""Log tables, fulltext MATCHES, log ingestion endpoints."""

import numpy as np
import pytest

from greptimedb_amd.engine.engine import EngineConfig, MitoEngine
from greptimedb_amd.engine.logstore import LogStore
from greptimedb_amd.query.executor import Executor


@pytest.fixture
def log_env(tmp_engine):
    ex = Executor(tmp_engine)
    ex.execute("CREATE TABLE logs (service STRING, ts TIMESTAMP TIME INDEX, "
               "message STRING, latency DOUBLE, PRIMARY KEY (service)) "
               "WITH ('append_mode'='true')")
    ex.execute("""INSERT INTO logs (service, ts, message, latency) VALUES
     ('api', 1000, 'GET /users returned 200 OK', 1.5),
     ('api', 2000, 'POST /login failed with timeout error', 30.0),
     ('db', 3000, 'connection pool exhausted error', 0.0),
     ('api', 4000, 'GET /users returned 200 OK', 2.0)""")
    return tmp_engine, ex


def test_matches_memtable(log_env):
    _, ex = log_env
    r = ex.execute("SELECT ts FROM logs WHERE matches(message, 'error') ORDER BY ts")
    assert list(r.columns[0]) == [2000, 3000]
    r = ex.execute("SELECT count(*) FROM logs WHERE matches(message, 'error timeout')")
    assert r.columns[0][0] == 1
    r = ex.execute("SELECT count(*) FROM logs WHERE matches(message, 'nosuchterm')")
    assert r.columns[0][0] == 0


def test_matches_after_flush_and_reopen(log_env, tmp_path):
    eng, ex = log_env
    eng.flush_all()
    r = ex.execute("SELECT message FROM logs WHERE matches(message, 'exhausted')")
    assert list(r.columns[0]) == ["connection pool exhausted error"]
    eng.close()
    eng2 = MitoEngine(EngineConfig(data_dir=eng.config.data_dir, device="cpu",
                                   background_flush=False))
    ex2 = Executor(eng2)
    r = ex2.execute("SELECT count(*) FROM logs WHERE matches(message, 'error')")
    assert r.columns[0][0] == 2
    r = ex2.execute("SELECT service, message, latency FROM logs ORDER BY ts LIMIT 1")
    assert r.rows()[0] == ("api", "GET /users returned 200 OK", 1.5)
    eng2.close()


def test_matches_mixed_predicates(log_env):
    _, ex = log_env
    r = ex.execute("SELECT count(*) FROM logs WHERE matches(message, 'error') "
                   "AND latency > 1 AND service = 'api'")
    assert r.columns[0][0] == 1


def test_matches_across_memtable_and_sst(log_env):
    eng, ex = log_env
    eng.flush_all()
    ex.execute("INSERT INTO logs (service, ts, message, latency) VALUES "
               "('api', 5000, 'disk error on volume 3', 9.0)")
    r = ex.execute("SELECT count(*) FROM logs WHERE matches(message, 'error')")
    assert r.columns[0][0] == 3


def test_logstore_json_ingest(tmp_engine):
    ls = LogStore(tmp_engine)
    n = ls.ingest("app_logs", [
        {"timestamp": 1000, "service": "web", "msg": "user alice logged in",
         "bytes": 512},
        {"timestamp": 2000, "service": "web", "msg": "user bob failed login",
         "bytes": 128, "extra": {"region": "eu"}},
    ], tag_keys=["service"])
    assert n == 2
    ex = Executor(tmp_engine)
    r = ex.execute("SELECT ts, msg, bytes FROM app_logs ORDER BY ts")
    assert len(r) == 2 and r.columns[2][0] == 512.0
    r = ex.execute("SELECT count(*) FROM app_logs WHERE matches(msg, 'failed login')")
    assert r.columns[0][0] == 1
    # nested key flattened
    r = ex.execute('SELECT "extra.region" FROM app_logs WHERE matches(msg, \'bob\')')
    assert list(r.columns[0]) == ["eu"]


def test_loki_push(tmp_engine):
    ls = LogStore(tmp_engine)
    n = ls.ingest_loki({"streams": [
        {"stream": {"app": "nginx", "env": "prod"},
         "values": [["1000000000", "GET / 200"], ["2000000000", "GET /x 404 not found"]]},
    ]})
    assert n == 2
    ex = Executor(tmp_engine)
    r = ex.execute("SELECT app, line FROM loki_logs WHERE matches(line, '404') ORDER BY ts")
    assert r.rows() == [("nginx", "GET /x 404 not found")]


def test_http_log_endpoints(tmp_engine):
    pytest.importorskip("fastapi")
    from fastapi.testclient import TestClient
    from greptimedb_amd.servers.http import ServerContext, build_app
    client = TestClient(build_app(ServerContext(tmp_engine)))
    r = client.post("/v1/events/logs", params={"table": "evt", "tag_keys": "svc"},
                    json=[{"timestamp": 1000, "svc": "a", "message": "hello world"}])
    assert r.json()["rows"] == 1
    r = client.post("/v1/loki/api/v1/push", json={"streams": [
        {"stream": {"app": "x"}, "values": [["1000000000", "log line one"]]}]})
    assert r.status_code == 204
    r = client.get("/v1/sql", params={"sql":
        "SELECT count(*) FROM evt WHERE matches(message, 'hello')"})
    assert r.json()["output"][0]["records"]["rows"][0][0] == 1


def test_fulltext_survives_unflushed_reopen(tmp_path):
    """WAL-only (never flushed) log data must keep its fulltext index
    across restart (regression: text_cols layout is persisted in the
    region strcols sidecar)."""
    from greptimedb_amd.engine.engine import EngineConfig, MitoEngine
    from greptimedb_amd.engine.logstore import LogStore
    from greptimedb_amd.query.executor import Executor
    d = str(tmp_path / "lg")
    eng = MitoEngine(EngineConfig(data_dir=d, device="cpu",
                                  background_flush=False))
    LogStore(eng).ingest("lg", [
        {"message": "disk failure on node7", "timestamp": 1000},
        {"message": "all good", "timestamp": 2000}])
    eng.close()
    eng2 = MitoEngine(EngineConfig(data_dir=d, device="cpu",
                                   background_flush=False))
    r = Executor(eng2).execute(
        "SELECT count(*) FROM lg WHERE matches(message, 'failure')")
    assert int(list(r.rows())[0][0]) == 1
    eng2.close()


def test_reopen_loads_persisted_ftindex_no_retokenize(log_env, monkeypatch):
    """region reopen must load the per-SST fulltext sidecar
    (Puffin analog) instead of re-tokenizing raw strings."""
    import os

    from greptimedb_amd.engine import ftindex
    from greptimedb_amd.engine.fulltext import FulltextColumn

    eng, ex = log_env
    eng.flush_all()
    st = eng.table("logs")
    sidecars = [p for r in st.regions
                for p in os.listdir(os.path.join(r.dir, "sst"))
                if p.endswith(".ftidx")]
    assert sidecars, "flush wrote no fulltext sidecar"
    eng.close()

    calls = {"n": 0}
    real = FulltextColumn.build_segment

    def counting(self, docs, device):
        calls["n"] += 1
        return real(self, docs, device)

    monkeypatch.setattr(FulltextColumn, "build_segment", counting)
    eng2 = MitoEngine(EngineConfig(data_dir=eng.config.data_dir, device="cpu",
                                   background_flush=False))
    assert calls["n"] == 0, "reopen re-tokenized despite sidecar"
    ex2 = Executor(eng2)
    r = ex2.execute("SELECT ts FROM logs WHERE matches(message, 'error') ORDER BY ts")
    assert list(r.columns[0]) == [2000, 3000]
    r = ex2.execute("SELECT count(*) FROM logs WHERE matches(message, 'exhausted pool')")
    assert r.columns[0][0] == 1
    r = ex2.execute("SELECT count(*) FROM logs WHERE matches(message, 'nosuchterm')")
    assert r.columns[0][0] == 0
    eng2.close()


def test_ftindex_sidecar_compaction(log_env):
    """Compaction rewrites one merged sidecar and purges the inputs."""
    import os

    eng, ex = log_env
    eng.flush_all()
    ex.execute("""INSERT INTO logs (service, ts, message, latency) VALUES
     ('api', 5000, 'disk full error on volume', 9.9)""")
    eng.flush_all()
    from greptimedb_amd.engine.compaction import Compactor
    st = eng.table("logs")
    for r in st.regions:
        Compactor(trigger_file_num=2).compact_region(r)
    for r in st.regions:
        d = os.path.join(r.dir, "sst")
        pq_files = [p for p in os.listdir(d) if p.endswith(".parquet")]
        ft_files = [p for p in os.listdir(d) if p.endswith(".ftidx")]
        assert len(ft_files) == len(pq_files)
    r = ex.execute("SELECT count(*) FROM logs WHERE matches(message, 'error')")
    assert r.columns[0][0] == 3
