"""Flow engine: continuous aggregation (batching mode)."""

import numpy as np

from greptimedb_amd.query.executor import Executor


def test_flow_create_tick_upsert(tmp_engine):
    ex = Executor(tmp_engine)
    ex.execute("CREATE TABLE src (h STRING, ts TIMESTAMP TIME INDEX, v DOUBLE, "
               "PRIMARY KEY (h)) WITH ('append_mode'='true')")
    ex.execute("CREATE FLOW f1 SINK TO sink1 AS "
               "SELECT date_trunc('minute', ts) AS minute, h, max(v), count(*) "
               "FROM src GROUP BY minute, h")
    fe = tmp_engine.flow_engine
    assert "f1" in fe.flows

    ex.execute("INSERT INTO src (h, ts, v) VALUES ('a', 1000, 1.0), "
               "('a', 2000, 5.0), ('b', 61000, 2.0)")
    out = fe.tick()
    assert out["f1"] == 2  # (minute0, a) + (minute1, b)
    r = ex.execute("SELECT h, ts, \"max(v)\", \"count(*)\" FROM sink1 ORDER BY ts, h")
    rows = r.rows()
    assert ("a", 0, 5.0, 2.0) in [tuple(x) for x in rows] or \
           ("a", 0, 5.0, 2) in [tuple(x) for x in rows]

    # incremental: new point in minute 0 updates the same sink row (upsert)
    ex.execute("INSERT INTO src (h, ts, v) VALUES ('a', 3000, 9.0)")
    fe.tick()
    r = ex.execute("SELECT \"max(v)\" FROM sink1 WHERE h = 'a' AND ts = 0")
    assert list(r.columns[0]) == [9.0]
    # no dirty data → no work
    assert fe.tick() == {}


def test_flow_show_drop(tmp_engine):
    ex = Executor(tmp_engine)
    ex.execute("CREATE TABLE s2 (h STRING, ts TIMESTAMP TIME INDEX, v DOUBLE, PRIMARY KEY (h))")
    ex.execute("CREATE FLOW f2 SINK TO sk2 AS SELECT date_trunc('hour', ts) AS hr, sum(v) FROM s2 GROUP BY hr")
    assert "f2" in ex.execute("SHOW FLOWS").columns[0]
    ex.execute("DROP FLOW f2")
    assert "f2" not in ex.execute("SHOW FLOWS").columns[0]


def test_flow_expire_after(tmp_engine):
    """EXPIRE AFTER clamps dirty windows: writes older than the TTL never
    re-aggregate (ref flow batching-mode expire)."""
    import time as _t
    from greptimedb_amd.query.executor import Executor
    ex = Executor(tmp_engine)
    ex.execute("CREATE TABLE src (h STRING, ts TIMESTAMP TIME INDEX, v DOUBLE, "
               "PRIMARY KEY (h)) WITH ('append_mode'='true')")
    ex.execute("CREATE FLOW f_exp SINK TO snk EXPIRE AFTER '1h' AS "
               "SELECT date_trunc('minute', ts) AS minute, max(v) FROM src "
               "GROUP BY minute")
    fe = tmp_engine.flow_engine
    assert fe.flows["f_exp"].expire_after_s == 3600
    now = int(_t.time() * 1000)
    old = now - 2 * 3600 * 1000          # beyond the TTL
    ex.execute(f"INSERT INTO src (h, ts, v) VALUES ('a', {old}, 99.0), "
               f"('a', {now}, 7.0)")
    out = fe.tick()
    assert out["f_exp"] >= 1
    r = ex.execute("SELECT count(*) FROM snk")
    # only the fresh window aggregated; the expired write is ignored
    assert int(list(r.rows())[0][0]) == 1
