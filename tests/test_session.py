"""Session / QueryContext: USE, SET, schemas, timezone buckets.

Reference parity: src/session QueryContext + USE/SET statements.
"""

import pytest

from greptimedb_amd.query.executor import Executor


def test_create_use_database(tmp_engine):
    ex = Executor(tmp_engine)
    ex.execute("CREATE DATABASE metrics_db")
    dbs = ex.execute("SHOW DATABASES").columns[0]
    assert "metrics_db" in list(dbs)
    ex.execute("USE metrics_db")
    ex.execute("CREATE TABLE m1 (h STRING, ts TIMESTAMP TIME INDEX, v DOUBLE,"
               " PRIMARY KEY (h))")
    assert "metrics_db.m1" in tmp_engine.tables   # flat-namespace key
    ex.execute("INSERT INTO m1 (h, ts, v) VALUES ('a', 1, 2.0)")
    assert ex.execute("SELECT count(*) FROM m1").rows() == [(1.0,)]
    # fully qualified access from another session
    ex2 = Executor(tmp_engine)
    r = ex2.execute("SELECT v FROM metrics_db.m1")
    assert r.rows() == [(2.0,)]
    # back to public: the bare name no longer resolves
    ex.execute("USE public")
    from greptimedb_amd.utils.errors import GreptimeError
    with pytest.raises(GreptimeError):
        ex.execute("SELECT * FROM m1")


def test_use_unknown_database_fails(tmp_engine):
    from greptimedb_amd.utils.errors import GreptimeError
    with pytest.raises(GreptimeError):
        Executor(tmp_engine).execute("USE no_such_db")


def test_drop_database_drops_tables(tmp_engine):
    ex = Executor(tmp_engine)
    ex.execute("CREATE DATABASE tmpdb")
    ex.execute("USE tmpdb")
    ex.execute("CREATE TABLE t1 (h STRING, ts TIMESTAMP TIME INDEX, v DOUBLE,"
               " PRIMARY KEY (h))")
    ex.execute("USE public")
    ex.execute("DROP DATABASE tmpdb")
    assert "tmpdb.t1" not in tmp_engine.tables
    assert "tmpdb" not in ex.execute("SHOW DATABASES").columns[0]


def test_set_and_show_variables(tmp_engine):
    ex = Executor(tmp_engine)
    ex.execute("SET time_zone = '+08:00'")
    assert ex.session.tz_offset_ms == 8 * 3600 * 1000
    r = ex.execute("SHOW VARIABLES LIKE 'time_zone'")
    assert r.rows() == [("time_zone", "+08:00")]


def test_timezone_shifts_day_truncation(tmp_engine):
    ex = Executor(tmp_engine)
    ex.execute("CREATE TABLE tz (h STRING, ts TIMESTAMP TIME INDEX, v DOUBLE,"
               " PRIMARY KEY (h))")
    # 2016-01-01T20:00Z and 2016-01-02T04:00Z: same UTC day? no — different;
    # in +08:00 both fall on 2016-01-02 local
    t1 = 1451678400000   # 2016-01-01T20:00:00Z
    t2 = 1451707200000   # 2016-01-02T04:00:00Z
    ex.execute(f"INSERT INTO tz (h, ts, v) VALUES ('a', {t1}, 1.0),"
               f" ('a', {t2}, 2.0)")
    r = ex.execute("SELECT date_trunc('day', ts) AS d, count(*) AS c FROM tz"
                   " GROUP BY d ORDER BY d")
    assert [int(c) for c in r.columns[1]] == [1, 1]   # UTC: split days
    ex.execute("SET time_zone = '+08:00'")
    r = ex.execute("SELECT date_trunc('day', ts) AS d, count(*) AS c FROM tz"
                   " GROUP BY d ORDER BY d")
    assert [int(c) for c in r.columns[1]] == [2]      # local: same day


def test_schema_survives_restart(tmp_path):
    from greptimedb_amd.engine.engine import EngineConfig, MitoEngine
    d = str(tmp_path / "data")
    eng = MitoEngine(EngineConfig(data_dir=d, device="cpu",
                                  background_flush=False))
    ex = Executor(eng)
    ex.execute("CREATE DATABASE persisted")
    eng.close()
    eng2 = MitoEngine(EngineConfig(data_dir=d, device="cpu",
                                   background_flush=False))
    assert "persisted" in eng2.schemas
    eng2.close()
