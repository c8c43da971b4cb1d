"""CTE (WITH), UNION/EXCEPT/INTERSECT, derived tables, views.

Reference parity: src/sql statements + DataFusion CTE/setop planning
(tests/cases/standalone/common/{cte,setops,view}).
"""

import pytest

from greptimedb_amd.query.executor import Executor


@pytest.fixture
def ex(tmp_engine):
    e = Executor(tmp_engine)
    e.execute("CREATE TABLE t (h STRING, ts TIMESTAMP TIME INDEX, v DOUBLE,"
              " PRIMARY KEY (h))")
    e.execute("INSERT INTO t (h, ts, v) VALUES ('a',1,1.0),('a',2,2.0),"
              "('b',1,3.0),('b',2,4.0),('c',1,5.0)")
    return e


def test_union_dedup_and_all(ex):
    r = ex.execute("SELECT h FROM t WHERE v < 2 UNION SELECT h FROM t WHERE v > 3")
    assert sorted(x[0] for x in r.rows()) == ["a", "b", "c"]
    r = ex.execute("SELECT h FROM t WHERE v <= 2 UNION ALL SELECT h FROM t WHERE v <= 2")
    assert len(r.rows()) == 4


def test_union_order_limit_applies_to_whole(ex):
    r = ex.execute("SELECT h, v FROM t WHERE h = 'a' UNION ALL "
                   "SELECT h, v FROM t WHERE h = 'c' ORDER BY v DESC LIMIT 2")
    assert [x[0] for x in r.rows()] == ["c", "a"]


def test_except_intersect(ex):
    r = ex.execute("SELECT h FROM t EXCEPT SELECT h FROM t WHERE v > 2.5")
    assert [x[0] for x in r.rows()] == ["a"]
    r = ex.execute("SELECT h FROM t INTERSECT SELECT h FROM t WHERE v >= 5")
    assert [x[0] for x in r.rows()] == ["c"]


def test_cte_chain(ex):
    r = ex.execute(
        "WITH mx AS (SELECT h, max(v) AS mv FROM t GROUP BY h), "
        "big AS (SELECT h, mv FROM mx WHERE mv >= 4) "
        "SELECT count(*) AS c, min(mv) AS lo FROM big")
    assert r.rows() == [(2.0, 4.0)]


def test_derived_table(ex):
    r = ex.execute("SELECT avg(mv) AS a FROM "
                   "(SELECT h, max(v) AS mv FROM t GROUP BY h) sub")
    assert r.rows()[0][0] == pytest.approx((2 + 4 + 5) / 3)


def test_view_create_query_drop(ex, tmp_engine):
    ex.execute("CREATE VIEW big AS SELECT h, v FROM t WHERE v >= 3")
    r = ex.execute("SELECT h, sum(v) AS s FROM big GROUP BY h ORDER BY h")
    assert r.rows() == [("b", 7.0), ("c", 5.0)]
    # persists across restart
    from greptimedb_amd.engine.engine import EngineConfig, MitoEngine
    d = tmp_engine.config.data_dir
    tmp_engine._save_catalog()
    eng2 = MitoEngine(EngineConfig(data_dir=d, device="cpu",
                                   background_flush=False))
    assert "big" in eng2.views
    r2 = Executor(eng2).execute("SELECT count(*) FROM big")
    assert r2.rows() == [(3.0,)]
    eng2.close()
    ex.execute("DROP VIEW big")
    from greptimedb_amd.utils.errors import GreptimeError
    with pytest.raises(GreptimeError):
        ex.execute("SELECT * FROM big")


def test_view_name_conflicts(ex):
    from greptimedb_amd.utils.errors import TableAlreadyExists
    with pytest.raises(TableAlreadyExists):
        ex.execute("CREATE VIEW t AS SELECT 1")
    ex.execute("CREATE VIEW w1 AS SELECT h FROM t")
    with pytest.raises(TableAlreadyExists):
        ex.execute("CREATE VIEW w1 AS SELECT h FROM t")
    ex.execute("CREATE OR REPLACE VIEW w1 AS SELECT v FROM t")
    assert ex.execute("SELECT count(*) FROM w1").rows() == [(5.0,)]


def test_cte_shadowing_scope_restored(ex):
    # inside WITH, `x` is the CTE; outside it's gone
    r = ex.execute("WITH x AS (SELECT 99 AS v) SELECT v FROM x")
    assert r.rows() == [(99,)]
    from greptimedb_amd.utils.errors import GreptimeError
    with pytest.raises(GreptimeError):
        ex.execute("SELECT v FROM x")
