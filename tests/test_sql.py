"""SQL parsing + execution correctness (CPU reference ops path)."""

import numpy as np
import pytest

from greptimedb_amd.engine.engine import EngineConfig, MitoEngine
from greptimedb_amd.engine.ingest import Ingestor
from greptimedb_amd.models.tsbs import CpuWorkload
from greptimedb_amd.query.executor import Executor
from greptimedb_amd.query.parser import parse_sql
from greptimedb_amd.utils.errors import InvalidSyntax


@pytest.fixture
def ex(tmp_engine):
    ing = Ingestor(tmp_engine)
    w = CpuWorkload(scale=10)
    for _ in range(10):
        ing.ingest_lines(w.next_batch(1000))
    return Executor(tmp_engine)


def test_parse_errors():
    with pytest.raises(InvalidSyntax):
        parse_sql("SELEC x")
    with pytest.raises(InvalidSyntax):
        parse_sql("SELECT FROM t WHERE")


def test_count_star(ex):
    assert ex.execute("SELECT count(*) FROM cpu").columns[0][0] == 10000


def test_constant_select(ex):
    r = ex.execute("SELECT 1 + 2 * 3")
    assert r.columns[0][0] == 7


def test_group_by_tag(ex):
    r = ex.execute("SELECT hostname, count(*) FROM cpu GROUP BY hostname ORDER BY hostname")
    assert len(r) == 10
    assert sum(r.columns[1]) == 10000


def test_time_bucket_agg_matches_raw(ex):
    r = ex.execute(
        "SELECT date_trunc('minute', ts) AS minute, max(usage_user) FROM cpu "
        "WHERE hostname = 'host_3' GROUP BY minute ORDER BY minute")
    raw = ex.execute("SELECT ts, usage_user FROM cpu WHERE hostname = 'host_3'")
    buckets = {}
    for t, v in zip(raw.columns[0], raw.columns[1]):
        b = (t // 60000) * 60000
        buckets[b] = max(buckets.get(b, -1e18), v)
    assert len(r) == len(buckets)
    for b, v in zip(r.columns[0], r.columns[1]):
        assert abs(buckets[int(b)] - v) < 1e-12


def test_all_aggs_consistency(ex):
    r = ex.execute(
        "SELECT count(usage_user), sum(usage_user), min(usage_user), "
        "max(usage_user), avg(usage_user) FROM cpu WHERE hostname IN ('host_1','host_2')")
    raw = ex.execute("SELECT usage_user FROM cpu WHERE hostname IN ('host_1','host_2')")
    vals = np.asarray(raw.columns[0], dtype=float)
    assert r.columns[0][0] == len(vals)
    assert abs(r.columns[1][0] - vals.sum()) < 1e-6
    assert r.columns[2][0] == vals.min()
    assert r.columns[3][0] == vals.max()
    assert abs(r.columns[4][0] - vals.mean()) < 1e-9


def test_residual_field_predicate(ex):
    r = ex.execute("SELECT count(*) FROM cpu WHERE usage_user > 50")
    raw = ex.execute("SELECT usage_user FROM cpu")
    exp = sum(1 for v in raw.columns[0] if v > 50)
    assert r.columns[0][0] == exp


def test_or_predicate(ex):
    r = ex.execute("SELECT count(*) FROM cpu WHERE hostname = 'host_1' OR hostname = 'host_2'")
    assert r.columns[0][0] == 2000


def test_between_and_limit(ex):
    r = ex.execute("SELECT ts, usage_user FROM cpu WHERE hostname='host_0' "
                   "ORDER BY ts LIMIT 5")
    assert len(r) == 5
    assert list(r.columns[0]) == sorted(r.columns[0])


def test_order_desc(ex):
    r = ex.execute("SELECT ts FROM cpu WHERE hostname='host_0' ORDER BY ts DESC LIMIT 3")
    assert list(r.columns[0]) == sorted(r.columns[0], reverse=True)


def test_having(ex):
    r = ex.execute("SELECT hostname, count(*) FROM cpu GROUP BY hostname "
                   "HAVING count(*) > 0 ORDER BY hostname")
    assert len(r) == 10


def test_double_groupby_shape(ex):
    """TSBS double-groupby-1: mean per (hour, hostname)."""
    r = ex.execute(
        "SELECT date_trunc('hour', ts) AS hour, hostname, avg(usage_user) "
        "FROM cpu GROUP BY hour, hostname ORDER BY hour, hostname")
    assert len(r) >= 10
    raw = ex.execute("SELECT ts, hostname, usage_user FROM cpu")
    acc = {}
    for t, h, v in zip(*raw.columns):
        k = ((t // 3600000) * 3600000, h)
        acc.setdefault(k, []).append(v)
    for hr, h, m in zip(*r.columns):
        assert abs(np.mean(acc[(int(hr), h)]) - m) < 1e-9
    assert len(r) == len(acc)


def test_show_describe_drop(ex):
    assert "cpu" in ex.execute("SHOW TABLES").columns[0]
    d = ex.execute("DESCRIBE cpu")
    assert "hostname" in d.columns[0] and "usage_user" in d.columns[0]
    ex.execute("CREATE TABLE tmp1 (a STRING, ts TIMESTAMP TIME INDEX, PRIMARY KEY(a))")
    assert "tmp1" in ex.execute("SHOW TABLES").columns[0]
    ex.execute("DROP TABLE tmp1")
    assert "tmp1" not in ex.execute("SHOW TABLES").columns[0]


def test_insert_select_roundtrip(tmp_engine):
    ex = Executor(tmp_engine)
    ex.execute("CREATE TABLE kv (k STRING, ts TIMESTAMP TIME INDEX, v DOUBLE, PRIMARY KEY (k))")
    ex.execute("INSERT INTO kv (k, ts, v) VALUES ('x', 1000, 1.5), ('y', 2000, 2.5)")
    r = ex.execute("SELECT k, ts, v FROM kv ORDER BY ts")
    assert [tuple(t) for t in r.rows()] == [("x", 1000, 1.5), ("y", 2000, 2.5)]


def test_agg_after_flush_equals_before(ex, tmp_engine):
    before = ex.execute("SELECT hostname, avg(usage_user) FROM cpu GROUP BY hostname ORDER BY hostname")
    tmp_engine.flush_all()
    after = ex.execute("SELECT hostname, avg(usage_user) FROM cpu GROUP BY hostname ORDER BY hostname")
    np.testing.assert_allclose(np.asarray(before.columns[1], dtype=float),
                               np.asarray(after.columns[1], dtype=float), rtol=1e-12)
