"""SQL window functions: func(...) OVER (PARTITION BY ... ORDER BY ...)
(ref: DataFusion window exprs via src/query; default-frame semantics)."""

import numpy as np
import pytest

from greptimedb_amd.query.executor import Executor


@pytest.fixture
def wex(tmp_engine):
    ex = Executor(tmp_engine)
    ex.execute("CREATE TABLE w (h STRING, ts TIMESTAMP TIME INDEX, v DOUBLE, "
               "PRIMARY KEY (h))")
    rows = []
    # host a: 10,20,30,40 ; host b: 5,5,15
    for i, v in enumerate([10.0, 20.0, 30.0, 40.0]):
        rows.append(f"('a', {1000 * (i + 1)}, {v})")
    for i, v in enumerate([5.0, 5.0, 15.0]):
        rows.append(f"('b', {1000 * (i + 1)}, {v})")
    ex.execute("INSERT INTO w (h, ts, v) VALUES " + ",".join(rows))
    return ex


def _col(r, name):
    return list(r.columns[r.names.index(name)])


def test_row_number(wex):
    r = wex.execute("SELECT h, ts, row_number() OVER (PARTITION BY h ORDER BY ts) "
                    "AS rn FROM w ORDER BY h, ts")
    assert _col(r, "rn") == [1, 2, 3, 4, 1, 2, 3]


def test_lag_lead_and_delta(wex):
    r = wex.execute("SELECT h, ts, v - lag(v) OVER (PARTITION BY h ORDER BY ts) "
                    "AS d FROM w ORDER BY h, ts")
    d = _col(r, "d")
    assert np.isnan(d[0]) and d[1:4] == [10.0, 10.0, 10.0]
    assert np.isnan(d[4]) and d[5] == 0.0 and d[6] == 10.0
    r = wex.execute("SELECT lead(v, 2, -1) OVER (PARTITION BY h ORDER BY ts) AS x "
                    "FROM w ORDER BY h, ts")
    assert _col(r, "x") == [30.0, 40.0, -1.0, -1.0, 15.0, -1.0, -1.0]


def test_running_sum_avg(wex):
    r = wex.execute("SELECT sum(v) OVER (PARTITION BY h ORDER BY ts) AS s "
                    "FROM w ORDER BY h, ts")
    assert _col(r, "s") == [10, 30, 60, 100, 5, 10, 25]
    r = wex.execute("SELECT avg(v) OVER (PARTITION BY h ORDER BY ts) AS a "
                    "FROM w ORDER BY h, ts")
    np.testing.assert_allclose(_col(r, "a"), [10, 15, 20, 25, 5, 5, 25 / 3])


def test_whole_partition_agg(wex):
    r = wex.execute("SELECT v / sum(v) OVER (PARTITION BY h) AS frac "
                    "FROM w ORDER BY h, ts")
    np.testing.assert_allclose(
        _col(r, "frac"), [0.1, 0.2, 0.3, 0.4, 0.2, 0.2, 0.6])
    r = wex.execute("SELECT max(v) OVER (PARTITION BY h) AS m FROM w ORDER BY h, ts")
    assert _col(r, "m") == [40.0] * 4 + [15.0] * 3


def test_rank_dense_rank_peers(wex):
    # host b has duplicate v=5 → peers share rank
    r = wex.execute("SELECT rank() OVER (PARTITION BY h ORDER BY v) AS rk, "
                    "dense_rank() OVER (PARTITION BY h ORDER BY v) AS dr "
                    "FROM w WHERE h = 'b' ORDER BY v, ts")
    assert _col(r, "rk") == [1, 1, 3]
    assert _col(r, "dr") == [1, 1, 2]


def test_first_last_value(wex):
    r = wex.execute("SELECT first_value(v) OVER (PARTITION BY h ORDER BY ts) AS f, "
                    "last_value(v) OVER (PARTITION BY h) AS l "
                    "FROM w ORDER BY h, ts")
    assert _col(r, "f") == [10.0] * 4 + [5.0] * 3
    assert _col(r, "l") == [40.0] * 4 + [15.0] * 3


def test_window_no_partition_running_min(wex):
    r = wex.execute("SELECT min(v) OVER (ORDER BY ts, h) AS m FROM w "
                    "ORDER BY ts, h")
    # interleaved by ts: a10,b5,a20,b5,a30,b15,a40 → running min
    assert _col(r, "m") == [10.0, 5.0, 5.0, 5.0, 5.0, 5.0, 5.0]


def test_window_desc_and_count(wex):
    r = wex.execute("SELECT row_number() OVER (PARTITION BY h ORDER BY ts DESC) "
                    "AS rn, count(*) OVER (PARTITION BY h) AS c "
                    "FROM w ORDER BY h, ts")
    assert _col(r, "rn") == [4, 3, 2, 1, 3, 2, 1]
    assert _col(r, "c") == [4] * 4 + [3] * 3
