"""SQL RANGE queries (ref src/query/src/range_select: window [t, t+range)
per ALIGN step; FILL NULL/PREV/LINEAR/const; ALIGN BY grouping)."""

import numpy as np
import pytest

from greptimedb_amd.query.executor import Executor


@pytest.fixture
def rex(tmp_engine):
    ex = Executor(tmp_engine)
    ex.execute("CREATE TABLE m (host STRING, ts TIMESTAMP TIME INDEX, "
               "v DOUBLE, PRIMARY KEY (host))")
    rows = []
    # host a: v = ts/1000 at 0,5,10,15,20s ; host b: 100+ts/1000 at 0,10,20s
    for t in range(0, 21, 5):
        rows.append(f"('a', {t * 1000}, {float(t)})")
    for t in range(0, 21, 10):
        rows.append(f"('b', {t * 1000}, {100.0 + t})")
    ex.execute("INSERT INTO m (host, ts, v) VALUES " + ",".join(rows))
    return ex


def _col(r, name):
    return list(r.columns[r.names.index(name)])


def test_range_tumbling(rex):
    # range == align → plain buckets [t, t+10s)
    r = rex.execute("SELECT ts, host, min(v) RANGE '10s' AS mn, "
                    "max(v) RANGE '10s' AS mx FROM m ALIGN '10s' "
                    "ORDER BY host, ts")
    assert _col(r, "ts") == [0, 10000, 20000] * 2
    assert _col(r, "mn") == [0.0, 10.0, 20.0, 100.0, 110.0, 120.0]
    assert _col(r, "mx") == [5.0, 15.0, 20.0, 100.0, 110.0, 120.0]


def test_range_sliding(rex):
    # range 20s, align 10s → overlapping windows
    r = rex.execute("SELECT ts, host, sum(v) RANGE '20s' AS s FROM m "
                    "ALIGN '10s' ORDER BY host, ts")
    a = {(h, t): s for h, t, s in zip(_col(r, "host"), _col(r, "ts"), _col(r, "s"))}
    # host a window [-10,10) → {0,5}; [0,20) → {0,5,10,15}; [10,30) → {10,15,20}
    assert a[("a", -10000)] == 5.0
    assert a[("a", 0)] == 30.0
    assert a[("a", 10000)] == 45.0
    assert a[("a", 20000)] == 20.0


def test_range_avg_count_fill(rex):
    r = rex.execute("SELECT ts, host, avg(v) RANGE '5s' AS av FROM m "
                    "WHERE host = 'b' ALIGN '5s' FILL PREV ORDER BY ts")
    # b has samples at 0,10,20 → 5s/15s slots fill from prev
    assert _col(r, "av") == [100.0, 100.0, 110.0, 110.0, 120.0]
    r = rex.execute("SELECT ts, host, count(v) RANGE '10s' AS c FROM m "
                    "WHERE host = 'a' ALIGN '10s' ORDER BY ts")
    assert _col(r, "c") == [2.0, 2.0, 1.0]


def test_range_fill_linear_and_const(rex):
    r = rex.execute("SELECT ts, host, max(v) RANGE '5s' FILL LINEAR AS mv "
                    "FROM m WHERE host = 'b' ALIGN '5s' ORDER BY ts")
    assert _col(r, "mv") == [100.0, 105.0, 110.0, 115.0, 120.0]
    r = rex.execute("SELECT ts, host, min(v) RANGE '5s' FILL 0 AS mv FROM m "
                    "WHERE host = 'b' ALIGN '5s' ORDER BY ts")
    assert _col(r, "mv") == [100.0, 0.0, 110.0, 0.0, 120.0]


def test_range_by_clause_and_expr(rex):
    # BY () → one global group
    r = rex.execute("SELECT ts, max(v) RANGE '10s' - min(v) RANGE '10s' AS d "
                    "FROM m ALIGN '10s' BY () ORDER BY ts")
    assert _col(r, "d") == [100.0, 100.0, 100.0]


def test_range_last_value(rex):
    r = rex.execute("SELECT ts, host, last_value(v) RANGE '20s' AS lv FROM m "
                    "WHERE host = 'a' ALIGN '20s' ORDER BY ts")
    assert _col(r, "lv") == [15.0, 20.0]
