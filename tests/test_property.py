"""Property-based tests (reference parity: tests-fuzz): random workloads
against a pure-Python oracle."""

import math

import numpy as np
import pytest

hyp = pytest.importorskip("hypothesis")
from hypothesis import HealthCheck, given, settings  # noqa: E402
from hypothesis import strategies as st_  # noqa: E402

from greptimedb_amd.engine.engine import EngineConfig, MitoEngine  # noqa: E402
from greptimedb_amd.query.executor import Executor  # noqa: E402

ROWS = st_.lists(
    st_.tuples(
        st_.sampled_from(["a", "b", "c", "d"]),            # tag
        st_.integers(min_value=0, max_value=100_000),      # ts ms
        st_.floats(min_value=-1e6, max_value=1e6,
                   allow_nan=False, allow_infinity=False),  # value
    ),
    min_size=1, max_size=60,
)


def _mk(tmp_path_factory, rows, append):
    eng = MitoEngine(EngineConfig(
        data_dir=str(tmp_path_factory.mktemp("prop")), device="cpu",
        background_flush=False, default_regions=3))
    ex = Executor(eng)
    mode = "true" if append else "false"
    ex.execute("CREATE TABLE p (h STRING, ts TIMESTAMP TIME INDEX, v DOUBLE, "
               f"PRIMARY KEY (h)) WITH ('append_mode'='{mode}')")
    vals = ", ".join(f"('{h}', {t}, {v!r})" for h, t, v in rows)
    ex.execute(f"INSERT INTO p (h, ts, v) VALUES {vals}")
    return eng, ex


def _oracle(rows, append):
    if append:
        return list(rows)
    last = {}
    for h, t, v in rows:
        last[(h, t)] = v
    return [(h, t, v) for (h, t), v in last.items()]


@settings(max_examples=40, deadline=None,
          suppress_health_check=[HealthCheck.function_scoped_fixture])
@given(rows=ROWS, append=st_.booleans(), flush=st_.booleans(),
       lo=st_.integers(min_value=0, max_value=100_000),
       span=st_.integers(min_value=1, max_value=100_000))
def test_agg_matches_oracle(tmp_path_factory, rows, append, flush, lo, span):
    eng, ex = _mk(tmp_path_factory, rows, append)
    try:
        if flush:
            eng.flush_all()
        hi = lo + span
        data = [(h, t, v) for h, t, v in _oracle(rows, append) if lo <= t < hi]
        r = ex.execute(f"SELECT count(*), sum(v), min(v), max(v) FROM p "
                       f"WHERE ts >= {lo} AND ts < {hi}")
        cnt = r.columns[0][0]
        assert cnt == len(data)
        if data:
            vs = [v for _h, _t, v in data]
            assert math.isclose(r.columns[1][0], sum(vs), rel_tol=1e-9, abs_tol=1e-6)
            assert math.isclose(r.columns[2][0], min(vs), rel_tol=0, abs_tol=0)
            assert math.isclose(r.columns[3][0], max(vs), rel_tol=0, abs_tol=0)
        # per-tag counts
        r = ex.execute("SELECT h, count(*) FROM p GROUP BY h ORDER BY h")
        exp = {}
        for h, _t, _v in _oracle(rows, append):
            exp[h] = exp.get(h, 0) + 1
        got = dict(zip(r.columns[0], (int(c) for c in r.columns[1])))
        assert got == exp
    finally:
        eng.close()


@settings(max_examples=25, deadline=None,
          suppress_health_check=[HealthCheck.function_scoped_fixture])
@given(rows=ROWS, flush=st_.booleans())
def test_raw_scan_matches_oracle(tmp_path_factory, rows, flush):
    eng, ex = _mk(tmp_path_factory, rows, append=True)
    try:
        if flush:
            eng.flush_all()
        r = ex.execute("SELECT h, ts, v FROM p ORDER BY ts, h, v")
        got = sorted((h, int(t), float(v)) for h, t, v in
                     zip(r.columns[0], r.columns[1], r.columns[2]))
        exp = sorted((h, t, float(np.float64(v))) for h, t, v in rows)
        assert len(got) == len(exp)
        for g, e in zip(got, exp):
            assert g[0] == e[0] and g[1] == e[1]
            assert math.isclose(g[2], e[2], rel_tol=1e-12, abs_tol=0)
    finally:
        eng.close()


@settings(max_examples=20, deadline=None,
          suppress_health_check=[HealthCheck.function_scoped_fixture])
@given(rows=ROWS)
def test_restart_preserves_data(tmp_path_factory, rows):
    eng, ex = _mk(tmp_path_factory, rows, append=True)
    d = eng.config.data_dir
    n = len(rows)
    eng.close()
    eng2 = MitoEngine(EngineConfig(data_dir=d, device="cpu",
                                   background_flush=False))
    try:
        ex2 = Executor(eng2)
        assert ex2.execute("SELECT count(*) FROM p").columns[0][0] == n
    finally:
        eng2.close()


@given(st_.integers(0, 6), st_.integers(1, 60), st_.integers(1, 4))
@settings(max_examples=25, deadline=None)
def test_window_running_aggregates_property(seed, n, nparts):
    """Window running sum/min/count vs a brute-force python oracle."""
    import numpy as np
    from greptimedb_amd.query import ast
    from greptimedb_amd.query.executor import _compute_window
    rng = np.random.RandomState(seed)
    part = rng.randint(0, nparts, n)
    order = rng.randint(0, 10, n)         # duplicate order keys → peers
    vals = np.where(rng.rand(n) < 0.2, np.nan, rng.randn(n).round(3))
    col_data = {"p": part.astype(object), "o": order.astype(float),
                "v": vals}
    for func in ("sum", "min", "count", "row_number"):
        node = ast.WindowFunc(func, [ast.Col("v")] if func != "row_number" else [],
                              [ast.Col("p")], [(ast.Col("o"), False)])
        got = _compute_window(node, col_data, n)
        # oracle: for each row, rows in same partition with order <= this
        # row's order (peer-inclusive default frame)
        for i in range(n):
            peers = [j for j in range(n)
                     if part[j] == part[i] and order[j] <= order[i]]
            pv = [vals[j] for j in peers if not np.isnan(vals[j])]
            if func == "sum":
                exp = sum(pv) if pv else np.nan
            elif func == "min":
                exp = min(pv) if pv else np.nan
            elif func == "count":
                exp = float(len(pv))
            else:
                continue   # row_number checked for shape only below
            if np.isnan(exp):
                assert np.isnan(got[i]), (func, i)
            else:
                assert abs(got[i] - exp) < 1e-9, (func, i, got[i], exp)
        if func == "row_number":
            # per partition: a permutation of 1..len(partition)
            for p in set(part):
                rns = sorted(got[part == p])
                assert rns == list(range(1, len(rns) + 1))


@settings(max_examples=25, deadline=None,
          suppress_health_check=[HealthCheck.function_scoped_fixture])
@given(rows=ROWS, flush=st_.booleans(), k=st_.floats(0.5, 3.0))
def test_expr_agg_fallback_matches_oracle(tmp_path_factory, rows, flush, k):
    """sum(v*k) / avg(v+k) / CASE aggregates through the general-shape
    fallback vs a python oracle."""
    eng, ex = _mk(tmp_path_factory, rows, append=True)
    try:
        if flush:
            eng.flush_all()
        r = ex.execute(f"SELECT h, sum(v * {k!r}), avg(v + {k!r}), "
                       f"sum(CASE WHEN v > 0 THEN 1 ELSE 0 END) "
                       f"FROM p GROUP BY h ORDER BY h")
        exp: dict = {}
        for h, _t, v in rows:
            s, c, pos = exp.get(h, (0.0, 0, 0))
            exp[h] = (s + v, c + 1, pos + (1 if v > 0 else 0))
        assert list(r.columns[0]) == sorted(exp)
        for i, h in enumerate(r.columns[0]):
            s, c, pos = exp[h]
            assert math.isclose(float(r.columns[1][i]), s * k,
                                rel_tol=1e-9, abs_tol=1e-6)
            assert math.isclose(float(r.columns[2][i]), s / c + k,
                                rel_tol=1e-9, abs_tol=1e-6)
            assert int(r.columns[3][i]) == pos
    finally:
        eng.close()


@settings(max_examples=25, deadline=None,
          suppress_health_check=[HealthCheck.function_scoped_fixture])
@given(rows=ROWS, flush=st_.booleans())
def test_first_last_value_matches_oracle(tmp_path_factory, rows, flush):
    """first_value/last_value per tag vs python argmin/argmax over ts
    (append mode: ties broken by insertion order → compare on unique ts)."""
    # dedupe (h, ts) keeping last so the oracle is unambiguous
    seen = {}
    for h, t, v in rows:
        seen[(h, t)] = v
    rows = [(h, t, v) for (h, t), v in seen.items()]
    eng, ex = _mk(tmp_path_factory, rows, append=False)
    try:
        if flush:
            eng.flush_all()
        r = ex.execute("SELECT h, first_value(v), last_value(v) FROM p "
                       "GROUP BY h ORDER BY h")
        first: dict = {}
        last: dict = {}
        for h, t, v in rows:
            if h not in first or t < first[h][0]:
                first[h] = (t, v)
            if h not in last or t > last[h][0]:
                last[h] = (t, v)
        assert list(r.columns[0]) == sorted(first)
        for i, h in enumerate(r.columns[0]):
            assert math.isclose(float(r.columns[1][i]), first[h][1],
                                rel_tol=1e-12, abs_tol=0)
            assert math.isclose(float(r.columns[2][i]), last[h][1],
                                rel_tol=1e-12, abs_tol=0)
    finally:
        eng.close()
