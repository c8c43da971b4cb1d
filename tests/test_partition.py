"""Multi-dim partition rules: PARTITION ON COLUMNS.

Reference parity: src/partition/src/multi_dim.rs (find_region,
split_record_batch) + PARTITION ON COLUMNS DDL.
"""

import numpy as np
import pytest

from greptimedb_amd.parallel.partition import (HashPartitionRule,
                                               MultiDimPartitionRule,
                                               PartitionExpr)
from greptimedb_amd.query.executor import Executor


def _expr(sql: str) -> PartitionExpr:
    from greptimedb_amd.query.parser import Parser
    return PartitionExpr.from_ast(
        Parser(f"SELECT {sql}").parse_statement().projections[0][0])


def test_rule_region_of_and_split():
    rule = MultiDimPartitionRule(
        ["host"],
        [_expr("host < 'h200'"),
         _expr("host >= 'h200' AND host < 'h600'"),
         _expr("host >= 'h600'")])
    # trailing catch-all default region is appended
    assert rule.n_regions == 4
    assert rule.region_of({"host": "h100"}) == 0
    assert rule.region_of({"host": "h200"}) == 1
    assert rule.region_of({"host": "h599"}) == 1
    assert rule.region_of({"host": "h900"}) == 2
    assert rule.region_of({"host": None}) == 3   # NULL → default region
    hosts = np.array(["h100", "h200", "h599", "h900", None], dtype=object)
    out = rule.split({"host": hosts}, 5)
    assert out.tolist() == [0, 1, 1, 2, 3]


def test_rule_numeric_coercion():
    rule = MultiDimPartitionRule(["rack"], [_expr("rack < 50"), _expr("rack >= 50")])
    assert rule.region_of({"rack": "7"}) == 0
    assert rule.region_of({"rack": "99"}) == 1
    out = rule.split({"rack": np.array(["7", "99", "50"], dtype=object)}, 3)
    assert out.tolist() == [0, 1, 1]


def test_rule_json_roundtrip_and_sql():
    rule = MultiDimPartitionRule(
        ["a", "b"], [_expr("a < 'm' AND b = 'x'"), _expr("a >= 'm' OR b != 'x'")])
    r2 = MultiDimPartitionRule.from_json(rule.to_json())
    assert r2.n_regions == rule.n_regions
    for vals in ({"a": "c", "b": "x"}, {"a": "z", "b": "x"}, {"a": "c", "b": "y"}):
        assert r2.region_of(vals) == rule.region_of(vals)
    assert "PARTITION ON COLUMNS (a, b)" in rule.to_sql()
    assert "a < 'm' AND b = 'x'" in rule.to_sql()


def test_hash_rule_stable():
    rule = HashPartitionRule(4, ["host"])
    r1 = rule.region_of({"host": "web-1"})
    assert 0 <= r1 < 4
    assert rule.region_of({"host": "web-1"}) == r1


def test_create_insert_query_partitioned(tmp_engine):
    ex = Executor(tmp_engine)
    ex.execute(
        "CREATE TABLE sensors ("
        "  host STRING, zone STRING, ts TIMESTAMP TIME INDEX, v DOUBLE,"
        "  PRIMARY KEY (host, zone)"
        ") PARTITION ON COLUMNS (host) ("
        "  host < 'h3', host >= 'h3' AND host < 'h6', host >= 'h6')")
    st = tmp_engine.table("sensors")
    assert len(st.regions) == 4  # 3 exprs + catch-all
    vals = ", ".join(f"('h{i}', 'z', {1000 + i}, {float(i)})" for i in range(9))
    ex.execute(f"INSERT INTO sensors (host, zone, ts, v) VALUES {vals}")
    per_region = [len(r.series) for r in st.regions]
    assert per_region == [3, 3, 3, 0]
    # scan over all partitions merges back
    res = ex.execute("SELECT count(*) AS c, min(v) AS lo, max(v) AS hi FROM sensors")
    row = res.rows()[0]
    assert (int(row[0]), float(row[1]), float(row[2])) == (9, 0.0, 8.0)
    res = ex.execute("SELECT host, v FROM sensors WHERE host = 'h7'")
    assert [list(r) for r in res.rows()] == [["h7", 7.0]]
    # rule survives restart (catalog round trip)
    ddl = ex.execute("SHOW CREATE TABLE sensors").rows()[0][1]
    assert "PARTITION ON COLUMNS (host)" in ddl
    assert "host < 'h3'" in ddl


def test_partitioned_table_restart(tmp_path):
    from greptimedb_amd.engine.engine import EngineConfig, MitoEngine
    d = str(tmp_path / "data")
    eng = MitoEngine(EngineConfig(data_dir=d, device="cpu", background_flush=False))
    ex = Executor(eng)
    ex.execute(
        "CREATE TABLE t (k STRING, ts TIMESTAMP TIME INDEX, v DOUBLE, PRIMARY KEY (k))"
        " PARTITION ON COLUMNS (k) (k < 'm', k >= 'm')")
    ex.execute("INSERT INTO t (k, ts, v) VALUES ('a', 1, 1.0), ('z', 2, 2.0)")
    eng.flush_all()
    eng.close()
    eng2 = MitoEngine(EngineConfig(data_dir=d, device="cpu", background_flush=False))
    ex2 = Executor(eng2)
    st = eng2.table("t")
    assert eng2.partition_rule(st).n_regions == 3
    assert isinstance(eng2.partition_rule(st), MultiDimPartitionRule)
    ex2.execute("INSERT INTO t (k, ts, v) VALUES ('b', 3, 3.0)")
    assert len(st.regions[0].series) == 2  # 'a' and 'b'
    res = ex2.execute("SELECT count(*) FROM t")
    assert int(res.rows()[0][0]) == 3
    eng2.close()


def test_influx_ingest_respects_rule(tmp_engine):
    from greptimedb_amd.engine.ingest import Ingestor
    ex = Executor(tmp_engine)
    ex.execute(
        "CREATE TABLE m (host STRING, ts TIMESTAMP TIME INDEX, u DOUBLE,"
        " PRIMARY KEY (host)) PARTITION ON COLUMNS (host)"
        " (host < 'k', host >= 'k')")
    ing = Ingestor(tmp_engine)
    lines = b"\n".join(b"m,host=%s u=1.0 %d" % (h, 1_000_000_000 + i)
                       for i, h in enumerate([b"alpha", b"kilo", b"zulu", b"beta"]))
    ing.ingest_lines(lines)
    st = tmp_engine.table("m")
    assert sorted(st.regions[0].series.tag_values) == [("alpha",), ("beta",)]
    assert sorted(st.regions[1].series.tag_values) == [("kilo",), ("zulu",)]


def test_partition_column_must_be_tag(tmp_engine):
    from greptimedb_amd.utils.errors import InvalidArguments
    ex = Executor(tmp_engine)
    with pytest.raises(InvalidArguments):
        ex.execute(
            "CREATE TABLE bad (host STRING, ts TIMESTAMP TIME INDEX, v DOUBLE,"
            " PRIMARY KEY (host)) PARTITION ON COLUMNS (v) (v < 1, v >= 1)")
