"""greptime.v1.GreptimeDatabase gRPC service (hand-rolled protobuf).

Reference parity: src/servers/src/grpc/database.rs (Handle row inserts),
builder.rs:138-172 (service assembly), health service.
"""

import pytest

from greptimedb_amd.servers.grpc_server import (DT_FLOAT64, DT_STRING,
                                                DT_TS_MILLI, SEM_FIELD,
                                                SEM_TAG, SEM_TIMESTAMP,
                                                GreptimeGrpcClient,
                                                GreptimeGrpcServer)

CPU_SCHEMA = [
    ("hostname", DT_STRING, SEM_TAG),
    ("ts", DT_TS_MILLI, SEM_TIMESTAMP),
    ("usage_user", DT_FLOAT64, SEM_FIELD),
]


@pytest.fixture
def grpc_pair(tmp_engine):
    srv = GreptimeGrpcServer(tmp_engine)
    cli = GreptimeGrpcClient("127.0.0.1", srv.port)
    yield tmp_engine, srv, cli
    cli.close()
    srv.shutdown()


def test_health(grpc_pair):
    _, _, cli = grpc_pair
    assert cli.health()


def test_insert_rows_roundtrip(grpc_pair):
    eng, srv, cli = grpc_pair
    rows = [[f"h{i % 3}", 1_600_000_000_000 + i * 1000, float(i)]
            for i in range(30)]
    n = cli.insert_rows("cpu_grpc", CPU_SCHEMA, rows)
    assert n == 30
    st = eng.table("cpu_grpc")
    assert st.schema.primary_key == ["hostname"]
    assert sum(r.num_rows for r in st.regions) == 30
    # query through the engine (results over gRPC travel via Flight)
    from greptimedb_amd.query.executor import Executor
    res = Executor(eng).execute(
        "SELECT hostname, count(*) AS c FROM cpu_grpc GROUP BY hostname "
        "ORDER BY hostname")
    assert res.to_dict()["c"] == [10, 10, 10]


def test_sql_over_grpc(grpc_pair):
    eng, srv, cli = grpc_pair
    cli.sql("CREATE TABLE g2 (k STRING, ts TIMESTAMP TIME INDEX, v DOUBLE,"
            " PRIMARY KEY (k))")
    assert "g2" in eng.tables
    cli.sql("INSERT INTO g2 (k, ts, v) VALUES ('a', 1, 2.0), ('b', 2, 3.0)")
    assert sum(r.num_rows for r in eng.table("g2").regions) == 2


def test_insert_null_values(grpc_pair):
    eng, srv, cli = grpc_pair
    rows = [["h0", 1_600_000_000_000, 1.0],
            ["h1", 1_600_000_001_000, None],
            [None, 1_600_000_002_000, 3.0]]
    assert cli.insert_rows("nulls", CPU_SCHEMA, rows) == 3
    from greptimedb_amd.query.executor import Executor
    res = Executor(eng).execute("SELECT count(*) AS c FROM nulls")
    assert res.to_dict()["c"] == [3]


def test_error_status_propagates(grpc_pair):
    _, _, cli = grpc_pair
    with pytest.raises(RuntimeError):
        cli.sql("SELECT * FROM no_such_table_xyz")


def test_ts_second_scaled(grpc_pair):
    eng, srv, cli = grpc_pair
    from greptimedb_amd.servers.grpc_server import DT_TS_SECOND
    schema = [("hostname", DT_STRING, SEM_TAG),
              ("ts", DT_TS_SECOND, SEM_TIMESTAMP),
              ("v", DT_FLOAT64, SEM_FIELD)]
    cli.insert_rows("tsec", schema, [["h", 1_600_000_000, 5.0]])
    from greptimedb_amd.query.executor import Executor
    res = Executor(eng).execute("SELECT ts FROM tsec")
    assert int(res.columns[0][0]) == 1_600_000_000_000  # ms
