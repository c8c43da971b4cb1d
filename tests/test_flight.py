"""Arrow Flight server: do_get SQL, do_put bulk ingest, actions.

Reference parity: src/servers/src/grpc/flight.rs:67-86 (query) and
:240-330 (do_put bulk ingest).
"""

import numpy as np
import pyarrow as pa
import pytest

from greptimedb_amd.servers.flight import FlightClient, GreptimeFlightServer


@pytest.fixture
def flight_pair(tmp_engine):
    from greptimedb_amd.query.executor import Executor
    srv = GreptimeFlightServer(tmp_engine, Executor(tmp_engine))
    cli = FlightClient("127.0.0.1", srv.port)
    yield tmp_engine, srv, cli
    cli.close()
    srv.shutdown()


def _mk_batch(n=100, hosts=4, t0=1_600_000_000_000):
    return pa.table({
        "hostname": pa.array([f"h{i % hosts}" for i in range(n)]),
        "ts": pa.array(np.arange(n, dtype=np.int64) * 1000 + t0,
                       type=pa.int64()).cast(pa.timestamp("ms")),
        "usage_user": pa.array(np.linspace(0, 99, n)),
        "usage_system": pa.array(np.linspace(1, 50, n)),
    })


def test_do_put_then_do_get_roundtrip(flight_pair):
    eng, srv, cli = flight_pair
    n = cli.put("cpu_flight", _mk_batch())
    assert n == 100
    st = eng.table("cpu_flight")
    assert sum(r.num_rows for r in st.regions) == 100
    res = cli.sql("SELECT hostname, count(*) AS c, max(usage_user) AS mx "
                  "FROM cpu_flight GROUP BY hostname ORDER BY hostname")
    d = res.to_pydict()
    assert d["hostname"] == ["h0", "h1", "h2", "h3"]
    assert d["c"] == [25, 25, 25, 25]
    assert max(d["mx"]) == pytest.approx(99.0)


def test_do_put_into_existing_sql_table(flight_pair):
    eng, srv, cli = flight_pair
    from greptimedb_amd.query.executor import Executor
    ex = Executor(eng)
    ex.execute("CREATE TABLE pre (hostname STRING, ts TIMESTAMP TIME INDEX,"
               " usage_user DOUBLE, PRIMARY KEY (hostname))"
               " PARTITION ON COLUMNS (hostname) (hostname < 'h2', hostname >= 'h2')")
    cli.put("pre", _mk_batch(40))
    st = eng.table("pre")
    # partition rule respected: h0/h1 → region 0, h2/h3 → region 1
    assert sorted(tv[0] for tv in st.regions[0].series.tag_values) == ["h0", "h1"]
    assert sorted(tv[0] for tv in st.regions[1].series.tag_values) == ["h2", "h3"]
    res = cli.sql("SELECT count(*) AS c FROM pre")
    assert res.to_pydict()["c"] == [40]


def test_do_get_timestamps_are_arrow_typed(flight_pair):
    eng, srv, cli = flight_pair
    cli.put("tt", _mk_batch(10))
    res = cli.sql("SELECT ts, usage_user FROM tt ORDER BY ts LIMIT 3")
    assert pa.types.is_timestamp(res.schema.field("ts").type)
    assert res.num_rows == 3


def test_flight_actions_flush(flight_pair):
    eng, srv, cli = flight_pair
    cli.put("fl", _mk_batch(50))
    list(cli.conn.do_action(pa.flight.Action("flush", b"")))
    st = eng.table("fl")
    assert sum(len(r.manifest.files) for r in st.regions) >= 1
    res = cli.sql("SELECT count(*) AS c FROM fl")
    assert res.to_pydict()["c"] == [50]


def test_get_flight_info(flight_pair):
    eng, srv, cli = flight_pair
    cli.put("gi", _mk_batch(20))
    import json
    import pyarrow.flight as flight
    desc = flight.FlightDescriptor.for_command(
        json.dumps({"sql": "SELECT count(*) AS c FROM gi"}).encode())
    info = cli.conn.get_flight_info(desc)
    assert info.total_records == 1
    tbl = cli.conn.do_get(info.endpoints[0].ticket).read_all()
    assert tbl.to_pydict()["c"] == [20]
