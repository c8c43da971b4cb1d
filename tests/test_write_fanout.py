"""Cross-rank write fan-out: any rank accepts any write.

Reference parity: src/operator/src/insert.rs:389-496 (partition-split +
per-peer fan-out). VERDICT r1 #2 done-criterion: a world>=4 test where ONE
rank ingests ALL the data and every rank's regions end up holding exactly
its hash shard.
"""

import multiprocessing as mp
import socket

import numpy as np
import pytest


def _free_ports(n: int) -> list[int]:
    socks, ports = [], []
    try:
        for _ in range(n):
            s = socket.socket()
            s.bind(("127.0.0.1", 0))
            socks.append(s)
            ports.append(s.getsockname()[1])
    finally:
        for s in socks:
            s.close()
    return ports


def _rank_proc(rank, world, ports, data_dir, barrier, results, ingest_all_rank):
    import torch  # noqa: F401  (loads libc10 for _native)
    from greptimedb_amd.engine.engine import EngineConfig, MitoEngine
    from greptimedb_amd.engine.ingest import Ingestor
    from greptimedb_amd.engine.series import tsid_hash
    from greptimedb_amd.engine import pk_codec
    from greptimedb_amd.parallel.write_fanout import WriteExchange

    eng = MitoEngine(EngineConfig(data_dir=f"{data_dir}/r{rank}", device="cpu",
                                  background_flush=False, default_regions=2))
    ing = Ingestor(eng, rank=rank, world=world)
    ex = WriteExchange(rank, world, handler=ing.handle_remote, ports=ports)
    ing.exchange = ex
    barrier.wait()  # all exchanges listening

    n_hosts, pts = 40, 5
    if rank == ingest_all_rank:
        lines = []
        for h in range(n_hosts):
            for p in range(pts):
                lines.append(b"cpu,hostname=host_%d,region=r%d usage_user=%f %d"
                             % (h, h % 3, h + p / 10.0, 1_000_000_000 * (p + 1)))
        ing.ingest_lines(b"\n".join(lines))
    barrier.wait()  # ingest done everywhere

    st = eng.tables.get("cpu")
    my_rows = sum(r.num_rows for r in st.regions) if st else 0
    my_series = sorted(tv for r in st.regions for tv in r.series.tag_values) \
        if st else []
    # expected shard: hosts whose pk hashes to this rank
    expected = []
    if st:
        for h in range(n_hosts):
            tags = (f"host_{h}", f"r{h % 3}")
            if tsid_hash(pk_codec.encode_pk(tags)) % world == rank:
                expected.append(tags)
    results.put((rank, my_rows, my_series, sorted(expected)))
    barrier.wait()  # keep exchanges alive until everyone reported
    ex.close()
    eng.close()


@pytest.mark.parametrize("world,ingest_rank", [(4, 0), (2, 1)])
def test_single_rank_ingests_all(tmp_path, world, ingest_rank):
    ctx = mp.get_context("spawn")
    barrier = ctx.Barrier(world, timeout=120)
    results = ctx.Queue()
    base = _free_ports(world)
    procs = [ctx.Process(target=_rank_proc,
                         args=(r, world, base, str(tmp_path), barrier, results,
                               ingest_rank))
             for r in range(world)]
    for p in procs:
        p.start()
    got = {}
    for _ in range(world):
        rank, rows, series, expected = results.get(timeout=180)
        got[rank] = (rows, series, expected)
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0
    total = sum(rows for rows, _, _ in got.values())
    assert total == 40 * 5
    all_series = []
    for rank, (rows, series, expected) in got.items():
        # every rank holds exactly its hash shard (disjoint ownership)
        assert series == expected, f"rank {rank} shard mismatch"
        assert rows == len(series) * 5
        all_series += series
    assert len(all_series) == 40 and len(set(all_series)) == 40


def _dual_ingest_proc(rank, world, ports, data_dir, barrier, results):
    import torch  # noqa: F401
    from greptimedb_amd.engine.engine import EngineConfig, MitoEngine
    from greptimedb_amd.engine.ingest import Ingestor
    from greptimedb_amd.parallel.write_fanout import WriteExchange

    eng = MitoEngine(EngineConfig(data_dir=f"{data_dir}/r{rank}", device="cpu",
                                  background_flush=False, default_regions=2))
    ing = Ingestor(eng, rank=rank, world=world)
    ex = WriteExchange(rank, world, handler=ing.handle_remote, ports=ports)
    ing.exchange = ex
    barrier.wait()
    # BOTH ranks write the SAME series set concurrently (ts offset per rank)
    lines = b"\n".join(
        b"mem,hostname=host_%d used=%f %d"
        % (h, float(h), 1_000_000_000 * (rank + 1) + h)
        for h in range(30))
    ing.ingest_lines(lines)
    barrier.wait()
    st = eng.tables.get("mem")
    rows = sum(r.num_rows for r in st.regions) if st else 0
    series = sorted(tv for r in st.regions for tv in r.series.tag_values) \
        if st else []
    results.put((rank, rows, series))
    barrier.wait()
    ex.close()
    eng.close()


def test_both_ranks_ingest_same_series(tmp_path):
    world = 2
    ctx = mp.get_context("spawn")
    barrier = ctx.Barrier(world, timeout=120)
    results = ctx.Queue()
    base = _free_ports(world)
    procs = [ctx.Process(target=_dual_ingest_proc,
                         args=(r, world, base, str(tmp_path), barrier, results))
             for r in range(world)]
    for p in procs:
        p.start()
    got = {}
    for _ in range(world):
        rank, rows, series = results.get(timeout=180)
        got[rank] = (rows, series)
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0
    # each series lives on exactly one rank, with BOTH ranks' points
    assert sum(r for r, _ in got.values()) == 2 * 30
    s0, s1 = got[0][1], got[1][1]
    assert not (set(s0) & set(s1))
    assert len(s0) + len(s1) == 30
    for rank, (rows, series) in got.items():
        assert rows == 2 * len(series)


@pytest.mark.gpu
def test_fanout_bulk_receive_gpu(tmp_path):
    """Owner-rank bulk apply (K16 scatter path) on device: two engines on
    cuda:0 in one process, rank 0 ingests everything, remote rows land via
    write_regions_bulk."""
    import torch
    from greptimedb_amd.engine.engine import EngineConfig, MitoEngine
    from greptimedb_amd.engine.ingest import Ingestor
    from greptimedb_amd.parallel.write_fanout import WriteExchange

    engines = [MitoEngine(EngineConfig(data_dir=str(tmp_path / f"r{r}"),
                                       device="cuda:0",
                                       background_flush=False,
                                       default_regions=2))
               for r in range(2)]
    ports = _free_ports(2)
    exchanges = []
    ings = []
    for r in range(2):
        ing = Ingestor(engines[r], rank=r, world=2)
        ex = WriteExchange(r, 2, handler=ing.handle_remote, ports=ports)
        ing.exchange = ex
        exchanges.append(ex)
        ings.append(ing)
    lines = b"\n".join(
        b"gpu_m,hostname=host_%d u=%f,v=%f %d"
        % (h, float(h), float(h) * 2, 1_000_000_000 * (p + 1))
        for h in range(16) for p in range(4))
    ings[0].ingest_lines(lines)
    torch.cuda.synchronize()
    totals = [sum(r.num_rows for r in e.table("gpu_m").regions)
              if "gpu_m" in e.tables else 0 for e in engines]
    assert sum(totals) == 64
    assert all(t > 0 for t in totals), totals
    # values survive the scatter: query each engine
    from greptimedb_amd.query.executor import Executor
    vals = []
    for e in engines:
        r = Executor(e).execute("SELECT sum(u) FROM gpu_m")
        vals.append(float(r.columns[0][0]))
    assert sum(vals) == sum(float(h) for h in range(16)) * 4
    for ex in exchanges:
        ex.close()
    for e in engines:
        e.close()


def test_exchange_explicit_peer_map(monkeypatch):
    """Multi-node seam: rank → (host, port) peer map (here both on
    loopback) routes requests by the per-rank address."""
    from greptimedb_amd.parallel.write_fanout import WriteExchange
    p0, p1 = _free_ports(2)
    peers = [("127.0.0.1", p0), ("127.0.0.1", p1)]
    got = []

    def handler(payload):
        got.append(payload)
        return b"OK"

    ex0 = WriteExchange(0, 2, handler=handler, peers=peers)
    ex1 = WriteExchange(1, 2, handler=handler, peers=peers)
    try:
        assert ex1.request(0, b"hello-from-1") == b"OK"
        assert ex0.request(1, b"hello-from-0") == b"OK"
        assert got == [b"hello-from-1", b"hello-from-0"]
    finally:
        ex0.close()
        ex1.close()


def test_exchange_peers_env(monkeypatch):
    from greptimedb_amd.parallel.write_fanout import WriteExchange
    p0, p1 = _free_ports(2)
    monkeypatch.setenv("GDB_FANOUT_PEERS",
                       f"127.0.0.1:{p0}, 127.0.0.1:{p1}")
    ex0 = WriteExchange(0, 2, handler=lambda b: b"OK")
    ex1 = WriteExchange(1, 2, handler=lambda b: b"OK")
    try:
        assert ex0.request(1, b"x") == b"OK"
    finally:
        ex0.close()
        ex1.close()
