"""Storage engine: ingest → memtable → flush → SST → reopen → WAL replay."""

import numpy as np
import pyarrow.parquet as pq

from greptimedb_amd.engine.engine import EngineConfig, MitoEngine
from greptimedb_amd.engine.ingest import Ingestor
from greptimedb_amd.models.tsbs import CpuWorkload


def _total(engine, table="cpu"):
    return sum(r.num_rows for r in engine.table(table).regions)


def test_ingest_flush_reopen(tmp_path):
    d = str(tmp_path / "data")
    eng = MitoEngine(EngineConfig(data_dir=d, device="cpu", background_flush=False))
    ing = Ingestor(eng)
    w = CpuWorkload(scale=50)
    for _ in range(5):
        ing.ingest_lines(w.next_batch(1000))
    assert _total(eng) == 5000
    eng.flush_all()
    assert _total(eng) == 5000
    assert sum(r.memtable.len for r in eng.table("cpu").regions) == 0
    eng.close()

    eng2 = MitoEngine(EngineConfig(data_dir=d, device="cpu", background_flush=False))
    assert _total(eng2) == 5000
    eng2.close()


def test_wal_replay_unflushed(tmp_path):
    d = str(tmp_path / "data")
    eng = MitoEngine(EngineConfig(data_dir=d, device="cpu", background_flush=False))
    ing = Ingestor(eng)
    w = CpuWorkload(scale=10)
    ing.ingest_lines(w.next_batch(500))
    eng.flush_all()
    ing.ingest_lines(w.next_batch(700))  # unflushed → only in WAL
    eng.close()
    eng2 = MitoEngine(EngineConfig(data_dir=d, device="cpu", background_flush=False))
    assert _total(eng2) == 1200
    # replay must not double-apply flushed entries
    eng2.flush_all()
    assert _total(eng2) == 1200
    eng2.close()


def test_series_codes_stable_across_restart(tmp_path):
    d = str(tmp_path / "data")
    eng = MitoEngine(EngineConfig(data_dir=d, device="cpu", background_flush=False))
    ing = Ingestor(eng)
    w = CpuWorkload(scale=20)
    ing.ingest_lines(w.next_batch(100))
    st = eng.table("cpu")
    before = {r.region_id: list(r.series.pks) for r in st.regions}
    eng.flush_all()
    eng.close()
    eng2 = MitoEngine(EngineConfig(data_dir=d, device="cpu", background_flush=False))
    for r in eng2.table("cpu").regions:
        assert list(r.series.pks) == before[r.region_id]
    eng2.close()


def test_sst_mito2_layout(tmp_path):
    """SST parquet must carry the mito2 internal columns
    (reference sst/parquet/format.rs:15-27)."""
    d = str(tmp_path / "data")
    eng = MitoEngine(EngineConfig(data_dir=d, device="cpu", background_flush=False))
    ing = Ingestor(eng)
    w = CpuWorkload(scale=5)
    ing.ingest_lines(w.next_batch(200))
    eng.flush_all()
    import glob
    files = glob.glob(f"{d}/region/*/sst/*.parquet")
    assert files
    t = pq.read_table(files[0])
    names = t.column_names
    assert "__primary_key" in names and "__sequence" in names and "__op_type" in names
    assert names[-3:] == ["__primary_key", "__sequence", "__op_type"]
    assert str(t.schema.field("__primary_key").type).startswith("dictionary")
    meta = pq.read_metadata(files[0])
    assert meta.row_group(0).column(0).compression in ("ZSTD",)
    eng.close()


def test_flush_sorted_and_deduped(tmp_path):
    from greptimedb_amd.query.executor import Executor
    d = str(tmp_path / "data")
    eng = MitoEngine(EngineConfig(data_dir=d, device="cpu", background_flush=False))
    ex = Executor(eng)
    ex.execute("CREATE TABLE t (h STRING, ts TIMESTAMP TIME INDEX, v DOUBLE, PRIMARY KEY (h))")
    ex.execute("INSERT INTO t (h, ts, v) VALUES ('b', 2, 1.0), ('a', 1, 2.0), "
               "('a', 1, 3.0), ('b', 1, 4.0)")
    eng.flush_all()
    import glob
    f = glob.glob(f"{d}/region/*/sst/*.parquet")
    rows = sum(pq.read_metadata(x).num_rows for x in f)
    assert rows == 3  # ('a',1) deduped last-wins
    r = ex.execute("SELECT h, ts, v FROM t ORDER BY h, ts")
    assert [tuple(x) for x in r.rows()] == [("a", 1, 3.0), ("b", 1, 4.0), ("b", 2, 1.0)]
    eng.close()


def test_wal_purged_after_flush(tmp_path):
    d = str(tmp_path / "data")
    eng = MitoEngine(EngineConfig(data_dir=d, device="cpu", background_flush=False,
                                  wal_segment_bytes=1 << 16))
    ing = Ingestor(eng)
    w = CpuWorkload(scale=10)
    for _ in range(20):
        ing.ingest_lines(w.next_batch(500))
    segs_before = len(eng.wal.segments())
    assert segs_before > 1
    eng.flush_all()
    # append one more batch to roll a new segment reference point
    ing.ingest_lines(w.next_batch(10))
    eng.flush_all()
    assert len(eng.wal.segments()) <= 2
    eng.close()


def test_schema_evolution_new_field(tmp_path):
    from greptimedb_amd.query.executor import Executor
    d = str(tmp_path / "data")
    eng = MitoEngine(EngineConfig(data_dir=d, device="cpu", background_flush=False))
    ing = Ingestor(eng)
    ing.ingest_lines(b"m,h=a f1=1 1000000000\n")
    ing.ingest_lines(b"m,h=a f1=2,f2=9 2000000000\n")
    ex = Executor(eng)
    r = ex.execute("SELECT ts, f1, f2 FROM m ORDER BY ts")
    assert list(r.columns[1]) == [1.0, 2.0]
    assert np.isnan(r.columns[2][0]) and r.columns[2][1] == 9.0
    eng.close()


def test_compaction_merges_small_ssts(tmp_path):
    from greptimedb_amd.query.executor import Executor
    d = str(tmp_path / "data")
    eng = MitoEngine(EngineConfig(data_dir=d, device="cpu", background_flush=False))
    ex = Executor(eng)
    ex.execute("CREATE TABLE t (h STRING, ts TIMESTAMP TIME INDEX, v DOUBLE, PRIMARY KEY (h))")
    for i in range(5):
        ex.execute(f"INSERT INTO t (h, ts, v) VALUES ('a', {i*1000}, {float(i)}), "
                   f"('b', {i*1000}, {float(i)+10})")
        for r in eng.table("t").regions:
            r.flush()
    before = ex.execute("SELECT h, ts, v FROM t ORDER BY h, ts").rows()
    r = ex.execute("ADMIN compact_table('t')")
    assert r.columns[0][0] >= 1
    after = ex.execute("SELECT h, ts, v FROM t ORDER BY h, ts").rows()
    assert before == after
    # files merged on disk + manifest
    import glob
    region = [rr for rr in eng.table("t").regions if rr.num_rows > 0][0]
    l1 = [m for m in region.manifest.files.values() if m["level"] == 1]
    assert l1, region.manifest.files
    eng.close()
    # reopen reads compacted layout
    eng2 = MitoEngine(EngineConfig(data_dir=d, device="cpu", background_flush=False))
    ex2 = Executor(eng2)
    assert ex2.execute("SELECT h, ts, v FROM t ORDER BY h, ts").rows() == before
    eng2.close()


def test_compaction_dedup_last_wins(tmp_path):
    from greptimedb_amd.query.executor import Executor
    d = str(tmp_path / "data")
    eng = MitoEngine(EngineConfig(data_dir=d, device="cpu", background_flush=False))
    ex = Executor(eng)
    ex.execute("CREATE TABLE t (h STRING, ts TIMESTAMP TIME INDEX, v DOUBLE, PRIMARY KEY (h))")
    for i in range(3):
        ex.execute(f"INSERT INTO t (h, ts, v) VALUES ('a', 1000, {float(i)})")
        for r in eng.table("t").regions:
            r.flush()
    ex.execute("ADMIN compact_table('t')")
    r = ex.execute("SELECT v FROM t")
    assert list(r.columns[0]) == [2.0]
    eng.close()


def test_explain(tmp_engine):
    from greptimedb_amd.query.executor import Executor
    ex = Executor(tmp_engine)
    ex.execute("CREATE TABLE t (h STRING, ts TIMESTAMP TIME INDEX, v DOUBLE, PRIMARY KEY (h))")
    ex.execute("INSERT INTO t (h, ts, v) VALUES ('a', 1000, 1.0)")
    r = ex.execute("EXPLAIN ANALYZE SELECT h, max(v) FROM t GROUP BY h")
    text = "\n".join(r.columns[0])
    assert "fused-ts-bucket-agg" in text and "Execution" in text


def test_repartition_table(tmp_path):
    from greptimedb_amd.query.executor import Executor
    d = str(tmp_path / "data")
    eng = MitoEngine(EngineConfig(data_dir=d, device="cpu", background_flush=False))
    ing = Ingestor(eng, default_regions=2)
    w = CpuWorkload(scale=30)
    for _ in range(4):
        ing.ingest_lines(w.next_batch(1000))
    ex = Executor(eng)
    before = ex.execute("SELECT hostname, count(*) FROM cpu GROUP BY hostname "
                        "ORDER BY hostname").rows()
    r = ex.execute("ADMIN repartition_table('cpu', 6)")
    assert r.columns[0][0] == 4000
    assert len(eng.table("cpu").regions) == 6
    after = ex.execute("SELECT hostname, count(*) FROM cpu GROUP BY hostname "
                       "ORDER BY hostname").rows()
    assert before == after
    # ingest continues through the same router after the epoch bump
    ing.ingest_lines(w.next_batch(500))
    assert ex.execute("SELECT count(*) FROM cpu").columns[0][0] == 4500
    # survives reopen
    eng.close()
    eng2 = MitoEngine(EngineConfig(data_dir=d, device="cpu", background_flush=False))
    ex2 = Executor(eng2)
    assert len(eng2.table("cpu").regions) == 6
    assert ex2.execute("SELECT count(*) FROM cpu").columns[0][0] == 4500
    eng2.close()


def test_promstore_restart_persistence(tmp_path):
    from greptimedb_amd.engine.promstore import PromStore
    from greptimedb_amd.query.promql.eval import PromEvaluator
    from tests.test_http import _ts_msg
    d = str(tmp_path / "data")
    eng = MitoEngine(EngineConfig(data_dir=d, device="cpu", background_flush=False))
    store = PromStore(eng)
    req = _ts_msg([("__name__", "m_up"), ("job", "a")], [(1.0, 1000), (2.0, 2000)])
    assert store.write(req, snappy=False) == 2
    eng.flush_all()
    store.write(_ts_msg([("__name__", "m_up"), ("job", "a")], [(3.0, 3000)]),
                snappy=False)  # WAL-only
    eng.close()
    eng2 = MitoEngine(EngineConfig(data_dir=d, device="cpu", background_flush=False))
    ev = PromEvaluator(eng2)
    m = ev.query_range('m_up{job="a"}', 3, 3, 1)
    assert m.S == 1 and float(m.values[0][0]) == 3.0
    # same store continues appending to the same series after restart
    store2 = PromStore(eng2)
    store2.write(_ts_msg([("__name__", "m_up"), ("job", "a")], [(4.0, 4000)]),
                 snappy=False)
    m = ev.query_range('m_up', 4, 4, 1)
    assert m.S == 1 and float(m.values[0][0]) == 4.0
    eng2.close()


def test_concurrent_ingest_threads(tmp_path):
    import threading
    d = str(tmp_path / "data")
    eng = MitoEngine(EngineConfig(data_dir=d, device="cpu", background_flush=False))
    def worker(wi):
        ing = Ingestor(eng)
        w = CpuWorkload(scale=10, seed=wi)
        w.tagsets = [t.replace(b"host_", b"w%d_host_" % wi) for t in w.tagsets]
        for _ in range(5):
            ing.ingest_lines(w.next_batch(500))
    threads = [threading.Thread(target=worker, args=(i,)) for i in range(4)]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    assert _total(eng) == 4 * 5 * 500
    # WAL replay after concurrent writes is consistent
    eng.close()
    eng2 = MitoEngine(EngineConfig(data_dir=d, device="cpu", background_flush=False))
    assert _total(eng2) == 10000
    eng2.close()


def test_bulk_scatter_append_matches_per_region(tmp_path):
    """K16 bulk path (engine.write_regions_bulk + ops.scatter_append) must
    produce byte-identical memtables + WAL state vs the per-region path."""
    engines = []
    for name, bulk in [("a", False), ("b", True)]:
        eng = MitoEngine(EngineConfig(data_dir=str(tmp_path / name),
                                      device="cpu", background_flush=False))
        ing = Ingestor(eng)
        ing._bulk = bulk  # force the K16 path on CPU (cpu_ref.scatter_append)
        w = CpuWorkload(scale=23, seed=5)
        for _ in range(4):
            ing.ingest_lines(w.next_batch(700))
        ing.engine.commit_wal()
        engines.append(eng)
    a, b = engines
    ra, rb = a.table("cpu").regions, b.table("cpu").regions
    assert [r.memtable.len for r in ra] == [r.memtable.len for r in rb]
    for x, y in zip(ra, rb):
        n = x.memtable.len
        assert (x.memtable.ts[:n] == y.memtable.ts[:n]).all()
        assert (x.memtable.series[:n] == y.memtable.series[:n]).all()
        fx, fy = x.memtable.fields[:, :n], y.memtable.fields[:, :n]
        assert ((fx == fy) | (fx.isnan() & fy.isnan())).all()
        assert x.memtable.min_ts == y.memtable.min_ts
        assert x.memtable.max_ts == y.memtable.max_ts
    # WAL seqs are assigned in a different (but equivalent) region order
    assert sorted(r.last_seq for r in ra) == sorted(r.last_seq for r in rb)
    # WAL replay of the bulk-written log reproduces the data
    b.close()
    b2 = MitoEngine(EngineConfig(data_dir=str(tmp_path / "b"), device="cpu",
                                 background_flush=False))
    assert _total(b2) == 2800
    b2.close()
    a.close()


def test_concurrent_bulk_ingest_with_flush(tmp_path):
    """4 writer threads on the K16 bulk path racing a flusher: no lost
    rows, consistent totals after final flush + reopen."""
    import threading
    eng = MitoEngine(EngineConfig(data_dir=str(tmp_path / "d"), device="cpu",
                                  background_flush=False,
                                  flush_bytes=1 << 20))
    NW, BATCHES, ROWS = 4, 12, 500
    def writer(wi):
        ing = Ingestor(eng)
        ing._bulk = True
        w = CpuWorkload(scale=11, seed=50 + wi)
        w.tagsets = [t.replace(b"host_", b"host_%d_" % wi) for t in w.tagsets]
        for _ in range(BATCHES):
            ing.ingest_lines(w.next_batch(ROWS))
    threads = [threading.Thread(target=writer, args=(i,)) for i in range(NW)]
    stop = threading.Event()
    def flusher():
        while not stop.is_set():
            try:
                eng.flush_all()
            except Exception:
                raise
    ft = threading.Thread(target=flusher)
    for t in threads:
        t.start()
    ft.start()
    for t in threads:
        t.join()
    stop.set()
    ft.join()
    eng.flush_all()
    total = sum(r.num_rows for r in eng.table("cpu").regions)
    assert total == NW * BATCHES * ROWS, total
    d = eng.config.data_dir
    eng.close()
    eng2 = MitoEngine(EngineConfig(data_dir=d, device="cpu",
                                   background_flush=False))
    assert sum(r.num_rows for r in eng2.table("cpu").regions) == NW * BATCHES * ROWS
    eng2.close()


def test_read_externally_written_mito2_sst(tmp_path):
    """Format compatibility: an SST written by ANOTHER writer (pyarrow here,
    standing in for a reference-written file — mito2 layout: field columns,
    time index, __primary_key dict<u32,binary>, __sequence, __op_type) must
    read back through read_sst (ref mito2/src/sst/parquet format.rs)."""
    import pyarrow as pa
    import pyarrow.parquet as pq
    from greptimedb_amd.engine import pk_codec
    from greptimedb_amd.engine.sst import read_sst
    from greptimedb_amd.models.schema import (ColumnSchema, DataType,
                                              SemanticType, TableSchema)
    schema = TableSchema(
        name="ext", columns=[
            ColumnSchema("host", DataType.STRING, SemanticType.TAG, 0),
            ColumnSchema("ts", DataType.TIMESTAMP_MS, SemanticType.TIMESTAMP, 1),
        ], primary_key=["host"])
    pks = [pk_codec.encode_pk(("h1",)), pk_codec.encode_pk(("h2",))]
    idx = pa.array([0, 0, 1, 1], type=pa.uint32())
    pk_col = pa.DictionaryArray.from_arrays(idx, pa.array(pks, type=pa.binary()))
    t = pa.table({
        "v": pa.array([1.0, 2.0, 3.0, 4.0]),
        "ts": pa.array([10, 20, 10, 30], type=pa.timestamp("ms")),
        "__primary_key": pk_col,
        "__sequence": pa.array([1, 2, 3, 4], type=pa.uint64()),
        "__op_type": pa.array([1, 1, 1, 1], type=pa.uint8()),
    })
    p = str(tmp_path / "ext.parquet")
    pq.write_table(t, p, compression="zstd")
    dict_values, indices, ts, fields, seq, str_cols = read_sst(p, schema, ["v"])
    assert dict_values == pks
    assert list(indices) == [0, 0, 1, 1]
    assert list(ts) == [10, 20, 10, 30]
    assert fields.shape == (1, 4) and list(fields[0]) == [1.0, 2.0, 3.0, 4.0]
    assert list(seq) == [1, 2, 3, 4]
    # and the pk decodes with our memcomparable codec
    assert pk_codec.decode_pk(dict_values[0], 1) == ("h1",)


def test_bulk_path_schema_evolution(tmp_path):
    """A new field appearing mid-stream grows the schema on the bulk path
    and WAL payloads stay replayable."""
    eng = MitoEngine(EngineConfig(data_dir=str(tmp_path / "se"), device="cpu",
                                  background_flush=False))
    ing = Ingestor(eng)
    ing._bulk = True
    ing.ingest_lines(b"m,host=a f1=1.0 1000000000\nm,host=b f1=2.0 2000000000\n")
    # new field f2 shows up later
    ing.ingest_lines(b"m,host=a f1=3.0,f2=30.0 3000000000\n")
    eng.commit_wal()
    from greptimedb_amd.query.executor import Executor
    ex = Executor(eng)
    r = ex.execute("SELECT f1, f2 FROM m ORDER BY ts")
    rows = [tuple(t) for t in r.rows()]
    assert rows[0][0] == 1.0 and np.isnan(rows[0][1])
    assert rows[2] == (3.0, 30.0)
    d = eng.config.data_dir
    eng.close()
    eng2 = MitoEngine(EngineConfig(data_dir=d, device="cpu",
                                   background_flush=False))
    ex2 = Executor(eng2)
    r = ex2.execute("SELECT count(*), max(f2) FROM m")
    row = list(r.rows())[0]
    assert int(row[0]) == 3 and float(row[1]) == 30.0
    eng2.close()


def test_flush_visibility_no_gap(tmp_path, monkeypatch):
    """a scan racing a flush must see the flushing
    memtable until its SstBatch is published into sst_cache."""
    import threading
    import time

    from greptimedb_amd.engine import region as region_mod
    from greptimedb_amd.engine import sst as sst_mod

    d = str(tmp_path / "data")
    eng = MitoEngine(EngineConfig(data_dir=d, device="cpu", background_flush=False))
    ing = Ingestor(eng)
    w = CpuWorkload(scale=10)
    ing.ingest_lines(w.next_batch(800))
    region = next(r for r in eng.table("cpu").regions if r.memtable.len > 0)
    n_before = region.num_rows

    gate = threading.Event()
    entered = threading.Event()
    real_write = sst_mod.write_sst

    def slow_write(*a, **k):
        entered.set()
        assert gate.wait(10)
        return real_write(*a, **k)

    monkeypatch.setattr(region_mod.sst_mod, "write_sst", slow_write)
    t = threading.Thread(target=region.flush)
    t.start()
    assert entered.wait(10)
    # mid-flush: memtable swapped, SST not yet published
    srcs = region.scan_sources()
    assert sum(s.n for s in srcs) == n_before
    assert region.num_rows == n_before
    gate.set()
    t.join(10)
    assert sum(s.n for s in region.scan_sources()) == n_before
    eng.close()


def test_row_sequences_monotonic_across_flushes(tmp_path):
    """per-row __sequence ranges of successive SSTs
    must not overlap, and compaction must preserve original sequences."""
    import os

    d = str(tmp_path / "data")
    eng = MitoEngine(EngineConfig(data_dir=d, device="cpu", background_flush=False,
                                  default_regions=1))
    ing = Ingestor(eng)
    w = CpuWorkload(scale=10)
    ranges = []
    for _ in range(4):
        ing.ingest_lines(w.next_batch(300))
        eng.flush_all()
    region = eng.table("cpu").regions[0]
    sst_dir = os.path.join(region.dir, "sst")
    for f in sorted(os.listdir(sst_dir)):
        t = pq.read_table(os.path.join(sst_dir, f))
        seq = t.column("__sequence").to_numpy(zero_copy_only=False)
        ranges.append((int(seq.min()), int(seq.max())))
    ranges.sort()
    for (lo1, hi1), (lo2, hi2) in zip(ranges, ranges[1:]):
        assert hi1 < lo2, f"overlapping SST seq ranges {ranges}"
    # compaction keeps real sequences (not arange-from-0)
    from greptimedb_amd.engine.compaction import Compactor
    Compactor(trigger_file_num=2).compact_region(region)
    files = sorted(os.listdir(sst_dir))
    assert len(files) == 1
    t = pq.read_table(os.path.join(sst_dir, files[0]))
    seq = t.column("__sequence").to_numpy(zero_copy_only=False)
    assert int(seq.max()) == max(hi for _, hi in ranges)
    eng.close()


def test_wal_purge_not_blocked_by_idle_region(tmp_path):
    """an idle (never-written) region must not pin WAL
    segments forever."""
    from greptimedb_amd.models.schema import (ColumnSchema, DataType,
                                              SemanticType, TableSchema)

    d = str(tmp_path / "data")
    eng = MitoEngine(EngineConfig(data_dir=d, device="cpu", background_flush=False,
                                  wal_segment_bytes=1 << 12, default_regions=2))
    # second, never-written table
    eng.create_table(TableSchema(
        name="idle",
        columns=[ColumnSchema("host", DataType.STRING, SemanticType.TAG, 0),
                 ColumnSchema("ts", DataType.TIMESTAMP_MS, SemanticType.TIMESTAMP, 1),
                 ColumnSchema("v", DataType.FLOAT64, SemanticType.FIELD, 2)],
        primary_key=["host"]), n_regions=2)
    ing = Ingestor(eng)
    w = CpuWorkload(scale=10)
    for _ in range(30):
        ing.ingest_lines(w.next_batch(200))
    assert len(eng.wal.segments()) > 1
    eng.flush_all()
    assert len(eng.wal.segments()) == 1  # everything purged but the tail
    eng.close()


def test_pk_codec_multiple_of_8():
    """strings of length 8k must match the reference
    memcomparable crate: full group + marker 9 + all-zero group + marker 0."""
    from greptimedb_amd.engine.pk_codec import decode_string, encode_string

    e8 = encode_string(b"abcdefgh")
    assert e8 == b"abcdefgh\x09" + bytes(8) + b"\x00"
    e16 = encode_string(b"abcdefgh01234567")
    assert e16 == b"abcdefgh\x09" + b"01234567\x09" + bytes(8) + b"\x00"
    for s in (b"", b"a", b"abcdefg", b"abcdefgh", b"abcdefgh0", b"x" * 16, b"x" * 17):
        enc = encode_string(s)
        dec, off = decode_string(enc, 0)
        assert dec == s and off == len(enc)
    # ordering property preserved
    vals = [b"", b"a", b"abcdefgh", b"abcdefgh\x00", b"abcdefghi", b"b"]
    assert sorted(vals) == [v for _, v in sorted((encode_string(v), v) for v in vals)]


def test_wal_sharded_replay_and_purge(tmp_path):
    """Sharded WAL: parallel writers, global seq order at
    replay, per-shard purge; shard-count changes between runs replay fine."""
    d = str(tmp_path / "data")
    from greptimedb_amd.engine.engine import EngineConfig, MitoEngine
    eng = MitoEngine(EngineConfig(data_dir=d, device="cpu", background_flush=False,
                                  wal_shards=4, wal_segment_bytes=1 << 12))
    ing = Ingestor(eng)
    w = CpuWorkload(scale=20)
    for _ in range(20):
        ing.ingest_lines(w.next_batch(200))
    import os as _os
    shard_files = eng.wal.segments()
    assert any(".s1." in f or ".s2." in f or ".s3." in f for f in shard_files)
    eng.close()
    # reopen with a DIFFERENT shard count: replay must still see all rows
    eng2 = MitoEngine(EngineConfig(data_dir=d, device="cpu", background_flush=False,
                                   wal_shards=2))
    assert _total(eng2) == 4000
    # seq-ordered replay: WAL-assigned last_seq must be the global max
    assert eng2.wal.next_seq > 1
    eng2.flush_all()
    assert _total(eng2) == 4000
    # purge leaves at most one (tail) segment per shard
    for sh in range(2):
        assert len(eng2.wal.segments(sh)) <= 1
    eng2.close()


def test_wal_sharded_concurrent_workers(tmp_path):
    import threading as _th
    d = str(tmp_path / "data")
    from greptimedb_amd.engine.engine import EngineConfig, MitoEngine
    eng = MitoEngine(EngineConfig(data_dir=d, device="cpu", background_flush=False,
                                  wal_shards=4))
    def work(wi):
        ing = Ingestor(eng)
        w = CpuWorkload(scale=10, seed=wi)
        w.tagsets = [t.replace(b"host_", b"host_%d_" % wi) for t in w.tagsets]
        for _ in range(10):
            ing.ingest_lines(w.next_batch(100))
    ts = [_th.Thread(target=work, args=(i,)) for i in range(4)]
    for t in ts:
        t.start()
    for t in ts:
        t.join()
    assert _total(eng) == 4000
    eng.close()
    eng2 = MitoEngine(EngineConfig(data_dir=d, device="cpu", background_flush=False,
                                   wal_shards=4))
    assert _total(eng2) == 4000
    eng2.close()


def test_gc_orphan_ssts(tmp_path):
    """Orphan SST files (crashed flush leftovers) are collected after the
    grace period (reference mito2 gc.rs)."""
    import os as _os
    d = str(tmp_path / "data")
    eng = MitoEngine(EngineConfig(data_dir=d, device="cpu", background_flush=False))
    ing = Ingestor(eng)
    w = CpuWorkload(scale=5)
    ing.ingest_lines(w.next_batch(200))
    eng.flush_all()
    region = next(r for st in eng.tables.values() for r in st.regions
                  if r.manifest.files)
    sdir = _os.path.join(region.dir, "sst")
    orphan = _os.path.join(sdir, "deadbeef00.parquet")
    open(orphan, "wb").write(b"not a real parquet")
    _os.utime(orphan, (0, 0))   # ancient mtime → past any grace period
    fresh = _os.path.join(sdir, "cafebabe01.parquet")
    open(fresh, "wb").write(b"in-flight flush artifact")
    n = eng.gc_orphan_ssts(grace_s=60.0)
    assert n == 1
    assert not _os.path.exists(orphan)
    assert _os.path.exists(fresh)          # grace period spares it
    # manifest-referenced files untouched
    for fid in region.manifest.files:
        assert _os.path.exists(_os.path.join(sdir, f"{fid}.parquet"))
    eng.close()


def test_gc_runs_periodically(tmp_path):
    """The background flush thread doubles as the GC ticker: orphans are
    collected without any ADMIN call when gc_interval_s elapses."""
    import os as _os
    import time as _time
    d = str(tmp_path / "data")
    eng = MitoEngine(EngineConfig(data_dir=d, device="cpu",
                                  background_flush=True,
                                  gc_interval_s=0.2, gc_grace_s=60.0))
    ing = Ingestor(eng)
    w = CpuWorkload(scale=5)
    ing.ingest_lines(w.next_batch(200))
    eng.flush_all()
    region = next(r for st in eng.tables.values() for r in st.regions
                  if r.manifest.files)
    orphan = _os.path.join(region.dir, "sst", "deadbeef00.parquet")
    open(orphan, "wb").write(b"leftover")
    _os.utime(orphan, (0, 0))
    deadline = _time.monotonic() + 5.0
    while _os.path.exists(orphan) and _time.monotonic() < deadline:
        _time.sleep(0.05)
    assert not _os.path.exists(orphan)
    eng.close()


def test_gorilla_compress_table_roundtrip(tmp_path):
    """K20 cold tier: ADMIN compress_table packs resident batches into
    Gorilla blocks; scans transparently re-materialize and results are
    bit-identical."""
    from greptimedb_amd.query.executor import Executor
    d = str(tmp_path / "data")
    eng = MitoEngine(EngineConfig(data_dir=d, device="cpu", background_flush=False))
    ing = Ingestor(eng)
    w = CpuWorkload(scale=20)
    for _ in range(3):
        ing.ingest_lines(w.next_batch(5000))
    eng.flush_all()
    ex = Executor(eng)
    before = ex.execute("SELECT hostname, count(*) c, avg(usage_user) a,"
                        " max(usage_system) m FROM cpu GROUP BY hostname"
                        " ORDER BY hostname").rows()
    r = ex.execute("ADMIN compress_table('cpu')")
    b0, b1 = int(r.columns[0][0]), int(r.columns[1][0])
    assert 0 < b1 < b0, (b0, b1)
    # batches are packed now
    assert any(b.ts is None for st in eng.tables.values()
               for reg in st.regions for b in reg.sst_cache.values())
    after = ex.execute("SELECT hostname, count(*) c, avg(usage_user) a,"
                       " max(usage_system) m FROM cpu GROUP BY hostname"
                       " ORDER BY hostname").rows()
    assert after == before
    eng.close()


def test_auto_created_fields_survive_reopen_after_wal_purge(tmp_path):
    """Round-2 fix: influx auto-ALTERed numeric fields persist in the
    region meta sidecar — reopen restores values even when the WAL entries
    that introduced the columns are gone."""
    from greptimedb_amd.query.executor import Executor
    d = str(tmp_path / "data")
    eng = MitoEngine(EngineConfig(data_dir=d, device="cpu", background_flush=False))
    ing = Ingestor(eng)
    w = CpuWorkload(scale=10)
    ing.ingest_lines(w.next_batch(2000))
    eng.flush_all()   # purges the WAL
    exp = Executor(eng).execute(
        "SELECT sum(usage_user) AS s, count(usage_idle) AS c FROM cpu").rows()
    fields_before = list(eng.table("cpu").regions[0].field_names)
    eng.close()
    eng2 = MitoEngine(EngineConfig(data_dir=d, device="cpu", background_flush=False))
    assert list(eng2.table("cpu").regions[0].field_names) == fields_before
    got = Executor(eng2).execute(
        "SELECT sum(usage_user) AS s, count(usage_idle) AS c FROM cpu").rows()
    assert got == exp
    eng2.close()


def test_ttl_expires_ssts(tmp_path):
    """SSTs wholly past the table ttl are dropped (reference: mito2
    compaction ttl expiry); fresh files survive."""
    from greptimedb_amd.query.executor import Executor
    import time as _time
    eng = MitoEngine(EngineConfig(data_dir=str(tmp_path / "d"), device="cpu",
                                  background_flush=False))
    ex = Executor(eng)
    ex.execute("CREATE TABLE tt (ts TIMESTAMP TIME INDEX, h STRING "
               "PRIMARY KEY, v DOUBLE) WITH (ttl='1h')")
    now = int(_time.time() * 1000)
    ex.execute(f"INSERT INTO tt VALUES (1000,'old',1), ({now},'new',2)")
    eng.flush_all()
    files_before = sum(len(r.manifest.files) for r in eng.table("tt").regions)
    assert files_before >= 2
    removed = eng.apply_ttl()
    # only files whose max_ts is fully past the cutoff drop; the file with
    # the fresh row survives
    assert removed >= 1
    rows = ex.execute("SELECT h FROM tt").rows()
    assert ("new",) in [tuple(r) for r in rows]
    assert ("old",) not in [tuple(r) for r in rows]
    # ADMIN surface
    r = ex.execute("ADMIN apply_ttl()")
    assert r.names == ["files_removed"]
    eng.close()


def test_cold_tier_auto_compress(tmp_path):
    """Idle SST batches Gorilla-pack via compress_cold and transparently
    re-decode on the next scan (K20 cold tier policy)."""
    from greptimedb_amd.query.executor import Executor
    eng = MitoEngine(EngineConfig(data_dir=str(tmp_path / "d"), device="cpu",
                                  background_flush=False))
    ex = Executor(eng)
    ex.execute("CREATE TABLE ct (ts TIMESTAMP TIME INDEX, h STRING "
               "PRIMARY KEY, v DOUBLE)")
    ex.execute("INSERT INTO ct VALUES (1000,'a',1.5),(2000,'a',2.5),"
               "(3000,'b',3.5)")
    eng.flush_all()
    before = ex.execute("SELECT h, v FROM ct ORDER BY ts").rows()
    # mark every batch ancient, then sweep
    for st in eng.tables.values():
        for r in st.regions:
            for b in r.sst_cache.values():
                b.last_access = -1e9
    n = eng.compress_cold(age_s=1.0)
    assert n >= 1
    packed = [b for st in eng.tables.values() for r in st.regions
              for b in r.sst_cache.values() if b.ts is None]
    assert packed
    # scans re-decode transparently and results are identical
    after = ex.execute("SELECT h, v FROM ct ORDER BY ts").rows()
    assert after == before
    assert all(b.ts is not None for b in packed)  # hot again
    eng.close()


def test_global_write_buffer_flush(tmp_path):
    """Node-wide memtable cap flushes the largest regions even when no
    single region hits its own threshold (reference:
    WriteBufferManagerImpl global accounting)."""
    eng = MitoEngine(EngineConfig(
        data_dir=str(tmp_path / "d"), device="cpu", background_flush=False,
        flush_bytes=1 << 30,                 # per-region threshold never hit
        global_write_buffer_bytes=200_000))  # tiny node-wide cap
    ing = Ingestor(eng)
    w = CpuWorkload(scale=20)
    for _ in range(10):
        ing.ingest_lines(w.next_batch(2000))
    total_mem = sum(r.memtable.bytes_used for st in eng.tables.values()
                    for r in st.regions)
    assert total_mem <= 200_000 + (1 << 16), total_mem
    files = sum(len(r.manifest.files) for st in eng.tables.values()
                for r in st.regions)
    assert files >= 1    # data went to SSTs, not lost
    assert _total(eng) == 20000
    eng.close()
