"""Line-protocol parser + WAL writer (csrc/native.cpp)."""

import os

import numpy as np
import pytest
import torch  # noqa: F401  (loads libc10 before the extension)

from greptimedb_amd import _native


def test_parse_basic():
    p = _native.LineParser()
    s, ts, fields, new = p.parse(
        b"cpu,hostname=h1,region=r f1=1.25,f2=-3i,f3=2e3 1451606400000000000\n"
        b"cpu,hostname=h2,region=r f1=7,f2=9i 1451606401000000000\n")
    assert list(s) == [0, 1]
    assert ts[0] == 1451606400000000000 and ts[1] == 1451606401000000000
    assert fields["f1"][0] == 1.25 and fields["f1"][1] == 7.0
    assert fields["f2"][0] == -3.0
    assert fields["f3"][0] == 2000.0
    assert np.isnan(fields["f3"][1])  # missing field → NaN
    assert [t[0] for t in new] == [0, 1]
    assert new[0][1] == b"cpu,hostname=h1,region=r"


def test_parse_interning():
    p = _native.LineParser()
    p.parse(b"m,a=1 f=1 1\n")
    s, ts, fields, new = p.parse(b"m,a=1 f=2 2\nm,a=2 f=3 3\n")
    assert list(s) == [0, 1]
    assert len(new) == 1 and new[0][0] == 1
    assert p.num_series() == 2


def test_parse_bool_string_fields():
    p = _native.LineParser()
    s, ts, fields, _ = p.parse(b'm,a=1 b=true,s="xy",v=5 9\n')
    assert fields["b"][0] == 1.0
    assert np.isnan(fields["s"][0])
    assert fields["v"][0] == 5.0


def test_parse_no_timestamp_and_garbage():
    p = _native.LineParser()
    s, ts, fields, _ = p.parse(b"m,a=1 f=1\n# comment\n\nbad-line-no-space\n")
    assert len(s) == 1
    assert ts[0] == 0


def test_wal_roundtrip(tmp_path):
    path = str(tmp_path / "0001.wal")
    w = _native.WalWriter()
    w.open_segment(path)
    w.append(7, 1, b"alpha")
    w.append(8, 2, b"beta" * 1000)
    w.commit(True)
    w.append(7, 3, b"gamma")
    w.commit(False)
    w.close_segment()
    ents = _native.wal_read_segment(path)
    assert [(r, s) for r, s, _ in ents] == [(7, 1), (8, 2), (7, 3)]
    assert ents[1][2] == b"beta" * 1000


def test_wal_torn_tail(tmp_path):
    path = str(tmp_path / "x.wal")
    w = _native.WalWriter()
    w.open_segment(path)
    w.append(1, 1, b"ok")
    w.commit(True)
    w.close_segment()
    with open(path, "ab") as f:
        f.write(b"\xff\x00\x00\x00garbage-torn")
    ents = _native.wal_read_segment(path)
    assert len(ents) == 1 and ents[0][2] == b"ok"


def test_wal_corrupt_crc(tmp_path):
    path = str(tmp_path / "y.wal")
    w = _native.WalWriter()
    w.open_segment(path)
    w.append(1, 1, b"aaaa")
    w.append(1, 2, b"bbbb")
    w.commit(True)
    w.close_segment()
    data = bytearray(open(path, "rb").read())
    data[10] ^= 0xFF  # flip a byte inside the first frame body
    open(path, "wb").write(bytes(data))
    ents = _native.wal_read_segment(path)
    assert len(ents) == 0  # stops at first corrupt frame


def test_parse_throughput_sanity():
    """Ingest parse must clearly exceed the reference's 327k rows/s node rate."""
    import time
    from greptimedb_amd.models.tsbs import CpuWorkload
    w = CpuWorkload(scale=100)
    p = _native.LineParser()
    p.parse(w.next_batch(3000))
    batches = [w.next_batch(3000) for _ in range(10)]
    t0 = time.perf_counter()
    n = 0
    for b in batches:
        s, *_ = p.parse(b)
        n += len(s)
    rate = n / (time.perf_counter() - t0)
    assert rate > 500_000, f"parse too slow: {rate:.0f} rows/s"
