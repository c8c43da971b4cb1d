"""mito2 SST format compatibility.

The fixture tests/fixtures/mito2_ref_style.parquet was produced by
scripts/make_mito2_fixture.py — an INDEPENDENT rendering of the format
spec (inline memcomparable + RegionMetadata JSON, no greptimedb_amd
imports) — so these tests cross-check reader and writer against the spec,
not against themselves. Reference: sst/parquet/format.rs:15-27,
flat_format.rs, parquet.rs:43, mito-codec row_converter.
"""

import json
import os

import numpy as np
import pyarrow.parquet as pq
import pytest

from greptimedb_amd.engine import pk_codec, sst as sst_mod
from greptimedb_amd.models.schema import (ColumnSchema, DataType,
                                          SemanticType, TableSchema)

FIXTURE = os.path.join(os.path.dirname(__file__), "fixtures",
                       "mito2_ref_style.parquet")


def _schema():
    return TableSchema(name="cpu_fix", columns=[
        ColumnSchema("host", DataType.STRING, SemanticType.TAG, 0),
        ColumnSchema("dc", DataType.STRING, SemanticType.TAG, 1),
        ColumnSchema("ts", DataType.TIMESTAMP_MS, SemanticType.TIMESTAMP, 2),
        ColumnSchema("usage_user", DataType.FLOAT64, SemanticType.FIELD, 3),
        ColumnSchema("usage_system", DataType.FLOAT64, SemanticType.FIELD, 4),
    ], primary_key=["host", "dc"])


def test_read_reference_style_fixture():
    dict_values, indices, ts, fields, seq, str_cols = sst_mod.read_sst(
        FIXTURE, _schema(), ["usage_user", "usage_system"])
    assert len(ts) == 12
    # pk bytes decode with OUR codec (byte-level compat incl. len-8 tags)
    tags = [pk_codec.decode_pk(pk, 2) for pk in dict_values]
    assert ("hostname", "regionx") in tags
    assert ("hostzzzz", "rz") in tags
    # and OUR encoder reproduces the fixture's bytes exactly
    for pk, tg in zip(dict_values, tags):
        assert pk_codec.encode_pk(tg) == pk
    assert str_cols == {}
    assert int(seq.max()) == 11


def test_fixture_region_metadata_parses():
    md = sst_mod.read_region_metadata(FIXTURE)
    assert md["primary_key"] == [0, 1]
    assert md["primary_key_encoding"] == "dense"
    names = [c["column_schema"]["name"] for c in md["column_metadatas"]]
    assert names == ["host", "dc", "ts", "usage_user", "usage_system"]
    sems = [c["semantic_type"] for c in md["column_metadatas"]]
    assert sems == ["Tag", "Tag", "Timestamp", "Field", "Field"]


def test_engine_opens_fixture_as_region(tmp_path):
    """Drop the fixture into a region dir + manifest: the engine must scan
    it (open path = real reference-file ingestion shape)."""
    from greptimedb_amd.engine.manifest import Manifest
    from greptimedb_amd.engine.region import Region
    rdir = tmp_path / "region"
    (rdir / "sst").mkdir(parents=True)
    import shutil
    shutil.copy(FIXTURE, rdir / "sst" / "fix01.parquet")
    man = Manifest(str(rdir / "manifest"))
    man.commit({"kind": "edit", "files_to_add": [{
        "file_id": "fix01", "level": 0, "min_ts": 1451606400000,
        "max_ts": 1451606430000, "num_rows": 12, "file_size": 5177,
        "seq_max": 11}], "files_to_remove": []})
    region = Region(123, _schema(), str(rdir), device="cpu")
    assert region.num_rows == 12
    srcs = region.scan_sources()
    assert sum(s.n for s in srcs) == 12
    assert sorted(region.series.tag_values) == [
        ("hostname", "regionx"), ("hosty", "regiony"), ("hostzzzz", "rz")]


def test_our_writer_matches_fixture_layout(tmp_path):
    """Writer-side: same data через write_sst must produce the same column
    layout, dictionary pk bytes, kv-metadata key and compression."""
    schema = _schema()
    series_tags = [("hostname", "regionx"), ("hosty", "regiony"),
                   ("hostzzzz", "rz")]
    pks = [pk_codec.encode_pk(t) for t in series_tags]
    order = np.argsort([pks[i] for i in range(3)])
    rows = []
    for si in order:
        for p in range(4):
            rows.append((int(si), 1451606400000 + p * 10_000,
                         10.0 * si + p, float(si)))
    se = np.array([r[0] for r in rows], dtype=np.int32)
    ts = np.array([r[1] for r in rows], dtype=np.int64)
    f = np.stack([np.array([r[2] for r in rows]),
                  np.array([r[3] for r in rows])])
    seq = np.arange(12, dtype=np.int64)
    out = str(tmp_path / "ours.parquet")
    sst_mod.write_sst(out, schema, pks, se, ts, f, seq,
                      ["usage_user", "usage_system"],
                      region_id=(77 << 32) | 0)
    ref = pq.read_table(FIXTURE)
    ours = pq.read_table(out)
    assert ours.column_names == ref.column_names
    assert [str(f_.type) for f_ in ours.schema] == \
        [str(f_.type) for f_ in ref.schema]
    def _dict(t):
        col = t.column("__primary_key").combine_chunks()
        if hasattr(col, "chunk"):
            col = col.chunk(0)
        return sorted(v.as_py() for v in col.dictionary)

    assert _dict(ours) == _dict(ref)
    m_ours = pq.read_metadata(out)
    assert m_ours.row_group(0).column(0).compression == "ZSTD"
    md = sst_mod.read_region_metadata(out)
    ref_md = sst_mod.read_region_metadata(FIXTURE)
    assert md["primary_key"] == ref_md["primary_key"]
    assert [c["column_schema"]["name"] for c in md["column_metadatas"]] == \
        [c["column_schema"]["name"] for c in ref_md["column_metadatas"]]


def test_flat_format_roundtrip(tmp_path):
    """flat_format.rs layout: raw tag columns + internal columns; our
    reader must reconstruct identical rows from both layouts."""
    schema = _schema()
    tags = [("alpha", "dc1"), ("beta", "dc2")]
    pks = [pk_codec.encode_pk(t) for t in tags]
    se = np.array([0, 0, 1, 1], dtype=np.int32)
    ts = np.array([1000, 2000, 1000, 2000], dtype=np.int64)
    f = np.stack([np.arange(4, dtype=np.float64),
                  np.arange(4, dtype=np.float64) * 10])
    seq = np.arange(4, dtype=np.int64)
    pk_path = str(tmp_path / "pk.parquet")
    flat_path = str(tmp_path / "flat.parquet")
    sst_mod.write_sst(pk_path, schema, pks, se, ts, f, seq,
                      ["usage_user", "usage_system"])
    sst_mod.write_sst(flat_path, schema, pks, se, ts, f, seq,
                      ["usage_user", "usage_system"], flat=True)
    t = pq.read_table(flat_path)
    assert t.column_names[:2] == ["host", "dc"]          # raw tag cols first
    assert t.column_names[-3:] == ["__primary_key", "__sequence", "__op_type"]
    r1 = sst_mod.read_sst(pk_path, schema, ["usage_user", "usage_system"])
    r2 = sst_mod.read_sst(flat_path, schema, ["usage_user", "usage_system"])
    assert list(r1[0]) == list(r2[0])        # same pk dictionary
    np.testing.assert_array_equal(r1[2], r2[2])
    np.testing.assert_array_equal(r1[3], r2[3])
    assert r1[5] == {} and r2[5] == {}       # tag cols NOT misread as strings
