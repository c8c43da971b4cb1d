#!/usr/bin/env python3
"""Build a reference-style mito2 SST fixture (tests/fixtures/).

This deliberately re-implements the memcomparable pk encoding and the
RegionMetadata JSON INLINE (no imports from greptimedb_amd) so the
committed fixture cross-checks the engine's reader/writer against an
independent rendering of the format spec:
  - sst/parquet/format.rs:15-27  column layout
  - mito-codec/src/row_converter  memcomparable dense pk
  - sst/parquet.rs:43            greptime:metadata key-value entry
"""

import json
import os

import numpy as np
import pyarrow as pa
import pyarrow.parquet as pq


def memcmp_string(b: bytes) -> bytes:
    # memcomparable crate: 8-byte groups, marker 9 = continue, else = len;
    # exact multiples of 8 end with an empty all-pad group marker 0
    out = bytearray()
    if not b:
        return bytes(8) + b"\x00"
    i = 0
    while i < len(b):
        g = b[i:i + 8]
        i += 8
        if i < len(b):
            out += g + b"\x09"
        elif len(g) == 8:
            out += g + b"\x09" + bytes(8) + b"\x00"
        else:
            out += g + bytes(8 - len(g)) + bytes([len(g)])
    return bytes(out)


def encode_pk(tags):
    out = bytearray()
    for v in tags:
        if v is None:
            out += b"\x00"
        else:
            out += b"\x01" + memcmp_string(v.encode())
    return bytes(out)


def main():
    here = os.path.dirname(os.path.abspath(__file__))
    fdir = os.path.join(here, "..", "tests", "fixtures")
    os.makedirs(fdir, exist_ok=True)
    # 3 series × 4 points; hostname length 8 exercises the 8-byte group edge
    series = [("hostname", "regionx"), ("hosty", "regiony"), ("hostzzzz", "rz")]
    rows = []
    for si, tags in enumerate(series):
        for p in range(4):
            rows.append((tags, 1451606400000 + p * 10_000, 10.0 * si + p,
                         float(si), si * 4 + p))
    rows.sort(key=lambda r: (encode_pk(r[0]), r[1], r[4]))
    pks = [encode_pk(r[0]) for r in rows]
    uniq = sorted(set(pks))
    idx = [uniq.index(p) for p in pks]

    cols = {
        "usage_user": pa.array([r[2] for r in rows], type=pa.float64()),
        "usage_system": pa.array([r[3] for r in rows], type=pa.float64()),
        "ts": pa.array(np.array([r[1] for r in rows], dtype="int64"),
                       type=pa.int64()).cast(pa.timestamp("ms")),
        "__primary_key": pa.DictionaryArray.from_arrays(
            pa.array(np.array(idx, dtype=np.uint32), type=pa.uint32()),
            pa.array(uniq, type=pa.binary())),
        "__sequence": pa.array(np.array([r[4] for r in rows], dtype="uint64"),
                               type=pa.uint64()),
        "__op_type": pa.array(np.ones(len(rows), dtype="uint8"),
                              type=pa.uint8()),
    }
    meta = {
        "column_metadatas": [
            {"column_schema": {"name": "host", "data_type": {"String": {"size_type": "Utf8"}},
                               "is_nullable": True, "is_time_index": False,
                               "default_constraint": None, "metadata": {}},
             "semantic_type": "Tag", "column_id": 0},
            {"column_schema": {"name": "dc", "data_type": {"String": {"size_type": "Utf8"}},
                               "is_nullable": True, "is_time_index": False,
                               "default_constraint": None, "metadata": {}},
             "semantic_type": "Tag", "column_id": 1},
            {"column_schema": {"name": "ts", "data_type": {"Timestamp": {"Millisecond": None}},
                               "is_nullable": False, "is_time_index": True,
                               "default_constraint": None, "metadata": {}},
             "semantic_type": "Timestamp", "column_id": 2},
            {"column_schema": {"name": "usage_user", "data_type": {"Float64": {}},
                               "is_nullable": True, "is_time_index": False,
                               "default_constraint": None, "metadata": {}},
             "semantic_type": "Field", "column_id": 3},
            {"column_schema": {"name": "usage_system", "data_type": {"Float64": {}},
                               "is_nullable": True, "is_time_index": False,
                               "default_constraint": None, "metadata": {}},
             "semantic_type": "Field", "column_id": 4},
        ],
        "primary_key": [0, 1],
        "region_id": (77 << 32) | 0,
        "schema_version": 0,
        "primary_key_encoding": "dense",
    }
    table = pa.Table.from_arrays(list(cols.values()), names=list(cols))
    table = table.replace_schema_metadata(
        {b"greptime:metadata": json.dumps(meta)})
    out = os.path.join(fdir, "mito2_ref_style.parquet")
    pq.write_table(table, out, row_group_size=102400, compression="zstd")
    print("wrote", out, os.path.getsize(out), "bytes")


if __name__ == "__main__":
    main()
