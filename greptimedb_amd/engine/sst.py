"""SST layer: mito2-format Parquet files + GPU-resident batch cache.

Reference parity: src/mito2/src/sst/parquet/ — primary-key format
`field..., time index, __primary_key (dict<u32,binary>), __sequence (u64),
__op_type (u8)` (format.rs:15-27), row group 102400 rows (parquet.rs:56),
ZSTD compression (parquet.rs:607). Files are written sorted by
(__primary_key, ts, seq) like the reference flush/compaction output.

MI355X design: the flush path already has sorted device tensors in HBM, so
every flushed SST keeps a device-resident `SstBatch` (288 GB per GPU makes
"keep everything hot" the default caching policy; the parquet file is the
durable/spill copy). Scans therefore hit HBM, not parquet decode — the
reference's page-cache/row-group-cache (mito2 cache.rs) collapses into this
one structure.
"""

from __future__ import annotations

import os
import uuid
from dataclasses import dataclass

import numpy as np
import pyarrow as pa
import pyarrow.parquet as pq
import torch

from greptimedb_amd.models.schema import TableSchema

ROW_GROUP_SIZE = 102400
OP_PUT = 1

# mito2 stores the region metadata JSON in the parquet key-value metadata
# under this key (reference sst/parquet.rs:43 PARQUET_METADATA_KEY)
PARQUET_METADATA_KEY = b"greptime:metadata"


def _dt_json(c) -> dict:
    """ConcreteDataType serde shape (reference src/datatypes types serde,
    e.g. manifest/action.rs:663 examples)."""
    from greptimedb_amd.models.schema import DataType
    m = {
        DataType.STRING: {"String": {"size_type": "Utf8"}},
        DataType.BINARY: {"Binary": None},
        DataType.JSON: {"String": {"size_type": "Utf8"}},
        DataType.VECTOR: {"Binary": None},
        DataType.FLOAT64: {"Float64": {}},
        DataType.FLOAT32: {"Float32": {}},
        DataType.INT64: {"Int64": {}},
        DataType.INT32: {"Int32": {}},
        DataType.UINT64: {"UInt64": {}},
        DataType.BOOL: {"Boolean": None},
        DataType.TIMESTAMP_MS: {"Timestamp": {"Millisecond": None}},
        DataType.TIMESTAMP_NS: {"Timestamp": {"Nanosecond": None}},
    }
    return m.get(c.dtype, {"Float64": {}})


def region_metadata_json(schema: TableSchema, region_id: int,
                         extra_fields: list[str] | None = None,
                         str_fields: list[str] | None = None) -> str:
    """RegionMetadata JSON exactly as the reference serializes it
    (store-api/src/metadata.rs serde; examples in manifest/action.rs:663)."""
    import json
    from greptimedb_amd.models.schema import SemanticType
    sem_name = {SemanticType.TAG: "Tag", SemanticType.FIELD: "Field",
                SemanticType.TIMESTAMP: "Timestamp"}
    cols = []
    next_id = 0
    name_to_id = {}
    for c in schema.columns:
        cols.append({
            "column_schema": {
                "name": c.name,
                "data_type": _dt_json(c),
                "is_nullable": bool(c.nullable) and
                               c.semantic != SemanticType.TIMESTAMP,
                "is_time_index": c.semantic == SemanticType.TIMESTAMP,
                "default_constraint": None,
                "metadata": {},
            },
            "semantic_type": sem_name[c.semantic],
            "column_id": c.column_id,
        })
        name_to_id[c.name] = c.column_id
        next_id = max(next_id, c.column_id + 1)
    for name in (extra_fields or []):
        if name in name_to_id:
            continue
        dt = {"String": {"size_type": "Utf8"}} if name in (str_fields or []) \
            else {"Float64": {}}
        cols.append({
            "column_schema": {"name": name, "data_type": dt,
                              "is_nullable": True, "is_time_index": False,
                              "default_constraint": None, "metadata": {}},
            "semantic_type": "Field",
            "column_id": next_id,
        })
        next_id += 1
    return json.dumps({
        "column_metadatas": cols,
        "primary_key": [name_to_id[t] for t in schema.primary_key],
        "region_id": region_id,
        "schema_version": 0,
        "primary_key_encoding": "dense",
    })


@dataclass
class SstMeta:
    file_id: str
    level: int
    min_ts: int
    max_ts: int
    num_rows: int
    file_size: int
    seq_max: int

    def to_dict(self):
        return self.__dict__.copy()

    @staticmethod
    def from_dict(d):
        return SstMeta(**d)


class SstBatch:
    """Device-resident, (series, ts, seq)-sorted columns of one SST."""

    def __init__(self, ts: torch.Tensor, series: torch.Tensor, fields: torch.Tensor,
                 seq: torch.Tensor | None, min_ts: int, max_ts: int,
                 field_names: list[str]):
        self.ts = ts
        self.series = series
        self.fields = fields  # [nf, n]
        self.seq = seq
        self.min_ts = min_ts
        self.max_ts = max_ts
        self.field_names = list(field_names)
        self.str_cols: dict = {}     # host string columns (log fields)
        self.text_index: dict = {}   # fulltext SegmentPostings per column
        import threading as _th
        import time as _time
        self._tier_lock = _th.Lock()  # compress vs decode exclusion
        self.last_access = _time.monotonic()

    @property
    def n(self) -> int:
        if self.ts is None and getattr(self, "_gorilla", None):
            return self._gorilla["__n"]
        return self.ts.numel()

    # ---------------------------------------------------- K20 cold tier
    def compress(self) -> int:
        """Pack ts+fields into Gorilla blocks (engine/gorilla.py), freeing
        the uncompressed HBM tensors. Returns bytes now used. Scans call
        ensure_decoded() which re-materializes via the GPU decode kernel —
        the resident-cold-tier policy the reference approximates with its
        page cache, except the compressed copy stays in HBM."""
        with self._tier_lock:
            return self._compress_locked()

    def _compress_locked(self) -> int:
        if self.ts is None:
            return sum(g.nbytes for k, g in self._gorilla.items()
                       if k != "__n")
        from greptimedb_amd.engine import gorilla
        ts_h = self.ts.cpu().numpy()
        packs = {"__n": int(self.ts.numel())}
        # ts packed ONCE; field blocks skip the ts section (ts_bits=0)
        packs["__ts"] = gorilla.GorillaBatch(ts_h, np.zeros(len(ts_h)))
        for i in range(self.fields.shape[0]):
            packs[f"f{i}"] = gorilla.GorillaBatch(
                ts_h, self.fields[i].cpu().numpy(), pack_ts=False)
        self._gorilla = packs
        self._series_h = self.series.cpu().numpy()
        self._seq_h = self.seq.cpu().numpy() if self.seq is not None else None
        self.ts = None
        self.series = None
        self.fields = None
        self.seq = None
        return sum(g.nbytes for k, g in packs.items() if k != "__n")

    def ensure_decoded(self, device) -> "SstBatch":
        import time as _time
        self.last_access = _time.monotonic()
        if self.ts is not None or not getattr(self, "_gorilla", None):
            return self
        with self._tier_lock:
            return self._decode_locked(device)

    def _decode_locked(self, device) -> "SstBatch":
        if self.ts is not None or not getattr(self, "_gorilla", None):
            return self
        import torch as _torch
        packs = self._gorilla
        nf = len([k for k in packs if k not in ("__n", "__ts")])
        n = packs["__n"]
        fields = _torch.empty((nf, n), dtype=_torch.float64, device=device)
        ts, _zero = packs["__ts"].decode(device)
        for i in range(nf):
            _t, v = packs[f"f{i}"].decode(device)
            fields[i] = v
        self.ts = ts.to(device)
        self.fields = fields
        self.series = _torch.as_tensor(self._series_h).to(device)
        if self._seq_h is not None:
            self.seq = _torch.as_tensor(self._seq_h).to(device)
        self._gorilla = None
        return self


def write_sst(path: str, schema: TableSchema, pks: list[bytes],
              series: np.ndarray, ts_ms: np.ndarray, fields: np.ndarray,
              seq: np.ndarray, field_names: list[str],
              str_cols: dict[str, np.ndarray] | None = None,
              region_id: int = 0, flat: bool = False) -> SstMeta:
    """Write one mito2-format parquet SST. Inputs are host arrays sorted by
    (series, ts); `pks[code]` gives the encoded primary key per local code.
    flat=True additionally stores decoded tag columns up front (reference
    sst/parquet/flat_format.rs: `pk cols..., fields..., time index,
    __primary_key, __sequence, __op_type`)."""
    n = len(ts_ms)
    # compact dictionary: unique codes in appearance order
    uniq, inv = np.unique(series, return_inverse=True)
    dict_values = pa.array([pks[int(c)] for c in uniq], type=pa.binary())
    pk_col = pa.DictionaryArray.from_arrays(pa.array(inv.astype(np.uint32), type=pa.uint32()), dict_values)

    cols, names = [], []
    if flat and schema.primary_key:
        from greptimedb_amd.engine import pk_codec
        tag_names = schema.primary_key
        decoded = [pk_codec.decode_pk(pks[int(c)], len(tag_names))
                   for c in uniq]
        for ti, tn in enumerate(tag_names):
            vals = np.array([decoded[int(i)][ti] for i in inv], dtype=object)
            cols.append(pa.array(list(vals), type=pa.string()))
            names.append(tn)
    for i, fn in enumerate(field_names):
        cols.append(pa.array(fields[i], type=pa.float64()))
        names.append(fn)
    for fn, vals in (str_cols or {}).items():
        vl = list(vals)
        is_bin = any(isinstance(v, (bytes, bytearray)) for v in vl)
        cols.append(pa.array(vl, type=pa.binary() if is_bin else pa.string()))
        names.append(fn)
    cols.append(pa.array(ts_ms, type=pa.timestamp("ms")))
    names.append(schema.time_index.name)
    cols.append(pk_col)
    names.append("__primary_key")
    cols.append(pa.array(seq.astype(np.uint64), type=pa.uint64()))
    names.append("__sequence")
    cols.append(pa.array(np.full(n, OP_PUT, dtype=np.uint8), type=pa.uint8()))
    names.append("__op_type")

    table = pa.Table.from_arrays(cols, names=names)
    meta_json = region_metadata_json(
        schema, region_id, extra_fields=field_names + list(str_cols or {}),
        str_fields=list(str_cols or {}))
    table = table.replace_schema_metadata({PARQUET_METADATA_KEY: meta_json})
    pq.write_table(table, path, row_group_size=ROW_GROUP_SIZE, compression="zstd")
    return SstMeta(
        file_id=os.path.basename(path).replace(".parquet", ""),
        level=0,
        min_ts=int(ts_ms.min()) if n else 0,
        max_ts=int(ts_ms.max()) if n else 0,
        num_rows=n,
        file_size=os.path.getsize(path),
        seq_max=int(seq.max()) if n else 0,
    )


def new_file_id() -> str:
    return uuid.uuid4().hex


def read_sst(path: str, schema: TableSchema, field_names: list[str]):
    """Read an SST back to host arrays: (pk_list per row-code, series codes
    i32 (dictionary indices), ts_ms i64, fields f64[nf, n], seq u64,
    str_cols {name: object ndarray}). The caller remaps dictionary indices
    into region-local codes."""
    t = pq.read_table(path)
    pk = t.column("__primary_key").combine_chunks()
    if isinstance(pk, pa.ChunkedArray):
        pk = pk.chunk(0)
    dict_values = [v.as_py() for v in pk.dictionary]
    indices = pk.indices.to_numpy(zero_copy_only=False).astype(np.int32)
    ts = t.column(schema.time_index.name).cast(pa.int64()).to_numpy(zero_copy_only=False)
    fields = np.stack([
        t.column(fn).to_numpy(zero_copy_only=False) if fn in t.column_names
        else np.full(len(ts), np.nan)
        for fn in field_names
    ]) if field_names else np.zeros((0, len(ts)))
    seq = t.column("__sequence").to_numpy(zero_copy_only=False).astype(np.int64)
    # flat-format files (sst/parquet/flat_format.rs) carry raw tag columns
    # too — tags are reconstructed from __primary_key, so skip them here
    internal = {schema.time_index.name, "__primary_key", "__sequence",
                "__op_type"} | set(schema.primary_key)
    str_cols = {}
    for cn in t.column_names:
        if cn in internal or cn in field_names:
            continue
        ftype = t.schema.field(cn).type
        if pa.types.is_string(ftype) or pa.types.is_large_string(ftype) or \
                pa.types.is_binary(ftype) or pa.types.is_large_binary(ftype):
            str_cols[cn] = t.column(cn).to_numpy(zero_copy_only=False)
    return dict_values, indices, ts, fields, seq, str_cols


def read_region_metadata(path: str) -> dict | None:
    """Parse the `greptime:metadata` key-value entry (reference:
    sst/parquet/metadata.rs reads the same key)."""
    import json
    md = pq.read_metadata(path).metadata or {}
    raw = md.get(PARQUET_METADATA_KEY)
    if raw is None:
        return None
    return json.loads(raw.decode() if isinstance(raw, bytes) else raw)
