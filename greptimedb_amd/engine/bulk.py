"""Bulk columnar ingest: Arrow RecordBatch → regions (C3 axis).

Reference parity: Arrow Flight `do_put` bulk path —
src/servers/src/grpc/flight.rs:240-330 (PutRecordBatchRequest) →
RegionRequest::BulkInserts → BulkMemtable (src/mito2/src/memtable/bulk.rs:391)
— rows skip the row-proto decode entirely. MI355X design: columns land as
numpy/torch arrays, series are registered in bulk, rows are partition-split
vectorized (parallel/partition.py) and appended per region with one WAL
group commit.
"""

from __future__ import annotations

import numpy as np

from greptimedb_amd.engine.engine import MitoEngine, TableState
from greptimedb_amd.models.schema import (ColumnSchema, DataType, SemanticType,
                                          TableSchema)

TS_NAMES = ("ts", "greptime_timestamp", "timestamp", "time")


def _infer_schema(name: str, batch) -> TableSchema:
    """No table yet: string columns become tags, the first timestamp-ish
    column the time index, numeric columns fields (greptime's gRPC
    auto-create infers the same way from semantic hints; here we only have
    arrow types)."""
    import pyarrow as pa
    cols = []
    cid = 0
    pk = []
    ts_col = None
    for f in batch.schema:
        if pa.types.is_timestamp(f.type) and ts_col is None:
            ts_col = f.name
    if ts_col is None:
        for f in batch.schema:
            if f.name.lower() in TS_NAMES:
                ts_col = f.name
                break
    if ts_col is None:
        raise ValueError(f"bulk ingest into new table {name}: no timestamp column")
    for f in batch.schema:
        if f.name == ts_col:
            cols.append(ColumnSchema(f.name, DataType.TIMESTAMP_MS,
                                     SemanticType.TIMESTAMP, cid))
        elif pa.types.is_string(f.type) or pa.types.is_large_string(f.type):
            cols.append(ColumnSchema(f.name, DataType.STRING, SemanticType.TAG, cid))
            pk.append(f.name)
        else:
            cols.append(ColumnSchema(f.name, DataType.FLOAT64, SemanticType.FIELD, cid))
        cid += 1
    return TableSchema(name=name, columns=cols, primary_key=pk)


def bulk_insert_arrow(engine: MitoEngine, table_name: str, batch,
                      durable: bool = True, append_mode: bool = True) -> int:
    """Ingest one Arrow RecordBatch (or Table). Returns rows written."""
    import pyarrow as pa
    if isinstance(batch, pa.Table):
        batch = batch.combine_chunks()
    n = batch.num_rows
    if n == 0:
        return 0
    try:
        st: TableState = engine.table(table_name)
    except Exception:
        st = engine.create_table(_infer_schema(table_name, batch),
                                 append_mode=append_mode, if_not_exists=True)
    schema = st.schema
    names = set(batch.schema.names)
    ts_name = schema.time_index.name
    if ts_name not in names:
        for cand in TS_NAMES:
            if cand in names:
                ts_name = cand
                break
    if ts_name not in names:
        raise ValueError(f"batch for {table_name} lacks time column {ts_name}")
    col = batch.column(batch.schema.get_field_index(ts_name))
    if pa.types.is_timestamp(col.type):
        ts_ms = col.cast(pa.timestamp("ms")).cast(pa.int64()).to_numpy(
            zero_copy_only=False)
    else:
        ts_ms = col.cast(pa.int64()).to_numpy(zero_copy_only=False)

    # tag columns → per-row value arrays
    tag_names = [c.name for c in schema.tag_columns]
    tag_cols = {}
    for tn in tag_names:
        if tn in names:
            arr = batch.column(batch.schema.get_field_index(tn))
            tag_cols[tn] = np.asarray(arr.to_pylist(), dtype=object)
        else:
            tag_cols[tn] = np.full(n, None, dtype=object)

    # vectorized partition split (multi-dim rule) or hash
    rule = engine.partition_rule(st)
    from greptimedb_amd.parallel.partition import MultiDimPartitionRule
    if isinstance(rule, MultiDimPartitionRule):
        region_of = rule.split(tag_cols, n)
    else:
        from greptimedb_amd.engine import pk_codec
        from greptimedb_amd.engine.series import tsid_hash
        pks = [pk_codec.encode_pk(tuple(tag_cols[tn][i] for tn in tag_names))
               for i in range(n)]
        region_of = np.array([tsid_hash(pk) % len(st.regions) for pk in pks],
                             dtype=np.int32)

    # numeric field columns in table order (auto-ALTER new ones)
    field_arrays = {}
    str_field_arrays = {}
    for f in batch.schema:
        if f.name == ts_name or f.name in tag_names:
            continue
        arr = batch.column(batch.schema.get_field_index(f.name))
        if pa.types.is_string(f.type) or pa.types.is_large_string(f.type) or \
                pa.types.is_binary(f.type):
            str_field_arrays[f.name] = arr.to_pylist()
        else:
            field_arrays[f.name] = arr.cast(pa.float64()).to_numpy(
                zero_copy_only=False)
    new_fields = [fn for fn in field_arrays
                  if fn not in st.regions[0].field_names]
    if new_fields:
        with engine._ddl_lock:
            new_fields = [fn for fn in new_fields
                          if fn not in st.regions[0].field_names]
            for r in st.regions:
                r.ensure_fields(new_fields)
    if str_field_arrays:
        for r in st.regions:
            r.ensure_str_fields(list(str_field_arrays),
                                fulltext=bool(r.text_cols))

    field_names = st.regions[0].field_names
    total = 0
    for ridx in np.unique(region_of):
        rows = np.flatnonzero(region_of == ridx)
        region = st.regions[int(ridx)]
        labels = [tuple(tag_cols[tn][i] for tn in tag_names) for i in rows]
        codes = region.register_series_bulk(labels)
        fmat = np.full((len(field_names), len(rows)), np.nan)
        for j, fn in enumerate(field_names):
            if fn in field_arrays:
                fmat[j] = field_arrays[fn][rows]
        strs = {fn: [vals[i] for i in rows]
                for fn, vals in str_field_arrays.items()} or None
        engine.write_region(st, int(ridx), codes.astype(np.int32),
                            ts_ms[rows], fmat, [], durable=durable,
                            str_fields=strs)
        total += len(rows)
    if durable:
        engine.commit_wal()
    engine.maybe_flush()
    return total
