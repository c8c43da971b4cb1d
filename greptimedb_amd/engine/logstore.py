"""Log ingestion: JSON events / Loki push → log tables.

Reference parity: src/pipeline (ETL engine — greptime_identity behavior:
JSON fields become columns) + the Loki push endpoint (servers http.rs:1129).
Round-1 pipeline surface: greptime_identity semantics with declared tag
keys; numeric values → f64 fields, strings → fulltext-indexed string
columns, nested JSON flattened one level with dotted keys.
"""

from __future__ import annotations

import time

import numpy as np

from greptimedb_amd.engine.engine import MitoEngine, TableState
from greptimedb_amd.engine.series import tsid_hash
from greptimedb_amd.engine import pk_codec
from greptimedb_amd.models.schema import ColumnSchema, DataType, SemanticType, TableSchema


class LogStore:
    def __init__(self, engine: MitoEngine, durable: bool = True):
        self.engine = engine
        self.durable = durable
        self.rows_ingested = 0

    def _get_table(self, name: str, tag_keys: list[str]) -> TableState:
        try:
            return self.engine.table(name)
        except Exception:
            cols = []
            cid = 0
            for t in tag_keys:
                cols.append(ColumnSchema(t, DataType.STRING, SemanticType.TAG, cid))
                cid += 1
            cols.append(ColumnSchema("ts", DataType.TIMESTAMP_MS,
                                     SemanticType.TIMESTAMP, cid))
            schema = TableSchema(name=name, columns=cols, primary_key=list(tag_keys),
                                 options={"append_mode": "true"})
            return self.engine.create_table(schema, append_mode=True,
                                            if_not_exists=True)

    def ingest_with_pipeline(self, table: str, entries: list[dict],
                             pipeline) -> int:
        """Run a compiled YAML Pipeline over the entries, then ingest each
        table-suffix group with the pipeline's tag/fulltext/timestamp hints
        (ref: src/pipeline dispatcher + tablesuffix routing)."""
        total = 0
        for suffix, rows in pipeline.run(entries).items():
            total += self.ingest(table + suffix, rows,
                                 tag_keys=pipeline.tag_keys,
                                 ts_key=pipeline.ts_key,
                                 fulltext_keys=pipeline.fulltext_keys)
        return total

    def ingest(self, table: str, entries: list[dict], tag_keys: list[str] | None = None,
               ts_key: str = "timestamp", fulltext_keys: list[str] | None = None) -> int:
        """greptime_identity-style ingestion: each entry's keys become
        columns. `tag_keys` values form the series; `ts_key` (ms epoch or
        ns > 1e15) is the time index (now() if absent)."""
        if not entries:
            return 0
        tag_keys = tag_keys or []
        st = self._get_table(table, tag_keys)
        # align to the table's established primary key; extra labels become
        # string fields
        tag_keys = [c.name for c in st.schema.tag_columns]
        n = len(entries)
        now_ms = int(time.time() * 1000)
        ts = np.empty(n, dtype=np.int64)
        num_vals: dict[str, list] = {}
        str_vals: dict[str, list] = {}
        tags_per_row = []
        for i, e in enumerate(entries):
            flat = {}
            for k, v in e.items():
                if isinstance(v, dict):
                    for k2, v2 in v.items():
                        flat[f"{k}.{k2}"] = v2
                else:
                    flat[k] = v
            tv = flat.get(ts_key)
            if tv is None:
                ts[i] = now_ms
            else:
                tvi = int(tv)
                ts[i] = tvi // 1_000_000 if tvi > 10 ** 15 else tvi
            tags_per_row.append(tuple(str(flat.get(t)) if flat.get(t) is not None
                                      else None for t in tag_keys))
            for k, v in flat.items():
                if k == ts_key or k in tag_keys:
                    continue
                if isinstance(v, bool):
                    num_vals.setdefault(k, [None] * n)[i] = float(v)
                elif isinstance(v, (int, float)):
                    num_vals.setdefault(k, [None] * n)[i] = float(v)
                elif v is not None:
                    str_vals.setdefault(k, [None] * n)[i] = str(v)

        # route rows to regions by series
        region_rows: dict[int, list[int]] = {}
        codes = np.empty(n, dtype=np.int32)
        for i, tags in enumerate(tags_per_row):
            pk = pk_codec.encode_pk(tags)
            ridx = tsid_hash(pk) % len(st.regions)
            codes[i] = st.regions[ridx].register_series(tags)
            region_rows.setdefault(ridx, []).append(i)

        with self.engine._ddl_lock:
            new_num = [k for k in num_vals if k not in st.regions[0].field_names]
            for r in (st.regions if new_num else []):
                r.ensure_fields(new_num)
            new_str = [k for k in str_vals
                       if k not in st.regions[0].str_field_names]
            if new_str:
                # fulltext_keys None → every string column fulltext (identity
                # pipeline default); else only the transform-declared ones
                ft = new_str if fulltext_keys is None else \
                    [k for k in new_str if k in fulltext_keys]
                plain = [k for k in new_str if k not in ft]
                for r in st.regions:
                    if ft:
                        r.ensure_str_fields(ft, fulltext=True)
                    if plain:
                        r.ensure_str_fields(plain, fulltext=False)
        fnames = st.regions[0].field_names
        for ridx, rows in region_rows.items():
            rows_a = np.array(rows)
            fmat = np.full((len(fnames), len(rows_a)), np.nan)
            for j, fn in enumerate(fnames):
                col = num_vals.get(fn)
                if col is not None:
                    fmat[j] = [col[r] if col[r] is not None else np.nan for r in rows_a]
            strs = {k: [v[r] for r in rows_a] for k, v in str_vals.items()}
            self.engine.write_region(st, ridx, codes[rows_a], ts[rows_a], fmat, [],
                                     durable=self.durable, str_fields=strs or None)
        if self.durable:
            self.engine.commit_wal()
        self.engine.maybe_flush()
        self.rows_ingested += n
        return n

    def ingest_loki(self, payload: dict) -> int:
        """Loki push API: {"streams":[{"stream":{label:value},
        "values":[[ts_ns, line], ...]}]} → table `loki_logs` with stream
        labels as tags and the line as fulltext `line` column."""
        total = 0
        for stream in payload.get("streams", []):
            labels = stream.get("stream", {})
            values = stream.get("values", [])
            entries = []
            for v in values:
                ts_ns = int(v[0])
                e = {"timestamp": ts_ns, "line": v[1]}
                e.update({k: str(val) for k, val in labels.items()})
                entries.append(e)
            total += self.ingest("loki_logs", entries,
                                 tag_keys=sorted(labels.keys()),
                                 ts_key="timestamp")
        return total
