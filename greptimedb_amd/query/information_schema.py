"""information_schema virtual tables.

Reference parity: src/catalog/src/system_schema/information_schema — the
tables/columns/partitions/region_statistics/flows/cluster_info views most
used by tooling and dashboards.
"""

from __future__ import annotations

import numpy as np

from greptimedb_amd.models.schema import SemanticType

VIRTUAL_TABLES = {
    "tables", "columns", "region_statistics", "flows", "cluster_info",
    "partitions", "region_peers", "build_info", "process_list", "views",
    "schemata", "ssts", "key_column_usage", "table_constraints",
    "procedure_info",
}

PG_CATALOG_TABLES = {"pg_tables", "pg_namespace", "pg_class"}


def is_information_schema(name: str) -> bool:
    low = name.lower()
    if low.startswith("information_schema.") and \
            low.split(".", 1)[1] in VIRTUAL_TABLES:
        return True
    return low.startswith("pg_catalog.") and \
        low.split(".", 1)[1] in PG_CATALOG_TABLES


def build(engine, name: str):
    """Return (names, columns) for the virtual table."""
    kind = name.split(".", 1)[1].lower()
    schema_ns = name.split(".", 1)[0].lower()
    if schema_ns == "pg_catalog":
        # pg_catalog compatibility views (reference
        # src/catalog/src/system_schema/pg_catalog.rs)
        schemas = sorted({"public", "information_schema", "pg_catalog"} |
                         set(getattr(engine, "schemas", set())))
        if kind == "pg_namespace":
            return _cols(["oid", "nspname"],
                         [(i + 1, s_) for i, s_ in enumerate(schemas)])
        rows = []
        for i, (t, st) in enumerate(sorted(engine.tables.items())):
            sch = t.split(".", 1)[0] if "." in t else "public"
            tn = t.split(".", 1)[1] if "." in t else t
            rows.append((i + 16384, tn, sch, "r"))
        if kind == "pg_class":
            return _cols(["oid", "relname", "relnamespace", "relkind"], rows)
        return _cols(["schemaname", "tablename", "tableowner"],
                     [(sch, tn, "greptime") for _o, tn, sch, _k in rows])
    if kind == "process_list":
        rows = [(pid, info["sql"][:200], round(info["elapsed_ms"], 1),
                 info["state"])
                for pid, info in sorted(
                    getattr(engine, "process_list", {}).items())]
        return _cols(["id", "query", "elapsed_ms", "state"], rows)
    if kind == "views":
        rows = [(v, sql) for v, sql in
                sorted(getattr(engine, "views", {}).items())]
        return _cols(["view_name", "definition"], rows)
    if kind == "schemata":
        schemas = sorted({"public", "information_schema", "greptime_private"} |
                         set(getattr(engine, "schemas", set())))
        return _cols(["catalog_name", "schema_name"],
                     [("greptime", s_) for s_ in schemas])
    if kind == "tables":
        rows = [("greptime", "public", t, "BASE TABLE", st.schema.table_id,
                 "mito-hip", len(st.regions))
                for t, st in sorted(engine.tables.items())]
        return _cols(["table_catalog", "table_schema", "table_name",
                      "table_type", "table_id", "engine", "region_count"], rows)
    if kind == "columns":
        rows = []
        for t, st in sorted(engine.tables.items()):
            for c in st.schema.columns:
                rows.append((t, c.name, c.dtype.value,
                             SemanticType(c.semantic).name))
            for fn in st.regions[0].field_names:
                if not st.schema.has_column(fn):
                    rows.append((t, fn, "float64", "FIELD"))
            for fn in st.regions[0].str_field_names:
                if not st.schema.has_column(fn):
                    rows.append((t, fn, "string", "FIELD"))
        return _cols(["table_name", "column_name", "data_type", "semantic_type"],
                     rows)
    if kind in ("region_statistics", "partitions"):
        rows = []
        for t, st in sorted(engine.tables.items()):
            for r in st.regions:
                rows.append((r.region_id, t, r.region_id & 0xFFFFFFFF,
                             r.memtable.len,
                             sum(b.n for b in r.sst_cache.values()),
                             len(r.sst_cache), len(r.series),
                             r.device, r.memtable.bytes_used))
        return _cols(["region_id", "table_name", "region_number",
                      "memtable_rows", "sst_rows", "sst_files", "series",
                      "device", "memtable_bytes"], rows)
    if kind == "flows":
        fe = getattr(engine, "flow_engine", None)
        flows = list(fe.flows.values()) if fe else []
        rows = [(f.name, f.source, f.sink, f.select_sql) for f in flows]
        return _cols(["flow_name", "source_table", "sink_table", "query"], rows)
    if kind == "region_peers":
        # leader/follower roles per region (ref information_schema
        # region_peers; followers appear when meta/replication is active)
        rows = []
        for t, st in sorted(engine.tables.items()):
            for r in st.regions:
                rows.append((r.region_id, t, 0, engine.config.device,
                             getattr(r, "role", "leader").upper(), "ALIVE"))
        return _cols(["region_id", "table_name", "peer_id", "peer_addr",
                      "role", "status"], rows)
    if kind == "ssts":
        # per-SST file inventory (ref information_schema ssts)
        rows = []
        for t, st in sorted(engine.tables.items()):
            for r in st.regions:
                for fid, meta in sorted(r.manifest.files.items()):
                    rows.append((t, r.region_id, fid,
                                 int(meta.get("num_rows", 0)),
                                 int(meta.get("file_size", 0)),
                                 int(meta.get("level", 0)),
                                 int(meta.get("min_ts", 0)),
                                 int(meta.get("max_ts", 0))))
        return _cols(["table_name", "region_id", "file_id", "num_rows",
                      "file_size", "level", "min_ts", "max_ts"], rows)
    if kind == "key_column_usage":
        rows = []
        for t, st in sorted(engine.tables.items()):
            for pos, pk in enumerate(st.schema.primary_key):
                rows.append(("greptime", "public", "PRIMARY", t, pk, pos + 1))
            rows.append(("greptime", "public", "TIME INDEX", t,
                         st.schema.time_index.name, 1))
        return _cols(["constraint_catalog", "constraint_schema",
                      "constraint_name", "table_name", "column_name",
                      "ordinal_position"], rows)
    if kind == "table_constraints":
        rows = []
        for t, st in sorted(engine.tables.items()):
            if st.schema.primary_key:
                rows.append(("greptime", "public", "PRIMARY", t, "PRIMARY KEY"))
            rows.append(("greptime", "public", "TIME INDEX", t, "TIME INDEX"))
        return _cols(["constraint_catalog", "constraint_schema",
                      "constraint_name", "table_name", "constraint_type"],
                     rows)
    if kind == "procedure_info":
        # persisted procedure-store entries (migrations etc.)
        rows = []
        store = getattr(engine, "procedure_store", None)
        if store is not None:
            for p in store.load_all():
                rows.append((p.get("pid", ""), p.get("type", ""),
                             p.get("status", ""),
                             str(p.get("state", ""))[:200]))
        return _cols(["procedure_id", "procedure_type", "status", "detail"],
                     rows)
    if kind == "build_info":
        from greptimedb_amd import __version__
        return _cols(["version", "arch", "backend"],
                     [(__version__, "gfx950", "rocm-hip")])
    if kind == "cluster_info":
        import torch
        rows = [(0, "standalone", engine.config.device,
                 torch.cuda.get_device_name(0) if torch.cuda.is_available() else "cpu")]
        return _cols(["peer_id", "peer_type", "device", "device_name"], rows)
    raise KeyError(kind)


def _cols(names, rows):
    cols = [np.array([r[i] for r in rows], dtype=object) for i in range(len(names))]
    return names, cols
