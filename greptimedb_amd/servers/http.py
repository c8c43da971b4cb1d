"""HTTP protocol server (L7).

Reference parity: src/servers/src/http.rs route tree — /v1/sql,
/v1/influxdb/write, /v1/prometheus/write (remote write),
/v1/prometheus/api/v1/* (Prometheus HTTP API), /health, /metrics.
Response shapes follow GreptimeDB's JSON (greptimedb output records /
Prometheus API formats) so existing clients/dashboards work.
"""

from __future__ import annotations

import math
import time

import numpy as np
from fastapi import FastAPI, Query, Request, Response

from greptimedb_amd.engine.engine import MitoEngine
from greptimedb_amd.engine.ingest import Ingestor
from greptimedb_amd.engine.promstore import PromStore
from greptimedb_amd.query.executor import Executor, QueryResult
from greptimedb_amd.query.promql.eval import PromEvaluator
from greptimedb_amd.query.promql.parser import parse_duration_s, parse_promql
from greptimedb_amd.utils.errors import GreptimeError
from greptimedb_amd.utils import metrics as metrics_mod
from greptimedb_amd.utils.timeutil import parse_ts_ms


class ServerContext:
    def __init__(self, engine: MitoEngine, dist=None):
        self.engine = engine
        self.ingestor = Ingestor(engine)
        self.promstore = PromStore(engine)
        from greptimedb_amd.engine.logstore import LogStore
        self.logstore = LogStore(engine)
        from greptimedb_amd.engine.tracestore import TraceStore
        self.tracestore = TraceStore(engine)
        self.executor = Executor(engine, dist=dist)
        self.prom = PromEvaluator(engine, dist=dist)
        self.started = time.time()


def _records_json(r: QueryResult) -> dict:
    def cell(v):
        if v is None:
            return None
        if isinstance(v, (np.floating, float)):
            f = float(v)
            return None if math.isnan(f) else f
        if isinstance(v, (np.integer, int)):
            return int(v)
        return str(v)
    return {
        "records": {
            "schema": {"column_schemas": [{"name": n, "data_type": "?"}
                                          for n in r.names]},
            "rows": [[cell(v) for v in row] for row in r.rows()],
        }
    }


def _parse_time_s(v: str | None, default=None):
    if v is None:
        return default
    try:
        return float(v)
    except ValueError:
        ms = parse_ts_ms(v)
        if ms is None:
            raise GreptimeError(f"bad time {v!r}")
        return ms / 1000.0


def _parse_step_s(v: str | None, default=60.0):
    if v is None:
        return default
    try:
        return float(v)
    except ValueError:
        return parse_duration_s(v)


def _prom_result(m, instant: bool) -> dict:
    result = []
    vals = m.values.cpu().numpy()
    for s in range(m.S):
        labels = {k: v for k, v in m.labels[s].items() if v is not None}
        if instant:
            v = vals[s, -1]
            if math.isnan(v):
                continue
            result.append({"metric": labels,
                           "value": [m.grid[-1] / 1000.0, str(v)]})
        else:
            pts = [[int(g) / 1000.0, str(vals[s, i])]
                   for i, g in enumerate(m.grid) if not math.isnan(vals[s, i])]
            if pts:
                result.append({"metric": labels, "values": pts})
    return {
        "status": "success",
        "data": {"resultType": "vector" if instant else "matrix", "result": result},
    }


def build_app(ctx: ServerContext) -> FastAPI:
    app = FastAPI(title="greptimedb-amd")

    @app.get("/health")
    @app.get("/ready")
    def health():
        return {}

    @app.get("/status")
    def status():
        tables = {
            name: sum(r.num_rows for r in st.regions)
            for name, st in ctx.engine.tables.items()
        }
        return {"uptime_s": time.time() - ctx.started, "tables": tables,
                "device": ctx.engine.config.device}

    # ---------------- SQL ----------------

    async def _sql(request: Request, sql: str | None):
        if sql is None:
            form = await request.form()
            sql = form.get("sql")
            if sql is None:
                body = (await request.body()).decode()
                sql = body or None
        t0 = time.perf_counter()
        try:
            r = ctx.executor.execute(sql)
        except GreptimeError as e:
            metrics_mod.counter("http_sql_errors").inc()
            return {"code": 3000, "error": str(e), "execution_time_ms":
                    round((time.perf_counter() - t0) * 1000, 3)}
        metrics_mod.counter("http_sql_requests").inc()
        return {"output": [_records_json(r)],
                "execution_time_ms": round((time.perf_counter() - t0) * 1000, 3)}

    @app.get("/v1/sql")
    async def sql_get(request: Request, sql: str = Query(None)):
        return await _sql(request, sql)

    @app.post("/v1/sql")
    async def sql_post(request: Request, sql: str = Query(None)):
        return await _sql(request, sql)

    # ---------------- influx line protocol ----------------

    async def _influx(request: Request, precision: str):
        body = await request.body()
        factor = {"ns": 1, "n": 1, "us": 1000, "u": 1000,
                  "ms": 1_000_000, "s": 1_000_000_000}.get(precision, 1)
        n = ctx.ingestor.ingest_lines(body, ts_scale_to_ns=factor)
        metrics_mod.counter("influx_rows").inc(n)
        return Response(status_code=204)

    @app.post("/v1/influxdb/write")
    async def influx_write(request: Request, precision: str = Query("ns")):
        return await _influx(request, precision)

    @app.post("/v1/influxdb/api/v2/write")
    async def influx_write_v2(request: Request, precision: str = Query("ns")):
        return await _influx(request, precision)

    # ---------------- logs ----------------

    @app.post("/v1/events/logs")
    async def events_logs(request: Request, table: str = Query("logs"),
                          tag_keys: str = Query(""), ts_key: str = Query("timestamp")):
        import json as _json
        body = await request.body()
        entries = _json.loads(body)
        if isinstance(entries, dict):
            entries = [entries]
        tags = [t for t in tag_keys.split(",") if t]
        n = ctx.logstore.ingest(table, entries, tag_keys=tags, ts_key=ts_key)
        metrics_mod.counter("log_events").inc(n)
        return {"rows": n}

    @app.post("/v1/loki/api/v1/push")
    async def loki_push(request: Request):
        import json as _json
        body = await request.body()
        n = ctx.logstore.ingest_loki(_json.loads(body))
        metrics_mod.counter("loki_lines").inc(n)
        return Response(status_code=204)

    # ---------------- OTLP ----------------

    @app.post("/v1/otlp/v1/traces")
    async def otlp_traces(request: Request):
        body = await request.body()
        n = ctx.tracestore.write(body)
        metrics_mod.counter("otlp_spans").inc(n)
        return {"partialSuccess": {}}

    # ---------------- prometheus remote write ----------------

    @app.post("/v1/prometheus/write")
    async def prom_write(request: Request):
        body = await request.body()
        snappy = request.headers.get("content-encoding", "snappy") != "identity"
        n = ctx.promstore.write(body, snappy=snappy)
        metrics_mod.counter("remote_write_samples").inc(n)
        return Response(status_code=204)

    # ---------------- prometheus query API ----------------

    async def _param(request: Request, name: str):
        v = request.query_params.get(name)
        if v is None and request.method == "POST":
            form = await request.form()
            v = form.get(name)
        return v

    @app.api_route("/v1/prometheus/api/v1/query", methods=["GET", "POST"])
    async def prom_query(request: Request):
        q = await _param(request, "query")
        t = _parse_time_s(await _param(request, "time"), time.time())
        try:
            m = ctx.prom.query_instant(q, t)
        except GreptimeError as e:
            return {"status": "error", "errorType": "bad_data", "error": str(e)}
        return _prom_result(m, instant=True)

    @app.api_route("/v1/prometheus/api/v1/query_range", methods=["GET", "POST"])
    async def prom_query_range(request: Request):
        q = await _param(request, "query")
        start = _parse_time_s(await _param(request, "start"))
        end = _parse_time_s(await _param(request, "end"))
        step = _parse_step_s(await _param(request, "step"))
        try:
            m = ctx.prom.query_range(q, start, end, step)
        except GreptimeError as e:
            return {"status": "error", "errorType": "bad_data", "error": str(e)}
        return _prom_result(m, instant=False)

    @app.api_route("/v1/prometheus/api/v1/labels", methods=["GET", "POST"])
    async def prom_labels(request: Request):
        names = {"__name__"}
        for st in ctx.engine.tables.values():
            for region in st.regions:
                names.update(region.series.tag_names)
        names.discard("__name__")
        return {"status": "success", "data": sorted(names) + ["__name__"]}

    @app.get("/v1/prometheus/api/v1/label/{name}/values")
    def prom_label_values(name: str):
        vals = set()
        if name == "__name__":
            vals.update(ctx.promstore.metrics)
            for tname, st in ctx.engine.tables.items():
                if tname != "greptime_metrics":
                    vals.add(tname)
        else:
            for st in ctx.engine.tables.values():
                for region in st.regions:
                    vals.update(region.series.inverted.get(name, {}).keys())
        return {"status": "success", "data": sorted(vals)}

    @app.api_route("/v1/prometheus/api/v1/series", methods=["GET", "POST"])
    async def prom_series(request: Request):
        matches = request.query_params.getlist("match[]")
        if not matches and request.method == "POST":
            form = await request.form()
            matches = form.getlist("match[]")
        out = []
        for q in matches:
            sel = parse_promql(q)
            try:
                st, _field = ctx.prom._resolve_table(sel)
            except GreptimeError:
                continue
            if st is None:
                continue
            for region in st.regions:
                codes = ctx.prom._match_codes(region, sel)
                it = range(len(region.series)) if codes is None else codes
                for c in it:
                    l = region.series.labels_of(c)
                    l.setdefault("__name__", sel.metric or st.schema.name)
                    out.append(l)
        return {"status": "success", "data": out}

    @app.get("/v1/prometheus/api/v1/metadata")
    def prom_metadata():
        return {"status": "success", "data": {}}

    # ---------------- metrics ----------------

    @app.get("/metrics")
    def metrics():
        return Response(metrics_mod.render_prometheus(), media_type="text/plain")

    return app
