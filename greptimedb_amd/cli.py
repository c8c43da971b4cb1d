"""`python -m greptimedb_amd` — the server binary.

Reference parity: src/cmd/src/bin/greptime.rs subcommands. Round-1 surface:
`standalone start` (engine + HTTP servers in one process, one GPU).
Multi-GPU serving runs one process per GPU via torch.distributed (see
parallel/), with rank 0 exposing HTTP.
"""

from __future__ import annotations

import argparse
import os


def main(argv=None):
    ap = argparse.ArgumentParser(prog="greptimedb_amd")
    sub = ap.add_subparsers(dest="role", required=True)
    st = sub.add_parser("standalone")
    st_sub = st.add_subparsers(dest="cmd", required=True)
    start = st_sub.add_parser("start")
    start.add_argument("--http-addr", default="0.0.0.0:4000")
    start.add_argument("--grpc-addr", default="0.0.0.0:4001")
    start.add_argument("--mysql-addr", default="0.0.0.0:4002")
    start.add_argument("--postgres-addr", default="0.0.0.0:4003")
    start.add_argument("--flight-addr", default="0.0.0.0:4005")
    start.add_argument("--data-dir", default="./greptimedb_data")
    start.add_argument("--device", default="auto",
                       help="cuda:N / cpu / auto")
    start.add_argument("--regions", type=int, default=4)
    start.add_argument("--wal-sync", action="store_true")
    start.add_argument("--user-provider", default=None,
                       help="static_user_provider:file:<path>")
    start.add_argument("--config", default=None, help="TOML config file")
    # `cli meta snapshot save/restore` (ref src/cli metadata snapshot tools)
    cli = sub.add_parser("cli")
    cli_sub = cli.add_subparsers(dest="cmd", required=True)
    meta = cli_sub.add_parser("meta")
    meta_sub = meta.add_subparsers(dest="action", required=True)
    for action in ("save", "restore"):
        p = meta_sub.add_parser(action)
        p.add_argument("--data-dir", required=True)
        p.add_argument("--file", required=True, help="snapshot tar path")
    # `cli data export/import` (ref src/cli data export/import tools)
    data = cli_sub.add_parser("data")
    data_sub = data.add_subparsers(dest="action", required=True)
    for action in ("export", "import"):
        p = data_sub.add_parser(action)
        p.add_argument("--data-dir", required=True)
        p.add_argument("--dir", required=True,
                       help="export/import directory (one parquet per "
                            "table + schema.sql)")
        p.add_argument("--tables", default=None,
                       help="comma-separated subset (default: all)")
    args = ap.parse_args(argv)

    if args.role == "cli" and args.cmd == "data":
        return _data_export_import(args)
    if args.role == "cli":
        return _meta_snapshot(args)

    import asyncio

    import torch
    import uvicorn

    from greptimedb_amd.engine.engine import EngineConfig, MitoEngine
    from greptimedb_amd.servers.http import ServerContext, build_app
    from greptimedb_amd.servers.mysql import MySQLServer
    from greptimedb_amd.servers.postgres import PostgresServer
    from greptimedb_amd.utils.config import load_config

    cfg = load_config(args.config) if args.config else {}
    device = cfg.get("device", args.device)
    if device == "auto":
        device = "cuda:0" if torch.cuda.is_available() else "cpu"
    engine = MitoEngine(EngineConfig(
        data_dir=cfg.get("data_dir", args.data_dir), device=device,
        default_regions=int(cfg.get("regions", args.regions)),
        wal_sync=bool(cfg.get("wal_sync", args.wal_sync)),
        record_events=True))
    user_provider = None
    if args.user_provider and args.user_provider.startswith("static_user_provider:file:"):
        from greptimedb_amd.servers.auth import StaticUserProvider
        user_provider = StaticUserProvider.from_file(
            args.user_provider.split("file:", 1)[1])
    ctx = ServerContext(engine, user_provider=user_provider)
    app = build_app(ctx)
    host, port = cfg.get("http_addr", args.http_addr).rsplit(":", 1)
    my_host, my_port = cfg.get("mysql_addr", args.mysql_addr).rsplit(":", 1)
    pg_host, pg_port = cfg.get("postgres_addr", args.postgres_addr).rsplit(":", 1)
    g_host, g_port = cfg.get("grpc_addr", args.grpc_addr).rsplit(":", 1)
    f_host, f_port = cfg.get("flight_addr", args.flight_addr).rsplit(":", 1)
    # gRPC GreptimeDatabase + Arrow Flight run on their own thread pools
    from greptimedb_amd.servers.flight import GreptimeFlightServer
    from greptimedb_amd.servers.grpc_server import GreptimeGrpcServer
    grpc_srv = GreptimeGrpcServer(engine, ctx.executor, g_host, int(g_port))
    # pyarrow Flight serves from its own threads as soon as it is built
    flight_srv = GreptimeFlightServer(engine, ctx.executor, f_host, int(f_port))
    print(f"greptimedb_amd standalone: device={device} data={args.data_dir} "
          f"http={host}:{port} grpc={grpc_srv.port} mysql={my_port} "
          f"postgres={pg_port} flight={flight_srv.port}", flush=True)

    async def serve():
        uv = uvicorn.Server(uvicorn.Config(app, host=host, port=int(port),
                                           log_level="warning"))
        my = MySQLServer(ctx.executor, my_host, int(my_port), user_provider)
        pg = PostgresServer(ctx.executor, pg_host, int(pg_port), user_provider)
        await asyncio.gather(uv.serve(), my.serve_forever(), pg.serve_forever())

    try:
        asyncio.run(serve())
    finally:
        grpc_srv.shutdown()
        flight_srv.shutdown()


def _data_export_import(args) -> int:
    """`cli data export/import`: whole-database backup as one parquet per
    table + a schema.sql of SHOW CREATE TABLE statements (reference:
    src/cli/src/data export/import tools)."""
    import os

    from greptimedb_amd.engine.engine import EngineConfig, MitoEngine
    from greptimedb_amd.query.executor import Executor

    eng = MitoEngine(EngineConfig(data_dir=args.data_dir, device="cpu",
                                  background_flush=False))
    ex = Executor(eng)
    try:
        if args.action == "export":
            os.makedirs(args.dir, exist_ok=True)
            names = sorted(eng.tables)
            if args.tables:
                wanted = {t.strip() for t in args.tables.split(",")}
                names = [n for n in names if n in wanted]
            ddls = []
            for t in names:
                r = ex.execute(f'SHOW CREATE TABLE "{t}"')
                ddls.append(str(r.columns[1][0]) + ";")
                n = ex.execute(
                    f"COPY \"{t}\" TO '{os.path.join(args.dir, t)}.parquet'")
                print(f"exported {t}: {n.rows()[0][0]} rows")
            with open(os.path.join(args.dir, "schema.sql"), "w") as f:
                f.write("\n\n".join(ddls) + "\n")
            print(f"export complete: {len(names)} tables -> {args.dir}")
        else:
            with open(os.path.join(args.dir, "schema.sql")) as f:
                for stmt in f.read().split(";"):
                    if stmt.strip():
                        ex.execute(stmt)
            names = sorted(
                fn[:-8] for fn in os.listdir(args.dir)
                if fn.endswith(".parquet"))
            if args.tables:
                wanted = {t.strip() for t in args.tables.split(",")}
                names = [n for n in names if n in wanted]
            for t in names:
                ex.execute(
                    f"COPY \"{t}\" FROM '{os.path.join(args.dir, t)}.parquet'")
                print(f"imported {t}")
            eng.flush_all()
            print(f"import complete: {len(names)} tables")
        return 0
    finally:
        eng.close()


def _meta_snapshot(args) -> int:
    """Metadata-only snapshot: catalog + manifests + series logs + pipelines
    (no SSTs/WAL — those are data; ref `greptime cli meta snapshot`)."""
    import tarfile

    def is_meta(path: str) -> bool:
        name = os.path.basename(path)
        return (name == "catalog.json" or name.endswith(".pipeline") or
                name == "series.log" or "manifest" in path or
                os.sep + "pipelines" + os.sep in path + os.sep)

    if args.action == "save":
        count = 0
        with tarfile.open(args.file, "w:gz") as tar:
            for root, _dirs, files in os.walk(args.data_dir):
                for f in files:
                    p = os.path.join(root, f)
                    if is_meta(p):
                        tar.add(p, arcname=os.path.relpath(p, args.data_dir))
                        count += 1
        print(f"meta snapshot: {count} files -> {args.file}", flush=True)
    else:
        os.makedirs(args.data_dir, exist_ok=True)
        with tarfile.open(args.file, "r:gz") as tar:
            for m in tar.getmembers():
                # refuse path escapes
                if m.name.startswith(("/", "..")) or ".." in m.name.split("/"):
                    raise ValueError(f"unsafe member {m.name}")
            tar.extractall(args.data_dir)
            n = len(tar.getmembers())
        print(f"meta snapshot: restored {n} files -> {args.data_dir}", flush=True)
    return 0


if __name__ == "__main__":
    main()
