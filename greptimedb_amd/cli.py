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
    start.add_argument("--data-dir", default="./greptimedb_data")
    start.add_argument("--device", default="auto",
                       help="cuda:N / cpu / auto")
    start.add_argument("--regions", type=int, default=4)
    start.add_argument("--wal-sync", action="store_true")
    args = ap.parse_args(argv)

    import torch
    import uvicorn

    from greptimedb_amd.engine.engine import EngineConfig, MitoEngine
    from greptimedb_amd.servers.http import ServerContext, build_app

    device = args.device
    if device == "auto":
        device = "cuda:0" if torch.cuda.is_available() else "cpu"
    engine = MitoEngine(EngineConfig(
        data_dir=args.data_dir, device=device,
        default_regions=args.regions, wal_sync=args.wal_sync))
    ctx = ServerContext(engine)
    app = build_app(ctx)
    host, port = args.http_addr.rsplit(":", 1)
    print(f"greptimedb_amd standalone: device={device} data={args.data_dir} "
          f"http={host}:{port}", flush=True)
    uvicorn.run(app, host=host, port=int(port), log_level="warning")


if __name__ == "__main__":
    main()
