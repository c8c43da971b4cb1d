"""Deterministic TSBS cpu-only fixture: direct columnar bulk load.

Reference parity: tests/perf/README.md + src/cmd query_perf_fixture —
query-perf benchmarking writes readable SSTs directly, skipping wire-protocol
ingestion. Here the fixture is generated ON DEVICE (torch) as sorted
SstBatches in the region scan cache: scale hosts × one point/10s, clamped
random-walk values like the TSBS host simulator. This is the C3 bulk-ingest
analog (reference Flight do_put → BulkMemtable, flight.rs:240-330): columnar
batches go straight to device memory, no row protos, no WAL (fixture data is
regenerable by seed).
"""

from __future__ import annotations

import numpy as np
import torch

from greptimedb_amd.engine import sst as sst_mod
from greptimedb_amd.engine.engine import MitoEngine
from greptimedb_amd.engine.series import tsid_hash
from greptimedb_amd.engine import pk_codec
from greptimedb_amd.models.tsbs import CPU_FIELDS, CPU_TAGS, host_tagsets
from greptimedb_amd.models.tsbs import CpuWorkload  # noqa: F401 (re-export convenience)

START_TS_S = 1451606400  # 2016-01-01T00:00:00Z (TSBS default)
INTERVAL_MS = 10_000


def load_cpu_fixture(engine: MitoEngine, scale: int = 4000, hours: int = 72,
                     seed: int = 11, rank: int = 0, world: int = 1,
                     epoch_chunk: int = 2048) -> int:
    """Create table `cpu` and bulk-load `scale` hosts × `hours` h of points
    on this rank's device. With world>1, hosts are sharded by hash across
    ranks (P1 region sharding, disjoint). Returns rows loaded locally."""
    from greptimedb_amd.models.tsbs import cpu_table_schema
    tagsets = host_tagsets(scale, seed=seed)
    device = engine.config.device
    # shard hosts: hash of tagset % world == rank
    my_hosts = [h for h in range(scale)
                if world == 1 or (tsid_hash(tagsets[h]) % world) == rank]
    if not my_hosts:
        return 0
    st = engine.create_table(cpu_table_schema(), append_mode=True, if_not_exists=True)
    field_names = st.regions[0].field_names
    f_map = [field_names.index(f) for f in CPU_FIELDS]

    # register series codes per region (vectorized-ish; rare path)
    per_region_hosts: dict[int, list[tuple[int, int]]] = {}
    for h in my_hosts:
        meas, tags = _parse_tags(tagsets[h])
        pk = pk_codec.encode_pk(tags)
        ridx = tsid_hash(pk) % len(st.regions)
        code = st.regions[ridx].register_series(tags)
        per_region_hosts.setdefault(ridx, []).append((h, code))

    epochs = hours * 360
    nf = len(CPU_FIELDS)
    total = 0
    gen_dev = device if str(device).startswith("cuda") else "cpu"
    g = torch.Generator(device=gen_dev).manual_seed(seed * 7919 + rank)
    for ridx, hosts in per_region_hosts.items():
        region = st.regions[ridx]
        codes = torch.tensor([c for _, c in hosts], dtype=torch.int32, device=gen_dev)
        H = len(hosts)
        state = torch.rand((H, nf), generator=g, dtype=torch.float64, device=gen_dev) * 100
        e0 = 0
        while e0 < epochs:
            E = min(epoch_chunk, epochs - e0)
            steps = (torch.rand((H, nf, E), generator=g, dtype=torch.float64,
                                device=gen_dev) * 2 - 1)
            vals = torch.clamp(steps.cumsum(dim=2) + state[:, :, None], 0, 100)
            state = vals[:, :, -1].clone()
            ts0 = START_TS_S * 1000 + e0 * INTERVAL_MS
            ts_e = torch.arange(E, dtype=torch.int64, device=gen_dev) * INTERVAL_MS + ts0
            # host-major flatten, hosts ordered by code → sorted by (series, ts)
            order = torch.argsort(codes)
            ts_flat = ts_e.repeat(H)                          # [H*E]
            se_flat = codes[order].repeat_interleave(E)
            v = vals[order]                                   # [H, nf, E]
            vf = v.permute(1, 0, 2).reshape(nf, H * E)        # [nf, H*E]
            if f_map == list(range(len(field_names))):
                fields_dev = vf.contiguous()
            else:
                fields_dev = torch.full((len(field_names), H * E), float("nan"),
                                        dtype=torch.float64, device=gen_dev)
                for j, dst in enumerate(f_map):
                    fields_dev[dst] = vf[j]
            batch = sst_mod.SstBatch(
                ts_flat.to(device), se_flat.to(device), fields_dev.to(device),
                None, int(ts0), int(ts0 + (E - 1) * INTERVAL_MS),
                list(field_names))
            region.sst_cache[f"fixture_{ridx}_{e0}"] = batch
            total += H * E
            e0 += E
    return total


def _parse_tags(tagset: bytes):
    parts = tagset.decode().split(",")
    tags = dict(p.split("=", 1) for p in parts)
    return "cpu", tuple(tags.get(t) for t in CPU_TAGS)


# ---------------------------------------------------------------- queries

def tsbs_queries(scale: int, hours: int, rng: np.random.RandomState | None = None,
                 hosts_prefix: str = "host_") -> dict[str, str]:
    """The TSBS DevOps query suite (cpu-only), SQL form
    (docs/benchmarks/tsbs in the reference; same shapes/windows)."""
    rng = rng or np.random.RandomState(5)
    t0 = START_TS_S * 1000
    t_end = t0 + hours * 3600_000

    def rand_window(h):
        span = h * 3600_000
        if t_end - span <= t0:
            return t0, t_end
        s = int(rng.randint(t0, t_end - span))
        return s, s + span

    def hosts(n):
        hs = rng.choice(scale, size=min(n, scale), replace=False)
        return ", ".join(f"'{hosts_prefix}{h}'" for h in hs)

    q = {}
    for nf, nh, nhr in [(1, 1, 1), (1, 1, 12), (1, 8, 1), (5, 1, 1), (5, 1, 12), (5, 8, 1)]:
        lo, hi = rand_window(nhr)
        fields = ", ".join(f"max({f}) AS max_{f}" for f in CPU_FIELDS[:nf])
        q[f"single-groupby-{nf}-{nh}-{nhr}"] = (
            f"SELECT date_trunc('minute', ts) AS minute, {fields} FROM cpu "
            f"WHERE hostname IN ({hosts(nh)}) AND ts >= {lo} AND ts < {hi} "
            f"GROUP BY minute ORDER BY minute")
    for nh, name in [(1, "cpu-max-all-1"), (8, "cpu-max-all-8")]:
        lo, hi = rand_window(8)
        fields = ", ".join(f"max({f}) AS max_{f}" for f in CPU_FIELDS)
        q[name] = (
            f"SELECT date_trunc('hour', ts) AS hour, {fields} FROM cpu "
            f"WHERE hostname IN ({hosts(nh)}) AND ts >= {lo} AND ts < {hi} "
            f"GROUP BY hour ORDER BY hour")
    for nf, name in [(1, "double-groupby-1"), (5, "double-groupby-5"),
                     (10, "double-groupby-all")]:
        lo, hi = rand_window(12)
        fields = ", ".join(f"avg({f}) AS avg_{f}" for f in CPU_FIELDS[:nf])
        q[name] = (
            f"SELECT date_trunc('hour', ts) AS hour, hostname, {fields} FROM cpu "
            f"WHERE ts >= {lo} AND ts < {hi} "
            f"GROUP BY hour, hostname ORDER BY hour, hostname")
    lo, hi = rand_window(12)
    q["high-cpu-all"] = (
        f"SELECT * FROM cpu WHERE usage_user > 90.0 AND ts >= {lo} AND ts < {hi}")
    q["high-cpu-1"] = (
        f"SELECT * FROM cpu WHERE usage_user > 90.0 AND ts >= {lo} AND ts < {hi} "
        f"AND hostname IN ({hosts(1)})")
    lo, hi = rand_window(1)
    q["groupby-orderby-limit"] = (
        f"SELECT date_trunc('minute', ts) AS minute, max(usage_user) FROM cpu "
        f"WHERE ts < {hi} GROUP BY minute ORDER BY minute DESC LIMIT 5")
    q["lastpoint"] = (
        "SELECT hostname, last_value(usage_user) FROM cpu GROUP BY hostname "
        "ORDER BY hostname")
    # GreptimeDB RANGE-query shapes (no TSBS reference number): sliding
    # 1h windows aligned every 10m — the range_select plan's signature load
    lo, hi = rand_window(12)
    q["range-sliding-8"] = (
        f"SELECT ts, hostname, avg(usage_user) RANGE '1h' AS a FROM cpu "
        f"WHERE hostname IN ({hosts(8)}) AND ts >= {lo} AND ts < {hi} "
        f"ALIGN '10m' ORDER BY hostname, ts LIMIT 100")
    q["range-sliding-all"] = (
        f"SELECT ts, hostname, max(usage_user) RANGE '1h' AS m FROM cpu "
        f"WHERE ts >= {lo} AND ts < {hi} ALIGN '10m' "
        f"ORDER BY hostname, ts LIMIT 100")
    return q
