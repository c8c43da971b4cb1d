"""TSBS DevOps `cpu-only` synthetic workload (the BASELINE.md headline config).

Matches the data shape of TSBS `cpu-only`: measurement `cpu`, 10 tags
(hostname..service_environment), 10 usage_* fields, one point per host per
10s interval (docs/benchmarks/tsbs/README.md in the reference). Values are a
clamped random walk in [0,100] like TSBS's host simulator. The generator
emits influx line protocol (the ingest wire format used by the reference's
TSBS runs) so the bench exercises the real parse path.
"""

from __future__ import annotations

import numpy as np

from greptimedb_amd.models.schema import ColumnSchema, DataType, SemanticType, TableSchema

CPU_TAGS = [
    "hostname", "region", "datacenter", "rack", "os", "arch",
    "team", "service", "service_version", "service_environment",
]

CPU_FIELDS = [
    "usage_user", "usage_system", "usage_idle", "usage_nice", "usage_iowait",
    "usage_irq", "usage_softirq", "usage_steal", "usage_guest", "usage_guest_nice",
]

_REGIONS = [
    "us-east-1", "us-west-1", "us-west-2", "eu-west-1", "eu-central-1",
    "ap-southeast-1", "ap-southeast-2", "ap-northeast-1", "sa-east-1",
]
_OS = ["Ubuntu16.10", "Ubuntu16.04LTS", "Ubuntu15.10"]
_ARCH = ["x64", "x86"]
_TEAMS = ["SF", "NYC", "LON", "CHI"]
_ENVS = ["production", "staging", "test"]


def cpu_table_schema(table_id: int = 1024) -> TableSchema:
    cols = []
    cid = 0
    for t in CPU_TAGS:
        cols.append(ColumnSchema(t, DataType.STRING, SemanticType.TAG, cid)); cid += 1
    cols.append(ColumnSchema("ts", DataType.TIMESTAMP_MS, SemanticType.TIMESTAMP, cid)); cid += 1
    for f in CPU_FIELDS:
        cols.append(ColumnSchema(f, DataType.FLOAT64, SemanticType.FIELD, cid)); cid += 1
    return TableSchema(name="cpu", columns=cols, primary_key=list(CPU_TAGS), table_id=table_id)


def host_tagsets(scale: int, seed: int = 1) -> list[bytes]:
    """Per-host influx tagset suffix `hostname=host_0,region=...` (sorted tag order
    is whatever TSBS emits — influx requires no sorting; we keep TSBS's order)."""
    rng = np.random.RandomState(seed)
    out = []
    for i in range(scale):
        vals = {
            "hostname": f"host_{i}",
            "region": _REGIONS[rng.randint(len(_REGIONS))],
            "datacenter": f"{_REGIONS[rng.randint(len(_REGIONS))]}{chr(ord('a') + rng.randint(3))}",
            "rack": str(rng.randint(100)),
            "os": _OS[rng.randint(len(_OS))],
            "arch": _ARCH[rng.randint(len(_ARCH))],
            "team": _TEAMS[rng.randint(len(_TEAMS))],
            "service": str(rng.randint(20)),
            "service_version": str(rng.randint(2)),
            "service_environment": _ENVS[rng.randint(len(_ENVS))],
        }
        out.append(",".join(f"{k}={vals[k]}" for k in CPU_TAGS).encode())
    return out


class CpuWorkload:
    """Streaming generator of influx-line-protocol batches for `cpu` points.

    One epoch = one point per host. `next_batch(n_rows)` returns bytes of
    n_rows lines, advancing time by `interval_s` per epoch like TSBS.
    """

    def __init__(self, scale: int = 100, start_ts_s: int = 1451606400,
                 interval_s: int = 10, seed: int = 7):
        self.scale = scale
        self.interval_ns = interval_s * 1_000_000_000
        self.ts_ns = start_ts_s * 1_000_000_000
        self.rng = np.random.RandomState(seed)
        self.tagsets = host_tagsets(scale, seed=seed)
        # random-walk state per host/field, clamped [0,100]
        self.values = self.rng.uniform(0, 100, size=(scale, len(CPU_FIELDS)))
        self._host = 0  # next host within current epoch

    def _advance(self, n: int) -> tuple[np.ndarray, np.ndarray]:
        """Return (host_idx[n], ts_ns[n]) advancing the epoch cursor."""
        idx = (self._host + np.arange(n)) % self.scale
        epoch = (self._host + np.arange(n)) // self.scale
        ts = self.ts_ns + epoch * self.interval_ns
        last = self._host + n
        self.ts_ns += (last // self.scale) * self.interval_ns
        self._host = last % self.scale
        return idx, ts

    def next_batch(self, n_rows: int) -> bytes:
        idx, ts = self._advance(n_rows)
        # step the random walk once per batch (cheap approximation of per-epoch walk)
        self.values += self.rng.uniform(-1, 1, size=self.values.shape)
        np.clip(self.values, 0, 100, out=self.values)
        vals = self.values[idx]  # (n, nf)
        lines = []
        fields = CPU_FIELDS
        for r in range(n_rows):
            h = idx[r]
            fv = vals[r]
            fstr = ",".join(f"{fields[j]}={fv[j]:.4f}" for j in range(len(fields)))
            lines.append(b"cpu," + self.tagsets[h] + b" " + fstr.encode() + b" " + str(ts[r]).encode())
        return b"\n".join(lines) + b"\n"
