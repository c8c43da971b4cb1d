"""kill -9 crash recovery over WAL + manifest + catalog.

A child process ingests continuously with fsync'd WAL commits and records
its durable high-water mark; the parent SIGKILLs it at a random moment and
asserts the reopened engine (a) recovers at least every acked row,
(b) is internally consistent, (c) reopens deterministically, and
(d) accepts new writes. Reference intent: tests-fuzz failover targets +
raft-engine crash semantics.
"""

import os
import signal
import subprocess
import sys
import time

import pytest

CHILD = r"""
import os, sys
import torch  # noqa
from greptimedb_amd.engine.engine import EngineConfig, MitoEngine
from greptimedb_amd.engine.ingest import Ingestor
from greptimedb_amd.models.tsbs import CpuWorkload

data_dir, ack_path, shards = sys.argv[1], sys.argv[2], int(sys.argv[3])
eng = MitoEngine(EngineConfig(data_dir=data_dir, device="cpu",
                              background_flush=True, flush_bytes=200_000,
                              wal_sync=True, wal_shards=shards,
                              wal_segment_bytes=1 << 16))
ing = Ingestor(eng)
w = CpuWorkload(scale=10)
total = 0
print("READY", flush=True)
while True:
    ing.ingest_lines(w.next_batch(500))   # fsync'd group commit inside
    total += 500
    with open(ack_path + ".tmp", "w") as f:
        f.write(str(total))
        f.flush()
        os.fsync(f.fileno())
    os.replace(ack_path + ".tmp", ack_path)
"""


def _total_rows(data_dir):
    from greptimedb_amd.engine.engine import EngineConfig, MitoEngine
    eng = MitoEngine(EngineConfig(data_dir=data_dir, device="cpu",
                                  background_flush=False))
    try:
        counts = {name: sum(r.num_rows for r in st.regions)
                  for name, st in eng.tables.items()}
        series = {name: sorted(pk for r in st.regions for pk in r.series.pks)
                  for name, st in eng.tables.items()}
        return counts, series, eng
    except Exception:
        eng.close()
        raise


@pytest.mark.parametrize("delay_ms,shards", [(150, 1), (400, 4), (800, 2)])
def test_kill9_recovers_acked_rows(tmp_path, delay_ms, shards):
    data_dir = str(tmp_path / "data")
    ack_path = str(tmp_path / "ack")
    proc = subprocess.Popen([sys.executable, "-c", CHILD, data_dir, ack_path,
                             str(shards)],
                            stdout=subprocess.PIPE, cwd=os.path.dirname(
                                os.path.dirname(os.path.abspath(__file__))))
    assert proc.stdout.readline().strip() == b"READY"
    deadline = time.monotonic() + 30
    while not os.path.exists(ack_path) and time.monotonic() < deadline:
        time.sleep(0.01)
    time.sleep(delay_ms / 1000)
    os.kill(proc.pid, signal.SIGKILL)
    proc.wait(timeout=30)
    acked = int(open(ack_path).read()) if os.path.exists(ack_path) else 0
    assert acked > 0, "child never acked a batch"

    counts, series, eng = _total_rows(data_dir)
    eng.close()
    recovered = sum(counts.values())
    # durability: every fsync-acked row survives kill -9
    assert recovered >= acked, f"lost acked rows: {recovered} < {acked}"
    # bounded over-recovery: at most the one in-flight batch beyond the ack
    assert recovered <= acked + 500

    # determinism: a second reopen sees the identical state
    counts2, series2, eng2 = _total_rows(data_dir)
    assert counts2 == counts and series2 == series
    # liveness: the recovered engine accepts new writes + flushes
    from greptimedb_amd.engine.ingest import Ingestor
    from greptimedb_amd.models.tsbs import CpuWorkload
    ing = Ingestor(eng2)
    w = CpuWorkload(scale=3, seed=99)
    ing.ingest_lines(w.next_batch(50))
    eng2.flush_all()
    counts3 = {name: sum(r.num_rows for r in st.regions)
               for name, st in eng2.tables.items()}
    assert sum(counts3.values()) == recovered + 50
    eng2.close()
