"""Multi-process distributed query combine (gloo, world_size=2, CPU).

Covers the RCCL partial-aggregate path (parallel/dist.py) with the same
torch.distributed code that runs over xGMI on the 8-GPU node — backend gloo
here, nccl(=RCCL) there.
"""

import json
import os
import subprocess
import sys

import numpy as np

WORKER = r"""
import json, os, sys
import numpy as np
import torch.distributed as dist

rank = int(os.environ["RANK"])
dist.init_process_group("gloo")

from greptimedb_amd.engine.engine import MitoEngine, EngineConfig
from greptimedb_amd.engine.ingest import Ingestor
from greptimedb_amd.models.tsbs import CpuWorkload
from greptimedb_amd.query.executor import Executor
from greptimedb_amd.parallel.dist import DistContext

base = sys.argv[1]
eng = MitoEngine(EngineConfig(data_dir=f"{base}/rank{rank}", device="cpu",
                              background_flush=False))
ing = Ingestor(eng)
# weak sharding: each rank owns a disjoint host range (P1 region sharding)
w = CpuWorkload(scale=10, seed=100 + rank)
# rename hosts to be disjoint across ranks
w.tagsets = [t.replace(b"host_", b"host_%d_" % rank) for t in w.tagsets]
for _ in range(5):
    ing.ingest_lines(w.next_batch(1000))

ex = Executor(eng, dist=DistContext(device="cpu"))
r1 = ex.execute("SELECT count(*) FROM cpu")
r2 = ex.execute("SELECT hostname, avg(usage_user) FROM cpu GROUP BY hostname ORDER BY hostname")
r3 = ex.execute("SELECT date_trunc('minute', ts) m, max(usage_user) FROM cpu GROUP BY m ORDER BY m")
r4 = ex.execute("SELECT ts, hostname, usage_user FROM cpu WHERE usage_user > 99")
r5 = ex.execute("SELECT hostname, last_value(usage_user) FROM cpu GROUP BY hostname ORDER BY hostname")
r5b = ex.execute("SELECT hostname, first_value(usage_user) FROM cpu GROUP BY hostname ORDER BY hostname")
# PromQL distributed aggregation (merge_prom_planes over gloo)
from greptimedb_amd.query.promql.eval import PromEvaluator
ev = PromEvaluator(eng, dist=DistContext(device="cpu"))
m = ev.query_range('sum({__field__="usage_user", __name__="cpu"})',
                   1451606450, 1451606450, 1)
# distributed RANGE query: partial planes merged over gloo
r6 = ex.execute("SELECT ts, hostname, avg(usage_user) RANGE '1m' AS a, "
                "max(usage_user) RANGE '1m' AS mx, count(usage_user) RANGE '1m' AS c "
                "FROM cpu ALIGN '30s' ORDER BY hostname, ts LIMIT 10000")
range_hosts = sorted(set(r6.columns[1]))
# distributed RANGE last_value (argmax-by-ts merge over gloo)
r7 = ex.execute("SELECT ts, hostname, last_value(usage_user) RANGE '1m' AS lv "
                "FROM cpu ALIGN '30s' ORDER BY hostname, ts LIMIT 10000")
# distributed quantile + count_values aggregations
mq = ev.query_range('quantile(0.5, {__field__="usage_user", __name__="cpu"})',
                    1451606450, 1451606450, 1)
mcv = ev.query_range('count_values("band", '
                     'floor({__field__="usage_user", __name__="cpu"} / 25))',
                     1451606450, 1451606450, 1)
cv_total = float(np.nansum(mcv.values.cpu().numpy()))
out = {
    "range_rows": len(r6),
    "range_hosts": len(range_hosts),
    "range_sum_a": round(float(np.nansum(np.asarray(r6.columns[2], dtype=float))), 4),
    "range_max": round(float(np.nanmax(np.asarray(r6.columns[3], dtype=float))), 4),
    "count": int(r1.columns[0][0]),
    "hosts": list(r2.columns[0]),
    "avgs": [float(x) for x in r2.columns[1]],
    "minutes": len(r3),
    "raw": len(r4),
    "lastpoint_hosts": len(r5),
    "firstpoint_hosts": len(r5b),
    "prom_sum": round(float(m.values[0][-1]), 6),
    "range_lv_rows": len(r7),
    "range_lv_hosts": len(sorted(set(r7.columns[1]))),
    "range_lv_sum": round(float(np.nansum(np.asarray(r7.columns[2], dtype=float))), 4),
    "quantile": round(float(mq.values[0][-1]), 6),
    "cv_series": mcv.S,
    "cv_total": cv_total,
}
print("RESULT" + str(rank) + json.dumps(out))
dist.destroy_process_group()
"""


def test_two_rank_query_combine(tmp_path):
    script = tmp_path / "worker.py"
    script.write_text(WORKER)
    env = dict(os.environ)
    repo_root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    env.update({
        "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": "29581",
        "WORLD_SIZE": "2",
        "PYTHONPATH": repo_root + os.pathsep + env.get("PYTHONPATH", ""),
    })
    procs = []
    for rank in range(2):
        e = dict(env)
        e["RANK"] = str(rank)
        procs.append(subprocess.Popen(
            [sys.executable, str(script), str(tmp_path)],
            env=e, stdout=subprocess.PIPE, stderr=subprocess.PIPE, text=True))
    outs = []
    for p in procs:
        out, err = p.communicate(timeout=180)
        assert p.returncode == 0, f"worker failed:\n{out}\n{err}"
        outs.append(out)
    results = []
    for rank, out in enumerate(outs):
        line = [ln for ln in out.splitlines() if ln.startswith(f"RESULT{rank}")][0]
        results.append(json.loads(line[len(f"RESULT{rank}"):]))
    # both ranks see the global result
    assert results[0] == results[1]
    assert results[0]["count"] == 10000      # 5000 per rank
    assert len(results[0]["hosts"]) == 20    # 10 hosts per rank, disjoint
    assert results[0]["raw"] >= 0
    assert results[0]["lastpoint_hosts"] == 20
    assert results[0]["firstpoint_hosts"] == 20
    assert results[0]["prom_sum"] != 0.0
    assert results[0]["range_hosts"] == 20   # all ranks' hosts in the plane
    assert results[0]["range_rows"] > 40
    assert results[0]["range_sum_a"] != 0.0
    assert results[0]["range_lv_hosts"] == 20
    assert results[0]["range_lv_sum"] != 0.0
    assert 0.0 <= results[0]["quantile"] <= 100.0
    assert results[0]["cv_total"] == 20.0    # one sample per series
    assert results[0]["cv_series"] >= 1
