import tempfile, time
import numpy as np, torch
from bench_promql import load_metric_fixture, START_MS
from greptimedb_amd.engine.engine import EngineConfig, MitoEngine
from greptimedb_amd.query.promql.eval import PromEvaluator

eng = MitoEngine(EngineConfig(data_dir=tempfile.mkdtemp(), device="cuda:0", background_flush=False))
store, n, n_jobs = load_metric_fixture(eng, 1000000, 100, 60)
ev = PromEvaluator(eng)
end_s = (START_MS + 60*60_000)/1000 - 60
start_s = end_s - 1800
queries = [
    ("by-job-all", 'sum by (job) (rate({__name__=~"metric_.*"}[5m]))'),
    ("by-name-all", 'sum by (__name__) (rate({__name__=~"metric_.*"}[5m]))'),
    ("plain-sum-all", 'sum(rate({__name__=~"metric_.*"}[5m]))'),
    ("rate-sum-one", 'sum(rate(metric_1[5m]))'),
]
for name, q in queries:
    for it in range(3):
        torch.cuda.synchronize(); t0 = time.perf_counter()
        m = ev.query_range(q, start_s, end_s, 60)
        torch.cuda.synchronize()
        print(f"{name} iter{it}: {time.perf_counter()-t0:.2f}s S={m.S}", flush=True)
