import cProfile, io, os, pstats, sys, tempfile, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from greptimedb_amd.engine.engine import EngineConfig, MitoEngine
from bench_promql import load_metric_fixture, START_MS
from greptimedb_amd.query.promql.eval import PromEvaluator

series = int(sys.argv[1]) if len(sys.argv) > 1 else 2_000_000
minutes = 30
dev = "cuda:0" if torch.cuda.is_available() else "cpu"
eng = MitoEngine(EngineConfig(data_dir=tempfile.mkdtemp(), device=dev,
                              background_flush=False))
t0 = time.perf_counter()
store, n, n_jobs = load_metric_fixture(eng, series, 100, minutes)
print("fixture", n, "samples in", round(time.perf_counter()-t0, 1), "s", flush=True)
ev = PromEvaluator(eng)
end_s = (START_MS + minutes*60_000)/1000 - 60
start_s = end_s - 1800
def run(q):
    m = ev.query_range(q, start_s, end_s, 60)
    if dev.startswith("cuda"): torch.cuda.synchronize()
    return m
run('sum by (job) ({__name__=~"metric_.*"})')   # warm
for q in ('sum by (job) ({__name__=~"metric_.*"})',
          'sum by (__name__) (rate({__name__=~"metric_.*"}[5m]))'):
    pr = cProfile.Profile()
    t0 = time.perf_counter()
    pr.enable()
    run(q)
    pr.disable()
    print("=== ", q, round((time.perf_counter()-t0)*1000, 1), "ms")
    s = io.StringIO()
    pstats.Stats(pr, stream=s).sort_stats("cumulative").print_stats(14)
    print("\n".join(s.getvalue().splitlines()[4:24]), flush=True)
