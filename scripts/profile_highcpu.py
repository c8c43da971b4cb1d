import cProfile, io, os, pstats, sys, tempfile, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from greptimedb_amd.engine.engine import EngineConfig, MitoEngine
from greptimedb_amd.models.tsbs_fixture import load_cpu_fixture, tsbs_queries
from greptimedb_amd.query.executor import Executor
dev = "cuda:0" if torch.cuda.is_available() else "cpu"
eng = MitoEngine(EngineConfig(data_dir=tempfile.mkdtemp(), device=dev,
                              background_flush=False))
load_cpu_fixture(eng, scale=4000, hours=72)
ex = Executor(eng)
qs = tsbs_queries(4000, 72)
for name in ("high-cpu-all", "range-sliding-all", "double-groupby-all"):
    q = qs[name]
    ex.execute(q)
    if dev.startswith("cuda"): torch.cuda.synchronize()
    pr = cProfile.Profile()
    t0 = time.perf_counter()
    pr.enable()
    r = ex.execute(q)
    if dev.startswith("cuda"): torch.cuda.synchronize()
    pr.disable()
    print("===", name, round((time.perf_counter()-t0)*1000, 1), "ms rows", len(r))
    s = io.StringIO()
    pstats.Stats(pr, stream=s).sort_stats("tottime").print_stats(10)
    print("\n".join(s.getvalue().splitlines()[4:20]), flush=True)
