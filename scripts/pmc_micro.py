"""Micro PMC target: the two hottest kernels on synthetic 100M-row data.
Run under rocprofv3 --pmc FETCH_SIZE WRITE_SIZE --stats (short + bounded)."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np, torch
from greptimedb_amd.ops import kernels
from greptimedb_amd.engine import gorilla
from greptimedb_amd import _hip_ops

dev = "cuda:0"
n = 100_000_000
n_series = 4000 * 10
g = torch.Generator(device=dev).manual_seed(1)
ts = torch.arange(n, dtype=torch.int64, device=dev) % 259_200_000
se = torch.randint(0, n_series, (n,), dtype=torch.int32, device=dev, generator=g)
fields = torch.rand((2, n), dtype=torch.float64, device=dev, generator=g) * 100
fi = torch.arange(2, dtype=torch.int32, device=dev)
lut = torch.arange(n_series, dtype=torch.int32, device=dev)
torch.cuda.synchronize()
for label, se_case in (("random-series", se),
                       ("sorted-series", torch.sort(se)[0].contiguous())):
    t0 = time.perf_counter()
    for _ in range(3):
        out = kernels.ts_bucket_agg(ts, se_case, fields.contiguous(), fi, lut,
                                    0, 259_200_000, 0, 3_600_000, n_series, 72)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / 3
    print(f"ts_bucket_agg[{label}]:", round(dt*1000, 1), "ms for", n,
          "rows x2 fields ->", round(n*(8+4+16)/dt/1e9, 0), "GB/s in")
# gorilla decode
rng = np.random.RandomState(0)
m = 50_000_000
tsh = 1451606400000 + np.arange(m, dtype=np.int64) * 10_000
vals = np.round(np.clip(np.cumsum(rng.uniform(-1, 1, m)) + 50, 0, 100), 4)
blob, bo, oo, nn = gorilla.pack(tsh, vals)
blob_t = torch.as_tensor(np.frombuffer(blob, np.uint8).copy()).to(dev)
bo_t = torch.as_tensor(bo).to(dev); oo_t = torch.as_tensor(oo).to(dev)
torch.cuda.synchronize()
t0 = time.perf_counter()
for _ in range(3):
    _hip_ops.gorilla_decode(blob_t, bo_t, oo_t, nn)
torch.cuda.synchronize()
dt = (time.perf_counter() - t0) / 3
print("gorilla_decode:", round(dt*1000, 2), "ms ->", round(m*16/dt/1e9, 0), "GB/s out")
