"""Minimal kernel-only loop for PMC counter collection (rocprofv3)."""
import numpy as np, torch
from greptimedb_amd.ops import kernels

n = 50_000_000
dev = "cuda:0"
ts = torch.arange(n, dtype=torch.int64, device=dev) % 86_400_000
series = (torch.arange(n, dtype=torch.int32, device=dev) // 10_000) % 4000
fields = torch.rand((10, n), dtype=torch.float64, device=dev)
fidx = torch.arange(10, dtype=torch.int32, device=dev)
lut = torch.arange(4000, dtype=torch.int32, device=dev)
torch.cuda.synchronize()
for _ in range(3):
    out = kernels.ts_bucket_agg(ts, series, fields, fidx, lut,
                                0, 86_400_000, 0, 3_600_000, 4000, 24)
torch.cuda.synchronize()
import time
t0 = time.perf_counter()
for _ in range(5):
    out = kernels.ts_bucket_agg(ts, series, fields, fidx, lut,
                                0, 86_400_000, 0, 3_600_000, 4000, 24)
torch.cuda.synchronize()
dt = (time.perf_counter() - t0) / 5
gb = n * (12 + 4 + 10 * 12) / 1e9
print(f"agg 50M rows x10 fields: {dt*1000:.2f} ms, ~{gb/dt:.0f} GB/s effective")
