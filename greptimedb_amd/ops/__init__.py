"""Device op dispatch.

Policy: on a GPU tensor the hand-written HIP kernels (greptimedb_amd._hip_ops,
gfx950) are the ONLY path — if the extension is missing we raise
NativeExtensionMissing rather than silently falling back to eager PyTorch.
CPU tensors use the reference implementations in ops.cpu_ref (plain
PyTorch/f64), which double as the numerics oracle for GPU tests.
"""

from greptimedb_amd.ops.kernels import (  # noqa: F401
    dedup_mark_last,
    filter_series_time,
    hip_ops_available,
    prom_range_eval,
    series_last,
    ts_bucket_agg,
    ts_bucket_agg_acc,
    ts_bucket_agg_finish,
)
