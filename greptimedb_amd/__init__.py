"""greptimedb_amd — MI355X-native observability database engine.

A from-scratch rebuild of GreptimeDB's capabilities (reference:
GreptimeTeam/greptimedb, surveyed in SURVEY.md) designed MI355X-first:

- columnar storage engine ("mito-hip") whose scan / filter / time-bucket
  aggregate / merge-dedup / PromQL range-vector hot paths are hand-written
  CDNA4 HIP kernels over GPU-resident column tensors,
- region shards partitioned across the GPUs of one node, partial aggregates
  combined with RCCL collectives over xGMI (torch.distributed backend "nccl"),
- host-side native (C++) ingest path: influx line-protocol parser + WAL with
  group commit,
- mito2-compatible Parquet SST format on disk/object storage.

Layer map mirrors SURVEY.md §1 (reference: src/cmd .. src/mito2) but the
implementation is new, GPU-first, and not a port.
"""

__version__ = "0.1.0"

from greptimedb_amd.utils.errors import GreptimeError  # noqa: F401
