"""Table / column data model.

Reference parity: src/datatypes/src/schema/ and src/store-api/src/metadata.rs
(RegionMetadata: column ids, semantic types Tag/Field/Timestamp, primary key
column ordering). We keep the same semantic-type model because the SST format
(mito2 parquet) depends on it, but the in-memory layout is
GPU-first: every column is a torch tensor; tag (string) columns are
dictionary-encoded to int32 codes against a per-table dictionary so the GPU
only ever sees fixed-width data.

RegionId = (table_id << 32) | region_number, identical to
src/store-api/src/storage/descriptors.rs:72-73 so tooling/tests can cite the
same identifiers.
"""

from __future__ import annotations

import enum
from dataclasses import dataclass, field

import numpy as np

from greptimedb_amd.utils.errors import InvalidArguments


class SemanticType(enum.IntEnum):
    TAG = 0
    FIELD = 1
    TIMESTAMP = 2


class DataType(enum.Enum):
    """Logical column types. `np` / `torch` views are fixed-width."""

    BOOL = "bool"
    INT8 = "int8"
    INT16 = "int16"
    INT32 = "int32"
    INT64 = "int64"
    UINT8 = "uint8"
    UINT32 = "uint32"
    UINT64 = "uint64"
    FLOAT32 = "float32"
    FLOAT64 = "float64"
    STRING = "string"          # dictionary-encoded int32 codes in memory
    BINARY = "binary"
    VECTOR = "vector"          # fixed-dim f32 embedding (bytes at rest)
    TIMESTAMP_MS = "timestamp_ms"    # int64 millis (greptime default)
    TIMESTAMP_NS = "timestamp_ns"    # int64 nanos
    JSON = "json"              # stored as string

    @property
    def storage_np(self) -> np.dtype:
        """numpy dtype of the in-memory (device) representation."""
        m = {
            DataType.BOOL: np.uint8,
            DataType.INT8: np.int8,
            DataType.INT16: np.int16,
            DataType.INT32: np.int32,
            DataType.INT64: np.int64,
            DataType.UINT8: np.uint8,
            DataType.UINT32: np.uint32,
            DataType.UINT64: np.uint64,
            DataType.FLOAT32: np.float32,
            DataType.FLOAT64: np.float64,
            DataType.STRING: np.int32,
            DataType.BINARY: np.int32,
            DataType.VECTOR: np.int32,
            DataType.JSON: np.int32,
            DataType.TIMESTAMP_MS: np.int64,
            DataType.TIMESTAMP_NS: np.int64,
        }
        return np.dtype(m[self])

    @property
    def is_timestamp(self) -> bool:
        return self in (DataType.TIMESTAMP_MS, DataType.TIMESTAMP_NS)

    @property
    def is_string_like(self) -> bool:
        return self in (DataType.STRING, DataType.BINARY, DataType.JSON,
                        DataType.VECTOR)

    @property
    def is_float(self) -> bool:
        return self in (DataType.FLOAT32, DataType.FLOAT64)


@dataclass(frozen=True)
class ColumnSchema:
    name: str
    dtype: DataType
    semantic: SemanticType
    column_id: int
    nullable: bool = True
    fulltext: bool = False   # build a fulltext index (string fields only)
    vector_dim: int = 0      # VECTOR(D) dimension

    def __post_init__(self):
        if self.semantic == SemanticType.TIMESTAMP and not self.dtype.is_timestamp:
            raise InvalidArguments(f"timestamp column {self.name} must be a timestamp type")


@dataclass
class TableSchema:
    """Schema + key structure of one table (≈ reference RegionMetadata).

    primary_key: ordered tag column names (the series key).
    """

    name: str
    columns: list[ColumnSchema]
    primary_key: list[str]
    table_id: int = 0
    options: dict = field(default_factory=dict)

    def __post_init__(self):
        names = [c.name for c in self.columns]
        if len(set(names)) != len(names):
            raise InvalidArguments(f"duplicate column names in {self.name}")
        ts = [c for c in self.columns if c.semantic == SemanticType.TIMESTAMP]
        if len(ts) != 1:
            raise InvalidArguments(f"table {self.name} must have exactly one timestamp column")
        self._by_name = {c.name: c for c in self.columns}
        for pk in self.primary_key:
            if pk not in self._by_name:
                raise InvalidArguments(f"primary key column {pk} not in schema")

    @property
    def time_index(self) -> ColumnSchema:
        return next(c for c in self.columns if c.semantic == SemanticType.TIMESTAMP)

    @property
    def tag_columns(self) -> list[ColumnSchema]:
        return [self._by_name[n] for n in self.primary_key]

    @property
    def field_columns(self) -> list[ColumnSchema]:
        return [c for c in self.columns if c.semantic == SemanticType.FIELD]

    def column(self, name: str) -> ColumnSchema:
        try:
            return self._by_name[name]
        except KeyError:
            raise InvalidArguments(f"no column {name!r} in table {self.name}") from None

    def has_column(self, name: str) -> bool:
        return name in self._by_name

    def to_dict(self) -> dict:
        return {
            "name": self.name,
            "table_id": self.table_id,
            "primary_key": list(self.primary_key),
            "options": dict(self.options),
            "columns": [
                {
                    "name": c.name,
                    "dtype": c.dtype.value,
                    "semantic": int(c.semantic),
                    "column_id": c.column_id,
                    "nullable": c.nullable,
                    "fulltext": c.fulltext,
                    "vector_dim": c.vector_dim,
                }
                for c in self.columns
            ],
        }

    @staticmethod
    def from_dict(d: dict) -> "TableSchema":
        cols = [
            ColumnSchema(
                name=c["name"],
                dtype=DataType(c["dtype"]),
                semantic=SemanticType(c["semantic"]),
                column_id=c["column_id"],
                nullable=c.get("nullable", True),
                fulltext=c.get("fulltext", False),
                vector_dim=c.get("vector_dim", 0),
            )
            for c in d["columns"]
        ]
        return TableSchema(
            name=d["name"],
            columns=cols,
            primary_key=list(d["primary_key"]),
            table_id=d.get("table_id", 0),
            options=d.get("options", {}),
        )


def region_id(table_id: int, region_number: int) -> int:
    """RegionId = table_id<<32 | region_number (store-api descriptors.rs:72)."""
    return (table_id << 32) | region_number


def region_table_id(rid: int) -> int:
    return rid >> 32


def region_number(rid: int) -> int:
    return rid & 0xFFFFFFFF
