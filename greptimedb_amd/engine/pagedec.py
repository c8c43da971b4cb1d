"""K11: GPU parquet page decode (hybrid host parse + device expand).

Reference hot loop: src/mito2/src/sst/parquet/reader.rs:152,766-860 (page
decode inside the parquet crate). MI355X split per SURVEY.md §7: the host
(csrc/pagedec.cpp) walks thrift PageHeaders and zstd-decompresses pages —
branchy, sequential; the device expands the bandwidth-bound parts — hybrid
RLE/bit-packed dictionary indices (rle_expand_indices_kernel) and the
dictionary gather — straight into HBM where the scan cache lives, so SST
open never round-trips decoded columns through host memory.

Covers the encodings this engine writes (and pyarrow defaults):
PLAIN + RLE_DICTIONARY over ZSTD, no nulls. Anything else raises and the
caller falls back to the pyarrow CPU reader.
"""

from __future__ import annotations

import numpy as np
import pyarrow.parquet as pq
import torch

from greptimedb_amd import _native

_CODEC = {"UNCOMPRESSED": 0, "ZSTD": 1}
_PHYS_NP = {"DOUBLE": np.float64, "INT64": np.int64, "INT32": np.int32,
            "FLOAT": np.float32}


def _expand_cpu(runs: np.ndarray, blob: bytes, bw: int, n: int) -> np.ndarray:
    out = np.empty(n, dtype=np.int32)
    for is_packed, off, val, start, count in runs:
        if not is_packed:
            out[start:start + count] = val
        else:
            nbytes = (count * bw + 7) // 8 + 8
            raw = np.frombuffer(blob[off:off + nbytes], dtype=np.uint8)
            bits = np.unpackbits(raw, bitorder="little")
            take = bits[: count * bw].reshape(count, bw).astype(np.int64)
            vals = (take * (1 << np.arange(bw, dtype=np.int64))).sum(axis=1)
            out[start:start + count] = vals
    return out


def read_numeric_column(path: str, name: str, device: str) -> torch.Tensor:
    """Decode one numeric column to a device tensor via the K11 path."""
    pf = pq.ParquetFile(path)
    md = pf.metadata
    col_idx = None
    for i in range(md.row_group(0).num_columns):
        if md.row_group(0).column(i).path_in_schema == name:
            col_idx = i
            break
    if col_idx is None:
        raise KeyError(name)
    phys = md.row_group(0).column(col_idx).physical_type
    np_dt = _PHYS_NP.get(phys)
    if np_dt is None:
        raise ValueError(f"unsupported physical type {phys}")
    max_def = pf.schema.column(col_idx).max_definition_level
    use_gpu = str(device).startswith("cuda")
    parts = []
    with open(path, "rb") as f:
        for rg in range(md.num_row_groups):
            c = md.row_group(rg).column(col_idx)
            codec = _CODEC.get(c.compression)
            if codec is None:
                raise ValueError(f"unsupported codec {c.compression}")
            start = c.data_page_offset
            if c.dictionary_page_offset is not None:
                start = min(start, c.dictionary_page_offset)
            f.seek(start)
            chunk = f.read(c.total_compressed_size)
            blob, pages, dict_off, dict_len = _native.parse_column_chunk(
                chunk, codec, max_def)
            dict_t = None
            if dict_off >= 0:
                dvals = np.frombuffer(blob, dtype=np_dt,
                                      count=dict_len // np.dtype(np_dt).itemsize,
                                      offset=dict_off)
                dict_t = torch.as_tensor(dvals.copy()).to(device)
            blob_t = None
            for kind, off, ln, nv in pages:
                if kind == 0:     # PLAIN values
                    vals = np.frombuffer(blob, dtype=np_dt, count=int(nv),
                                         offset=int(off))
                    parts.append(torch.as_tensor(vals.copy()).to(device))
                else:             # RLE_DICTIONARY indices
                    runs, bw = _native.rle_run_table(blob, int(off), int(ln),
                                                     int(nv))
                    if use_gpu:
                        from greptimedb_amd import _hip_ops
                        if blob_t is None:
                            padded = blob + b"\x00" * 16
                            blob_t = torch.as_tensor(
                                np.frombuffer(padded, dtype=np.uint8).copy()
                            ).to(device)
                        runs_t = torch.as_tensor(
                            np.ascontiguousarray(runs)).to(device)
                        idx = _hip_ops.rle_expand_indices(
                            runs_t, blob_t, int(bw), int(nv))
                    else:
                        idx = torch.as_tensor(
                            _expand_cpu(np.asarray(runs), blob, int(bw),
                                        int(nv)))
                    parts.append(dict_t[idx.long()])
    if not parts:
        return torch.zeros(0, dtype=torch.float64, device=device)
    return torch.cat(parts)


def read_columns(path: str, names: list[str], device: str) -> dict:
    return {n: read_numeric_column(path, n, device) for n in names}
