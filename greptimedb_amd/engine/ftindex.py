"""Persisted per-SST fulltext index sidecar (Puffin analog).

Reference parity: src/puffin (blob container attached to SSTs) +
src/index/src/inverted_index/format.rs:15-34 — indexes are written next to
each SST and LOADED at region open instead of being rebuilt from raw
strings. MI355X design: the blob is the exact GPU posting layout
(term dictionary + CSR postings); open = read file → intern terms into the
region tokenizer (no doc re-tokenization) → upload the rows array to HBM.

File `<fid>.ftidx`, little-endian:
  magic  b"GFTX1\\n"
  u32 header_len + JSON header:
    {col: {"k": n_terms, "total": n_postings, "n_rows": rows_in_sst}}
  per column (header key order):
    term_lens  u32[k]
    term_blob  bytes (concatenated utf8, lengths above)
    starts     i64[k+1]   (CSR offsets into rows)
    rows       i64[total] (ascending doc ids per term)
"""

from __future__ import annotations

import json
import os
import struct

import numpy as np
import torch

MAGIC = b"GFTX1\n"


def sidecar_path(sst_path: str) -> str:
    return sst_path[: -len(".parquet")] + ".ftidx" \
        if sst_path.endswith(".parquet") else sst_path + ".ftidx"


def save_sidecar(sst_path: str, text_index: dict, text_cols: dict) -> str | None:
    """Write the sidecar for one SST. text_index: {col: SegmentPostings};
    text_cols: {col: FulltextColumn} (for tid → term string)."""
    if not text_index:
        return None
    header = {}
    bodies = []
    for col, seg in text_index.items():
        tok = text_cols[col].tokenizer
        terms = [tok.term_str(int(t)).encode() for t in seg.uniq_tids]
        lens = np.array([len(t) for t in terms], dtype=np.uint32)
        blob = b"".join(terms)
        rows = seg.rows.cpu().numpy().astype(np.int64)
        header[col] = {"k": len(terms), "total": int(rows.size),
                       "n_rows": seg.n_rows}
        bodies.append((lens.tobytes(), blob,
                       seg.starts.astype(np.int64).tobytes(), rows.tobytes()))
    hdr = json.dumps(header).encode()
    path = sidecar_path(sst_path)
    tmp = path + ".tmp"
    with open(tmp, "wb") as f:
        f.write(MAGIC)
        f.write(struct.pack("<I", len(hdr)))
        f.write(hdr)
        for parts in bodies:
            for p in parts:
                f.write(p)
    os.replace(tmp, path)
    return path


def load_sidecar(sst_path: str, text_cols: dict, device,
                 row_remap: np.ndarray | None = None) -> dict | None:
    """Load a sidecar → {col: SegmentPostings} with tids interned into the
    region tokenizers. row_remap (old sst row → current batch row) applies
    when the open path re-permuted rows. Returns None if absent/invalid."""
    from greptimedb_amd.engine.fulltext import SegmentPostings

    path = sidecar_path(sst_path)
    if not os.path.exists(path):
        return None
    with open(path, "rb") as f:
        buf = f.read()
    if not buf.startswith(MAGIC):
        return None
    off = len(MAGIC)
    (hlen,) = struct.unpack_from("<I", buf, off)
    off += 4
    header = json.loads(buf[off:off + hlen].decode())
    off += hlen
    out = {}
    for col, meta in header.items():
        k, total, n_rows = meta["k"], meta["total"], meta["n_rows"]
        lens = np.frombuffer(buf, dtype=np.uint32, count=k, offset=off)
        off += 4 * k
        blob_len = int(lens.sum())
        blob = buf[off:off + blob_len]
        off += blob_len
        starts = np.frombuffer(buf, dtype=np.int64, count=k + 1, offset=off).copy()
        off += 8 * (k + 1)
        rows = np.frombuffer(buf, dtype=np.int64, count=total, offset=off).copy()
        off += 8 * total
        ft = text_cols.get(col)
        if ft is None:
            continue
        tids = ft.tokenizer.intern_blob(lens.astype(np.int32), blob) if k else \
            np.zeros(0, dtype=np.int32)
        tids = np.asarray(tids, dtype=np.int32)
        # probe binary-searches uniq_tids → re-sort the CSR by tid
        # (fully vectorized gather: O(total), no per-term python loop)
        order = np.argsort(tids, kind="stable")
        if k and not np.array_equal(order, np.arange(k)):
            counts = np.diff(starts)
            lens = counts[order]
            new_starts = np.zeros(k + 1, dtype=np.int64)
            np.cumsum(lens, out=new_starts[1:])
            seg_id = np.repeat(np.arange(k), lens)
            within = np.arange(int(lens.sum()), dtype=np.int64) - \
                np.repeat(new_starts[:-1], lens)
            gather = starts[order][seg_id] + within
            tids, starts, rows = tids[order], new_starts, rows[gather]
        if row_remap is not None and rows.size:
            rows = row_remap[rows]
        out[col] = SegmentPostings(tids, starts,
                                   torch.as_tensor(rows).to(device), n_rows)
    return out
