"""Fulltext index for log string columns (K13/K15).

Reference parity: src/index fulltext_index (tantivy-based) + the
matches/matches_term UDFs (src/common/function). MI355X redesign: the
tokenizer (csrc/native.cpp, lowercase alnum runs ≈ tantivy default) interns
terms region-wide; flushed segments hold CSR posting lists as DEVICE
tensors, so a MATCHES probe is a handful of index_fill_ scatters on GPU —
the FST/posting-file probe of the reference becomes device bitmap work, and
the term dictionary itself is the exact "skipping index": a segment whose
dictionary lacks a query term is skipped entirely (strictly stronger than
the reference's bloom filter skipping index, no false positives).
"""

from __future__ import annotations

import numpy as np
import torch

from greptimedb_amd import _native


class SegmentPostings:
    """Immutable per-segment posting lists: term id → row ids (device)."""

    def __init__(self, uniq_tids: np.ndarray, starts: np.ndarray,
                 rows: torch.Tensor, n_rows: int):
        self.uniq_tids = uniq_tids      # sorted i32 [k] host
        self.starts = starts            # i64 [k+1] host
        self.rows = rows                # i64 [total] device
        self.n_rows = n_rows

    @staticmethod
    def build(offsets: np.ndarray, tids: np.ndarray, n_rows: int, device):
        """From tokenizer CSR (doc offsets + term ids per doc)."""
        if len(tids) == 0:
            return SegmentPostings(np.zeros(0, np.int32), np.zeros(1, np.int64),
                                   torch.zeros(0, dtype=torch.int64, device=device),
                                   n_rows)
        doc_of = np.repeat(np.arange(n_rows, dtype=np.int64),
                           np.diff(offsets).astype(np.int64))
        order = np.argsort(tids, kind="stable")
        st = tids[order]
        rows_sorted = doc_of[order]
        uniq, starts = np.unique(st, return_index=True)
        starts = np.concatenate([starts, [len(st)]]).astype(np.int64)
        return SegmentPostings(uniq.astype(np.int32), starts,
                               torch.as_tensor(rows_sorted).to(device), n_rows)

    def probe(self, tids: list[int], device) -> torch.Tensor:
        """Rows containing ALL terms → bool [n_rows] (device)."""
        mask = None
        for tid in tids:
            i = np.searchsorted(self.uniq_tids, tid)
            if i >= len(self.uniq_tids) or self.uniq_tids[i] != tid:
                return torch.zeros(self.n_rows, dtype=torch.bool, device=device)
            rows = self.rows[int(self.starts[i]): int(self.starts[i + 1])]
            m = torch.zeros(self.n_rows, dtype=torch.bool, device=device)
            m.index_fill_(0, rows, True)
            mask = m if mask is None else (mask & m)
        if mask is None:
            return torch.ones(self.n_rows, dtype=torch.bool, device=device)
        return mask


class MemPostings:
    """Mutable postings over the live memtable rows (host side)."""

    def __init__(self):
        self.lists: dict[int, list] = {}
        self.n_rows = 0

    def append_batch(self, offsets: np.ndarray, tids: np.ndarray):
        n = len(offsets) - 1
        if len(tids):
            doc_of = np.repeat(np.arange(n, dtype=np.int64) + self.n_rows,
                               np.diff(offsets).astype(np.int64))
            order = np.argsort(tids, kind="stable")
            st = tids[order]
            rows_sorted = doc_of[order]
            uniq, starts = np.unique(st, return_index=True)
            starts = np.concatenate([starts, [len(st)]])
            for i, t in enumerate(uniq):
                self.lists.setdefault(int(t), []).append(
                    rows_sorted[starts[i]:starts[i + 1]])
        self.n_rows += n

    def probe(self, tids: list[int], n_rows: int, device) -> torch.Tensor:
        mask = None
        for tid in tids:
            parts = self.lists.get(tid)
            if not parts:
                return torch.zeros(n_rows, dtype=torch.bool, device=device)
            rows = np.concatenate(parts)
            rows = rows[rows < n_rows]
            m = torch.zeros(n_rows, dtype=torch.bool, device=device)
            m.index_fill_(0, torch.as_tensor(rows).to(device), True)
            mask = m if mask is None else (mask & m)
        if mask is None:
            return torch.ones(n_rows, dtype=torch.bool, device=device)
        return mask


class FulltextColumn:
    """Per-region, per-column fulltext state: shared term dict + memtable
    postings (segments live on the SstBatch)."""

    def __init__(self):
        self.tokenizer = _native.Tokenizer()
        self.mem = MemPostings()

    def index_batch(self, docs: list):
        offsets, tids, _new = self.tokenizer.tokenize(docs)
        self.mem.append_batch(offsets, tids)

    def reset_mem(self):
        self.mem = MemPostings()

    def query_tids(self, terms: list[str]) -> list[int] | None:
        """None ⇒ some term unknown region-wide (no rows can match)."""
        out = []
        for t in terms:
            tid = self.tokenizer.term_id(t)
            if tid < 0:
                return None
            out.append(tid)
        return out

    def build_segment(self, docs_sorted: list, device) -> SegmentPostings:
        offsets, tids, _ = self.tokenizer.tokenize(docs_sorted)
        return SegmentPostings.build(offsets, tids, len(docs_sorted), device)
