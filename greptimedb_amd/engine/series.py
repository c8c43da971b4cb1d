"""Per-region series index: local dense codes ↔ tag values ↔ encoded pk.

Reference parity: mito2 memtable/time_series.rs keys its BTreeMap by encoded
primary key and src/mito2/src/series_index.rs tracks series; the metric
engine hashes labels to __tsid (row_modifier.rs:98). Here every region keeps
a dense local code per series (what the GPU columns store) plus:
  - encoded pk bytes (SST __primary_key dictionary values),
  - decoded tag value tuple (for group-by output),
  - per-tag inverted map value → [codes] (host-side tag predicate probe —
    the K13 analog for tag equality/IN filters).
"""

from __future__ import annotations

import numpy as np

from greptimedb_amd.engine import pk_codec


def tsid_hash(pk: bytes) -> int:
    """FNV-1a 64 of the encoded pk (stable __tsid; reference uses a label
    hash in metric-engine row_modifier.rs:98)."""
    h = 0xCBF29CE484222325
    for b in pk:
        h = ((h ^ b) * 0x100000001B3) & 0xFFFFFFFFFFFFFFFF
    return h


class SeriesIndex:
    def __init__(self, tag_names: list[str]):
        self.tag_names = list(tag_names)
        self.pk_to_code: dict[bytes, int] = {}
        self.pks: list[bytes] = []
        self.tag_values: list[tuple] = []
        # tag name -> value -> list[int] codes
        self.inverted: dict[str, dict[str, list[int]]] = {t: {} for t in tag_names}
        self._codes_cache: dict = {}

    def __len__(self) -> int:
        return len(self.pks)

    def get_or_create(self, tags: tuple) -> int:
        pk = pk_codec.encode_pk(tags)
        code = self.pk_to_code.get(pk)
        if code is None:
            code = self.add(pk, tags)
        return code

    def add(self, pk: bytes, tags: tuple) -> int:
        code = len(self.pks)
        self.pk_to_code[pk] = code
        self.pks.append(pk)
        self.tag_values.append(tags)
        for t, v in zip(self.tag_names, tags):
            if v is not None:
                self.inverted[t].setdefault(v, []).append(code)
        return code

    def add_encoded(self, pk: bytes) -> int:
        """Register a series seen only as encoded pk (SST load / series log)."""
        code = self.pk_to_code.get(pk)
        if code is not None:
            return code
        if pk_codec.is_sparse(pk):
            return self.add_labels(pk, pk_codec.decode_sparse(pk))
        tags = pk_codec.decode_pk(pk, len(self.tag_names))
        return self.add(pk, tags)

    # -------- sparse / metric-engine mode: dynamic label sets --------

    def get_or_create_labels(self, labels: dict) -> int:
        pk = pk_codec.encode_sparse(labels)
        code = self.pk_to_code.get(pk)
        if code is None:
            code = self.add_labels(pk, labels)
        return code

    def add_labels(self, pk: bytes, labels: dict) -> int:
        for name in labels:
            if name not in self.inverted:
                self.tag_names.append(name)
                self.inverted[name] = {}
        code = len(self.pks)
        self.pk_to_code[pk] = code
        self.pks.append(pk)
        self.tag_values.append(tuple(labels.get(n) for n in self.tag_names))
        for n, v in labels.items():
            if v is not None:
                self.inverted[n].setdefault(v, []).append(code)
        return code

    # ---------------- tag predicate → slot LUT helpers ----------------

    def codes_for_eq(self, tag: str, value: str) -> list[int]:
        return self.inverted.get(tag, {}).get(value, [])

    def codes_for_in(self, tag: str, values: list[str]) -> list[int]:
        out: list[int] = []
        inv = self.inverted.get(tag, {})
        for v in values:
            out.extend(inv.get(v, []))
        return out

    def tag_array(self, tag: str) -> np.ndarray:
        """Object array of this tag's value per code (group-by output).
        Sparse mode: rows registered before a label first appeared have
        shorter tuples → None."""
        i = self.tag_names.index(tag)
        return np.array([t[i] if i < len(t) else None for t in self.tag_values],
                        dtype=object)

    def labels_of(self, code: int) -> dict:
        t = self.tag_values[code]
        return {n: v for n, v in zip(self.tag_names, t) if v is not None}

    def tag_codes(self, tag: str):
        """Factorized tag column: (value_id i32[n_series] (-1 = absent),
        values list). Built from the inverted index (vectorized over unique
        values, not series) and cached until new series arrive."""
        cached = self._codes_cache.get(tag)
        n = len(self.pks)
        if cached is not None and cached[2] == n:
            return cached[0], cached[1]
        arr = np.full(n, -1, dtype=np.int32)
        values = []
        inv = self.inverted.get(tag, {})
        for i, (v, codes) in enumerate(inv.items()):
            values.append(v)
            arr[np.asarray(codes, dtype=np.int64)] = i
        self._codes_cache[tag] = (arr, values, n)
        return arr, values
