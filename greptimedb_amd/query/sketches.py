"""Approximate aggregation sketches: HyperLogLog + UDDSketch.

Reference parity: src/common/function/src/aggrs/approximate/{hll.rs,
uddsketch.rs} (aggregates `hll`, `hll_merge`, `uddsketch_state`,
`uddsketch_merge`) and the scalar calculators `hll_count` /
`uddsketch_calc` (src/common/function/src/scalars/). States are opaque
bytes, mergeable across ranks/windows — the same two-level (state → calc)
shape the reference uses with flows.
"""

from __future__ import annotations

import math
import struct

import numpy as np
import xxhash

HLL_P = 14                   # 2^14 registers (reference hll.rs precision)
HLL_M = 1 << HLL_P
_HLL_ALPHA = 0.7213 / (1 + 1.079 / HLL_M)


class Hll:
    __slots__ = ("regs",)

    def __init__(self, regs: np.ndarray | None = None):
        self.regs = regs if regs is not None else np.zeros(HLL_M, dtype=np.uint8)

    def add_values(self, values) -> "Hll":
        for v in values:
            if v is None or (isinstance(v, float) and math.isnan(v)):
                continue
            h = xxhash.xxh64_intdigest(str(v).encode())
            idx = h & (HLL_M - 1)
            w = h >> HLL_P
            # rank = leading-zero count of the remaining 64-P bits + 1
            rank = (64 - HLL_P) - w.bit_length() + 1
            if rank > self.regs[idx]:
                self.regs[idx] = rank
        return self

    def merge(self, other: "Hll") -> "Hll":
        np.maximum(self.regs, other.regs, out=self.regs)
        return self

    def count(self) -> float:
        regs = self.regs.astype(np.float64)
        est = _HLL_ALPHA * HLL_M * HLL_M / np.sum(np.exp2(-regs))
        zeros = int(np.count_nonzero(self.regs == 0))
        if est <= 2.5 * HLL_M and zeros:
            est = HLL_M * math.log(HLL_M / zeros)   # linear counting
        return float(est)

    def dumps(self) -> bytes:
        return b"HLL1" + self.regs.tobytes()

    @staticmethod
    def loads(b: bytes) -> "Hll":
        assert b[:4] == b"HLL1", "not an hll state"
        return Hll(np.frombuffer(b[4:], dtype=np.uint8).copy())


class UddSketch:
    """Log-bucketed quantile sketch (UDDSketch paper; reference
    uddsketch.rs). State: {bucket_index: count} at error rate alpha."""

    __slots__ = ("alpha", "gamma", "buckets", "n", "zero_count", "max_buckets")

    def __init__(self, max_buckets: int = 128, alpha: float = 0.01):
        self.max_buckets = max_buckets
        self.alpha = alpha
        self.gamma = (1 + alpha) / (1 - alpha)
        self.buckets: dict[int, int] = {}
        self.zero_count = 0
        self.n = 0

    def _key(self, v: float) -> int:
        av = abs(v)
        k = math.ceil(math.log(av) / math.log(self.gamma))
        return k if v > 0 else -k - (1 << 30)   # negatives in a shifted space

    def add_values(self, values) -> "UddSketch":
        for v in values:
            if v is None or (isinstance(v, float) and math.isnan(v)):
                continue
            v = float(v)
            self.n += 1
            if v == 0.0:
                self.zero_count += 1
                continue
            k = self._key(v)
            self.buckets[k] = self.buckets.get(k, 0) + 1
            if len(self.buckets) > self.max_buckets:
                self._compact()
        return self

    def _compact(self):
        """Double alpha (halve resolution) — UDDSketch's uniform collapse."""
        self.alpha = 2 * self.alpha / (1 + self.alpha ** 2)
        self.gamma = (1 + self.alpha) / (1 - self.alpha)
        old = self.buckets
        self.buckets = {}
        for k, c in old.items():
            neg = k < -(1 << 29)
            kk = (-(k + (1 << 30))) if neg else k
            nk = math.ceil(kk / 2)
            nk = (-nk - (1 << 30)) if neg else nk
            self.buckets[nk] = self.buckets.get(nk, 0) + c

    def merge(self, other: "UddSketch") -> "UddSketch":
        while abs(self.alpha - other.alpha) > 1e-12:
            if self.alpha < other.alpha:
                self._compact()
            else:
                other = other._copy_compacted()
        for k, c in other.buckets.items():
            self.buckets[k] = self.buckets.get(k, 0) + c
        self.zero_count += other.zero_count
        self.n += other.n
        return self

    def _copy_compacted(self) -> "UddSketch":
        c = UddSketch(self.max_buckets, self.alpha)
        c.buckets = dict(self.buckets)
        c.zero_count = self.zero_count
        c.n = self.n
        c._compact()
        return c

    def quantile(self, q: float) -> float:
        if self.n == 0:
            return float("nan")
        rank = q * (self.n - 1)
        # order: negatives (descending |v|), zeros, positives
        neg = sorted((k for k in self.buckets if k < -(1 << 29)),
                     key=lambda k: -(-(k + (1 << 30))))
        pos = sorted(k for k in self.buckets if k >= -(1 << 29))
        acc = 0
        for k in neg:
            acc += self.buckets[k]
            if acc > rank:
                kk = -(k + (1 << 30))
                return -2 * self.gamma ** kk / (self.gamma + 1)
        if acc + self.zero_count > rank:
            return 0.0
        acc += self.zero_count
        for k in pos:
            acc += self.buckets[k]
            if acc > rank:
                return 2 * self.gamma ** k / (self.gamma + 1)
        return float("nan")

    def dumps(self) -> bytes:
        items = sorted(self.buckets.items())
        out = [b"UDD1", struct.pack("<dqq", self.alpha, self.n,
                                    self.zero_count),
               struct.pack("<q", len(items))]
        for k, c in items:
            out.append(struct.pack("<qq", k, c))
        return b"".join(out)

    @staticmethod
    def loads(b: bytes) -> "UddSketch":
        assert b[:4] == b"UDD1", "not a uddsketch state"
        alpha, n, zc = struct.unpack_from("<dqq", b, 4)
        (k_n,) = struct.unpack_from("<q", b, 28)
        s = UddSketch(alpha=alpha)
        s.n, s.zero_count = n, zc
        off = 36
        for _ in range(k_n):
            k, c = struct.unpack_from("<qq", b, off)
            off += 16
            s.buckets[k] = c
        return s


# ------------------------------------------------------------- calc UDFs

def hll_count(state: bytes) -> float:
    return Hll.loads(state).count()


def uddsketch_calc(q: float, state: bytes) -> float:
    return UddSketch.loads(state).quantile(q)
