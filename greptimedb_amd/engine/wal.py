"""WAL: sharded segment management + batch payload codec over the native writer.

Reference parity: src/log-store raft_engine backend + mito2/src/wal.rs
(WalWriter group commit, per-region entries, replay from entry id,
obsolete/purge). Segments are `{first_seq:020d}[.s{shard}].wal`; frames are
written by the C++ WalWriter (crc32'd, torn-tail safe). Payloads carry a
parsed columnar write batch (see encode_batch) so replay does not re-parse
wire protocol.

Sharding (P5 write workers, round-1 VERDICT #8): with `shards > 1` each
region maps to one shard (own writer + own lock + own segment files), so
parallel ingest workers stop serializing on a single append lock — only
the global sequence counter stays shared (a two-instruction critical
section). Sequences are globally monotonic, so replay heap-merges the
per-shard streams by seq and all flushed_seq/purge logic is unchanged.
"""

from __future__ import annotations

import heapq
import json
import os
import re
import struct
import threading

import numpy as np

from greptimedb_amd import _native


def encode_batch(series: np.ndarray, ts_ms: np.ndarray, fields: np.ndarray,
                 field_names: list[str], new_series: list[tuple[int, bytes]],
                 str_cols: dict[str, list] | None = None) -> bytes:
    """[u32 hdr_len][hdr json][series i32][ts i64][fields f64 nf*n]
    [per str col: lengths i32[n] (-1=None) + utf8 blob]"""
    str_cols = str_cols or {}
    bin_cols = [name for name, vals in str_cols.items()
                if any(isinstance(v, (bytes, bytearray)) for v in vals)]
    hdr = json.dumps({
        "n": int(len(ts_ms)),
        "fields": field_names,
        "strs": list(str_cols),
        "bins": bin_cols,
        "new_series": [[c, pk.hex()] for c, pk in new_series],
    }).encode()
    parts = [struct.pack("<I", len(hdr)), hdr,
             np.ascontiguousarray(series, dtype=np.int32).tobytes(),
             np.ascontiguousarray(ts_ms, dtype=np.int64).tobytes(),
             np.ascontiguousarray(fields, dtype=np.float64).tobytes()]
    for name in str_cols:
        vals = str_cols[name]
        if not isinstance(vals, list):
            vals = list(vals)
        lens, blob = _native.pack_str_col(vals)
        parts.append(lens.tobytes())
        parts.append(blob)
    return b"".join(parts)


def decode_batch(buf: bytes):
    (hlen,) = struct.unpack_from("<I", buf, 0)
    hdr = json.loads(buf[4:4 + hlen].decode())
    n = hdr["n"]
    nf = len(hdr["fields"])
    off = 4 + hlen
    series = np.frombuffer(buf, dtype=np.int32, count=n, offset=off); off += 4 * n
    ts = np.frombuffer(buf, dtype=np.int64, count=n, offset=off); off += 8 * n
    fields = np.frombuffer(buf, dtype=np.float64, count=nf * n, offset=off).reshape(nf, n)
    off += 8 * nf * n
    str_cols = {}
    bins = set(hdr.get("bins", []))
    for name in hdr.get("strs", []):
        lens = np.frombuffer(buf, dtype=np.int32, count=n, offset=off); off += 4 * n
        vals = []
        raw = name in bins
        for ln in lens:
            if ln < 0:
                vals.append(None)
            else:
                b = buf[off:off + ln]
                vals.append(b if raw else b.decode())
                off += ln
        str_cols[name] = vals
    new_series = [(c, bytes.fromhex(h)) for c, h in hdr["new_series"]]
    return series, ts, fields, hdr["fields"], new_series, str_cols


_SEG_RE = re.compile(r"^(\d{20})(?:\.s(\d+))?\.wal$")


class _Shard:
    __slots__ = ("idx", "dir", "writer", "lock", "dirty")

    def __init__(self, idx: int, dir: str):
        self.idx = idx
        self.dir = dir
        self.writer = _native.WalWriter()
        self.lock = threading.Lock()
        self.dirty = False

    def seg_path(self, first_seq: int) -> str:
        if self.idx == 0:
            return os.path.join(self.dir, f"{first_seq:020d}.wal")
        return os.path.join(self.dir, f"{first_seq:020d}.s{self.idx}.wal")


class Wal:
    def __init__(self, dir: str, segment_bytes: int = 128 << 20,
                 sync_on_commit: bool = False, shards: int = 1):
        self.dir = dir
        os.makedirs(dir, exist_ok=True)
        self.segment_bytes = segment_bytes
        self.sync_on_commit = sync_on_commit
        self.next_seq = 1
        # last appended seq per region — updated atomically with the seq
        # assignment so purge decisions never race an in-flight append
        self.region_last: dict[int, int] = {}
        self._lock = threading.Lock()   # guards next_seq + region_last
        self.shards = [_Shard(i, dir) for i in range(max(shards, 1))]
        if self.segments():
            # resume: next_seq = last replayed seq + 1 (caller replays first)
            for _, rid, seq, _ in self.replay():
                self.next_seq = max(self.next_seq, seq + 1)
                self.region_last[rid] = seq
        for sh in self.shards:
            sh.writer.open_segment(sh.seg_path(self.next_seq))

    # ------------------------------------------------------------- layout
    def segments(self, shard: int | None = None) -> list[str]:
        out = []
        for f in os.listdir(self.dir):
            m = _SEG_RE.match(f)
            if not m:
                continue
            if shard is not None and int(m.group(2) or 0) != shard:
                continue
            out.append(f)
        return sorted(out)

    def _shard_of(self, region_id: int) -> _Shard:
        return self.shards[region_id % len(self.shards)]

    # ------------------------------------------------------------- append
    def append(self, region_id: int, payload: bytes) -> int:
        with self._lock:
            seq = self.next_seq
            self.next_seq += 1
            self.region_last[region_id] = seq
        sh = self._shard_of(region_id)
        with sh.lock:
            sh.writer.append(region_id, seq, payload)
            sh.dirty = True
        return seq

    def commit(self):
        for sh in self.shards:
            if not sh.dirty:
                continue
            with sh.lock:
                if not sh.dirty:
                    continue
                size = sh.writer.commit(self.sync_on_commit)
                sh.dirty = False
                if size >= self.segment_bytes:
                    sh.writer.close_segment()
                    sh.writer.open_segment(sh.seg_path(self.next_seq))

    # ------------------------------------------------------------- replay
    def replay(self):
        """Yield (seg_name, region_id, seq, payload) in GLOBAL seq order
        (heap-merge of the per-shard seq-ascending streams — a region's
        entries replay in order even if the shard count changed)."""
        def shard_stream(shard_idx):
            for seg in self.segments(shard_idx):
                for region, seq, payload in _native.wal_read_segment(
                        os.path.join(self.dir, seg)):
                    yield seq, seg, region, payload

        streams = [shard_stream(i) for i in range(len(self.shards))]
        # segments of shards beyond the current count (count was lowered):
        present = {int(_SEG_RE.match(f).group(2) or 0) for f in self.segments()}
        for extra in sorted(present - set(range(len(self.shards)))):
            streams.append(shard_stream(extra))
        for seq, seg, region, payload in heapq.merge(*streams):
            yield seg, region, seq, payload

    def purge_before(self, seq: int):
        """Delete whole segments whose every entry has seq < `seq` — per
        shard, a segment is obsolete when the shard's NEXT segment's first
        seq is <= `seq` (reference: WAL truncation after flush)."""
        for sh in range(len(self.shards)):
            segs = self.segments(sh)
            for i, seg in enumerate(segs[:-1]):
                nxt_first = int(_SEG_RE.match(segs[i + 1]).group(1))
                if nxt_first <= seq:
                    os.unlink(os.path.join(self.dir, seg))

    def close(self):
        for sh in self.shards:
            if sh.writer is None:
                continue
            try:
                sh.writer.commit(self.sync_on_commit)
            except RuntimeError:
                pass  # already closed
            sh.writer.close_segment()
            sh.writer = None
        self.shards = []
