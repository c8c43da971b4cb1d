"""WAL: segment management + batch payload codec over the native writer.

Reference parity: src/log-store raft_engine backend + mito2/src/wal.rs
(WalWriter group commit, per-region entries, replay from entry id,
obsolete/purge). Segments are `{first_seq:020d}.wal`; frames are written by
the C++ WalWriter (crc32'd, torn-tail safe). Payloads carry a parsed
columnar write batch (see encode_batch) so replay does not re-parse wire
protocol.
"""

from __future__ import annotations

import json
import os
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


class Wal:
    def __init__(self, dir: str, segment_bytes: int = 128 << 20, sync_on_commit: bool = False):
        self.dir = dir
        os.makedirs(dir, exist_ok=True)
        self.segment_bytes = segment_bytes
        self.sync_on_commit = sync_on_commit
        self.writer = _native.WalWriter()
        self.next_seq = 1
        # last appended seq per region — updated atomically with the seq
        # assignment so purge decisions never race an in-flight append
        self.region_last: dict[int, int] = {}
        self._lock = threading.Lock()  # multi-worker ingest (P5 write workers)
        segs = self.segments()
        if segs:
            # resume: next_seq = last replayed seq + 1 (caller replays first)
            for _, rid, seq, _ in self.replay():
                self.next_seq = max(self.next_seq, seq + 1)
                self.region_last[rid] = seq
            self._open_new_segment()
        else:
            self._open_new_segment()

    def _seg_path(self, first_seq: int) -> str:
        return os.path.join(self.dir, f"{first_seq:020d}.wal")

    def segments(self) -> list[str]:
        return sorted(f for f in os.listdir(self.dir) if f.endswith(".wal"))

    def _open_new_segment(self):
        self.writer.open_segment(self._seg_path(self.next_seq))

    def append(self, region_id: int, payload: bytes) -> int:
        with self._lock:
            seq = self.next_seq
            self.next_seq += 1
            self.writer.append(region_id, seq, payload)
            self.region_last[region_id] = seq
            return seq

    def commit(self):
        with self._lock:
            size = self.writer.commit(self.sync_on_commit)
            if size >= self.segment_bytes:
                self.writer.close_segment()
                self._open_new_segment()

    def replay(self):
        """Yield (seg_name, region_id, seq, payload) in order."""
        for seg in self.segments():
            for region, seq, payload in _native.wal_read_segment(os.path.join(self.dir, seg)):
                yield seg, region, seq, payload

    def purge_before(self, seq: int):
        """Delete whole segments whose every entry has seq < `seq`.
        A segment named by its first seq is obsolete when the NEXT segment's
        first seq is <= `seq` (reference: WAL truncation after flush)."""
        segs = self.segments()
        for i, seg in enumerate(segs[:-1]):
            nxt_first = int(segs[i + 1].split(".")[0])
            if nxt_first <= seq:
                os.unlink(os.path.join(self.dir, seg))

    def close(self):
        if self.writer is not None:
            try:
                self.writer.commit(self.sync_on_commit)
            except RuntimeError:
                pass  # already closed
            self.writer.close_segment()
            self.writer = None
