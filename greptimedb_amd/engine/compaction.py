"""Compaction: TWCS-style time-window SST merging (K18).

Reference parity: src/mito2/src/compaction/twcs.rs — SSTs are bucketed by
time window; when a window accumulates >= trigger_file_num files they are
merged into one (two levels: L0 → L1 = LEVEL_COMPACTED, max 32 inputs).
MI355X redesign: inputs are already device-resident sorted batches, so the
merge is a device concat + sort + LastRow dedup (K3+K4 composed) and one
parquet rewrite — no read-back from storage.
"""

from __future__ import annotations

import os

import numpy as np
import torch

from greptimedb_amd.engine import sst as sst_mod
from greptimedb_amd.ops import dedup_mark_last

TRIGGER_FILE_NUM = 4
MAX_INPUTS = 32


def pick_window_s(span_s: float) -> int:
    """TWCS window sizing (reference compaction/twcs.rs time window infer):
    bucket the region's time span into hour-scale windows."""
    for w in (3600, 2 * 3600, 12 * 3600, 24 * 3600, 7 * 24 * 3600):
        if span_s <= w * 8:
            return w
    return 14 * 24 * 3600


class Compactor:
    def __init__(self, trigger_file_num: int = TRIGGER_FILE_NUM):
        self.trigger = trigger_file_num

    def pick(self, region) -> list[list[str]]:
        """Group L0 file ids by time window; windows with >= trigger files
        are compaction candidates."""
        files = region.manifest.files
        if len(files) < self.trigger:
            return []
        tr = region.time_range()
        if tr is None:
            return []
        window_ms = pick_window_s((tr[1] - tr[0]) / 1000 + 1) * 1000
        buckets: dict[int, list[str]] = {}
        for fid, meta in files.items():
            if meta.get("level", 0) >= 1:
                continue
            if fid not in region.sst_cache:
                continue
            w = meta["min_ts"] // window_ms
            buckets.setdefault(w, []).append(fid)
        return [fids[:MAX_INPUTS] for fids in buckets.values()
                if len(fids) >= self.trigger]

    def compact_region(self, region) -> int:
        """Run all picked merges; returns number of merges performed."""
        done = 0
        for fids in self.pick(region):
            self._merge(region, fids)
            done += 1
        return done

    def _merge(self, region, fids: list[str]):
        batches = [region.sst_cache[f] for f in fids]
        device = region.device
        ts = torch.cat([b.ts for b in batches])
        se = torch.cat([b.series for b in batches])
        # per-row sequences order rows across overlapping files; batches
        # without one (pre-upgrade fixtures) get file-order synthetic seqs
        seq_parts, synth_base = [], 0
        for b in batches:
            if b.seq is not None:
                seq_parts.append(b.seq.to(torch.int64))
                synth_base = max(synth_base, int(b.seq.max().item()) + 1 if b.n else 0)
            else:
                seq_parts.append(torch.arange(synth_base, synth_base + b.n,
                                              dtype=torch.int64, device=device))
                synth_base += b.n
        seq = torch.cat(seq_parts)
        # unify field layout
        fnames = []
        for b in batches:
            for fn in b.field_names:
                if fn not in fnames:
                    fnames.append(fn)
        n = ts.numel()
        fields = torch.full((len(fnames), n), float("nan"), dtype=torch.float64,
                            device=device)
        off = 0
        str_parts: dict[str, list] = {}
        for b in batches:
            for i, fn in enumerate(b.field_names):
                fields[fnames.index(fn), off:off + b.n] = b.fields[i][: b.n]
            for sn in getattr(b, "str_cols", {}):
                str_parts.setdefault(sn, [])
            off += b.n
        # string columns (aligned, None-padded)
        off = 0
        for b in batches:
            for sn in str_parts:
                col = getattr(b, "str_cols", {}).get(sn)
                if col is None:
                    str_parts[sn].append(np.full(b.n, None, dtype=object))
                else:
                    str_parts[sn].append(np.asarray(col, dtype=object))
            off += b.n
        # sort by (series, ts, seq): stable sorts applied innermost-first so
        # the newest (highest-seq) row of each (series, ts) group lands last
        o0 = torch.argsort(seq, stable=True)
        o1 = o0[torch.argsort(ts[o0], stable=True)]
        perm = o1[torch.argsort(se[o1], stable=True)]
        ts, se, seq = ts[perm], se[perm], seq[perm]
        fields = fields[:, perm]
        if not region.append_mode:
            keep = dedup_mark_last(se.contiguous(), ts.contiguous())
            kidx = keep.nonzero(as_tuple=True)[0]
            ts, se, seq, fields = ts[kidx], se[kidx], seq[kidx], fields[:, kidx]
            perm = perm[kidx]
        perm_h = perm.cpu().numpy()
        str_cols_sorted = {sn: np.concatenate(parts)[perm_h]
                           for sn, parts in str_parts.items()}

        ts_h = ts.cpu().numpy()
        se_h = se.cpu().numpy()
        f_h = fields.cpu().numpy()
        seq_h = seq.cpu().numpy()
        fid = sst_mod.new_file_id()
        path = os.path.join(region.dir, "sst", f"{fid}.parquet")
        meta = sst_mod.write_sst(path, region.schema, region.series.pks,
                                 se_h, ts_h, f_h, seq_h, fnames,
                                 str_cols=str_cols_sorted,
                                 region_id=region.region_id)
        meta.level = 1  # LEVEL_COMPACTED
        region.manifest.commit({
            "kind": "edit",
            "files_to_add": [meta.to_dict()],
            "files_to_remove": list(fids),
        })
        new_batch = sst_mod.SstBatch(ts.contiguous(), se.contiguous(),
                                     fields.contiguous(), seq.contiguous(),
                                     meta.min_ts, meta.max_ts, fnames)
        new_batch.str_cols = str_cols_sorted
        for sn, arr in str_cols_sorted.items():
            ft = region.text_cols.get(sn)
            if ft is not None:
                new_batch.text_index[sn] = ft.build_segment(list(arr), device)
        if new_batch.text_index:
            from greptimedb_amd.engine import ftindex
            ftindex.save_sidecar(path, new_batch.text_index, region.text_cols)
        with region.lock:
            for f in fids:
                region.sst_cache.pop(f, None)
            region.sst_cache[fid] = new_batch
        # remove merged files from disk (reference: file purger)
        for f in fids:
            for suffix in (".parquet", ".ftidx"):
                p = os.path.join(region.dir, "sst", f"{f}{suffix}")
                if os.path.exists(p):
                    os.unlink(p)
