"""Repartition: change a table's region count, redistributing resident data.

Reference parity: docs/rfcs/2025-06-20-repartition.md + mito2 remap_manifest
(the reference remaps manifests and shares SST files across regions; here a
region's data IS device tensors, so repartition gathers each old region's
rows by their new pk-hash assignment, writes them as fresh sorted SSTs in
staged region directories, then atomically swaps directories and bumps the
table epoch so ingest routers re-resolve series)."""

from __future__ import annotations

import os
import shutil

import numpy as np
import torch

from greptimedb_amd.engine import sst as sst_mod
from greptimedb_amd.engine.region import Region
from greptimedb_amd.engine.series import tsid_hash
from greptimedb_amd.models.schema import region_id as make_region_id


def repartition_table(engine, name: str, n_new: int) -> int:
    st = engine.table(name)
    if n_new == len(st.regions):
        return 0
    schema = st.schema
    device = engine.config.device
    for r in st.regions:
        r.flush()
    staging_base = os.path.join(engine.config.data_dir, "region_staging")
    shutil.rmtree(staging_base, ignore_errors=True)
    new_regions = []
    for rn in range(n_new):
        rid = make_region_id(schema.table_id, rn)
        new_regions.append(Region(rid, schema, os.path.join(staging_base, str(rid)),
                                  device=device, append_mode=st.append_mode))
    moved = 0
    for old in st.regions:
        ncodes = len(old.series)
        if ncodes == 0:
            continue
        tgt_region = np.empty(ncodes, dtype=np.int32)
        tgt_code = np.empty(ncodes, dtype=np.int32)
        for code in range(ncodes):
            pk = old.series.pks[code]
            ri = tsid_hash(pk) % n_new
            tgt_region[code] = ri
            tgt_code[code] = new_regions[ri].series.add_encoded(pk)
        # persist new series logs in bulk
        for nr in new_regions:
            import struct as _struct
            buf = bytearray()
            for pk in nr.series.pks[getattr(nr, "_logged", 0):]:
                buf += _struct.pack("<I", len(pk)) + pk
            if buf:
                nr._series_log.write(bytes(buf))
                nr._series_log.flush()
            nr._logged = len(nr.series.pks)
        t_region = torch.as_tensor(tgt_region, device=device)
        t_code = torch.as_tensor(tgt_code, device=device)
        for batch in old.sst_cache.values():
            se_l = batch.series.long()
            for ri, nr in enumerate(new_regions):
                mask = t_region[se_l] == ri
                idx = mask.nonzero(as_tuple=True)[0]
                if idx.numel() == 0:
                    continue
                ts_t = batch.ts[idx].contiguous()
                se_t = t_code[se_l[idx]].int().contiguous()
                f_t = batch.fields[:, idx].contiguous()
                idx_h = idx.cpu().numpy()
                strs = {sn: np.asarray(col, dtype=object)[idx_h]
                        for sn, col in getattr(batch, "str_cols", {}).items()}
                # sort by (series, ts) for the new region
                o = torch.argsort(ts_t, stable=True)
                perm = o[torch.argsort(se_t[o], stable=True)]
                ts_t, se_t, f_t = ts_t[perm], se_t[perm], f_t[:, perm]
                perm_h = perm.cpu().numpy()
                strs = {sn: a[perm_h] for sn, a in strs.items()}
                fid = sst_mod.new_file_id()
                path = os.path.join(nr.dir, "sst", f"{fid}.parquet")
                meta = sst_mod.write_sst(
                    path, schema, nr.series.pks, se_t.cpu().numpy(),
                    ts_t.cpu().numpy(), f_t.cpu().numpy(),
                    np.arange(ts_t.numel(), dtype=np.int64),
                    batch.field_names, str_cols=strs)
                nr.manifest.commit({"kind": "edit",
                                    "files_to_add": [meta.to_dict()],
                                    "files_to_remove": []})
                nb = sst_mod.SstBatch(ts_t.contiguous(), se_t.contiguous(),
                                      f_t.contiguous(), None, meta.min_ts,
                                      meta.max_ts, batch.field_names)
                nb.str_cols = strs
                for sn, arr in strs.items():
                    ft = nr.text_cols.get(sn)
                    if ft is not None:
                        nb.text_index[sn] = ft.build_segment(list(arr), device)
                nr.sst_cache[fid] = nb
                moved += int(idx.numel())
    # swap directories: old region dirs out, staged dirs in
    for old in st.regions:
        old._series_log.close()
        shutil.rmtree(old.dir, ignore_errors=True)
    for nr in new_regions:
        final_dir = os.path.join(engine.config.data_dir, "region",
                                 str(nr.region_id))
        shutil.rmtree(final_dir, ignore_errors=True)
        nr._series_log.close()
        shutil.move(nr.dir, final_dir)
    # reopen from final locations (fresh fds, validated manifests)
    st.regions = [
        Region(make_region_id(schema.table_id, rn), schema,
               os.path.join(engine.config.data_dir, "region",
                            str(make_region_id(schema.table_id, rn))),
               device=device, append_mode=st.append_mode)
        for rn in range(n_new)
    ]
    # all pre-repartition data is durable in the new SSTs: advance
    # flushed_seq past every existing WAL entry so replay never re-applies
    # old-layout mutations to the new regions
    seq_now = engine.wal.next_seq - 1
    for nr in st.regions:
        nr.manifest.commit({"kind": "edit", "files_to_add": [],
                            "files_to_remove": [], "flushed_seq": seq_now})
        nr.flushed_seq = seq_now
        nr.last_seq = max(nr.last_seq, seq_now)
    engine._purge_wal()
    engine.routing_epoch = getattr(engine, "routing_epoch", 0) + 1
    engine._save_catalog()
    return moved
