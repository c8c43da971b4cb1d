"""Prometheus remote read (reference: src/servers prom_store read path).

Control-plane sized: request protobuf decoded in Python, series resolved
through the PromQL matcher machinery, raw samples gathered from device
columns, response framed as snappy(literal)-compressed ReadResponse.
"""

from __future__ import annotations

import struct

import numpy as np
import torch

from greptimedb_amd.query.promql import ast as past


def _varint_read(buf, off):
    v = 0
    shift = 0
    while True:
        b = buf[off]
        off += 1
        v |= (b & 0x7F) << shift
        if not (b & 0x80):
            return v, off
        shift += 7


def _varint(v: int) -> bytes:
    out = b""
    while True:
        b7 = v & 0x7F
        v >>= 7
        out += bytes([b7 | (0x80 if v else 0)])
        if not v:
            return out


def _ld(f, payload: bytes) -> bytes:
    return _varint((f << 3) | 2) + _varint(len(payload)) + payload


def snappy_compress_literal(data: bytes) -> bytes:
    """Valid (all-literal) snappy block framing."""
    out = [_varint(len(data))]
    i = 0
    while i < len(data):
        chunk = data[i:i + (1 << 16)]
        n = len(chunk) - 1
        if n < 60:
            out.append(bytes([n << 2]))
        elif n < (1 << 8):
            out.append(bytes([60 << 2, n]))
        elif n < (1 << 16):
            out.append(bytes([61 << 2]) + struct.pack("<H", n))
        i += len(chunk)
        out.append(chunk)
    return b"".join(out)


_MATCH_OPS = {0: "=", 1: "!=", 2: "=~", 3: "!~"}


def parse_read_request(body: bytes) -> list[dict]:
    """→ [{start_ms, end_ms, matchers: [Matcher]}]"""
    queries = []
    off = 0
    while off < len(body):
        key, off = _varint_read(body, off)
        if (key >> 3) == 1 and (key & 7) == 2:
            ln, off = _varint_read(body, off)
            q = {"start_ms": 0, "end_ms": 0, "matchers": []}
            qend = off + ln
            while off < qend:
                k2, off = _varint_read(body, off)
                f, wt = k2 >> 3, k2 & 7
                if f == 1 and wt == 0:
                    q["start_ms"], off = _varint_read(body, off)
                elif f == 2 and wt == 0:
                    q["end_ms"], off = _varint_read(body, off)
                elif f == 3 and wt == 2:
                    ml, off = _varint_read(body, off)
                    mend = off + ml
                    typ, name, value = 0, "", ""
                    while off < mend:
                        k3, off = _varint_read(body, off)
                        f3, wt3 = k3 >> 3, k3 & 7
                        if f3 == 1 and wt3 == 0:
                            typ, off = _varint_read(body, off)
                        elif f3 == 2 and wt3 == 2:
                            sl, off = _varint_read(body, off)
                            name = body[off:off + sl].decode()
                            off += sl
                        elif f3 == 3 and wt3 == 2:
                            sl, off = _varint_read(body, off)
                            value = body[off:off + sl].decode()
                            off += sl
                        else:
                            off = mend
                    q["matchers"].append(past.Matcher(name, _MATCH_OPS.get(typ, "="),
                                                      value))
                else:
                    if wt == 0:
                        _, off = _varint_read(body, off)
                    elif wt == 2:
                        sl, off = _varint_read(body, off)
                        off += sl
                    else:
                        off = qend
            queries.append(q)
        else:
            break
    return queries


def execute_read(prom_eval, queries: list[dict]) -> bytes:
    """Evaluate queries → snappy-compressed ReadResponse protobuf."""
    results = []
    for q in queries:
        sel = past.Selector(None, q["matchers"])
        st, field = prom_eval._resolve_table(sel)
        series_msgs = []
        if st is not None:
            for region in st.regions:
                codes = prom_eval._match_codes(region, sel)
                it = range(len(region.series)) if codes is None else codes
                it = list(it)
                if not it:
                    continue
                lut = np.full(len(region.series), -1, dtype=np.int32)
                lut[np.asarray(it, dtype=np.int64)] = np.arange(len(it), dtype=np.int32)
                device = prom_eval.engine.config.device
                lut_t = torch.as_tensor(lut, device=device)
                per_series: dict[int, list] = {}
                for src in region.scan_sources(q["start_ms"], q["end_ms"] + 1):
                    p = src.field_pos.get(field)
                    if p is None:
                        continue
                    from greptimedb_amd.ops import filter_series_time
                    mask = filter_series_time(src.ts, src.series, lut_t,
                                              q["start_ms"], q["end_ms"] + 1)
                    idx = mask.nonzero(as_tuple=True)[0]
                    if idx.numel() == 0:
                        continue
                    se = lut_t[src.series[idx].long()].cpu().numpy()
                    ts = src.ts[idx].cpu().numpy()
                    vals = src.fields[p][idx].cpu().numpy()
                    for s, t, v in zip(se, ts, vals):
                        per_series.setdefault(int(s), []).append((int(t), float(v)))
                for sl, samples in per_series.items():
                    code = it[sl]
                    labels = region.series.labels_of(code)
                    labels.setdefault("__name__", st.schema.name)
                    body = b""
                    for n2, v2 in sorted(labels.items()):
                        body += _ld(1, _ld(1, n2.encode()) + _ld(2, str(v2).encode()))
                    for t, v in sorted(samples):
                        body += _ld(2, _varint((1 << 3) | 1) +
                                    struct.pack("<d", v) +
                                    _varint(2 << 3) + _varint(t))
                    series_msgs.append(_ld(1, body))
        results.append(_ld(1, b"".join(series_msgs)))
    return snappy_compress_literal(b"".join(results))
