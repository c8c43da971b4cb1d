"""Internal span tracing, exported as OTLP into the engine's own trace
table (self-hosted observability).

Reference parity: the reference exports its tokio-tracing spans via an
OTLP exporter (src/common/telemetry). Here the tracer buffers spans in
memory and `export_to(tracestore)` encodes a standard OTLP
ExportTraceServiceRequest protobuf that flows through the SAME native
parser + ingest path as external traces (csrc OtlpTraceParser) — the
database observes itself with its own wire format.

Usage:
    from greptimedb_amd.utils.tracing import tracer
    with tracer.span("sql.execute", statement=sql): ...
ADMIN enable_tracing(1) / ADMIN flush_tracing() control it via SQL.
"""

from __future__ import annotations

import os
import struct
import threading
import time
from contextlib import contextmanager

_SERVICE = "greptimedb_amd"
_MAX_BUFFER = 100_000


def _varint(v: int) -> bytes:
    out = b""
    while True:
        b7 = v & 0x7F
        v >>= 7
        out += bytes([b7 | (0x80 if v else 0)])
        if not v:
            return out


def _len_field(fnum: int, payload: bytes) -> bytes:
    return _varint((fnum << 3) | 2) + _varint(len(payload)) + payload


def _fixed64(fnum: int, v: int) -> bytes:
    return _varint((fnum << 3) | 1) + struct.pack("<Q", v)


class Span:
    __slots__ = ("trace_id", "span_id", "parent_id", "name", "start_ns",
                 "end_ns", "attrs", "status")

    def __init__(self, trace_id, span_id, parent_id, name, start_ns, attrs):
        self.trace_id = trace_id
        self.span_id = span_id
        self.parent_id = parent_id
        self.name = name
        self.start_ns = start_ns
        self.end_ns = 0
        self.attrs = attrs
        self.status = 0   # 0 unset, 1 ok, 2 error


class Tracer:
    def __init__(self):
        self.enabled = False
        self._buf: list[Span] = []
        self._lock = threading.Lock()
        self._local = threading.local()

    # ------------------------------------------------------------ record

    @contextmanager
    def span(self, name: str, **attrs):
        if not self.enabled:
            yield None
            return
        stack = getattr(self._local, "stack", None)
        if stack is None:
            stack = self._local.stack = []
        trace_id = stack[0].trace_id if stack else os.urandom(16)
        parent_id = stack[-1].span_id if stack else b""
        s = Span(trace_id, os.urandom(8), parent_id, name,
                 time.time_ns(), attrs)
        stack.append(s)
        try:
            yield s
            s.status = 1
        except BaseException:
            s.status = 2
            raise
        finally:
            s.end_ns = time.time_ns()
            stack.pop()
            with self._lock:
                if len(self._buf) < _MAX_BUFFER:
                    self._buf.append(s)

    # ------------------------------------------------------------ export

    def _encode_otlp(self, spans: list[Span]) -> bytes:
        def kv(key: str, val: str) -> bytes:
            return _len_field(1, _len_field(1, key.encode()) +
                              _len_field(2, _len_field(1, val.encode())))
        span_msgs = b""
        for s in spans:
            msg = (_len_field(1, s.trace_id) + _len_field(2, s.span_id) +
                   (_len_field(4, s.parent_id) if s.parent_id else b"") +
                   _len_field(5, s.name.encode()) +
                   _fixed64(7, s.start_ns) + _fixed64(8, s.end_ns))
            for k, v in s.attrs.items():
                msg += _len_field(9, _len_field(1, str(k).encode()) +
                                  _len_field(2, _len_field(1, str(v).encode())))
            if s.status:
                msg += _len_field(15, _varint((2 << 3) | 0) + _varint(s.status))
            span_msgs += _len_field(2, msg)
        resource = _len_field(1, kv("service.name", _SERVICE))
        scope_spans = _len_field(2, span_msgs)
        return _len_field(1, resource + scope_spans)

    def export_to(self, tracestore) -> int:
        """Drain the buffer into a TraceStore via OTLP bytes; returns spans."""
        with self._lock:
            spans, self._buf = self._buf, []
        if not spans:
            return 0
        return tracestore.write(self._encode_otlp(spans))

    def drop(self):
        with self._lock:
            self._buf.clear()


tracer = Tracer()
