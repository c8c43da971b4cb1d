"""Internal metrics registry exposed at /metrics (reference parity:
per-crate metrics.rs lazy_static Prometheus registries + servers http.rs
/metrics route)."""

from __future__ import annotations

import threading

_lock = threading.Lock()
_counters: dict[str, float] = {}
_gauges: dict[str, float] = {}


class _Counter:
    def __init__(self, name):
        self.name = name

    def inc(self, v: float = 1.0):
        with _lock:
            _counters[self.name] = _counters.get(self.name, 0.0) + v


def counter(name: str) -> _Counter:
    return _Counter(name)


def set_gauge(name: str, v: float):
    with _lock:
        _gauges[name] = v


def render_prometheus() -> str:
    lines = []
    with _lock:
        for k, v in sorted(_counters.items()):
            lines.append(f"# TYPE greptime_{k} counter")
            lines.append(f"greptime_{k} {v}")
        for k, v in sorted(_gauges.items()):
            lines.append(f"# TYPE greptime_{k} gauge")
            lines.append(f"greptime_{k} {v}")
    return "\n".join(lines) + "\n"
