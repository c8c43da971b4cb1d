"""Timestamp parsing/formatting (UTC, epoch milliseconds)."""

from __future__ import annotations

import datetime as _dt
import re

_TS_RE = re.compile(
    r"^(\d{4})-(\d{2})-(\d{2})(?:[ T](\d{2}):(\d{2})(?::(\d{2})(?:\.(\d+))?)?)?(Z|[+-]\d{2}:?\d{2})?$"
)


def parse_ts_ms(s: str) -> int | None:
    """'2016-01-01 00:00:00[.123][Z]' → epoch ms (UTC). None if not a timestamp."""
    m = _TS_RE.match(s.strip())
    if not m:
        return None
    y, mo, d = int(m.group(1)), int(m.group(2)), int(m.group(3))
    hh = int(m.group(4) or 0)
    mm = int(m.group(5) or 0)
    ss = int(m.group(6) or 0)
    frac = m.group(7) or ""
    ms = int((frac + "000")[:3]) if frac else 0
    tz = m.group(8)
    dt = _dt.datetime(y, mo, d, hh, mm, ss, tzinfo=_dt.timezone.utc)
    epoch = int(dt.timestamp() * 1000) + ms
    if tz and tz != "Z":
        sign = 1 if tz[0] == "+" else -1
        t = tz[1:].replace(":", "")
        off = int(t[:2]) * 60 + int(t[2:4] or 0)
        epoch -= sign * off * 60_000
    return epoch


def format_ts_ms(ms: int) -> str:
    dt = _dt.datetime.fromtimestamp(ms / 1000, tz=_dt.timezone.utc)
    if ms % 1000:
        return dt.strftime("%Y-%m-%dT%H:%M:%S.") + f"{ms % 1000:03d}"
    return dt.strftime("%Y-%m-%dT%H:%M:%S")


_TRUNC_MS = {
    "millisecond": 1, "second": 1000, "minute": 60_000, "hour": 3_600_000,
    "day": 86_400_000, "week": 604_800_000,
}


def trunc_unit_ms(unit: str) -> int | None:
    return _TRUNC_MS.get(unit.lower())
