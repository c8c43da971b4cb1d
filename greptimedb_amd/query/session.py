"""Session / QueryContext: current schema, timezone, session variables.

Reference parity: src/session (QueryContext {current_catalog,
current_schema, timezone, channel}) + the USE / SET statements. MI355X
model keeps one flat engine namespace; non-public schemas prefix table
names (`<schema>.<table>`), so every protocol server can hold its own
Session while sharing the engine.
"""

from __future__ import annotations

import re

DEFAULT_CATALOG = "greptime"
DEFAULT_SCHEMA = "public"
SYSTEM_SCHEMAS = {"public", "greptime_private", "information_schema"}


def parse_tz_offset_ms(tz: str) -> int:
    """'+08:00' / '-05:30' / 'UTC' / 'SYSTEM' → offset in ms."""
    tz = tz.strip().strip("'\"")
    if tz.upper() in ("UTC", "SYSTEM", "Z", ""):
        return 0
    m = re.fullmatch(r"([+-])(\d{1,2}):(\d{2})", tz)
    if not m:
        raise ValueError(f"bad timezone {tz!r}")
    sign = 1 if m.group(1) == "+" else -1
    return sign * (int(m.group(2)) * 3600 + int(m.group(3)) * 60) * 1000


class Session:
    def __init__(self, catalog: str = DEFAULT_CATALOG,
                 schema: str = DEFAULT_SCHEMA, timezone: str = "UTC"):
        self.catalog = catalog
        self.schema = schema
        self.timezone = timezone
        self.tz_offset_ms = 0
        self.vars: dict[str, str] = {}
        self.cursors: dict = {}   # name -> (QueryResult, position)

    def set_var(self, name: str, value: str):
        name = name.lower().lstrip("@")
        self.vars[name] = value
        if name in ("time_zone", "timezone", "session.time_zone"):
            self.tz_offset_ms = parse_tz_offset_ms(str(value))
            self.timezone = str(value).strip("'\"")

    def resolve_table(self, name: str) -> str:
        """Schema-qualify a table name into the engine's flat namespace."""
        if "." in name:
            schema, table = name.split(".", 1)
            if schema in (DEFAULT_SCHEMA, self.catalog):
                return table
            return name  # '<schema>.<table>' is the flat key
        if self.schema != DEFAULT_SCHEMA:
            return f"{self.schema}.{name}"
        return name
