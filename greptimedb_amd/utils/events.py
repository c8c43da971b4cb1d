"""Event recorder: persist cluster events into a system table.

Reference parity: src/common/event-recorder (recorder.rs:311) — DDL,
migration and maintenance events land in a queryable system events table.
Here the table is `greptime_events` (type tag + ts + json payload string),
written through the normal engine write path so it flushes/compacts/
replays like any other table.
"""

from __future__ import annotations

import json
import time

import numpy as np

EVENTS_TABLE = "greptime_events"


class EventRecorder:
    def __init__(self, engine):
        self.engine = engine
        self._recording = False   # guard against self-recursion

    def _table(self):
        from greptimedb_amd.models.schema import (ColumnSchema, DataType,
                                                  SemanticType, TableSchema)
        st = self.engine.tables.get(EVENTS_TABLE)
        if st is not None:
            return st
        schema = TableSchema(name=EVENTS_TABLE, columns=[
            ColumnSchema("event_type", DataType.STRING, SemanticType.TAG, 0),
            ColumnSchema("ts", DataType.TIMESTAMP_MS, SemanticType.TIMESTAMP, 1),
            ColumnSchema("payload", DataType.STRING, SemanticType.FIELD, 2,
                         fulltext=False),
        ], primary_key=["event_type"])
        return self.engine.create_table(schema, n_regions=1, append_mode=True,
                                        if_not_exists=True)

    def record(self, event_type: str, payload: dict | None = None):
        if self._recording:
            return
        self._recording = True
        try:
            st = self._table()
            region_idx = 0
            code = st.regions[0].register_series((event_type,))
            now = int(time.time() * 1000)
            self.engine.write_region(
                st, region_idx, np.array([code], dtype=np.int32),
                np.array([now], dtype=np.int64),
                np.zeros((len(st.regions[0].field_names), 1)), [],
                str_fields={"payload": [json.dumps(payload or {})]})
            self.engine.commit_wal()
        except Exception:
            pass  # event recording must never fail the triggering operation
        finally:
            self._recording = False


def recorder_of(engine) -> EventRecorder:
    r = getattr(engine, "event_recorder", None)
    if r is None:
        r = engine.event_recorder = EventRecorder(engine)
    return r
