"""Arrow Flight service: do_get (SQL query) + do_put (bulk ingest).

Reference parity: src/servers/src/grpc/flight.rs — FlightCraft do_get
(:67-86) streams query results as Arrow record batches; do_put (:240-330)
streams RecordBatch bulk ingestion (PutRecordBatchRequest) into the bulk
memtable path. MI355X design: the real Arrow Flight wire protocol (gRPC +
Arrow IPC via pyarrow.flight, interoperable with any Flight client);
queries run on the GPU engine, do_put feeds engine/bulk.py's columnar
path. Tickets are JSON: {"sql": "..."}; put descriptors name the target
table (path[0] or a JSON command).
"""

from __future__ import annotations

import json
import threading

import numpy as np
import pyarrow as pa
import pyarrow.flight as flight


def query_result_to_table(res) -> pa.Table:
    """QueryResult (columnar: names/columns/kinds) → Arrow table."""
    arrays = []
    for name, col, kind in zip(res.names, res.columns, res.kinds):
        if kind == "ts":
            a = pa.array(np.asarray(col, dtype=np.int64), type=pa.int64()) \
                .cast(pa.timestamp("ms"))
        else:
            vals = list(col)
            if vals and all(isinstance(v, (bytes, bytearray)) for v in vals
                            if v is not None):
                a = pa.array(vals, type=pa.binary())
            else:
                try:
                    a = pa.array(np.asarray(col))
                except (pa.ArrowInvalid, pa.ArrowTypeError, ValueError):
                    a = pa.array([None if v is None else str(v) for v in vals],
                                 type=pa.string())
        arrays.append(a)
    return pa.Table.from_arrays(arrays, names=list(res.names))


class GreptimeFlightServer(flight.FlightServerBase):
    def __init__(self, engine, executor=None, host: str = "127.0.0.1",
                 port: int = 0):
        location = flight.Location.for_grpc_tcp(host, port)
        super().__init__(location)
        self.engine = engine
        if executor is None:
            from greptimedb_amd.query.executor import Executor
            executor = Executor(engine)
        self.executor = executor
        self._lock = threading.Lock()

    # ------------------------------------------------------------- query
    def do_get(self, context, ticket):
        req = json.loads(ticket.ticket.decode())
        sql = req.get("sql")
        if not sql:
            raise flight.FlightServerError("ticket must carry {'sql': ...}")
        res = self.executor.execute(sql)
        table = query_result_to_table(res)
        return flight.RecordBatchStream(table)

    def get_flight_info(self, context, descriptor):
        if descriptor.descriptor_type == flight.DescriptorType.CMD:
            ticket = flight.Ticket(descriptor.command)
            req = json.loads(descriptor.command.decode())
            res = self.executor.execute(req["sql"])
            table = query_result_to_table(res)
            endpoints = [flight.FlightEndpoint(ticket, [])]
            return flight.FlightInfo(table.schema, descriptor, endpoints,
                                     table.num_rows, -1)
        raise flight.FlightServerError("only CMD descriptors supported")

    def list_flights(self, context, criteria):
        for name, st in self.engine.tables.items():
            desc = flight.FlightDescriptor.for_path(name)
            info = flight.FlightInfo(
                pa.schema([]), desc, [],
                sum(r.num_rows for r in st.regions), -1)
            yield info

    # ------------------------------------------------------------- ingest
    def do_put(self, context, descriptor, reader, writer):
        """Bulk ingest: descriptor path[0] = table name (or CMD JSON
        {"table": ..., "append_mode": bool}); each incoming RecordBatch
        goes through the columnar bulk path (reference BulkInserts)."""
        from greptimedb_amd.engine.bulk import bulk_insert_arrow
        append_mode = True
        if descriptor.descriptor_type == flight.DescriptorType.PATH:
            table = descriptor.path[0].decode()
        else:
            cmd = json.loads(descriptor.command.decode())
            table = cmd["table"]
            append_mode = bool(cmd.get("append_mode", True))
        total = 0
        for chunk in reader:
            batch = chunk.data
            if batch is None or batch.num_rows == 0:
                continue
            total += bulk_insert_arrow(self.engine, table, batch,
                                       append_mode=append_mode)
        writer.write(json.dumps({"affected_rows": total}).encode())

    # ------------------------------------------------------------- actions
    def do_action(self, context, action):
        if action.type == "flush":
            self.engine.flush_all()
            yield flight.Result(b"ok")
        elif action.type == "compact":
            self.engine.compact_all()
            yield flight.Result(b"ok")
        else:
            raise flight.FlightServerError(f"unknown action {action.type}")

    def list_actions(self, context):
        return [("flush", "flush all regions"), ("compact", "compact all regions")]


class FlightClient:
    """Thin convenience client (tests, tools): sql() and put()."""

    def __init__(self, host: str, port: int):
        self.conn = flight.connect(f"grpc://{host}:{port}")

    def sql(self, q: str) -> pa.Table:
        ticket = flight.Ticket(json.dumps({"sql": q}).encode())
        return self.conn.do_get(ticket).read_all()

    def put(self, table: str, data: pa.Table | pa.RecordBatch) -> int:
        if isinstance(data, pa.RecordBatch):
            data = pa.Table.from_batches([data])
        desc = flight.FlightDescriptor.for_path(table)
        writer, meta_reader = self.conn.do_put(desc, data.schema)
        writer.write_table(data)
        writer.done_writing()
        buf = meta_reader.read()
        writer.close()
        if buf is None:
            return -1
        return json.loads(buf.to_pybytes().decode())["affected_rows"]

    def close(self):
        self.conn.close()
