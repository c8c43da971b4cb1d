"""Heartbeat + mailbox wire protocol (rank ⇄ metasrv).

Reference parity: the heartbeat bidi stream (src/datanode/src/heartbeat.rs
:58-205, src/meta-client/src/client.rs:734) carrying region stats UP and
mailbox instructions DOWN (src/common/meta/src/instruction.rs piggybacked
by src/meta-srv/src/handler/mailbox_handler.rs). MI355X shape: the metasrv
role is a thread in one of the rank processes (or its own process); the
transport reuses the loopback frame exchange (parallel/write_fanout.py) —
a request/ack per heartbeat, instructions returned in the ack like the
reference piggybacks them on the stream.

Frames are pickled dicts:
  up:   {"type": "heartbeat", "node": str, "regions": [...], "ts": ms}
  down: {"instructions": [{"kind": "downgrade_region"|"upgrade_region"|
                           "close_region", "table": str, "region_idx": int},
                          ...]}
"""

from __future__ import annotations

import pickle
import threading
import time

from greptimedb_amd.meta.supervisor import RegionSupervisor
from greptimedb_amd.parallel.write_fanout import WriteExchange


class MetaServer:
    """Metasrv side: accepts heartbeats, runs the φ supervisor, queues
    mailbox instructions per node."""

    def __init__(self, port: int, on_failover=None, threshold: float = 8.0,
                 acceptable_pause_ms: float = 10_000.0,
                 host: str = "127.0.0.1"):
        self.supervisor = RegionSupervisor(
            on_failover=on_failover, threshold=threshold,
            acceptable_pause_ms=acceptable_pause_ms)
        self.mailbox: dict[str, list] = {}
        self.stats: dict[str, dict] = {}
        self._lock = threading.Lock()
        # single-slot exchange: rank 0 = this server
        self.exchange = WriteExchange(0, 1, handler=self._handle,
                                      ports=[port], host=host)
        self.port = self.exchange.ports[0]

    def _handle(self, payload: bytes) -> bytes:
        msg = pickle.loads(payload)
        if msg.get("type") != "heartbeat":
            return pickle.dumps({"error": "unknown message"})
        node = msg["node"]
        with self._lock:
            self.supervisor.heartbeat(node, msg.get("ts"))
            self.stats[node] = {"regions": msg.get("regions", []),
                                "ts": msg.get("ts")}
            instr = self.mailbox.pop(node, [])
        return pickle.dumps({"instructions": instr})

    def send_instruction(self, node: str, instruction: dict):
        """Queue a mailbox instruction; delivered on the node's next
        heartbeat ack (reference mailbox piggyback)."""
        with self._lock:
            self.mailbox.setdefault(node, []).append(instruction)

    def check(self):
        return self.supervisor.check()

    def close(self):
        self.exchange.close()


class HeartbeatTask:
    """Datanode side: periodic heartbeat with region stats; applies
    returned mailbox instructions against the local engine."""

    def __init__(self, node_id: str, engine, server_port: int,
                 interval_s: float = 1.0, host: str = "127.0.0.1"):
        self.node_id = node_id
        self.engine = engine
        self.interval_s = interval_s
        self.applied: list = []
        # client-only exchange: world of 1 pointing at the server port
        self._ex = WriteExchange(0, 1, ports=[0], host=host)
        self._ex.ports = [server_port]
        self._ex._srv.close()          # no server side on the client
        self._stop = threading.Event()
        self._thread = None

    def beat_once(self) -> list:
        regions = [
            {"table": name, "region_id": r.region_id,
             "rows": r.num_rows, "writable": r.writable}
            for name, st in self.engine.tables.items()
            for r in st.regions
        ]
        payload = pickle.dumps({"type": "heartbeat", "node": self.node_id,
                                "regions": regions,
                                "ts": time.time() * 1000})
        resp = pickle.loads(self._ex.request(0, payload))
        instrs = resp.get("instructions", [])
        for ins in instrs:
            self._apply(ins)
        return instrs

    def _apply(self, ins: dict):
        kind = ins.get("kind")
        try:
            st = self.engine.table(ins["table"])
            region = st.regions[ins["region_idx"]]
        except Exception:
            return
        if kind == "downgrade_region":
            region.writable = False
        elif kind == "upgrade_region":
            region.writable = True
        elif kind == "close_region":
            region.writable = False
            with region.lock:
                region.sst_cache.clear()
        self.applied.append(ins)

    def start(self):
        def loop():
            while not self._stop.wait(self.interval_s):
                try:
                    self.beat_once()
                except Exception:
                    pass   # metasrv unreachable → φ rises server-side
        self._thread = threading.Thread(target=loop, daemon=True)
        self._thread.start()

    def stop(self):
        self._stop.set()
        if self._thread:
            self._thread.join(timeout=5)
        self._ex.close()
