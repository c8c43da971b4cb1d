"""Cross-rank write fan-out: any rank accepts any write.

Reference parity: src/operator/src/insert.rs:389-496 — the Inserter
partition-splits every row batch and fans the slices out to the owning
datanodes over gRPC (`group_requests_by_peer` → per-peer
RegionServer.handle). MI355X redesign: one process per GPU on one node, so
the peer hop is a loopback TCP exchange between ranks (the data starts on
the host — lines arrive over the wire — so xGMI/RCCL buys nothing for the
row shipping itself; device work starts at the owner's memtable append,
like the reference where WAL+memtable live on the owning datanode).

Ownership: a series' global partition index p (PartitionRule) maps to rank
`p % world` for multi-dim rules; hash tables use `tsid_hash(pk) % world`.
The receiving rank re-resolves its local region with its own engine's rule,
WAL-commits, then acks — the sender's ingest call returns only after every
peer ack (reference: Inserter joins all per-peer futures).

Frames are length-prefixed pickles of numpy arrays (loopback-only internal
transport, never exposed on a public port; the external write surface
remains HTTP/MySQL/PG/gRPC).
"""

from __future__ import annotations

import pickle
import socket
import struct
import threading
import time


def fanout_port(rank: int, base: int | None = None) -> int:
    """Deterministic per-rank loopback port. Base derives from the torch
    rendezvous port so concurrent jobs on one box don't collide."""
    import os
    if base is None:
        base = int(os.environ.get("GDB_FANOUT_BASE",
                                  int(os.environ.get("MASTER_PORT", "29400")) + 500))
    return base + rank


def _read_exact(sock: socket.socket, n: int) -> bytes:
    buf = bytearray()
    while len(buf) < n:
        chunk = sock.recv(n - len(buf))
        if not chunk:
            raise ConnectionError("peer closed")
        buf += chunk
    return bytes(buf)


def _read_frame(sock: socket.socket) -> bytes:
    (ln,) = struct.unpack("<I", _read_exact(sock, 4))
    return _read_exact(sock, ln)


def _write_frame(sock: socket.socket, payload: bytes):
    sock.sendall(struct.pack("<I", len(payload)) + payload)


class WriteExchange:
    """Per-rank loopback exchange: a listener thread accepting peer write
    batches + pooled client connections to every peer."""

    def __init__(self, rank: int, world: int, handler=None,
                 base_port: int | None = None, host: str = "127.0.0.1",
                 ports: list[int] | None = None, handler_factory=None,
                 peers: list[tuple[str, int]] | None = None):
        """`peers` maps rank → (host, port) for multi-node layouts; the
        GDB_FANOUT_PEERS env ("h1:p1,h2:p2,…", one entry per rank) sets it
        for torchrun-launched jobs. Default: loopback, one port per rank
        (single-node — xGMI box — layout)."""
        import os as _os
        self.rank = rank
        self.world = world
        if peers is None and _os.environ.get("GDB_FANOUT_PEERS"):
            peers = []
            for ent in _os.environ["GDB_FANOUT_PEERS"].split(","):
                h, _, p = ent.strip().rpartition(":")
                peers.append((h, int(p)))
        if peers is not None:
            if len(peers) != world:
                raise ValueError(f"peers needs {world} entries")
            self.peers = list(peers)
        else:
            base_ports = list(ports) if ports is not None else \
                [fanout_port(r, base_port) for r in range(world)]
            self.peers = [(host, p) for p in base_ports]
        self.host = self.peers[rank][0]
        self.ports = [p for _h, p in self.peers]
        self.handler = handler      # callable(payload: bytes) -> bytes
        # handler_factory() -> fresh handler per connection: each peer
        # connection gets its own receive pipeline (no shared lock), like
        # the reference's per-connection tonic service instances
        self.handler_factory = handler_factory
        # free-connection pool per peer: a conn is checked OUT for the
        # whole request/ack round trip, so concurrent callers (the 6 TSBS
        # ingest workers) each get their own socket — and the peer gets
        # one handler thread (and receive pipeline) per socket. A single
        # shared conn would serialize every remote apply per sender/peer
        # pair, collapsing whole-node throughput.
        self._pool: dict[int, list[socket.socket]] = {r: [] for r in range(world)}
        self._pool_lock = threading.Lock()
        self._closing = False
        self._srv = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
        self._srv.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        own_host = self.peers[rank][0]
        bind_host = own_host if own_host.startswith("127.") else "0.0.0.0"
        self._srv.bind((bind_host, self.ports[rank]))
        self._srv.listen(64)
        self._accept_thread = threading.Thread(target=self._accept_loop,
                                               daemon=True)
        self._accept_thread.start()

    # ------------------------------------------------------------- server
    def _accept_loop(self):
        while not self._closing:
            try:
                conn, _ = self._srv.accept()
            except OSError:
                return
            conn.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
            threading.Thread(target=self._serve_conn, args=(conn,),
                             daemon=True).start()

    def _serve_conn(self, conn: socket.socket):
        handler = self.handler_factory() if self.handler_factory else self.handler
        try:
            while not self._closing:
                req = _read_frame(conn)
                try:
                    resp = handler(req) if handler else b"OK"
                except Exception as e:  # report the error to the sender
                    resp = b"ERR " + repr(e).encode()
                _write_frame(conn, resp or b"OK")
        except (ConnectionError, OSError):
            pass
        finally:
            conn.close()

    # ------------------------------------------------------------- client
    def _connect(self, peer: int) -> socket.socket:
        deadline = time.monotonic() + 30
        last = None
        while time.monotonic() < deadline:
            try:
                # host from the peer map, port from self.ports — callers
                # (meta heartbeat) patch .ports after construction
                s = socket.create_connection(
                    (self.peers[peer][0], self.ports[peer]), timeout=30)
                s.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
                return s
            except OSError as e:
                last = e
                time.sleep(0.1)
        raise ConnectionError(f"cannot reach rank {peer}: {last}")

    def request(self, peer: int, payload: bytes) -> bytes:
        """Send one frame to `peer`, wait for its ack frame. Concurrent
        callers use distinct pooled connections (no lock across the RTT)."""
        with self._pool_lock:
            sock = self._pool[peer].pop() if self._pool[peer] else None
        if sock is None:
            sock = self._connect(peer)
        try:
            try:
                _write_frame(sock, payload)
                resp = _read_frame(sock)
            except (ConnectionError, OSError):
                # one reconnect attempt (peer restarted)
                sock.close()
                sock = self._connect(peer)
                _write_frame(sock, payload)
                resp = _read_frame(sock)
        except BaseException:
            try:
                sock.close()
            except OSError:
                pass
            raise
        with self._pool_lock:
            if self._closing:
                sock.close()
            else:
                self._pool[peer].append(sock)
        if resp.startswith(b"ERR"):
            raise RuntimeError(f"rank {peer} write failed: {resp[4:].decode()}")
        return resp

    def close(self):
        self._closing = True
        try:
            self._srv.close()
        except OSError:
            pass
        with self._pool_lock:
            for conns in self._pool.values():
                for s in conns:
                    try:
                        s.close()
                    except OSError:
                        pass
                conns.clear()


# ------------------------------------------------------------------ codec

def encode_routed_batch(tagsets: list, srow, ts_ms, fields_mat,
                        field_names: list[str], str_cols=None) -> bytes:
    """(unique tagset keys, row→tagset idx, ts, fields[nf,n], names)."""
    return pickle.dumps(
        ("write", tagsets, srow, ts_ms, fields_mat, field_names,
         str_cols or {}),
        protocol=pickle.HIGHEST_PROTOCOL)


def decode_routed_batch(payload: bytes):
    kind, *rest = pickle.loads(payload)
    assert kind == "write", kind
    return rest
