"""Remote storage backends: S3 object store, Kafka-style remote WAL,
etcd-style metadata KV — all over HTTP clients.

Reference parity:
  * S3ObjectStore — src/object-store factory.rs S3 backend (OpenDAL).
    Minimal S3 REST surface: PUT/GET/DELETE object, GET ?list-type=2 with
    prefix (no auth signing in this offline environment — the seam is the
    wire protocol + retry semantics, exercised by tests/test_remote.py's
    in-process fake endpoint).
  * RemoteLogStore — src/log-store/src/kafka/log_store.rs:68 shape: many
    regions multiplexed onto topics, append returns per-topic offsets,
    replay reads a topic from an offset, purge truncates below an offset.
  * HttpKvBackend — src/common/meta/src/kv_backend (etcd.rs) shape:
    range/put/delete_range/CAS over a KV endpoint.

A single in-process `FakeRemoteServer` (tests/test_remote.py) serves all
three protocols, so every seam is exercised over a REAL network hop
without egress.
"""

from __future__ import annotations

import json
import time
import urllib.error
import urllib.parse
import urllib.request

from greptimedb_amd.engine.objstore import ObjectStore


def _http(method: str, url: str, body: bytes | None = None,
          attempts: int = 3, backoff_s: float = 0.05) -> bytes:
    last = None
    for i in range(attempts):
        try:
            req = urllib.request.Request(url, data=body, method=method)
            with urllib.request.urlopen(req, timeout=10) as resp:
                return resp.read()
        except urllib.error.HTTPError as e:
            if e.code == 404:
                raise FileNotFoundError(url) from None
            last = e
        except (urllib.error.URLError, ConnectionError, OSError) as e:
            last = e
        time.sleep(backoff_s * (2 ** i))
    raise last


class S3ObjectStore(ObjectStore):
    """Bucket-scoped object store over the S3 REST surface."""

    def __init__(self, endpoint: str, bucket: str):
        self.base = endpoint.rstrip("/") + "/" + bucket
        self.bucket = bucket

    def _url(self, key: str) -> str:
        return self.base + "/" + urllib.parse.quote(key)

    def put(self, key: str, data: bytes):
        _http("PUT", self._url(key), data)

    def get(self, key: str) -> bytes:
        return _http("GET", self._url(key))

    def delete(self, key: str):
        try:
            _http("DELETE", self._url(key))
        except FileNotFoundError:
            pass

    def list(self, prefix: str = "") -> list[str]:
        q = urllib.parse.urlencode({"list-type": "2", "prefix": prefix})
        body = _http("GET", self.base + "?" + q)
        return json.loads(body.decode())["keys"]

    def exists(self, key: str) -> bool:
        try:
            _http("GET", self._url(key))
            return True
        except FileNotFoundError:
            return False


class RemoteLogStore:
    """Kafka-remote-WAL shape: regions multiplexed onto topics."""

    def __init__(self, endpoint: str, topic: str):
        self.base = endpoint.rstrip("/") + "/wal/" + topic

    def append(self, region_id: int, seq: int, payload: bytes) -> int:
        """Produce one entry; returns the topic offset."""
        hdr = json.dumps({"region": region_id, "seq": seq}).encode()
        body = len(hdr).to_bytes(4, "little") + hdr + payload
        resp = _http("POST", self.base + "/append", body)
        return json.loads(resp.decode())["offset"]

    def replay(self, from_offset: int = 0):
        """Yield (offset, region_id, seq, payload) from `from_offset`."""
        body = _http("GET", f"{self.base}/read?from={from_offset}")
        off = 0
        while off < len(body):
            total = int.from_bytes(body[off:off + 4], "little")
            rec = body[off + 4: off + 4 + total]
            off += 4 + total
            hlen = int.from_bytes(rec[:4], "little")
            hdr = json.loads(rec[4:4 + hlen].decode())
            yield hdr["offset"], hdr["region"], hdr["seq"], rec[4 + hlen:]

    def purge_before(self, offset: int):
        _http("POST", f"{self.base}/purge?before={offset}")


class HttpKvBackend:
    """etcd-shaped metadata KV over HTTP (range/put/delete/CAS)."""

    def __init__(self, endpoint: str):
        self.base = endpoint.rstrip("/") + "/kv"

    def put(self, key: str, value: str):
        _http("PUT", self.base + "/" + urllib.parse.quote(key),
              value.encode())

    def get(self, key: str) -> str | None:
        try:
            return _http("GET", self.base + "/" +
                         urllib.parse.quote(key)).decode()
        except FileNotFoundError:
            return None

    def delete(self, key: str):
        try:
            _http("DELETE", self.base + "/" + urllib.parse.quote(key))
        except FileNotFoundError:
            pass

    def range(self, prefix: str) -> dict[str, str]:
        q = urllib.parse.urlencode({"prefix": prefix})
        body = _http("GET", self.base + "?" + q)
        return json.loads(body.decode())

    def compare_and_put(self, key: str, expect: str | None, value: str) -> bool:
        """etcd txn: put iff current == expect (None = must be absent)."""
        payload = json.dumps({"expect": expect, "value": value}).encode()
        resp = _http("POST", self.base + "/" + urllib.parse.quote(key) +
                     "?cas=1", payload)
        return json.loads(resp.decode())["ok"]
