"""MySQL + PostgreSQL wire protocol servers, exercised over real sockets
with minimal spec-conformant clients."""

import asyncio
import struct

import pytest

from greptimedb_amd.query.executor import Executor
from greptimedb_amd.servers.mysql import MySQLServer
from greptimedb_amd.servers.postgres import PostgresServer


@pytest.fixture
def ex(tmp_engine):
    ex = Executor(tmp_engine)
    ex.execute("CREATE TABLE t (h STRING, ts TIMESTAMP TIME INDEX, v DOUBLE, PRIMARY KEY (h))")
    ex.execute("INSERT INTO t (h, ts, v) VALUES ('a', 1000, 1.5), ('b', 2000, 2.5)")
    return ex


# ------------------------------------------------------------------- mysql

async def _mysql_read_packet(reader):
    hdr = await reader.readexactly(4)
    ln = hdr[0] | (hdr[1] << 8) | (hdr[2] << 16)
    return hdr[3], await reader.readexactly(ln)


def _mysql_lenenc(buf, off):
    b = buf[off]
    if b < 251:
        return b, off + 1
    if b == 0xFC:
        return struct.unpack_from("<H", buf, off + 1)[0], off + 3
    if b == 0xFD:
        return int.from_bytes(buf[off + 1:off + 4], "little"), off + 4
    return struct.unpack_from("<Q", buf, off + 1)[0], off + 9


async def _mysql_session(port, queries):
    reader, writer = await asyncio.open_connection("127.0.0.1", port)
    seq, greeting = await _mysql_read_packet(reader)
    assert greeting[0] == 0x0A  # protocol v10
    # handshake response 41
    resp = (struct.pack("<IIB", 0x0200, 1 << 24, 33) + b"\x00" * 23 +
            b"root\x00" + b"\x00")
    ln = len(resp)
    writer.write(bytes([ln & 0xFF, (ln >> 8) & 0xFF, (ln >> 16) & 0xFF, seq + 1]) + resp)
    await writer.drain()
    _seq, ok = await _mysql_read_packet(reader)
    assert ok[0] == 0x00
    results = []
    for q in queries:
        payload = b"\x03" + q.encode()
        ln = len(payload)
        writer.write(bytes([ln & 0xFF, (ln >> 8) & 0xFF, (ln >> 16) & 0xFF, 0]) + payload)
        await writer.drain()
        _seq, first = await _mysql_read_packet(reader)
        if first[0] in (0x00, 0xFF):
            results.append(("status", first))
            continue
        ncols, _ = _mysql_lenenc(first, 0)
        for _ in range(ncols):
            await _mysql_read_packet(reader)
        _seq, eof = await _mysql_read_packet(reader)
        assert eof[0] == 0xFE
        rows = []
        while True:
            _seq, pkt = await _mysql_read_packet(reader)
            if pkt[0] == 0xFE and len(pkt) < 9:
                break
            row, off = [], 0
            for _ in range(ncols):
                if pkt[off] == 0xFB:
                    row.append(None)
                    off += 1
                else:
                    ln2, off = _mysql_lenenc(pkt, off)
                    row.append(pkt[off:off + ln2].decode())
                    off += ln2
            rows.append(row)
        results.append(("rows", rows))
    writer.write(b"\x01\x00\x00\x00\x01")  # COM_QUIT
    writer.close()
    return results


def test_mysql_protocol(ex):
    async def run():
        srv = MySQLServer(ex, host="127.0.0.1", port=0)
        s = await srv.start()
        port = s.sockets[0].getsockname()[1]
        out = await _mysql_session(port, [
            "SELECT h, ts, v FROM t ORDER BY ts",
            "SELECT count(*) FROM t",
            "SET NAMES utf8",
            "SELEKT broken",
        ])
        s.close()
        return out
    out = asyncio.run(run())
    assert out[0][0] == "rows"
    assert out[0][1] == [["a", "1970-01-01 00:00:01", "1.5"],
                         ["b", "1970-01-01 00:00:02", "2.5"]]
    assert out[1][1] == [["2"]]
    assert out[2][0] == "status" and out[2][1][0] == 0x00
    assert out[3][0] == "status" and out[3][1][0] == 0xFF  # error packet


# ------------------------------------------------------------------- postgres

async def _pg_session(port, queries):
    reader, writer = await asyncio.open_connection("127.0.0.1", port)
    params = b"user\x00tester\x00database\x00public\x00\x00"
    body = struct.pack("!I", 196608) + params
    writer.write(struct.pack("!I", len(body) + 4) + body)
    await writer.drain()
    # read until ReadyForQuery
    async def read_msg():
        tag = await reader.readexactly(1)
        (ln,) = struct.unpack("!I", await reader.readexactly(4))
        return tag, await reader.readexactly(ln - 4)
    while True:
        tag, body = await read_msg()
        if tag == b"Z":
            break
    results = []
    for q in queries:
        payload = q.encode() + b"\x00"
        writer.write(b"Q" + struct.pack("!I", len(payload) + 4) + payload)
        await writer.drain()
        rows, names, status = [], [], None
        while True:
            tag, body = await read_msg()
            if tag == b"T":
                (ncols,) = struct.unpack_from("!h", body, 0)
                off = 2
                for _ in range(ncols):
                    end = body.index(b"\x00", off)
                    names.append(body[off:end].decode())
                    off = end + 1 + 18
            elif tag == b"D":
                (ncols,) = struct.unpack_from("!h", body, 0)
                off = 2
                row = []
                for _ in range(ncols):
                    (ln2,) = struct.unpack_from("!i", body, off)
                    off += 4
                    if ln2 < 0:
                        row.append(None)
                    else:
                        row.append(body[off:off + ln2].decode())
                        off += ln2
                rows.append(row)
            elif tag == b"C":
                status = body.rstrip(b"\x00").decode()
            elif tag == b"E":
                status = "ERROR"
            elif tag == b"Z":
                break
        results.append((status, names, rows))
    writer.write(b"X" + struct.pack("!I", 4))
    writer.close()
    return results


def test_postgres_protocol(ex):
    async def run():
        srv = PostgresServer(ex, host="127.0.0.1", port=0)
        s = await srv.start()
        port = s.sockets[0].getsockname()[1]
        out = await _pg_session(port, [
            "SELECT h, v FROM t ORDER BY h",
            "SELECT count(*) FROM t",
            "SELEKT nope",
        ])
        s.close()
        return out
    out = asyncio.run(run())
    assert out[0][0] == "SELECT 2"
    assert out[0][1] == ["h", "v"]
    assert out[0][2] == [["a", "1.5"], ["b", "2.5"]]
    assert out[1][2] == [["2"]]
    assert out[2][0] == "ERROR"


async def _pg_extended_session(port, sql, params):
    """Parse/Bind/Describe/Execute/Sync round trip (text params)."""
    reader, writer = await asyncio.open_connection("127.0.0.1", port)
    body = struct.pack("!I", 196608) + b"user\x00tester\x00\x00"
    writer.write(struct.pack("!I", len(body) + 4) + body)
    await writer.drain()

    async def read_msg():
        tag = await reader.readexactly(1)
        (ln,) = struct.unpack("!I", await reader.readexactly(4))
        return tag, await reader.readexactly(ln - 4)

    while (await read_msg())[0] != b"Z":
        pass

    def msg(tag, payload):
        return tag + struct.pack("!I", len(payload) + 4) + payload

    p = b"\x00" + sql.encode() + b"\x00" + struct.pack("!h", len(params))
    p += struct.pack(f"!{len(params)}I", *([0] * len(params)))
    writer.write(msg(b"P", p))
    b = b"\x00\x00" + struct.pack("!h", 0) + struct.pack("!h", len(params))
    for v in params:
        ev = str(v).encode()
        b += struct.pack("!i", len(ev)) + ev
    b += struct.pack("!h", 0)
    writer.write(msg(b"B", b))
    writer.write(msg(b"D", b"P\x00"))
    writer.write(msg(b"E", b"\x00" + struct.pack("!i", 0)))
    writer.write(msg(b"S", b""))
    await writer.drain()
    names, rows, seen = [], [], []
    while True:
        tag, body = await read_msg()
        seen.append(tag)
        if tag == b"T":
            (ncols,) = struct.unpack_from("!h", body, 0)
            off = 2
            for _ in range(ncols):
                end = body.index(b"\x00", off)
                names.append(body[off:end].decode())
                off = end + 1 + 18
        elif tag == b"D":
            (ncols,) = struct.unpack_from("!h", body, 0)
            off = 2
            row = []
            for _ in range(ncols):
                (ln2,) = struct.unpack_from("!i", body, off)
                off += 4
                if ln2 < 0:
                    row.append(None)
                else:
                    row.append(body[off:off + ln2].decode())
                    off += ln2
            rows.append(row)
        elif tag == b"Z":
            break
    writer.write(b"X" + struct.pack("!I", 4))
    writer.close()
    return seen, names, rows


def test_postgres_extended_protocol(ex):
    async def run():
        srv = PostgresServer(ex, host="127.0.0.1", port=0)
        s = await srv.start()
        port = s.sockets[0].getsockname()[1]
        out = await _pg_extended_session(
            port, "SELECT h, v FROM t WHERE v > $1 AND h != $2 ORDER BY h",
            [1.0, "zzz"])
        s.close()
        return out
    seen, names, rows = asyncio.run(run())
    assert b"1" in seen and b"2" in seen and b"C" in seen  # Parse/Bind complete
    assert names == ["h", "v"]
    assert rows == [["a", "1.5"], ["b", "2.5"]]


async def _mysql_prepared_session(port, sql, params):
    """COM_STMT_PREPARE + COM_STMT_EXECUTE with typed binary params."""
    reader, writer = await asyncio.open_connection("127.0.0.1", port)
    seq, greeting = await _mysql_read_packet(reader)
    resp = (struct.pack("<IIB", 0x0200, 1 << 24, 33) + b"\x00" * 23 +
            b"tester\x00" + b"\x00")
    ln = len(resp)
    writer.write(bytes([ln & 0xFF, (ln >> 8) & 0xFF, (ln >> 16) & 0xFF,
                        seq + 1]) + resp)
    await writer.drain()
    _s, ok = await _mysql_read_packet(reader)
    assert ok[0] == 0x00

    def send(payload):
        writer.write(struct.pack("<I", len(payload))[:3] + b"\x00" + payload)

    send(b"\x16" + sql.encode())
    await writer.drain()
    seq, first = await _mysql_read_packet(reader)
    assert first[0] == 0x00
    (sid,) = struct.unpack_from("<I", first, 1)
    (ncols, nparams) = struct.unpack_from("<HH", first, 5)
    for _ in range(nparams + (1 if nparams else 0)):    # param defs + eof
        await _mysql_read_packet(reader)
    # execute: null bitmap + bound flag + types + values
    body = b"\x17" + struct.pack("<IBI", sid, 0, 1)
    bm = bytearray((len(params) + 7) // 8)
    types = b""
    vals = b""
    for i, p in enumerate(params):
        if p is None:
            bm[i // 8] |= 1 << (i % 8)
            types += bytes([6, 0])
        elif isinstance(p, int):
            types += bytes([8, 0]); vals += struct.pack("<q", p)
        elif isinstance(p, float):
            types += bytes([5, 0]); vals += struct.pack("<d", p)
        else:
            b = str(p).encode()
            types += bytes([0xFD, 0]); vals += bytes([len(b)]) + b
    body += bytes(bm) + b"\x01" + types + vals
    send(body)
    await writer.drain()
    # read binary resultset
    seq, head = await _mysql_read_packet(reader)
    ncols = head[0]
    names = []
    for _ in range(ncols):
        _s, col = await _mysql_read_packet(reader)
        # parse 5th lenenc string (name)
        off = 0
        for k in range(5):
            ln = col[off]; off += 1
            if k == 4:
                names.append(col[off:off + ln].decode())
            off += ln
    await _mysql_read_packet(reader)                    # eof
    rows = []
    while True:
        _s, pkt = await _mysql_read_packet(reader)
        if pkt[0] == 0xFE and len(pkt) < 9:
            break
        nb_len = (ncols + 2 + 7) // 8
        bm2 = pkt[1:1 + nb_len]
        off = 1 + nb_len
        row = []
        for i in range(ncols):
            if bm2[(i + 2) // 8] & (1 << ((i + 2) % 8)):
                row.append(None)
                continue
            ln = pkt[off]; off += 1
            row.append(pkt[off:off + ln].decode()); off += ln
        rows.append(row)
    writer.write(b"\x01\x00\x00\x00\x01")
    writer.close()
    return names, rows


def test_mysql_prepared_statements(ex):
    async def run():
        from greptimedb_amd.servers.mysql import MySQLServer
        srv = MySQLServer(ex, host="127.0.0.1", port=0)
        s = await srv.start()
        port = s.sockets[0].getsockname()[1]
        out = await _mysql_prepared_session(
            port, "SELECT h, v FROM t WHERE v > ? AND h != ? ORDER BY h",
            [1.0, "zzz"])
        s.close()
        return out
    names, rows = asyncio.run(run())
    assert names == ["h", "v"]
    assert rows == [["a", "1.5"], ["b", "2.5"]]


def test_postgres_scram_auth(ex):
    """SCRAM-SHA-256 SASL exchange (RFC 7677; reference pgwire SCRAM)."""
    import struct as _st
    from greptimedb_amd.servers.auth import (StaticUserProvider,
                                             scram_client_messages)

    async def run(password):
        provider = StaticUserProvider({"alice": "s3cret"})
        srv = PostgresServer(ex, host="127.0.0.1", port=0,
                             user_provider=provider)
        s = await srv.start()
        port = s.sockets[0].getsockname()[1]
        reader, writer = await asyncio.open_connection("127.0.0.1", port)
        body = _st.pack("!I", 196608) + b"user\x00alice\x00\x00"
        writer.write(_st.pack("!I", len(body) + 4) + body)
        await writer.drain()

        async def read_msg():
            tag = await reader.readexactly(1)
            (ln,) = _st.unpack("!I", await reader.readexactly(4))
            return tag, await reader.readexactly(ln - 4)

        tag, payload = await read_msg()
        assert tag == b"R" and _st.unpack("!I", payload[:4])[0] == 10
        assert b"SCRAM-SHA-256" in payload

        sfirst_holder = {}

        async def do_exchange():
            def server_first_fn(client_first):
                return sfirst_holder["v"]
            # client-first
            import base64, hashlib, hmac as _hmac, os as _os
            cnonce = base64.b64encode(_os.urandom(18)).decode()
            bare = f"n=alice,r={cnonce}"
            cfirst = ("n,," + bare).encode()
            writer.write(b"p" + _st.pack(
                "!I", 4 + len(b"SCRAM-SHA-256\x00") + 4 + len(cfirst)) +
                b"SCRAM-SHA-256\x00" + _st.pack("!i", len(cfirst)) + cfirst)
            await writer.drain()
            t2, pl2 = await read_msg()
            assert t2 == b"R" and _st.unpack("!I", pl2[:4])[0] == 11
            sfirst = pl2[4:].decode()
            attrs = dict(kv.split("=", 1) for kv in sfirst.split(","))
            salt = base64.b64decode(attrs["s"])
            salted = hashlib.pbkdf2_hmac("sha256", password.encode(), salt,
                                         int(attrs["i"]))
            ckey = _hmac.new(salted, b"Client Key", hashlib.sha256).digest()
            skey = hashlib.sha256(ckey).digest()
            wo_proof = f"c=biws,r={attrs['r']}"
            auth_msg = ",".join([bare, sfirst, wo_proof]).encode()
            csig = _hmac.new(skey, auth_msg, hashlib.sha256).digest()
            proof = base64.b64encode(bytes(a ^ b for a, b in
                                           zip(ckey, csig))).decode()
            cfinal = f"{wo_proof},p={proof}".encode()
            writer.write(b"p" + _st.pack("!I", 4 + len(cfinal)) + cfinal)
            await writer.drain()
            return await read_msg()

        t3, pl3 = await do_exchange()
        if password == "s3cret":
            assert t3 == b"R" and _st.unpack("!I", pl3[:4])[0] == 12
            assert pl3[4:].startswith(b"v=")
            t4, pl4 = await read_msg()
            assert t4 == b"R" and _st.unpack("!I", pl4[:4])[0] == 0
        else:
            assert t3 == b"E"
        writer.close()
        s.close()
        return True

    assert asyncio.run(run("s3cret"))
    assert asyncio.run(run("wrongpw"))


def test_mysql_caching_sha2_auth(ex):
    """caching_sha2_password fast-auth (MySQL 8 default plugin)."""
    import hashlib
    import struct as _st

    from greptimedb_amd.servers.auth import StaticUserProvider
    from greptimedb_amd.servers.mysql import MySQLServer

    async def run(password):
        provider = StaticUserProvider({"bob": "pw123"})
        srv = MySQLServer(ex, "127.0.0.1", 0, provider)
        s = await srv.start()
        port = s.sockets[0].getsockname()[1]
        reader, writer = await asyncio.open_connection("127.0.0.1", port)
        _seq, greeting = await _mysql_read_packet(reader)
        nonce = b"12345678123456789012"
        p1 = hashlib.sha256(password.encode()).digest()
        p2 = hashlib.sha256(p1).digest()
        token = bytes(a ^ b for a, b in zip(
            p1, hashlib.sha256(p2 + nonce).digest()))
        caps = 0x00000200 | 0x00080000 | 0x00000008  # proto41|plugin_auth|connect_with_db
        body = (_st.pack("<I", caps) + _st.pack("<I", 1 << 24) + bytes([33]) +
                b"\x00" * 23 + b"bob\x00" + bytes([len(token)]) + token +
                b"testdb\x00" + b"caching_sha2_password\x00")
        writer.write(_st.pack("<I", len(body) | (1 << 24))[:3] + bytes([1]) + body)
        await writer.drain()
        _seq2, more = await _mysql_read_packet(reader)
        ok_pkt = None
        if more[:1] == b"\x01":       # fast-auth success marker
            _seq3, ok_pkt = await _mysql_read_packet(reader)
        else:
            ok_pkt = more
        writer.close()
        s.close()
        return ok_pkt[:1]

    assert asyncio.run(run("pw123")) == b"\x00"       # OK
    assert asyncio.run(run("wrong")) == b"\xff"       # ERR
