"""MySQL wire-protocol server (L7).

Reference parity: src/servers/src/mysql/ (opensrv-mysql based). Implements
protocol v10: handshake, CLIENT_PROTOCOL_41 text protocol, COM_QUERY /
COM_PING / COM_QUIT / COM_INIT_DB, lenenc text resultsets with EOF framing
(CLIENT_DEPRECATE_EOF intentionally not advertised). Auth accepts any
credentials unless a UserProvider is configured (reference auth seam).
"""

from __future__ import annotations

import asyncio
import math
import struct

import numpy as np

from greptimedb_amd.utils.errors import GreptimeError

CLIENT_PROTOCOL_41 = 0x00000200
CLIENT_CONNECT_WITH_DB = 0x00000008
CLIENT_PLUGIN_AUTH = 0x00080000

MYSQL_TYPE_DOUBLE = 5
MYSQL_TYPE_LONGLONG = 8
MYSQL_TYPE_DATETIME = 12
MYSQL_TYPE_VAR_STRING = 253


def lenenc_int(n: int) -> bytes:
    if n < 251:
        return bytes([n])
    if n < (1 << 16):
        return b"\xfc" + struct.pack("<H", n)
    if n < (1 << 24):
        return b"\xfd" + struct.pack("<I", n)[:3]
    return b"\xfe" + struct.pack("<Q", n)


def lenenc_str(s: bytes) -> bytes:
    return lenenc_int(len(s)) + s


class MySQLServer:
    def __init__(self, executor, host="0.0.0.0", port=4002, user_provider=None):
        self.executor = executor
        self.host = host
        self.port = port
        self.user_provider = user_provider
        self._server = None

    async def start(self):
        self._server = await asyncio.start_server(self._handle, self.host, self.port)
        return self._server

    async def serve_forever(self):
        await self.start()
        async with self._server:
            await self._server.serve_forever()

    # ------------------------------------------------------- packet IO

    @staticmethod
    async def _read_packet(reader) -> tuple[int, bytes]:
        hdr = await reader.readexactly(4)
        ln = hdr[0] | (hdr[1] << 8) | (hdr[2] << 16)
        seq = hdr[3]
        return seq, await reader.readexactly(ln)

    @staticmethod
    def _packet(seq: int, payload: bytes) -> bytes:
        ln = len(payload)
        return bytes([ln & 0xFF, (ln >> 8) & 0xFF, (ln >> 16) & 0xFF, seq]) + payload

    def _ok(self, seq, affected=0) -> bytes:
        return self._packet(seq, b"\x00" + lenenc_int(affected) + lenenc_int(0) +
                            struct.pack("<HH", 0x0002, 0))

    def _eof(self, seq) -> bytes:
        return self._packet(seq, b"\xfe" + struct.pack("<HH", 0, 0x0002))

    def _err(self, seq, msg: str, code=1064) -> bytes:
        return self._packet(seq, b"\xff" + struct.pack("<H", code) + b"#42000" +
                            msg.encode()[:400])

    # ------------------------------------------------------- session

    async def _handle(self, reader, writer):
        try:
            # handshake v10
            greeting = (b"\x0a" + b"greptimedb-amd-mysql\x00" +
                        struct.pack("<I", 1) +          # thread id
                        b"12345678\x00" +               # auth-plugin-data part1
                        struct.pack("<H", (CLIENT_PROTOCOL_41 | CLIENT_CONNECT_WITH_DB |
                                           CLIENT_PLUGIN_AUTH) & 0xFFFF) +
                        bytes([33]) +                   # charset utf8
                        struct.pack("<H", 0x0002) +     # status
                        struct.pack("<H", ((CLIENT_PROTOCOL_41 | CLIENT_PLUGIN_AUTH)
                                           >> 16) & 0xFFFF) +
                        bytes([21]) + b"\x00" * 10 +
                        b"123456789012\x00" +
                        b"mysql_native_password\x00")
            writer.write(self._packet(0, greeting))
            await writer.drain()
            seq, resp = await self._read_packet(reader)
            if len(resp) >= 32:
                user_end = resp.find(b"\x00", 32)
                user = resp[32:user_end].decode(errors="replace") if user_end > 0 else ""
            else:
                user = ""
            if self.user_provider is not None:
                from greptimedb_amd.servers.auth import (
                    mysql_caching_sha2_check, mysql_native_check, password_of)
                ok = self.user_provider.allow(user)
                stored = password_of(self.user_provider, user)
                if ok and stored:
                    # auth token + (optional) client auth plugin name
                    token = b""
                    plugin = b"mysql_native_password"
                    if user_end > 0 and user_end + 1 < len(resp):
                        tl = resp[user_end + 1]
                        token = resp[user_end + 2: user_end + 2 + tl]
                        rest = resp[user_end + 2 + tl:]
                        # optional database name, then plugin name (both NUL)
                        caps = struct.unpack("<I", resp[:4])[0]
                        if caps & CLIENT_CONNECT_WITH_DB and b"\x00" in rest:
                            rest = rest.split(b"\x00", 1)[1]
                        if rest:
                            plugin = rest.split(b"\x00", 1)[0] or plugin
                    nonce = b"12345678123456789012"
                    if plugin == b"caching_sha2_password":
                        ok = mysql_caching_sha2_check(stored, nonce, token)
                    else:
                        ok = mysql_native_check(stored, nonce, token)
                if not ok:
                    writer.write(self._err(seq + 1, f"access denied for {user}", 1045))
                    await writer.drain()
                    writer.close()
                    return
            if self.user_provider is not None and 'plugin' in dir() and \
                    plugin == b"caching_sha2_password":
                # fast-auth success marker precedes OK (MySQL 8 protocol)
                writer.write(self._packet(seq + 1, b"\x01\x03"))
                writer.write(self._ok(seq + 2))
            else:
                writer.write(self._ok(seq + 1))
            await writer.drain()

            stmts: dict = {}     # stmt_id → {"sql", "nparams", "types"}
            next_stmt = [1]
            while True:
                try:
                    seq, cmd = await self._read_packet(reader)
                except (asyncio.IncompleteReadError, ConnectionResetError):
                    break
                if not cmd:
                    break
                op = cmd[0]
                if op == 0x01:       # COM_QUIT
                    break
                if op in (0x0E, 0x02):   # COM_PING / COM_INIT_DB
                    writer.write(self._ok(1))
                elif op == 0x03:     # COM_QUERY
                    sql = cmd[1:].decode(errors="replace")
                    writer.write(self._run_query(sql))
                elif op == 0x16:     # COM_STMT_PREPARE
                    writer.write(self._stmt_prepare(cmd[1:], stmts, next_stmt))
                elif op == 0x17:     # COM_STMT_EXECUTE
                    writer.write(self._stmt_execute(cmd[1:], stmts))
                elif op in (0x19, 0x1A):  # COM_STMT_CLOSE / COM_STMT_RESET
                    (sid,) = struct.unpack_from("<I", cmd, 1)
                    if op == 0x19:
                        stmts.pop(sid, None)   # CLOSE: no response
                    else:
                        stmts.get(sid, {}).pop("types", None)
                        writer.write(self._ok(1))
                else:
                    writer.write(self._err(1, f"unsupported command {op:#x}", 1047))
                await writer.drain()
        except (asyncio.IncompleteReadError, ConnectionResetError):
            pass
        finally:
            try:
                writer.close()
            except Exception:
                pass

    # -------------------------------------------------- prepared statements

    def _stmt_prepare(self, body: bytes, stmts: dict, next_stmt: list) -> bytes:
        """COM_STMT_PREPARE: '?' placeholders, params typed at execute.
        num_columns reported 0 — the execute response carries the real
        resultset metadata (clients re-read it; Connector/J & friends do)."""
        sql = body.decode(errors="replace")
        nparams = sql.count("?")
        sid = next_stmt[0]
        next_stmt[0] += 1
        stmts[sid] = {"sql": sql, "nparams": nparams}
        out = [self._packet(1, b"\x00" + struct.pack("<I", sid) +
                            struct.pack("<HH", 0, nparams) + b"\x00" +
                            struct.pack("<H", 0))]
        seq = 2
        for i in range(nparams):
            nb = f"?{i}".encode()
            col = (lenenc_str(b"def") + lenenc_str(b"") + lenenc_str(b"") +
                   lenenc_str(b"") + lenenc_str(nb) + lenenc_str(nb) +
                   b"\x0c" + struct.pack("<H", 33) + struct.pack("<I", 1024) +
                   bytes([MYSQL_TYPE_VAR_STRING]) + struct.pack("<H", 0) +
                   bytes([0]) + b"\x00\x00")
            out.append(self._packet(seq, col))
            seq += 1
        if nparams:
            out.append(self._eof(seq))
        return b"".join(out)

    @staticmethod
    def _decode_params(body: bytes, off: int, n: int, st: dict):
        null_bm = body[off: off + (n + 7) // 8]
        off += (n + 7) // 8
        bound = body[off]
        off += 1
        if bound:
            st["types"] = [struct.unpack_from("<BB", body, off + 2 * i)[0]
                           for i in range(n)]
            off += 2 * n
        types = st.get("types") or [MYSQL_TYPE_VAR_STRING] * n
        vals = []
        for i in range(n):
            if null_bm[i // 8] & (1 << (i % 8)):
                vals.append(None)
                continue
            t = types[i]
            if t in (1,):                         # TINY
                vals.append(struct.unpack_from("<b", body, off)[0]); off += 1
            elif t == 2:                          # SHORT
                vals.append(struct.unpack_from("<h", body, off)[0]); off += 2
            elif t == 3:                          # LONG
                vals.append(struct.unpack_from("<i", body, off)[0]); off += 4
            elif t == 8:                          # LONGLONG
                vals.append(struct.unpack_from("<q", body, off)[0]); off += 8
            elif t == 4:                          # FLOAT
                vals.append(struct.unpack_from("<f", body, off)[0]); off += 4
            elif t == 5:                          # DOUBLE
                vals.append(struct.unpack_from("<d", body, off)[0]); off += 8
            elif t == 6:                          # NULL
                vals.append(None)
            else:                                 # lenenc string-ish
                ln = body[off]; off += 1
                if ln == 0xFC:
                    (ln,) = struct.unpack_from("<H", body, off); off += 2
                elif ln == 0xFD:
                    ln = int.from_bytes(body[off:off + 3], "little"); off += 3
                elif ln == 0xFE:
                    (ln,) = struct.unpack_from("<Q", body, off); off += 8
                vals.append(body[off:off + ln].decode(errors="replace"))
                off += ln
        return vals

    def _stmt_execute(self, body: bytes, stmts: dict) -> bytes:
        (sid,) = struct.unpack_from("<I", body, 0)
        st = stmts.get(sid)
        if st is None:
            return self._err(1, f"unknown statement {sid}", 1243)
        off = 9                                   # id(4) + flags(1) + iter(4)
        try:
            vals = self._decode_params(body, off, st["nparams"], st) \
                if st["nparams"] else []
        except (IndexError, struct.error):
            return self._err(1, "malformed COM_STMT_EXECUTE", 1210)
        sql = ""
        it = iter(vals)
        for part in st["sql"].split("?"):
            sql += part
            try:
                v = next(it)
            except StopIteration:
                continue
            if v is None:
                sql += "NULL"
            elif isinstance(v, (int, float)):
                sql += repr(v)
            else:
                sql += "'" + str(v).replace("'", "''") + "'"
        return self._run_query(sql, binary=True)

    def _run_query(self, sql: str, binary: bool = False) -> bytes:
        s = sql.strip().rstrip(";").lower()
        # common client handshake queries
        if s.startswith(("set ", "set@", "use ")) or s in ("commit", "rollback", "begin"):
            return self._ok(1)
        if s.startswith("select @@") or s == "select version()":
            return self._text_resultset(["version()"], [["8.4.0-greptimedb-amd"]],
                                        [MYSQL_TYPE_VAR_STRING])
        if s in ("select database()", "select schema()"):
            return self._text_resultset(["database()"], [["public"]],
                                        [MYSQL_TYPE_VAR_STRING])
        if s.startswith("show variables") or s.startswith("show session variables"):
            return self._text_resultset(
                ["Variable_name", "Value"],
                [["version", "8.4.0-greptimedb-amd"],
                 ["character_set_client", "utf8mb4"],
                 ["max_allowed_packet", "16777216"]],
                [MYSQL_TYPE_VAR_STRING, MYSQL_TYPE_VAR_STRING])
        if s.startswith("show collation") or s.startswith("show character set"):
            return self._text_resultset(["Charset"], [["utf8mb4"]],
                                        [MYSQL_TYPE_VAR_STRING])
        try:
            r = self.executor.execute(sql)
        except GreptimeError as e:
            return self._err(1, str(e) or type(e).__name__)
        except Exception as e:  # pragma: no cover
            return self._err(1, f"{type(e).__name__}: {e}")
        types = []
        for col, kind in zip(r.columns, r.kinds):
            if kind == "ts":
                types.append(MYSQL_TYPE_DATETIME)
            elif len(col) and isinstance(col[0], (int, np.integer)):
                types.append(MYSQL_TYPE_LONGLONG)
            elif len(col) and isinstance(col[0], (float, np.floating)):
                types.append(MYSQL_TYPE_DOUBLE)
            else:
                types.append(MYSQL_TYPE_VAR_STRING)
        rows = []
        from greptimedb_amd.utils.timeutil import format_ts_ms
        for row in r.rows():
            out = []
            for v, kind in zip(row, r.kinds):
                if v is None or (isinstance(v, float) and math.isnan(v)):
                    out.append(None)
                elif kind == "ts":
                    out.append(format_ts_ms(int(v)).replace("T", " "))
                else:
                    out.append(str(v))
            rows.append(out)
        if binary:
            return self._binary_resultset(r.names, rows)
        return self._text_resultset(r.names, rows, types)

    def _binary_resultset(self, names, rows) -> bytes:
        """Binary-protocol resultset (COM_STMT_EXECUTE response). Every
        column is declared VAR_STRING, so values are lenenc strings — the
        client coerces (this is what stringified text mode looks like)."""
        out = [self._packet(1, lenenc_int(len(names)))]
        seq = 2
        for name in names:
            nb = name.encode()
            col = (lenenc_str(b"def") + lenenc_str(b"") + lenenc_str(b"") +
                   lenenc_str(b"") + lenenc_str(nb) + lenenc_str(nb) +
                   b"\x0c" + struct.pack("<H", 33) + struct.pack("<I", 1024) +
                   bytes([MYSQL_TYPE_VAR_STRING]) + struct.pack("<H", 0) +
                   bytes([0]) + b"\x00\x00")
            out.append(self._packet(seq, col))
            seq += 1
        out.append(self._eof(seq)); seq += 1
        nb_len = (len(names) + 2 + 7) // 8
        for row in rows:
            bm = bytearray(nb_len)
            payload = b""
            for i, v in enumerate(row):
                if v is None:
                    bm[(i + 2) // 8] |= 1 << ((i + 2) % 8)
                else:
                    payload += lenenc_str(str(v).encode())
            out.append(self._packet(seq, b"\x00" + bytes(bm) + payload))
            seq += 1
        out.append(self._eof(seq))
        return b"".join(out)

    def _text_resultset(self, names, rows, types) -> bytes:
        out = [self._packet(1, lenenc_int(len(names)))]
        seq = 2
        for name, typ in zip(names, types):
            nb = name.encode()
            col = (lenenc_str(b"def") + lenenc_str(b"") + lenenc_str(b"") +
                   lenenc_str(b"") + lenenc_str(nb) + lenenc_str(nb) +
                   b"\x0c" + struct.pack("<H", 33) + struct.pack("<I", 1024) +
                   bytes([typ]) + struct.pack("<H", 0) + bytes([0]) + b"\x00\x00")
            out.append(self._packet(seq, col))
            seq += 1
        out.append(self._eof(seq)); seq += 1
        for row in rows:
            payload = b"".join(b"\xfb" if v is None else lenenc_str(str(v).encode())
                               for v in row)
            out.append(self._packet(seq, payload)); seq += 1
        out.append(self._eof(seq))
        return b"".join(out)
