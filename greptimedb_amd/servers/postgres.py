"""PostgreSQL wire-protocol server (L7).

Reference parity: src/servers/src/postgres/ (pgwire based). Implements the
v3 protocol simple-query flow: StartupMessage → AuthenticationOk +
ParameterStatus + ReadyForQuery; 'Q' → RowDescription/DataRow/
CommandComplete; errors as ErrorResponse; SSLRequest politely declined.
Extended protocol: Parse/Bind/Describe/Execute/Close/Sync with text-format
parameters ($1…) substituted at Bind (the query runs at Bind so Describe
can return the real RowDescription, like the reference plans at bind).
"""

from __future__ import annotations

import asyncio
import math
import re
import struct

import numpy as np

from greptimedb_amd.utils.errors import GreptimeError

OID_TEXT = 25
OID_INT8 = 20
OID_FLOAT8 = 701
OID_TIMESTAMP = 1114


def _msg(tag: bytes, payload: bytes) -> bytes:
    return tag + struct.pack("!I", len(payload) + 4) + payload


class PostgresServer:
    def __init__(self, executor, host="0.0.0.0", port=4003, user_provider=None):
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

    async def _handle(self, reader, writer):
        try:
            # startup (possibly preceded by SSLRequest)
            while True:
                hdr = await reader.readexactly(4)
                (ln,) = struct.unpack("!I", hdr)
                body = await reader.readexactly(ln - 4)
                if len(body) >= 4:
                    (code,) = struct.unpack("!I", body[:4])
                else:
                    code = 0
                if code == 80877103:      # SSLRequest → decline
                    writer.write(b"N")
                    await writer.drain()
                    continue
                if code == 80877102:      # CancelRequest → ignore
                    writer.close()
                    return
                break  # StartupMessage
            params = {}
            parts = body[4:].split(b"\x00")
            for i in range(0, len(parts) - 1, 2):
                if parts[i]:
                    params[parts[i].decode()] = parts[i + 1].decode()
            user = params.get("user", "")
            if self.user_provider is not None:
                from greptimedb_amd.servers.auth import (ScramSha256Server,
                                                         password_of,
                                                         pg_md5_check)
                ok = self.user_provider.allow(user)
                stored = password_of(self.user_provider, user)
                if ok and stored:
                    # SASL SCRAM-SHA-256 (RFC 7677; reference: pgwire SCRAM)
                    writer.write(_msg(b"R", struct.pack("!I", 10) +
                                      b"SCRAM-SHA-256\x00\x00"))
                    await writer.drain()
                    tag = await reader.readexactly(1)
                    (ln,) = struct.unpack("!I", await reader.readexactly(4))
                    body = await reader.readexactly(ln - 4)
                    ok = False
                    if tag == b"p":
                        mech_end = body.index(b"\x00")
                        mech = body[:mech_end].decode()
                        (rlen,) = struct.unpack("!i", body[mech_end + 1:
                                                           mech_end + 5])
                        initial = body[mech_end + 5: mech_end + 5 + rlen] \
                            .decode() if rlen >= 0 else ""
                        if mech == "SCRAM-SHA-256" and initial:
                            scram = ScramSha256Server(stored)
                            sfirst = scram.server_first(initial)
                            writer.write(_msg(b"R", struct.pack("!I", 11) +
                                              sfirst.encode()))
                            await writer.drain()
                            tag2 = await reader.readexactly(1)
                            (ln2,) = struct.unpack(
                                "!I", await reader.readexactly(4))
                            cfinal = (await reader.readexactly(ln2 - 4)).decode()
                            if tag2 == b"p":
                                v = scram.verify_client_final(cfinal)
                                if v is not None:
                                    writer.write(_msg(
                                        b"R", struct.pack("!I", 12) + v.encode()))
                                    ok = True
                        elif mech == "PLAIN" or not initial:
                            # legacy fallback: md5 challenge-response
                            salt = b"\x9a\x17\x2e\x41"
                            writer.write(_msg(b"R", struct.pack("!I", 5) + salt))
                            await writer.drain()
                            tag3 = await reader.readexactly(1)
                            (ln3,) = struct.unpack(
                                "!I", await reader.readexactly(4))
                            pw = (await reader.readexactly(ln3 - 4)).rstrip(b"\x00")
                            ok = tag3 == b"p" and pg_md5_check(
                                stored, user, salt, pw.decode(errors="replace"))
                if not ok:
                    writer.write(_msg(b"E", b"SSFATAL\x00C28000\x00M" +
                                      f"auth failed for {user}".encode() + b"\x00\x00"))
                    await writer.drain()
                    writer.close()
                    return
            writer.write(_msg(b"R", struct.pack("!I", 0)))  # AuthenticationOk
            for k, v in (("server_version", "16.0 (greptimedb-amd)"),
                         ("client_encoding", "UTF8"),
                         ("DateStyle", "ISO"),
                         ("server_encoding", "UTF8")):
                writer.write(_msg(b"S", k.encode() + b"\x00" + v.encode() + b"\x00"))
            writer.write(_msg(b"K", struct.pack("!II", 1, 1)))  # BackendKeyData
            writer.write(_msg(b"Z", b"I"))
            await writer.drain()

            stmts: dict = {}    # name → (sql, n_params)   (extended protocol)
            portals: dict = {}  # name → QueryResult | GreptimeError

            while True:
                try:
                    tag = await reader.readexactly(1)
                except (asyncio.IncompleteReadError, ConnectionResetError):
                    break
                hdr = await reader.readexactly(4)
                (ln,) = struct.unpack("!I", hdr)
                body = await reader.readexactly(ln - 4)
                if tag == b"X":  # Terminate
                    break
                if tag == b"Q":
                    sql = body.rstrip(b"\x00").decode(errors="replace")
                    for chunk in self._run_query(sql):
                        writer.write(chunk)
                    writer.write(_msg(b"Z", b"I"))
                    await writer.drain()
                elif tag in (b"P", b"B", b"D", b"E", b"C", b"S", b"H"):
                    for chunk in self._extended(tag, body, stmts, portals):
                        writer.write(chunk)
                    await writer.drain()
                else:
                    writer.write(_msg(b"Z", b"I"))
                    await writer.drain()
        except (asyncio.IncompleteReadError, ConnectionResetError):
            pass
        finally:
            try:
                writer.close()
            except Exception:
                pass

    # ---------------------------------------------- extended query protocol

    _NUM = re.compile(r"-?\d+(\.\d+)?([eE][+-]?\d+)?$")

    def _extended(self, tag: bytes, body: bytes, stmts: dict, portals: dict):
        _NUM = self._NUM

        def err(msg: str, code: bytes = b"42601"):
            return _msg(b"E", b"SERROR\x00C" + code + b"\x00M" +
                        msg.encode()[:400] + b"\x00\x00")

        if tag == b"S":                       # Sync
            yield _msg(b"Z", b"I")
            return
        if tag == b"H":                       # Flush — nothing buffered
            return
        if tag == b"P":                       # Parse: name\0 sql\0 n oids…
            name, rest = body.split(b"\x00", 1)
            sql, rest = rest.split(b"\x00", 1)
            (np_,) = struct.unpack("!h", rest[:2])
            stmts[name] = (sql.decode(errors="replace"), np_)
            yield _msg(b"1", b"")             # ParseComplete
            return
        if tag == b"B":                       # Bind: portal\0 stmt\0 fmts, params
            portal, rest = body.split(b"\x00", 1)
            sname, rest = rest.split(b"\x00", 1)
            off = 0
            (nfmt,) = struct.unpack_from("!h", rest, off); off += 2
            fmts = struct.unpack_from(f"!{nfmt}h", rest, off); off += 2 * nfmt
            (nparams,) = struct.unpack_from("!h", rest, off); off += 2
            params = []
            for i in range(nparams):
                (plen,) = struct.unpack_from("!i", rest, off); off += 4
                if plen < 0:
                    params.append(None)
                else:
                    params.append(rest[off:off + plen]); off += plen
                if (fmts[i] if i < nfmt else (fmts[0] if nfmt == 1 else 0)) == 1:
                    yield err("binary parameters not supported", b"0A000")
                    return
            got = stmts.get(sname)
            if got is None:
                yield err(f"unknown prepared statement {sname!r}", b"26000")
                return
            sql, _np = got
            for i in range(len(params), 0, -1):   # $10 before $1
                p = params[i - 1]
                if p is None:
                    lit = "NULL"
                else:
                    s = p.decode(errors="replace")
                    lit = s if _NUM.match(s) else "'" + s.replace("'", "''") + "'"
                sql = sql.replace(f"${i}", lit)
            try:
                portals[portal] = self.executor.execute(sql)
            except GreptimeError as e:
                portals[portal] = e
            except Exception as e:              # pragma: no cover
                portals[portal] = GreptimeError(f"{type(e).__name__}: {e}")
            yield _msg(b"2", b"")               # BindComplete
            return
        if tag == b"D":                        # Describe 'S'|'P' + name
            kind, name = body[:1], body[1:].split(b"\x00")[0]
            if kind == b"S":
                got = stmts.get(name)
                np_ = got[1] if got else 0
                yield _msg(b"t", struct.pack("!h", np_) +
                           struct.pack(f"!{np_}I", *([OID_TEXT] * np_)))
                yield _msg(b"n", b"")           # NoData (schema known at Bind)
                return
            r = portals.get(name)
            if isinstance(r, GreptimeError) or r is None:
                yield _msg(b"n", b"")
                return
            yield self._row_description(r)
            return
        if tag == b"E":                        # Execute: portal\0 maxrows
            name = body.split(b"\x00")[0]
            r = portals.get(name)
            if r is None:
                yield err(f"unknown portal {name!r}", b"34000")
                return
            if isinstance(r, GreptimeError):
                yield err(str(r) or type(r).__name__)
                return
            nrows = 0
            for chunk in self._data_rows(r):
                nrows += 1
                yield chunk
            yield _msg(b"C", f"SELECT {nrows}".encode() + b"\x00")
            return
        if tag == b"C":                        # Close statement/portal
            kind, name = body[:1], body[1:].split(b"\x00")[0]
            (stmts if kind == b"S" else portals).pop(name, None)
            yield _msg(b"3", b"")
            return

    def _run_query(self, sql: str):
        s = sql.strip().rstrip(";").lower()
        if not s or s.startswith(("set ", "begin", "commit", "rollback")):
            yield _msg(b"C", b"SET\x00")
            return
        try:
            r = self.executor.execute(sql)
        except GreptimeError as e:
            yield _msg(b"E", b"SERROR\x00C42601\x00M" +
                       (str(e) or type(e).__name__).encode()[:400] + b"\x00\x00")
            return
        except Exception as e:  # pragma: no cover
            yield _msg(b"E", b"SERROR\x00CXX000\x00M" +
                       f"{type(e).__name__}: {e}".encode()[:400] + b"\x00\x00")
            return
        yield self._row_description(r)
        nrows = 0
        for chunk in self._data_rows(r):
            nrows += 1
            yield chunk
        yield _msg(b"C", f"SELECT {nrows}".encode() + b"\x00")

    def _row_description(self, r) -> bytes:
        fields = b""
        for col, kind, name in zip(r.columns, r.kinds, r.names):
            if kind == "ts":
                oid = OID_TIMESTAMP
            elif len(col) and isinstance(col[0], (int, np.integer)):
                oid = OID_INT8
            elif len(col) and isinstance(col[0], (float, np.floating)):
                oid = OID_FLOAT8
            else:
                oid = OID_TEXT
            fields += (name.encode() + b"\x00" + struct.pack("!IhIhih", 0, 0, oid,
                                                             -1, -1, 0))
        return _msg(b"T", struct.pack("!h", len(r.names)) + fields)

    def _data_rows(self, r):
        from greptimedb_amd.utils.timeutil import format_ts_ms
        for row in r.rows():
            payload = struct.pack("!h", len(row))
            for v, kind in zip(row, r.kinds):
                if v is None or (isinstance(v, float) and math.isnan(v)):
                    payload += struct.pack("!i", -1)
                else:
                    if kind == "ts":
                        sv = format_ts_ms(int(v)).replace("T", " ")
                    else:
                        sv = str(v)
                    b = sv.encode()
                    payload += struct.pack("!i", len(b)) + b
            yield _msg(b"D", payload)
