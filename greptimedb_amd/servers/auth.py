"""Authentication (reference parity: src/auth — UserProvider trait,
static file provider `user=password` lines, permission checks)."""

from __future__ import annotations

import hmac


class UserProvider:
    def allow(self, user: str, password: str | None = None) -> bool:
        raise NotImplementedError


class StaticUserProvider(UserProvider):
    """`user=password[:ro]` map (reference: static_user_provider file
    format + permission checker seam; `:ro` marks a read-only user)."""

    def __init__(self, users: dict[str, str],
                 modes: dict[str, str] | None = None):
        self.users = dict(users)
        self.modes = dict(modes or {})

    @staticmethod
    def from_file(path: str) -> "StaticUserProvider":
        users, modes = {}, {}
        with open(path) as f:
            for line in f:
                line = line.strip()
                if line and "=" in line and not line.startswith("#"):
                    u, p = line.split("=", 1)
                    u = u.strip()
                    p = p.strip()
                    if p.endswith(":ro"):
                        p = p[:-3]
                        modes[u] = "ro"
                    users[u] = p
        return StaticUserProvider(users, modes)

    def mode(self, user: str) -> str:
        return self.modes.get(user, "rw")

    def allow(self, user: str, password: str | None = None) -> bool:
        if user not in self.users:
            return False
        if password is None:
            return True  # transport did not carry a password (trusted channel)
        return hmac.compare_digest(self.users[user], password)


class AllowAllProvider(UserProvider):
    def allow(self, user: str, password: str | None = None) -> bool:
        return True


def password_of(provider, user: str) -> str | None:
    """Stored password for challenge-response schemes (None = unknown or
    provider has no secrets — transport should fall back to trust)."""
    users = getattr(provider, "users", None)
    if isinstance(users, dict):
        return users.get(user)
    return None


def mysql_native_check(password: str, scramble: bytes, token: bytes) -> bool:
    """mysql_native_password: token == SHA1(pass) XOR SHA1(scramble +
    SHA1(SHA1(pass))) (MySQL secure auth handshake)."""
    import hashlib
    if not token:
        return password == ""
    h1 = hashlib.sha1(password.encode()).digest()
    h2 = hashlib.sha1(h1).digest()
    expect = bytes(a ^ b for a, b in
                   zip(h1, hashlib.sha1(scramble + h2).digest()))
    return hmac.compare_digest(expect, token)


def pg_md5_check(password: str, user: str, salt: bytes, response: str) -> bool:
    """PostgreSQL md5 auth: 'md5' + md5(md5(password + user) + salt)."""
    import hashlib
    inner = hashlib.md5((password + user).encode()).hexdigest()
    expect = "md5" + hashlib.md5(inner.encode() + salt).hexdigest()
    return hmac.compare_digest(expect, response)


# ---------------------------------------------------------------- SCRAM
# SCRAM-SHA-256 (RFC 5802/7677) server side — reference: pgwire's SCRAM
# support in src/servers/src/postgres (builder enables SCRAM auth).

import base64 as _b64
import hashlib as _hashlib
import os as _os


def _h(data: bytes) -> bytes:
    return _hashlib.sha256(data).digest()


def _hmac(key: bytes, msg: bytes) -> bytes:
    return hmac.new(key, msg, _hashlib.sha256).digest()


def _xor(a: bytes, b: bytes) -> bytes:
    return bytes(x ^ y for x, y in zip(a, b))


class ScramSha256Server:
    """One SCRAM-SHA-256 exchange (server side).

    usage:
        s = ScramSha256Server(password)
        server_first = s.server_first(client_first_bare_message)
        server_final = s.verify_client_final(client_final_message)  # or None
    """

    ITERATIONS = 4096

    def __init__(self, password: str):
        self.salt = _os.urandom(16)
        salted = _hashlib.pbkdf2_hmac("sha256", password.encode(), self.salt,
                                      self.ITERATIONS)
        self.stored_key = _h(_hmac(salted, b"Client Key"))
        self.server_key = _hmac(salted, b"Server Key")
        self.server_nonce = _b64.b64encode(_os.urandom(18)).decode()
        self.client_first_bare = None
        self.server_first_msg = None

    @staticmethod
    def parse_client_first(msg: str) -> tuple[str, str]:
        """gs2-header,client-first-bare → (username, client_nonce)."""
        # e.g. "n,,n=user,r=nonce"
        parts = msg.split(",", 2)
        bare = parts[2] if len(parts) >= 3 else msg
        attrs = dict(kv.split("=", 1) for kv in bare.split(",") if "=" in kv)
        return attrs.get("n", ""), attrs.get("r", "")

    def server_first(self, client_first: str) -> str:
        parts = client_first.split(",", 2)
        self.client_first_bare = parts[2] if len(parts) >= 3 else client_first
        _user, cnonce = self.parse_client_first(client_first)
        self.full_nonce = cnonce + self.server_nonce
        self.server_first_msg = (
            f"r={self.full_nonce},s={_b64.b64encode(self.salt).decode()},"
            f"i={self.ITERATIONS}")
        return self.server_first_msg

    def verify_client_final(self, client_final: str) -> str | None:
        """Returns the server-final-message ('v=...') or None on failure."""
        attrs = dict(kv.split("=", 1) for kv in client_final.split(",")
                     if "=" in kv)
        proof_b64 = attrs.get("p")
        if proof_b64 is None or attrs.get("r") != self.full_nonce:
            return None
        without_proof = client_final[: client_final.rfind(",p=")]
        auth_message = ",".join([self.client_first_bare,
                                 self.server_first_msg, without_proof]).encode()
        client_sig = _hmac(self.stored_key, auth_message)
        client_key = _xor(_b64.b64decode(proof_b64), client_sig)
        if not hmac.compare_digest(_h(client_key), self.stored_key):
            return None
        server_sig = _hmac(self.server_key, auth_message)
        return "v=" + _b64.b64encode(server_sig).decode()


def scram_client_messages(user: str, password: str, server_first_fn):
    """Test/client helper: runs the client side of SCRAM-SHA-256.
    `server_first_fn(client_first) -> server_first`; returns
    (client_final, expected_server_sig_checker)."""
    cnonce = _b64.b64encode(_os.urandom(18)).decode()
    client_first_bare = f"n={user},r={cnonce}"
    client_first = "n,," + client_first_bare
    server_first = server_first_fn(client_first)
    attrs = dict(kv.split("=", 1) for kv in server_first.split(",") if "=" in kv)
    full_nonce, salt, iters = attrs["r"], _b64.b64decode(attrs["s"]), int(attrs["i"])
    assert full_nonce.startswith(cnonce)
    salted = _hashlib.pbkdf2_hmac("sha256", password.encode(), salt, iters)
    client_key = _hmac(salted, b"Client Key")
    stored_key = _h(client_key)
    without_proof = f"c=biws,r={full_nonce}"
    auth_message = ",".join([client_first_bare, server_first,
                             without_proof]).encode()
    client_sig = _hmac(stored_key, auth_message)
    proof = _b64.b64encode(_xor(client_key, client_sig)).decode()
    client_final = f"{without_proof},p={proof}"
    server_key = _hmac(salted, b"Server Key")
    expected_v = "v=" + _b64.b64encode(_hmac(server_key, auth_message)).decode()
    return client_first, client_final, expected_v


def mysql_caching_sha2_check(password: str, nonce: bytes, token: bytes) -> bool:
    """caching_sha2_password fast-auth scramble (MySQL 8 default):
    token = XOR(SHA256(pwd), SHA256(SHA256(SHA256(pwd)) || nonce))."""
    if not token:
        return False
    p1 = _hashlib.sha256(password.encode()).digest()
    p2 = _hashlib.sha256(p1).digest()
    expected = bytes(a ^ b for a, b in zip(
        p1, _hashlib.sha256(p2 + nonce).digest()))
    return hmac.compare_digest(expected, token)


# ------------------------------------------------------------ permissions

WRITE_STMT_NAMES = {
    "InsertValues", "Delete", "CreateTable", "DropTable", "AlterTable",
    "TruncateTable", "CreateView", "DropView", "CreateDatabase",
    "DropDatabase", "CreateFlow", "DropFlow", "Admin", "Copy",
}


def check_permission(provider, user: str, sql: str) -> bool:
    """Reference parity: src/auth permission checker — deny DML/DDL to
    read-only users. Statement kind comes from the parser (COPY TO is a
    read; COPY FROM writes)."""
    if provider is None or getattr(provider, "mode", None) is None:
        return True
    if provider.mode(user) != "ro":
        return True
    from greptimedb_amd.query.parser import parse_sql
    try:
        stmt = parse_sql(sql)
    except Exception:
        return True  # executor reports the real syntax error
    name = type(stmt).__name__
    if name == "Copy":
        return getattr(stmt, "direction", "to") == "to"
    return name not in WRITE_STMT_NAMES
