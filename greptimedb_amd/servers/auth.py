"""Authentication (reference parity: src/auth — UserProvider trait,
static file provider `user=password` lines, permission checks)."""

from __future__ import annotations

import hmac


class UserProvider:
    def allow(self, user: str, password: str | None = None) -> bool:
        raise NotImplementedError


class StaticUserProvider(UserProvider):
    """`user=password` map (reference: static_user_provider file format)."""

    def __init__(self, users: dict[str, str]):
        self.users = dict(users)

    @staticmethod
    def from_file(path: str) -> "StaticUserProvider":
        users = {}
        with open(path) as f:
            for line in f:
                line = line.strip()
                if line and "=" in line and not line.startswith("#"):
                    u, p = line.split("=", 1)
                    users[u.strip()] = p.strip()
        return StaticUserProvider(users)

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
