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
