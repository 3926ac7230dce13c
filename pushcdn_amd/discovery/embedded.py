"""Embedded discovery backend: SQLite with expiry columns and prune-on-read —
semantics-identical to the reference's sqlx implementation
(``cdn-proto/src/discovery/embedded.rs``): brokers table with TTL'd
heartbeats + connection counts, one-time permits (GETDEL semantics),
whitelist (empty whitelist = allow all).

A shared on-disk file (or shared ``file::memory:?cache=shared`` URI) plays
the Redis/KeyDB role across in-process services for tests — exactly how the
reference's integration tests fake a cluster (tests/src/tests/mod.rs:62-143).
"""

from __future__ import annotations

import asyncio
import secrets
import sqlite3
import time
from typing import List, Optional, Set

from . import BrokerIdentifier, DiscoveryClient
from ..proto.errors import DiscoveryError

_SCHEMA = """
CREATE TABLE IF NOT EXISTS brokers (
    identifier TEXT PRIMARY KEY,
    num_connections INTEGER NOT NULL DEFAULT 0,
    expiry REAL NOT NULL
);
CREATE TABLE IF NOT EXISTS permits (
    permit INTEGER PRIMARY KEY,
    broker TEXT NOT NULL,
    user_pubkey BLOB NOT NULL,
    expiry REAL NOT NULL
);
CREATE TABLE IF NOT EXISTS whitelist (
    user_pubkey BLOB PRIMARY KEY
);
"""


class EmbeddedDiscovery(DiscoveryClient):
    def __init__(self, path: str, identity: Optional[BrokerIdentifier],
                 global_permits: bool = False) -> None:
        # global_permits: permits are valid at ANY broker (the reference's
        # `global-permits` cargo feature, discovery/mod.rs:50-63)
        self.path = path
        self.identity = identity
        self.global_permits = global_permits
        self._conn = sqlite3.connect(path, timeout=10, check_same_thread=False)
        self._conn.executescript(_SCHEMA)
        self._conn.commit()
        self._lock = asyncio.Lock()

    def _prune(self) -> None:
        now = time.time()
        self._conn.execute("DELETE FROM brokers WHERE expiry < ?", (now,))
        self._conn.execute("DELETE FROM permits WHERE expiry < ?", (now,))

    async def perform_heartbeat(self, num_connections: int, expiry_s: float) -> None:
        if self.identity is None:
            raise DiscoveryError("heartbeat requires an identity")
        async with self._lock:
            self._prune()
            self._conn.execute(
                "INSERT INTO brokers(identifier, num_connections, expiry) VALUES(?,?,?) "
                "ON CONFLICT(identifier) DO UPDATE SET num_connections=?, expiry=?",
                (str(self.identity), num_connections, time.time() + expiry_s,
                 num_connections, time.time() + expiry_s),
            )
            self._conn.commit()

    async def get_with_least_connections(self) -> BrokerIdentifier:
        async with self._lock:
            self._prune()
            # load = num_connections + outstanding permits (reference redis.rs:122-172)
            rows = self._conn.execute(
                "SELECT b.identifier, b.num_connections + "
                "  (SELECT COUNT(*) FROM permits p WHERE p.broker = b.identifier) AS load "
                "FROM brokers b ORDER BY load ASC, b.identifier ASC LIMIT 1"
            ).fetchall()
            self._conn.commit()
        if not rows:
            raise DiscoveryError("no brokers available")
        return BrokerIdentifier.parse(rows[0][0])

    async def get_other_brokers(self) -> Set[BrokerIdentifier]:
        async with self._lock:
            self._prune()
            rows = self._conn.execute("SELECT identifier FROM brokers").fetchall()
            self._conn.commit()
        out = {BrokerIdentifier.parse(r[0]) for r in rows}
        if self.identity is not None:
            out.discard(self.identity)
        return out

    async def issue_permit(
        self, broker: BrokerIdentifier, expiry_s: float, user_pubkey: bytes
    ) -> int:
        async with self._lock:
            self._prune()
            for _ in range(16):
                # permit > 1: 0 = failed, 1 = success-flag (reference message.rs:338-345).
                # CSPRNG (secrets): permits are a credential; Mersenne Twister
                # output observable via a client's own permits must not let it
                # predict others' (reference StdRng::from_entropy()).
                permit = secrets.randbelow(2**63 - 2) + 2
                try:
                    self._conn.execute(
                        "INSERT INTO permits(permit, broker, user_pubkey, expiry) "
                        "VALUES(?,?,?,?)",
                        (permit, str(broker), user_pubkey, time.time() + expiry_s),
                    )
                    self._conn.commit()
                    return permit
                except sqlite3.IntegrityError:
                    continue
        raise DiscoveryError("failed to issue permit")

    async def validate_permit(
        self, broker: BrokerIdentifier, permit: int
    ) -> Optional[bytes]:
        """One-time validation (GETDEL): returns the user pubkey or None."""
        async with self._lock:
            self._prune()
            # Permits are broker-bound unless global_permits is on (reference
            # redis.rs:219-265).  The broker binding is part of the SELECT and
            # DELETE predicates so that a permit presented to the WRONG broker
            # is rejected without being consumed — it stays redeemable at the
            # broker it was issued for, matching the redis backend's
            # broker-scoped key.
            if self.global_permits:
                row = self._conn.execute(
                    "SELECT user_pubkey FROM permits WHERE permit = ?", (permit,)
                ).fetchone()
                if row is None:
                    self._conn.commit()
                    return None
                self._conn.execute("DELETE FROM permits WHERE permit = ?", (permit,))
            else:
                row = self._conn.execute(
                    "SELECT user_pubkey FROM permits WHERE permit = ? AND broker = ?",
                    (permit, str(broker)),
                ).fetchone()
                if row is None:
                    self._conn.commit()
                    return None
                self._conn.execute(
                    "DELETE FROM permits WHERE permit = ? AND broker = ?",
                    (permit, str(broker)),
                )
            self._conn.commit()
        return bytes(row[0])

    async def set_whitelist(self, users: List[bytes]) -> None:
        async with self._lock:
            self._conn.execute("DELETE FROM whitelist")
            self._conn.executemany(
                "INSERT OR IGNORE INTO whitelist(user_pubkey) VALUES(?)",
                [(u,) for u in users],
            )
            self._conn.commit()

    async def check_whitelist(self, user: bytes) -> bool:
        async with self._lock:
            n = self._conn.execute("SELECT COUNT(*) FROM whitelist").fetchone()[0]
            if n == 0:
                return True  # empty whitelist = allow all (reference redis.rs:303-314)
            row = self._conn.execute(
                "SELECT 1 FROM whitelist WHERE user_pubkey = ?", (user,)
            ).fetchone()
        return row is not None
