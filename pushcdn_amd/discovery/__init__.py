"""Discovery / coordination layer.

Mirrors the reference trait surface (``cdn-proto/src/discovery/mod.rs:28-76``):
heartbeats with TTL, least-connections broker selection, one-time permits,
whitelist.  Backends: ``Embedded`` (SQLite, tests/local — reference
embedded.rs) and ``Redis`` (KeyDB, prod — reference redis.rs; requires a
reachable server, gated at runtime).
"""

from __future__ import annotations

from abc import ABC, abstractmethod
from dataclasses import dataclass
from typing import List, Optional, Set


@dataclass(frozen=True, order=True)
class BrokerIdentifier:
    """{public_advertise_endpoint}/{private_advertise_endpoint} — ordered so
    it can serve as the CRDT conflict identity (reference discovery/mod.rs:80-129)."""

    public_advertise_endpoint: str
    private_advertise_endpoint: str

    def __str__(self) -> str:
        return f"{self.public_advertise_endpoint}/{self.private_advertise_endpoint}"

    @classmethod
    def parse(cls, s: str) -> "BrokerIdentifier":
        pub, sep, priv = s.partition("/")
        if not sep:
            raise ValueError(f"bad broker identifier {s!r}")
        return cls(pub, priv)


class DiscoveryClient(ABC):
    """The coordination API every backend implements."""

    @abstractmethod
    async def perform_heartbeat(self, num_connections: int, expiry_s: float) -> None: ...

    @abstractmethod
    async def get_with_least_connections(self) -> BrokerIdentifier: ...

    @abstractmethod
    async def get_other_brokers(self) -> Set[BrokerIdentifier]: ...

    @abstractmethod
    async def issue_permit(
        self, broker: BrokerIdentifier, expiry_s: float, user_pubkey: bytes
    ) -> int: ...

    @abstractmethod
    async def validate_permit(
        self, broker: BrokerIdentifier, permit: int
    ) -> Optional[bytes]: ...

    @abstractmethod
    async def set_whitelist(self, users: List[bytes]) -> None: ...

    @abstractmethod
    async def check_whitelist(self, user: bytes) -> bool: ...


def new_discovery_client(path: str, identity: Optional[BrokerIdentifier],
                         global_permits: bool = False) -> DiscoveryClient:
    """Factory: redis:// URLs get the Redis client, anything else Embedded.
    global_permits mirrors the reference's cargo feature of the same name."""
    if path.startswith("redis://") or path.startswith("rediss://"):
        from .redis import RedisDiscovery

        return RedisDiscovery(path, identity, global_permits)
    from .embedded import EmbeddedDiscovery

    return EmbeddedDiscovery(path, identity, global_permits)
