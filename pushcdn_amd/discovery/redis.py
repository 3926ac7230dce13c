"""Redis/KeyDB discovery backend (prod path, reference discovery/redis.rs).

Implemented over a minimal asyncio RESP client (no external redis-py
dependency in this image).  Uses plain ``EXPIRE``-able keys instead of
KeyDB's ``EXPIREMEMBER`` extension: each broker gets its own TTL'd key
``broker:{id}`` holding its connection count, so liveness falls out of key
expiry exactly as the reference's set-member expiry does (redis.rs:81-115).
Permits are ``SET ... EX`` + ``GETDEL`` (redis.rs:207-265); whitelist is a
set (redis.rs:271-326; empty set = allow all).
"""

from __future__ import annotations

import asyncio
import secrets
from typing import List, Optional, Set
from urllib.parse import urlparse

from . import BrokerIdentifier, DiscoveryClient
from ..proto.errors import DiscoveryError


class _Resp:
    """Tiny RESP2 client."""

    def __init__(self, host: str, port: int) -> None:
        self.host, self.port = host, port
        self.reader: Optional[asyncio.StreamReader] = None
        self.writer: Optional[asyncio.StreamWriter] = None
        self._lock = asyncio.Lock()

    async def _ensure(self) -> None:
        if self.writer is None or self.writer.is_closing():
            self.reader, self.writer = await asyncio.open_connection(self.host, self.port)

    async def cmd(self, *args):
        async with self._lock:
            await self._ensure()
            out = [f"*{len(args)}\r\n".encode()]
            for a in args:
                if isinstance(a, str):
                    a = a.encode()
                elif isinstance(a, int):
                    a = str(a).encode()
                out.append(f"${len(a)}\r\n".encode() + a + b"\r\n")
            self.writer.write(b"".join(out))
            await self.writer.drain()
            return await self._read_reply()

    # a compromised/byzantine discovery server must not be able to OOM the
    # broker (a "$<huge>" bulk length would readexactly() unboundedly) or
    # escape the DiscoveryError contract with ValueError/UnicodeDecodeError
    _MAX_BULK = 64 << 20
    _MAX_ARRAY = 1 << 20

    @staticmethod
    def _int(rest: bytes) -> int:
        try:
            return int(rest)
        except ValueError as e:
            raise DiscoveryError(f"bad RESP integer {rest!r}") from e

    async def _read_reply(self):
        line = await self.reader.readline()
        if not line:
            raise DiscoveryError("redis connection closed")
        kind, rest = line[:1], line[1:].strip()
        if kind == b"+":
            return rest.decode("utf-8", "replace")
        if kind == b"-":
            raise DiscoveryError(f"redis error: {rest.decode('utf-8', 'replace')}")
        if kind == b":":
            return self._int(rest)
        if kind == b"$":
            n = self._int(rest)
            if n == -1:
                return None
            if not 0 <= n <= self._MAX_BULK:
                raise DiscoveryError(f"RESP bulk length {n} out of bounds")
            data = await self.reader.readexactly(n + 2)
            return data[:-2]
        if kind == b"*":
            n = self._int(rest)
            if n == -1:
                return None
            if not 0 <= n <= self._MAX_ARRAY:
                raise DiscoveryError(f"RESP array length {n} out of bounds")
            return [await self._read_reply() for _ in range(n)]
        raise DiscoveryError(f"bad RESP reply {line!r}")


class RedisDiscovery(DiscoveryClient):
    def __init__(self, url: str, identity: Optional[BrokerIdentifier],
                 global_permits: bool = False) -> None:
        u = urlparse(url)
        self._r = _Resp(u.hostname or "127.0.0.1", u.port or 6379)
        self.identity = identity
        self.global_permits = global_permits

    async def perform_heartbeat(self, num_connections: int, expiry_s: float) -> None:
        if self.identity is None:
            raise DiscoveryError("heartbeat requires an identity")
        key = f"broker:{self.identity}"
        await self._r.cmd("SET", key, num_connections, "EX", int(max(1, expiry_s)))

    async def _broker_keys(self) -> List[str]:
        cursor = "0"
        keys: List[str] = []
        while True:
            reply = await self._r.cmd("SCAN", cursor, "MATCH", "broker:*", "COUNT", "100")
            cursor = reply[0].decode() if isinstance(reply[0], bytes) else reply[0]
            keys.extend(k.decode() if isinstance(k, bytes) else k for k in reply[1])
            if cursor == "0":
                return keys

    async def get_with_least_connections(self) -> BrokerIdentifier:
        best = None
        for key in await self._broker_keys():
            raw = await self._r.cmd("GET", key)
            if raw is None:
                continue
            ident = key[len("broker:"):]
            permits = await self._r.cmd("SCARD", f"permits:{ident}") or 0
            load = int(raw) + int(permits)
            if best is None or load < best[0] or (load == best[0] and ident < best[1]):
                best = (load, ident)
        if best is None:
            raise DiscoveryError("no brokers available")
        return BrokerIdentifier.parse(best[1])

    async def get_other_brokers(self) -> Set[BrokerIdentifier]:
        out = set()
        for key in await self._broker_keys():
            ident = BrokerIdentifier.parse(key[len("broker:"):])
            if self.identity is None or ident != self.identity:
                out.add(ident)
        return out

    async def issue_permit(
        self, broker: BrokerIdentifier, expiry_s: float, user_pubkey: bytes
    ) -> int:
        # CSPRNG: permits are the sole broker-side credential (the reference
        # uses StdRng::from_entropy(), redis.rs:207-214) — a predictable RNG
        # would let one client forecast other users' permits.
        permit = secrets.randbelow(2**63 - 2) + 2
        scope = "any" if self.global_permits else str(broker)
        await self._r.cmd("SET", f"permit:{scope}:{permit}", user_pubkey,
                          "EX", int(max(1, expiry_s)))
        await self._r.cmd("SADD", f"permits:{broker}", permit)
        await self._r.cmd("EXPIRE", f"permits:{broker}", int(max(1, expiry_s)))
        return permit

    async def validate_permit(
        self, broker: BrokerIdentifier, permit: int
    ) -> Optional[bytes]:
        scope = "any" if self.global_permits else str(broker)
        key = f"permit:{scope}:{permit}"
        raw = await self._r.cmd("GETDEL", key)
        await self._r.cmd("SREM", f"permits:{broker}", permit)
        return bytes(raw) if raw is not None else None

    async def set_whitelist(self, users: List[bytes]) -> None:
        await self._r.cmd("DEL", "whitelist")
        for u in users:
            await self._r.cmd("SADD", "whitelist", u)

    async def check_whitelist(self, user: bytes) -> bool:
        n = await self._r.cmd("SCARD", "whitelist")
        if not n:
            return True
        return bool(await self._r.cmd("SISMEMBER", "whitelist", user))
