"""RedisDiscovery exercised against a minimal in-process RESP2 server (no
Redis binary in the image — the fake implements exactly the command subset
the client uses, with real TTL semantics).  Mirrors the reference's
KeyDB-backed semantics (``cdn-proto/src/discovery/redis.rs``): TTL'd
heartbeats, least-connections placement counting outstanding permits,
one-time permits, whitelist (empty set = allow all)."""

import asyncio
import shutil

import pytest
import time


from pushcdn_amd.discovery import BrokerIdentifier
from pushcdn_amd.discovery.redis import RedisDiscovery


class FakeRedis:
    """RESP2 server: SET[EX]/GET/GETDEL/DEL/EXPIRE/SCAN/SADD/SREM/SCARD/
    SISMEMBER with lazy TTL expiry."""

    def __init__(self):
        self.kv = {}       # key -> (bytes value, expires_at | None)
        self.sets = {}     # key -> (set of bytes, expires_at | None)
        self.server = None
        self.port = None

    def _alive(self, store, key):
        ent = store.get(key)
        if ent is None:
            return None
        if ent[1] is not None and time.monotonic() >= ent[1]:
            del store[key]
            return None
        return ent

    async def start(self):
        self.server = await asyncio.start_server(self._serve, "127.0.0.1", 0)
        self.port = self.server.sockets[0].getsockname()[1]

    async def close(self):
        self.server.close()
        await self.server.wait_closed()

    async def _serve(self, reader, writer):
        try:
            while True:
                line = await reader.readline()
                if not line:
                    return
                assert line[:1] == b"*", line
                nargs = int(line[1:].strip())
                args = []
                for _ in range(nargs):
                    hdr = await reader.readline()
                    assert hdr[:1] == b"$"
                    n = int(hdr[1:].strip())
                    args.append((await reader.readexactly(n + 2))[:-2])
                writer.write(self._dispatch(args))
                await writer.drain()
        except (ConnectionResetError, asyncio.IncompleteReadError):
            pass
        finally:
            writer.close()

    @staticmethod
    def _bulk(v):
        return b"$-1\r\n" if v is None else b"$%d\r\n%s\r\n" % (len(v), v)

    def _dispatch(self, args):
        cmd = args[0].upper().decode()
        if cmd == "SET":
            key = args[1]
            val = args[2]
            exp = None
            if len(args) >= 5 and args[3].upper() == b"EX":
                exp = time.monotonic() + int(args[4])
            self.kv[key] = (val, exp)
            return b"+OK\r\n"
        if cmd == "GET":
            ent = self._alive(self.kv, args[1])
            return self._bulk(None if ent is None else ent[0])
        if cmd == "GETDEL":
            ent = self._alive(self.kv, args[1])
            if ent is not None:
                del self.kv[args[1]]
            return self._bulk(None if ent is None else ent[0])
        if cmd == "DEL":
            n = 0
            for k in args[1:]:
                n += self.kv.pop(k, None) is not None
                n += self.sets.pop(k, None) is not None
            return b":%d\r\n" % n
        if cmd == "EXPIRE":
            for store in (self.kv, self.sets):
                ent = self._alive(store, args[1])
                if ent is not None:
                    store[args[1]] = (ent[0], time.monotonic() + int(args[2]))
                    return b":1\r\n"
            return b":0\r\n"
        if cmd == "SCAN":
            import fnmatch

            pattern = b"*"
            if b"MATCH" in [a.upper() for a in args]:
                pattern = args[[a.upper() for a in args].index(b"MATCH") + 1]
            keys = [k for k in list(self.kv) if self._alive(self.kv, k)
                    and fnmatch.fnmatch(k.decode(), pattern.decode())]
            out = b"*2\r\n" + self._bulk(b"0") + b"*%d\r\n" % len(keys)
            for k in keys:
                out += self._bulk(k)
            return out
        if cmd == "SADD":
            ent = self._alive(self.sets, args[1]) or (set(), None)
            ent[0].add(args[2])
            self.sets[args[1]] = ent
            return b":1\r\n"
        if cmd == "SREM":
            ent = self._alive(self.sets, args[1])
            if ent is None or args[2] not in ent[0]:
                return b":0\r\n"
            ent[0].discard(args[2])
            return b":1\r\n"
        if cmd == "SCARD":
            ent = self._alive(self.sets, args[1])
            return b":%d\r\n" % (0 if ent is None else len(ent[0]))
        if cmd == "SISMEMBER":
            ent = self._alive(self.sets, args[1])
            return b":%d\r\n" % (1 if ent and args[2] in ent[0] else 0)
        return b"-ERR unknown command %s\r\n" % cmd.encode()


def run(coro):
    return asyncio.run(asyncio.wait_for(coro, timeout=30))


def ident(n):
    return BrokerIdentifier(f"pub-{n}", f"priv-{n}")


def test_heartbeat_discovery_and_least_connections():
    async def go():
        srv = FakeRedis()
        await srv.start()
        url = f"redis://127.0.0.1:{srv.port}"
        a = RedisDiscovery(url, ident("a"))
        b = RedisDiscovery(url, ident("b"))
        await a.perform_heartbeat(5, 60)
        await b.perform_heartbeat(2, 60)
        # peers see each other, not themselves
        assert await a.get_other_brokers() == {ident("b")}
        assert await b.get_other_brokers() == {ident("a")}
        # placement picks the least-loaded broker (b: 2 < a: 5)
        assert await a.get_with_least_connections() == ident("b")
        # outstanding permits count toward load (reference redis.rs:117-158)
        for _ in range(4):
            await a.issue_permit(ident("b"), 60, b"userkey")
        assert await a.get_with_least_connections() == ident("a")
        await srv.close()

    run(go())


def test_heartbeat_ttl_expires():
    async def go():
        srv = FakeRedis()
        await srv.start()
        d = RedisDiscovery(f"redis://127.0.0.1:{srv.port}", ident("x"))
        await d.perform_heartbeat(0, 1)
        assert await d.get_other_brokers() == set()  # only self
        other = RedisDiscovery(f"redis://127.0.0.1:{srv.port}", ident("y"))
        await other.perform_heartbeat(0, 1)
        assert await d.get_other_brokers() == {ident("y")}
        await asyncio.sleep(1.1)  # TTL lapses -> broker disappears
        assert await d.get_other_brokers() == set()
        await srv.close()

    run(go())


def test_permit_one_time_and_scope():
    async def go():
        srv = FakeRedis()
        await srv.start()
        url = f"redis://127.0.0.1:{srv.port}"
        d = RedisDiscovery(url, ident("a"))
        permit = await d.issue_permit(ident("a"), 60, b"user-pk")
        # wrong broker scope: invalid (per-broker permits)
        assert await d.validate_permit(ident("b"), permit) is None
        assert await d.validate_permit(ident("a"), permit) == b"user-pk"
        # one-time: second validation fails (GETDEL)
        assert await d.validate_permit(ident("a"), permit) is None

        g = RedisDiscovery(url, ident("a"), global_permits=True)
        permit = await g.issue_permit(ident("a"), 60, b"user-pk")
        # global permits validate at ANY broker (reference global-permits)
        assert await g.validate_permit(ident("b"), permit) == b"user-pk"
        assert await g.validate_permit(ident("b"), permit) is None
        await srv.close()

    run(go())


def test_whitelist_semantics():
    async def go():
        srv = FakeRedis()
        await srv.start()
        d = RedisDiscovery(f"redis://127.0.0.1:{srv.port}", ident("a"))
        # empty whitelist allows everyone (reference redis.rs:308-326)
        assert await d.check_whitelist(b"anyone")
        await d.set_whitelist([b"alice", b"bob"])
        assert await d.check_whitelist(b"alice")
        assert not await d.check_whitelist(b"mallory")
        await d.set_whitelist([b"alice"])
        assert not await d.check_whitelist(b"bob")
        await srv.close()

    run(go())


@pytest.mark.skipif(
    not (shutil.which("redis-server") or shutil.which("keydb-server")),
    reason="no redis/keydb server binary in this image (opt-in lane; the "
           "deploy compose stands up eqalpha/keydb)")
def test_redis_discovery_against_real_server(tmp_path):
    """Drives RedisDiscovery against a REAL redis/keydb server (opt-in):
    heartbeat TTL expiry, least-connections scan, broker-scoped GETDEL
    permits, and whitelist semantics (reference redis.rs:81-326)."""
    import socket as _socket
    import subprocess
    import time as _time

    server_bin = shutil.which("keydb-server") or shutil.which("redis-server")
    s = _socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    proc = subprocess.Popen(
        [server_bin, "--port", str(port), "--save", "", "--appendonly", "no",
         "--dir", str(tmp_path)],
        stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL)
    try:
        deadline = _time.time() + 10
        while _time.time() < deadline:
            try:
                probe = _socket.create_connection(("127.0.0.1", port), timeout=0.5)
                probe.close()
                break
            except OSError:
                _time.sleep(0.1)

        async def go():
            from pushcdn_amd.discovery import BrokerIdentifier
            from pushcdn_amd.discovery.redis import RedisDiscovery

            a = BrokerIdentifier("a-pub", "a-priv")
            b = BrokerIdentifier("b-pub", "b-priv")
            da = RedisDiscovery(f"redis://127.0.0.1:{port}", a)
            db = RedisDiscovery(f"redis://127.0.0.1:{port}", b)

            # heartbeat + least-connections
            await da.perform_heartbeat(5, 60)
            await db.perform_heartbeat(1, 60)
            assert await da.get_with_least_connections() == b
            assert await da.get_other_brokers() == {b}

            # permits: broker-scoped, one-shot GETDEL
            permit = await da.issue_permit(b, 30, b"user-pk")
            assert permit > 1
            assert await db.validate_permit(a, permit) is None  # wrong broker
            assert await db.validate_permit(b, permit) == b"user-pk"
            assert await db.validate_permit(b, permit) is None  # consumed

            # permit load influences placement (conns + outstanding permits)
            for _ in range(10):
                await da.issue_permit(b, 30, b"u")
            assert await da.get_with_least_connections() == a

            # whitelist: empty = allow all; set restricts
            assert await da.check_whitelist(b"anyone")
            await da.set_whitelist([b"alice"])
            assert await da.check_whitelist(b"alice")
            assert not await da.check_whitelist(b"mallory")

            # heartbeat TTL expiry prunes dead brokers
            await da.perform_heartbeat(0, 1)
            await asyncio.sleep(1.5)
            assert a not in await db.get_other_brokers()

        asyncio.run(asyncio.wait_for(go(), 30))
    finally:
        proc.terminate()
        proc.wait(timeout=10)


def test_resp_parser_rejects_byzantine_replies():
    """A malicious/corrupted discovery server must surface DiscoveryError —
    never OOM (unbounded bulk/array lengths), ValueError, or
    UnicodeDecodeError."""
    import asyncio

    from pushcdn_amd.discovery.redis import _Resp
    from pushcdn_amd.proto.errors import DiscoveryError

    async def one(reply: bytes):
        srv_reader = None

        async def handler(reader, writer):
            nonlocal srv_reader
            srv_reader = reader
            await reader.read(256)  # the command
            writer.write(reply)
            await writer.drain()

        server = await asyncio.start_server(handler, "127.0.0.1", 0)
        port = server.sockets[0].getsockname()[1]
        r = _Resp("127.0.0.1", port)
        try:
            return await asyncio.wait_for(r.cmd("PING"), 5)
        finally:
            close = getattr(r, "close", None)
            if close:
                try:
                    res = close()
                    if asyncio.iscoroutine(res):
                        await res
                except Exception:
                    pass
            server.close()
            await server.wait_closed()

    async def go():
        import pytest

        # huge bulk length: bounded, not an attempted 1 TiB readexactly
        with pytest.raises(DiscoveryError):
            await one(b"$1099511627776\r\n")
        # huge array length
        with pytest.raises(DiscoveryError):
            await one(b"*2147483647\r\n")
        # non-integer length
        with pytest.raises(DiscoveryError):
            await one(b"$abc\r\n")
        # invalid UTF-8 in a simple string must not raise UnicodeDecodeError
        out = await one(b"+\xff\xfe\r\n")
        assert isinstance(out, str)
        # negative bulk other than -1
        with pytest.raises(DiscoveryError):
            await one(b"$-7\r\n")

    asyncio.run(go())
