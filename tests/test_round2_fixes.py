"""Round-2 hardening regression tests (ADVICE.md + VERDICT.md items):

- G2 verkey subgroup membership enforced on deserialize (ADVICE medium;
  reference ark deserialize_uncompressed, signature.rs:92-110)
- future timestamps rejected in marshal/broker auth (ADVICE low;
  reference u64-wrap semantics, marshal.rs:66-83)
- embedded permits are broker-bound and NOT consumed by the wrong broker
  (ADVICE low; reference redis.rs:219-265 broker-scoped keys)
- permits come from a CSPRNG (ADVICE high)
- keyed routing hash + registration collision refusal (ADVICE low;
  reference DirectMap keys on the full pubkey, direct/mod.rs:14)
- departed users' direct entries removed (no slot-reuse misdelivery)
- GPU-broker capacity exhaustion refuses cleanly instead of orphaning a
  registered user (VERDICT weak 4; reference connections/mod.rs:278-304)
"""

import asyncio
import time
import uuid

import pytest

from pushcdn_amd.crypto import bls
from pushcdn_amd.proto import message as m
from pushcdn_amd.proto.transports.memory import Memory
from pushcdn_amd.utils.keyhash import derive_routing_seed, fnv1a64

P = 21888242871839275222246405745257275088696311157297823662689037894645226208583
R = 21888242871839275222246405745257275088548364400416034343698204186575808495617


def run(coro):
    return asyncio.run(asyncio.wait_for(coro, timeout=60))


# --------------------------- G2 subgroup check ---------------------------

def _f2mul(a, b):
    return ((a[0] * b[0] - a[1] * b[1]) % P, (a[0] * b[1] + a[1] * b[0]) % P)


def _f2add(a, b):
    return ((a[0] + b[0]) % P, (a[1] + b[1]) % P)


def _f2sub(a, b):
    return ((a[0] - b[0]) % P, (a[1] - b[1]) % P)


def _f2inv(a):
    n = (a[0] * a[0] + a[1] * a[1]) % P
    ni = pow(n, P - 2, P)
    return ((a[0] * ni) % P, (-a[1] * ni) % P)


def _fsqrt(a):
    y = pow(a, (P + 1) // 4, P)
    return y if y * y % P == a else None


def _f2sqrt(z):
    a, b = z
    if b == 0:
        s = _fsqrt(a)
        if s is not None:
            return (s, 0)
        s = _fsqrt((-a) % P)
        return (0, s) if s is not None else None
    al = _fsqrt((a * a + b * b) % P)
    if al is None:
        return None
    inv2 = pow(2, P - 2, P)
    de = (a + al) * inv2 % P
    x0 = _fsqrt(de)
    if x0 is None:
        x0 = _fsqrt((a - al) * inv2 % P)
        if x0 is None:
            return None
    x1 = b * pow(2 * x0 % P, P - 2, P) % P
    c = (x0, x1)
    return c if _f2mul(c, c) == z else None


_B2 = _f2mul((3, 0), _f2inv((9, 1)))


def _ec_add(Pt, Q):
    if Pt is None:
        return Q
    if Q is None:
        return Pt
    if Pt[0] == Q[0]:
        if Pt[1] != Q[1]:
            return None
        lam = _f2mul(_f2mul((3, 0), _f2mul(Pt[0], Pt[0])),
                     _f2inv(_f2mul((2, 0), Pt[1])))
    else:
        lam = _f2mul(_f2sub(Q[1], Pt[1]), _f2inv(_f2sub(Q[0], Pt[0])))
    x3 = _f2sub(_f2sub(_f2mul(lam, lam), Pt[0]), Q[0])
    y3 = _f2sub(_f2mul(lam, _f2sub(Pt[0], x3)), Pt[1])
    return (x3, y3)


def _ec_mul(Pt, k):
    acc = None
    while k:
        if k & 1:
            acc = _ec_add(acc, Pt)
        Pt = _ec_add(Pt, Pt)
        k >>= 1
    return acc


def _find_cofactor_point(seed):
    """An on-curve E'(Fp2) point OUTSIDE the r-order subgroup (the huge-
    cofactor part is ~all of the curve, so a random point qualifies)."""
    import random

    rng = random.Random(seed)
    while True:
        x = (rng.randrange(P), rng.randrange(P))
        rhs = _f2add(_f2mul(_f2mul(x, x), x), _B2)
        y = _f2sqrt(rhs)
        if y is None:
            continue
        if _ec_mul((x, y), R) is not None:  # r*P != infinity -> not in subgroup
            return (x, y)


def _core():
    from pushcdn_amd.ops.build import build_core

    return build_core()


def test_verkey_subgroup_membership_enforced():
    core = _core()
    x, y = _find_cofactor_point(1)
    bad = b"".join(c.to_bytes(32, "little") for c in (x[0], x[1], y[0], y[1]))
    assert core._verkey_ok(bad) is False
    # a real verkey still accepted
    kp = bls.KeyPair.from_seed(3)
    assert core._verkey_ok(kp.public_key) is True
    # and verify() refuses the bad key outright
    sig = bls.sign(kp.private_key, "ns", b"msg")
    assert bls.verify(bad, "ns", b"msg", sig) is False


def test_verify_still_roundtrips_after_subgroup_check():
    kp = bls.KeyPair.from_seed(11)
    sig = bls.sign(kp.private_key, "espresso-cdn-user-marshal-auth", b"payload")
    assert bls.verify(kp.public_key, "espresso-cdn-user-marshal-auth", b"payload", sig)
    assert not bls.verify(kp.public_key, "wrong-namespace", b"payload", sig)


# --------------------------- timestamp window ---------------------------

async def _auth_pair(name):
    """(client_conn, server_conn) over the Memory transport."""
    from pushcdn_amd.proto.limiter import Limiter

    listener = await Memory.bind(name, None, None)
    client = await Memory.connect(name, True, Limiter(None))
    unf = await listener.accept()
    server = await unf.finalize(Limiter(None))
    return client, server


def test_marshal_rejects_future_timestamp(tmp_path):
    from pushcdn_amd.auth.marshal import MarshalAuth
    from pushcdn_amd.discovery.embedded import EmbeddedDiscovery

    async def go():
        db = str(tmp_path / f"{uuid.uuid4().hex}.db")
        disc = EmbeddedDiscovery(db, None)
        client, server = await _auth_pair(f"fut-{uuid.uuid4().hex[:6]}")
        kp = bls.KeyPair.from_seed(5)
        ts = int(time.time()) + 3600  # pre-signed for the future
        sig = bls.sign_timestamp(kp.private_key, bls.USER_MARSHAL_NAMESPACE, ts)
        await client.send_message(m.AuthenticateWithKey(
            public_key=kp.public_key, timestamp=ts, signature=sig))
        result = await MarshalAuth.verify_user(server, disc)
        assert result is None
        resp = await client.recv_message()
        assert isinstance(resp, m.AuthenticateResponse) and resp.permit == 0

    run(go())


def test_broker_auth_rejects_future_timestamp():
    from pushcdn_amd.auth.broker import BrokerAuth
    from pushcdn_amd.discovery import BrokerIdentifier

    async def go():
        client, server = await _auth_pair(f"futb-{uuid.uuid4().hex[:6]}")
        kp = bls.KeyPair.from_seed(7)
        ts = int(time.time()) + 3600
        sig = bls.sign_timestamp(kp.private_key, bls.BROKER_BROKER_NAMESPACE, ts)
        await client.send_message(m.AuthenticateWithKey(
            public_key=kp.public_key, timestamp=ts, signature=sig))
        ident = BrokerIdentifier("a", "b")
        ok = await BrokerAuth.verify_broker(server, ident, kp)
        assert ok is False

    run(go())


# --------------------------- permit semantics ---------------------------

def test_embedded_permit_wrong_broker_not_consumed(tmp_path):
    from pushcdn_amd.discovery import BrokerIdentifier
    from pushcdn_amd.discovery.embedded import EmbeddedDiscovery

    async def go():
        db = str(tmp_path / f"{uuid.uuid4().hex}.db")
        a = BrokerIdentifier("a-pub", "a-priv")
        b = BrokerIdentifier("b-pub", "b-priv")
        disc = EmbeddedDiscovery(db, a)
        permit = await disc.issue_permit(a, 30.0, b"userkey")
        assert permit > 1
        # presented to the WRONG broker: rejected AND not consumed
        assert await disc.validate_permit(b, permit) is None
        # still redeemable at the right broker, exactly once
        assert await disc.validate_permit(a, permit) == b"userkey"
        assert await disc.validate_permit(a, permit) is None

    run(go())


def test_permits_use_csprng(tmp_path):
    """Permits must come from the secrets module, not random (ADVICE high).
    Behavioral smoke: range is [2, 2^63] and values differ across issues."""
    from pushcdn_amd.discovery import BrokerIdentifier
    from pushcdn_amd.discovery.embedded import EmbeddedDiscovery
    import inspect

    import pushcdn_amd.discovery.embedded as emb
    import pushcdn_amd.discovery.redis as rds

    assert "secrets" in inspect.getsource(emb.EmbeddedDiscovery.issue_permit) or \
        "secrets" in inspect.getsource(emb)
    assert "import secrets" in inspect.getsource(rds)
    assert "random.randrange" not in inspect.getsource(emb)
    assert "random.randrange" not in inspect.getsource(rds)

    async def go():
        db = str(tmp_path / f"{uuid.uuid4().hex}.db")
        a = BrokerIdentifier("a-pub", "a-priv")
        disc = EmbeddedDiscovery(db, a)
        seen = set()
        for _ in range(16):
            permit = await disc.issue_permit(a, 30.0, b"u")
            assert 2 <= permit <= 2**63
            seen.add(permit)
        assert len(seen) == 16

    run(go())


# --------------------------- keyed routing hash ---------------------------

def test_seeded_fnv_and_derive():
    assert fnv1a64(b"abc") == fnv1a64(b"abc", 0)
    assert fnv1a64(b"abc", 1) != fnv1a64(b"abc", 0)
    s1 = derive_routing_seed(b"\x01" * 32)
    s2 = derive_routing_seed(b"\x02" * 32)
    assert s1 != s2 and 0 <= s1 < 2**64
    # deterministic (cluster-wide agreement)
    assert derive_routing_seed(b"\x01" * 32) == s1


def test_reference_parse_uses_seed():
    from pushcdn_amd.ops import reference as ref

    raw = m.serialize(m.Direct(recipient=b"R" * 64, message=b"x" * 16))
    pr0 = ref.parse_batch(raw, [0, len(raw)], 0)
    pr1 = ref.parse_batch(raw, [0, len(raw)], 12345)
    assert int(pr0.recip_hash[0]) != int(pr1.recip_hash[0])
    want = fnv1a64(b"R" * 64, 12345)
    got = int(pr1.recip_hash[0]) & ((1 << 64) - 1)
    assert got == want


def test_direct_hash_collision_refused():
    from pushcdn_amd.broker.gpu_engine import GpuBrokerEngine

    eng = GpuBrokerEngine(device="cpu", n_users=8, ring_bytes=1 << 12,
                          use_gpu_ops=False, hash_seed=99)
    eng.register_direct(b"alice-key", 0)
    # same key re-registers fine (reconnect)
    eng.register_direct(b"alice-key", 1)
    # a DIFFERENT key landing on the same 64-bit hash is refused
    h = fnv1a64(b"bob-key", 99)
    eng._direct_pubkeys[h] = b"someone-else"
    with pytest.raises(ValueError):
        eng.register_direct(b"bob-key", 2)


def test_unregister_direct_removes_entry():
    import torch

    from pushcdn_amd.broker.gpu_engine import GpuBrokerEngine
    from pushcdn_amd.ops import reference as ref

    eng = GpuBrokerEngine(device="cpu", n_users=8, ring_bytes=1 << 12,
                          use_gpu_ops=False, hash_seed=5)
    eng.register_direct(b"carol", 3)
    q = torch.tensor([ref._i64(fnv1a64(b"carol", 5))], dtype=torch.int64)
    assert int(ref.direct_lookup(eng.direct_keys, eng.direct_vals, q)[0]) == 3
    eng.unregister_direct(b"carol")
    assert int(ref.direct_lookup(eng.direct_keys, eng.direct_vals, q)[0]) < 0


# --------------------------- capacity refusal ---------------------------

def test_gpu_capacity_refused_cleanly(tmp_path):
    """gpu_max_users+1 connects: the extra user is refused without
    orphaning state; existing users keep working."""
    from tests.test_integration import make_client, make_marshal, new_db, stop_stack
    from pushcdn_amd.broker.service import Broker, BrokerConfig

    async def go():
        db = new_db(tmp_path)
        cfg = BrokerConfig(
            public_bind_endpoint="cap-pub",
            public_advertise_endpoint="cap-pub",
            private_bind_endpoint="cap-priv",
            private_advertise_endpoint="cap-priv",
            discovery_endpoint=db,
            keypair=bls.KeyPair.from_seed(1000),
            user_protocol=Memory,
            broker_protocol=Memory,
            heartbeat_interval_s=0.2,
            data_plane="gpu",
            gpu_device="cpu",
            gpu_max_users=2,
            gpu_ring_bytes=1 << 14,
            gpu_tick_interval_s=0.01,
        )
        broker = Broker(cfg)
        await broker.start()
        await broker.discovery.perform_heartbeat(0, 60)
        marshal, endpoint = make_marshal(db)
        await marshal.start()

        a = make_client(endpoint, seed=71, topics=[1])
        b = make_client(endpoint, seed=72, topics=[1])
        await a.ensure_initialized()
        await b.ensure_initialized()
        await asyncio.sleep(0.1)
        assert len(broker.connections.users) == 2

        # third connect: refused cleanly (removed from connections, slot
        # accounting intact, no orphaned handler)
        c = make_client(endpoint, seed=73, topics=[1])
        try:
            await asyncio.wait_for(c.ensure_initialized(), timeout=5)
        except Exception:
            pass
        await asyncio.sleep(0.2)
        assert len(broker.connections.users) == 2
        assert len(broker._free_gpu_slots) == 0
        assert len(broker._gpu_user_by_slot) == 2

        # the surviving users still route traffic
        await a.send_broadcast_message([1], b"still-alive")
        msg = await asyncio.wait_for(b.receive_message(), timeout=10)
        assert msg.message == b"still-alive"

        # a departing user frees capacity for a new one
        b.close()
        await asyncio.sleep(0.3)
        d = make_client(endpoint, seed=74, topics=[1])
        await asyncio.wait_for(d.ensure_initialized(), timeout=10)
        await asyncio.sleep(0.1)
        assert len(broker._gpu_user_by_slot) == 2

        c.close()
        await stop_stack([broker], marshal, a, d)

    run(go())


def test_native_ingest_blob_path_cpu(tmp_path):
    """The C++ ingest path end-to-end on the CPU reference engine: native
    pump classification -> blob tick queue -> engine routing -> K7-style
    compact drain -> batched send_rings_batch egress, over real TCP."""
    import uuid as _uuid

    from pushcdn_amd.broker.service import Broker, BrokerConfig
    from pushcdn_amd.client import Client, ClientConfig
    from pushcdn_amd.discovery import BrokerIdentifier
    from pushcdn_amd.marshal import Marshal, MarshalConfig
    from pushcdn_amd.proto.transports.tcp_native import TcpNative

    async def go():
        db = str(tmp_path / f"blob-{_uuid.uuid4().hex}.db")
        broker = Broker(BrokerConfig(
            public_bind_endpoint="127.0.0.1:0",
            public_advertise_endpoint="127.0.0.1:0",
            private_bind_endpoint="127.0.0.1:0",
            private_advertise_endpoint="127.0.0.1:0",
            discovery_endpoint=db,
            keypair=bls.KeyPair.from_seed(1000),
            user_protocol=TcpNative,
            broker_protocol=TcpNative,
            data_plane="gpu",
            gpu_device="cpu",
            gpu_max_users=16,
            gpu_ring_bytes=1 << 16,
            gpu_tick_interval_s=0.005,
        ))
        await broker.start()
        pub = f"127.0.0.1:{broker._user_listener.port}"
        priv = f"127.0.0.1:{broker._broker_listener.port}"
        broker.config.public_advertise_endpoint = pub
        broker.config.private_advertise_endpoint = priv
        broker.identity = BrokerIdentifier(pub, priv)
        broker.discovery.identity = broker.identity
        broker.connections.identity = broker.identity
        await broker.discovery.perform_heartbeat(0, 600)
        marshal = Marshal(MarshalConfig(bind_endpoint="127.0.0.1:0",
                                        discovery_endpoint=db, protocol=TcpNative))
        await marshal.start()
        ep = f"127.0.0.1:{marshal._listener.port}"

        a = Client(ClientConfig(endpoint=ep, keypair=bls.KeyPair.from_seed(81),
                                subscribed_topics=[3], protocol=TcpNative))
        b = Client(ClientConfig(endpoint=ep, keypair=bls.KeyPair.from_seed(82),
                                subscribed_topics=[3], protocol=TcpNative))
        await a.ensure_initialized()
        await b.ensure_initialized()
        await asyncio.sleep(0.3)

        for i in range(40):
            await a.send_broadcast_message([3], f"blob-{i}".encode())
        got = [(await asyncio.wait_for(b.receive_message(), timeout=15)).message
               for _ in range(40)]
        assert got == [f"blob-{i}".encode() for i in range(40)]

        # direct through the blob path
        await a.send_direct_message(b.public_key, b"blob-direct")
        msg = await asyncio.wait_for(b.receive_message(), timeout=15)
        assert msg.message == b"blob-direct"

        # runtime subscribe through the ingest loop (disc 5 inline)
        await b.subscribe([9])
        await asyncio.sleep(0.2)
        await a.send_broadcast_message([9], b"post-sub")
        msg = await asyncio.wait_for(b.receive_message(), timeout=15)
        assert msg.message == b"post-sub"
        # malformed frame disconnects (eviction semantics preserved)
        conn = await a._get_connection()
        pump, cid = conn.pump_handle()
        pump.send(cid, b"\xff" * 24)
        await asyncio.sleep(0.5)
        assert len(broker.connections.users) == 1

        a.close()
        b.close()
        await marshal.close()
        await broker.close()

    run(go())


def test_socket_bench_quic_native_cpu_smoke():
    """The multi-process socket bench's --transport flag drives the full
    stack (marshal + GPU-plane broker on the CPU fallback + subprocess
    clients) over the QUIC-profile native datapath end to end."""
    import json
    import subprocess
    import sys
    from pathlib import Path

    root = Path(__file__).resolve().parent.parent
    out = subprocess.run(
        [sys.executable, "scripts/bench_socket.py", "--subs", "2",
         "--sub-procs", "1", "--senders", "1", "--rate", "100",
         "--seconds", "6", "--transport", "quic-native", "--device", "cpu",
         "--tag", "pysmoke"],
        capture_output=True, text=True, timeout=180, cwd=root)
    lines = [l for l in out.stdout.splitlines() if l.startswith("{")]
    assert lines, out.stdout[-2000:] + out.stderr[-2000:]
    d = json.loads(lines[-1])
    assert d["msgs_sent"] > 0
    # CPU-fallback broker ticks are ~1 s each; just require that the path
    # moved real messages through broker -> QUIC TLS -> subscribers
    assert d["deliveries_counted"] > 0, d
