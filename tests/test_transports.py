"""Generic transport conformance test — one shared scenario instantiated per
transport (the reference's tier-2 test, ``protocols/mod.rs:396-481``):
bind -> accept/finalize -> bidirectional send/recv -> soft close."""

import asyncio

import pytest

from pushcdn_amd.proto import message as m
from pushcdn_amd.proto.errors import ConnectionError_
from pushcdn_amd.proto.limiter import Bytes, Limiter
from pushcdn_amd.proto.transports.memory import Memory, gen_testing_connection_pair
from pushcdn_amd.proto.transports.tcp import Tcp


def run(coro):
    return asyncio.run(coro)


async def _conformance(protocol, endpoint):
    limiter = Limiter(global_memory_pool_size=1 << 20)
    listener = await protocol.bind(endpoint, None, None)
    bound_endpoint = endpoint
    if hasattr(listener, "port"):
        bound_endpoint = f"127.0.0.1:{listener.port}"

    async def server():
        unfinalized = await listener.accept()
        conn = await unfinalized.finalize(limiter)
        msg = await conn.recv_message()
        assert msg == m.Direct(b"client", b"hello from client")
        await conn.send_message(m.Direct(b"server", b"hello from server"))
        # raw path too
        await conn.send_message_raw(Bytes(m.serialize(m.Subscribe([1, 2]))))
        await conn.soft_close()

    async def client():
        conn = await protocol.connect(bound_endpoint, True, limiter)
        await conn.send_message(m.Direct(b"client", b"hello from client"))
        reply = await conn.recv_message()
        assert reply == m.Direct(b"server", b"hello from server")
        raw = await conn.recv_message_raw()
        assert m.deserialize(raw.data) == m.Subscribe([1, 2])
        raw.drop()
        await conn.soft_close()

    await asyncio.wait_for(asyncio.gather(server(), client()), timeout=10)
    await listener.close()


def test_tcp_conformance():
    run(_conformance(Tcp, "127.0.0.1:0"))


def test_memory_conformance():
    run(_conformance(Memory, "test-endpoint-1"))


def test_connect_refused():
    async def go():
        limiter = Limiter()
        with pytest.raises(ConnectionError_):
            await Tcp.connect("127.0.0.1:1", False, limiter)
        with pytest.raises(ConnectionError_):
            await Memory.connect("nonexistent", False, limiter)

    run(go())


def test_memory_pair_send_recv():
    async def go():
        limiter = Limiter()
        a, b = gen_testing_connection_pair(limiter)
        await a.send_message(m.Broadcast([0], b"x" * 10))
        got = await b.recv_message()
        assert got == m.Broadcast([0], b"x" * 10)

    run(go())


def test_large_message_over_tcp():
    async def go():
        limiter = Limiter(global_memory_pool_size=1 << 24)
        listener = await Tcp.bind("127.0.0.1:0", None, None)
        endpoint = f"127.0.0.1:{listener.port}"
        payload = bytes(range(256)) * 8192  # 2 MiB

        async def server():
            conn = await (await listener.accept()).finalize(limiter)
            msg = await conn.recv_message()
            assert msg.message == payload
            await conn.soft_close()

        async def client():
            conn = await Tcp.connect(endpoint, True, limiter)
            await conn.send_message(m.Broadcast([1], payload))
            await conn.soft_close()

        await asyncio.wait_for(asyncio.gather(server(), client()), timeout=15)
        await listener.close()

    run(go())


def test_recv_on_dead_connection_errors():
    async def go():
        limiter = Limiter()
        a, b = gen_testing_connection_pair(limiter)
        await a.soft_close()
        with pytest.raises(ConnectionError_):
            await asyncio.wait_for(b.recv_message(), timeout=5)

    run(go())


def test_tcp_tls_conformance():
    from pushcdn_amd.proto.transports.tcp_tls import TcpTls

    run(_conformance(TcpTls, "127.0.0.1:0"))


def test_tls_rejects_untrusted_ca(tmp_path):
    """A client trusting a DIFFERENT CA must fail the handshake."""
    from pushcdn_amd.crypto import tls as tlslib
    from pushcdn_amd.proto.transports.tcp_tls import TcpTls

    async def go():
        limiter = Limiter()
        listener = await TcpTls.bind("127.0.0.1:0", None, None)
        endpoint = f"127.0.0.1:{listener.port}"

        other_ca_cert, _ = tlslib.generate_ca(str(tmp_path / "other-ca"))

        class UntrustingClient(TcpTls):
            ca_cert_path = other_ca_cert

        # the handshake fails before accept() ever yields a connection, so
        # only the client side needs checking
        with pytest.raises(ConnectionError_):
            await asyncio.wait_for(
                UntrustingClient.connect(endpoint, False, limiter), timeout=15
            )
        await listener.close()

    run(go())


def test_metrics_endpoint():
    """Prometheus /metrics endpoint parity (reference metrics.rs:18-39)."""
    import urllib.request

    from pushcdn_amd.utils.metrics import serve_metrics, BYTES_SENT

    async def go():
        server = await serve_metrics("127.0.0.1", 0)
        port = server.sockets[0].getsockname()[1]
        BYTES_SENT.inc(17)

        def fetch():
            with urllib.request.urlopen(f"http://127.0.0.1:{port}/metrics", timeout=5) as r:
                return r.read().decode()

        body = await asyncio.get_running_loop().run_in_executor(None, fetch)
        for metric in ("total_bytes_sent", "total_bytes_recv", "running_latency",
                       "num_users_connected", "num_brokers_connected"):
            assert metric in body, metric
        server.close()
        await server.wait_closed()

    run(go())


def test_oversize_frame_rejected():
    """A frame header larger than MAX_MESSAGE_SIZE must error, not allocate
    (reference protocols/mod.rs:322-325)."""
    import struct

    async def go():
        limiter = Limiter()
        listener = await Tcp.bind("127.0.0.1:0", None, None)
        endpoint_port = listener.port

        async def server():
            conn = await (await listener.accept()).finalize(limiter)
            with pytest.raises(ConnectionError_):
                await asyncio.wait_for(conn.recv_message_raw(), timeout=5)

        async def client():
            reader, writer = await asyncio.open_connection("127.0.0.1", endpoint_port)
            writer.write(struct.pack(">I", m.MAX_MESSAGE_SIZE + 1))
            await writer.drain()
            await asyncio.sleep(0.2)
            writer.close()

        await asyncio.wait_for(asyncio.gather(server(), client()), timeout=15)
        await listener.close()

    run(go())


def test_limiter_backpressure_on_connection():
    """With a tiny global pool, a second message blocks until the first's
    allocation is released (reference protocols/mod.rs:328)."""

    async def go():
        limiter = Limiter(global_memory_pool_size=64)
        a, b = gen_testing_connection_pair(limiter)
        await a.send_message_raw(Bytes(b"x" * 60))
        await a.send_message_raw(Bytes(b"y" * 60))
        first = await asyncio.wait_for(b.recv_message_raw(), timeout=5)
        # second recv must block while the first allocation is held
        second_task = asyncio.ensure_future(b.recv_message_raw())
        await asyncio.sleep(0.1)
        assert not second_task.done(), "backpressure did not hold"
        first.drop()  # release 60 bytes
        second = await asyncio.wait_for(second_task, timeout=5)
        second.drop()

    run(go())


def test_tcp_native_conformance():
    """The C++ epoll pump transport passes the same conformance contract
    as the asyncio transports."""
    from pushcdn_amd.proto.transports.tcp_native import TcpNative

    run(_conformance(TcpNative, "127.0.0.1:0"))


def test_tcp_native_large_and_burst():
    from pushcdn_amd.proto.transports.tcp_native import TcpNative

    async def go():
        limiter = Limiter(global_memory_pool_size=1 << 30)
        listener = await TcpNative.bind("127.0.0.1:0", None, None)
        endpoint = f"127.0.0.1:{listener.port}"
        payload = bytes(range(256)) * (4 << 12)  # 4 MiB patterned

        async def server():
            conn = await (await listener.accept()).finalize(limiter)
            # burst of 200 small frames arrives intact and in order
            for i in range(200):
                msg = await conn.recv_message()
                assert msg.message == f"burst-{i}".encode()
            msg = await conn.recv_message()
            assert msg.message == payload  # 4 MiB frame reassembled
            await conn.send_message(m.Direct(b"s", b"done"))
            await conn.soft_close()

        async def client():
            conn = await TcpNative.connect(endpoint, True, limiter)
            for i in range(200):
                await conn.send_message(m.Direct(b"c", f"burst-{i}".encode()))
            await conn.send_message(m.Direct(b"c", payload))
            reply = await conn.recv_message()
            assert reply.message == b"done"
            await conn.soft_close()

        await asyncio.wait_for(asyncio.gather(server(), client()), timeout=30)
        await listener.close()

    run(go())


def test_tcp_native_peer_disappears():
    from pushcdn_amd.proto.transports.tcp_native import TcpNative
    from pushcdn_amd.proto.errors import ConnectionError_

    async def go():
        limiter = Limiter(global_memory_pool_size=1 << 20)
        listener = await TcpNative.bind("127.0.0.1:0", None, None)
        endpoint = f"127.0.0.1:{listener.port}"

        async def server():
            conn = await (await listener.accept()).finalize(limiter)
            conn.close()  # hard close, no flush

        async def client():
            conn = await TcpNative.connect(endpoint, True, limiter)
            await asyncio.sleep(0.2)
            with pytest.raises(ConnectionError_):
                await asyncio.wait_for(conn.recv_message(), timeout=5)
            conn.close()

        await asyncio.wait_for(asyncio.gather(server(), client()), timeout=15)
        await listener.close()

    run(go())


def test_full_stack_over_native_tcp(tmp_path):
    """marshal + broker + 2 clients end-to-end with the C++ pump transport
    on every hop (user plane, broker plane, marshal)."""
    import uuid as _uuid

    from pushcdn_amd.broker.service import Broker, BrokerConfig
    from pushcdn_amd.client import Client, ClientConfig
    from pushcdn_amd.crypto import bls
    from pushcdn_amd.discovery import BrokerIdentifier
    from pushcdn_amd.marshal import Marshal, MarshalConfig
    from pushcdn_amd.proto.transports.tcp_native import TcpNative

    async def go():
        db = str(tmp_path / f"native-{_uuid.uuid4().hex}.db")
        broker = Broker(BrokerConfig(
            public_bind_endpoint="127.0.0.1:0",
            public_advertise_endpoint="127.0.0.1:0",
            private_bind_endpoint="127.0.0.1:0",
            private_advertise_endpoint="127.0.0.1:0",
            discovery_endpoint=db,
            keypair=bls.KeyPair.from_seed(1000),
            user_protocol=TcpNative,
            broker_protocol=TcpNative,
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

        alice = Client(ClientConfig(endpoint=ep, keypair=bls.KeyPair.from_seed(1),
                                    subscribed_topics=[0], protocol=TcpNative))
        bob = Client(ClientConfig(endpoint=ep, keypair=bls.KeyPair.from_seed(2),
                                  subscribed_topics=[0], protocol=TcpNative))
        await alice.ensure_initialized()
        await bob.ensure_initialized()
        await asyncio.sleep(0.2)

        await alice.send_broadcast_message([0], b"native-broadcast")
        msg = await asyncio.wait_for(bob.receive_message(), timeout=10)
        assert msg.message == b"native-broadcast"
        await bob.send_direct_message(alice.public_key, b"native-direct")
        got = await asyncio.wait_for(alice.receive_message(), timeout=10)
        while not (hasattr(got, "recipient")):  # skip alice's own broadcast echo
            got = await asyncio.wait_for(alice.receive_message(), timeout=10)
        assert got.message == b"native-direct"

        alice.close()
        bob.close()
        await marshal.close()
        await broker.close()

    run(go())


def test_gpu_engine_drain_over_native_tcp(tmp_path):
    """data_plane=gpu (CPU reference engine here) + native TCP user plane:
    the egress drain rides pump.send_ring — the pump parses the drained
    ring records and enqueues each wire frame in C++."""
    import uuid as _uuid

    from pushcdn_amd.broker.service import Broker, BrokerConfig
    from pushcdn_amd.client import Client, ClientConfig
    from pushcdn_amd.crypto import bls
    from pushcdn_amd.discovery import BrokerIdentifier
    from pushcdn_amd.marshal import Marshal, MarshalConfig
    from pushcdn_amd.proto.transports.tcp_native import TcpNative

    async def go():
        db = str(tmp_path / f"natgpu-{_uuid.uuid4().hex}.db")
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
            gpu_max_users=32,
            gpu_ring_bytes=1 << 14,
            gpu_tick_interval_s=0.01,
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

        alice = Client(ClientConfig(endpoint=ep, keypair=bls.KeyPair.from_seed(1),
                                    subscribed_topics=[5], protocol=TcpNative))
        bob = Client(ClientConfig(endpoint=ep, keypair=bls.KeyPair.from_seed(2),
                                  subscribed_topics=[5], protocol=TcpNative))
        await alice.ensure_initialized()
        await bob.ensure_initialized()
        await asyncio.sleep(0.2)

        # burst: every delivery drains through send_ring, order preserved
        for i in range(20):
            await alice.send_broadcast_message([5], f"ring-{i}".encode())
        for i in range(20):
            msg = await asyncio.wait_for(bob.receive_message(), timeout=10)
            assert msg.message == f"ring-{i}".encode()

        alice.close()
        bob.close()
        await marshal.close()
        await broker.close()

    run(go())


def test_two_broker_mesh_over_native_tcp(tmp_path):
    """Two brokers whose BROKER plane (mutual auth, CRDT syncs, forwarded
    traffic) also rides the C++ pump; cross-broker broadcast + direct."""
    import uuid as _uuid

    from pushcdn_amd.broker.service import Broker, BrokerConfig
    from pushcdn_amd.client import Client, ClientConfig
    from pushcdn_amd.crypto import bls
    from pushcdn_amd.discovery import BrokerIdentifier
    from pushcdn_amd.marshal import Marshal, MarshalConfig
    from pushcdn_amd.proto.transports.tcp_native import TcpNative

    async def mk_broker(db, kp):
        b = Broker(BrokerConfig(
            public_bind_endpoint="127.0.0.1:0",
            public_advertise_endpoint="127.0.0.1:0",
            private_bind_endpoint="127.0.0.1:0",
            private_advertise_endpoint="127.0.0.1:0",
            discovery_endpoint=db,
            keypair=kp,
            user_protocol=TcpNative,
            broker_protocol=TcpNative,
            heartbeat_interval_s=0.2,
            sync_interval_s=0.2,
        ))
        await b.start()
        pub = f"127.0.0.1:{b._user_listener.port}"
        priv = f"127.0.0.1:{b._broker_listener.port}"
        b.config.public_advertise_endpoint = pub
        b.config.private_advertise_endpoint = priv
        b.identity = BrokerIdentifier(pub, priv)
        b.discovery.identity = b.identity
        b.connections.identity = b.identity
        await b.discovery.perform_heartbeat(0, 600)
        return b

    async def go():
        db = str(tmp_path / f"mesh-nat-{_uuid.uuid4().hex}.db")
        kp = bls.KeyPair.from_seed(1000)
        b1 = await mk_broker(db, kp)
        b2 = await mk_broker(db, kp)
        await asyncio.sleep(0.8)  # heartbeats dial the mesh
        assert len(b1.connections.brokers) == 1
        assert len(b2.connections.brokers) == 1

        marshal = Marshal(MarshalConfig(bind_endpoint="127.0.0.1:0",
                                        discovery_endpoint=db, protocol=TcpNative))
        await marshal.start()
        ep = f"127.0.0.1:{marshal._listener.port}"

        # steer alice -> b1, bob -> b2 with artificial load reports
        await b1.discovery.perform_heartbeat(0, 60)
        await b2.discovery.perform_heartbeat(10, 60)
        alice = Client(ClientConfig(endpoint=ep, keypair=bls.KeyPair.from_seed(11),
                                    subscribed_topics=[3], protocol=TcpNative))
        await alice.ensure_initialized()
        # steering races the brokers' own 0.2 s heartbeats — retry until
        # bob lands on b2
        bob = None
        for attempt in range(20):
            await b1.discovery.perform_heartbeat(10, 60)
            await b2.discovery.perform_heartbeat(0, 60)
            bob = Client(ClientConfig(endpoint=ep, keypair=bls.KeyPair.from_seed(12),
                                      subscribed_topics=[3], protocol=TcpNative))
            await bob.ensure_initialized()
            await asyncio.sleep(0.1)
            if len(b2.connections.users) == 1:
                break
            bob.close()
            await asyncio.sleep(0.2)
        assert len(b1.connections.users) == 1 and len(b2.connections.users) == 1
        await asyncio.sleep(0.6)  # topic/user syncs over the pump

        await alice.send_broadcast_message([3], b"pump-mesh-broadcast")
        msg = await asyncio.wait_for(bob.receive_message(), timeout=10)
        assert msg.message == b"pump-mesh-broadcast"
        await alice.send_direct_message(bob.public_key, b"pump-mesh-direct")
        msg = await asyncio.wait_for(bob.receive_message(), timeout=10)
        assert msg.message == b"pump-mesh-direct"

        alice.close()
        bob.close()
        await marshal.close()
        await b1.close()
        await b2.close()

    run(go())


def test_tls_explicit_ca_trust(tmp_path):
    """A client trusting an EXPLICIT CA file (not the process-local CA)
    handshakes with a server using a leaf minted from that CA
    (reference tls.rs:100-126 load_ca + root store wiring)."""
    from pushcdn_amd.crypto.tls import generate_ca
    from pushcdn_amd.proto.transports.tcp_tls import TcpTls

    ca_cert, ca_key = generate_ca(str(tmp_path / "ca"))

    class TlsWithCa(TcpTls):  # RunDef-style CA wiring via class attrs
        ca_cert_path = ca_cert
        ca_key_path = ca_key

    async def go():
        limiter = Limiter(global_memory_pool_size=1 << 20)
        listener = await TlsWithCa.bind("127.0.0.1:0", None, None)
        endpoint = f"127.0.0.1:{listener.port}"

        async def server():
            conn = await (await listener.accept()).finalize(limiter)
            msg = await conn.recv_message()
            await conn.send_message(m.Direct(b"s", msg.message[::-1]))
            await conn.soft_close()

        async def client():
            # use_local_authority=False -> trust the EXPLICIT CA file
            conn = await TlsWithCa.connect(endpoint, False, limiter)
            await conn.send_message(m.Direct(b"c", b"abc"))
            reply = await conn.recv_message()
            assert reply.message == b"cba"
            await conn.soft_close()

        await asyncio.wait_for(asyncio.gather(server(), client()), timeout=15)
        await listener.close()

    run(go())


def test_pump_many_connections_stress():
    """One pump, 60 concurrent native-loopback connections, interleaved
    bursts — every frame arrives intact, in per-connection order, no
    cross-connection bleed."""
    from pushcdn_amd.proto.transports.tcp_native import TcpNative

    async def go():
        limiter = Limiter(global_memory_pool_size=1 << 28)
        listener = await TcpNative.bind("127.0.0.1:0", None, None)
        endpoint = f"127.0.0.1:{listener.port}"
        N, MSGS = 60, 40

        async def server():
            async def serve_one(idx):
                conn = await (await listener.accept()).finalize(limiter)
                ident = (await conn.recv_message()).message  # b"conn-<i>"
                for j in range(MSGS):
                    msg = await conn.recv_message()
                    assert msg.message == ident + b":%d" % j
                await conn.send_message(m.Direct(b"s", ident + b":done"))
                await conn.soft_close()

            await asyncio.gather(*[serve_one(i) for i in range(N)])

        async def client(i):
            conn = await TcpNative.connect(endpoint, True, limiter)
            ident = b"conn-%d" % i
            await conn.send_message(m.Direct(b"c", ident))
            for j in range(MSGS):
                await conn.send_message(m.Direct(b"c", ident + b":%d" % j))
                if j % 7 == 0:
                    await asyncio.sleep(0)  # interleave with other clients
            reply = await conn.recv_message()
            assert reply.message == ident + b":done"
            await conn.soft_close()

        await asyncio.wait_for(
            asyncio.gather(server(), *[client(i) for i in range(N)]), timeout=60)
        await listener.close()

    run(go())


def test_leaf_cert_san_is_espresso(tmp_path):
    """Leaf certificates carry the fixed SAN/SNI name "espresso" — wire
    parity with the reference's pinned name (tls.rs:63-71, tcp_tls.rs:91-95:
    clients always connect with server_hostname=espresso)."""
    import subprocess

    from pushcdn_amd.crypto.tls import generate_ca, generate_cert_from_ca

    ca_cert, ca_key = generate_ca(str(tmp_path / "ca"))
    cert, _key = generate_cert_from_ca(ca_cert, ca_key, str(tmp_path / "leaf"))
    text = subprocess.run(["openssl", "x509", "-in", cert, "-noout", "-text"],
                          capture_output=True, text=True, check=True).stdout
    assert "DNS:espresso" in text
    assert "CN = espresso" in text or "CN=espresso" in text


def test_queued_messages_drain_after_peer_close():
    """Messages already received before the peer closed are still readable;
    only AFTER the queue drains does recv fail (base.py recv_message_raw's
    reader-death branch — the reference's channel close semantics)."""
    async def go():
        limiter = Limiter(global_memory_pool_size=1 << 20)
        listener = await Tcp.bind("127.0.0.1:0", None, None)
        endpoint = f"127.0.0.1:{listener.port}"

        got = []

        async def server():
            conn = await (await listener.accept()).finalize(limiter)
            await conn.send_message(m.Direct(b"s", b"one"))
            await conn.send_message(m.Direct(b"s", b"two"))
            await conn.soft_close()  # flush, then the peer sees EOF

        async def client():
            conn = await Tcp.connect(endpoint, True, limiter)
            await asyncio.sleep(0.3)  # let both frames land and the reader die
            got.append((await conn.recv_message()).message)
            got.append((await conn.recv_message()).message)
            try:
                await asyncio.wait_for(conn.recv_message(), timeout=2)
                raise AssertionError("expected ConnectionError after drain")
            except ConnectionError_:
                pass
            conn.close()

        await asyncio.wait_for(asyncio.gather(server(), client()), timeout=15)
        await listener.close()
        assert got == [b"one", b"two"]

    run(go())


def test_quic_conformance():
    """The QUIC-profile transport (reliable TLS stream over UDP) passes the
    same conformance contract (reference quic.rs:279-298)."""
    from pushcdn_amd.proto.transports.quic import Quic

    run(_conformance(Quic, "127.0.0.1:0"))


def test_quic_rejects_untrusted_ca(tmp_path):
    """A client with a DIFFERENT trust root must refuse the server's cert —
    same trust model as TcpTls (reference tls.rs:100-126)."""
    from pushcdn_amd.crypto import tls as tlslib
    from pushcdn_amd.proto.transports.quic import Quic

    async def go():
        limiter = Limiter(global_memory_pool_size=1 << 20)
        listener = await Quic.bind("127.0.0.1:0", None, None)
        ep = f"127.0.0.1:{listener.port}"

        async def server():
            try:
                unf = await asyncio.wait_for(listener.accept(), 8)
                await unf.finalize(limiter)
            except Exception:
                pass

        st = asyncio.get_running_loop().create_task(server())
        other_ca_cert, other_ca_key = tlslib.generate_ca(str(tmp_path / "otherca"))

        class UntrustingClient(Quic):
            ca_cert_path = other_ca_cert
            ca_key_path = other_ca_key

        from pushcdn_amd.proto.errors import ConnectionError_
        try:
            await UntrustingClient.connect(ep, False, limiter)
            raise AssertionError("handshake against the wrong CA succeeded")
        except ConnectionError_:
            pass
        st.cancel()
        await listener.close()

    run(go())


def test_quic_packet_loss_recovery():
    """Retransmission: drop 20% of datagrams in each direction and the
    stream still delivers everything in order."""
    import random

    from pushcdn_amd.proto.transports import quic as quicmod

    async def go():
        rng = random.Random(42)
        orig = quicmod._QuicEndpoint.send_pkt

        def lossy(self, addr, ptype, cid, payload):
            # never drop the handshake/teardown control packets, only data
            if ptype in (quicmod.PKT_STREAM, quicmod.PKT_ACK) and rng.random() < 0.2:
                return
            orig(self, addr, ptype, cid, payload)

        quicmod._QuicEndpoint.send_pkt = lossy
        try:
            limiter = Limiter(global_memory_pool_size=1 << 22)
            listener = await quicmod.Quic.bind("127.0.0.1:0", None, None)
            ep = f"127.0.0.1:{listener.port}"

            async def server():
                unf = await listener.accept()
                conn = await unf.finalize(limiter)
                for i in range(20):
                    msg = await conn.recv_message()
                    assert msg.message == bytes([i]) * 3000, f"msg {i}"
                await conn.send_message(m.Direct(b"s", b"all-received"))
                await conn.soft_close()

            st = asyncio.get_running_loop().create_task(server())
            conn = await quicmod.Quic.connect(ep, True, limiter)
            for i in range(20):
                await conn.send_message(m.Direct(b"c", bytes([i]) * 3000))
            reply = await asyncio.wait_for(conn.recv_message(), 30)
            assert reply.message == b"all-received"
            await conn.soft_close()
            await st
            await listener.close()
        finally:
            quicmod._QuicEndpoint.send_pkt = orig

    run(go())


def test_quic_full_service_stack(tmp_path):
    """marshal + broker + two clients entirely over the QUIC transport:
    auth/permits, subscribe, broadcast and direct round-trips."""
    import uuid as _uuid

    from pushcdn_amd.broker.service import Broker, BrokerConfig
    from pushcdn_amd.client import Client, ClientConfig
    from pushcdn_amd.crypto import bls
    from pushcdn_amd.discovery import BrokerIdentifier
    from pushcdn_amd.marshal import Marshal, MarshalConfig
    from pushcdn_amd.proto.transports.quic import Quic

    async def go():
        db = str(tmp_path / f"quic-{_uuid.uuid4().hex}.db")
        broker = Broker(BrokerConfig(
            public_bind_endpoint="127.0.0.1:0",
            public_advertise_endpoint="127.0.0.1:0",
            private_bind_endpoint="127.0.0.1:0",
            private_advertise_endpoint="127.0.0.1:0",
            discovery_endpoint=db,
            keypair=bls.KeyPair.from_seed(1000),
            user_protocol=Quic,
            broker_protocol=Quic,
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
                                        discovery_endpoint=db, protocol=Quic))
        await marshal.start()
        ep = f"127.0.0.1:{marshal._listener.port}"

        alice = Client(ClientConfig(endpoint=ep, keypair=bls.KeyPair.from_seed(91),
                                    subscribed_topics=[2], protocol=Quic))
        bob = Client(ClientConfig(endpoint=ep, keypair=bls.KeyPair.from_seed(92),
                                  subscribed_topics=[2], protocol=Quic))
        await alice.ensure_initialized()
        await bob.ensure_initialized()
        await asyncio.sleep(0.2)

        await alice.send_broadcast_message([2], b"quic-broadcast")
        msg = await asyncio.wait_for(bob.receive_message(), timeout=10)
        assert msg.message == b"quic-broadcast"
        await bob.send_direct_message(alice.public_key, b"quic-direct")
        got = await asyncio.wait_for(alice.receive_message(), timeout=10)
        while got.message == b"quic-broadcast":  # alice's own echo first
            got = await asyncio.wait_for(alice.receive_message(), timeout=10)
        assert got.message == b"quic-direct"

        alice.close()
        bob.close()
        await marshal.close()
        await broker.close()

    run(go())


def test_quic_packet_chaos_drop_dup_reorder():
    """Reliability under combined datagram loss, duplication, AND
    reordering: random per-packet chaos on both directions; the stream
    must still deliver every message, in order, byte-exact."""
    import random

    from pushcdn_amd.proto.transports import quic as quicmod

    async def go():
        rng = random.Random(1234)
        orig = quicmod._QuicEndpoint.send_pkt
        delayed = []

        def chaotic(self, addr, ptype, cid, payload):
            if ptype in (quicmod.PKT_STREAM, quicmod.PKT_ACK):
                r = rng.random()
                if r < 0.10:
                    return  # drop
                if r < 0.18:
                    orig(self, addr, ptype, cid, payload)  # duplicate
                if r < 0.30:
                    # delay/reorder: hold the packet, release a previous one
                    delayed.append((self, addr, ptype, cid, payload))
                    if len(delayed) > 4:
                        args = delayed.pop(rng.randrange(len(delayed)))
                        orig(*args)
                    return
            orig(self, addr, ptype, cid, payload)

        quicmod._QuicEndpoint.send_pkt = chaotic
        try:
            limiter = Limiter(global_memory_pool_size=1 << 22)
            listener = await quicmod.Quic.bind("127.0.0.1:0", None, None)
            ep = f"127.0.0.1:{listener.port}"
            rng2 = random.Random(99)
            payloads = [bytes(rng2.randrange(256) for _ in range(rng2.randrange(1, 5000)))
                        for _ in range(25)]

            async def server():
                unf = await listener.accept()
                conn = await unf.finalize(limiter)
                for i, want in enumerate(payloads):
                    msg = await conn.recv_message()
                    assert msg.message == want, f"msg {i} corrupted"
                await conn.send_message(m.Direct(b"s", b"chaos-ok"))
                await conn.soft_close()

            st = asyncio.get_running_loop().create_task(server())
            conn = await quicmod.Quic.connect(ep, True, limiter)
            for p in payloads:
                await conn.send_message(m.Direct(b"c", p))
            reply = await asyncio.wait_for(conn.recv_message(), 60)
            assert reply.message == b"chaos-ok"
            await conn.soft_close()
            await st
            await listener.close()
        finally:
            quicmod._QuicEndpoint.send_pkt = orig

    run(go())


def test_quic_native_conformance():
    """The native-datapath QUIC profile (C++ UdpPump reliability layer,
    csrc/net/udp_stream.h) passes the same conformance contract."""
    from pushcdn_amd.proto.transports.quic import QuicNative

    run(_conformance(QuicNative, "127.0.0.1:0"))


def test_quic_native_interop_with_python_endpoint():
    """The native and pure-Python endpoints speak the SAME profile wire
    format: native server <-> python client and python server <-> native
    client both carry framed messages over the TLS stream."""
    from pushcdn_amd.proto import message as m
    from pushcdn_amd.proto.transports.quic import Quic, QuicNative

    async def one_pair(server_proto, client_proto, tag):
        limiter = Limiter(global_memory_pool_size=1 << 28)
        listener = await server_proto.bind("127.0.0.1:0", None, None)
        endpoint = f"127.0.0.1:{listener.port}"

        async def server():
            conn = await (await listener.accept()).finalize(limiter)
            msg = await conn.recv_message()
            await conn.send_message(m.Broadcast([1], msg.message + b"-echo"))
            await conn.soft_close()

        async def client():
            conn = await client_proto.connect(endpoint, True, limiter)
            await conn.send_message(m.Direct(b"u", tag))
            echo = await conn.recv_message()
            assert echo.message == tag + b"-echo"
            await conn.soft_close()

        await asyncio.wait_for(asyncio.gather(server(), client()), timeout=20)
        await listener.close()

    async def go():
        await one_pair(QuicNative, Quic, b"native-server")
        await one_pair(Quic, QuicNative, b"python-server")

    run(go())


def test_quic_native_packet_loss_recovery():
    """15% deterministic datagram loss in BOTH directions (the pump's
    debug_set_loss LCG hook): a multi-megabyte framed transfer still
    completes intact through retransmission."""
    import os as _os

    from pushcdn_amd.proto import message as m
    from pushcdn_amd.proto.transports import quic as quicmod

    async def go():
        limiter = Limiter(global_memory_pool_size=1 << 28)
        listener = await quicmod.QuicNative.bind("127.0.0.1:0", None, None)
        listener._ep.pump.debug_set_loss(150)
        endpoint = f"127.0.0.1:{listener.port}"
        payload = _os.urandom(3 << 20)

        async def server():
            conn = await (await listener.accept()).finalize(limiter)
            got = await conn.recv_message()
            assert got.message == payload
            await conn.send_message(m.Direct(b"s", b"ok"))
            await conn.soft_close()

        async def client():
            conn = await quicmod.QuicNative.connect(endpoint, True, limiter)
            # reach through to the client pump for symmetric loss
            for c in conn_pumps():
                c.debug_set_loss(150)
            await conn.send_message(m.Broadcast([1], payload))
            assert (await conn.recv_message()).message == b"ok"
            await conn.soft_close()

        def conn_pumps():
            # every live native endpoint except the listener's
            eps = [listener._ep.pump]
            return [p for p in _NATIVE_PUMPS if p is not listener._ep.pump]

        # track client endpoints created during this test
        _NATIVE_PUMPS = []
        orig_init = quicmod._NativeEndpoint.__init__

        def patched(self, server):
            orig_init(self, server)
            _NATIVE_PUMPS.append(self.pump)

        quicmod._NativeEndpoint.__init__ = patched
        try:
            await asyncio.wait_for(asyncio.gather(server(), client()),
                                   timeout=60)
        finally:
            quicmod._NativeEndpoint.__init__ = orig_init
        await listener.close()

    run(go())


def test_quic_native_large_transfer_integrity():
    """A 32 MiB framed message survives the windowed/retransmitting stream
    byte-for-byte (exercises deferred trim, watermark pacing, reordering)."""
    import hashlib
    import os as _os

    from pushcdn_amd.proto import message as m
    from pushcdn_amd.proto.transports.quic import QuicNative

    async def go():
        limiter = Limiter(global_memory_pool_size=1 << 30)
        listener = await QuicNative.bind("127.0.0.1:0", None, None)
        endpoint = f"127.0.0.1:{listener.port}"
        payload = _os.urandom(32 << 20)
        digest = hashlib.sha256(payload).digest()

        async def server():
            conn = await (await listener.accept()).finalize(limiter)
            got = await conn.recv_message()
            assert hashlib.sha256(got.message).digest() == digest
            await conn.send_message(m.Direct(b"s", b"ok"))
            await conn.soft_close()

        async def client():
            conn = await QuicNative.connect(endpoint, True, limiter)
            await conn.send_message(m.Broadcast([1], payload))
            assert (await conn.recv_message()).message == b"ok"
            await conn.soft_close()

        await asyncio.wait_for(asyncio.gather(server(), client()), timeout=60)
        await listener.close()

    run(go())


def test_quic_native_full_service_stack(tmp_path):
    """marshal + broker + two clients over the NATIVE QUIC datapath
    (C++ UdpPump reliability; same auth/permit/subscribe/route flows as
    test_quic_full_service_stack)."""
    import uuid as _uuid

    from pushcdn_amd.broker.service import Broker, BrokerConfig
    from pushcdn_amd.client import Client, ClientConfig
    from pushcdn_amd.crypto import bls
    from pushcdn_amd.discovery import BrokerIdentifier
    from pushcdn_amd.marshal import Marshal, MarshalConfig
    from pushcdn_amd.proto.transports.quic import QuicNative

    async def go():
        db = str(tmp_path / f"quicn-{_uuid.uuid4().hex}.db")
        broker = Broker(BrokerConfig(
            public_bind_endpoint="127.0.0.1:0",
            public_advertise_endpoint="127.0.0.1:0",
            private_bind_endpoint="127.0.0.1:0",
            private_advertise_endpoint="127.0.0.1:0",
            discovery_endpoint=db,
            keypair=bls.KeyPair.from_seed(1001),
            user_protocol=QuicNative,
            broker_protocol=QuicNative,
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
                                        discovery_endpoint=db,
                                        protocol=QuicNative))
        await marshal.start()
        ep = f"127.0.0.1:{marshal._listener.port}"

        alice = Client(ClientConfig(endpoint=ep, keypair=bls.KeyPair.from_seed(93),
                                    subscribed_topics=[3], protocol=QuicNative))
        bob = Client(ClientConfig(endpoint=ep, keypair=bls.KeyPair.from_seed(94),
                                  subscribed_topics=[3], protocol=QuicNative))
        await alice.ensure_initialized()
        await bob.ensure_initialized()
        await asyncio.sleep(0.2)

        await alice.send_broadcast_message([3], b"quicn-broadcast")
        msg = await asyncio.wait_for(bob.receive_message(), timeout=10)
        assert msg.message == b"quicn-broadcast"
        await bob.send_direct_message(alice.public_key, b"quicn-direct")
        got = await asyncio.wait_for(alice.receive_message(), timeout=10)
        while got.message == b"quicn-broadcast":
            got = await asyncio.wait_for(alice.receive_message(), timeout=10)
        assert got.message == b"quicn-direct"

        alice.close()
        bob.close()
        await marshal.close()
        await broker.close()

    run(go())


def test_quic_native_reconnect_churn_no_leaks():
    """30 connect/close cycles against one listener: every client endpoint
    (UDP socket + pump thread) must be torn down — the reconnect path must
    not accumulate per-attempt resources (regression for the broker-auth
    failure leak)."""
    import gc
    import threading

    from pushcdn_amd.proto import message as m
    from pushcdn_amd.proto.transports.quic import QuicNative

    async def go():
        limiter = Limiter(global_memory_pool_size=1 << 28)
        listener = await QuicNative.bind("127.0.0.1:0", None, None)
        endpoint = f"127.0.0.1:{listener.port}"

        async def server():
            while True:
                unf = await listener.accept()
                conn = await unf.finalize(limiter)
                msg = await conn.recv_message()
                await conn.send_message(msg)
                await conn.soft_close()

        st = asyncio.create_task(server())
        base_threads = threading.active_count()
        for i in range(30):
            conn = await QuicNative.connect(endpoint, True, limiter)
            await conn.send_message(m.Direct(b"u", b"ping-%d" % i))
            echo = await conn.recv_message()
            assert echo.message == b"ping-%d" % i
            await conn.soft_close()
        st.cancel()
        await listener.close()
        # teardown is async-ish (pump threads join in stop()); allow a beat
        await asyncio.sleep(0.3)
        gc.collect()
        leaked = threading.active_count() - base_threads
        assert leaked <= 2, f"{leaked} lingering threads after 30 cycles"

    run(go())
