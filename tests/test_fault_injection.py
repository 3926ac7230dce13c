"""Fault-injection churn, in-process: the reference ships these as live-stack
injector binaries (``bad-broker.rs`` — a fresh random-key broker every 300 ms,
aborted; ``bad-connector.rs`` — connection churn), run under process-compose.
Here the same chaos runs inside the test while a stable broker+client pair
must keep delivering.  (Our CLI also ships the injector binaries themselves:
``pushcdn_amd.cli bad-broker / bad-connector / bad-sender``.)"""

import asyncio

from pushcdn_amd.crypto import bls
from pushcdn_amd.proto import message as m

from test_integration import make_broker, make_client, make_marshal, new_db, run


def test_broker_churn_does_not_disrupt_delivery(tmp_path):
    """Random-key brokers appear and die every ~50 ms (reference
    bad-broker.rs:37-98); they fail cluster auth against the stable broker
    but pollute discovery. Delivery through the stable broker must keep
    working the whole time."""

    async def go():
        db = new_db(tmp_path)
        stable = make_broker(db, tag="stable")
        await stable.start()
        await stable.discovery.perform_heartbeat(0, 60)
        marshal, endpoint = make_marshal(db)
        await marshal.start()
        client = make_client(endpoint, seed=1, topics=[0])
        await client.ensure_initialized()

        async def churn():
            for i in range(10):
                bad = make_broker(db, keypair=bls.KeyPair.from_seed(5000 + i),
                                  tag=f"bad-{i}")
                try:
                    await bad.start()
                    await bad.discovery.perform_heartbeat(10_000, 2)
                    await asyncio.sleep(0.05)
                finally:
                    await bad.close()  # the reference aborts the task

        churn_task = asyncio.get_running_loop().create_task(churn())
        delivered = 0
        while not churn_task.done():
            await client.send_direct_message(client.public_key, b"under-churn")
            msg = await asyncio.wait_for(client.receive_message(), timeout=10)
            assert isinstance(msg, m.Direct) and msg.message == b"under-churn"
            delivered += 1
            await asyncio.sleep(0.02)
        await churn_task
        assert delivered >= 10
        # the stable broker never accepted a bad broker into its mesh
        assert len(stable.connections.brokers) == 0
        client.close()
        await marshal.close()
        await stable.close()

    run(go())


def test_connection_churn_no_leaks(tmp_path):
    """Clients connect and vanish in a loop (reference bad-connector.rs:33-73);
    the broker must not leak user slots and the long-lived client must stay
    functional."""

    async def go():
        db = new_db(tmp_path)
        broker = make_broker(db, tag="churn-target")
        await broker.start()
        await broker.discovery.perform_heartbeat(0, 60)
        marshal, endpoint = make_marshal(db)
        await marshal.start()
        stable = make_client(endpoint, seed=1, topics=[0])
        await stable.ensure_initialized()

        for i in range(15):
            c = make_client(endpoint, seed=100 + i, topics=[0, 1])
            await c.ensure_initialized()
            c.close()
        # give the broker's receive loops a moment to observe the closes
        for _ in range(50):
            if len(broker.connections.users) == 1:
                break
            await asyncio.sleep(0.1)
        assert len(broker.connections.users) == 1  # only the stable client

        await stable.send_direct_message(stable.public_key, b"still-alive")
        msg = await asyncio.wait_for(stable.receive_message(), timeout=10)
        assert msg.message == b"still-alive"
        stable.close()
        await marshal.close()
        await broker.close()

    run(go())


def test_connection_churn_native_transport(tmp_path):
    """The C++ pump under connection churn: clients over tcp-native appear
    and vanish; no fd/slot leaks, survivor stays functional."""
    import uuid as _uuid

    from pushcdn_amd.broker.service import Broker, BrokerConfig
    from pushcdn_amd.client import Client, ClientConfig
    from pushcdn_amd.discovery import BrokerIdentifier
    from pushcdn_amd.marshal import Marshal, MarshalConfig
    from pushcdn_amd.proto.transports.tcp_native import TcpNative

    async def go():
        db = str(tmp_path / f"churn-nat-{_uuid.uuid4().hex}.db")
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

        stable = Client(ClientConfig(endpoint=ep, keypair=bls.KeyPair.from_seed(1),
                                     subscribed_topics=[0], protocol=TcpNative))
        await stable.ensure_initialized()
        for i in range(12):
            c = Client(ClientConfig(endpoint=ep, keypair=bls.KeyPair.from_seed(200 + i),
                                    subscribed_topics=[0], protocol=TcpNative))
            await c.ensure_initialized()
            c.close()
        for _ in range(50):
            if len(broker.connections.users) == 1:
                break
            await asyncio.sleep(0.1)
        assert len(broker.connections.users) == 1
        await stable.send_direct_message(stable.public_key, b"pump-survives")
        msg = await asyncio.wait_for(stable.receive_message(), timeout=10)
        assert msg.message == b"pump-survives"
        stable.close()
        await marshal.close()
        await broker.close()

    run(go())


def test_injector_clis_run(tmp_path):
    """The fault-injector binaries start against a live stack and do their
    thing for a couple of seconds (reference bad-broker.rs / bad-connector.rs
    / bad-sender.rs run under process-compose)."""
    import os
    import signal
    import subprocess
    import sys
    import time

    from pathlib import Path

    REPO = str(Path(__file__).resolve().parent.parent)
    db = str(tmp_path / "inj.db")
    env = dict(os.environ)
    env["PYTHONPATH"] = REPO
    procs = []

    def spawn(*args):
        p = subprocess.Popen([sys.executable, "-m", "pushcdn_amd.cli", *args],
                             cwd=REPO, env=env, stdout=subprocess.PIPE,
                             stderr=subprocess.STDOUT, text=True,
                             start_new_session=True)
        procs.append(p)
        return p

    try:
        spawn("marshal", "-d", db, "-b", "127.0.0.1:42737")
        spawn("broker", "-d", db,
              "--public-bind-endpoint", "127.0.0.1:42738",
              "--public-advertise-endpoint", "127.0.0.1:42738",
              "--private-bind-endpoint", "127.0.0.1:42739",
              "--private-advertise-endpoint", "127.0.0.1:42739")
        time.sleep(4)
        bb = spawn("bad-broker", "-d", db)
        bc = spawn("bad-connector", "-m", "127.0.0.1:42737")
        time.sleep(4)
        # injectors are alive and chaosing; stack processes alive too
        for p in procs:
            assert p.poll() is None, (p.args, p.stdout.read()[-500:])
    finally:
        for p in procs:
            try:
                os.killpg(p.pid, signal.SIGKILL)
            except Exception:
                p.kill()
        for p in procs:
            p.wait(timeout=10)
