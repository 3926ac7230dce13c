"""CLI surface — the reference's binaries as subcommands:

  python -m pushcdn_amd.cli broker         prod broker daemon
  python -m pushcdn_amd.cli marshal        prod marshal daemon
  python -m pushcdn_amd.cli client         demo loop (direct+broadcast to self)
  python -m pushcdn_amd.cli bad-broker     fault injector: churn brokers
  python -m pushcdn_amd.cli bad-connector  churn injector: client connects
  python -m pushcdn_amd.cli bad-sender     load generator: big messages to self

Default ports mirror the reference: broker 1738/1739, marshal 1737
(cdn-broker/src/binaries/broker.rs:24-131, marshal.rs:20-86).
"""

from __future__ import annotations

import argparse
import asyncio
import random


def _transport(name: str):
    """Map a --transport flag to a Protocol class.  tcp-native is the C++
    epoll pump (csrc/net/pump.h) — the production per-connection IO path."""
    if name == "tcp":
        from .proto.transports.tcp import Tcp

        return Tcp
    if name == "tcp-tls":
        from .proto.transports.tcp_tls import TcpTls

        return TcpTls
    if name == "tcp-native":
        from .proto.transports.tcp_native import TcpNative

        return TcpNative
    if name == "quic":
        from .proto.transports.quic import Quic

        return Quic
    if name == "quic-native":
        from .proto.transports.quic import QuicNative

        return QuicNative
    raise SystemExit(f"unknown transport {name!r}")


def _broker_args(p: argparse.ArgumentParser) -> None:
    p.add_argument("-d", "--discovery-endpoint", default="/tmp/pushcdn-discovery.db")
    p.add_argument("--public-bind-endpoint", default="0.0.0.0:1738")
    p.add_argument("--public-advertise-endpoint", default="local_ip:1738")
    p.add_argument("--private-bind-endpoint", default="0.0.0.0:1739")
    p.add_argument("--private-advertise-endpoint", default="local_ip:1739")
    p.add_argument("--metrics-bind-endpoint", default=None)
    p.add_argument("--ca-cert-path", default=None)
    p.add_argument("--ca-key-path", default=None)
    p.add_argument("--key-seed", type=int, default=0, help="BLS cluster keypair seed")
    p.add_argument("--global-memory-pool-size", type=int, default=1 << 30)
    p.add_argument("--data-plane", choices=["host", "gpu"], default="host")
    p.add_argument("--gpu-device", default="cuda:0")
    p.add_argument("--user-transport", choices=["tcp", "tcp-tls", "tcp-native", "quic", "quic-native"],
                   default="tcp", help="user-plane transport (tcp-native = C++ epoll "
                                       "pump; quic[-native] = QUIC-profile over UDP)")
    p.add_argument("--broker-transport", choices=["tcp", "tcp-native"], default="tcp")


def cmd_broker(args) -> None:
    from .broker.service import Broker, BrokerConfig
    from .crypto import bls

    cfg = BrokerConfig(
        public_bind_endpoint=args.public_bind_endpoint,
        public_advertise_endpoint=args.public_advertise_endpoint,
        private_bind_endpoint=args.private_bind_endpoint,
        private_advertise_endpoint=args.private_advertise_endpoint,
        discovery_endpoint=args.discovery_endpoint,
        keypair=bls.KeyPair.from_seed(args.key_seed),
        metrics_bind_endpoint=args.metrics_bind_endpoint,
        global_memory_pool_size=args.global_memory_pool_size,
        ca_cert_path=args.ca_cert_path,
        ca_key_path=args.ca_key_path,
        data_plane=args.data_plane,
        gpu_device=args.gpu_device,
        user_protocol=_transport(args.user_transport),
        broker_protocol=_transport(args.broker_transport),
    )
    asyncio.run(Broker(cfg).run_forever())


def cmd_mesh_broker(args) -> None:
    """GPU broker with the RCCL/xGMI broker-plane — launch one per GPU:
    torchrun --nproc-per-node 8 -m pushcdn_amd.cli mesh-broker ..."""
    import os

    from .broker.mesh_service import MeshBroker
    from .broker.service import BrokerConfig
    from .crypto import bls

    rank = int(os.environ.get("LOCAL_RANK", os.environ.get("RANK", "0")))

    def port_shift(ep: str) -> str:
        host, _, port = ep.rpartition(":")
        return f"{host}:{int(port) + 2 * rank}"

    cfg = BrokerConfig(
        public_bind_endpoint=port_shift(args.public_bind_endpoint),
        public_advertise_endpoint=port_shift(args.public_advertise_endpoint),
        private_bind_endpoint=port_shift(args.private_bind_endpoint),
        private_advertise_endpoint=port_shift(args.private_advertise_endpoint),
        discovery_endpoint=args.discovery_endpoint,
        keypair=bls.KeyPair.from_seed(args.key_seed),
        global_memory_pool_size=args.global_memory_pool_size,
        data_plane="gpu",
        gpu_device=f"cuda:{rank}",
    )
    asyncio.run(MeshBroker(cfg).run_forever())


def cmd_marshal(args) -> None:
    from .marshal import Marshal, MarshalConfig

    cfg = MarshalConfig(
        bind_endpoint=args.bind_endpoint,
        discovery_endpoint=args.discovery_endpoint,
        metrics_bind_endpoint=args.metrics_bind_endpoint,
        global_memory_pool_size=args.global_memory_pool_size,
        ca_cert_path=args.ca_cert_path,
        ca_key_path=args.ca_key_path,
        protocol=_transport(getattr(args, "transport", "tcp")),
    )
    asyncio.run(Marshal(cfg).run_forever())


def cmd_client(args) -> None:
    """Demo loop: direct-to-self + broadcast-to-self, assert the echo, sleep
    (reference cdn-client/src/binaries/client.rs:29-123)."""
    from .client import Client, ClientConfig
    from .crypto import bls
    from .proto import message as m

    async def go() -> None:
        client = Client(
            ClientConfig(
                endpoint=args.marshal_endpoint,
                keypair=bls.KeyPair.from_seed(random.randrange(2**63)),
                subscribed_topics=[0],
                protocol=_transport(getattr(args, "transport", "tcp")),
            )
        )
        while True:
            await client.send_direct_message(client.public_key, b"hello direct")
            msg = await client.receive_message()
            assert isinstance(msg, m.Direct) and msg.message == b"hello direct"
            await client.send_broadcast_message([0], b"hello broadcast")
            msg = await client.receive_message()
            assert isinstance(msg, m.Broadcast) and msg.message == b"hello broadcast"
            print("echo ok", flush=True)
            await asyncio.sleep(5)

    asyncio.run(go())


def cmd_bad_broker(args) -> None:
    """Spawn a fresh random-key broker every 300 ms, then kill it
    (reference bad-broker.rs:37-98)."""
    from .broker.service import Broker, BrokerConfig
    from .crypto import bls

    async def go() -> None:
        n = 0
        while True:
            n += 1
            port_a, port_b = 20000 + (n * 2) % 20000, 20001 + (n * 2) % 20000
            cfg = BrokerConfig(
                public_bind_endpoint=f"127.0.0.1:{port_a}",
                public_advertise_endpoint=f"127.0.0.1:{port_a}",
                private_bind_endpoint=f"127.0.0.1:{port_b}",
                private_advertise_endpoint=f"127.0.0.1:{port_b}",
                discovery_endpoint=args.discovery_endpoint,
                keypair=bls.KeyPair.from_seed(random.randrange(2**63)),
            )
            broker = Broker(cfg)
            try:
                await broker.start()
                await asyncio.sleep(0.3)
            finally:
                await broker.close()

    asyncio.run(go())


def cmd_bad_connector(args) -> None:
    """New client connection every 200 ms (reference bad-connector.rs:33-73)."""
    from .client import Client, ClientConfig
    from .crypto import bls

    async def go() -> None:
        while True:
            client = Client(
                ClientConfig(
                    endpoint=args.marshal_endpoint,
                    keypair=bls.KeyPair.from_seed(random.randrange(2**63)),
                    subscribed_topics=[0],
                )
            )
            try:
                await asyncio.wait_for(client.ensure_initialized(), 5)
            except Exception:
                pass
            client.close()
            await asyncio.sleep(0.2)

    asyncio.run(go())


def cmd_bad_sender(args) -> None:
    """Big direct+broadcast to self in a loop (reference bad-sender.rs:24-105)."""
    from .client import Client, ClientConfig
    from .crypto import bls

    async def go() -> None:
        client = Client(
            ClientConfig(
                endpoint=args.marshal_endpoint,
                keypair=bls.KeyPair.from_seed(random.randrange(2**63)),
                subscribed_topics=[0],
                protocol=_transport(getattr(args, "transport", "tcp")),
            )
        )
        payload = bytes(args.message_size)
        n = 0
        while True:
            await client.send_direct_message(client.public_key, payload)
            await client.receive_message()
            await client.send_broadcast_message([0], payload)
            await client.receive_message()
            n += 2
            if n % 10 == 0:
                print(f"{n} messages echoed", flush=True)

    asyncio.run(go())


def main(argv=None) -> None:
    p = argparse.ArgumentParser(prog="pushcdn")
    sub = p.add_subparsers(dest="cmd", required=True)

    b = sub.add_parser("broker")
    _broker_args(b)
    b.set_defaults(fn=cmd_broker)

    mb = sub.add_parser("mesh-broker")
    _broker_args(mb)
    mb.set_defaults(fn=cmd_mesh_broker)

    ms = sub.add_parser("marshal")
    ms.add_argument("-d", "--discovery-endpoint", default="/tmp/pushcdn-discovery.db")
    ms.add_argument("-b", "--bind-endpoint", default="0.0.0.0:1737")
    ms.add_argument("--metrics-bind-endpoint", default=None)
    ms.add_argument("--ca-cert-path", default=None)
    ms.add_argument("--ca-key-path", default=None)
    ms.add_argument("--global-memory-pool-size", type=int, default=1 << 30)
    ms.add_argument("--transport", choices=["tcp", "tcp-tls", "tcp-native", "quic", "quic-native"], default="tcp")
    ms.set_defaults(fn=cmd_marshal)

    c = sub.add_parser("client")
    c.add_argument("-m", "--marshal-endpoint", default="127.0.0.1:1737")
    c.add_argument("--transport", choices=["tcp", "tcp-tls", "tcp-native", "quic", "quic-native"], default="tcp")
    c.set_defaults(fn=cmd_client)

    bb = sub.add_parser("bad-broker")
    bb.add_argument("-d", "--discovery-endpoint", default="/tmp/pushcdn-discovery.db")
    bb.set_defaults(fn=cmd_bad_broker)

    bc = sub.add_parser("bad-connector")
    bc.add_argument("-m", "--marshal-endpoint", default="127.0.0.1:1737")
    bc.set_defaults(fn=cmd_bad_connector)

    bs = sub.add_parser("bad-sender")
    bs.add_argument("-m", "--marshal-endpoint", default="127.0.0.1:1737")
    bs.add_argument("--message-size", type=int, default=9_000_000)
    bs.set_defaults(fn=cmd_bad_sender)

    args = p.parse_args(argv)
    args.fn(args)


if __name__ == "__main__":
    main()
