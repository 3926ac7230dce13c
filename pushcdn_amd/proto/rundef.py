"""RunDef — the compile-time dependency-injection surface of the reference
(``cdn-proto/src/def.rs:54-168``) as init-time policy objects: a RunDef picks
the signature scheme, the user/broker transports, the discovery backend and
the topic space; every service takes one (or the equivalent explicit config
fields, which these bundles populate).
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Callable, Optional, Type

from .topic import ALL_TOPICS, TEST_TOPIC_SPACE, TopicSpace
from .transports.base import Protocol
from .transports.memory import Memory
from .transports.tcp import Tcp
from .transports.tcp_tls import TcpTls


@dataclass(frozen=True)
class ConnectionDef:
    """Scheme x Protocol x MessageHook (reference def.rs:62-66).  The
    signature scheme is always BLS-over-BN254 (pushcdn_amd.crypto.bls);
    hooks are callables Message -> "process"|"skip"|raise."""

    protocol: Type[Protocol]
    message_hook: Optional[Callable] = None


@dataclass(frozen=True)
class RunDef:
    """Broker x User x Discovery x Topic wiring (reference def.rs:54-59)."""

    broker: ConnectionDef
    user: ConnectionDef
    discovery_endpoint: str
    topic_space: TopicSpace = field(default_factory=lambda: ALL_TOPICS)


def production_run_def(discovery_endpoint: str) -> RunDef:
    """BLS + TCP (brokers) + TCP/TLS (users) + Redis/KeyDB
    (reference ProductionRunDef, def.rs:101-136)."""
    return RunDef(
        broker=ConnectionDef(protocol=Tcp),
        user=ConnectionDef(protocol=TcpTls),
        discovery_endpoint=discovery_endpoint,
        topic_space=ALL_TOPICS,
    )


def quic_run_def(discovery_endpoint: str) -> RunDef:
    """BLS + TCP (brokers) + QUIC-profile (users) + Redis/KeyDB — the
    reference's QUIC-capable generics (def.rs:101-136 are generic over
    Protocol; quic.rs is a first-class user transport).  See
    transports/quic.py for the QUIC-profile scope note."""
    from .transports.quic import Quic

    return RunDef(
        broker=ConnectionDef(protocol=Tcp),
        user=ConnectionDef(protocol=Quic),
        discovery_endpoint=discovery_endpoint,
        topic_space=ALL_TOPICS,
    )


def testing_run_def(discovery_endpoint: str) -> RunDef:
    """Memory transports + embedded SQLite (reference TestingRunDef,
    def.rs:140-159)."""
    return RunDef(
        broker=ConnectionDef(protocol=Memory),
        user=ConnectionDef(protocol=Memory),
        discovery_endpoint=discovery_endpoint,
        topic_space=TEST_TOPIC_SPACE,
    )
