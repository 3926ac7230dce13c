"""The broker service (reference ``cdn-broker/src/lib.rs`` + ``tasks/``).

``Broker.start()`` runs the same five long-lived tasks as the reference
(lib.rs:269-319) — heartbeat, sync, whitelist, user listener, broker
listener (+ optional metrics server) — and fails fast if any dies.

Routing runs in one of two data planes:
  - host: per-message routing against the Connections tables (the reference's
    path, used for control-plane scale and CPU tests)
  - gpu: incoming user messages are batched per tick through the CDNA4
    kernel pipeline (GpuBrokerEngine) and egress rings are drained back to
    the per-user connections — the MI355X-native hot path.
"""

from __future__ import annotations

import asyncio
import random
import socket
from dataclasses import dataclass, field
from typing import Callable, Dict, List, Optional, Sequence

from ..auth import BrokerAuth
from ..auth.broker import BrokerAuth as _BA
from ..crypto import bls
from ..discovery import BrokerIdentifier, new_discovery_client
from ..proto import message as m
from ..proto.errors import ConnectionError_, TopicError
from ..proto.limiter import Bytes, Limiter
from ..proto.topic import TopicSpace, ALL_TOPICS
from ..proto.transports.base import Connection
from ..utils.log import get_logger, ident
from .connections import Connections

log = get_logger("broker")

HEARTBEAT_INTERVAL_S = 10.0
HEARTBEAT_EXPIRY_S = 60.0
SYNC_INTERVAL_S = 10.0
WHITELIST_INTERVAL_S = 60.0

# ProcessMessage / SkipMessage hook results (reference def.rs:69-97)
PROCESS_MESSAGE = "process"
SKIP_MESSAGE = "skip"


def resolve_local_ip(endpoint: str) -> str:
    """'local_ip' substitution in advertise endpoints (reference lib.rs:157-168)."""
    if not endpoint.startswith("local_ip"):
        return endpoint
    try:
        s = socket.socket(socket.AF_INET, socket.SOCK_DGRAM)
        s.connect(("8.8.8.8", 80))
        ip = s.getsockname()[0]
        s.close()
    except OSError:
        ip = "127.0.0.1"
    return endpoint.replace("local_ip", ip)


@dataclass
class BrokerConfig:
    public_bind_endpoint: str
    public_advertise_endpoint: str
    private_bind_endpoint: str
    private_advertise_endpoint: str
    discovery_endpoint: str = ""
    keypair: Optional[bls.KeyPair] = None
    metrics_bind_endpoint: Optional[str] = None
    global_memory_pool_size: Optional[int] = 1 << 30
    user_message_hook: Optional[Callable] = None
    broker_message_hook: Optional[Callable] = None
    topic_space: TopicSpace = field(default_factory=lambda: ALL_TOPICS)
    user_protocol: Optional[type] = None    # Protocol class for users
    broker_protocol: Optional[type] = None  # Protocol class for brokers
    ca_cert_path: Optional[str] = None
    ca_key_path: Optional[str] = None
    # strong-consistency feature (reference cargo feature, on by default for
    # brokers): push partial syncs immediately on user connect
    strong_consistency: bool = True
    # task intervals (reference: 10 s heartbeat/sync, 60 s whitelist);
    # configurable so tests can run fast
    heartbeat_interval_s: float = HEARTBEAT_INTERVAL_S
    sync_interval_s: float = SYNC_INTERVAL_S
    whitelist_interval_s: float = WHITELIST_INTERVAL_S
    # data plane: "host" or "gpu"
    data_plane: str = "host"
    gpu_device: str = "cuda:0"
    gpu_tick_interval_s: float = 0.002
    gpu_max_users: int = 16384
    gpu_ring_bytes: int = 1 << 21
    # MeshBroker: collective timeout (peer-failure detection latency) and
    # rebuild-attempt window (communicator rebuild on
    # membership change with host-TCP fallback meanwhile)
    mesh_timeout_s: Optional[float] = 30.0
    mesh_rebuild_timeout_s: float = 10.0
    # HBM message-pool budget for ingest staging (reference
    # --global-memory-pool-size default 1 GiB, broker.rs:71-73); 0 = off
    gpu_pool_bytes: int = 1 << 30


@dataclass
class UserHandle:
    connection: Connection
    task: Optional[asyncio.Task] = None
    gpu_index: Optional[int] = None


@dataclass
class BrokerHandle:
    connection: Connection
    task: Optional[asyncio.Task] = None


class Broker:
    def __init__(self, config: BrokerConfig) -> None:
        from ..proto.transports.tcp import Tcp

        self.config = config
        config.public_advertise_endpoint = resolve_local_ip(config.public_advertise_endpoint)
        config.private_advertise_endpoint = resolve_local_ip(config.private_advertise_endpoint)
        self.identity = BrokerIdentifier(
            config.public_advertise_endpoint, config.private_advertise_endpoint
        )
        self.keypair = config.keypair or bls.KeyPair.from_seed(0)
        self.discovery = new_discovery_client(config.discovery_endpoint, self.identity)
        self.limiter = Limiter(config.global_memory_pool_size)
        self.connections = Connections(self.identity)
        self.user_protocol = config.user_protocol or Tcp
        self.broker_protocol = config.broker_protocol or Tcp
        self._tasks: List[asyncio.Task] = []
        self._listeners = []
        self._closed = False
        self._engine = None
        self._gpu_queue: Optional[asyncio.Queue] = None
        self._free_gpu_slots: List[int] = []
        self._drain_by_buffer: Dict[int, asyncio.Task] = {}
        self._last_drain: Optional[asyncio.Task] = None
        if config.data_plane == "gpu":
            from .gpu_engine import GpuBrokerEngine

            from ..utils.keyhash import derive_routing_seed

            self._engine = GpuBrokerEngine(
                device=config.gpu_device,
                n_users=config.gpu_max_users,
                ring_bytes=config.gpu_ring_bytes,
                fanout_wire=True,  # forward raw wire bytes verbatim
                # keyed routing hash: cluster-wide secret derived from the
                # shared broker private key (see keyhash.derive_routing_seed)
                hash_seed=derive_routing_seed(self.keypair.private_key),
            )
            self._gpu_queue = asyncio.Queue()
            self._free_gpu_slots = list(range(config.gpu_max_users - 1, -1, -1))
            self._gpu_user_by_slot: Dict[int, bytes] = {}
            if config.gpu_pool_bytes:
                from .hbm_pool import HbmMessagePool

                # the CPU reference engine (tests) caps the arena so suites
                # that stand up many brokers stay light; on cuda the full
                # budget is carved out of the 288 GB HBM up front
                pool_bytes = config.gpu_pool_bytes
                if not self._engine.is_cuda:
                    pool_bytes = min(pool_bytes, 32 << 20)
                self._hbm_pool = HbmMessagePool(pool_bytes, device=config.gpu_device)
            else:
                self._hbm_pool = None
        else:
            self._hbm_pool = None

    # ------------------------------ lifecycle ------------------------------

    async def start(self) -> None:
        self._user_listener = await self.user_protocol.bind(
            self.config.public_bind_endpoint, None, None
        )
        self._broker_listener = await self.broker_protocol.bind(
            self.config.private_bind_endpoint, None, None
        )
        self._listeners = [self._user_listener, self._broker_listener]
        loop = asyncio.get_running_loop()
        self._tasks = [
            loop.create_task(self._heartbeat_task(), name="heartbeat"),
            loop.create_task(self._sync_task(), name="sync"),
            loop.create_task(self._whitelist_task(), name="whitelist"),
            loop.create_task(self._user_listener_task(), name="user-listener"),
            loop.create_task(self._broker_listener_task(), name="broker-listener"),
        ]
        if self._engine is not None:
            self._tasks.append(loop.create_task(self._gpu_tick_task(), name="gpu-tick"))
        if self.config.metrics_bind_endpoint:
            from ..utils.metrics import serve_metrics
            from ..proto.transports.tcp import parse_endpoint

            host, port = parse_endpoint(self.config.metrics_bind_endpoint)
            self._metrics_server = await serve_metrics(host, port)

    async def run_forever(self) -> None:
        """Crash the service if any task dies (reference lib.rs:302-318)."""
        await self.start()
        done, _pending = await asyncio.wait(self._tasks, return_when=asyncio.FIRST_COMPLETED)
        for t in done:
            exc = t.exception()
            if exc:
                raise exc
        raise RuntimeError(f"broker task {next(iter(done)).get_name()} exited")

    async def close(self) -> None:
        self._closed = True
        for t in self._tasks:
            t.cancel()
        for handle in list(self.connections.users.values()):
            handle.connection.close()
            if handle.task:
                handle.task.cancel()
        for handle in list(self.connections.brokers.values()):
            handle.connection.close()
            if handle.task:
                handle.task.cancel()
        for l in self._listeners:
            await l.close()

    # ------------------------------ senders ------------------------------

    async def try_send_to_user(self, pubkey: bytes, raw: Bytes) -> None:
        """Send; evict the user on failure (reference user/sender.rs:16-33)."""
        handle = self.connections.users.get(pubkey)
        if handle is None:
            raw.drop()
            return
        try:
            await handle.connection.send_message_raw(raw)
        except Exception:
            await self.remove_user(pubkey)

    async def try_send_to_broker(self, broker: BrokerIdentifier, raw: Bytes) -> None:
        handle = self.connections.brokers.get(broker)
        if handle is None:
            raw.drop()
            return
        try:
            await handle.connection.send_message_raw(raw)
        except Exception:
            await self.remove_broker(broker)

    async def try_send_to_brokers(self, raw: Bytes) -> None:
        """Fan-out to every connected broker (reference broker/sender.rs:49-58)."""
        for broker in self.connections.all_brokers():
            await self.try_send_to_broker(broker, raw.clone())
        raw.drop()

    async def remove_user(self, pubkey: bytes) -> None:
        handle = self.connections.remove_user(pubkey)
        if handle is not None:
            log.info("user disconnected: %s", ident(pubkey))
            if handle.task:
                handle.task.cancel()
            handle.connection.close()
            if self._engine is not None and handle.gpu_index is not None:
                self._release_gpu_slot(handle.gpu_index)

    async def remove_broker(self, broker: BrokerIdentifier) -> None:
        handle = self.connections.remove_broker(broker)
        if handle is not None:
            log.info("broker disconnected: %s", broker)
            if handle.task:
                handle.task.cancel()
            handle.connection.close()

    # ------------------------------ user path ------------------------------

    async def _user_listener_task(self) -> None:
        while not self._closed:
            unfinalized = await self._user_listener.accept()
            asyncio.get_running_loop().create_task(self._handle_user_connection(unfinalized))

    async def _handle_user_connection(self, unfinalized) -> None:
        """accept -> finalize -> permit auth -> add + receive loop
        (reference user/handler.rs:26-91)."""
        try:
            connection = await asyncio.wait_for(unfinalized.finalize(self.limiter), 5)
        except Exception:
            return
        result = await BrokerAuth.verify_user(connection, self.identity, self.discovery)
        if result is None:
            connection.close()
            return
        pubkey, raw_topics = result
        # the INITIAL subscribe tolerates an all-invalid list — the user
        # just connects with no topics (reference user/handler.rs:47
        # discards the prune error); later Subscribe/Unsubscribe frames
        # disconnect on it (handler.rs:140-156)
        try:
            topics = self.config.topic_space.prune(raw_topics)
        except TopicError:
            topics = []
        handle = UserHandle(connection=connection)
        old = self.connections.add_user(pubkey, handle, topics)
        log.info("user connected: %s topics=%s%s", ident(pubkey), topics,
                 " (kicked old session)" if old else "")
        if old is not None:
            # duplicate key kicks the old session (connections/mod.rs:290-298)
            if old.task:
                old.task.cancel()
            old.connection.close()
            if self._engine is not None and old.gpu_index is not None:
                self._release_gpu_slot(old.gpu_index)
        if self._engine is not None:
            # A capacity or routing-hash-collision refusal must clean up the
            # just-registered user — round 1 raised here AFTER add_user,
            # leaving a registered user with no receive loop and no
            # eviction (reference eviction semantics connections/mod.rs:278-304).
            try:
                handle.gpu_index = self._claim_gpu_slot(pubkey)
                self._engine.subscribe(handle.gpu_index, topics)
                self._engine.register_direct(pubkey, handle.gpu_index)
            except (RuntimeError, ValueError) as exc:
                log.warning("refusing user %s: %s", ident(pubkey), exc)
                await self.remove_user(pubkey)
                return
        handle.task = asyncio.get_running_loop().create_task(
            self._user_receive_loop(pubkey, handle)
        )
        if self.config.strong_consistency:
            # immediate partial syncs on connect (user/handler.rs:79-90)
            await self._send_partial_syncs()

    # MeshBroker overrides to False: its collective pack path consumes
    # per-message Bytes, not ingest blobs
    BLOB_INGEST = True

    async def _user_receive_loop_ingest(self, pubkey: bytes, handle: UserHandle) -> None:
        """C++-ingest receive loop: the pump accumulates classified frames
        in one contiguous buffer; Python pulls a TICK's worth per call and
        enqueues the whole blob — no per-message interpreter work at all on
        the Broadcast/Direct hot path.  Semantics match the reference loop
        (user/handler.rs:95-163): malformed/unexpected frames and
        no-valid-topic broadcasts disconnect; Subscribe/Unsubscribe apply
        inline."""
        from ..utils.metrics import BYTES_RECV

        connection = handle.connection
        connection.enable_ingest()
        valid_topic = [False] * 256
        for t in getattr(self.config.topic_space, "valid", range(256)):
            valid_topic[t & 0xFF] = True
        try:
            while True:
                blob, ends, discs, toffs, tcnts, roffs, rlens = \
                    await connection.recv_ingest_batch()
                BYTES_RECV.inc(len(blob))
                fwds = []
                start = 0
                for i, d in enumerate(discs):
                    end = ends[i]
                    if d == 4:  # Broadcast
                        ab = start + toffs[i]
                        if not any(valid_topic[b] for b in blob[ab:ab + tcnts[i]]):
                            raise ConnectionError_("no valid topics")
                        # FRAME-RELATIVE field ranges: consumers (framed
                        # forwarding, mesh digests) slice per-message views
                        fwds.append(("b", toffs[i], toffs[i] + tcnts[i]))
                    elif d == 3:  # Direct
                        fwds.append(("d", roffs[i], roffs[i] + rlens[i]))
                    elif d in (5, 6):  # Subscribe / Unsubscribe (rare, inline)
                        ab = start + toffs[i]
                        try:
                            topics = self.config.topic_space.prune(
                                list(blob[ab:ab + tcnts[i]]))
                        except TopicError:
                            raise ConnectionError_("no valid topics")
                        if d == 5:
                            self.connections.subscribe_user(pubkey, topics)
                            if handle.gpu_index is not None:
                                self._engine.subscribe(handle.gpu_index, topics)
                        else:
                            self.connections.unsubscribe_user(pubkey, topics)
                            if handle.gpu_index is not None:
                                self._engine.unsubscribe(handle.gpu_index, topics)
                        fwds.append(None)
                    else:
                        raise ConnectionError_("malformed or unexpected frame")
                    start = end
                await self._gpu_queue.put(("blob", blob, ends, fwds))
        except (ConnectionError_, asyncio.CancelledError):
            pass
        finally:
            if self.connections.users.get(pubkey) is handle:
                await self.remove_user(pubkey)
            else:
                connection.close()

    async def _user_receive_loop_fast(self, pubkey: bytes, handle: UserHandle) -> None:
        """Batched ingest fast path for the GPU data plane on the native
        pump: drain every frame already queued on the connection in one
        pass and route by a STRUCTURAL parse (proto.message.parse_offsets —
        the host mirror of K4) instead of a full deserialize per message.
        Semantics match the slow loop: Broadcast with no valid topic or any
        malformed/unexpected frame disconnects (reference
        user/handler.rs:95-163); Subscribe/Unsubscribe are handled inline.
        Used only when no user message hook is installed — hooks see
        deserialized Messages, so they take the general loop."""
        from ..proto.message import parse_offsets

        connection = handle.connection
        q = connection._recv_q
        valid_topic = [False] * 256
        for t in getattr(self.config.topic_space, "valid", range(256)):
            valid_topic[t & 0xFF] = True
        try:
            while True:
                raw = await connection.recv_message_raw()
                frames = [raw]
                while not q.empty() and len(frames) < 2048:
                    frames.append(q.get_nowait())
                for raw in frames:
                    try:
                        r = parse_offsets(raw.data)
                    except Exception:
                        raise ConnectionError_("malformed frame")
                    disc = r["disc"]
                    if disc == 4:  # Broadcast
                        toff, tcnt = r["topics_off"], r["topics_cnt"]
                        topics = raw.data[toff:toff + tcnt]
                        # reference prune semantics: error (disconnect) if no
                        # valid topic remains, including the empty list
                        if not any(valid_topic[t] for t in topics):
                            raise ConnectionError_("no valid topics")
                        await self._gpu_queue.put((raw, ("b", bytes(topics))))
                    elif disc == 3:  # Direct
                        await self._gpu_queue.put((raw, ("d", r["recipient"])))
                    elif disc in (5, 6):  # Subscribe / Unsubscribe
                        toff, tcnt = r["topics_off"], r["topics_cnt"]
                        try:
                            topics = self.config.topic_space.prune(
                                list(raw.data[toff:toff + tcnt]))
                        except TopicError:
                            raise ConnectionError_("no valid topics")
                        if disc == 5:
                            self.connections.subscribe_user(pubkey, topics)
                            if handle.gpu_index is not None:
                                self._engine.subscribe(handle.gpu_index, topics)
                        else:
                            self.connections.unsubscribe_user(pubkey, topics)
                            if handle.gpu_index is not None:
                                self._engine.unsubscribe(handle.gpu_index, topics)
                        raw.drop()
                    else:
                        raise ConnectionError_("unexpected message type")
        except (ConnectionError_, asyncio.CancelledError):
            pass
        finally:
            if self.connections.users.get(pubkey) is handle:
                await self.remove_user(pubkey)
            else:
                connection.close()

    async def _user_receive_loop(self, pubkey: bytes, handle: UserHandle) -> None:
        """The per-user hot loop (reference user/handler.rs:95-163)."""
        if (self._engine is not None
                and self.config.user_message_hook is None
                and type(self).BLOB_INGEST
                and hasattr(handle.connection, "enable_ingest")):
            await self._user_receive_loop_ingest(pubkey, handle)
            return
        if (self._engine is not None
                and self.config.user_message_hook is None
                and hasattr(handle.connection, "_recv_q")):
            await self._user_receive_loop_fast(pubkey, handle)
            return
        connection = handle.connection
        try:
            while True:
                raw = await connection.recv_message_raw()
                try:
                    msg = m.deserialize(raw.data)
                except Exception:
                    break
                hook = self.config.user_message_hook
                if hook is not None:
                    verdict = hook(msg)
                    if verdict == SKIP_MESSAGE:
                        raw.drop()
                        continue
                    if verdict not in (PROCESS_MESSAGE, None):
                        break
                if isinstance(msg, m.Broadcast):
                    try:
                        topics = self.config.topic_space.prune(msg.topics)
                    except TopicError:
                        break
                    if self._engine is not None:
                        await self._gpu_queue.put((raw, ("b", bytes(topics))))
                    else:
                        await self.handle_broadcast_message(topics, raw, to_users_only=False)
                elif isinstance(msg, m.Direct):
                    if self._engine is not None:
                        await self._gpu_queue.put((raw, ("d", msg.recipient)))
                    else:
                        await self.handle_direct_message(msg.recipient, raw, to_user_only=False)
                elif isinstance(msg, m.Subscribe):
                    try:
                        topics = self.config.topic_space.prune(msg.topics)
                    except TopicError:
                        break
                    self.connections.subscribe_user(pubkey, topics)
                    if self._engine is not None and handle.gpu_index is not None:
                        self._engine.subscribe(handle.gpu_index, topics)
                    raw.drop()
                elif isinstance(msg, m.Unsubscribe):
                    # pruned like Subscribe: an all-invalid unsubscribe
                    # disconnects (reference user/handler.rs:150-156)
                    try:
                        topics = self.config.topic_space.prune(msg.topics)
                    except TopicError:
                        break
                    self.connections.unsubscribe_user(pubkey, topics)
                    if self._engine is not None and handle.gpu_index is not None:
                        self._engine.unsubscribe(handle.gpu_index, topics)
                    raw.drop()
                else:
                    break  # unexpected message type: disconnect
        except (ConnectionError_, asyncio.CancelledError):
            pass
        finally:
            # only remove if WE are still the registered session — a
            # duplicate-key connect may have already replaced this handle
            # (reference connections/mod.rs:290-298 kick semantics)
            if self.connections.users.get(pubkey) is handle:
                await self.remove_user(pubkey)
            else:
                connection.close()

    # ------------------------------ routing ------------------------------

    async def handle_broadcast_message(
        self, topics: Sequence[int], raw: Bytes, to_users_only: bool
    ) -> None:
        """Forward raw bytes verbatim to interested brokers + users
        (reference broker/handler.rs:240-272)."""
        users, brokers = self.connections.get_interested_by_topic(topics, to_users_only)
        for broker in brokers:
            await self.try_send_to_broker(broker, raw.clone())
        for user in users:
            await self.try_send_to_user(user, raw.clone())
        raw.drop()

    async def handle_direct_message(
        self, recipient: bytes, raw: Bytes, to_user_only: bool
    ) -> None:
        """DirectMap lookup -> local delivery or 1-hop forward
        (reference broker/handler.rs:197-237)."""
        owner = self.connections.get_broker_identifier_of_user(recipient)
        if owner is None:
            raw.drop()  # unknown user: silently dropped (handler.rs:209-236)
            return
        if owner == self.identity:
            await self.try_send_to_user(recipient, raw)
        elif not to_user_only:
            await self.try_send_to_broker(owner, raw)
        else:
            raw.drop()

    # ------------------------------ broker path ------------------------------

    async def _broker_listener_task(self) -> None:
        while not self._closed:
            unfinalized = await self._broker_listener.accept()
            asyncio.get_running_loop().create_task(
                self._handle_broker_connection(unfinalized, is_outbound=False)
            )

    async def _handle_broker_connection(self, conn_or_unfinalized, is_outbound: bool,
                                        peer: Optional[BrokerIdentifier] = None) -> None:
        """Mutual auth (direction-dependent ordering) + initial full syncs +
        receive loop (reference broker/handler.rs:31-118)."""
        try:
            if is_outbound:
                connection = conn_or_unfinalized
                peer_identity = await _BA.authenticate_with_broker(connection, self.keypair)
                ok = await _BA.verify_broker(connection, self.identity, self.keypair)
                if not ok:
                    connection.close()
                    return
            else:
                connection = await asyncio.wait_for(conn_or_unfinalized.finalize(self.limiter), 5)
                ok = await _BA.verify_broker(connection, self.identity, self.keypair)
                if not ok:
                    connection.close()
                    return
                peer_identity = await _BA.authenticate_with_broker(connection, self.keypair)
        except Exception:
            return
        handle = BrokerHandle(connection=connection)
        old = self.connections.add_broker(peer_identity, handle)
        log.info("broker connected: %s (outbound=%s)%s", peer_identity, is_outbound,
                 " (replaced old connection)" if old else "")
        if old is not None:
            if old.task:
                old.task.cancel()
            old.connection.close()
        # initial full syncs (broker/handler.rs:98-117)
        await self.try_send_to_broker(
            peer_identity, Bytes(m.serialize(m.TopicSync(self.connections.get_full_topic_sync())))
        )
        await self.try_send_to_broker(
            peer_identity, Bytes(m.serialize(m.UserSync(self.connections.get_full_user_sync())))
        )
        handle.task = asyncio.get_running_loop().create_task(
            self._broker_receive_loop(peer_identity, connection)
        )

    async def _broker_receive_loop(
        self, peer: BrokerIdentifier, connection: Connection
    ) -> None:
        """reference broker/handler.rs:121-194."""
        try:
            while True:
                raw = await connection.recv_message_raw()
                try:
                    msg = m.deserialize(raw.data)
                except Exception:
                    break
                hook = self.config.broker_message_hook
                if hook is not None:
                    verdict = hook(msg)
                    if verdict == SKIP_MESSAGE:
                        raw.drop()
                        continue
                    if verdict not in (PROCESS_MESSAGE, None):
                        break
                if isinstance(msg, m.Broadcast):
                    try:
                        topics = self.config.topic_space.prune(msg.topics)
                    except TopicError:
                        raw.drop()
                        continue
                    # single-hop mesh: deliver only to local users.  On the
                    # GPU plane that means routing through the ENGINE with
                    # no re-forwarding (fwd=None) — the kernels only ever
                    # write local rings, which IS to_users_only.
                    if self._engine is not None:
                        await self._gpu_queue.put((raw, None))
                    else:
                        await self.handle_broadcast_message(topics, raw, to_users_only=True)
                elif isinstance(msg, m.Direct):
                    if self._engine is not None:
                        # K5 drops non-local recipients = to_user_only
                        await self._gpu_queue.put((raw, None))
                    else:
                        await self.handle_direct_message(msg.recipient, raw, to_user_only=True)
                elif isinstance(msg, m.UserSync):
                    to_kick = self.connections.apply_user_sync(msg.data)
                    for pubkey in to_kick:
                        await self.remove_user(pubkey)
                    raw.drop()
                elif isinstance(msg, m.TopicSync):
                    self.connections.apply_topic_sync(peer, msg.data)
                    raw.drop()
                else:
                    break
        except (ConnectionError_, asyncio.CancelledError):
            pass
        finally:
            await self.remove_broker(peer)

    # ------------------------------ tasks ------------------------------

    async def _heartbeat_task(self) -> None:
        """Every 10 s: publish load, discover peers, dial brokers with
        identifier >= ours, shuffled (reference heartbeat.rs:28-108)."""
        while True:
            try:
                await self.discovery.perform_heartbeat(
                    len(self.connections.users), HEARTBEAT_EXPIRY_S
                )
                others = await self.discovery.get_other_brokers()
                to_dial = [
                    b for b in others
                    if b not in self.connections.brokers and str(b) >= str(self.identity)
                ]
                random.shuffle(to_dial)
                for peer in to_dial:
                    asyncio.get_running_loop().create_task(self._dial_broker(peer))
            except Exception:
                pass
            await asyncio.sleep(self.config.heartbeat_interval_s)

    async def _dial_broker(self, peer: BrokerIdentifier) -> None:
        try:
            connection = await self.broker_protocol.connect(
                peer.private_advertise_endpoint, True, self.limiter
            )
        except Exception:
            return
        await self._handle_broker_connection(connection, is_outbound=True, peer=peer)

    async def _send_partial_syncs(self) -> None:
        user_delta = self.connections.get_partial_user_sync()
        topic_delta = self.connections.get_partial_topic_sync()
        if user_delta:
            await self.try_send_to_brokers(Bytes(m.serialize(m.UserSync(user_delta))))
        if topic_delta:
            await self.try_send_to_brokers(Bytes(m.serialize(m.TopicSync(topic_delta))))

    async def _sync_task(self) -> None:
        """Every 10 s: ship CRDT diffs to every peer (reference sync.rs:129-144)."""
        while True:
            await asyncio.sleep(self.config.sync_interval_s)
            try:
                await self._send_partial_syncs()
            except Exception:
                pass

    async def _whitelist_task(self) -> None:
        """Every 60 s: kick users no longer whitelisted
        (reference whitelist.rs:19-46)."""
        while True:
            await asyncio.sleep(self.config.whitelist_interval_s)
            try:
                for pubkey in self.connections.all_users():
                    if not await self.discovery.check_whitelist(pubkey):
                        await self.remove_user(pubkey)
            except Exception:
                pass

    # ------------------------------ GPU data plane ------------------------------

    def _claim_gpu_slot(self, pubkey: bytes) -> int:
        if not self._free_gpu_slots:
            raise RuntimeError("gpu broker user capacity exceeded")
        slot = self._free_gpu_slots.pop()
        self._gpu_user_by_slot[slot] = pubkey
        return slot

    def _release_gpu_slot(self, slot: int) -> None:
        pubkey = self._gpu_user_by_slot.pop(slot, None)
        if pubkey is not None:
            # drop the direct-table entry: a Direct to the departed key must
            # be dropped, not delivered to the slot's next occupant
            self._engine.unregister_direct(pubkey)
        # clear all subscriptions for the slot
        self._engine.unsubscribe(slot, list(range(256)))
        self._free_gpu_slots.append(slot)

    async def _drain_egress(self) -> None:
        """One tick's egress: K7-compact every used ring into one staging
        buffer (one D2H), then hand the WHOLE tick to the C++ pump in one
        send_rings_batch call.  Per-user fallback (memory/asyncio transports
        or a foreign pump) parses records in Python as before.  Eviction on
        delivery failure is preserved (reference user/sender.rs:16-33).

        Drains are PIPELINED: the socket dispatch of tick N runs as a
        background task (chained after N-1 so per-user frame order holds)
        while tick N+1 ingests and routes; host staging is double-buffered
        and a buffer is only reused once its drain completed."""
        idx = self._engine.next_staging_index()
        busy = self._drain_by_buffer.get(idx)
        if busy is not None:
            try:
                await asyncio.shield(busy)  # buffer still being written out
            except Exception:
                # a failed dispatch (e.g. eviction errors) must not kill the
                # tick loop; the buffer is no longer being written either way
                pass
        wpos, offsets, staging = self._engine.drain_compact()
        prev = self._last_drain

        async def _dispatch_chained():
            if prev is not None:
                try:
                    await asyncio.shield(prev)
                except Exception:
                    pass
            await self._dispatch_egress(wpos, offsets, staging)

        task = asyncio.get_running_loop().create_task(_dispatch_chained())
        self._drain_by_buffer[idx] = task
        self._last_drain = task

    async def _dispatch_egress(self, wpos, offsets, staging) -> None:
        from .gpu_engine import parse_ring_records

        by_pump = {}        # pump -> [(pubkey, cid, start, end)]
        fallback = []       # (slot, pubkey, nbytes)
        for slot, pubkey in list(self._gpu_user_by_slot.items()):
            n = int(wpos[slot])
            if n == 0:
                continue
            handle = self.connections.users.get(pubkey)
            ph = getattr(handle.connection, "pump_handle", None) \
                if handle is not None else None
            if ph is not None:
                try:
                    p, cid = ph()
                except Exception:
                    await self.remove_user(pubkey)
                    continue
                by_pump.setdefault(p, []).append(
                    (pubkey, cid, int(offsets[slot]), int(offsets[slot + 1])))
                continue
            fallback.append((slot, pubkey, n))
        if by_pump:
            # off-loop AND parallel: the coalescing memcpy releases the GIL
            # in C++ (frame building is lock-free), so sharding the users
            # across executor threads — and across pump shards
            # (PUSHCDN_PUMP_SHARDS epoll threads) — scales the drain with
            # cores while the event loop stays responsive
            loop = asyncio.get_running_loop()
            view = staging.numpy()
            jobs = []   # (chunk, future)
            for pump, entries in by_pump.items():
                nshards = min(4, len(entries))
                per = (len(entries) + nshards - 1) // nshards
                for i in range(0, len(entries), per):
                    chunk = entries[i:i + per]
                    jobs.append((chunk, loop.run_in_executor(
                        None, pump.send_rings_batch, view,
                        [e[1] for e in chunk],
                        [e[2] for e in chunk],
                        [e[3] for e in chunk],
                    )))
            results = await asyncio.gather(*(f for _c, f in jobs))
            from ..utils.metrics import BYTES_SENT

            total_payload = 0
            for (chunk, _f), flat in zip(jobs, results):
                for i, (pubkey, _cid, _s, _e) in enumerate(chunk):
                    cnt = flat[2 * i]
                    total_payload += flat[2 * i + 1]
                    if cnt < 0:
                        await self.remove_user(pubkey)
            if total_payload:
                BYTES_SENT.inc(total_payload)
        for slot, pubkey, n in fallback:
            ring = bytes(staging[int(offsets[slot]):int(offsets[slot + 1])].numpy()
                         .tobytes())
            handle = self.connections.users.get(pubkey)
            sink = getattr(handle.connection, "send_ring_records", None) \
                if handle is not None else None
            if sink is not None:
                try:
                    sink(ring, n)
                except Exception:
                    await self.remove_user(pubkey)
            else:
                for _seq, payload in parse_ring_records(ring, n):
                    await self.try_send_to_user(pubkey, Bytes(payload))

    async def _forward_to_mesh(self, raw: Bytes, fwd) -> None:
        """Broker-plane forwarding for a LOCAL-origin message on the GPU
        data plane — the framed-TCP side of the unified broker plane
        (reference broker/handler.rs:197-272): Broadcasts go to every peer
        with subscribers on the message's topics, Directs to the DirectMap
        owner; remote-origin messages (fwd=None) are never re-forwarded
        (single-hop mesh)."""
        kind, data = fwd
        if kind == "b":
            _users, brokers = self.connections.get_interested_by_topic(
                list(data), to_users_only=False)
            for broker in brokers:
                await self.try_send_to_broker(broker, raw.clone())
        else:
            owner = self.connections.get_broker_identifier_of_user(data)
            if owner is not None and owner != self.identity:
                await self.try_send_to_broker(owner, raw.clone())

    async def _gpu_tick_task(self) -> None:
        """Batch queued user messages through the kernel pipeline each tick,
        then drain egress rings back to the user connections.  Queue items
        are either per-message (Bytes, fwd) pairs (asyncio transports,
        broker-plane inbound) or whole ingest blobs
        ("blob", bytes, end_offsets, fwds) from the C++ pump path."""
        import time as _time

        ts = self.tick_stats = {"assemble": 0.0, "ingest": 0.0, "tick": 0.0,
                                "drain": 0.0, "sleep": 0.0, "msgs": 0, "ticks": 0}
        while True:
            item = await self._gpu_queue.get()
            t0 = _time.perf_counter()
            batch = [item]
            # bound the tick by ITEMS and BYTES: blob items can each carry
            # thousands of messages, and an unbounded catch-up tick would
            # balloon the ingest staging + drain
            tick_bytes = len(item[1]) if item[0] == "blob" else len(item[0].data)
            while (not self._gpu_queue.empty() and len(batch) < 4096
                   and tick_bytes < (64 << 20)):
                nxt = self._gpu_queue.get_nowait()
                tick_bytes += len(nxt[1]) if nxt[0] == "blob" else len(nxt[0].data)
                batch.append(nxt)
            buf = bytearray()
            offsets = [0]
            have_peers = bool(self.connections.brokers)
            legacy = []  # Bytes to drop after the tick
            for item in batch:
                if item[0] == "blob":
                    _tag, blob, ends, fwds = item
                    if have_peers:
                        fs = 0
                        for i, fwd in enumerate(fwds):
                            fe = ends[i]
                            if fwd is not None:
                                kind, s, e = fwd  # frame-relative
                                if kind == "b":
                                    _u, brokers = self.connections.get_interested_by_topic(
                                        list(blob[fs + s:fs + e]), to_users_only=False)
                                    for b in brokers:
                                        await self.try_send_to_broker(
                                            b, Bytes(blob[fs:fe]))
                                else:
                                    owner = self.connections.get_broker_identifier_of_user(
                                        blob[fs + s:fs + e])
                                    if owner is not None and owner != self.identity:
                                        await self.try_send_to_broker(
                                            owner, Bytes(blob[fs:fe]))
                            fs = fe
                    base = len(buf)
                    buf += blob
                    offsets.extend(base + e for e in ends)
                else:
                    raw, fwd = item
                    legacy.append(raw)
                    if have_peers and fwd is not None:
                        await self._forward_to_mesh(raw, fwd)
                    buf += raw.data
                    offsets.append(len(buf))
            t1 = _time.perf_counter()
            # ingest staging from the bounded HBM pool: exhaustion WAITS
            # here (backpressure up through the tick queue to the sockets),
            # matching the reference limiter (protocols/mod.rs:328)
            pb = await self._hbm_pool.alloc(len(buf)) if self._hbm_pool else None
            host = bytes(buf)
            dbuf, doff = self._engine.ingest(
                host, offsets, staging=pb.tensor if pb else None)
            t2 = _time.perf_counter()
            self._engine.tick(dbuf, doff, host_batch=host, host_offsets=offsets)
            t3 = _time.perf_counter()
            await self._drain_egress()
            t4 = _time.perf_counter()
            if pb is not None:
                pb.drop()
            for raw in legacy:
                raw.drop()
            ts["assemble"] += t1 - t0
            ts["ingest"] += t2 - t1
            ts["tick"] += t3 - t2
            ts["drain"] += t4 - t3
            ts["msgs"] += len(offsets) - 1
            ts["ticks"] += 1
            await asyncio.sleep(self.config.gpu_tick_interval_s)
            ts["sleep"] += _time.perf_counter() - t4
