"""Broker connection/routing state — host-side authoritative tables.

Mirror of the reference's ``Connections`` (cdn-broker/src/connections/mod.rs):
  - users: pubkey -> live user connection handle
  - brokers: BrokerIdentifier -> live broker connection handle
  - direct_map: CRDT VersionedMap pubkey -> owning broker
  - broadcast maps: RelationalMap pubkey<->topics, broker<->topics, plus the
    CRDT TopicSyncMap for replicating *this broker's* topic interest set

On a GPU broker these tables are also projected into the device engine
(subscription bitmap + direct hash table) by the service layer; this module
stays torch-free so the control plane is testable anywhere.
"""

from __future__ import annotations

from typing import Dict, List, Optional, Sequence, Set, Tuple

from ..discovery import BrokerIdentifier
from ..utils.metrics import NUM_BROKERS_CONNECTED, NUM_USERS_CONNECTED
from .relational_map import RelationalMap
from .versioned_map import VersionedMap, Versioned, deserialize_delta, serialize_delta

Topic = int
UserPubKey = bytes


class SyncMap:
    """bytes->bytes replicated map over the NATIVE CRDT (pushcdn_core
    VersionedMap, csrc/state/versioned_map.h) with a pure-Python fallback.
    Deltas travel as the documented length-prefixed encoding — the two
    implementations are wire-compatible (cross-checked in test_state.py)."""

    def __init__(self, conflict_id: str) -> None:
        self._native = False
        try:
            from ..ops.build import build_core

            self._m = build_core().VersionedMap(conflict_id)
            self._native = True
        except Exception:
            self._m = VersionedMap(conflict_id)

    def insert(self, key: bytes, value: bytes) -> None:
        self._m.insert(key, value)

    def remove(self, key: bytes) -> None:
        self._m.remove(key)

    def get(self, key: bytes) -> Optional[bytes]:
        return self._m.get(key)

    def diff_bytes(self) -> bytes:
        if self._native:
            return self._m.diff()
        return serialize_delta(self._m.diff(), lambda k: k, lambda v: v)

    def full_bytes(self) -> bytes:
        if self._native:
            return self._m.get_full()
        return serialize_delta(self._m.get_full(), lambda k: k, lambda v: v)

    def merge_bytes(self, raw: bytes) -> List[Tuple[bytes, Optional[bytes], Optional[bytes]]]:
        if self._native:
            return list(self._m.merge(raw))
        delta = deserialize_delta(raw, lambda k: k, lambda v: v)
        return self._m.merge(delta)


class Connections:
    def __init__(self, identity: BrokerIdentifier) -> None:
        self.identity = identity
        # live connection handles (opaque to this module; services store
        # their Connection + task handles here)
        self.users: Dict[UserPubKey, object] = {}
        self.brokers: Dict[BrokerIdentifier, object] = {}
        # routing state (CRDTs are native C++ when pushcdn_core is built)
        self.direct_map = SyncMap(str(identity))
        self.user_topics: RelationalMap[UserPubKey, Topic] = RelationalMap()
        self.broker_topics: RelationalMap[str, Topic] = RelationalMap()
        self.topic_sync_map = SyncMap(str(identity))
        self._previous_local_topics: Set[Topic] = set()

    # ------------------------------ users ------------------------------

    def add_user(self, pubkey: UserPubKey, handle: object, topics: Sequence[Topic]) -> Optional[object]:
        """Add a user; returns the OLD handle if a duplicate key was kicked
        (reference connections/mod.rs:278-304)."""
        old = self.users.pop(pubkey, None)
        self.users[pubkey] = handle
        self.user_topics.associate_key_with_values(pubkey, topics)
        self.direct_map.insert(pubkey, str(self.identity).encode())
        NUM_USERS_CONNECTED.set(len(self.users))
        return old

    def remove_user(self, pubkey: UserPubKey) -> Optional[object]:
        handle = self.users.pop(pubkey, None)
        if handle is not None:
            self.user_topics.remove_key(pubkey)
            # only remove from the direct map if we still own the user
            if self.direct_map.get(pubkey) == str(self.identity).encode():
                self.direct_map.remove(pubkey)
            NUM_USERS_CONNECTED.set(len(self.users))
        return handle

    def subscribe_user(self, pubkey: UserPubKey, topics: Sequence[Topic]) -> None:
        if pubkey in self.users:
            self.user_topics.associate_key_with_values(pubkey, topics)

    def unsubscribe_user(self, pubkey: UserPubKey, topics: Sequence[Topic]) -> None:
        self.user_topics.dissociate_key_from_values(pubkey, topics)

    def all_users(self) -> List[UserPubKey]:
        return list(self.users.keys())

    # ------------------------------ brokers ------------------------------

    def add_broker(self, broker: BrokerIdentifier, handle: object) -> Optional[object]:
        """Duplicate broker connections replace the old one
        (reference connections/mod.rs:262-273)."""
        old = self.brokers.pop(broker, None)
        self.brokers[broker] = handle
        NUM_BROKERS_CONNECTED.set(len(self.brokers))
        return old

    def remove_broker(self, broker: BrokerIdentifier) -> Optional[object]:
        handle = self.brokers.pop(broker, None)
        if handle is not None:
            self.broker_topics.remove_key(str(broker))
            NUM_BROKERS_CONNECTED.set(len(self.brokers))
        return handle

    def all_brokers(self) -> List[BrokerIdentifier]:
        return list(self.brokers.keys())

    # ------------------------------ routing ------------------------------

    def get_interested_by_topic(
        self, topics: Sequence[Topic], to_users_only: bool
    ) -> Tuple[List[UserPubKey], List[BrokerIdentifier]]:
        """Union of subscriber sets over topics
        (reference connections/mod.rs:94-124)."""
        users: Set[UserPubKey] = set()
        brokers: Set[str] = set()
        for t in topics:
            users |= self.user_topics.get_keys_by_value(t)
            if not to_users_only:
                brokers |= self.broker_topics.get_keys_by_value(t)
        return (
            [u for u in users],
            [] if to_users_only else [BrokerIdentifier.parse(b) for b in brokers],
        )

    def get_broker_identifier_of_user(self, pubkey: UserPubKey) -> Optional[BrokerIdentifier]:
        owner = self.direct_map.get(pubkey)
        return BrokerIdentifier.parse(owner.decode()) if owner else None

    # ------------------------------ sync ------------------------------

    def get_partial_user_sync(self) -> bytes:
        return self.direct_map.diff_bytes()

    def get_full_user_sync(self) -> bytes:
        return self.direct_map.full_bytes()

    def apply_user_sync(self, data: bytes) -> List[UserPubKey]:
        """Merge a remote user delta; returns local users to kick (now owned
        elsewhere — reference connections/mod.rs:154-162)."""
        changed = self.direct_map.merge_bytes(data)
        me = str(self.identity).encode()
        to_kick = []
        for key, _old, new in changed:
            if new is not None and new != me and key in self.users:
                to_kick.append(key)
        return to_kick

    def _local_topic_updates(self) -> None:
        """Refresh the TopicSyncMap from the current local user interest set
        (reference connections/mod.rs:205-237)."""
        current = self.user_topics.get_values()
        for t in current - self._previous_local_topics:
            self.topic_sync_map.insert(bytes([t]), b"\x01")
        for t in self._previous_local_topics - current:
            self.topic_sync_map.remove(bytes([t]))
        self._previous_local_topics = current

    def get_partial_topic_sync(self) -> bytes:
        self._local_topic_updates()
        return self.topic_sync_map.diff_bytes()

    def get_full_topic_sync(self) -> bytes:
        self._local_topic_updates()
        return self.topic_sync_map.full_bytes()

    def apply_topic_sync(self, broker: BrokerIdentifier, data: bytes) -> None:
        """Apply a remote broker's topic interests: subscribe/unsubscribe the
        broker per changed topic (reference connections/mod.rs:165-191).
        The delta is applied directly (not merged into our own map: it
        describes the PEER's interests, keyed per peer in broker_topics)."""
        delta = deserialize_delta(data, lambda b: b[0], lambda b: True)
        for topic, e in delta.items():
            if e.value:
                self.broker_topics.associate_key_with_values(str(broker), [topic])
            else:
                self.broker_topics.dissociate_key_from_values(str(broker), [topic])
