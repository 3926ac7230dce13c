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

from dataclasses import dataclass, field
from typing import Callable, Dict, List, Optional, Sequence, Set, Tuple

from ..discovery import BrokerIdentifier
from ..utils.metrics import NUM_BROKERS_CONNECTED, NUM_USERS_CONNECTED
from .relational_map import RelationalMap
from .versioned_map import VersionedMap, Versioned, deserialize_delta, serialize_delta

Topic = int
UserPubKey = bytes


class Connections:
    def __init__(self, identity: BrokerIdentifier) -> None:
        self.identity = identity
        # live connection handles (opaque to this module; services store
        # their Connection + task handles here)
        self.users: Dict[UserPubKey, object] = {}
        self.brokers: Dict[BrokerIdentifier, object] = {}
        # routing state
        self.direct_map: VersionedMap[UserPubKey, str, str] = VersionedMap(str(identity))
        self.user_topics: RelationalMap[UserPubKey, Topic] = RelationalMap()
        self.broker_topics: RelationalMap[str, Topic] = RelationalMap()
        self.topic_sync_map: VersionedMap[Topic, bool, str] = VersionedMap(str(identity))
        self._previous_local_topics: Set[Topic] = set()

    # ------------------------------ users ------------------------------

    def add_user(self, pubkey: UserPubKey, handle: object, topics: Sequence[Topic]) -> Optional[object]:
        """Add a user; returns the OLD handle if a duplicate key was kicked
        (reference connections/mod.rs:278-304)."""
        old = self.users.pop(pubkey, None)
        self.users[pubkey] = handle
        self.user_topics.associate_key_with_values(pubkey, topics)
        self.direct_map.insert(pubkey, str(self.identity))
        NUM_USERS_CONNECTED.set(len(self.users))
        return old

    def remove_user(self, pubkey: UserPubKey) -> Optional[object]:
        handle = self.users.pop(pubkey, None)
        if handle is not None:
            self.user_topics.remove_key(pubkey)
            # only remove from the direct map if we still own the user
            if self.direct_map.get(pubkey) == str(self.identity):
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
        return BrokerIdentifier.parse(owner) if owner else None

    # ------------------------------ sync ------------------------------

    def get_partial_user_sync(self) -> bytes:
        return serialize_delta(self.direct_map.diff(), bytes, lambda v: v.encode())

    def get_full_user_sync(self) -> bytes:
        return serialize_delta(self.direct_map.get_full(), bytes, lambda v: v.encode())

    def apply_user_sync(self, data: bytes) -> List[UserPubKey]:
        """Merge a remote user delta; returns local users to kick (now owned
        elsewhere — reference connections/mod.rs:154-162)."""
        delta = deserialize_delta(data, bytes, lambda b: b.decode())
        changed = self.direct_map.merge(delta)
        to_kick = []
        for key, _old, new in changed:
            if new is not None and new != str(self.identity) and key in self.users:
                to_kick.append(key)
        return to_kick

    def _local_topic_updates(self) -> None:
        """Refresh the TopicSyncMap from the current local user interest set
        (reference connections/mod.rs:205-237)."""
        current = self.user_topics.get_values()
        for t in current - self._previous_local_topics:
            self.topic_sync_map.insert(t, True)
        for t in self._previous_local_topics - current:
            self.topic_sync_map.remove(t)
        self._previous_local_topics = current

    def get_partial_topic_sync(self) -> bytes:
        self._local_topic_updates()
        return serialize_delta(
            self.topic_sync_map.diff(), lambda t: bytes([t]), lambda v: b"\x01"
        )

    def get_full_topic_sync(self) -> bytes:
        self._local_topic_updates()
        return serialize_delta(
            self.topic_sync_map.get_full(), lambda t: bytes([t]), lambda v: b"\x01"
        )

    def apply_topic_sync(self, broker: BrokerIdentifier, data: bytes) -> None:
        """Apply a remote broker's topic interests: subscribe/unsubscribe the
        broker per changed topic (reference connections/mod.rs:165-191)."""
        delta = deserialize_delta(data, lambda b: b[0], lambda b: True)
        for topic, e in delta.items():
            if e.value:
                self.broker_topics.associate_key_with_values(str(broker), [topic])
            else:
                self.broker_topics.dissociate_key_from_values(str(broker), [topic])
