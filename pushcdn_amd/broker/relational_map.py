"""RelationalMap — bidirectional multimap key<->values used for the broadcast
(topic) subscription state (reference
``cdn-broker/src/connections/broadcast/relational_map.rs``).

Keys are connection identities (user pubkey bytes / broker id strings),
values are topics.  Both directions are kept consistent on every mutation.
"""

from __future__ import annotations

from typing import Dict, Generic, Iterable, Set, TypeVar

K = TypeVar("K")
V = TypeVar("V")


class RelationalMap(Generic[K, V]):
    def __init__(self) -> None:
        self._by_key: Dict[K, Set[V]] = {}
        self._by_value: Dict[V, Set[K]] = {}

    def associate_key_with_values(self, key: K, values: Iterable[V]) -> None:
        ks = self._by_key.setdefault(key, set())
        for v in values:
            ks.add(v)
            self._by_value.setdefault(v, set()).add(key)

    def dissociate_key_from_values(self, key: K, values: Iterable[V]) -> None:
        ks = self._by_key.get(key)
        if ks is None:
            return
        for v in values:
            ks.discard(v)
            vs = self._by_value.get(v)
            if vs is not None:
                vs.discard(key)
                if not vs:
                    del self._by_value[v]
        if not ks:
            del self._by_key[key]

    def remove_key(self, key: K) -> Set[V]:
        ks = self._by_key.pop(key, set())
        for v in ks:
            vs = self._by_value.get(v)
            if vs is not None:
                vs.discard(key)
                if not vs:
                    del self._by_value[v]
        return ks

    def get_values_by_key(self, key: K) -> Set[V]:
        return set(self._by_key.get(key, set()))

    def get_keys_by_value(self, value: V) -> Set[K]:
        return set(self._by_value.get(value, set()))

    def get_values(self) -> Set[V]:
        return set(self._by_value.keys())

    def get_keys(self) -> Set[K]:
        return set(self._by_key.keys())

    def __contains__(self, key: K) -> bool:
        return key in self._by_key
