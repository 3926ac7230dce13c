"""VersionedMap — the eventually-consistent replicated map used for broker
state sync (reference ``cdn-broker/src/connections/versioned_map.rs``):
per-key u64 version + tombstones; ``diff()`` returns changes since the last
diff; ``merge()`` applies a remote delta with last-writer-wins on version,
ties broken by a conflict identity (larger wins).

Wire form for sync messages: a compact length-prefixed binary encoding
(this framework's documented format — the reference used rkyv, a Rust-only
layout; byte-compat there is neither possible nor required since both ends
are ours)."""

from __future__ import annotations

import struct
from dataclasses import dataclass
from typing import Callable, Dict, Generic, List, Optional, Tuple, TypeVar

K = TypeVar("K")
V = TypeVar("V")
C = TypeVar("C")


@dataclass
class Versioned(Generic[V, C]):
    value: Optional[V]        # None = tombstone (removed)
    version: int
    conflict_id: C


class VersionedMap(Generic[K, V, C]):
    def __init__(self, local_conflict_id: C) -> None:
        self.local_conflict_id = local_conflict_id
        self._map: Dict[K, Versioned[V, C]] = {}
        self._dirty: Dict[K, Versioned[V, C]] = {}

    # ------------------------- local mutation -------------------------

    def insert(self, key: K, value: V) -> None:
        self._modify_local(key, value)

    def remove(self, key: K) -> None:
        if key in self._map and self._map[key].value is not None:
            self._modify_local(key, None)

    def _modify_local(self, key: K, value: Optional[V]) -> None:
        prev = self._map.get(key)
        version = (prev.version + 1) if prev else 1
        entry = Versioned(value, version, self.local_conflict_id)
        self._map[key] = entry
        self._dirty[key] = entry

    def get(self, key: K) -> Optional[V]:
        e = self._map.get(key)
        return e.value if e else None

    def items(self) -> List[Tuple[K, V]]:
        return [(k, e.value) for k, e in self._map.items() if e.value is not None]

    def __len__(self) -> int:
        return sum(1 for e in self._map.values() if e.value is not None)

    # ------------------------- replication -------------------------

    def diff(self) -> Dict[K, Versioned[V, C]]:
        """Changes since the last diff() (reference versioned_map.rs:168-194);
        also purges tombstones that have now been shipped."""
        d = self._dirty
        self._dirty = {}
        # purge shipped tombstones from the map (keep versions monotone by
        # retaining version info only while the entry is live)
        for k, e in list(self._map.items()):
            if e.value is None and k in d:
                del self._map[k]
        return d

    def get_full(self) -> Dict[K, Versioned[V, C]]:
        return dict(self._map)

    def merge(self, remote: Dict[K, Versioned[V, C]]) -> List[Tuple[K, Optional[V], Optional[V]]]:
        """Apply a remote delta. Returns [(key, old_value, new_value)] for
        entries that changed (the broker uses this to kick moved users —
        reference connections/mod.rs:154-162)."""
        changed: List[Tuple[K, Optional[V], Optional[V]]] = []
        for k, re in remote.items():
            le = self._map.get(k)
            take = False
            if le is None:
                take = True
            elif re.version > le.version:
                take = True
            elif re.version == le.version and re.conflict_id > le.conflict_id:
                take = True
            if take:
                old = le.value if le else None
                if re.value is None:
                    # tombstone: remove
                    if k in self._map:
                        del self._map[k]
                else:
                    self._map[k] = Versioned(re.value, re.version, re.conflict_id)
                if old != re.value:
                    changed.append((k, old, re.value))
        return changed


# ---------------------------------------------------------------------------
# Serialization of sync deltas where K = bytes, V = str (DirectMap) or
# K = int topic, V = bool (TopicSyncMap); conflict id = str.
# Record: [u8 has_value][u64 version][u16 cid_len][cid][u32 key_len][key]
#         [u32 val_len][val]
# ---------------------------------------------------------------------------

def serialize_delta(delta: Dict, key_enc: Callable, val_enc: Callable) -> bytes:
    out = bytearray()
    out += struct.pack("<I", len(delta))
    for k, e in delta.items():
        kb = key_enc(k)
        cid = str(e.conflict_id).encode()
        out += struct.pack("<BQH", 1 if e.value is not None else 0, e.version, len(cid))
        out += cid
        out += struct.pack("<I", len(kb))
        out += kb
        if e.value is not None:
            vb = val_enc(e.value)
            out += struct.pack("<I", len(vb))
            out += vb
    return bytes(out)


def deserialize_delta(data: bytes, key_dec: Callable, val_dec: Callable) -> Dict:
    off = 0
    (n,) = struct.unpack_from("<I", data, off)
    off += 4
    out: Dict = {}
    for _ in range(n):
        has_value, version, cid_len = struct.unpack_from("<BQH", data, off)
        off += 11
        cid = data[off : off + cid_len].decode()
        off += cid_len
        (klen,) = struct.unpack_from("<I", data, off)
        off += 4
        k = key_dec(data[off : off + klen])
        off += klen
        value = None
        if has_value:
            (vlen,) = struct.unpack_from("<I", data, off)
            off += 4
            value = val_dec(data[off : off + vlen])
            off += vlen
        out[k] = Versioned(value, version, cid)
    return out
