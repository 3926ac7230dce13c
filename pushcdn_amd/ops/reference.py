"""CPU reference implementations of the data-plane kernels.

Pure Python/torch mirrors of csrc/hip/dataplane.hip, used for:
  - golden-testing each HIP kernel (GPU result == this, bit-for-bit)
  - running the broker engine on CPU (tests without a GPU)

Semantics match the reference broker: topic match = union of subscriber sets
(connections/mod.rs:94-124), fan-out copies the raw payload per recipient
(sender.rs), per-(sender->recipient) FIFO order preserved.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import List, Tuple

import torch

from ..utils.keyhash import fnv1a64
from ..broker.gpu_engine import ring_rec


@dataclass
class ParseResult:
    disc: torch.Tensor         # int32 [M]
    payload_off: torch.Tensor  # int64 [M]
    payload_len: torch.Tensor  # int32 [M]
    topics_off: torch.Tensor   # int64 [M]
    topics_cnt: torch.Tensor   # int32 [M]
    recip_hash: torch.Tensor   # int64 [M] (bit-cast u64)
    timestamp: torch.Tensor    # int64 [M]


def _i64(x: int) -> int:
    """Bit-cast u64 -> i64 for storage in torch int64 tensors."""
    return x - (1 << 64) if x >= (1 << 63) else x


def parse_batch(buf: bytes, offsets: List[int], hash_seed: int = 0) -> ParseResult:
    """Mirror of k4_parse_batch. `buf` is concatenated serialized Messages."""
    from ..proto import message as msglib

    M = len(offsets) - 1
    disc = torch.full((M,), -1, dtype=torch.int32)
    payload_off = torch.zeros(M, dtype=torch.int64)
    payload_len = torch.zeros(M, dtype=torch.int32)
    topics_off = torch.zeros(M, dtype=torch.int64)
    topics_cnt = torch.zeros(M, dtype=torch.int32)
    recip_hash = torch.zeros(M, dtype=torch.int64)
    timestamp = torch.zeros(M, dtype=torch.int64)

    for i in range(M):
        raw = buf[offsets[i] : offsets[i + 1]]
        base = offsets[i]
        try:
            r = msglib.parse_offsets(raw)
        except Exception:
            continue
        disc[i] = r["disc"]
        payload_off[i] = base + r["payload_off"] if r["payload_len"] else 0
        payload_len[i] = r["payload_len"]
        topics_off[i] = base + r["topics_off"] if r["topics_cnt"] else 0
        topics_cnt[i] = r["topics_cnt"]
        timestamp[i] = _i64(r["timestamp"])
        if r["disc"] == 3:
            recip_hash[i] = _i64(fnv1a64(r["recipient"], hash_seed))
    return ParseResult(disc, payload_off, payload_len, topics_off, topics_cnt, recip_hash, timestamp)


def topic_mask(
    sub_bitmap: torch.Tensor,  # int64 [256][W] (bit-cast u64)
    buf: bytes,
    topics_off: torch.Tensor,
    topics_cnt: torch.Tensor,
    disc: torch.Tensor,
) -> torch.Tensor:
    """Mirror of k2a_topic_mask: OR of topic rows for broadcast messages."""
    M = disc.shape[0]
    W = sub_bitmap.shape[1]
    mask = torch.zeros((M, W), dtype=torch.int64)
    for m in range(M):
        if int(disc[m]) != 4:
            continue
        off, cnt = int(topics_off[m]), int(topics_cnt[m])
        acc = torch.zeros(W, dtype=torch.int64)
        for t in buf[off : off + cnt]:
            acc |= sub_bitmap[t]
        mask[m] = acc
    return mask


def assign_emit(
    mask: torch.Tensor,         # int64 [M][W]
    payload_len: torch.Tensor,  # int32 [M]
    ring_wpos: torch.Tensor,    # int64 [n_users]
    ring_bytes: int,
    n_users: int,
) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor, int]:
    """Mirror of k2b: per-user ordered ring assignment. Returns
    (pair_user, pair_msg, pair_dst, drops); mutates ring_wpos."""
    M, W = mask.shape
    pair_user: List[int] = []
    pair_msg: List[int] = []
    pair_dst: List[int] = []
    drops = 0
    # Same grouping as the kernel: pairs grouped by user, messages in order.
    for u in range(n_users):
        w, bit = u >> 6, 1 << (u & 63)
        wpos = int(ring_wpos[u])
        for m in range(M):
            # Python's & on negative ints is two's-complement with infinite
            # sign extension, so this is correct for bit 63 too.
            if not (int(mask[m, w]) & bit):
                continue
            length = int(payload_len[m])
            rec = ring_rec(length)
            if wpos + rec > ring_bytes:
                drops += 1
                pair_user.append(-1)
                pair_msg.append(m)
                pair_dst.append(0)
                continue
            pair_user.append(u)
            pair_msg.append(m)
            pair_dst.append(u * ring_bytes + wpos)
            wpos += rec
        ring_wpos[u] = wpos
    return (
        torch.tensor(pair_user, dtype=torch.int32),
        torch.tensor(pair_msg, dtype=torch.int32),
        torch.tensor(pair_dst, dtype=torch.int64),
        drops,
    )


def fanout(
    buf: bytes,
    payload_off: torch.Tensor,
    payload_len: torch.Tensor,
    pair_user: torch.Tensor,
    pair_msg: torch.Tensor,
    pair_dst: torch.Tensor,
    msg_seq: torch.Tensor,
    egress: bytearray,
) -> None:
    """Mirror of k3_fanout: write {u32 len, u32 seq, 8B pad} + payload."""
    import struct

    for p in range(pair_user.shape[0]):
        u = int(pair_user[p])
        if u < 0:
            continue
        m = int(pair_msg[p])
        off, length = int(payload_off[m]), int(payload_len[m])
        dst = int(pair_dst[p])
        struct.pack_into("<IIII", egress, dst, length, int(msg_seq[m]) & 0xFFFFFFFF, 0, 0)
        egress[dst + 16 : dst + 16 + length] = buf[off : off + length]


def direct_lookup(table_keys: torch.Tensor, table_vals: torch.Tensor, query: torch.Tensor) -> torch.Tensor:
    """Mirror of k5_direct_lookup (linear probe, 0 = empty key)."""
    S = table_keys.shape[0]
    maskv = S - 1
    out = torch.full((query.shape[0],), -(2**31), dtype=torch.int32)
    for i in range(query.shape[0]):
        h = int(query[i]) & ((1 << 64) - 1)
        if h == 0:
            continue
        s = h & maskv
        for probe in range(S):
            k = int(table_keys[(s + probe) & maskv]) & ((1 << 64) - 1)
            if k == h:
                out[i] = table_vals[(s + probe) & maskv]
                break
            if k == 0:
                break
    return out


def build_direct_table(entries: List[Tuple[int, int]], size: int) -> Tuple[torch.Tensor, torch.Tensor]:
    """Host-side construction of the open-addressing table (u64 hash -> owner)."""
    assert size & (size - 1) == 0
    keys = torch.zeros(size, dtype=torch.int64)
    vals = torch.zeros(size, dtype=torch.int32)
    maskv = size - 1
    for h, owner in entries:
        h &= (1 << 64) - 1
        assert h != 0
        s = h & maskv
        for probe in range(size):
            slot = (s + probe) & maskv
            k = int(keys[slot]) & ((1 << 64) - 1)
            if k == 0 or k == h:
                keys[slot] = _i64(h)
                vals[slot] = owner
                break
        else:
            raise RuntimeError("direct table full")
    return keys, vals
