"""Wire-format message layer: the 9-variant ``Message`` union.

Wire format is Cap'n Proto segment encoding, byte-compatible with the
reference schema (``/root/reference/cdn-proto/schema/messages.capnp``) and the
reference wrapper (``cdn-proto/src/message.rs:83-312``):

- ``Message``            struct: data 1 word (u16 discriminant @ byte 0), 1 pointer
- ``AuthenticateWithKey``        : data 1 word (timestamp u64), 2 pointers (publicKey, signature)
- ``AuthenticateWithPermit``     : data 1 word (permit u64), 0 pointers
- ``AuthenticateResponse``       : data 1 word (permit u64), 1 pointer (context Text)
- ``Direct``                     : data 0, 2 pointers (recipient, message)
- ``Broadcast``                  : data 0, 2 pointers (topics List(UInt8), message)
- ``subscribe``/``unsubscribe``  : List(UInt8) directly in the union pointer
- ``userSync``/``topicSync``     : Data directly in the union pointer

Discriminants (generated bindings ``messages_capnp.rs:77-117``):
0=AuthenticateWithKey 1=AuthenticateWithPermit 2=AuthenticateResponse
3=Direct 4=Broadcast 5=Subscribe 6=Unsubscribe 7=UserSync 8=TopicSync

This module is the *reference implementation* used to golden-test the C++
(`csrc/wire/`) and on-device (K4) serializers.  The hot path never runs this
Python code.
"""

from __future__ import annotations

import struct
from dataclasses import dataclass
from typing import List, Union

from .errors import SerializeError, DeserializeError

# Topic is a single u8 on the wire (reference message.rs:26)
Topic = int

# Maximum message size: u32::MAX / 8 (reference cdn-proto/src/lib.rs:25)
MAX_MESSAGE_SIZE = (2**32 - 1) // 8


@dataclass
class AuthenticateWithKey:
    public_key: bytes
    timestamp: int
    signature: bytes


@dataclass
class AuthenticateWithPermit:
    permit: int


@dataclass
class AuthenticateResponse:
    permit: int
    context: str


@dataclass
class Direct:
    recipient: bytes
    message: bytes


@dataclass
class Broadcast:
    topics: List[Topic]
    message: bytes


@dataclass
class Subscribe:
    topics: List[Topic]


@dataclass
class Unsubscribe:
    topics: List[Topic]


@dataclass
class UserSync:
    data: bytes


@dataclass
class TopicSync:
    data: bytes


Message = Union[
    AuthenticateWithKey,
    AuthenticateWithPermit,
    AuthenticateResponse,
    Direct,
    Broadcast,
    Subscribe,
    Unsubscribe,
    UserSync,
    TopicSync,
]

_DISCRIMINANT = {
    AuthenticateWithKey: 0,
    AuthenticateWithPermit: 1,
    AuthenticateResponse: 2,
    Direct: 3,
    Broadcast: 4,
    Subscribe: 5,
    Unsubscribe: 6,
    UserSync: 7,
    TopicSync: 8,
}


class _SegmentBuilder:
    """Single-segment Cap'n Proto builder (word-granular bump allocator)."""

    def __init__(self) -> None:
        self.words: bytearray = bytearray()

    def alloc(self, nwords: int) -> int:
        off = len(self.words) // 8
        self.words.extend(b"\x00" * (8 * nwords))
        return off

    def put_u64(self, word_off: int, value: int) -> None:
        struct.pack_into("<Q", self.words, word_off * 8, value)

    def put_u16(self, word_off: int, byte_in_word: int, value: int) -> None:
        struct.pack_into("<H", self.words, word_off * 8 + byte_in_word, value)

    def put_bytes(self, word_off: int, data: bytes) -> None:
        self.words[word_off * 8 : word_off * 8 + len(data)] = data

    def struct_ptr(self, ptr_word: int, target_word: int, data_words: int, ptr_words: int) -> None:
        b = target_word - (ptr_word + 1)
        val = (0 & 3) | ((b & 0x3FFFFFFF) << 2) | ((data_words & 0xFFFF) << 32) | ((ptr_words & 0xFFFF) << 48)
        self.put_u64(ptr_word, val)

    def list_ptr(self, ptr_word: int, target_word: int, elt_size_code: int, count: int) -> None:
        b = target_word - (ptr_word + 1)
        val = 1 | ((b & 0x3FFFFFFF) << 2) | ((elt_size_code & 7) << 32) | ((count & 0x1FFFFFFF) << 35)
        self.put_u64(ptr_word, val)

    def write_byte_list(self, ptr_word: int, data: bytes) -> None:
        """Allocate and write a Data/List(UInt8) (element size code 2).
        Appends the payload directly (one copy) instead of zero-filling the
        allocation first — a 100 MiB payload was paying 3 extra passes."""
        tgt = len(self.words) // 8
        self.words += data
        pad = (-len(data)) % 8
        if pad:
            self.words += b"\x00" * pad
        self.list_ptr(ptr_word, tgt, 2, len(data))

    def write_text(self, ptr_word: int, text: str) -> None:
        """Text = byte list with NUL terminator included in element count."""
        raw = text.encode("utf-8") + b"\x00"
        nwords = (len(raw) + 7) // 8
        tgt = self.alloc(nwords)
        self.put_bytes(tgt, raw)
        self.list_ptr(ptr_word, tgt, 2, len(raw))


_core = None
_core_failed = False


def _get_core():
    """Lazy-load the native C++ wire codec (pushcdn_core); None if unavailable."""
    global _core, _core_failed
    if _core is None and not _core_failed:
        try:
            from ..ops.build import build_core

            _core = build_core()
        except Exception:
            _core_failed = True
    return _core


def serialize(msg: Message) -> bytes:
    """Serialize via the native C++ codec when available (csrc/wire/message.h,
    byte-identical to serialize_py — cross-checked in tests/test_wire.py)."""
    core = _get_core()
    if core is None:
        return serialize_py(msg)
    t = type(msg)
    if t is Broadcast:
        return core.wire_serialize_broadcast(bytes(bytearray(x & 0xFF for x in msg.topics)),
                                             bytes(msg.message))
    if t is Direct:
        return core.wire_serialize_direct(bytes(msg.recipient), bytes(msg.message))
    if t is AuthenticateWithKey:
        return core.wire_serialize_authenticate_with_key(
            bytes(msg.public_key), msg.timestamp & 0xFFFFFFFFFFFFFFFF, bytes(msg.signature))
    if t is AuthenticateWithPermit:
        return core.wire_serialize_authenticate_with_permit(msg.permit & 0xFFFFFFFFFFFFFFFF)
    if t is AuthenticateResponse:
        return core.wire_serialize_authenticate_response(
            msg.permit & 0xFFFFFFFFFFFFFFFF, msg.context)
    if t is Subscribe:
        return core.wire_serialize_topics(5, bytes(bytearray(x & 0xFF for x in msg.topics)))
    if t is Unsubscribe:
        return core.wire_serialize_topics(6, bytes(bytearray(x & 0xFF for x in msg.topics)))
    if t is UserSync:
        return core.wire_serialize_sync(7, bytes(msg.data))
    if t is TopicSync:
        return core.wire_serialize_sync(8, bytes(msg.data))
    raise SerializeError(f"unknown message type {t!r}")


def deserialize(data: bytes) -> Message:
    """Parse via the native C++ codec when available."""
    core = _get_core()
    if core is None:
        return deserialize_py(data)
    d = core.wire_deserialize(bytes(data))
    if d is None:
        raise DeserializeError("malformed message")
    disc = d["disc"]
    if disc == 0:
        return AuthenticateWithKey(d["public_key"], d["timestamp"], d["signature"])
    if disc == 1:
        return AuthenticateWithPermit(d["timestamp"])
    if disc == 2:
        try:
            context = d["context"].decode("utf-8")
        except UnicodeDecodeError as e:
            raise DeserializeError(f"invalid utf-8 in Text: {e}") from e
        return AuthenticateResponse(d["timestamp"], context)
    if disc == 3:
        return Direct(d["recipient"], d["payload"])
    if disc == 4:
        return Broadcast(list(d["topics"]), d["payload"])
    if disc == 5:
        return Subscribe(list(d["topics"]))
    if disc == 6:
        return Unsubscribe(list(d["topics"]))
    if disc == 7:
        return UserSync(d["payload"])
    return TopicSync(d["payload"])


def serialize_py(msg: Message) -> bytes:
    """Pure-Python reference serializer (single capnp segment).

    Mirrors reference ``Message::serialize`` (message.rs:116-204): stream
    header ``[u32 segcount-1 = 0][u32 nwords]`` then the segment.
    """
    seg = _SegmentBuilder()
    root_ptr = seg.alloc(1)
    msg_struct = seg.alloc(2)  # Message: data 1 + ptrs 1
    seg.struct_ptr(root_ptr, msg_struct, 1, 1)
    disc = _DISCRIMINANT.get(type(msg))
    if disc is None:
        raise SerializeError(f"unknown message type {type(msg)!r}")
    seg.put_u16(msg_struct, 0, disc)
    union_ptr = msg_struct + 1

    if isinstance(msg, AuthenticateWithKey):
        inner = seg.alloc(3)  # data 1 + ptrs 2
        seg.struct_ptr(union_ptr, inner, 1, 2)
        seg.write_byte_list(inner + 1, bytes(msg.public_key))
        seg.put_u64(inner, msg.timestamp & 0xFFFFFFFFFFFFFFFF)
        seg.write_byte_list(inner + 2, bytes(msg.signature))
    elif isinstance(msg, AuthenticateWithPermit):
        inner = seg.alloc(1)  # data 1 + ptrs 0
        seg.struct_ptr(union_ptr, inner, 1, 0)
        seg.put_u64(inner, msg.permit & 0xFFFFFFFFFFFFFFFF)
    elif isinstance(msg, AuthenticateResponse):
        inner = seg.alloc(2)  # data 1 + ptrs 1
        seg.struct_ptr(union_ptr, inner, 1, 1)
        seg.put_u64(inner, msg.permit & 0xFFFFFFFFFFFFFFFF)
        seg.write_text(inner + 1, msg.context)
    elif isinstance(msg, Broadcast):
        inner = seg.alloc(2)  # data 0 + ptrs 2
        seg.struct_ptr(union_ptr, inner, 0, 2)
        seg.write_byte_list(inner, bytes(bytearray(t & 0xFF for t in msg.topics)))
        seg.write_byte_list(inner + 1, bytes(msg.message))
    elif isinstance(msg, Direct):
        inner = seg.alloc(2)  # data 0 + ptrs 2
        seg.struct_ptr(union_ptr, inner, 0, 2)
        seg.write_byte_list(inner, bytes(msg.recipient))
        seg.write_byte_list(inner + 1, bytes(msg.message))
    elif isinstance(msg, (Subscribe, Unsubscribe)):
        seg.write_byte_list(union_ptr, bytes(bytearray(t & 0xFF for t in msg.topics)))
    elif isinstance(msg, (UserSync, TopicSync)):
        seg.write_byte_list(union_ptr, bytes(msg.data))
    else:  # pragma: no cover
        raise SerializeError(f"unhandled message type {type(msg)!r}")

    nwords = len(seg.words) // 8
    return struct.pack("<II", 0, nwords) + bytes(seg.words)


class _SegmentReader:
    """Bounds-checked reader over one capnp segment."""

    def __init__(self, data: bytes) -> None:
        if len(data) % 8 != 0:
            raise DeserializeError("segment not word-aligned")
        self.data = data
        self.nwords = len(data) // 8

    def u64(self, word: int) -> int:
        if word < 0 or word >= self.nwords:
            raise DeserializeError("word offset out of bounds")
        return struct.unpack_from("<Q", self.data, word * 8)[0]

    def read_struct_ptr(self, ptr_word: int):
        val = self.u64(ptr_word)
        if val == 0:
            raise DeserializeError("null struct pointer")
        if val & 3 != 0:
            raise DeserializeError("expected struct pointer")
        b = (val >> 2) & 0x3FFFFFFF
        if b & 0x20000000:  # sign-extend 30-bit
            b -= 0x40000000
        data_words = (val >> 32) & 0xFFFF
        ptr_words = (val >> 48) & 0xFFFF
        tgt = ptr_word + 1 + b
        if tgt < 0 or tgt + data_words + ptr_words > self.nwords:
            raise DeserializeError("struct out of bounds")
        return tgt, data_words, ptr_words

    def read_byte_list(self, ptr_word: int) -> bytes:
        val = self.u64(ptr_word)
        if val == 0:
            return b""
        if val & 3 != 1:
            raise DeserializeError("expected list pointer")
        b = (val >> 2) & 0x3FFFFFFF
        if b & 0x20000000:
            b -= 0x40000000
        code = (val >> 32) & 7
        count = (val >> 35) & 0x1FFFFFFF
        if code != 2:
            raise DeserializeError(f"expected byte list, got element code {code}")
        tgt = ptr_word + 1 + b
        if tgt < 0 or tgt * 8 + count > len(self.data):
            raise DeserializeError("list out of bounds")
        return self.data[tgt * 8 : tgt * 8 + count]

    def read_text(self, ptr_word: int) -> str:
        raw = self.read_byte_list(ptr_word)
        if raw and raw[-1] == 0:
            raw = raw[:-1]
        try:
            return raw.decode("utf-8")
        except UnicodeDecodeError as e:
            raise DeserializeError(f"invalid utf-8 in Text: {e}") from e


def deserialize_py(data: bytes) -> Message:
    """Pure-Python reference parser.

    Mirrors reference ``Message::deserialize`` (message.rs:212-312); the
    traversal limit there equals the buffer length, which bounds work the same
    way our explicit bounds checks do.
    """
    if len(data) < 8:
        raise DeserializeError("short buffer")
    seg_count_m1, nwords = struct.unpack_from("<II", data, 0)
    if seg_count_m1 != 0:
        raise DeserializeError("multi-segment messages unsupported")
    if 8 + nwords * 8 > len(data):
        raise DeserializeError("segment extends past buffer")
    seg = _SegmentReader(data[8 : 8 + nwords * 8])

    tgt, dw, pw = seg.read_struct_ptr(0)
    if dw < 1 or pw < 1:
        raise DeserializeError("malformed Message struct")
    disc = struct.unpack_from("<H", seg.data, tgt * 8)[0]
    union_ptr = tgt + dw

    if disc == 0:
        itgt, idw, ipw = seg.read_struct_ptr(union_ptr)
        if idw < 1 or ipw < 2:
            raise DeserializeError("malformed AuthenticateWithKey")
        timestamp = seg.u64(itgt)
        public_key = seg.read_byte_list(itgt + idw)
        signature = seg.read_byte_list(itgt + idw + 1)
        return AuthenticateWithKey(public_key, timestamp, signature)
    if disc == 1:
        itgt, idw, _ = seg.read_struct_ptr(union_ptr)
        if idw < 1:
            raise DeserializeError("malformed AuthenticateWithPermit")
        return AuthenticateWithPermit(seg.u64(itgt))
    if disc == 2:
        itgt, idw, ipw = seg.read_struct_ptr(union_ptr)
        if idw < 1 or ipw < 1:
            raise DeserializeError("malformed AuthenticateResponse")
        return AuthenticateResponse(seg.u64(itgt), seg.read_text(itgt + idw))
    if disc == 3:
        itgt, idw, ipw = seg.read_struct_ptr(union_ptr)
        if ipw < 2:
            raise DeserializeError("malformed Direct")
        return Direct(seg.read_byte_list(itgt + idw), seg.read_byte_list(itgt + idw + 1))
    if disc == 4:
        itgt, idw, ipw = seg.read_struct_ptr(union_ptr)
        if ipw < 2:
            raise DeserializeError("malformed Broadcast")
        topics = list(seg.read_byte_list(itgt + idw))
        return Broadcast(topics, seg.read_byte_list(itgt + idw + 1))
    if disc == 5:
        return Subscribe(list(seg.read_byte_list(union_ptr)))
    if disc == 6:
        return Unsubscribe(list(seg.read_byte_list(union_ptr)))
    if disc == 7:
        return UserSync(seg.read_byte_list(union_ptr))
    if disc == 8:
        return TopicSync(seg.read_byte_list(union_ptr))
    raise DeserializeError(f"unknown discriminant {disc}")


def parse_offsets(data: bytes):
    """Structural parse returning byte offsets into ``data`` — the host mirror
    of the K4 device kernel (csrc/hip/dataplane.hip k4_parse_batch).

    Returns dict with: disc, payload_off, payload_len, topics_off, topics_cnt,
    recipient (bytes), timestamp.  Offsets are 0 whenever the length is 0
    (canonicalized, matching the kernel).  Raises DeserializeError on garbage.
    """

    def span(seg: _SegmentReader, seg_base: int, ptr_word: int):
        val = seg.u64(ptr_word)
        if val == 0:
            return 0, 0
        raw = seg.read_byte_list(ptr_word)
        if not raw:
            return 0, 0
        b = (val >> 2) & 0x3FFFFFFF
        if b & 0x20000000:
            b -= 0x40000000
        tgt = ptr_word + 1 + b
        return seg_base + tgt * 8, len(raw)

    out = {
        "disc": -1, "payload_off": 0, "payload_len": 0,
        "topics_off": 0, "topics_cnt": 0, "recipient": b"", "timestamp": 0,
    }
    if len(data) < 16:
        raise DeserializeError("short buffer")
    seg_count_m1, nwords = struct.unpack_from("<II", data, 0)
    if seg_count_m1 != 0 or 8 + nwords * 8 > len(data):
        raise DeserializeError("bad stream header")
    seg = _SegmentReader(data[8 : 8 + nwords * 8])
    seg_base = 8
    tgt, dw, pw = seg.read_struct_ptr(0)
    if dw < 1 or pw < 1:
        raise DeserializeError("malformed Message struct")
    disc = struct.unpack_from("<H", seg.data, tgt * 8)[0]
    up = tgt + dw
    if disc == 0:
        it, idw, ipw = seg.read_struct_ptr(up)
        if idw < 1 or ipw < 2:
            raise DeserializeError("malformed AuthenticateWithKey")
        out["timestamp"] = seg.u64(it)
        out["payload_off"], out["payload_len"] = span(seg, seg_base, it + idw)
        out["topics_off"], out["topics_cnt"] = span(seg, seg_base, it + idw + 1)
    elif disc == 1:
        it, idw, _ = seg.read_struct_ptr(up)
        if idw < 1:
            raise DeserializeError("malformed AuthenticateWithPermit")
        out["timestamp"] = seg.u64(it)
    elif disc == 2:
        it, idw, ipw = seg.read_struct_ptr(up)
        if idw < 1 or ipw < 1:
            raise DeserializeError("malformed AuthenticateResponse")
        out["timestamp"] = seg.u64(it)
        out["payload_off"], out["payload_len"] = span(seg, seg_base, it + idw)
    elif disc == 3:
        it, idw, ipw = seg.read_struct_ptr(up)
        if ipw < 2:
            raise DeserializeError("malformed Direct")
        out["recipient"] = seg.read_byte_list(it + idw)
        out["payload_off"], out["payload_len"] = span(seg, seg_base, it + idw + 1)
    elif disc == 4:
        it, idw, ipw = seg.read_struct_ptr(up)
        if ipw < 2:
            raise DeserializeError("malformed Broadcast")
        out["topics_off"], out["topics_cnt"] = span(seg, seg_base, it + idw)
        out["payload_off"], out["payload_len"] = span(seg, seg_base, it + idw + 1)
    elif disc in (5, 6):
        out["topics_off"], out["topics_cnt"] = span(seg, seg_base, up)
    elif disc in (7, 8):
        out["payload_off"], out["payload_len"] = span(seg, seg_base, up)
    else:
        raise DeserializeError(f"unknown discriminant {disc}")
    out["disc"] = disc
    return out
