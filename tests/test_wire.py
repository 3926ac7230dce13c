"""Serde round-trip parity for every message type — the tier-1 tests of the
reference (``cdn-proto/src/message.rs:397-457``), plus capnp-format structural
checks and malformed-input rejection."""

import struct

import pytest

from pushcdn_amd.proto import message as m
from pushcdn_amd.proto.errors import DeserializeError


ALL_MESSAGES = [
    m.AuthenticateWithKey(public_key=b"\x01" * 64, timestamp=345234, signature=b"\x02" * 64),
    m.AuthenticateWithPermit(permit=1234),
    m.AuthenticateResponse(permit=0, context="some failure reason"),
    m.AuthenticateResponse(permit=2**64 - 1, context=""),
    m.Direct(recipient=b"12345", message=b"123456789"),
    m.Direct(recipient=b"", message=b""),
    m.Broadcast(topics=[99, 112], message=b"hello bye"),
    m.Broadcast(topics=[], message=b"x"),
    m.Subscribe(topics=[0, 1, 2, 255]),
    m.Unsubscribe(topics=[]),
    m.UserSync(data=b"123456273834"),
    m.TopicSync(data=b"\x00" * 17),
]


@pytest.mark.parametrize("msg", ALL_MESSAGES, ids=lambda x: type(x).__name__)
def test_roundtrip(msg):
    assert m.deserialize(m.serialize(msg)) == msg


def test_stream_header_is_single_segment():
    data = m.serialize(m.Direct(b"r", b"m"))
    seg_count_m1, nwords = struct.unpack_from("<II", data, 0)
    assert seg_count_m1 == 0
    assert len(data) == 8 + nwords * 8


def test_discriminants_match_reference():
    # messages_capnp.rs:77-117 — the wire discriminants must be stable.
    expected = [
        (m.AuthenticateWithKey(b"", 0, b""), 0),
        (m.AuthenticateWithPermit(0), 1),
        (m.AuthenticateResponse(0, ""), 2),
        (m.Direct(b"", b""), 3),
        (m.Broadcast([], b""), 4),
        (m.Subscribe([]), 5),
        (m.Unsubscribe([]), 6),
        (m.UserSync(b""), 7),
        (m.TopicSync(b""), 8),
    ]
    for msg, disc in expected:
        data = m.serialize(msg)
        # Root struct pointer at word 0 points at the Message struct whose
        # first data word's low u16 is the discriminant.
        root = struct.unpack_from("<Q", data, 8)[0]
        assert root & 3 == 0
        b = (root >> 2) & 0x3FFFFFFF
        msg_word = 0 + 1 + b
        got = struct.unpack_from("<H", data, 8 + msg_word * 8)[0]
        assert got == disc, type(msg).__name__


def test_rejects_garbage():
    with pytest.raises(DeserializeError):
        m.deserialize(b"")
    with pytest.raises(DeserializeError):
        m.deserialize(b"\x00" * 7)
    with pytest.raises(DeserializeError):
        m.deserialize(b"\xff" * 64)


def test_rejects_truncated():
    data = m.serialize(m.Broadcast([1, 2], b"y" * 100))
    for cut in (9, 17, len(data) - 8, len(data) - 1):
        with pytest.raises(DeserializeError):
            m.deserialize(data[:cut])


def test_rejects_out_of_bounds_pointer():
    data = bytearray(m.serialize(m.UserSync(b"abcd")))
    # Corrupt the union pointer to point far out of the segment.
    # Find message struct: root ptr at word 0.
    root = struct.unpack_from("<Q", data, 8)[0]
    b = (root >> 2) & 0x3FFFFFFF
    union_ptr_word = 1 + b + 1  # msg struct data word + 1
    bad = 1 | ((0x1000 & 0x3FFFFFFF) << 2) | (2 << 32) | (4 << 35)
    struct.pack_into("<Q", data, 8 + union_ptr_word * 8, bad)
    with pytest.raises(DeserializeError):
        m.deserialize(bytes(data))


def test_large_payload_roundtrip():
    payload = bytes(range(256)) * 4096  # 1 MiB
    msg = m.Broadcast(topics=[7], message=payload)
    assert m.deserialize(m.serialize(msg)) == msg


def test_max_message_size_constant():
    # reference cdn-proto/src/lib.rs:25
    assert m.MAX_MESSAGE_SIZE == (2**32 - 1) // 8


def test_cpp_codec_byte_identical_to_python():
    """The native C++ wire codec (csrc/wire/message.h) must produce the
    exact bytes of the Python reference serializer, and parse them back."""
    from pushcdn_amd.proto.message import _get_core, serialize_py, deserialize_py

    core = _get_core()
    assert core is not None, "native codec must build in this environment"
    for msg in ALL_MESSAGES:
        py_bytes = serialize_py(msg)
        cpp_bytes = m.serialize(msg)
        assert cpp_bytes == py_bytes, type(msg).__name__
        assert m.deserialize(cpp_bytes) == msg
        assert deserialize_py(cpp_bytes) == msg


def test_cpp_codec_rejects_garbage():
    from pushcdn_amd.proto.message import _get_core

    core = _get_core()
    assert core.wire_deserialize(b"") is None
    assert core.wire_deserialize(b"\xff" * 64) is None
    data = m.serialize(m.Broadcast([1], b"x" * 50))
    assert core.wire_deserialize(bytes(data)) is not None
    for cut in (9, 17, len(data) - 8):
        assert core.wire_deserialize(bytes(data[:cut])) is None


def test_error_taxonomy_hierarchy():
    """The error taxonomy drives reconnect policy (reference error.rs:21-44):
    connection/auth/parse errors are distinct and catchable as the shared
    base, so callers can branch like the reference's Error enum."""
    from pushcdn_amd.proto import errors as e

    for exc in (e.ConnectionError_, e.AuthenticationError, e.ParseError,
                e.DeserializeError, e.SerializeError, e.TopicError,
                e.CryptoError, e.DiscoveryError):
        assert issubclass(exc, e.CdnError)
        try:
            raise exc("boom")
        except e.CdnError as caught:
            assert "boom" in str(caught)
