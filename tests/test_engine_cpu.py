"""Broker engine semantics on the CPU reference path (no GPU):
broadcast fan-out to subscribers only, per-user FIFO order, direct routing,
ring records, drops on ring-full."""


from pushcdn_amd.broker.gpu_engine import GpuBrokerEngine, parse_ring_records
from pushcdn_amd.proto import message as m


def make_batch(msgs):
    offsets = [0]
    buf = b""
    for msg in msgs:
        b = m.serialize(msg)
        buf += b
        offsets.append(len(buf))
    return buf, offsets


def new_engine(n_users=130, ring_bytes=1 << 14):
    return GpuBrokerEngine(device="cpu", n_users=n_users, ring_bytes=ring_bytes,
                           direct_table_size=256, use_gpu_ops=False)


def test_broadcast_reaches_only_subscribers():
    eng = new_engine()
    eng.subscribe(3, [7])
    eng.subscribe(77, [7])   # crosses the 64-bit word boundary
    eng.subscribe(5, [9])
    batch, offsets = make_batch([m.Broadcast([7], b"hello-broadcast")])
    buf, off = eng.ingest(batch, offsets)
    stats = eng.tick(buf, off, host_batch=batch, host_offsets=offsets)
    assert stats.n_messages == 1
    wpos = eng.drain_cursors()
    for u in range(eng.n_users):
        recs = parse_ring_records(eng.read_ring(u), int(wpos[u]))
        if u in (3, 77):
            assert [p for _, p in recs] == [b"hello-broadcast"]
        else:
            assert recs == []


def test_per_user_fifo_order():
    eng = new_engine()
    eng.subscribe(10, [1, 2])
    msgs = [m.Broadcast([1], b"first"), m.Broadcast([2], b"second"), m.Broadcast([1], b"third")]
    batch, offsets = make_batch(msgs)
    buf, off = eng.ingest(batch, offsets)
    eng.tick(buf, off, host_batch=batch, host_offsets=offsets)
    wpos = eng.drain_cursors()
    recs = parse_ring_records(eng.read_ring(10), int(wpos[10]))
    assert [p for _, p in recs] == [b"first", b"second", b"third"]
    assert [s for s, _ in recs] == [0, 1, 2]


def test_direct_routing_local():
    eng = new_engine()
    key = b"user-key-42"
    eng.register_direct(key, 42)
    batch, offsets = make_batch([m.Direct(key, b"direct-payload")])
    buf, off = eng.ingest(batch, offsets)
    eng.tick(buf, off, host_batch=batch, host_offsets=offsets)
    wpos = eng.drain_cursors()
    recs = parse_ring_records(eng.read_ring(42), int(wpos[42]))
    assert [p for _, p in recs] == [b"direct-payload"]


def test_direct_unknown_recipient_dropped():
    eng = new_engine()
    batch, offsets = make_batch([m.Direct(b"nobody", b"x")])
    buf, off = eng.ingest(batch, offsets)
    eng.tick(buf, off, host_batch=batch, host_offsets=offsets)
    wpos = eng.drain_cursors()
    assert int(wpos.sum()) == 0


def test_ring_full_drops():
    eng = new_engine(n_users=2, ring_bytes=1 << 8)  # 256-byte rings
    eng.subscribe(0, [1])
    payload = b"z" * 200  # record = 16 + 208 = 224 bytes; second one drops
    msgs = [m.Broadcast([1], payload), m.Broadcast([1], payload)]
    batch, offsets = make_batch(msgs)
    buf, off = eng.ingest(batch, offsets)
    stats = eng.tick(buf, off, host_batch=batch, host_offsets=offsets)
    assert stats.n_drops == 1
    wpos = eng.drain_cursors()
    recs = parse_ring_records(eng.read_ring(0), int(wpos[0]))
    assert len(recs) == 1


def test_mixed_batch():
    eng = new_engine()
    eng.subscribe(1, [5])
    eng.register_direct(b"k2", 2)
    msgs = [
        m.Broadcast([5], b"bcast"),
        m.Direct(b"k2", b"direct"),
        m.Subscribe([1, 2]),   # no delivery
        m.UserSync(b"sync"),   # no delivery
    ]
    batch, offsets = make_batch(msgs)
    buf, off = eng.ingest(batch, offsets)
    eng.tick(buf, off, host_batch=batch, host_offsets=offsets)
    wpos = eng.drain_cursors()
    assert [p for _, p in parse_ring_records(eng.read_ring(1), int(wpos[1]))] == [b"bcast"]
    assert [p for _, p in parse_ring_records(eng.read_ring(2), int(wpos[2]))] == [b"direct"]


def test_unsubscribe_stops_delivery():
    eng = new_engine()
    eng.subscribe(4, [3])
    eng.unsubscribe(4, [3])
    batch, offsets = make_batch([m.Broadcast([3], b"gone")])
    buf, off = eng.ingest(batch, offsets)
    eng.tick(buf, off, host_batch=batch, host_offsets=offsets)
    wpos = eng.drain_cursors()
    assert int(wpos[4]) == 0


def test_subscribe_all_bulk():
    eng = new_engine(n_users=70)
    eng.subscribe_all([0])
    batch, offsets = make_batch([m.Broadcast([0], b"everyone")])
    buf, off = eng.ingest(batch, offsets)
    stats = eng.tick(buf, off, host_batch=batch, host_offsets=offsets)
    wpos = eng.drain_cursors()
    delivered = sum(1 for u in range(70) if int(wpos[u]) > 0)
    assert delivered == 70


def test_parse_ring_records_seq_sort_with_wrap():
    """Drained ring records come back in seq order even when the u32
    sequence counter wraps mid-batch (the ring WRITE order is claim order;
    K5b's atomic direct claims interleave within a tick)."""
    import struct

    from pushcdn_amd.broker.gpu_engine import parse_ring_records, ring_rec

    def rec(seq, payload):
        r = struct.pack("<IIII", len(payload), seq, 0, 0) + payload
        return r + b"\x00" * (ring_rec(len(payload)) - len(r))

    ring = (rec(1, b"c") + rec(0xFFFFFFFE, b"a")
            + rec(0, b"b2") + rec(0xFFFFFFFF, b"b1"))
    out = parse_ring_records(ring, len(ring))
    assert [p for _, p in out] == [b"a", b"b1", b"b2", b"c"]

    # no-wrap ordinary case
    ring2 = rec(7, b"y") + rec(5, b"x") + rec(9, b"z")
    assert [p for _, p in parse_ring_records(ring2, len(ring2))] == [b"x", b"y", b"z"]
