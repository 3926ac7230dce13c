// Native C++ unit tests for the host core (wire codec + BLS), built with
// ASan/UBSan by scripts/native_sanitize.sh — the C++ analog of the
// reference's safety posture (#![forbid(unsafe_code)] + clippy lanes,
// "plan TSan/ASan CI lanes" for the C++ rewrite).

#include <cassert>
#include <cstdio>
#include <cstring>
#include <random>
#include <string>
#include <vector>

#include "../bls/bls.h"
#include "../net/pump.h"
#include "../net/udp_stream.h"
#include "../state/versioned_map.h"
#include "../wire/message.h"

#include <sys/socket.h>
#include <unistd.h>

using namespace bn254;

static int checks = 0;
#define CHECK(x)                                                                    \
    do {                                                                            \
        if (!(x)) {                                                                 \
            fprintf(stderr, "CHECK failed at %s:%d: %s\n", __FILE__, __LINE__, #x); \
            return 1;                                                               \
        }                                                                           \
        ++checks;                                                                   \
    } while (0)

static int test_wire_roundtrip() {
    std::vector<uint8_t> topics = {1, 2, 255};
    std::vector<uint8_t> payload(1024);
    for (size_t i = 0; i < payload.size(); ++i) payload[i] = (uint8_t)i;
    auto raw = wire::serialize_broadcast(topics.data(), topics.size(), payload.data(),
                                         payload.size());
    wire::Parsed p;
    CHECK(wire::deserialize(raw.data(), raw.size(), &p));
    CHECK(p.disc == wire::BROADCAST);
    CHECK(p.topics == topics);
    CHECK(p.payload == payload);

    auto raw2 = wire::serialize_direct((const uint8_t*)"rcpt", 4, payload.data(), 16);
    CHECK(wire::deserialize(raw2.data(), raw2.size(), &p));
    CHECK(p.disc == wire::DIRECT);
    CHECK(std::string(p.recipient.begin(), p.recipient.end()) == "rcpt");

    auto raw3 = wire::serialize_authenticate_response(42, "ok:1738");
    CHECK(wire::deserialize(raw3.data(), raw3.size(), &p));
    CHECK(p.disc == wire::AUTHENTICATE_RESPONSE && p.timestamp == 42 && p.context == "ok:1738");
    return 0;
}

static int test_wire_fuzz_no_crash() {
    // Bounds-checked parser must reject arbitrary garbage without UB
    // (this is what ASan/UBSan actually verifies here).
    std::mt19937_64 rng(7);
    auto seed_msg = wire::serialize_broadcast(nullptr, 0, (const uint8_t*)"x", 1);
    for (int iter = 0; iter < 20000; ++iter) {
        std::vector<uint8_t> buf = seed_msg;
        int flips = 1 + (int)(rng() % 8);
        for (int f = 0; f < flips; ++f) buf[rng() % buf.size()] ^= (uint8_t)(rng() & 0xff);
        size_t len = (iter % 3 == 0) ? rng() % (buf.size() + 1) : buf.size();
        wire::Parsed p;
        (void)wire::deserialize(buf.data(), len, &p);  // may fail; must not crash
    }
    ++checks;
    return 0;
}

static int test_field_arithmetic() {
    Fp a = Fp::from_u64(123456789);
    Fp b = Fp::from_u64(987654321);
    CHECK(Fp::mul(a, b) == Fp::mul(b, a));
    CHECK(Fp::mul(a, a.inv()) == Fp::one());
    Fp2 x{a, b};
    CHECK(Fp2::mul(x, x.inv()) == Fp2::one());
    Fp12 f{{Fp2{a, b}, Fp2{b, a}, Fp2{a, a}}, {Fp2{b, b}, Fp2{a, b}, Fp2{b, a}}};
    CHECK(Fp12::mul(f, f.inv()) == Fp12::one());
    return 0;
}

static int test_bls_end_to_end() {
    // pairing bilinearity + verify flow, exercised under sanitizers
    G1 P = g1_generator();
    G2 Q = g2_generator();
    Fp px, py;
    P.to_affine(px, py);
    Fp2 qx, qy;
    Q.to_affine(qx, qy);
    Fp12 e = pairing(px, py, G2Affine{qx, qy});
    CHECK(!(e == Fp12::one()));

    uint8_t scratch[64];
    memcpy(scratch, "sanitizer-test-message", 22);
    Fp hx, hy;
    CHECK(bls::hash_to_g1_with_scratch(scratch, 22, hx, hy));
    CHECK(g1_on_curve(hx, hy));
    return 0;
}

static int test_pump_frames_and_close() {
    // socketpair through the epoll pump: framed roundtrip, burst order,
    // soft-close flush, hard-close detection — under ASan with the pump
    // thread live (threaded memory-safety exercise)
    net::Pump pump;
    int fds[2];
    CHECK(socketpair(AF_UNIX, SOCK_STREAM, 0, fds) == 0);
    int64_t a = pump.add(fds[0]);
    int64_t b = pump.add(fds[1]);
    // burst of 100 frames lands intact and in order
    for (int i = 0; i < 100; ++i) {
        std::string msg = "frame-" + std::to_string(i);
        CHECK(pump.send(a, msg.data(), msg.size()));
    }
    std::vector<std::string> got;
    for (int spins = 0; spins < 2000 && got.size() < 100; ++spins) {
        auto r = pump.recv_batch(b, 100);
        for (auto& f : r.first) got.emplace_back(std::move(f));
        usleep(1000);
    }
    CHECK(got.size() == 100);
    for (int i = 0; i < 100; ++i) CHECK(got[i] == "frame-" + std::to_string(i));
    // big frame (1 MiB) reassembles across many reads
    std::string big(1 << 20, 'x');
    CHECK(pump.send(b, big.data(), big.size()));
    std::string rx;
    for (int spins = 0; spins < 2000 && rx.empty(); ++spins) {
        auto r = pump.recv_batch(a, 4);
        if (!r.first.empty()) rx = std::move(r.first[0]);
        usleep(1000);
    }
    CHECK(rx == big);
    // soft close flushes the last frame before closing
    CHECK(pump.send(a, "last", 4));
    pump.soft_close(a);
    pump.forget(a);
    bool saw_last = false, closed = false;
    for (int spins = 0; spins < 2000 && !closed; ++spins) {
        auto r = pump.recv_batch(b, 4);
        for (auto& f : r.first) saw_last = saw_last || f == "last";
        closed = r.second;
        usleep(1000);
    }
    CHECK(saw_last);
    CHECK(closed);
    pump.hard_close(b);
    pump.forget(b);
    pump.stop();
    return 0;
}


static int test_pump_ingest_and_batch_drain() {
    // round-2 paths under ASan/TSan: frame classification + contiguous
    // ingest accumulation, recv_drain counting, send_rings_batch with the
    // lock-free frame build, and send_raw pre-framed bursts
    net::Pump pump;
    int fds[2];
    CHECK(socketpair(AF_UNIX, SOCK_STREAM, 0, fds) == 0);
    int64_t a = pump.add(fds[0]);
    int64_t b = pump.add(fds[1]);
    pump.set_ingest(b);

    // a Broadcast wire message, framed; classify_frame must see disc=4
    uint8_t topics[2] = {3, 9};
    auto bc = wire::serialize_broadcast(topics, 2, (const uint8_t*)"payload!", 8);
    std::string burst;
    for (int i = 0; i < 64; ++i) {
        uint32_t be = htonl((uint32_t)bc.size());
        burst.append((const char*)&be, 4);
        burst.append((const char*)bc.data(), bc.size());
    }
    CHECK(pump.send_raw(a, burst.data(), burst.size()));
    net::Pump::IngestBatch ib;
    for (int spins = 0; spins < 2000 && ib.offs.size() < 64; ++spins) {
        auto got = pump.recv_ingest(b);
        if (!got.blob.empty()) {
            ib.blob += got.blob;
            for (auto o : got.offs) ib.offs.push_back((int64_t)ib.blob.size() -
                                                      (int64_t)got.blob.size() + o);
            for (auto& m : got.meta) ib.meta.push_back(m);
        }
        usleep(1000);
    }
    CHECK(ib.offs.size() == 64);
    CHECK(ib.blob.size() == 64 * bc.size());
    for (auto& m : ib.meta) {
        CHECK(m.disc == wire::BROADCAST);
        CHECK(m.topics_cnt == 2);
        CHECK(ib.blob[m.topics_off] == 3 && ib.blob[m.topics_off + 1] == 9);
    }

    // batched egress drain: two fake rings (16 B headers + padded payloads)
    // through send_rings_batch; receiver counts them via recv_drain
    auto make_rec = [](uint32_t seq, const std::string& p) {
        std::string r(16, '\0');
        uint32_t len = (uint32_t)p.size();
        memcpy(&r[0], &len, 4);
        memcpy(&r[4], &seq, 4);
        r += p;
        r.resize(16 + ((p.size() + 15) & ~(size_t)15), '\0');
        return r;
    };
    std::string ring1 = make_rec(0, "alpha") + make_rec(1, "beta");
    std::string ring2 = make_rec(7, "gamma-longer-payload");
    std::string base = ring1 + ring2;
    int fds2[2];
    CHECK(socketpair(AF_UNIX, SOCK_STREAM, 0, fds2) == 0);
    int64_t c = pump.add(fds2[0]);
    int64_t d = pump.add(fds2[1]);
    auto counts = pump.send_rings_batch(
        (const uint8_t*)base.data(), {c, c},
        {0, (int64_t)ring1.size()},
        {(int64_t)ring1.size(), (int64_t)base.size()});
    CHECK(counts.size() == 4);  // [count, payload] x 2
    CHECK(counts[0] == 2 && counts[1] == (int64_t)(5 + 4));
    CHECK(counts[2] == 1 && counts[3] == 20);
    int64_t n_total = 0;
    std::string last;
    for (int spins = 0; spins < 2000 && n_total < 3; ++spins) {
        auto r = pump.recv_drain(d);
        n_total += std::get<0>(r);
        if (std::get<0>(r)) last = std::get<2>(r);
        usleep(1000);
    }
    CHECK(n_total == 3);
    CHECK(last == "gamma-longer-payload");
    pump.hard_close(c);
    pump.forget(c);
    pump.forget(d);

    // gone connection reports -1 in the batch result
    pump.hard_close(b);
    auto counts2 = pump.send_rings_batch((const uint8_t*)base.data(), {b},
                                         {0}, {(int64_t)ring1.size()});
    CHECK(counts2[0] == -1);
    pump.forget(a);
    pump.forget(b);
    pump.stop();
    return 0;
}

static int test_crdt_delta_fuzz() {
    // corrupted sync payloads under ASan: bit flips, truncations and raw
    // garbage must be rejected without any out-of-bounds access
    std::mt19937 rng(1234);
    state::VersionedMap vm("cid-asan");
    for (int i = 0; i < 8; ++i)
        vm.insert("key-" + std::to_string(i), std::string(i, 'v'));
    auto delta = state::VersionedMap::serialize_delta(vm.get_full());
    for (int trial = 0; trial < 4000; ++trial) {
        std::vector<uint8_t> blob(delta.begin(), delta.end());
        int kind = rng() % 3;
        if (kind == 0 && !blob.empty()) {
            for (int f = 0; f < 6; ++f)
                blob[rng() % blob.size()] ^= 1u << (rng() % 8);
        } else if (kind == 1) {
            blob.resize(rng() % (blob.size() + 1));
        } else {
            blob.resize(rng() % 64);
            for (auto& b : blob) b = (uint8_t)rng();
        }
        std::map<std::string, state::Versioned> d;
        bool ok = state::VersionedMap::deserialize_delta(
            blob.data(), blob.size(), &d);
        if (ok) vm.merge(d);  // parsed garbage must still merge safely
    }
    vm.insert("probe", "ok");
    CHECK(vm.get("probe").has_value());  // map still alive
    ++checks;
    return 0;
}


// UDP reliable stream (the QUIC-profile native datapath): handshake,
// bidirectional multi-MB transfer with 10% deterministic loss both ways,
// FIN/eof, and abort — under ASan/UBSan/TSan (live epoll threads).
static int test_udp_stream_reliability() {
    net::UdpPump srv, cli;
    int port = srv.bind("127.0.0.1", 0);
    CHECK(port > 0);
    srv.debug_set_loss(100);
    cli.debug_set_loss(100);
    const uint64_t cid = 0x1122334455667788ull;
    CHECK(cli.connect("127.0.0.1", port, cid, 9));
    for (int spins = 0; spins < 4000 && cli.client_status(cid) != 1; ++spins)
        usleep(1000);
    CHECK(cli.client_status(cid) == 1);
    auto acc = srv.accept_poll();
    for (int spins = 0; spins < 4000 && acc.empty(); ++spins) {
        usleep(1000);
        acc = srv.accept_poll();
    }
    CHECK(acc.size() == 1 && acc[0].first == cid && acc[0].second == 9);
    ++checks;

    // 2 MiB each way, patterned payloads, verified byte-for-byte
    std::string big(2u << 20, '\0');
    for (size_t i = 0; i < big.size(); ++i) big[i] = (char)(i * 31 + 7);
    CHECK(cli.stream_write(cid, big.data(), big.size()));
    std::string got;
    for (int spins = 0; spins < 20000 && got.size() < big.size(); ++spins) {
        auto r = srv.recv_stream(cid);
        got += std::get<0>(r);
        if (std::get<0>(r).empty()) usleep(500);
    }
    CHECK(got == big);
    ++checks;
    std::string big2(3u << 20, '\0');
    for (size_t i = 0; i < big2.size(); ++i) big2[i] = (char)(i * 13 + 1);
    CHECK(srv.stream_write(cid, big2.data(), big2.size()));
    got.clear();
    for (int spins = 0; spins < 20000 && got.size() < big2.size(); ++spins) {
        auto r = cli.recv_stream(cid);
        got += std::get<0>(r);
        if (std::get<0>(r).empty()) usleep(500);
    }
    CHECK(got == big2);
    CHECK(cli.tx_backlog(cid) >= 0);
    ++checks;

    // graceful close: FIN survives the lossy path and surfaces as eof
    cli.graceful_close(cid);
    bool eof = false, closed = false;
    for (int spins = 0; spins < 8000 && !eof && !closed; ++spins) {
        auto r = srv.recv_stream(cid);
        eof = std::get<1>(r);
        closed = std::get<2>(r);
        usleep(1000);
    }
    CHECK(eof || closed);
    ++checks;
    srv.abort_conn(cid);
    srv.forget(cid);
    cli.forget(cid);
    cli.stop();
    srv.stop();
    return 0;
}


// the single-pass serializers must emit BYTE-IDENTICAL wire to the Builder
// path for every padding case, and the view parser must agree with the
// copying parser
static int test_wire_single_pass_equivalence() {
    std::mt19937 rng(77);
    for (int trial = 0; trial < 300; ++trial) {
        size_t rlen = rng() % 40;
        size_t mlen = (trial < 50) ? rng() % 9 : rng() % 5000;
        std::vector<uint8_t> rcpt(rlen), msg(mlen);
        for (auto& b : rcpt) b = (uint8_t)rng();
        for (auto& b : msg) b = (uint8_t)rng();
        for (uint16_t disc : {wire::DIRECT, wire::BROADCAST}) {
            auto old_bytes = disc == wire::DIRECT
                ? wire::serialize_direct(rcpt.data(), rlen, msg.data(), mlen)
                : wire::serialize_broadcast(rcpt.data(), rlen, msg.data(), mlen);
            std::vector<uint8_t> neu(wire::payload_msg_wire_bytes(rlen, mlen), 0xAB);
            wire::serialize_payload_msg_into(neu.data(), disc, rcpt.data(), rlen,
                                             msg.data(), mlen);
            CHECK(old_bytes == neu);
            wire::Parsed p1;
            wire::ParsedView p2;
            CHECK(wire::deserialize(neu.data(), neu.size(), &p1));
            CHECK(wire::deserialize_views(neu.data(), neu.size(), &p2));
            CHECK(p1.disc == p2.disc);
            CHECK(p1.payload.size() == p2.payload_len);
            CHECK(p1.payload == std::vector<uint8_t>(p2.payload,
                                                     p2.payload + p2.payload_len));
            if (disc == wire::DIRECT) {
                CHECK(p1.recipient ==
                      std::vector<uint8_t>(p2.recipient, p2.recipient + p2.recipient_len));
            } else {
                CHECK(p1.topics ==
                      std::vector<uint8_t>(p2.topics, p2.topics + p2.topics_len));
            }
        }
    }
    ++checks;
    return 0;
}


// garbage datagrams at a live server endpoint: truncated headers, huge
// bogus offsets, unknown ptypes, STREAM floods for unknown cids — the
// parser must neither crash nor accept a connection
static int test_udp_pump_datagram_fuzz() {
    net::UdpPump srv;
    int port = srv.bind("127.0.0.1", 0);
    CHECK(port > 0);
    int fd = ::socket(AF_INET, SOCK_DGRAM, 0);
    CHECK(fd >= 0);
    sockaddr_in a{};
    a.sin_family = AF_INET;
    a.sin_port = htons((uint16_t)port);
    a.sin_addr.s_addr = inet_addr("127.0.0.1");
    std::mt19937 rng(99);
    std::vector<uint8_t> pkt;
    for (int i = 0; i < 3000; ++i) {
        size_t len = rng() % 128;
        if (i % 7 == 0) len = rng() % 2000;
        pkt.resize(len);
        for (auto& b : pkt) b = (uint8_t)rng();
        if (i % 3 == 0 && len >= 9) pkt[0] = (uint8_t)(rng() % 8);  // plausible ptype
        (void)::sendto(fd, pkt.data(), pkt.size(), 0, (sockaddr*)&a, sizeof(a));
    }
    usleep(50 * 1000);
    // INIT-shaped garbage DOES accept (that is the wire contract); pure
    // garbage must not have wedged the endpoint: a real handshake works
    net::UdpPump cli;
    CHECK(cli.connect("127.0.0.1", port, 0xF00Dull, 1));
    for (int spins = 0; spins < 4000 && cli.client_status(0xF00Dull) != 1; ++spins)
        usleep(1000);
    CHECK(cli.client_status(0xF00Dull) == 1);
    ++checks;
    ::close(fd);
    cli.stop();
    srv.stop();
    return 0;
}


// adversarial lanes: the pump's structural classifier, the ring-record
// parser, and the two wire parsers must be memory-safe on garbage AND the
// copying/view parsers must AGREE on accept/reject + field bytes
static int test_adversarial_parsers() {
    std::mt19937 rng(4242);

    // 1) classify_frame on random/mutated buffers
    {
        auto valid = wire::serialize_broadcast((const uint8_t*)"\x01\x02", 2,
                                               (const uint8_t*)"payload", 7);
        for (int trial = 0; trial < 4000; ++trial) {
            std::vector<uint8_t> b;
            if (trial % 2 == 0) {
                b.assign(valid.begin(), valid.end());
                for (int f = 0; f < 5; ++f)
                    if (!b.empty()) b[rng() % b.size()] ^= 1u << (rng() % 8);
                b.resize(rng() % (b.size() + 1));
            } else {
                b.resize(rng() % 96);
                for (auto& x : b) x = (uint8_t)rng();
            }
            net::FrameMeta m = net::classify_frame(b.data(), b.size());
            if (m.disc >= 0) {  // any views handed out must stay in-bounds
                CHECK((size_t)m.topics_off + m.topics_cnt <= b.size());
                CHECK((size_t)m.recip_off + m.recip_len <= b.size());
            }
        }
        ++checks;
    }

    // 2) build_ring_frames on mutated ring bytes (via a pump batch drain
    //    against a closed-conn id: builds frames, then drops them)
    //    — exercised directly through the static helper's caller
    //    send_rings_batch with id -1 (gone): counts become -1, but the
    //    frame build runs first on the adversarial bytes.
    {
        net::Pump pump;
        for (int trial = 0; trial < 600; ++trial) {
            std::string ring;
            int recs = rng() % 4;
            for (int r = 0; r < recs; ++r) {
                uint32_t len = rng() % 64;
                uint32_t seq = rng();
                std::string payload(len, (char)(rng() & 0xFF));
                char hdr[16] = {0};
                memcpy(hdr, &len, 4);
                memcpy(hdr + 4, &seq, 4);
                ring += std::string(hdr, 16) + payload;
                ring += std::string((16 - (payload.size() % 16)) % 16, '\0');
            }
            // mutate: flip bytes incl. the length fields
            std::vector<char> buf(ring.begin(), ring.end());
            for (int f = 0; f < 6 && !buf.empty(); ++f)
                buf[rng() % buf.size()] ^= 1 << (rng() % 8);
            size_t wpos = buf.empty() ? 0 : rng() % (buf.size() + 1);
            auto counts = pump.send_rings_batch(
                (const uint8_t*)buf.data(), {99999},
                {0}, {(int64_t)wpos});
            CHECK(counts.size() == 2 && counts[0] == -1);  // conn unknown
        }
        ++checks;
    }

    // 3) deserialize vs deserialize_views agreement on mutated wire
    {
        auto base = wire::serialize_direct((const uint8_t*)"user-1", 6,
                                           (const uint8_t*)"msg-payload", 11);
        for (int trial = 0; trial < 6000; ++trial) {
            std::vector<uint8_t> b(base.begin(), base.end());
            int kind = rng() % 3;
            if (kind == 0) {
                for (int f = 0; f < 4; ++f)
                    b[rng() % b.size()] ^= 1u << (rng() % 8);
            } else if (kind == 1) {
                b.resize(rng() % (b.size() + 1));
            } else {
                b.resize(16 + rng() % 128);
                for (auto& x : b) x = (uint8_t)rng();
            }
            wire::Parsed p1;
            wire::ParsedView p2;
            bool ok1 = wire::deserialize(b.data(), b.size(), &p1);
            bool ok2 = wire::deserialize_views(b.data(), b.size(), &p2);
            CHECK(ok1 == ok2);
            if (ok1) {
                CHECK(p1.disc == p2.disc);
                CHECK(p1.payload ==
                      std::vector<uint8_t>(p2.payload, p2.payload + p2.payload_len));
                CHECK(p1.recipient ==
                      std::vector<uint8_t>(p2.recipient,
                                           p2.recipient + p2.recipient_len));
                CHECK(p1.topics ==
                      std::vector<uint8_t>(p2.topics, p2.topics + p2.topics_len));
                CHECK(p1.context == p2.context);
            }
        }
        ++checks;
    }
    return 0;
}

int main() {
    if (test_wire_roundtrip()) return 1;
    if (test_wire_fuzz_no_crash()) return 1;
    if (test_field_arithmetic()) return 1;
    if (test_bls_end_to_end()) return 1;
    if (test_pump_frames_and_close()) return 1;
    if (test_pump_ingest_and_batch_drain()) return 1;
    if (test_crdt_delta_fuzz()) return 1;
    if (test_udp_stream_reliability()) return 1;
    if (test_wire_single_pass_equivalence()) return 1;
    if (test_udp_pump_datagram_fuzz()) return 1;
    if (test_adversarial_parsers()) return 1;
    printf("native tests OK (%d checks)\n", checks);
    return 0;
}
