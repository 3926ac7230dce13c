// Torch extension bindings for the MI355X broker data-plane kernels
// (csrc/hip/dataplane.hip). PyTorch owns every HBM buffer; these calls
// invoke the extern "C" launchers defined next to the kernels.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <hip/hip_runtime.h>

#include <cstdint>

struct ParseOut {
    int32_t* disc;
    int64_t* payload_off;
    int32_t* payload_len;
    int64_t* topics_off;
    int32_t* topics_cnt;
    uint64_t* recip_hash;
    uint64_t* timestamp;
};

struct alignas(16) PairRec { int32_t user; int32_t msg; int64_t dst; };

extern "C" {
void launch_k4_parse(const uint8_t*, const int64_t*, int32_t, uint64_t, ParseOut, hipStream_t);
void launch_k2a_topic_mask(const uint64_t*, const uint8_t*, const int64_t*, const int32_t*,
                           const int32_t*, uint64_t*, int32_t, int32_t, hipStream_t);
void launch_k2b_count(const uint64_t*, int32_t, int32_t, int32_t, int32_t*, hipStream_t);
void launch_k2b_emit(const uint64_t*, const int64_t*, const int32_t*, const int32_t*, int32_t,
                     int32_t, int32_t, int64_t, uint64_t*, PairRec*, uint32_t*, hipStream_t);
void launch_k3_fanout(const uint8_t*, const int64_t*, const int32_t*, const PairRec*,
                      const uint32_t*, int32_t, uint8_t*, hipStream_t);
void launch_k5_direct_lookup(const uint64_t*, const int32_t*, int64_t, const uint64_t*, int32_t,
                             int32_t*, hipStream_t);
void launch_k2c_apply_subs(uint64_t*, const uint8_t*, const int64_t*, const int32_t*,
                           const int32_t*, const int32_t*, int32_t, int32_t, hipStream_t);
void launch_k1_bls_verify(const uint8_t*, const uint8_t*, uint8_t*, const int64_t*, int32_t,
                          int32_t*, hipStream_t);
void launch_k1_hash_to_g1(uint8_t*, const int64_t*, int32_t, uint8_t*, hipStream_t);
void launch_k1_precompute_g2_lines(uint8_t*, int32_t*, hipStream_t);
void launch_k7_compact_rings(const uint8_t*, int64_t, const int64_t*, const int64_t*,
                             uint8_t*, int32_t, int32_t, hipStream_t);
void launch_k1_bls_verify2(const uint8_t*, const uint8_t*, uint8_t*, const int64_t*,
                           const uint8_t*, int32_t, int32_t*, hipStream_t);
void launch_k1_bls_verify_wave(const uint8_t*, const uint8_t*, uint8_t*, const int64_t*,
                               const uint8_t*, const uint64_t*, int32_t, int32_t*,
                               hipStream_t);
void launch_k1_dbg_wave(const uint8_t*, const uint8_t*, uint8_t*, const int64_t*,
                        const uint8_t*, const uint64_t*, int32_t, int32_t, int32_t*,
                        hipStream_t);
void launch_k3_fanout_wave(const uint8_t*, const int64_t*, const int32_t*, const PairRec*,
                           const uint32_t*, const int32_t*, int32_t, uint8_t*, int, int,
                           hipStream_t);
void launch_k3_fanout_flat2(const uint8_t*, const int64_t*, const int32_t*, const PairRec*,
                            uint32_t, const int32_t*, int32_t, int32_t, int32_t, uint8_t*,
                            int, int, hipStream_t);
void launch_k3_fanout_flat3(const uint8_t*, const int64_t*, const int32_t*, const PairRec*,
                            const uint32_t*, const int32_t*, int32_t, int32_t, int32_t,
                            uint8_t*, int, int, hipStream_t);
void launch_k_seq_advance(uint32_t*, int32_t, hipStream_t);
void launch_k5b_emit_direct(const int32_t*, const int32_t*, const int64_t*, const int32_t*,
                            int32_t, int64_t, int32_t, uint64_t*, int32_t*, PairRec*,
                            uint32_t*, hipStream_t);
void launch_k2a_topic_mask_t(const uint64_t*, const uint8_t*, const int64_t*, const int32_t*,
                             const int32_t*, uint64_t*, int32_t, int32_t, hipStream_t);
void launch_k2b_fused_t(const uint64_t*, const int32_t*, int32_t, int32_t, int32_t, int64_t,
                        int32_t, int32_t, uint64_t*, int32_t*, PairRec*, uint32_t*,
                        hipStream_t);
void launch_k2b_blocks_t(const uint64_t*, int32_t, int32_t, int32_t, int64_t, int32_t,
                         int32_t, uint64_t*, int32_t*, int32_t*, int32_t*, int32_t*,
                         int32_t*, int64_t*, PairRec*, uint32_t*, hipStream_t);
}

#define CHECK_DEV(x) TORCH_CHECK(x.is_cuda(), #x " must be on the GPU")
#define CHECK_CONTIG(x) TORCH_CHECK(x.is_contiguous(), #x " must be contiguous")

static inline PairRec* pair_ptr(torch::Tensor& pairs) {
    TORCH_CHECK(pairs.dim() == 2 && pairs.size(1) == 4 && pairs.dtype() == torch::kInt32
                && pairs.is_contiguous(), "pairs must be a contiguous int32 [cap, 4] tensor");
    return (PairRec*)pairs.data_ptr<int32_t>();
}

static inline hipStream_t cur_stream() {
    return at::hip::getCurrentHIPStream().stream();
}

std::vector<torch::Tensor> parse_batch(torch::Tensor buf, torch::Tensor offsets,
                                       uint64_t hash_seed) {
    CHECK_DEV(buf); CHECK_CONTIG(buf); CHECK_DEV(offsets); CHECK_CONTIG(offsets);
    TORCH_CHECK(buf.dtype() == torch::kUInt8 && offsets.dtype() == torch::kInt64);
    int32_t M = (int32_t)offsets.size(0) - 1;
    TORCH_CHECK(M >= 0);
    auto o32 = torch::TensorOptions().dtype(torch::kInt32).device(buf.device());
    auto o64 = torch::TensorOptions().dtype(torch::kInt64).device(buf.device());
    auto disc = torch::empty({M}, o32);
    auto payload_off = torch::empty({M}, o64);
    auto payload_len = torch::empty({M}, o32);
    auto topics_off = torch::empty({M}, o64);
    auto topics_cnt = torch::empty({M}, o32);
    auto recip_hash = torch::empty({M}, o64);  // bit-cast u64
    auto timestamp = torch::empty({M}, o64);
    if (M > 0) {
        ParseOut out{disc.data_ptr<int32_t>(), payload_off.data_ptr<int64_t>(),
                     payload_len.data_ptr<int32_t>(), topics_off.data_ptr<int64_t>(),
                     topics_cnt.data_ptr<int32_t>(),
                     (uint64_t*)recip_hash.data_ptr<int64_t>(),
                     (uint64_t*)timestamp.data_ptr<int64_t>()};
        launch_k4_parse(buf.data_ptr<uint8_t>(), offsets.data_ptr<int64_t>(), M, hash_seed,
                        out, cur_stream());
    }
    return {disc, payload_off, payload_len, topics_off, topics_cnt, recip_hash, timestamp};
}

torch::Tensor topic_mask(torch::Tensor sub_bitmap, torch::Tensor buf, torch::Tensor topics_off,
                         torch::Tensor topics_cnt, torch::Tensor disc) {
    CHECK_DEV(sub_bitmap); CHECK_CONTIG(sub_bitmap);
    TORCH_CHECK(sub_bitmap.dim() == 2 && sub_bitmap.size(0) == 256);
    int32_t W = (int32_t)sub_bitmap.size(1);
    int32_t M = (int32_t)disc.size(0);
    auto mask = torch::empty({M, (int64_t)W},
                             torch::TensorOptions().dtype(torch::kInt64).device(buf.device()));
    if (M > 0) {
        launch_k2a_topic_mask((const uint64_t*)sub_bitmap.data_ptr<int64_t>(),
                              buf.data_ptr<uint8_t>(), topics_off.data_ptr<int64_t>(),
                              topics_cnt.data_ptr<int32_t>(), disc.data_ptr<int32_t>(),
                              (uint64_t*)mask.data_ptr<int64_t>(), M, W, cur_stream());
    }
    return mask;
}

std::vector<torch::Tensor> assign_emit(torch::Tensor mask, torch::Tensor payload_off,
                                       torch::Tensor payload_len, torch::Tensor ring_wpos,
                                       int64_t ring_bytes, int64_t n_users) {
    CHECK_DEV(mask); CHECK_CONTIG(mask);
    int32_t M = (int32_t)mask.size(0);
    int32_t W = (int32_t)mask.size(1);
    TORCH_CHECK(ring_bytes % 16 == 0, "ring_bytes must be a multiple of 16");
    auto o32 = torch::TensorOptions().dtype(torch::kInt32).device(mask.device());
    auto counts = torch::zeros({n_users}, o32);
    launch_k2b_count((const uint64_t*)mask.data_ptr<int64_t>(), M, W, (int32_t)n_users,
                     counts.data_ptr<int32_t>(), cur_stream());
    auto cum = counts.cumsum(0).to(torch::kInt32);
    auto pair_base = (cum - counts).contiguous();  // exclusive scan
    int64_t total = cum[n_users - 1].item<int32_t>();  // one small D2H sync
    auto pairs = torch::empty({total, 4}, o32);
    auto drops = torch::zeros({1}, o32);
    launch_k2b_emit((const uint64_t*)mask.data_ptr<int64_t>(), payload_off.data_ptr<int64_t>(),
                    payload_len.data_ptr<int32_t>(), pair_base.data_ptr<int32_t>(), M, W,
                    (int32_t)n_users, ring_bytes, (uint64_t*)ring_wpos.data_ptr<int64_t>(),
                    pair_ptr(pairs), (uint32_t*)drops.data_ptr<int32_t>(), cur_stream());
    // split the AoS records back into the golden-test API's three tensors
    auto pair_user = pairs.select(1, 0).contiguous();
    auto pair_msg = pairs.select(1, 1).contiguous();
    auto pair_dst = pairs.slice(1, 2, 4).contiguous().view(torch::kInt64).squeeze(1);
    return {pair_user, pair_msg, pair_dst, drops};
}

void fanout(torch::Tensor buf, torch::Tensor payload_off, torch::Tensor payload_len,
            torch::Tensor pair_user, torch::Tensor pair_msg, torch::Tensor pair_dst,
            torch::Tensor msg_seq, torch::Tensor egress) {
    CHECK_DEV(egress); CHECK_CONTIG(egress);
    int32_t n_pairs = (int32_t)pair_user.size(0);
    if (n_pairs == 0) return;
    auto pairs = torch::empty({n_pairs, 4},
                              torch::TensorOptions().dtype(torch::kInt32).device(buf.device()));
    pairs.select(1, 0).copy_(pair_user);
    pairs.select(1, 1).copy_(pair_msg);
    pairs.slice(1, 2, 4).view(torch::kInt64).squeeze(1).copy_(pair_dst);
    launch_k3_fanout(buf.data_ptr<uint8_t>(), payload_off.data_ptr<int64_t>(),
                     payload_len.data_ptr<int32_t>(), pair_ptr(pairs),
                     (const uint32_t*)msg_seq.data_ptr<int32_t>(), n_pairs,
                     egress.data_ptr<uint8_t>(), cur_stream());
}

torch::Tensor direct_lookup(torch::Tensor table_keys, torch::Tensor table_vals,
                            torch::Tensor query) {
    CHECK_DEV(table_keys); CHECK_DEV(query);
    int64_t S = table_keys.size(0);
    TORCH_CHECK((S & (S - 1)) == 0, "table size must be a power of two");
    int32_t N = (int32_t)query.size(0);
    auto owner = torch::empty({N},
                              torch::TensorOptions().dtype(torch::kInt32).device(query.device()));
    if (N > 0) {
        launch_k5_direct_lookup((const uint64_t*)table_keys.data_ptr<int64_t>(),
                                table_vals.data_ptr<int32_t>(), S,
                                (const uint64_t*)query.data_ptr<int64_t>(), N,
                                owner.data_ptr<int32_t>(), cur_stream());
    }
    return owner;
}

void apply_subs(torch::Tensor sub_bitmap, torch::Tensor buf, torch::Tensor topics_off,
                torch::Tensor topics_cnt, torch::Tensor disc, torch::Tensor user_idx) {
    CHECK_DEV(sub_bitmap); CHECK_CONTIG(sub_bitmap);
    int32_t W = (int32_t)sub_bitmap.size(1);
    int32_t M = (int32_t)disc.size(0);
    if (M == 0) return;
    launch_k2c_apply_subs((uint64_t*)sub_bitmap.data_ptr<int64_t>(), buf.data_ptr<uint8_t>(),
                          topics_off.data_ptr<int64_t>(), topics_cnt.data_ptr<int32_t>(),
                          disc.data_ptr<int32_t>(), user_idx.data_ptr<int32_t>(), M, W,
                          cur_stream());
}

torch::Tensor bls_verify_batch(torch::Tensor vks, torch::Tensor sigs, torch::Tensor msgs,
                               torch::Tensor moff) {
    CHECK_DEV(vks); CHECK_DEV(sigs); CHECK_DEV(msgs); CHECK_DEV(moff);
    CHECK_CONTIG(vks); CHECK_CONTIG(sigs); CHECK_CONTIG(msgs); CHECK_CONTIG(moff);
    int32_t N = (int32_t)moff.size(0) - 1;
    TORCH_CHECK(vks.numel() == (int64_t)N * 128 && sigs.numel() == (int64_t)N * 64);
    auto ok = torch::zeros({N}, torch::TensorOptions().dtype(torch::kInt32).device(vks.device()));
    if (N > 0) {
        launch_k1_bls_verify(vks.data_ptr<uint8_t>(), sigs.data_ptr<uint8_t>(),
                             msgs.data_ptr<uint8_t>(), moff.data_ptr<int64_t>(), N,
                             ok.data_ptr<int32_t>(), cur_stream());
    }
    return ok;
}

void compact_rings(torch::Tensor egress, int64_t ring_bytes, torch::Tensor wpos,
                   torch::Tensor dst_off, torch::Tensor staging, int64_t max_chunks) {
    CHECK_DEV(egress); CHECK_DEV(wpos); CHECK_DEV(dst_off); CHECK_DEV(staging);
    CHECK_CONTIG(egress); CHECK_CONTIG(wpos); CHECK_CONTIG(dst_off); CHECK_CONTIG(staging);
    TORCH_CHECK(wpos.dtype() == torch::kInt64 && dst_off.dtype() == torch::kInt64);
    int32_t n = (int32_t)wpos.size(0);
    launch_k7_compact_rings(egress.data_ptr<uint8_t>(), ring_bytes,
                            wpos.data_ptr<int64_t>(), dst_off.data_ptr<int64_t>(),
                            staging.data_ptr<uint8_t>(), n, (int32_t)max_chunks,
                            cur_stream());
}

torch::Tensor precompute_g2_lines(torch::Tensor device_probe) {
    // fixed-g2 Miller-loop line coefficients (128 records x 192 B); computed
    // once per process by a 1-thread kernel, consumed by bls_verify_batch2
    auto out = torch::zeros({128 * 192},
                            torch::TensorOptions().dtype(torch::kUInt8)
                                .device(device_probe.device()));
    auto n = torch::zeros({1}, torch::TensorOptions().dtype(torch::kInt32)
                                   .device(device_probe.device()));
    launch_k1_precompute_g2_lines(out.data_ptr<uint8_t>(), n.data_ptr<int32_t>(),
                                  cur_stream());
    return out;
}

torch::Tensor bls_verify_batch2(torch::Tensor vks, torch::Tensor sigs, torch::Tensor msgs,
                                torch::Tensor moff, torch::Tensor g2_lines) {
    CHECK_DEV(vks); CHECK_DEV(sigs); CHECK_DEV(msgs); CHECK_DEV(moff); CHECK_DEV(g2_lines);
    CHECK_CONTIG(vks); CHECK_CONTIG(sigs); CHECK_CONTIG(msgs); CHECK_CONTIG(moff);
    CHECK_CONTIG(g2_lines);
    int32_t N = (int32_t)moff.size(0) - 1;
    TORCH_CHECK(vks.numel() == (int64_t)N * 128 && sigs.numel() == (int64_t)N * 64);
    auto ok = torch::zeros({N}, torch::TensorOptions().dtype(torch::kInt32).device(vks.device()));
    if (N > 0) {
        launch_k1_bls_verify2(vks.data_ptr<uint8_t>(), sigs.data_ptr<uint8_t>(),
                              msgs.data_ptr<uint8_t>(), moff.data_ptr<int64_t>(),
                              g2_lines.data_ptr<uint8_t>(), N, ok.data_ptr<int32_t>(),
                              cur_stream());
    }
    return ok;
}

torch::Tensor bls_verify_batch_wave(torch::Tensor vks, torch::Tensor sigs,
                                    torch::Tensor msgs, torch::Tensor moff,
                                    torch::Tensor g2_lines, torch::Tensor rand_r) {
    CHECK_DEV(vks); CHECK_DEV(sigs); CHECK_DEV(msgs); CHECK_DEV(moff); CHECK_DEV(g2_lines);
    CHECK_DEV(rand_r);
    CHECK_CONTIG(vks); CHECK_CONTIG(sigs); CHECK_CONTIG(msgs); CHECK_CONTIG(moff);
    CHECK_CONTIG(g2_lines); CHECK_CONTIG(rand_r);
    int32_t N = (int32_t)moff.size(0) - 1;
    TORCH_CHECK(vks.numel() == (int64_t)N * 128 && sigs.numel() == (int64_t)N * 64);
    TORCH_CHECK(rand_r.numel() >= N && rand_r.dtype() == torch::kInt64);
    auto ok = torch::zeros({N}, torch::TensorOptions().dtype(torch::kInt32).device(vks.device()));
    if (N > 0) {
        launch_k1_bls_verify_wave(vks.data_ptr<uint8_t>(), sigs.data_ptr<uint8_t>(),
                                  msgs.data_ptr<uint8_t>(), moff.data_ptr<int64_t>(),
                                  g2_lines.data_ptr<uint8_t>(),
                                  (const uint64_t*)rand_r.data_ptr<int64_t>(), N,
                                  ok.data_ptr<int32_t>(), cur_stream());
    }
    return ok;
}

torch::Tensor hash_to_g1_batch(torch::Tensor msgs, torch::Tensor moff) {
    CHECK_DEV(msgs); CHECK_DEV(moff);
    int32_t N = (int32_t)moff.size(0) - 1;
    auto out = torch::zeros({N, 64},
                            torch::TensorOptions().dtype(torch::kUInt8).device(msgs.device()));
    if (N > 0) {
        launch_k1_hash_to_g1(msgs.data_ptr<uint8_t>(), moff.data_ptr<int64_t>(), N,
                             out.data_ptr<uint8_t>(), cur_stream());
    }
    return out;
}

void fanout_wave(torch::Tensor buf, torch::Tensor payload_off, torch::Tensor payload_len,
                 torch::Tensor pairs,
                 torch::Tensor msg_seq, torch::Tensor n_pairs, torch::Tensor egress,
                 int64_t nt, int64_t grid) {
    CHECK_DEV(egress); CHECK_CONTIG(egress);
    launch_k3_fanout_wave(buf.data_ptr<uint8_t>(), payload_off.data_ptr<int64_t>(),
                          payload_len.data_ptr<int32_t>(), pair_ptr(pairs),
                          (const uint32_t*)msg_seq.data_ptr<int32_t>(),
                          n_pairs.data_ptr<int32_t>(), (int32_t)pairs.size(0),
                          egress.data_ptr<uint8_t>(), (int)nt,
                          (int)grid, cur_stream());
}

void fanout_flat2(torch::Tensor buf, torch::Tensor payload_off, torch::Tensor payload_len,
                  torch::Tensor pairs,
                  int64_t seq_base, torch::Tensor n_pairs, int64_t units_per_pair,
                  torch::Tensor egress, int64_t nt, int64_t grid, int64_t uniform_len) {
    CHECK_DEV(egress); CHECK_CONTIG(egress);
    int32_t capacity = (int32_t)pairs.size(0);
    launch_k3_fanout_flat2(buf.data_ptr<uint8_t>(), payload_off.data_ptr<int64_t>(),
                           payload_len.data_ptr<int32_t>(), pair_ptr(pairs),
                           (uint32_t)seq_base, n_pairs.data_ptr<int32_t>(), capacity,
                           (int32_t)units_per_pair, (int32_t)uniform_len,
                           egress.data_ptr<uint8_t>(), (int)nt, (int)grid, cur_stream());
}

void fanout_flat3(torch::Tensor buf, torch::Tensor payload_off, torch::Tensor payload_len,
                  torch::Tensor pairs,
                  torch::Tensor seq_state, torch::Tensor n_pairs, int64_t units_per_pair,
                  torch::Tensor egress, int64_t nt, int64_t grid, int64_t uniform_len) {
    CHECK_DEV(egress); CHECK_CONTIG(egress);
    int32_t capacity = (int32_t)pairs.size(0);
    launch_k3_fanout_flat3(buf.data_ptr<uint8_t>(), payload_off.data_ptr<int64_t>(),
                           payload_len.data_ptr<int32_t>(), pair_ptr(pairs),
                           (const uint32_t*)seq_state.data_ptr<int32_t>(),
                           n_pairs.data_ptr<int32_t>(), capacity, (int32_t)units_per_pair,
                           (int32_t)uniform_len, egress.data_ptr<uint8_t>(), (int)nt,
                           (int)grid, cur_stream());
}

void seq_advance(torch::Tensor seq_state, int64_t m) {
    launch_k_seq_advance((uint32_t*)seq_state.data_ptr<int32_t>(), (int32_t)m, cur_stream());
}

torch::Tensor topic_mask_t(torch::Tensor sub_bitmap, torch::Tensor buf,
                           torch::Tensor topics_off, torch::Tensor topics_cnt,
                           torch::Tensor disc) {
    CHECK_DEV(sub_bitmap); CHECK_CONTIG(sub_bitmap);
    int32_t W = (int32_t)sub_bitmap.size(1);
    int32_t M = (int32_t)disc.size(0);
    auto mask_t = torch::empty({(int64_t)W, M},
                               torch::TensorOptions().dtype(torch::kInt64).device(buf.device()));
    if (M > 0) {
        launch_k2a_topic_mask_t((const uint64_t*)sub_bitmap.data_ptr<int64_t>(),
                                buf.data_ptr<uint8_t>(), topics_off.data_ptr<int64_t>(),
                                topics_cnt.data_ptr<int32_t>(), disc.data_ptr<int32_t>(),
                                (uint64_t*)mask_t.data_ptr<int64_t>(), M, W, cur_stream());
    }
    return mask_t;
}

void assign_emit_fused_t(torch::Tensor mask_t, torch::Tensor payload_len,
                         torch::Tensor ring_wpos, int64_t ring_bytes, int64_t n_users,
                         torch::Tensor pairs, torch::Tensor drops, torch::Tensor n_pairs,
                         int64_t uniform_rec) {
    CHECK_DEV(mask_t); CHECK_CONTIG(mask_t);
    int32_t W = (int32_t)mask_t.size(0);
    int32_t M = (int32_t)mask_t.size(1);
    TORCH_CHECK(ring_bytes % 16 == 0);
    int32_t capacity = (int32_t)pairs.size(0);
    launch_k2b_fused_t((const uint64_t*)mask_t.data_ptr<int64_t>(),
                       payload_len.data_ptr<int32_t>(), M, W, (int32_t)n_users, ring_bytes,
                       capacity, (int32_t)uniform_rec,
                       (uint64_t*)ring_wpos.data_ptr<int64_t>(),
                       n_pairs.data_ptr<int32_t>(), pair_ptr(pairs),
                       (uint32_t*)drops.data_ptr<int32_t>(), cur_stream());
}

void assign_emit_blocks_t(torch::Tensor mask_t, torch::Tensor ring_wpos,
                          int64_t ring_bytes, int64_t n_users,
                          torch::Tensor bcount, torch::Tensor pprefix, torch::Tensor ubase,
                          torch::Tensor ufit, torch::Tensor udst,
                          torch::Tensor pairs, torch::Tensor drops, torch::Tensor n_pairs,
                          int64_t uniform_rec) {
    CHECK_DEV(mask_t); CHECK_CONTIG(mask_t);
    int32_t W = (int32_t)mask_t.size(0);
    int32_t M = (int32_t)mask_t.size(1);
    TORCH_CHECK(ring_bytes % 16 == 0 && uniform_rec > 0);
    int64_t NB = (M + 31) / 32;
    TORCH_CHECK(bcount.numel() >= NB * W * 64 && pprefix.numel() >= NB * W * 64,
                "k2b block scratch too small");
    TORCH_CHECK(ubase.numel() >= W * 64 && ufit.numel() >= W * 64 && udst.numel() >= W * 64);
    int32_t capacity = (int32_t)pairs.size(0);
    launch_k2b_blocks_t((const uint64_t*)mask_t.data_ptr<int64_t>(), M, W, (int32_t)n_users,
                        ring_bytes, capacity, (int32_t)uniform_rec,
                        (uint64_t*)ring_wpos.data_ptr<int64_t>(),
                        n_pairs.data_ptr<int32_t>(),
                        bcount.data_ptr<int32_t>(), pprefix.data_ptr<int32_t>(),
                        ubase.data_ptr<int32_t>(), ufit.data_ptr<int32_t>(),
                        udst.data_ptr<int64_t>(),
                        pair_ptr(pairs),
                        (uint32_t*)drops.data_ptr<int32_t>(), cur_stream());
}

void emit_direct(torch::Tensor disc, torch::Tensor owner, torch::Tensor payload_off,
                 torch::Tensor payload_len, int64_t ring_bytes, torch::Tensor ring_wpos,
                 torch::Tensor n_pairs, torch::Tensor pairs, torch::Tensor drops) {
    int32_t M = (int32_t)disc.size(0);
    if (M == 0) return;
    int32_t capacity = (int32_t)pairs.size(0);
    launch_k5b_emit_direct(disc.data_ptr<int32_t>(), owner.data_ptr<int32_t>(),
                           payload_off.data_ptr<int64_t>(), payload_len.data_ptr<int32_t>(),
                           M, ring_bytes, capacity, (uint64_t*)ring_wpos.data_ptr<int64_t>(),
                           n_pairs.data_ptr<int32_t>(), pair_ptr(pairs),
                           (uint32_t*)drops.data_ptr<int32_t>(), cur_stream());
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
    m.def("parse_batch", &parse_batch, "K4: on-device capnp parse of a message batch",
          py::arg("buf"), py::arg("offsets"), py::arg("hash_seed") = 0);
    m.def("topic_mask", &topic_mask, "K2a: per-message subscriber mask");
    m.def("assign_emit", &assign_emit, "K2b: per-user FIFO ring assignment + pair list");
    m.def("fanout", &fanout, "K3: N-way payload fan-out into egress rings");
    m.def("direct_lookup", &direct_lookup, "K5: batched direct-route hash probe");
    m.def("apply_subs", &apply_subs, "K2c: apply subscribe/unsubscribe batch to bitmap");
    m.def("bls_verify_batch", &bls_verify_batch, "K1: batched BLS-over-BN254 verification");
    m.def("compact_rings", &compact_rings,
          "K7: gather used egress-ring prefixes into one staging buffer");
    m.def("precompute_g2_lines", &precompute_g2_lines,
          "fixed-g2 Miller line coefficients for K1 v2 (once per process)");
    m.def("bls_verify_batch2", &bls_verify_batch2,
          "K1 v2: 2-lane Fp2-decomposed batched BLS verification");
    m.def("_k1_dbg_wave",
          [](torch::Tensor vks, torch::Tensor sigs, torch::Tensor msgs, torch::Tensor moff,
             torch::Tensor g2_lines, torch::Tensor rand_r, int64_t mode) {
              int32_t N = (int32_t)moff.size(0) - 1;
              auto ok = torch::zeros({N}, torch::TensorOptions().dtype(torch::kInt32)
                                              .device(vks.device()));
              launch_k1_dbg_wave(vks.data_ptr<uint8_t>(), sigs.data_ptr<uint8_t>(),
                                 msgs.data_ptr<uint8_t>(), moff.data_ptr<int64_t>(),
                                 g2_lines.data_ptr<uint8_t>(),
                                 (const uint64_t*)rand_r.data_ptr<int64_t>(), N,
                                 (int32_t)mode, ok.data_ptr<int32_t>(), cur_stream());
              return ok;
          });
    m.def("bls_verify_batch_wave", &bls_verify_batch_wave,
          "K1 v3: wave-batched product verification (shared final exp, "
          "exact per-item fallback)");
    m.def("hash_to_g1_batch", &hash_to_g1_batch, "K1 helper: batched hash-to-G1");
    m.def("fanout_wave", &fanout_wave, "K3v2: wave-per-pair fan-out (nt flag, device count)");
    m.def("fanout_flat2", &fanout_flat2, "K3v4: flat fan-out, seq from base, capacity clamp");
    m.def("fanout_flat3", &fanout_flat3, "K3v5: graph-capturable (device seq counter)");
    m.def("seq_advance", &seq_advance, "bump the device seq counter (inside the graph)");
    m.def("topic_mask_t", &topic_mask_t, "K2a transposed: mask[W][M]");
    m.def("emit_direct", &emit_direct,
          "K5b: on-device direct-delivery pair emission (no host sync)");
    m.def("assign_emit_fused_t", &assign_emit_fused_t,
          "K2b fused on transposed mask (wave-aggregated claims, uniform-rec fast path)");
    m.def("assign_emit_blocks_t", &assign_emit_blocks_t,
          "K2b block-parallel (P1 count / P2 bases / P3 emit) for uniform records");
}
