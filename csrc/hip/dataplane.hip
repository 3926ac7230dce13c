// MI355X (gfx950) data-plane kernels for the GPU broker.
//
// These implement the per-message hot path of the reference broker
// (cdn-broker/src/tasks/user/handler.rs:95-163 + broker/handler.rs:197-272)
// as batched CDNA4 kernels over HBM-resident tables:
//
//   K4  k4_parse_batch — on-device Cap'n Proto validation/field extraction
//                        (reference cdn-proto/src/message.rs:212-312)
//   K2a k2a_topic_mask[_t] — per-message OR of subscription-bitmap topic
//                        rows (reference connections/mod.rs:94-124);
//                        the _t variant writes the mask TRANSPOSED [W][M]
//                        so K2b scans contiguous memory
//   K2b k2b_p1_count / k2b_p2_bases / k2b_p3_emit — the production emit
//                        for uniform records: per-(user, 32-msg block)
//                        counts, then per-user closed-form ring math with
//                        wave-aggregated span claims (one atomic per
//                        wave64), then block-parallel emission — fills the
//                        chip at any population (one-lane-per-user runs
//                        only W waves).  k2b_fused_t is the one-kernel
//                        variant (non-uniform records + golden tests);
//                        k2b_count/k2b_emit the two-pass reference pair.
//                        All preserve the per-connection FIFO the
//                        reference gets from its per-conn channel actors
//                        (protocols/mod.rs:139-217).
//   K3  k3_fanout_flat_t — the production fan-out for uniform records
//                        (unit-per-lane flat index, ~100% lane utilization,
//                        non-temporal 16 B stores; seq from value or device
//                        counter for hipGraph capture); k3_fanout_wave_t is
//                        the general mixed-size path (one wave per pair,
//                        best for records >= 4 KiB); k3_fanout is the plain
//                        reference variant for golden tests. All mirror the
//                        reference's raw-bytes Arc-clone push
//                        (user/sender.rs:16-33).
//   K5  k5_direct_lookup — open-addressing probe user-hash -> owner
//                        (reference connections/mod.rs:69-71 DirectMap get)
//   K5b k5b_emit_direct — on-device direct-delivery pair emission (appends
//                        to the same pair list; no host sync)
//   K2c k2c_apply_subs — ordered subscription updates
//
// Design notes (per /opt/skills/guides/cdna_hip_programming.md):
//  - wave = 64; all copies are uint4 (16 B/lane) vectorized
//  - K3 uses a grid-stride loop over delivery pairs with >> 256 workgroups so
//    all 8 XCDs fill; the payload source is usually L2/LLC-hit after the
//    first recipient of a message on that XCD
//  - no cross-workgroup hand-off inside a launch: every kernel's outputs are
//    consumed by the *next* launch on the stream (boundary ≈1.5 us), so no
//    agent-scope fencing is needed
//  - tables are torch tensors: PyTorch owns the HBM, kernels are raw HIP

#include <hip/hip_runtime.h>
#include <stdint.h>

#define WAVE 64
#ifndef CDN_CHECK
#define CDN_CHECK 1
#endif

// Egress ring record stride: 16 B header + payload padded to 16 B, so every
// record start stays uint4-aligned.  Pairs are emitted grouped by user with
// messages in order, so a user's records are back-to-back in the ring and
// the flat K3 writes them as one contiguous streaming range (full-stride,
// zero-padded tail) — 64 B alignment was MEASURED SLOWER (261k vs 285k
// msgs/s headline: +8.6% stored bytes buys no RMW savings).  Host mirrors:
// pushcdn_amd/broker/gpu_engine.py ring_rec / ops/reference.py.
#define RING_ALIGN 16ull
__host__ __device__ inline uint64_t ring_rec(int32_t len) {
    return ((uint64_t)len + 16 + (RING_ALIGN - 1)) & ~(RING_ALIGN - 1);
}

// A delivery pair as ONE 16-byte record {user, msg, ring dst}: emitters do a
// single dword4 store per pair and K3 a single dword4 load per unit.  The
// SoA layout (three arrays) cost ~3x scattered sub-line stores per pair —
// measured 100 us/tick in k2b_p3_emit at 2.56M pairs before this change.
struct alignas(16) PairRec { int32_t user; int32_t msg; int64_t dst; };
static_assert(sizeof(PairRec) == 16, "PairRec must be one dword4");
typedef unsigned int cdn_v4u __attribute__((ext_vector_type(4)));
__device__ inline void store_pair(PairRec* p, int32_t u, int32_t m, int64_t d) {
    PairRec r{u, m, d};
    cdn_v4u v; memcpy(&v, &r, 16);
    *(cdn_v4u*)p = v;
}

// ---------------------------------------------------------------------------
// 64-bit FNV-1a — the routing hash for user public keys. Host mirror in
// pushcdn_amd/utils/keyhash.py must match bit-for-bit.
// `seed` XORs into the offset basis: brokers derive it from the cluster
// private key, so an attacker cannot grind out hash collisions offline
// (FNV alone is byte-invertible). seed=0 == classic FNV-1a.
// ---------------------------------------------------------------------------
__host__ __device__ inline uint64_t fnv1a64(const uint8_t* data, uint32_t len,
                                            uint64_t seed = 0) {
    uint64_t h = 0xcbf29ce484222325ull ^ seed;
    for (uint32_t i = 0; i < len; ++i) {
        h ^= (uint64_t)data[i];
        h *= 0x100000001b3ull;
    }
    return h;
}

// ---------------------------------------------------------------------------
// K4: parse a batch of concatenated serialized Messages (capnp stream format,
// single segment each). One thread per message: the pointer graph is 3-5
// words deep, so a thread walks it; payload bytes are never touched here.
//
// Outputs (all int32/int64 tensors, -1/0 on invalid):
//   disc[i]         discriminant 0..8, or -1 if malformed
//   payload_off/len payload ("message" field / sync Data / topic list) byte
//                   range within the batch buffer
//   topics_off/cnt  topic list byte range (Broadcast/Subscribe/Unsubscribe)
//   recip_hash      fnv1a64 of the recipient key (Direct), else 0
//   timestamp       AuthenticateWithKey.timestamp, else 0
// ---------------------------------------------------------------------------

struct ParseOut {
    int32_t* disc;
    int64_t* payload_off;
    int32_t* payload_len;
    int64_t* topics_off;
    int32_t* topics_cnt;
    uint64_t* recip_hash;
    uint64_t* timestamp;
};

__device__ inline uint64_t ld_u64(const uint8_t* p) {
    uint64_t v;
    memcpy(&v, p, 8);
    return v;
}

// Decode a struct pointer at word `pw` (relative to segment base `seg` of
// `nwords`). Returns false if malformed. Out: target word, data words, ptr words.
__device__ inline bool read_struct_ptr(const uint8_t* seg, int64_t nwords, int64_t pw,
                                       int64_t* tgt, int32_t* dw, int32_t* ptrw) {
    if (pw < 0 || pw >= nwords) return false;
    uint64_t v = ld_u64(seg + pw * 8);
    if (v == 0 || (v & 3) != 0) return false;
    int64_t b = (v >> 2) & 0x3fffffff;
    if (b & 0x20000000) b -= 0x40000000;
    int32_t d = (v >> 32) & 0xffff;
    int32_t p = (v >> 48) & 0xffff;
    int64_t t = pw + 1 + b;
    if (t < 0 || t + d + p > nwords) return false;
    *tgt = t; *dw = d; *ptrw = p;
    return true;
}

// Decode a byte-list pointer (Data / Text / List(UInt8), element code 2).
__device__ inline bool read_byte_list(const uint8_t* seg, int64_t nwords, int64_t pw,
                                      int64_t* off, int32_t* len) {
    if (pw < 0 || pw >= nwords) return false;
    uint64_t v = ld_u64(seg + pw * 8);
    if (v == 0) { *off = pw; *len = 0; return true; }  // null ptr -> empty
    if ((v & 3) != 1) return false;
    int64_t b = (v >> 2) & 0x3fffffff;
    if (b & 0x20000000) b -= 0x40000000;
    uint32_t code = (v >> 32) & 7;
    int64_t count = (v >> 35) & 0x1fffffff;
    if (code != 2) return false;
    int64_t t = pw + 1 + b;
    if (t < 0 || t * 8 + count > nwords * 8) return false;
    *off = t * 8;
    *len = (int32_t)count;
    return true;
}

extern "C" __global__ void k4_parse_batch(
    const uint8_t* __restrict__ buf,
    const int64_t* __restrict__ offsets,  // [M+1] byte offsets into buf
    int32_t M, uint64_t hash_seed, ParseOut out)
{
    int i = blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= M) return;
    // defaults; field values are buffered in locals and committed only
    // once the whole frame validates, so rejected records are all-zero
    // (bit-identical to the host mirror, proto/message.py parse_offsets)
    out.disc[i] = -1;
    out.payload_off[i] = 0; out.payload_len[i] = 0;
    out.topics_off[i] = 0;  out.topics_cnt[i] = 0;
    out.recip_hash[i] = 0;  out.timestamp[i] = 0;
    int64_t o_poff = 0, o_toff = 0, o_ts = 0;
    uint64_t o_rhash = 0;
    int32_t o_plen = 0, o_tcnt = 0;

    int64_t beg = offsets[i], end = offsets[i + 1];
    if (end - beg < 8 + 8) return;  // header + root ptr
    const uint8_t* p = buf + beg;
    uint32_t seg_m1, nw;
    memcpy(&seg_m1, p, 4); memcpy(&nw, p + 4, 4);
    if (seg_m1 != 0) return;
    if (8 + (int64_t)nw * 8 > end - beg) return;
    const uint8_t* seg = p + 8;
    int64_t nwords = nw;

    int64_t mt; int32_t mdw, mpw;
    if (!read_struct_ptr(seg, nwords, 0, &mt, &mdw, &mpw)) return;
    if (mdw < 1 || mpw < 1) return;
    uint16_t disc;
    memcpy(&disc, seg + mt * 8, 2);
    int64_t up = mt + mdw;  // union pointer word

    int64_t seg_base = (seg - buf);  // byte offset of segment within buf

    switch (disc) {
    case 0: {  // AuthenticateWithKey
        int64_t it; int32_t idw, ipw;
        if (!read_struct_ptr(seg, nwords, up, &it, &idw, &ipw)) return;
        if (idw < 1 || ipw < 2) return;
        o_ts = (int64_t)ld_u64(seg + it * 8);
        int64_t ko; int32_t kl;
        if (!read_byte_list(seg, nwords, it + idw, &ko, &kl)) return;
        o_poff = kl ? seg_base + ko : 0;  // public key bytes
        o_plen = kl;
        int64_t so; int32_t sl;
        if (!read_byte_list(seg, nwords, it + idw + 1, &so, &sl)) return;
        o_toff = sl ? seg_base + so : 0;   // signature bytes (reused slot)
        o_tcnt = sl;
        break;
    }
    case 1: {  // AuthenticateWithPermit
        int64_t it; int32_t idw, ipw;
        if (!read_struct_ptr(seg, nwords, up, &it, &idw, &ipw)) return;
        if (idw < 1) return;
        o_ts = (int64_t)ld_u64(seg + it * 8);  // permit
        break;
    }
    case 2: {  // AuthenticateResponse
        int64_t it; int32_t idw, ipw;
        if (!read_struct_ptr(seg, nwords, up, &it, &idw, &ipw)) return;
        if (idw < 1 || ipw < 1) return;
        o_ts = (int64_t)ld_u64(seg + it * 8);  // permit
        int64_t co; int32_t cl;
        if (!read_byte_list(seg, nwords, it + idw, &co, &cl)) return;
        o_poff = cl ? seg_base + co : 0; o_plen = cl;
        break;
    }
    case 3: {  // Direct
        int64_t it; int32_t idw, ipw;
        if (!read_struct_ptr(seg, nwords, up, &it, &idw, &ipw)) return;
        if (ipw < 2) return;
        int64_t ro; int32_t rl;
        if (!read_byte_list(seg, nwords, it + idw, &ro, &rl)) return;
        o_rhash = fnv1a64(seg + ro, rl, hash_seed);
        int64_t mo; int32_t ml;
        if (!read_byte_list(seg, nwords, it + idw + 1, &mo, &ml)) return;
        o_poff = ml ? seg_base + mo : 0; o_plen = ml;
        break;
    }
    case 4: {  // Broadcast
        int64_t it; int32_t idw, ipw;
        if (!read_struct_ptr(seg, nwords, up, &it, &idw, &ipw)) return;
        if (ipw < 2) return;
        int64_t to; int32_t tc;
        if (!read_byte_list(seg, nwords, it + idw, &to, &tc)) return;
        o_toff = tc ? seg_base + to : 0; o_tcnt = tc;
        int64_t mo; int32_t ml;
        if (!read_byte_list(seg, nwords, it + idw + 1, &mo, &ml)) return;
        o_poff = ml ? seg_base + mo : 0; o_plen = ml;
        break;
    }
    case 5: case 6: {  // Subscribe / Unsubscribe
        int64_t to; int32_t tc;
        if (!read_byte_list(seg, nwords, up, &to, &tc)) return;
        o_toff = tc ? seg_base + to : 0; o_tcnt = tc;
        break;
    }
    case 7: case 8: {  // UserSync / TopicSync
        int64_t dof; int32_t dl;
        if (!read_byte_list(seg, nwords, up, &dof, &dl)) return;
        o_poff = dl ? seg_base + dof : 0; o_plen = dl;
        break;
    }
    default:
        return;
    }
    out.payload_off[i] = o_poff; out.payload_len[i] = o_plen;
    out.topics_off[i] = o_toff;  out.topics_cnt[i] = o_tcnt;
    out.recip_hash[i] = (int64_t)o_rhash; out.timestamp[i] = o_ts;
    out.disc[i] = (int32_t)disc;
}

// ---------------------------------------------------------------------------
// K2a: per-message recipient mask.
//   sub_bitmap: [256][W] uint64 — bit u of word w set iff user (w*64+u)
//               subscribes to the topic
//   mask:       [M][W] uint64 out — OR of the message's topic rows
// Grid: one thread per (message, word): M*W threads. Broadcast-only messages
// (disc==4) produce a mask; everything else produces zeros.
// exclude_user: for no-echo semantics the caller can exclude the sender
// (reference tests: no echo to the originating connection is NOT default —
// the reference DOES echo to the sender if subscribed; keep -1 to disable).
// ---------------------------------------------------------------------------
extern "C" __global__ void k2a_topic_mask(
    const uint64_t* __restrict__ sub_bitmap,  // [256][W]
    const uint8_t* __restrict__ buf,
    const int64_t* __restrict__ topics_off,
    const int32_t* __restrict__ topics_cnt,
    const int32_t* __restrict__ disc,
    uint64_t* __restrict__ mask,              // [M][W]
    int32_t M, int32_t W)
{
    int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (idx >= (int64_t)M * W) return;
    int m = idx / W;
    int w = idx % W;
    uint64_t acc = 0;
    if (disc[m] == 4) {
        const uint8_t* topics = buf + topics_off[m];
        int n = topics_cnt[m];
        for (int t = 0; t < n; ++t) {
            acc |= sub_bitmap[(int64_t)topics[t] * W + w];
        }
    }
    mask[idx] = acc;
}

// ---------------------------------------------------------------------------
// K2b: per-user ordered assignment + pair emission.
// One thread per user walks the M messages IN ORDER (per-(sender,recipient)
// FIFO), assigns consecutive ring offsets in its egress ring, and appends
// delivery pairs. Pair slots are claimed per-user via an exclusive scan over
// delivery counts computed in pass 0 (count_only=1), so the pair list is
// grouped by user and ordered by message within each user.
//
// egress ring layout: ring_bytes per user within a single [N_users][ring_bytes]
// tensor. Each delivery writes an 8-byte record header {u32 len, u32 msg_seq}
// followed by the payload, 8-byte aligned (the host-side drain parses this).
// If the ring fills, the remaining deliveries for that user are dropped and
// counted (best-effort semantics = reference eviction-on-full, sender.rs).
// ---------------------------------------------------------------------------
extern "C" __global__ void k2b_count(
    const uint64_t* __restrict__ mask,   // [M][W]
    int32_t M, int32_t W, int32_t n_users,
    int32_t* __restrict__ counts)        // [n_users]
{
    int u = blockIdx.x * blockDim.x + threadIdx.x;
    if (u >= n_users) return;
    int w = u >> 6;
    uint64_t bit = 1ull << (u & 63);
    int c = 0;
    for (int m = 0; m < M; ++m) c += (mask[(int64_t)m * W + w] & bit) ? 1 : 0;
    counts[u] = c;
}

extern "C" __global__ void k2b_emit(
    const uint64_t* __restrict__ mask,        // [M][W]
    const int64_t* __restrict__ payload_off,  // [M]
    const int32_t* __restrict__ payload_len,  // [M]
    const int32_t* __restrict__ pair_base,    // [n_users] exclusive scan of counts
    int32_t M, int32_t W, int32_t n_users,
    int64_t ring_bytes,
    uint64_t* __restrict__ ring_wpos,         // [n_users] persistent write cursor
    PairRec* __restrict__ pairs,              // [total] {user,msg,dst}
    uint32_t* __restrict__ drops)             // [1] dropped deliveries (ring full)
{
    int u = blockIdx.x * blockDim.x + threadIdx.x;
    if (u >= n_users) return;
    int w = u >> 6;
    uint64_t bit = 1ull << (u & 63);
    int slot = pair_base[u];
    uint64_t wpos = ring_wpos[u];
    uint32_t dropped = 0;
    for (int m = 0; m < M; ++m) {
        if (!(mask[(int64_t)m * W + w] & bit)) continue;
        int32_t len = payload_len[m];
        // 64-aligned record stride (see ring_rec)
        uint64_t rec = ring_rec(len);
        if (wpos + rec > (uint64_t)ring_bytes) {
            store_pair(pairs + slot, -1, m, 0);
            slot++; dropped++; continue;
        }
        store_pair(pairs + slot, u, m, (int64_t)u * ring_bytes + wpos);
        slot++;
        wpos += rec;
    }
    ring_wpos[u] = wpos;
    if (dropped) atomicAdd(drops, dropped);
}

// ---------------------------------------------------------------------------
// K3: fan-out payload copy. Grid-stride over delivery pairs; each workgroup
// copies one pair per iteration with 16 B/lane vector loads/stores. Writes
// the 8-byte record header, then the payload. Source bytes for a hot message
// are L2/LLC-resident after the first few recipients.
// ---------------------------------------------------------------------------
extern "C" __global__ void k3_fanout(
    const uint8_t* __restrict__ buf,
    const int64_t* __restrict__ payload_off,
    const int32_t* __restrict__ payload_len,
    const PairRec* __restrict__ pairs,
    const uint32_t* __restrict__ msg_seq,   // [M] global sequence numbers
    int32_t n_pairs,
    uint8_t* __restrict__ egress)
{
    for (int p = blockIdx.x; p < n_pairs; p += gridDim.x) {
        const PairRec pr = pairs[p];
        if (pr.user < 0) continue;  // dropped (ring full)
        int m = pr.msg;
        int32_t len = payload_len[m];
        const uint8_t* src = buf + payload_off[m];
        uint8_t* dst = egress + pr.dst;
        if (threadIdx.x == 0) {
            uint32_t hdr[4] = {(uint32_t)len, msg_seq[m], 0, 0};
            memcpy(dst, hdr, 16);
        }
        dst += 16;  // 16-aligned: ring base and every record start are 16-aligned
        int32_t nvec = len >> 4;          // full 16-byte chunks
        const bool src16 = (((uintptr_t)src) & 15) == 0;
        if (src16) {
            const uint4* s4 = (const uint4*)src;
            uint4* d4 = (uint4*)dst;
            for (int k = threadIdx.x; k < nvec; k += blockDim.x) d4[k] = s4[k];
        } else {
            for (int k = threadIdx.x; k < nvec; k += blockDim.x) {
                uint8_t tmp[16];
                memcpy(tmp, src + (size_t)k * 16, 16);
                memcpy(dst + (size_t)k * 16, tmp, 16);
            }
        }
        for (int k = (nvec << 4) + threadIdx.x; k < len; k += blockDim.x)
            dst[k] = src[k];
    }
}

// ---------------------------------------------------------------------------
// K5: batched direct-route lookup. Open-addressing (linear probe) hash table
// over HBM: keys[S] = fnv1a64(pubkey) (0 = empty), vals[S] = owner id
// (>=0: local user index; <0: -(broker_rank+2); reference DirectMap).
// ---------------------------------------------------------------------------
extern "C" __global__ void k5_direct_lookup(
    const uint64_t* __restrict__ table_keys,
    const int32_t* __restrict__ table_vals,
    int64_t table_size,                      // power of two
    const uint64_t* __restrict__ query,      // [N]
    int32_t N,
    int32_t* __restrict__ owner)             // [N] out; INT32_MIN = not found
{
    int i = blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= N) return;
    uint64_t h = query[i];
    if (h == 0) { owner[i] = INT32_MIN; return; }
    uint64_t mask = (uint64_t)table_size - 1;
    uint64_t s = h & mask;
    for (int64_t probe = 0; probe < table_size; ++probe) {
        uint64_t k = table_keys[(s + probe) & mask];
        if (k == h) { owner[i] = table_vals[(s + probe) & mask]; return; }
        if (k == 0) { owner[i] = INT32_MIN; return; }
    }
    owner[i] = INT32_MIN;
}

// ---------------------------------------------------------------------------
// Subscription updates (Subscribe/Unsubscribe batches) applied on-device.
// One thread per (update). Bitmap bit flips must be atomic (two users in one
// word updated concurrently).
// ---------------------------------------------------------------------------
extern "C" __global__ void k2c_apply_subs(
    uint64_t* __restrict__ sub_bitmap,       // [256][W]
    const uint8_t* __restrict__ buf,
    const int64_t* __restrict__ topics_off,
    const int32_t* __restrict__ topics_cnt,
    const int32_t* __restrict__ disc,        // 5=subscribe 6=unsubscribe
    const int32_t* __restrict__ user_idx,    // [M] local user index per message
    int32_t M, int32_t W)
{
    // One thread applies the whole batch IN MESSAGE ORDER: a subscribe and
    // a later unsubscribe of the same (user, topic) in one batch must land
    // in order (subscription updates are control-plane-rate, so the serial
    // walk is microseconds; the data-plane kernels stay fully parallel).
    if (blockIdx.x != 0 || threadIdx.x != 0) return;
    for (int m = 0; m < M; ++m) {
        int d = disc[m];
        if (d != 5 && d != 6) continue;
        int u = user_idx[m];
        if (u < 0) continue;
        int w = u >> 6;
        uint64_t bit = 1ull << (u & 63);
        const uint8_t* topics = buf + topics_off[m];
        int n = topics_cnt[m];
        for (int t = 0; t < n; ++t) {
            uint64_t* word = &sub_bitmap[(int64_t)topics[t] * W + w];
            if (d == 5) *word |= bit;
            else        *word &= ~bit;
        }
    }
}

// ---------------------------------------------------------------------------
// Host-side launchers (called from the torch-extension bindings, which are
// compiled as plain host C++ and cannot reference __global__ symbols).
// ---------------------------------------------------------------------------
extern "C" {

void launch_k4_parse(const uint8_t* buf, const int64_t* offsets, int32_t M, uint64_t hash_seed,
                     ParseOut out, hipStream_t s) {
    if (M <= 0) return;
    int threads = 256, blocks = (M + threads - 1) / threads;
    hipLaunchKernelGGL(k4_parse_batch, dim3(blocks), dim3(threads), 0, s, buf, offsets, M,
                       hash_seed, out);
}

void launch_k2a_topic_mask(const uint64_t* sub_bitmap, const uint8_t* buf,
                           const int64_t* topics_off, const int32_t* topics_cnt,
                           const int32_t* disc, uint64_t* mask, int32_t M, int32_t W,
                           hipStream_t s) {
    if (M <= 0) return;
    int64_t total = (int64_t)M * W;
    int threads = 256;
    int64_t blocks = (total + threads - 1) / threads;
    hipLaunchKernelGGL(k2a_topic_mask, dim3((uint32_t)blocks), dim3(threads), 0, s, sub_bitmap,
                       buf, topics_off, topics_cnt, disc, mask, M, W);
}

void launch_k2b_count(const uint64_t* mask, int32_t M, int32_t W, int32_t n_users,
                      int32_t* counts, hipStream_t s) {
    int threads = 256, blocks = (n_users + threads - 1) / threads;
    hipLaunchKernelGGL(k2b_count, dim3(blocks), dim3(threads), 0, s, mask, M, W, n_users, counts);
}

void launch_k2b_emit(const uint64_t* mask, const int64_t* payload_off,
                     const int32_t* payload_len, const int32_t* pair_base, int32_t M, int32_t W,
                     int32_t n_users, int64_t ring_bytes, uint64_t* ring_wpos, PairRec* pairs,
                     uint32_t* drops, hipStream_t s) {
    int threads = 256, blocks = (n_users + threads - 1) / threads;
    hipLaunchKernelGGL(k2b_emit, dim3(blocks), dim3(threads), 0, s, mask, payload_off,
                       payload_len, pair_base, M, W, n_users, ring_bytes, ring_wpos, pairs,
                       drops);
}

void launch_k3_fanout(const uint8_t* buf, const int64_t* payload_off,
                      const int32_t* payload_len, const PairRec* pairs, const uint32_t* msg_seq,
                      int32_t n_pairs, uint8_t* egress, hipStream_t s) {
    int blocks = n_pairs < 16384 ? n_pairs : 16384;
    if (blocks == 0) return;
    hipLaunchKernelGGL(k3_fanout, dim3(blocks), dim3(128), 0, s, buf, payload_off, payload_len,
                       pairs, msg_seq, n_pairs, egress);
}

void launch_k5_direct_lookup(const uint64_t* table_keys, const int32_t* table_vals,
                             int64_t table_size, const uint64_t* query, int32_t N,
                             int32_t* owner, hipStream_t s) {
    int threads = 256, blocks = (N + threads - 1) / threads;
    hipLaunchKernelGGL(k5_direct_lookup, dim3(blocks), dim3(threads), 0, s, table_keys,
                       table_vals, table_size, query, N, owner);
}

void launch_k2c_apply_subs(uint64_t* sub_bitmap, const uint8_t* buf, const int64_t* topics_off,
                           const int32_t* topics_cnt, const int32_t* disc,
                           const int32_t* user_idx, int32_t M, int32_t W, hipStream_t s) {
    int threads = 256, blocks = (M + threads - 1) / threads;
    hipLaunchKernelGGL(k2c_apply_subs, dim3(blocks), dim3(threads), 0, s, sub_bitmap, buf,
                       topics_off, topics_cnt, disc, user_idx, M, W);
}

}  // extern "C"

// ---------------------------------------------------------------------------
// K3v2: wave-per-pair fan-out. 256-thread workgroups = 4 waves; each wave
// owns one delivery pair per grid-stride iteration (64 lanes x 16 B = 1 KiB
// per pass — matches the 1 KiB-payload sweet spot; a 64 KiB payload takes 64
// passes). n_pairs is read from a device pointer so the host never syncs on
// the pair count. NT=1 uses non-temporal stores for the egress payload
// (written once, consumed by the D2H drain) to keep the per-XCD L2 for the
// hot message source bytes instead of the streaming egress.
// ---------------------------------------------------------------------------
template <int NT>
__global__ void __launch_bounds__(256) k3_fanout_wave_t(
    const uint8_t* __restrict__ buf,
    const int64_t* __restrict__ payload_off,
    const int32_t* __restrict__ payload_len,
    const PairRec* __restrict__ pairs,
    const uint32_t* __restrict__ msg_seq,
    const int32_t* __restrict__ n_pairs_ptr,
    int32_t capacity,
    uint8_t* __restrict__ egress)
{
    // clamp: the claim counter keeps counting past the pair buffer when a
    // tick oversubscribes pair_capacity (claims beyond it are recorded as
    // drops, never stored) — reading pairs[] to the raw counter walked off
    // the buffer and faulted (batch 1024 x 12.5k-subscriber mixed config)
    int n_pairs = *n_pairs_ptr;
    if (n_pairs > capacity) n_pairs = capacity;
    const int lane = threadIdx.x & 63;
    const int wave_in_wg = threadIdx.x >> 6;
    const int waves_total = gridDim.x * 4;
    for (int p = blockIdx.x * 4 + wave_in_wg; p < n_pairs; p += waves_total) {
        const PairRec pr = pairs[p];
        if (pr.user < 0) continue;
        const int mi = pr.msg;
        const int32_t len = payload_len[mi];
        const uint8_t* src = buf + payload_off[mi];
        uint8_t* dst = egress + pr.dst;
        if (lane == 0) {
            uint32_t hdr[4] = {(uint32_t)len, msg_seq[mi], 0, 0};
            if (NT) {
                typedef unsigned int v4u __attribute__((ext_vector_type(4)));
                v4u h; memcpy(&h, hdr, 16);
                __builtin_nontemporal_store(h, (v4u*)dst);
            } else {
                memcpy(dst, hdr, 16);
            }
        }
        dst += 16;
        const int32_t nvec = len >> 4;
        const bool src16 = (((uintptr_t)src) & 15) == 0;
        if (src16) {
            typedef unsigned int v4u __attribute__((ext_vector_type(4)));
            const v4u* s4 = (const v4u*)src;
            v4u* d4 = (v4u*)dst;
            for (int k = lane; k < nvec; k += 64) {
                v4u v = s4[k];
                if (NT) __builtin_nontemporal_store(v, d4 + k);
                else d4[k] = v;
            }
        } else {
            for (int k = lane; k < nvec; k += 64) {
                uint8_t tmp[16];
                memcpy(tmp, src + (size_t)k * 16, 16);
                memcpy(dst + (size_t)k * 16, tmp, 16);
            }
        }
        for (int k = (nvec << 4) + lane; k < len; k += 64) dst[k] = src[k];
    }
}

extern "C" {

void launch_k3_fanout_wave(const uint8_t* buf, const int64_t* payload_off,
                           const int32_t* payload_len, const PairRec* pairs,
                           const uint32_t* msg_seq, const int32_t* n_pairs_ptr,
                           int32_t capacity,
                           uint8_t* egress, int nt, int grid, hipStream_t s) {
    if (grid <= 0) grid = 4096;  // 4096 WGs x 4 waves = 16384 concurrent pairs
    if (nt)
        hipLaunchKernelGGL((k3_fanout_wave_t<1>), dim3(grid), dim3(256), 0, s, buf, payload_off,
                           payload_len, pairs, msg_seq, n_pairs_ptr, capacity,
                           egress);
    else
        hipLaunchKernelGGL((k3_fanout_wave_t<0>), dim3(grid), dim3(256), 0, s, buf, payload_off,
                           payload_len, pairs, msg_seq, n_pairs_ptr, capacity,
                           egress);
}


}  // extern "C"

// ---------------------------------------------------------------------------
// K3 flat fan-out for UNIFORM record sizes: work unit = one 16 B chunk of
// one delivery record, lane -> consecutive flat units (~100% lane
// utilization at any payload size; the wave-per-pair variant idles lanes on
// the tail pass). Two entry points share the template:
//   flat2: seq base passed by value (eager path)
//   flat3: seq base read from a device counter (hipGraph-capturable; a
//          host-passed base would be frozen into the captured graph)
// n_pairs is read from a device pointer and clamped to the buffer capacity.
// ---------------------------------------------------------------------------
template <int NT, bool SEQ_FROM_PTR>
__global__ void __launch_bounds__(256) k3_fanout_flat_t(
    const uint8_t* __restrict__ buf,
    const int64_t* __restrict__ payload_off,
    const int32_t* __restrict__ payload_len,
    const PairRec* __restrict__ pairs,
    uint32_t seq_base_val,
    const uint32_t* __restrict__ seq_state,
    const int32_t* __restrict__ n_pairs_ptr,
    int32_t capacity,
    int32_t units_per_pair,
    uint32_t div_magic,                   // Granlund-Montgomery multiplier for
    int32_t div_shift,                    // f/units_per_pair (exact for f<2^31,
                                          // d<=2^21; a per-unit 64-bit idiv was
                                          // ~25 VALU ops on the hottest loop)
    int32_t uniform_len,                  // >0: every record is this long —
                                          // skips the per-unit length load
    uint8_t* __restrict__ egress)
{
    typedef unsigned int v4u __attribute__((ext_vector_type(4)));
    const uint32_t seq_base = SEQ_FROM_PTR ? seq_state[0] : seq_base_val;
    int np = *n_pairs_ptr;
    if (np > capacity) np = capacity;
    const int64_t n_units = (int64_t)np * units_per_pair;
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (int64_t f = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; f < n_units; f += stride) {
        const uint32_t f32 = (uint32_t)f;  // n_units < 2^31 (capacity*units)
        const int p = (int)(((uint64_t)f32 * div_magic) >> div_shift);
        const int unit = (int)(f32 - (uint32_t)p * (uint32_t)units_per_pair);
        const PairRec pr = pairs[p];   // one dword4 load
        if (pr.user < 0) continue;
        const int mi = pr.msg;
        const int32_t len = uniform_len > 0 ? uniform_len : payload_len[mi];
        uint8_t* dst = egress + pr.dst + (size_t)unit * 16;
        if (unit == 0) {
            uint32_t hdr[4] = {(uint32_t)len, seq_base + (uint32_t)mi, 0, 0};
            v4u h; memcpy(&h, hdr, 16);
            if (NT) __builtin_nontemporal_store(h, (v4u*)dst);
            else memcpy(dst, hdr, 16);
            continue;
        }
        const uint8_t* src = buf + payload_off[mi] + (size_t)(unit - 1) * 16;
        const int32_t coff = (unit - 1) * 16;
        if (coff + 16 <= len && (((uintptr_t)src) & 15) == 0) {
            v4u v = *(const v4u*)src;
            if (NT) __builtin_nontemporal_store(v, (v4u*)dst);
            else *(v4u*)dst = v;
        } else if (coff >= len) {
            // pure pad unit: zero store so the record stride is fully
            // written (full-line NT writes, no RMW at record boundaries)
            v4u z = {0, 0, 0, 0};
            if (NT) __builtin_nontemporal_store(z, (v4u*)dst);
            else *(v4u*)dst = z;
        } else {
            // partial tail unit: zero-fill to one full 16 B store
            uint8_t tmp[16];
            #pragma unroll
            for (int b = 0; b < 16; ++b) tmp[b] = (coff + b < len) ? src[b] : 0;
            v4u v; memcpy(&v, tmp, 16);
            if (NT) __builtin_nontemporal_store(v, (v4u*)dst);
            else *(v4u*)dst = v;
        }
    }
}

// exact u32 division-by-constant: p = (f * m) >> sh for all f < 2^31,
// 1 <= d <= 2^21 (Granlund-Montgomery with one headroom bit: l = ceil_log2 d,
// m = ceil(2^(31+l)/d) < 2^32, sh = 31+l; m*d - 2^(31+l) < d <= 2^l)
static inline void u32_div_magic(int32_t d, uint32_t* m, int32_t* sh) {
    int l = 0;
    while ((1u << l) < (uint32_t)d) ++l;  // ceil_log2(d), d >= 1
    *m = (uint32_t)(((1ull << (31 + l)) + (uint64_t)d - 1) / (uint64_t)d);
    *sh = 31 + l;
}

extern "C" void launch_k3_fanout_flat2(
    const uint8_t* buf, const int64_t* payload_off, const int32_t* payload_len,
    const PairRec* pairs,
    uint32_t seq_base, const int32_t* n_pairs_ptr, int32_t capacity, int32_t units_per_pair, int32_t uniform_len,
    uint8_t* egress, int nt, int grid, hipStream_t s) {
    if (grid <= 0) grid = 16384;  // swept: 16384 > 8192 > 4096 (~0.5% each)
    uint32_t dm; int32_t dsh;
    u32_div_magic(units_per_pair, &dm, &dsh);
    if (nt)
        hipLaunchKernelGGL((k3_fanout_flat_t<1, false>), dim3(grid), dim3(256), 0, s, buf,
                           payload_off, payload_len, pairs, seq_base,
                           nullptr, n_pairs_ptr, capacity, units_per_pair, dm, dsh,
                           uniform_len, egress);
    else
        hipLaunchKernelGGL((k3_fanout_flat_t<0, false>), dim3(grid), dim3(256), 0, s, buf,
                           payload_off, payload_len, pairs, seq_base,
                           nullptr, n_pairs_ptr, capacity, units_per_pair, dm, dsh,
                           uniform_len, egress);
}

extern "C" void launch_k3_fanout_flat3(
    const uint8_t* buf, const int64_t* payload_off, const int32_t* payload_len,
    const PairRec* pairs,
    const uint32_t* seq_state, const int32_t* n_pairs_ptr, int32_t capacity,
    int32_t units_per_pair, int32_t uniform_len, uint8_t* egress, int nt, int grid,
    hipStream_t s) {
    if (grid <= 0) grid = 16384;
    uint32_t dm; int32_t dsh;
    u32_div_magic(units_per_pair, &dm, &dsh);
    if (nt)
        hipLaunchKernelGGL((k3_fanout_flat_t<1, true>), dim3(grid), dim3(256), 0, s, buf,
                           payload_off, payload_len, pairs, 0u,
                           seq_state, n_pairs_ptr, capacity, units_per_pair, dm, dsh,
                           uniform_len, egress);
    else
        hipLaunchKernelGGL((k3_fanout_flat_t<0, true>), dim3(grid), dim3(256), 0, s, buf,
                           payload_off, payload_len, pairs, 0u,
                           seq_state, n_pairs_ptr, capacity, units_per_pair, dm, dsh,
                           uniform_len, egress);
}

extern "C" __global__ void k_seq_advance(uint32_t* seq_state, int32_t m) {
    if (blockIdx.x == 0 && threadIdx.x == 0) seq_state[0] += (uint32_t)m;
}

extern "C" {


void launch_k_seq_advance(uint32_t* seq_state, int32_t m, hipStream_t s) {
    hipLaunchKernelGGL(k_seq_advance, dim3(1), dim3(1), 0, s, seq_state, m);
}

}  // extern "C"

// ---------------------------------------------------------------------------
// Transposed-mask pipeline: mask_t[W][M] instead of [M][W] so K2b's per-user
// scan over messages reads CONTIGUOUS memory (wide loads + ILP instead of a
// 1.2 KB stride per iteration — K2b is latency-bound at ~157 waves).
// ---------------------------------------------------------------------------
extern "C" __global__ void k2a_topic_mask_t(
    const uint64_t* __restrict__ sub_bitmap,  // [256][W]
    const uint8_t* __restrict__ buf,
    const int64_t* __restrict__ topics_off,
    const int32_t* __restrict__ topics_cnt,
    const int32_t* __restrict__ disc,
    uint64_t* __restrict__ mask_t,            // [W][M]
    int32_t M, int32_t W)
{
    int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (idx >= (int64_t)M * W) return;
    int m = idx / W;
    int w = idx % W;
    uint64_t acc = 0;
    if (disc[m] == 4) {
        const uint8_t* topics = buf + topics_off[m];
        int n = topics_cnt[m];
        for (int t = 0; t < n; ++t) acc |= sub_bitmap[(int64_t)topics[t] * W + w];
    }
    mask_t[(int64_t)w * M + m] = acc;
}

// uniform_rec != 0: every record in this tick is uniform_rec bytes (the
// uniform-wire broadcast shape) — ring math is closed-form and payload_len
// is never loaded. Slot claims are WAVE-AGGREGATED: a wave64 prefix-scan of
// per-lane counts and ONE atomicAdd per wave (157 atomics for 10k users
// instead of 10k — a single counter word saturates at ~88 atomics/us,
// MI355X_MICROARCH 'dequeue' row) keep the pair list contiguous.
extern "C" __global__ void k2b_fused_t(
    const uint64_t* __restrict__ mask_t,      // [W][M]
    const int32_t* __restrict__ payload_len,
    int32_t M, int32_t W, int32_t n_users,
    int64_t ring_bytes, int32_t capacity, int32_t uniform_rec,
    uint64_t* __restrict__ ring_wpos,
    int32_t* __restrict__ n_pairs,            // [1] global pair counter (zeroed)
    PairRec* __restrict__ pairs,
    uint32_t* __restrict__ drops)
{
    const int u = blockIdx.x * blockDim.x + threadIdx.x;
    const bool active = u < n_users;
    const int w = u >> 6;
    const uint64_t bit = 1ull << (u & 63);
    const uint64_t* col = mask_t + (int64_t)w * M;
    int count = 0;
    if (active) {
#pragma unroll 8
        for (int m = 0; m < M; ++m) count += (col[m] & bit) ? 1 : 0;
    }
    // wave64 inclusive scan of counts; one atomic per wave claims the span
    const int lane = threadIdx.x & 63;
    int incl = count;
    for (int d = 1; d < 64; d <<= 1) {
        int ngh = __shfl_up(incl, d);
        if (lane >= d) incl += ngh;
    }
    int wave_total = __shfl(incl, 63);
    int base = 0;
    if (lane == 0 && wave_total > 0) base = atomicAdd(n_pairs, wave_total);
    base = __shfl(base, 0);
    int slot = base + incl - count;
    if (!active || count == 0) return;

    if (uniform_rec) {
        uint64_t wpos = ring_wpos[u];
        const int32_t fit = (int32_t)((ring_bytes - wpos) / (uint64_t)uniform_rec);
        uint32_t dropped = 0;
        int emitted = 0;
        const int64_t dst_base = (int64_t)u * ring_bytes + (int64_t)wpos;
        for (int m = 0; m < M; ++m) {
            if (!(col[m] & bit)) continue;
            const bool ok = (emitted < fit) && (slot < capacity);
            if (!ok) {
                if (slot < capacity) { store_pair(pairs + slot, -1, m, 0); slot++; }
                dropped++;
                continue;
            }
            store_pair(pairs + slot, u, m, dst_base + (int64_t)emitted * uniform_rec);
            slot++;
            emitted++;
        }
        ring_wpos[u] = wpos + (uint64_t)emitted * uniform_rec;
        if (dropped) atomicAdd(drops, dropped);
        return;
    }

    uint64_t wpos = ring_wpos[u];
    uint32_t dropped = 0;
    for (int m = 0; m < M; ++m) {
        if (!(col[m] & bit)) continue;
        int32_t len = payload_len[m];
        uint64_t rec = ring_rec(len);
        bool fits_ring = (wpos + rec <= (uint64_t)ring_bytes);
        bool fits_cap = (slot < capacity);
        if (!fits_ring || !fits_cap) {
            if (fits_cap) { store_pair(pairs + slot, -1, m, 0); slot++; }
            dropped++;
            continue;
        }
        store_pair(pairs + slot, u, m, (int64_t)u * ring_bytes + wpos);
        slot++;
        wpos += rec;
    }
    ring_wpos[u] = wpos;
    if (dropped) atomicAdd(drops, dropped);
}

extern "C" {

void launch_k2a_topic_mask_t(const uint64_t* sub_bitmap, const uint8_t* buf,
                             const int64_t* topics_off, const int32_t* topics_cnt,
                             const int32_t* disc, uint64_t* mask_t, int32_t M, int32_t W,
                             hipStream_t s) {
    if (M <= 0) return;
    int64_t total = (int64_t)M * W;
    int threads = 256;
    int64_t blocks = (total + threads - 1) / threads;
    hipLaunchKernelGGL(k2a_topic_mask_t, dim3((uint32_t)blocks), dim3(threads), 0, s,
                       sub_bitmap, buf, topics_off, topics_cnt, disc, mask_t, M, W);
}

// ---------------------------------------------------------------------------
// K2b block-parallel variant for UNIFORM records.  The fused kernel above
// runs one lane per user, so a 10k-user broker fills only ~157 wave slots
// of the 1024 SIMDs (measured 142 us/tick = 15% of the broadcast tick).
// This three-launch pipeline parallelizes over (user-wave, message-block)
// while producing a BIT-IDENTICAL pair list (grouped per user, messages in
// order, same placeholder/drop semantics):
//   P1  per-(user, 32-msg block) set-bit counts
//   P2  per-user totals -> wave-aggregated pair-span claim + closed-form
//       ring math (emitted = min(total, ring fit, pair-capacity headroom))
//       + per-block exclusive prefixes
//   P3  per-(user, block) emission at precomputed offsets
// ---------------------------------------------------------------------------
#define K2B_BLK 32

extern "C" __global__ void k2b_p1_count(
    const uint64_t* __restrict__ mask_t,   // [W][M]
    int32_t M, int32_t W, int32_t NB,
    int32_t* __restrict__ bcount)          // [NB][W*64]
{
    const int wave = (blockIdx.x * blockDim.x + threadIdx.x) >> 6;
    if (wave >= W * NB) return;
    const int w = wave / NB, b = wave - (int64_t)(wave / NB) * NB;
    const int lane = threadIdx.x & 63;
    const uint64_t bit = 1ull << lane;
    const uint64_t* col = mask_t + (int64_t)w * M;
    const int m0 = b * K2B_BLK, m1 = min(M, m0 + K2B_BLK);
    int c = 0;
    for (int m = m0; m < m1; ++m) c += (col[m] & bit) ? 1 : 0;
    bcount[(int64_t)b * (W * 64) + (w * 64 + lane)] = c;
}

extern "C" __global__ void k2b_p2_bases(
    const int32_t* __restrict__ bcount,    // [NB][W*64]
    int32_t W, int32_t NB, int32_t n_users,
    int64_t ring_bytes, int32_t capacity, int32_t uniform_rec,
    uint64_t* __restrict__ ring_wpos,
    int32_t* __restrict__ n_pairs,
    int32_t* __restrict__ pprefix,         // [NB][W*64] out: per-block p-offset
    int32_t* __restrict__ ubase,           // [W*64] out: pair-slot base
    int32_t* __restrict__ ufit,            // [W*64] out: ring fit (records)
    int64_t* __restrict__ udst,            // [W*64] out: ring dst base
    uint32_t* __restrict__ drops)
{
    const int u = blockIdx.x * blockDim.x + threadIdx.x;
    const bool active = u < n_users;
    const int lane = threadIdx.x & 63;
    int total = 0;
    if (active) {
        for (int b = 0; b < NB; ++b) {
            pprefix[(int64_t)b * (W * 64) + u] = total;
            total += bcount[(int64_t)b * (W * 64) + u];
        }
    }
    // wave-aggregated claim of the contiguous per-user span
    int incl = total;
    for (int d = 1; d < 64; d <<= 1) {
        int ngh = __shfl_up(incl, d);
        if (lane >= d) incl += ngh;
    }
    int wave_total = __shfl(incl, 63);
    int base = 0;
    if (lane == 0 && wave_total > 0) base = atomicAdd(n_pairs, wave_total);
    base = __shfl(base, 0);
    const int my_base = base + incl - total;
    uint32_t dropped = 0;
    if (active) {
        ubase[u] = my_base;
        const uint64_t wpos = ring_wpos[u];
        const int32_t fit = (int32_t)((ring_bytes - wpos) / (uint64_t)uniform_rec);
        ufit[u] = fit;
        udst[u] = (int64_t)u * ring_bytes + (int64_t)wpos;
        int cap_room = capacity - my_base; if (cap_room < 0) cap_room = 0;
        int emitted = total; if (emitted > fit) emitted = fit; if (emitted > cap_room) emitted = cap_room;
        ring_wpos[u] = wpos + (uint64_t)emitted * uniform_rec;
        dropped = (uint32_t)(total - emitted);
    }
    // one drops atomic per wave
    for (int d = 1; d < 64; d <<= 1) dropped += __shfl_down(dropped, d);
    if (lane == 0 && dropped) atomicAdd(drops, dropped);
}

extern "C" __global__ void k2b_p3_emit(
    const uint64_t* __restrict__ mask_t,   // [W][M]
    const int32_t* __restrict__ pprefix,   // [NB][W*64]
    const int32_t* __restrict__ ubase,
    const int32_t* __restrict__ ufit,
    const int64_t* __restrict__ udst,
    int32_t M, int32_t W, int32_t NB, int32_t n_users,
    int32_t capacity, int32_t uniform_rec,
    PairRec* __restrict__ pairs)
{
    const int wave = (blockIdx.x * blockDim.x + threadIdx.x) >> 6;
    if (wave >= W * NB) return;
    const int w = wave / NB, b = wave - (int64_t)(wave / NB) * NB;
    const int lane = threadIdx.x & 63;
    const int u = w * 64 + lane;
    if (u >= n_users) return;
    const uint64_t bit = 1ull << lane;
    const uint64_t* col = mask_t + (int64_t)w * M;
    int p = pprefix[(int64_t)b * (W * 64) + u];
    const int base = ubase[u], fit = ufit[u];
    const int64_t dstb = udst[u];
    const int m0 = b * K2B_BLK, m1 = min(M, m0 + K2B_BLK);
    for (int m = m0; m < m1; ++m) {
        if (!(col[m] & bit)) continue;
        const int slot = base + p;
        if (slot < capacity) {
            if (p < fit)
                store_pair(pairs + slot, u, m, dstb + (int64_t)p * uniform_rec);
            else  // ring full: placeholder, counted in P2
                store_pair(pairs + slot, -1, m, 0);
        }
        p++;
    }
}

extern "C" void launch_k2b_blocks_t(
    const uint64_t* mask_t, int32_t M, int32_t W, int32_t n_users,
    int64_t ring_bytes, int32_t capacity, int32_t uniform_rec,
    uint64_t* ring_wpos, int32_t* n_pairs,
    int32_t* bcount, int32_t* pprefix, int32_t* ubase, int32_t* ufit, int64_t* udst,
    PairRec* pairs,
    uint32_t* drops, hipStream_t s) {
    if (M <= 0) return;
    const int NB = (M + K2B_BLK - 1) / K2B_BLK;
    const int waves = W * NB;
    const int threads = 256;
    const int blocks_wb = (waves * 64 + threads - 1) / threads;
    const int blocks_u = (n_users + threads - 1) / threads;
    hipLaunchKernelGGL(k2b_p1_count, dim3(blocks_wb), dim3(threads), 0, s,
                       mask_t, M, W, NB, bcount);
    hipLaunchKernelGGL(k2b_p2_bases, dim3(blocks_u), dim3(threads), 0, s,
                       bcount, W, NB, n_users, ring_bytes, capacity, uniform_rec,
                       ring_wpos, n_pairs, pprefix, ubase, ufit, udst, drops);
    hipLaunchKernelGGL(k2b_p3_emit, dim3(blocks_wb), dim3(threads), 0, s,
                       mask_t, pprefix, ubase, ufit, udst, M, W, NB, n_users,
                       capacity, uniform_rec, pairs);
}

void launch_k2b_fused_t(const uint64_t* mask_t, const int32_t* payload_len, int32_t M,
                        int32_t W, int32_t n_users, int64_t ring_bytes, int32_t capacity,
                        int32_t uniform_rec, uint64_t* ring_wpos, int32_t* n_pairs,
                        PairRec* pairs,
                        uint32_t* drops, hipStream_t s) {
    int threads = 256, blocks = (n_users + threads - 1) / threads;
    hipLaunchKernelGGL(k2b_fused_t, dim3(blocks), dim3(threads), 0, s, mask_t, payload_len, M,
                       W, n_users, ring_bytes, capacity, uniform_rec, ring_wpos, n_pairs,
                       pairs, drops);
}

}  // extern "C"

// ---------------------------------------------------------------------------
// K5b: on-device direct-message delivery pair emission. For each Direct
// message whose K5 owner lookup resolved to a LOCAL user, claim ring space
// (atomic on the user's cursor — per-sender order within a tick is
// preserved by message index only per thread; cross-sender order is
// unspecified, as in the reference's independent per-conn tasks) and append
// a delivery pair. Replaces the host-side direct routing loop (which cost a
// D2H sync per tick).
// ---------------------------------------------------------------------------
extern "C" __global__ void k5b_emit_direct(
    const int32_t* __restrict__ disc,
    const int32_t* __restrict__ owner,        // K5 output; >=0 = local user
    const int64_t* __restrict__ payload_off,  // wire offsets (fanout_wire)
    const int32_t* __restrict__ payload_len,
    int32_t M,
    int64_t ring_bytes, int32_t capacity,
    uint64_t* __restrict__ ring_wpos,
    int32_t* __restrict__ n_pairs,
    PairRec* __restrict__ pairs,
    uint32_t* __restrict__ drops)
{
    int i = blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= M) return;
    if (disc[i] != 3) return;
    int u = owner[i];
    if (u < 0) return;
    int32_t len = payload_len[i];
    uint64_t rec = ring_rec(len);
    int slot = atomicAdd(n_pairs, 1);
    if (slot >= capacity) {
        atomicAdd(drops, 1u);
        return;
    }
    // CAS claim: the round-1 add-then-rollback scheme could, with two
    // concurrent overshooters and an interleaved accept, leave an accepted
    // interval ABOVE a rolled-back one and let later claims overlap it.
    // CAS never inflates the cursor, so accepted intervals are exact and
    // a full ring drops cleanly (counted) with no spurious rollback race.
    unsigned long long cur = atomicAdd((unsigned long long*)&ring_wpos[u], 0ull);
    while (true) {
        if (cur + rec > (uint64_t)ring_bytes) {
            store_pair(pairs + slot, -1, i, 0);  // K3 skips user<0 pairs
            atomicAdd(drops, 1u);
            return;
        }
        unsigned long long prev = atomicCAS((unsigned long long*)&ring_wpos[u], cur,
                                            cur + (unsigned long long)rec);
        if (prev == cur) break;
        cur = prev;
    }
    store_pair(pairs + slot, u, i, (int64_t)u * ring_bytes + (int64_t)cur);
}

extern "C" void launch_k5b_emit_direct(
    const int32_t* disc, const int32_t* owner, const int64_t* payload_off,
    const int32_t* payload_len, int32_t M, int64_t ring_bytes, int32_t capacity,
    uint64_t* ring_wpos, int32_t* n_pairs, PairRec* pairs, uint32_t* drops, hipStream_t s) {
    int threads = 256, blocks = (M + threads - 1) / threads;
    hipLaunchKernelGGL(k5b_emit_direct, dim3(blocks), dim3(threads), 0, s, disc, owner,
                       payload_off, payload_len, M, ring_bytes, capacity, ring_wpos, n_pairs,
                       pairs, drops);
}

// ---------------------------------------------------------------------------
// K7: egress-ring compaction. Gathers every ring's used prefix into one
// contiguous staging buffer so the per-tick drain is ONE D2H copy + ONE
// C++ pump call instead of one of each per user (the socket-path
// bottleneck: reference Arc-clone forwarding has no per-recipient copies
// either, sender.rs:16-33).  wpos/dst_off are device tensors; records are
// 16-aligned so 16 B vector copies cover exactly the used bytes.
//   grid.x = ring (user), grid.y = 64 KiB chunk of that ring
// ---------------------------------------------------------------------------
#define K7_CHUNK (64 * 1024)

extern "C" __global__ void k7_compact_rings(
    const uint8_t* __restrict__ egress,
    int64_t ring_bytes,
    const int64_t* __restrict__ wpos,     // [N] used bytes per ring
    const int64_t* __restrict__ dst_off,  // [N] exclusive scan of wpos
    uint8_t* __restrict__ staging)
{
    int u = blockIdx.x;
    int64_t used = wpos[u];
    int64_t chunk = (int64_t)blockIdx.y * K7_CHUNK;
    if (chunk >= used) return;
    int64_t end = used < chunk + K7_CHUNK ? used : chunk + K7_CHUNK;
    const uint8_t* src = egress + (int64_t)u * ring_bytes;
    uint8_t* dst = staging + dst_off[u];
    for (int64_t pos = chunk + (int64_t)threadIdx.x * 16; pos < end;
         pos += (int64_t)blockDim.x * 16) {
        cdn_v4u v;
        memcpy(&v, src + pos, 16);
        *(cdn_v4u*)(dst + pos) = v;
    }
}

extern "C" void launch_k7_compact_rings(const uint8_t* egress, int64_t ring_bytes,
                                        const int64_t* wpos, const int64_t* dst_off,
                                        uint8_t* staging, int32_t n_rings,
                                        int32_t max_chunks, hipStream_t s) {
    if (n_rings <= 0 || max_chunks <= 0) return;
    hipLaunchKernelGGL(k7_compact_rings, dim3(n_rings, max_chunks), dim3(256), 0, s,
                       egress, ring_bytes, wpos, dst_off, staging);
}
