// K1: batched BLS-over-BN254 signature verification on MI355X (gfx950).
//
// The auth-storm hot path (BASELINE.json config 2: 10k subscriber auths):
// the reference verifies one signature per connection on the CPU
// (cdn-proto/src/connection/auth/marshal.rs:66-72, broker.rs:266-273); here a
// batch of N pending auths is verified in one launch, one wavefront LANE per
// verification — the field tower (csrc/common/bn254*.h) is shared source
// with the host implementation, so device results are golden-tested
// bit-for-bit against host results.
//
// Register pressure: a full pairing needs several Fp12 temporaries
// (12 * 4 = 48 u64 each), far beyond the 512-VGPR file — the kernel spills
// to scratch by design; throughput comes from running 10k+ independent
// verifications across 256 CUs. A lane-cooperative Fp-parallel variant is a
// later optimization if profiling warrants it.

#include <hip/hip_runtime.h>
#include <stdint.h>

#include "../bls/bls.h"
#include "bn254_pair2.h"

using namespace bn254;

// __launch_bounds__(64, 1): one wave per SIMD unlocks the full 512-VGPR
// budget per lane — the pairing's working set (~110 u64 live values) spills
// ~5.4 KB/lane to scratch at the default budget, and scratch waits were
// ~74% of wave time (profiles/pmc_r01.txt). Occupancy is irrelevant here:
// the kernel is latency-bound per lane, not bandwidth- or wave-limited.
extern "C" __global__ void __launch_bounds__(64, 1)
k1_bls_verify(
    const uint8_t* __restrict__ vks,     // [N][128]
    const uint8_t* __restrict__ sigs,    // [N][64]
    uint8_t* __restrict__ msgs,          // flat namespaced messages, each with 1 spare byte
    const int64_t* __restrict__ moff,    // [N+1] offsets (end includes spare byte)
    int32_t N,
    int32_t* __restrict__ ok)            // [N] out: 1 valid, 0 invalid
{
    int i = blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= N) return;

    bls::VerKey vk;
    if (!bls::verkey_deserialize(vks + (size_t)i * 128, vk)) { ok[i] = 0; return; }
    Fp sx, sy;
    if (!bls::sig_deserialize(sigs + (size_t)i * 64, sx, sy)) { ok[i] = 0; return; }
    uint8_t* scratch = msgs + moff[i];
    uint32_t msg_len = (uint32_t)(moff[i + 1] - moff[i] - 1);  // spare byte excluded
    ok[i] = bls::verify_core(vk, scratch, msg_len, sx, sy) ? 1 : 0;
}

// ---------------------------------------------------------------------------
// K1 v2: 2-lane Fp2-decomposed verification (see bn254_pair2.h header
// comment) — lane pair (2i, 2i+1) verifies signature i; the fixed-g2 side
// of the pairing product reads precomputed Miller-loop line coefficients.
// ---------------------------------------------------------------------------

// Runs the single-lane Miller-loop R-evolution ONCE for Q = g2 generator and
// serializes every line's raw Montgomery limbs in NAF-walk order (the exact
// order miller_loop2_lines replays). One thread; launched once per process.
extern "C" __global__ void k1_precompute_g2_lines(uint8_t* __restrict__ out,
                                                  int32_t* __restrict__ n_out) {
    if (blockIdx.x != 0 || threadIdx.x != 0) return;
    Fp two_inv = Fp::from_u64(2).inv();
    G2 gen = g2_generator();
    G2Affine q{gen.X, gen.Y};
    G2Affine negq{q.x, Fp2::neg(q.y)};
    G2Proj r{q.x, q.y, Fp2::one()};
    int cur = 0;
    auto emit = [&](const LineCoeffs& l) {
        u64* rec = (u64*)(out + (size_t)cur * bn254p2::LINE_REC_BYTES);
        for (int i = 0; i < 4; ++i) rec[i] = l.c0.c0.n.v[i];
        for (int i = 0; i < 4; ++i) rec[4 + i] = l.c0.c1.n.v[i];
        for (int i = 0; i < 4; ++i) rec[8 + i] = l.c1.c0.n.v[i];
        for (int i = 0; i < 4; ++i) rec[12 + i] = l.c1.c1.n.v[i];
        for (int i = 0; i < 4; ++i) rec[16 + i] = l.c2.c0.n.v[i];
        for (int i = 0; i < 4; ++i) rec[20 + i] = l.c2.c1.n.v[i];
        ++cur;
    };
    BN_NOUNROLL for (int i = bn254c::ATE_NAF_LEN - 2; i >= 0; --i) {
        emit(doubling_step(r, two_inv));
        int8_t d = bn254c::ATE_NAF[i];
        if (d == 1) emit(addition_step(r, q));
        else if (d == -1) emit(addition_step(r, negq));
    }
    G2Affine q1 = g2_frobenius(q);
    G2Affine q2 = g2_frobenius(q1);
    q2.y = Fp2::neg(q2.y);
    emit(addition_step(r, q1));
    emit(addition_step(r, q2));
    *n_out = cur;
}

extern "C" __global__ void __launch_bounds__(64, 1)
k1_bls_verify2(
    const uint8_t* __restrict__ vks,      // [N][128]
    const uint8_t* __restrict__ sigs,     // [N][64]
    uint8_t* __restrict__ msgs,           // namespaced messages + 1 spare byte each
    const int64_t* __restrict__ moff,     // [N+1]
    const uint8_t* __restrict__ g2_lines, // precomputed fixed-g2 line coeffs
    int32_t N,
    int32_t* __restrict__ ok)
{
    int lane = blockIdx.x * blockDim.x + threadIdx.x;
    int v = lane >> 1;  // verification index: one PAIR of lanes per signature
    if (v >= N) return;
    bn254p2::PL L = bn254p2::PL::self();

    // verkey: this lane loads its Fp2 components; checks combined pairwise
    Fp vkx, vky;
    if (!bn254p2::verkey_load2(L, vks + (size_t)v * 128, vkx, vky)) {
        if (!L.hi) ok[v] = 0;
        return;
    }
    // signature (G1): plain Fp, duplicated across the pair (lockstep-free)
    Fp sx, sy;
    if (!bls::sig_deserialize(sigs + (size_t)v * 64, sx, sy)) {
        if (!L.hi) ok[v] = 0;
        return;
    }
    // hash-to-G1: identical data on both lanes of the pair
    uint8_t* scratch = msgs + moff[v];
    uint32_t msg_len = (uint32_t)(moff[v + 1] - moff[v] - 1);
    Fp hx, hy;
    if (!bls::hash_to_g1_with_scratch(scratch, msg_len, hx, hy)) {
        if (!L.hi) ok[v] = 0;
        return;
    }
    // e(H, pk) * e(-sig, g2) == 1, one shared final exponentiation;
    // the fixed-g2 loop replays precomputed lines (no G2 arithmetic)
    bn254p2::F12 ml1 = bn254p2::miller_loop2(L, hx, hy, {vkx, vky});
    bn254p2::F12 ml2 = bn254p2::miller_loop2_lines(L, sx, Fp::neg(sy), g2_lines);
    bn254p2::F12 f = L.f12mul(ml1, ml2);
    bool good = L.f12is_one(bn254p2::final_exponentiation2(L, f));
    if (!L.hi) ok[v] = good ? 1 : 0;
}

// ---------------------------------------------------------------------------
// K1 v3: wave-batched product verification.  Standard small-exponent batch
// verification: with per-item secret random coefficients r_i in [1, 2^64),
//     prod_i [ e([r_i]H_i, pk_i) * e(-[r_i]sig_i, g2) ] == 1
// holds iff every item verifies, except with probability ~2^-64 (an
// adversary cannot craft cancelling cofactors without predicting r_i).
// Each WAVE (32 lane-pairs) multiplies its items' Miller products via a
// shuffle butterfly and runs ONE shared final exponentiation; a wave that
// fails the batched check falls back to exact per-item final
// exponentiations, so per-item verdicts are always exact.  Marshal-side
// auth storms are overwhelmingly valid, so the common path amortizes the
// final exponentiation 32x.  (Reference semantics stay per-item:
// marshal.rs:66-72 accepts/rejects each connection individually.)
// ---------------------------------------------------------------------------
extern "C" __global__ void __launch_bounds__(64, 1)
k1_bls_verify_wave(
    const uint8_t* __restrict__ vks,      // [N][128]
    const uint8_t* __restrict__ sigs,     // [N][64]
    uint8_t* __restrict__ msgs,
    const int64_t* __restrict__ moff,     // [N+1]
    const uint8_t* __restrict__ g2_lines,
    const uint64_t* __restrict__ rand_r,  // [N] secret batch coefficients
    int32_t N,
    int32_t* __restrict__ ok)
{
    int lane = blockIdx.x * blockDim.x + threadIdx.x;
    int v = lane >> 1;
    bn254p2::PL L = bn254p2::PL::self();
    bool active = v < N;
    bool item_ok = false;
    Fp vkx, vky, hx, hy, sx, sy;
    if (active) {
        item_ok = bn254p2::verkey_load2(L, vks + (size_t)v * 128, vkx, vky);
        if (item_ok) item_ok = bls::sig_deserialize(sigs + (size_t)v * 64, sx, sy);
        if (item_ok) {
            uint8_t* scratch = msgs + moff[v];
            uint32_t msg_len = (uint32_t)(moff[v + 1] - moff[v] - 1);
            item_ok = bls::hash_to_g1_with_scratch(scratch, msg_len, hx, hy);
        }
    }
    bn254p2::F12 fi = L.f12one();
    if (item_ok) {
        // scale BOTH G1 inputs by the item's secret coefficient
        uint64_t r = rand_r[v] | 1;  // never zero
        Fp shx, shy, ssx, ssy;
        bn254p2::g1_smul_affine(hx, hy, r, shx, shy);
        bn254p2::g1_smul_affine(sx, Fp::neg(sy), r, ssx, ssy);
        bn254p2::F12 ml1 = bn254p2::miller_loop2(L, shx, shy, {vkx, vky});
        bn254p2::F12 ml2 = bn254p2::miller_loop2_lines(L, ssx, ssy, g2_lines);
        fi = L.f12mul(ml1, ml2);
    }
    // butterfly product across the wave's 32 pairs (all lanes converged;
    // masks >= 2 keep the c0/c1 pair parity aligned)
    bn254p2::F12 f = fi;
    for (int mask = 2; mask <= 32; mask <<= 1)
        f = L.f12mul(f, bn254p2::shfl_f12(f, mask));
    bool wave_ok = L.f12is_one(bn254p2::final_exponentiation2(L, f));
    int verdict;
    if (wave_ok) {
        verdict = item_ok ? 1 : 0;
    } else {
        // rare path: exact per-item check (invalid signature in the wave)
        verdict = (item_ok &&
                   L.f12is_one(bn254p2::final_exponentiation2(L, fi))) ? 1 : 0;
    }
    if (active && !L.hi) ok[v] = verdict;
}

// debug bisect of the v3 pipeline (mode gates how far each lane goes):
// 1 = parse+hash, 2 = +G1 scalar muls/to_affine, 3 = +millers,
// 4 = +per-item FE (no butterfly), 5 = full v3
extern "C" __global__ void __launch_bounds__(64, 1)
k1_dbg_wave(const uint8_t* vks, const uint8_t* sigs, uint8_t* msgs,
            const int64_t* moff, const uint8_t* g2_lines, const uint64_t* rand_r,
            int32_t N, int32_t mode, int32_t* ok)
{
    int lane = blockIdx.x * blockDim.x + threadIdx.x;
    int v = lane >> 1;
    bn254p2::PL L = bn254p2::PL::self();
    bool active = v < N;
    bool item_ok = false;
    Fp vkx, vky, hx, hy, sx, sy;
    if (active) {
        item_ok = bn254p2::verkey_load2(L, vks + (size_t)v * 128, vkx, vky);
        if (item_ok) item_ok = bls::sig_deserialize(sigs + (size_t)v * 64, sx, sy);
        if (item_ok) {
            uint8_t* scratch = msgs + moff[v];
            uint32_t msg_len = (uint32_t)(moff[v + 1] - moff[v] - 1);
            item_ok = bls::hash_to_g1_with_scratch(scratch, msg_len, hx, hy);
        }
    }
    if (mode <= 1) { if (active && !L.hi) ok[v] = item_ok ? 1 : 0; return; }
    if (mode == 20 || mode == 21) {  // scalar_mul only, no to_affine
        Fp outb = Fp::zero();
        if (item_ok) {
            uint64_t r = (mode == 20) ? 3ull : (rand_r[v] | 1);
            Fp ax, ay;
            bn254p2::g1_smul_affine(hx, hy, r, ax, ay);
            outb = ax;
        }
        if (active && !L.hi) ok[v] = (int32_t)(outb.n.v[0] & 1);
        return;
    }
    if (mode == 22) {  // to_affine only (inv of a known-good Z=1 point? use hx)
        Fp outb = Fp::zero();
        if (item_ok) {
            G1 p{hx, hy, Fp::from_u64(2)};
            Fp ax, ay;
            p.to_affine(ax, ay);
            outb = ax;
        }
        if (active && !L.hi) ok[v] = (int32_t)(outb.n.v[0] & 1);
        return;
    }
    bn254p2::F12 fi = L.f12one();
    Fp shx = hx, shy = hy, ssx = sx, ssy = Fp::neg(sy);
    if (item_ok) {
        uint64_t r = rand_r[v] | 1;
        bn254p2::g1_smul_affine(hx, hy, r, shx, shy);
        bn254p2::g1_smul_affine(sx, Fp::neg(sy), r, ssx, ssy);
    }
    if (mode <= 2) { if (active && !L.hi) ok[v] = (int32_t)(shx.n.v[0] & 1); return; }
    if (item_ok) {
        bn254p2::F12 ml1 = bn254p2::miller_loop2(L, shx, shy, {vkx, vky});
        bn254p2::F12 ml2 = bn254p2::miller_loop2_lines(L, ssx, ssy, g2_lines);
        fi = L.f12mul(ml1, ml2);
    }
    if (mode <= 3) { if (active && !L.hi) ok[v] = (int32_t)(fi.c0.c0.n.v[0] & 1); return; }
    if (mode <= 4) {
        bool one = L.f12is_one(bn254p2::final_exponentiation2(L, fi));
        if (active && !L.hi) ok[v] = (item_ok && one) ? 1 : 0;
        return;
    }
    bn254p2::F12 f = fi;
    for (int mask = 2; mask <= 32; mask <<= 1)
        f = L.f12mul(f, bn254p2::shfl_f12(f, mask));
    bool wave_ok = L.f12is_one(bn254p2::final_exponentiation2(L, f));
    if (mode == 6) { if (active && !L.hi) ok[v] = wave_ok ? 1 : 0; return; }
    int verdict;
    if (wave_ok) verdict = item_ok ? 1 : 0;
    else verdict = (item_ok && L.f12is_one(bn254p2::final_exponentiation2(L, fi))) ? 1 : 0;
    if (active && !L.hi) ok[v] = verdict;
}

extern "C" void launch_k1_dbg_wave(const uint8_t* vks, const uint8_t* sigs, uint8_t* msgs,
                                   const int64_t* moff, const uint8_t* g2_lines,
                                   const uint64_t* rand_r, int32_t N, int32_t mode,
                                   int32_t* ok, hipStream_t s) {
    int blocks = (2 * N + 63) / 64;
    hipLaunchKernelGGL(k1_dbg_wave, dim3(blocks), dim3(64), 0, s, vks, sigs, msgs, moff,
                       g2_lines, rand_r, N, mode, ok);
}

// Device self-test: hash_to_g1 + sign-shaped scalar mul, for golden tests.
extern "C" __global__ void __launch_bounds__(64)
k1_hash_to_g1(
    uint8_t* __restrict__ msgs, const int64_t* __restrict__ moff, int32_t N,
    uint8_t* __restrict__ out)  // [N][64]
{
    int i = blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= N) return;
    Fp hx, hy;
    uint8_t* scratch = msgs + moff[i];
    uint32_t len = (uint32_t)(moff[i + 1] - moff[i] - 1);
    if (bls::hash_to_g1_with_scratch(scratch, len, hx, hy)) {
        bls::sig_serialize(hx, hy, out + (size_t)i * 64);
    } else {
        for (int j = 0; j < 64; ++j) out[(size_t)i * 64 + j] = 0;
    }
}

extern "C" {

void launch_k1_bls_verify(const uint8_t* vks, const uint8_t* sigs, uint8_t* msgs,
                          const int64_t* moff, int32_t N, int32_t* ok, hipStream_t s) {
    int threads = 64;  // one wave per block: scratch-heavy kernel, keep blocks small
    int blocks = (N + threads - 1) / threads;
    hipLaunchKernelGGL(k1_bls_verify, dim3(blocks), dim3(threads), 0, s, vks, sigs, msgs,
                       moff, N, ok);
}

void launch_k1_hash_to_g1(uint8_t* msgs, const int64_t* moff, int32_t N, uint8_t* out,
                          hipStream_t s) {
    int threads = 64;
    int blocks = (N + threads - 1) / threads;
    hipLaunchKernelGGL(k1_hash_to_g1, dim3(blocks), dim3(threads), 0, s, msgs, moff, N, out);
}

void launch_k1_precompute_g2_lines(uint8_t* out, int32_t* n_out, hipStream_t s) {
    hipLaunchKernelGGL(k1_precompute_g2_lines, dim3(1), dim3(1), 0, s, out, n_out);
}

void launch_k1_bls_verify2(const uint8_t* vks, const uint8_t* sigs, uint8_t* msgs,
                           const int64_t* moff, const uint8_t* g2_lines, int32_t N,
                           int32_t* ok, hipStream_t s) {
    int threads = 64;                       // 32 verifications per wave (2 lanes each)
    int lanes = 2 * N;
    int blocks = (lanes + threads - 1) / threads;
    hipLaunchKernelGGL(k1_bls_verify2, dim3(blocks), dim3(threads), 0, s, vks, sigs, msgs,
                       moff, g2_lines, N, ok);
}

void launch_k1_bls_verify_wave(const uint8_t* vks, const uint8_t* sigs, uint8_t* msgs,
                               const int64_t* moff, const uint8_t* g2_lines,
                               const uint64_t* rand_r, int32_t N, int32_t* ok,
                               hipStream_t s) {
    int threads = 64;
    int lanes = 2 * N;
    int blocks = (lanes + threads - 1) / threads;
    hipLaunchKernelGGL(k1_bls_verify_wave, dim3(blocks), dim3(threads), 0, s, vks, sigs,
                       msgs, moff, g2_lines, rand_r, N, ok);
}

}  // extern "C"
