// BN254 optimal-ate pairing, 2-lane Fp2-decomposed (device-only, gfx950).
//
// K1's round-1 form ran one whole pairing per lane: the Fp12 working set
// (~110 live u64) blew past the 512-VGPR file and spilled ~5.4 KB/lane to
// scratch; PMC showed ~38% of wave time parked on scratch/memory waits
// (profiles/pmc_r01.txt) and a __launch_bounds__ sweep proved occupancy was
// not the lever.  This header halves the live state per lane by splitting
// every Fp2 across a LANE PAIR: lane (2i) holds c0 and lane (2i+1) holds c1
// of each Fp2 of verification i.  Consequences on CDNA4:
//   - Fp2 mul: 2 Fp mults/lane (Karatsuba across the pair with one
//     __shfl_xor(1) exchange) vs 3 serial — 1.5x on the mul stream;
//   - Fp2 sqr: 1 Fp mult/lane vs 2;
//   - Fp12 live state: 6 Fp = 192 B/lane vs 384 — the spill driver;
//   - control flow is identical on both lanes of a pair (same NAF, same
//     data-dependent hash-to-curve counter), so pair divergence is zero.
// The wavefront is 64 wide -> 32 verifications per wave.
//
// The e(-sig, g2) side of the verification always pairs against the FIXED
// group generator, so its Miller-loop line coefficients depend on nothing
// per-verification: k1_precompute_g2_lines (bls_kernels.hip) runs the
// R-evolution once and every verification replays the NAF walk reading the
// stored lines — the G2 point arithmetic disappears from the hot loop and
// the loads are pair-uniform (all pairs read the same record per step).
//
// Mirrors the math of csrc/common/bn254_pairing.h (shared host/device,
// golden-tested against Python bignums); outputs are bit-identical to the
// host path by construction — tests/test_gpu_kernels.py asserts it.
// Reference behavior: jellyfish BLS verify, cdn-proto/src/crypto/
// signature.rs:155-174.

#pragma once
#include "../bls/bls.h"
#include "../common/bn254.h"
#include "../common/bn254_pairing.h"

namespace bn254p2 {

using namespace bn254;

// lane-paired Fp6 / Fp12: each Fp is THIS lane's component of the Fp2
struct F6 {
    Fp c0, c1, c2;
};
struct F12 {
    F6 c0, c1;
};
struct Line {
    Fp c0, c1, c2;
};
struct G2Av {
    Fp x, y;
};
struct G2Pv {
    Fp x, y, z;
};

// serialized Line record in the precomputed-lines buffer: 6 raw Montgomery
// Fp (c0.c0, c0.c1, c1.c0, c1.c1, c2.c0, c2.c1) = 192 bytes
constexpr int LINE_REC_BYTES = 192;
// NAF doubling steps + nonzero-NAF additions + 2 Frobenius additions,
// bounded generously (actual ~92 for BN254's 6x+2 NAF)
constexpr int MAX_LINES = 128;

struct PL {
    bool hi;  // lane parity: false -> c0 component, true -> c1

    __device__ static PL self() { return PL{(threadIdx.x & 1) != 0}; }

    // exchange a 32-bit value with the partner lane: DPP quad_perm(1,0,3,2)
    // is a single full-rate VALU op vs ds_bpermute (LDS pipeline) for
    // __shfl_xor — the exchange sits on f2mul's critical path.
    __device__ static uint32_t xchg32(uint32_t v) {
        return (uint32_t)__builtin_amdgcn_mov_dpp((int)v, 0xB1 /*quad_perm 1,0,3,2*/,
                                                  0xF, 0xF, true);
    }

    // exchange an Fp with the partner lane
    __device__ static Fp xchg(const Fp& a) {
        Fp r;
#pragma unroll
        for (int i = 0; i < 4; ++i) {
            uint32_t lo = xchg32((uint32_t)a.n.v[i]);
            uint32_t hi32 = xchg32((uint32_t)(a.n.v[i] >> 32));
            r.n.v[i] = ((u64)hi32 << 32) | lo;
        }
        return r;
    }

    // ------------------------- Fp2 (lane-paired) -------------------------

    __device__ Fp f2one() const { return hi ? Fp::zero() : Fp::one(); }

    __device__ bool f2is_zero(const Fp& a) const {
        unsigned z = a.is_zero() ? 1u : 0u;
        return (z & xchg32(z)) != 0;
    }

    __device__ bool f2eq(const Fp& a, const Fp& b) const {
        unsigned e = (a == b) ? 1u : 0u;
        return (e & xchg32(e)) != 0;
    }

    // Karatsuba across the pair: lane0 computes v0=a0b0, lane1 v1=a1b1,
    // both compute v2=(a0+a1)(b0+b1); c0 = v0-v1, c1 = v2-v0-v1.
    __device__ Fp f2mul(const Fp& a, const Fp& b) const {
        Fp m1 = Fp::mul(a, b);
        Fp sa = Fp::add(a, xchg(a));
        Fp sb = Fp::add(b, xchg(b));
        Fp m2 = Fp::mul(sa, sb);
        Fp m1x = xchg(m1);
        return hi ? Fp::sub(Fp::sub(m2, m1), m1x) : Fp::sub(m1, m1x);
    }

    // (a0+a1)(a0-a1) on lane0, 2*a1*a0 on lane1 — ONE Fp mult per lane
    __device__ Fp f2sqr(const Fp& a) const {
        Fp ax = xchg(a);
        Fp s = Fp::add(a, ax);
        Fp d = Fp::sub(a, ax);
        Fp x = hi ? Fp::dbl(a) : s;
        Fp y = hi ? ax : d;
        return Fp::mul(x, y);
    }

    __device__ Fp f2mul_fp(const Fp& a, const Fp& s) const { return Fp::mul(a, s); }

    // xi = 9 + u: (9a0 - a1) + (9a1 + a0)u
    __device__ Fp f2mul_xi(const Fp& a) const {
        Fp ax = xchg(a);
        Fp nine = Fp::add(Fp::dbl(Fp::dbl(Fp::dbl(a))), a);
        return hi ? Fp::add(nine, ax) : Fp::sub(nine, ax);
    }

    __device__ Fp f2conj(const Fp& a) const { return hi ? Fp::neg(a) : a; }

    __device__ Fp f2inv(const Fp& a) const {
        Fp m = Fp::sqr(a);
        Fp n = Fp::add(m, xchg(m));  // a0^2 + a1^2 (both lanes)
        Fp ni = n.inv();
        Fp r = Fp::mul(a, ni);
        return hi ? Fp::neg(r) : r;
    }

    // this lane's component of an Fp2 constant
    __device__ Fp f2const(const Limbs4& c0, const Limbs4& c1) const {
        return Fp::from_u256(from_limbs(hi ? c1 : c0));
    }

    // ------------------------- Fp6 -------------------------

    __device__ F6 f6zero() const { return {Fp::zero(), Fp::zero(), Fp::zero()}; }
    __device__ F6 f6one() const { return {f2one(), Fp::zero(), Fp::zero()}; }

    __device__ F6 f6add(const F6& a, const F6& b) const {
        return {Fp::add(a.c0, b.c0), Fp::add(a.c1, b.c1), Fp::add(a.c2, b.c2)};
    }
    __device__ F6 f6sub(const F6& a, const F6& b) const {
        return {Fp::sub(a.c0, b.c0), Fp::sub(a.c1, b.c1), Fp::sub(a.c2, b.c2)};
    }
    __device__ F6 f6neg(const F6& a) const {
        return {Fp::neg(a.c0), Fp::neg(a.c1), Fp::neg(a.c2)};
    }

    __device__ F6 f6mul(const F6& a, const F6& b) const {
        Fp v0 = f2mul(a.c0, b.c0);
        Fp v1 = f2mul(a.c1, b.c1);
        Fp v2 = f2mul(a.c2, b.c2);
        Fp t0 = f2mul(Fp::add(a.c1, a.c2), Fp::add(b.c1, b.c2));
        t0 = Fp::sub(Fp::sub(t0, v1), v2);
        Fp r0 = Fp::add(v0, f2mul_xi(t0));
        Fp t1 = f2mul(Fp::add(a.c0, a.c1), Fp::add(b.c0, b.c1));
        t1 = Fp::sub(Fp::sub(t1, v0), v1);
        Fp r1 = Fp::add(t1, f2mul_xi(v2));
        Fp t2 = f2mul(Fp::add(a.c0, a.c2), Fp::add(b.c0, b.c2));
        t2 = Fp::sub(Fp::sub(t2, v0), v2);
        Fp r2 = Fp::add(t2, v1);
        return {r0, r1, r2};
    }

    __device__ F6 f6sqr(const F6& a) const { return f6mul(a, a); }

    __device__ F6 f6mul_f2(const F6& a, const Fp& b) const {
        return {f2mul(a.c0, b), f2mul(a.c1, b), f2mul(a.c2, b)};
    }

    __device__ F6 f6mul_by_01(const F6& f, const Fp& b0, const Fp& b1) const {
        Fp f0b0 = f2mul(f.c0, b0);
        Fp f1b1 = f2mul(f.c1, b1);
        Fp f2b0 = f2mul(f.c2, b0);
        Fp f2b1 = f2mul(f.c2, b1);
        Fp f0b1 = f2mul(f.c0, b1);
        Fp f1b0 = f2mul(f.c1, b0);
        return {Fp::add(f0b0, f2mul_xi(f2b1)), Fp::add(f0b1, f1b0), Fp::add(f1b1, f2b0)};
    }

    __device__ F6 f6mul_v(const F6& a) const { return {f2mul_xi(a.c2), a.c0, a.c1}; }

    __device__ F6 f6inv(const F6& a) const {
        Fp A = Fp::sub(f2sqr(a.c0), f2mul_xi(f2mul(a.c1, a.c2)));
        Fp B = Fp::sub(f2mul_xi(f2sqr(a.c2)), f2mul(a.c0, a.c1));
        Fp C = Fp::sub(f2sqr(a.c1), f2mul(a.c0, a.c2));
        Fp den = Fp::add(f2mul(a.c0, A),
                         f2mul_xi(Fp::add(f2mul(a.c2, B), f2mul(a.c1, C))));
        Fp di = f2inv(den);
        return {f2mul(A, di), f2mul(B, di), f2mul(C, di)};
    }

    // ------------------------- Fp12 -------------------------

    __device__ F12 f12one() const { return {f6one(), f6zero()}; }

    __device__ bool f12is_one(const F12& a) const {
        unsigned e = ((a.c0.c0 == f2one()) & a.c0.c1.is_zero() & a.c0.c2.is_zero() &
                      a.c1.c0.is_zero() & a.c1.c1.is_zero() & a.c1.c2.is_zero())
                         ? 1u
                         : 0u;
        return (e & xchg32(e)) != 0;
    }

    __device__ F12 f12mul(const F12& a, const F12& b) const {
        F6 v0 = f6mul(a.c0, b.c0);
        F6 v1 = f6mul(a.c1, b.c1);
        F6 t = f6mul(f6add(a.c0, a.c1), f6add(b.c0, b.c1));
        return {f6add(v0, f6mul_v(v1)), f6sub(f6sub(t, v0), v1)};
    }

    __device__ F12 f12sqr(const F12& a) const {
        F6 v0 = f6mul(a.c0, a.c1);
        F6 t = f6mul(f6add(a.c0, a.c1), f6add(a.c0, f6mul_v(a.c1)));
        F6 c0n = f6sub(f6sub(t, v0), f6mul_v(v0));
        return {c0n, f6add(v0, v0)};
    }

    __device__ F12 f12conj(const F12& a) const { return {a.c0, f6neg(a.c1)}; }

    __device__ F12 f12inv(const F12& a) const {
        F6 d = f6sub(f6sqr(a.c0), f6mul_v(f6sqr(a.c1)));
        F6 di = f6inv(d);
        return {f6mul(a.c0, di), f6neg(f6mul(a.c1, di))};
    }

    __device__ F12 f12mul_by_034(const F12& f, const Fp& a0, const Fp& a3,
                                 const Fp& a4) const {
        F6 x = f6mul_f2(f.c0, a0);
        F6 y = f6mul_by_01(f.c1, a3, a4);
        Fp s03 = Fp::add(a0, a3);
        F6 e = f6mul_by_01(f6add(f.c0, f.c1), s03, a4);
        return {f6add(x, f6mul_v(y)), f6sub(e, f6add(x, y))};
    }

    __device__ F12 f12cyclo_sqr(const F12& f) const {
        Fp z0 = f.c0.c0, z4 = f.c0.c1, z3 = f.c0.c2;
        Fp z2 = f.c1.c0, z1 = f.c1.c1, z5 = f.c1.c2;
        Fp t0, t1, t2, t3;
        {
            Fp a2 = f2sqr(z0), b2 = f2sqr(z1);
            t0 = Fp::add(a2, f2mul_xi(b2));
            t1 = Fp::sub(Fp::sub(f2sqr(Fp::add(z0, z1)), a2), b2);
        }
        Fp r0 = Fp::add(Fp::dbl(Fp::sub(t0, z0)), t0);
        Fp r1 = Fp::add(Fp::dbl(Fp::add(t1, z1)), t1);
        {
            Fp a2 = f2sqr(z2), b2 = f2sqr(z3);
            t0 = Fp::add(a2, f2mul_xi(b2));
            t1 = Fp::sub(Fp::sub(f2sqr(Fp::add(z2, z3)), a2), b2);
        }
        {
            Fp a2 = f2sqr(z4), b2 = f2sqr(z5);
            t2 = Fp::add(a2, f2mul_xi(b2));
            t3 = Fp::sub(Fp::sub(f2sqr(Fp::add(z4, z5)), a2), b2);
        }
        Fp r4 = Fp::add(Fp::dbl(Fp::sub(t0, z4)), t0);
        Fp r5 = Fp::add(Fp::dbl(Fp::add(t1, z5)), t1);
        Fp xt3 = f2mul_xi(t3);
        Fp r2 = Fp::add(Fp::dbl(Fp::add(xt3, z2)), xt3);
        Fp r3 = Fp::add(Fp::dbl(Fp::sub(t2, z3)), t2);
        return {{r0, r4, r3}, {r2, r1, r5}};
    }

    __device__ F12 f12frob1(const F12& a) const {
        Fp g1 = f2const(bn254c::GAMMA1_1_C0, bn254c::GAMMA1_1_C1);
        Fp g2 = f2const(bn254c::GAMMA1_2_C0, bn254c::GAMMA1_2_C1);
        Fp g3 = f2const(bn254c::GAMMA1_3_C0, bn254c::GAMMA1_3_C1);
        Fp g4 = f2const(bn254c::GAMMA1_4_C0, bn254c::GAMMA1_4_C1);
        Fp g5 = f2const(bn254c::GAMMA1_5_C0, bn254c::GAMMA1_5_C1);
        return {{f2conj(a.c0.c0), f2mul(f2conj(a.c0.c1), g2), f2mul(f2conj(a.c0.c2), g4)},
                {f2mul(f2conj(a.c1.c0), g1), f2mul(f2conj(a.c1.c1), g3),
                 f2mul(f2conj(a.c1.c2), g5)}};
    }

    __device__ F12 f12frob2(const F12& a) const {
        Fp g1 = Fp::from_u256(from_limbs(bn254c::GAMMA2_1));
        Fp g2 = Fp::from_u256(from_limbs(bn254c::GAMMA2_2));
        Fp g3 = Fp::from_u256(from_limbs(bn254c::GAMMA2_3));
        Fp g4 = Fp::from_u256(from_limbs(bn254c::GAMMA2_4));
        Fp g5 = Fp::from_u256(from_limbs(bn254c::GAMMA2_5));
        return {{a.c0.c0, Fp::mul(a.c0.c1, g2), Fp::mul(a.c0.c2, g4)},
                {Fp::mul(a.c1.c0, g1), Fp::mul(a.c1.c1, g3), Fp::mul(a.c1.c2, g5)}};
    }

    __device__ F12 f12frob3(const F12& a) const { return f12frob1(f12frob2(a)); }

    // ------------------------- Miller loop -------------------------

    __device__ Line dbl_step(G2Pv& r, const Fp& two_inv) const {
        Fp a = f2mul_fp(f2mul(r.x, r.y), two_inv);
        Fp b = f2sqr(r.y);
        Fp c = f2sqr(r.z);
        Fp bt = f2const(bn254c::B2_C0, bn254c::B2_C1);
        Fp e = f2mul(bt, Fp::add(Fp::dbl(c), c));
        Fp f = Fp::add(Fp::dbl(e), e);
        Fp g = f2mul_fp(Fp::add(b, f), two_inv);
        Fp h = Fp::sub(f2sqr(Fp::add(r.y, r.z)), Fp::add(b, c));
        Fp i = Fp::sub(e, b);
        Fp j = f2sqr(r.x);
        Fp e2 = f2sqr(e);
        r.x = f2mul(a, Fp::sub(b, f));
        r.y = Fp::sub(f2sqr(g), Fp::add(Fp::dbl(e2), e2));
        r.z = f2mul(b, h);
        return {Fp::neg(h), Fp::add(Fp::dbl(j), j), i};
    }

    __device__ Line add_step(G2Pv& r, const G2Av& q) const {
        Fp theta = Fp::sub(r.y, f2mul(q.y, r.z));
        Fp lambda = Fp::sub(r.x, f2mul(q.x, r.z));
        Fp c = f2sqr(theta);
        Fp d = f2sqr(lambda);
        Fp e = f2mul(lambda, d);
        Fp f = f2mul(r.z, c);
        Fp g = f2mul(r.x, d);
        Fp h = Fp::add(Fp::sub(e, Fp::dbl(g)), f);
        r.x = f2mul(lambda, h);
        r.y = Fp::sub(f2mul(theta, Fp::sub(g, h)), f2mul(e, r.y));
        r.z = f2mul(r.z, e);
        Fp j = Fp::sub(f2mul(theta, q.x), f2mul(lambda, q.y));
        return {lambda, Fp::neg(theta), j};
    }

    __device__ void ell(F12& f, const Line& l, const Fp& px, const Fp& py) const {
        Fp c0 = f2mul_fp(l.c0, py);
        Fp c1 = f2mul_fp(l.c1, px);
        f = f12mul_by_034(f, c0, c1, l.c2);
    }

    __device__ G2Av g2frob(const G2Av& q) const {
        Fp fx = f2const(bn254c::FROB_X_C0, bn254c::FROB_X_C1);
        Fp fy = f2const(bn254c::FROB_Y_C0, bn254c::FROB_Y_C1);
        return {f2mul(f2conj(q.x), fx), f2mul(f2conj(q.y), fy)};
    }
};

// ---------------------------------------------------------------------------
// big building blocks (noinline: a fully-inlined pairing explodes compile
// time — round-1 lesson, see ROUND1_NOTES.md)
// ---------------------------------------------------------------------------

__device__ __attribute__((noinline)) F12 miller_loop2(const PL& L, const Fp& px,
                                                      const Fp& py, const G2Av& q) {
    Fp two_inv = Fp::from_u64(2).inv();
    G2Pv r{q.x, q.y, L.f2one()};
    G2Av negq{q.x, Fp::neg(q.y)};
    F12 f = L.f12one();
    BN_NOUNROLL for (int i = bn254c::ATE_NAF_LEN - 2; i >= 0; --i) {
        f = L.f12sqr(f);
        Line l = L.dbl_step(r, two_inv);
        L.ell(f, l, px, py);
        int8_t d = bn254c::ATE_NAF[i];
        if (d == 1) {
            l = L.add_step(r, q);
            L.ell(f, l, px, py);
        } else if (d == -1) {
            l = L.add_step(r, negq);
            L.ell(f, l, px, py);
        }
    }
    G2Av q1 = L.g2frob(q);
    G2Av q2 = L.g2frob(q1);
    q2.y = Fp::neg(q2.y);
    Line l = L.add_step(r, q1);
    L.ell(f, l, px, py);
    l = L.add_step(r, q2);
    L.ell(f, l, px, py);
    return f;
}

// load this lane's components of line record `idx` from the precomputed
// buffer (raw Montgomery limbs; pair-uniform address stream)
__device__ inline Line load_line(const PL& L, const uint8_t* lines, int idx) {
    const u64* rec = (const u64*)(lines + (size_t)idx * LINE_REC_BYTES);
    int o = L.hi ? 4 : 0;
    Line r;
#pragma unroll
    for (int i = 0; i < 4; ++i) r.c0.n.v[i] = rec[o + i];
#pragma unroll
    for (int i = 0; i < 4; ++i) r.c1.n.v[i] = rec[8 + o + i];
#pragma unroll
    for (int i = 0; i < 4; ++i) r.c2.n.v[i] = rec[16 + o + i];
    return r;
}

// Miller loop against the FIXED g2 generator: replays the NAF walk reading
// precomputed line coefficients — no G2 arithmetic at all.
__device__ __attribute__((noinline)) F12 miller_loop2_lines(const PL& L, const Fp& px,
                                                            const Fp& py,
                                                            const uint8_t* lines) {
    F12 f = L.f12one();
    int cur = 0;
    BN_NOUNROLL for (int i = bn254c::ATE_NAF_LEN - 2; i >= 0; --i) {
        f = L.f12sqr(f);
        Line l = load_line(L, lines, cur++);
        L.ell(f, l, px, py);
        if (bn254c::ATE_NAF[i] != 0) {
            l = load_line(L, lines, cur++);
            L.ell(f, l, px, py);
        }
    }
    Line l = load_line(L, lines, cur++);
    L.ell(f, l, px, py);
    l = load_line(L, lines, cur++);
    L.ell(f, l, px, py);
    return f;
}

__device__ __attribute__((noinline)) F12 pow_by_x2(const PL& L, const F12& a) {
    uint64_t e = bn254c::BN_X;
    F12 result = L.f12one();
    bool started = false;
    BN_NOUNROLL for (int b = 63; b >= 0; --b) {
        if (started) result = L.f12cyclo_sqr(result);
        if ((e >> b) & 1) {
            if (started) result = L.f12mul(result, a);
            else {
                result = a;
                started = true;
            }
        }
    }
    return result;
}

__device__ inline F12 easy_part2(const PL& L, const F12& f) {
    F12 f1 = L.f12conj(f);
    F12 f2 = L.f12inv(f);
    F12 r = L.f12mul(f1, f2);
    return L.f12mul(L.f12frob2(r), r);
}

__device__ __attribute__((noinline)) F12 hard_exp_chain2(const PL& L, const F12& r_in) {
    F12 r = r_in;
    F12 y0 = L.f12conj(pow_by_x2(L, r));
    F12 y1 = L.f12cyclo_sqr(y0);
    F12 y2 = L.f12cyclo_sqr(y1);
    F12 y3 = L.f12mul(y2, y1);
    F12 y4 = L.f12conj(pow_by_x2(L, y3));
    F12 y5 = L.f12cyclo_sqr(y4);
    F12 y6 = L.f12conj(pow_by_x2(L, y5));
    y3 = L.f12conj(y3);
    y6 = L.f12conj(y6);
    F12 y7 = L.f12mul(y6, y4);
    F12 y8 = L.f12mul(y7, y3);
    F12 y9 = L.f12mul(y8, y1);
    F12 y10 = L.f12mul(y8, y4);
    F12 y11 = L.f12mul(y10, r);
    F12 y12 = L.f12frob1(y9);
    F12 y13 = L.f12mul(y12, y11);
    y8 = L.f12frob2(y8);
    F12 y14 = L.f12mul(y8, y13);
    r = L.f12conj(r);
    F12 y15 = L.f12mul(r, y9);
    y15 = L.f12frob3(y15);
    return L.f12mul(y15, y14);
}

__device__ inline F12 final_exponentiation2(const PL& L, const F12& f) {
    return hard_exp_chain2(L, easy_part2(L, f));
}

// ---------------------------------------------------------------------------
// G2 jacobian arithmetic over lane-paired Fp2 — for the subgroup check
// ---------------------------------------------------------------------------

__device__ inline G2Pv g2pv_dbl(const PL& L, const G2Pv& p) {
    if (L.f2is_zero(p.z)) return p;
    Fp A = L.f2sqr(p.x);
    Fp B = L.f2sqr(p.y);
    Fp C = L.f2sqr(B);
    Fp t = L.f2sqr(Fp::add(p.x, B));
    Fp D = Fp::dbl(Fp::sub(Fp::sub(t, A), C));
    Fp E = Fp::add(Fp::dbl(A), A);
    Fp Fq = L.f2sqr(E);
    Fp X3 = Fp::sub(Fq, Fp::dbl(D));
    Fp eight_c = Fp::dbl(Fp::dbl(Fp::dbl(C)));
    Fp Y3 = Fp::sub(L.f2mul(E, Fp::sub(D, X3)), eight_c);
    Fp Z3 = Fp::dbl(L.f2mul(p.y, p.z));
    return {X3, Y3, Z3};
}

__device__ inline G2Pv g2pv_add(const PL& L, const G2Pv& p, const G2Pv& q) {
    if (L.f2is_zero(p.z)) return q;
    if (L.f2is_zero(q.z)) return p;
    Fp Z1Z1 = L.f2sqr(p.z);
    Fp Z2Z2 = L.f2sqr(q.z);
    Fp U1 = L.f2mul(p.x, Z2Z2);
    Fp U2 = L.f2mul(q.x, Z1Z1);
    Fp S1 = L.f2mul(L.f2mul(p.y, q.z), Z2Z2);
    Fp S2 = L.f2mul(L.f2mul(q.y, p.z), Z1Z1);
    if (L.f2eq(U1, U2)) {
        if (L.f2eq(S1, S2)) return g2pv_dbl(L, p);
        return {L.f2one(), L.f2one(), Fp::zero()};
    }
    Fp H = Fp::sub(U2, U1);
    Fp I = L.f2sqr(Fp::dbl(H));
    Fp J = L.f2mul(H, I);
    Fp rr = Fp::dbl(Fp::sub(S2, S1));
    Fp V = L.f2mul(U1, I);
    Fp X3 = Fp::sub(Fp::sub(L.f2sqr(rr), J), Fp::dbl(V));
    Fp Y3 = Fp::sub(L.f2mul(rr, Fp::sub(V, X3)), Fp::dbl(L.f2mul(S1, J)));
    Fp Z3 = L.f2mul(Fp::dbl(L.f2mul(p.z, q.z)), H);
    return {X3, Y3, Z3};
}

// subgroup membership: psi(P) == [6x^2]P  (see csrc/bls/bls.h g2_in_subgroup
// for the soundness argument; identical math, lane-paired)
__device__ __attribute__((noinline)) bool g2_in_subgroup2(const PL& L, const Fp& x,
                                                          const Fp& y) {
    U256 k = from_limbs(bn254c::SIX_X_SQ);
    G2Pv base{x, y, L.f2one()};
    G2Pv acc{L.f2one(), L.f2one(), Fp::zero()};
    bool started = false;
    BN_NOUNROLL for (int i = 3; i >= 0; --i) {
        BN_NOUNROLL for (int b = 63; b >= 0; --b) {
            if (started) acc = g2pv_dbl(L, acc);
            if ((k.v[i] >> b) & 1) {
                if (started) acc = g2pv_add(L, acc, base);
                else {
                    acc = base;
                    started = true;
                }
            }
        }
    }
    if (L.f2is_zero(acc.z)) return false;
    // to affine
    Fp zi = L.f2inv(acc.z);
    Fp zi2 = L.f2sqr(zi);
    Fp ax = L.f2mul(acc.x, zi2);
    Fp ay = L.f2mul(acc.y, L.f2mul(zi2, zi));
    G2Av psi = L.g2frob({x, y});
    return L.f2eq(ax, psi.x) && L.f2eq(ay, psi.y);
}

// ---------------------------------------------------------------------------
// verkey load: this lane reads ITS components; range/curve/subgroup checks
// combined across the pair
// ---------------------------------------------------------------------------
__device__ inline bool verkey_load2(const PL& L, const uint8_t* in, Fp& x, Fp& y) {
    U256 p = from_limbs(bn254c::P);
    U256 xv = bls::u256_from_le(in + (L.hi ? 32 : 0));
    U256 yv = bls::u256_from_le(in + 64 + (L.hi ? 32 : 0));
    unsigned ok = (!u256_gte(xv, p) && !u256_gte(yv, p)) ? 1u : 0u;
    if ((ok & PL::xchg32(ok)) == 0) return false;
    x = Fp::from_u256(xv);
    y = Fp::from_u256(yv);
    if (L.f2is_zero(x) && L.f2is_zero(y)) return false;  // infinity
    // on curve: y^2 == x^3 + b2
    Fp b2 = L.f2const(bn254c::B2_C0, bn254c::B2_C1);
    Fp lhs = L.f2sqr(y);
    Fp rhs = Fp::add(L.f2mul(L.f2sqr(x), x), b2);
    if (!L.f2eq(lhs, rhs)) return false;
    return g2_in_subgroup2(L, x, y);
}

// ---------------------------------------------------------------------------
// G1 scalar multiplication by a SMALL positive scalar (the batch
// coefficients): left-to-right jacobian double-and-add with NO infinity
// branches — impossible here because 1 <= r < 2^64 << group order and the
// base has prime order, so no intermediate is ever the identity and the
// mixed add never sees equal points after the first doubling.  (The
// generic Point<Fp>::scalar_mul template hangs on gfx950 in this kernel —
// compiler pathology around its data-dependent early returns under
// divergence; this straight-line form also drops ~2x the work.)
// ---------------------------------------------------------------------------

__device__ inline void g1_dbl_j(Fp& X, Fp& Y, Fp& Z) {
    // dbl-2009-l
    Fp A = Fp::sqr(X);
    Fp B = Fp::sqr(Y);
    Fp C = Fp::sqr(B);
    Fp t = Fp::sqr(Fp::add(X, B));
    Fp D = Fp::dbl(Fp::sub(Fp::sub(t, A), C));
    Fp E = Fp::add(Fp::dbl(A), A);
    Fp F2 = Fp::sqr(E);
    Fp X3 = Fp::sub(F2, Fp::dbl(D));
    Fp eight_c = Fp::dbl(Fp::dbl(Fp::dbl(C)));
    Fp Y3 = Fp::sub(Fp::mul(E, Fp::sub(D, X3)), eight_c);
    Fp Z3 = Fp::dbl(Fp::mul(Y, Z));
    X = X3; Y = Y3; Z = Z3;
}

__device__ inline void g1_addmixed_j(Fp& X, Fp& Y, Fp& Z, const Fp& px, const Fp& py) {
    // madd-2007-bl (Z2 = 1)
    Fp Z1Z1 = Fp::sqr(Z);
    Fp U2 = Fp::mul(px, Z1Z1);
    Fp S2 = Fp::mul(Fp::mul(py, Z), Z1Z1);
    Fp H = Fp::sub(U2, X);
    Fp HH = Fp::sqr(H);
    Fp I = Fp::dbl(Fp::dbl(HH));
    Fp J = Fp::mul(H, I);
    Fp rr = Fp::dbl(Fp::sub(S2, Y));
    Fp V = Fp::mul(X, I);
    Fp X3 = Fp::sub(Fp::sub(Fp::sqr(rr), J), Fp::dbl(V));
    Fp Y3 = Fp::sub(Fp::mul(rr, Fp::sub(V, X3)), Fp::dbl(Fp::mul(Y, J)));
    Fp Z3 = Fp::sub(Fp::sub(Fp::sqr(Fp::add(Z, H)), Z1Z1), HH);
    X = X3; Y = Y3; Z = Z3;
}

__device__ __attribute__((noinline)) void g1_smul_affine(const Fp& px, const Fp& py,
                                                         uint64_t r, Fp& ox, Fp& oy) {
    int top = 63 - __builtin_clzll(r | 1);
    Fp X = px, Y = py, Z = Fp::one();
    BN_NOUNROLL for (int b = top - 1; b >= 0; --b) {
        g1_dbl_j(X, Y, Z);
        if ((r >> b) & 1) g1_addmixed_j(X, Y, Z, px, py);
    }
    Fp zi = Z.inv();
    Fp zi2 = Fp::sqr(zi);
    ox = Fp::mul(X, zi2);
    oy = Fp::mul(Y, Fp::mul(zi2, zi));
}

// ---------------------------------------------------------------------------
// wave-level helpers for batched product verification
// ---------------------------------------------------------------------------

// exchange an Fp with the lane at (lane ^ mask) — used by the wave butterfly
// product (mask >= 2 keeps pair parity aligned, so c0/c1 lanes line up)
__device__ inline Fp xchg_mask(const Fp& a, int mask) {
    Fp r;
#pragma unroll
    for (int i = 0; i < 4; ++i)
        r.n.v[i] = (u64)__shfl_xor((unsigned long long)a.n.v[i], mask, 64);
    return r;
}

__device__ inline F12 shfl_f12(const F12& a, int mask) {
    F12 r;
    r.c0.c0 = xchg_mask(a.c0.c0, mask);
    r.c0.c1 = xchg_mask(a.c0.c1, mask);
    r.c0.c2 = xchg_mask(a.c0.c2, mask);
    r.c1.c0 = xchg_mask(a.c1.c0, mask);
    r.c1.c1 = xchg_mask(a.c1.c1, mask);
    r.c1.c2 = xchg_mask(a.c1.c2, mask);
    return r;
}

}  // namespace bn254p2
