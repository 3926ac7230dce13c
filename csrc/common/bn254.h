// BN254 (alt_bn128) field and curve arithmetic — shared host/device header.
//
// Implements the algebra behind the reference's BLS-over-BN254 signature
// scheme (jellyfish bls_over_bn254; reference cdn-proto/src/crypto/
// signature.rs:113-175): Fp/Fr Montgomery arithmetic (4x64 limbs), Fp2,
// G1/G2 Jacobian points, scalar multiplication.  The pairing tower lives in
// bn254_pairing.h.
//
// The same source compiles for host C++ (pushcdn_core) and for gfx950 device
// code (K1 batched verification, csrc/hip/bls_kernels.hip): all functions are
// BN_HOSTDEV, all constants come from the generated bn254_constants.h.

#pragma once
#include <stdint.h>

#include "bn254_constants.h"

#if defined(__HIPCC__) || defined(__HIP_DEVICE_COMPILE__)
// plain inline (not forceinline): a fully-inlined pairing explodes compile
// time; BN_BIGFUNC marks the large building blocks noinline outright.
#define BN_INLINE __host__ __device__ inline
#define BN_BIGFUNC __host__ __device__ __attribute__((noinline))
#define BN_NOUNROLL _Pragma("clang loop unroll(disable)")
#else
#define BN_INLINE inline
#define BN_BIGFUNC inline __attribute__((noinline))
#define BN_NOUNROLL
#endif

namespace bn254 {

using u64 = uint64_t;
using u128 = unsigned __int128;

// ---------------------------------------------------------------------------
// 4-limb helpers
// ---------------------------------------------------------------------------
struct U256 {
    u64 v[4];
};

BN_INLINE bool u256_eq(const U256& a, const U256& b) {
    return a.v[0] == b.v[0] && a.v[1] == b.v[1] && a.v[2] == b.v[2] && a.v[3] == b.v[3];
}

BN_INLINE bool u256_is_zero(const U256& a) {
    return (a.v[0] | a.v[1] | a.v[2] | a.v[3]) == 0;
}

BN_INLINE bool u256_gte(const U256& a, const U256& b) {
    for (int i = 3; i >= 0; --i) {
        if (a.v[i] > b.v[i]) return true;
        if (a.v[i] < b.v[i]) return false;
    }
    return true;
}

// a += b, returns carry
BN_INLINE u64 u256_add(U256& a, const U256& b) {
    u128 c = 0;
    for (int i = 0; i < 4; ++i) {
        c += (u128)a.v[i] + b.v[i];
        a.v[i] = (u64)c;
        c >>= 64;
    }
    return (u64)c;
}

// a -= b, returns borrow
BN_INLINE u64 u256_sub(U256& a, const U256& b) {
    u64 borrow = 0;
    for (int i = 0; i < 4; ++i) {
        u64 bi = b.v[i];
        u64 t = a.v[i] - bi;
        u64 borrow2 = (a.v[i] < bi) ? 1 : 0;
        u64 t2 = t - borrow;
        borrow2 |= (t < borrow) ? 1 : 0;
        a.v[i] = t2;
        borrow = borrow2;
    }
    return borrow;
}

BN_INLINE U256 from_limbs(const Limbs4& l) {
    U256 r;
    for (int i = 0; i < 4; ++i) r.v[i] = l.v[i];
    return r;
}

// ---------------------------------------------------------------------------
// Montgomery field element, parameterized by modulus constants.
// MOD selects Fp (base field) or Fr (scalar field).
// ---------------------------------------------------------------------------
struct FpTag {};
struct FrTag {};

template <class Tag>
BN_INLINE U256 field_mod() {
    return from_limbs(Tag{}.mod());
}

struct FpParams {
    BN_INLINE static U256 mod() { return from_limbs(bn254c::P); }
    BN_INLINE static u64 ninv() { return bn254c::PINV; }
    BN_INLINE static U256 r2() { return from_limbs(bn254c::R2_P); }
    BN_INLINE static U256 one_mont() { return from_limbs(bn254c::ONE_MONT_P); }
};

struct FrParams {
    BN_INLINE static U256 mod() { return from_limbs(bn254c::R_MOD); }
    BN_INLINE static u64 ninv() { return bn254c::RINV_FR; }
    BN_INLINE static U256 r2() { return from_limbs(bn254c::R2_FR); }
    BN_INLINE static U256 one_mont() { return from_limbs(bn254c::ONE_MONT_FR); }
};

template <class Params>
struct Fe {
    U256 n;  // Montgomery form

    BN_INLINE static Fe zero() { return Fe{{{0, 0, 0, 0}}}; }
    BN_INLINE static Fe one() { return Fe{Params::one_mont()}; }

    BN_INLINE bool is_zero() const { return u256_is_zero(n); }
    BN_INLINE bool operator==(const Fe& o) const { return u256_eq(n, o.n); }

    BN_INLINE static Fe add(const Fe& a, const Fe& b) {
        Fe r = a;
        u64 carry = u256_add(r.n, b.n);
        U256 mod = Params::mod();
        if (carry || u256_gte(r.n, mod)) u256_sub(r.n, mod);
        return r;
    }

    BN_INLINE static Fe sub(const Fe& a, const Fe& b) {
        Fe r = a;
        if (u256_sub(r.n, b.n)) u256_add(r.n, Params::mod());
        return r;
    }

    BN_INLINE static Fe neg(const Fe& a) {
        if (a.is_zero()) return a;
        Fe r{Params::mod()};
        u256_sub(r.n, a.n);
        return r;
    }

    BN_INLINE static Fe dbl(const Fe& a) { return add(a, a); }

    // CIOS Montgomery multiplication
    BN_INLINE static Fe mul(const Fe& a, const Fe& b) {
        u64 t[6] = {0, 0, 0, 0, 0, 0};
        const U256 mod = Params::mod();
        const u64 ninv = Params::ninv();
        for (int i = 0; i < 4; ++i) {
            u128 c = 0;
            for (int j = 0; j < 4; ++j) {
                c = (u128)a.n.v[j] * b.n.v[i] + t[j] + (u64)c;
                t[j] = (u64)c;
                c >>= 64;
            }
            u128 c2 = (u128)t[4] + (u64)c;
            t[4] = (u64)c2;
            t[5] = (u64)(c2 >> 64);

            u64 m = t[0] * ninv;
            c = (u128)m * mod.v[0] + t[0];
            c >>= 64;
            for (int j = 1; j < 4; ++j) {
                c = (u128)m * mod.v[j] + t[j] + (u64)c;
                t[j - 1] = (u64)c;
                c >>= 64;
            }
            c2 = (u128)t[4] + (u64)c;
            t[3] = (u64)c2;
            t[4] = t[5] + (u64)(c2 >> 64);
        }
        Fe r{{{t[0], t[1], t[2], t[3]}}};
        if (t[4] || u256_gte(r.n, mod)) u256_sub(r.n, mod);
        return r;
    }

    BN_INLINE static Fe sqr(const Fe& a) { return mul(a, a); }

    // to/from Montgomery
    BN_INLINE static Fe from_u256(const U256& standard) {
        Fe t{standard};
        Fe r2{Params::r2()};
        return mul(t, r2);
    }

    BN_INLINE U256 to_u256() const {
        // Montgomery reduce: multiply by 1
        Fe one_raw{{{1, 0, 0, 0}}};
        return mul(*this, one_raw).n;
    }

    BN_INLINE static Fe from_u64(u64 x) { return from_u256(U256{{x, 0, 0, 0}}); }

    // exponentiation by a standard-form 4-limb exponent (MSB scan)
    BN_BIGFUNC static Fe pow(const Fe& a, const U256& e) {
        Fe result = one();
        bool started = false;
        BN_NOUNROLL for (int i = 3; i >= 0; --i) {
            BN_NOUNROLL for (int b = 63; b >= 0; --b) {
                if (started) result = sqr(result);
                if ((e.v[i] >> b) & 1) {
                    if (started) result = mul(result, a);
                    else { result = a; started = true; }
                }
            }
        }
        return started ? result : one();
    }

    BN_INLINE Fe inv() const {
        U256 pm2 = Params::mod();
        // subtract 2
        U256 two{{2, 0, 0, 0}};
        u256_sub(pm2, two);
        return pow(*this, pm2);
    }
};

using Fp = Fe<FpParams>;
using Fr = Fe<FrParams>;

// ---------------------------------------------------------------------------
// Fp2 = Fp[u] / (u^2 + 1)
// ---------------------------------------------------------------------------
struct Fp2 {
    Fp c0, c1;

    BN_INLINE static Fp2 zero() { return {Fp::zero(), Fp::zero()}; }
    BN_INLINE static Fp2 one() { return {Fp::one(), Fp::zero()}; }
    BN_INLINE bool is_zero() const { return c0.is_zero() && c1.is_zero(); }
    BN_INLINE bool operator==(const Fp2& o) const { return c0 == o.c0 && c1 == o.c1; }

    BN_INLINE static Fp2 add(const Fp2& a, const Fp2& b) {
        return {Fp::add(a.c0, b.c0), Fp::add(a.c1, b.c1)};
    }
    BN_INLINE static Fp2 sub(const Fp2& a, const Fp2& b) {
        return {Fp::sub(a.c0, b.c0), Fp::sub(a.c1, b.c1)};
    }
    BN_INLINE static Fp2 neg(const Fp2& a) { return {Fp::neg(a.c0), Fp::neg(a.c1)}; }
    BN_INLINE static Fp2 dbl(const Fp2& a) { return add(a, a); }
    BN_INLINE static Fp2 conj(const Fp2& a) { return {a.c0, Fp::neg(a.c1)}; }

    BN_INLINE static Fp2 mul(const Fp2& a, const Fp2& b) {
        // Karatsuba: (a0 b0 - a1 b1) + ((a0+a1)(b0+b1) - a0b0 - a1b1) u
        Fp v0 = Fp::mul(a.c0, b.c0);
        Fp v1 = Fp::mul(a.c1, b.c1);
        Fp s = Fp::mul(Fp::add(a.c0, a.c1), Fp::add(b.c0, b.c1));
        return {Fp::sub(v0, v1), Fp::sub(Fp::sub(s, v0), v1)};
    }

    BN_INLINE static Fp2 sqr(const Fp2& a) {
        // (a0+a1)(a0-a1) + (2 a0 a1) u
        Fp p = Fp::mul(Fp::add(a.c0, a.c1), Fp::sub(a.c0, a.c1));
        Fp q = Fp::dbl(Fp::mul(a.c0, a.c1));
        return {p, q};
    }

    BN_INLINE static Fp2 mul_fp(const Fp2& a, const Fp& b) {
        return {Fp::mul(a.c0, b), Fp::mul(a.c1, b)};
    }

    // multiply by xi = 9 + u
    BN_INLINE static Fp2 mul_xi(const Fp2& a) {
        // (9 a0 - a1) + (9 a1 + a0) u
        Fp a0x8 = Fp::dbl(Fp::dbl(Fp::dbl(a.c0)));
        Fp a1x8 = Fp::dbl(Fp::dbl(Fp::dbl(a.c1)));
        Fp nine_a0 = Fp::add(a0x8, a.c0);
        Fp nine_a1 = Fp::add(a1x8, a.c1);
        return {Fp::sub(nine_a0, a.c1), Fp::add(nine_a1, a.c0)};
    }

    BN_INLINE Fp2 inv() const {
        // 1 / (c0 + c1 u) = (c0 - c1 u) / (c0^2 + c1^2)
        Fp d = Fp::add(Fp::sqr(c0), Fp::sqr(c1));
        Fp di = d.inv();
        return {Fp::mul(c0, di), Fp::neg(Fp::mul(c1, di))};
    }
};

// ---------------------------------------------------------------------------
// Curve points (Jacobian: x = X/Z^2, y = Y/Z^3)
// ---------------------------------------------------------------------------
template <class F>
struct Point {
    F X, Y, Z;

    BN_INLINE static Point infinity() { return {F::one(), F::one(), F::zero()}; }
    BN_INLINE bool is_infinity() const { return Z.is_zero(); }

    BN_INLINE static Point dbl(const Point& p) {
        if (p.is_infinity()) return p;
        // dbl-2009-l: A = X^2, B = Y^2, C = B^2, D = 2((X+B)^2 - A - C),
        // E = 3A, F = E^2; X3 = F - 2D, Y3 = E(D - X3) - 8C, Z3 = 2 Y Z
        F A = F::sqr(p.X);
        F B = F::sqr(p.Y);
        F C = F::sqr(B);
        F t = F::sqr(F::add(p.X, B));
        F D = F::dbl(F::sub(F::sub(t, A), C));
        F E = F::add(F::dbl(A), A);
        F Fq = F::sqr(E);
        F X3 = F::sub(Fq, F::dbl(D));
        F eight_c = F::dbl(F::dbl(F::dbl(C)));
        F Y3 = F::sub(F::mul(E, F::sub(D, X3)), eight_c);
        F Z3 = F::dbl(F::mul(p.Y, p.Z));
        return {X3, Y3, Z3};
    }

    BN_INLINE static Point add(const Point& p, const Point& q) {
        if (p.is_infinity()) return q;
        if (q.is_infinity()) return p;
        // add-2007-bl
        F Z1Z1 = F::sqr(p.Z);
        F Z2Z2 = F::sqr(q.Z);
        F U1 = F::mul(p.X, Z2Z2);
        F U2 = F::mul(q.X, Z1Z1);
        F S1 = F::mul(F::mul(p.Y, q.Z), Z2Z2);
        F S2 = F::mul(F::mul(q.Y, p.Z), Z1Z1);
        if (U1 == U2) {
            if (S1 == S2) return dbl(p);
            return infinity();
        }
        F H = F::sub(U2, U1);
        F I = F::sqr(F::dbl(H));
        F J = F::mul(H, I);
        F r = F::dbl(F::sub(S2, S1));
        F V = F::mul(U1, I);
        F X3 = F::sub(F::sub(F::sqr(r), J), F::dbl(V));
        F Y3 = F::sub(F::mul(r, F::sub(V, X3)), F::dbl(F::mul(S1, J)));
        F Z3 = F::mul(F::mul(F::dbl(F::mul(p.Z, q.Z)), H), F::one());
        // Z3 = ((Z1+Z2)^2 - Z1Z1 - Z2Z2) * H  — use the simpler 2*Z1*Z2*H
        return {X3, Y3, Z3};
    }

    BN_INLINE static Point neg(const Point& p) { return {p.X, F::neg(p.Y), p.Z}; }

    // scalar multiplication by a standard-form scalar (not Montgomery)
    BN_BIGFUNC static Point scalar_mul(const Point& p, const U256& k) {
        Point result = infinity();
        bool started = false;
        BN_NOUNROLL for (int i = 3; i >= 0; --i) {
            BN_NOUNROLL for (int b = 63; b >= 0; --b) {
                if (started) result = dbl(result);
                if ((k.v[i] >> b) & 1) {
                    if (started) result = add(result, p);
                    else { result = p; started = true; }
                }
            }
        }
        return started ? result : infinity();
    }

    // normalize to affine (x, y); infinity -> (0, 0)
    BN_INLINE void to_affine(F& x, F& y) const {
        if (is_infinity()) { x = F::zero(); y = F::zero(); return; }
        F zi = Z.inv();
        F zi2 = F::sqr(zi);
        x = F::mul(X, zi2);
        y = F::mul(Y, F::mul(zi2, zi));
    }
};

using G1 = Point<Fp>;
using G2 = Point<Fp2>;

BN_INLINE G1 g1_generator() {
    return {Fp::from_u256(from_limbs(bn254c::G1_X)),
            Fp::from_u256(from_limbs(bn254c::G1_Y)), Fp::one()};
}

BN_INLINE G2 g2_generator() {
    Fp2 x{Fp::from_u256(from_limbs(bn254c::G2_X_C0)), Fp::from_u256(from_limbs(bn254c::G2_X_C1))};
    Fp2 y{Fp::from_u256(from_limbs(bn254c::G2_Y_C0)), Fp::from_u256(from_limbs(bn254c::G2_Y_C1))};
    return {x, y, Fp2::one()};
}

// y^2 = x^3 + 3 membership for affine G1
BN_INLINE bool g1_on_curve(const Fp& x, const Fp& y) {
    Fp b3 = Fp::from_u64(3);
    Fp lhs = Fp::sqr(y);
    Fp rhs = Fp::add(Fp::mul(Fp::sqr(x), x), b3);
    return lhs == rhs;
}

BN_INLINE bool g2_on_curve(const Fp2& x, const Fp2& y) {
    Fp2 b2{Fp::from_u256(from_limbs(bn254c::B2_C0)), Fp::from_u256(from_limbs(bn254c::B2_C1))};
    Fp2 lhs = Fp2::sqr(y);
    Fp2 rhs = Fp2::add(Fp2::mul(Fp2::sqr(x), x), b2);
    return lhs == rhs;
}

}  // namespace bn254
