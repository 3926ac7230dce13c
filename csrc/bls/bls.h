// BLS signatures over BN254 — host/device shared implementation core.
//
// Mirrors the structure of the reference's scheme (jellyfish bls_over_bn254,
// used at cdn-proto/src/crypto/signature.rs:113-175): signing key = Fr
// scalar, verification key = G2 point, signature = G1 point,
//   sign:   sigma = sk * H(namespace || message)      (H: hash-to-G1)
//   verify: e(H(m), pk) == e(sigma, g2)
// implemented as the product check e(H(m), pk) * e(-sigma, g2) == 1 with one
// shared final exponentiation.
//
// Namespace domain separation is byte-prefixing, exactly as the reference
// does (signature.rs:126-129: namespaced_message = namespace || message).
//
// Serialization (documented wire format of THIS framework):
//   SignKey   32 B  Fr scalar, little-endian
//   VerKey   128 B  G2 affine: x.c0 || x.c1 || y.c0 || y.c1, 32 B LE each
//   Signature 64 B  G1 affine: x || y, 32 B LE each;  all-zero = infinity
// (ark-serialize-uncompressed uses the same coordinate order/endianness for
// these curves; flag bits are unused here because neither valid signatures
// nor valid verkeys are ever the point at infinity.)
//
// Hash-to-G1: try-and-increment with SHA-256 —
//   for ctr = 0..254: d = SHA256(msg || ctr_u8); x = d mod p (bytes LE);
//   if x^3 + 3 is a QR: y = sqrt, pick parity by d[0] & 1 -> done.
// Deterministic, constant across host and device (K1 golden-tests this).

#pragma once
#include "../common/bn254.h"
#include "../common/bn254_pairing.h"
#include "../common/sha256.h"

namespace bls {

using namespace bn254;

// interpret 32 LE bytes as U256
BN_INLINE U256 u256_from_le(const uint8_t* b) {
    U256 r;
    for (int i = 0; i < 4; ++i) {
        u64 w = 0;
        for (int j = 7; j >= 0; --j) w = (w << 8) | b[8 * i + j];
        r.v[i] = w;
    }
    return r;
}

BN_INLINE void u256_to_le(const U256& a, uint8_t* out) {
    for (int i = 0; i < 4; ++i)
        for (int j = 0; j < 8; ++j) out[8 * i + j] = (uint8_t)(a.v[i] >> (8 * j));
}

// reduce a 256-bit value mod `mod` (value < 2^256 < 6*mod for both p and r)
BN_INLINE U256 u256_mod(U256 a, const U256& mod) {
    while (u256_gte(a, mod)) u256_sub(a, mod);
    return a;
}

// y = sqrt(a) if it exists (p = 3 mod 4): y = a^((p+1)/4); verify y^2 == a.
BN_INLINE bool fp_sqrt(const Fp& a, Fp& out) {
    Fp y = Fp::pow(a, from_limbs(bn254c::SQRT_EXP));
    if (Fp::sqr(y) == a) { out = y; return true; }
    return false;
}

// hash (already namespaced) message bytes to a G1 point.
// msg buffer must have one spare byte at msg[len] for the counter (the
// callers build "namespace || message || ctr" in a scratch buffer).
BN_BIGFUNC bool hash_to_g1_with_scratch(uint8_t* scratch, uint32_t len, Fp& outx, Fp& outy) {
    BN_NOUNROLL for (uint32_t ctr = 0; ctr < 255; ++ctr) {
        scratch[len] = (uint8_t)ctr;
        uint8_t d[32];
        sha256(scratch, len + 1, d);
        U256 xv = u256_mod(u256_from_le(d), from_limbs(bn254c::P));
        Fp x = Fp::from_u256(xv);
        Fp rhs = Fp::add(Fp::mul(Fp::sqr(x), x), Fp::from_u64(3));
        Fp y;
        if (fp_sqrt(rhs, y)) {
            // canonical sign: take y if (y mod 2) == (d[0] & 1) else -y,
            // parity taken on the standard-form representation
            U256 ys = y.to_u256();
            if ((ys.v[0] & 1) != (u64)(d[0] & 1)) y = Fp::neg(y);
            outx = x;
            outy = y;
            return true;
        }
    }
    return false;  // cryptographically unreachable
}

struct VerKey {
    Fp2 x, y;
};

BN_INLINE void verkey_serialize(const VerKey& vk, uint8_t out[128]) {
    u256_to_le(vk.x.c0.to_u256(), out);
    u256_to_le(vk.x.c1.to_u256(), out + 32);
    u256_to_le(vk.y.c0.to_u256(), out + 64);
    u256_to_le(vk.y.c1.to_u256(), out + 96);
}

// Prime-order subgroup membership for a non-infinity on-curve G2 point.
// BN254 E'(Fp2) has a large cofactor, so on-curve alone is NOT enough (the
// reference's ark deserialize_uncompressed enforces subgroup membership).
// Fast endomorphism check: P is in the r-order subgroup iff
//     psi(P) == [t-1]P          with t-1 = 6x^2
// where psi is the untwist-Frobenius endomorphism (g2_frobenius).
// Soundness: psi^2 - [t]psi + [p] = 0 holds on all of E'(Fp2); substituting
// psi(P) = [t-1]P yields [p+1-t]P = O, and p+1-t = r exactly for BN curves
// (G1 cofactor is 1).  ~127-bit scalar mul instead of the naive 254-bit [r]P.
BN_BIGFUNC bool g2_in_subgroup(const Fp2& x, const Fp2& y) {
    G2 s = G2::scalar_mul({x, y, Fp2::one()}, from_limbs(bn254c::SIX_X_SQ));
    if (s.is_infinity()) return false;  // psi(P) of a finite point is finite
    G2Affine psi = g2_frobenius({x, y});
    Fp2 sx, sy;
    s.to_affine(sx, sy);
    return sx == psi.x && sy == psi.y;
}

BN_BIGFUNC bool verkey_deserialize(const uint8_t in[128], VerKey& vk) {
    U256 p = from_limbs(bn254c::P);
    U256 xc0 = u256_from_le(in), xc1 = u256_from_le(in + 32);
    U256 yc0 = u256_from_le(in + 64), yc1 = u256_from_le(in + 96);
    if (u256_gte(xc0, p) || u256_gte(xc1, p) || u256_gte(yc0, p) || u256_gte(yc1, p))
        return false;
    vk.x = {Fp::from_u256(xc0), Fp::from_u256(xc1)};
    vk.y = {Fp::from_u256(yc0), Fp::from_u256(yc1)};
    if (vk.x.is_zero() && vk.y.is_zero()) return false;  // infinity not a valid key
    if (!g2_on_curve(vk.x, vk.y)) return false;
    return g2_in_subgroup(vk.x, vk.y);
}

BN_INLINE void sig_serialize(const Fp& x, const Fp& y, uint8_t out[64]) {
    u256_to_le(x.to_u256(), out);
    u256_to_le(y.to_u256(), out + 32);
}

BN_INLINE bool sig_deserialize(const uint8_t in[64], Fp& x, Fp& y) {
    U256 p = from_limbs(bn254c::P);
    U256 xv = u256_from_le(in), yv = u256_from_le(in + 32);
    if (u256_gte(xv, p) || u256_gte(yv, p)) return false;
    x = Fp::from_u256(xv);
    y = Fp::from_u256(yv);
    if (x.is_zero() && y.is_zero()) return false;  // infinity not a valid sig
    return g1_on_curve(x, y);
}

// Core verification given parsed inputs. msg scratch = namespace||message
// with a spare byte (see hash_to_g1_with_scratch).
BN_BIGFUNC bool verify_core(const VerKey& vk, uint8_t* scratch, uint32_t msg_len,
                           const Fp& sig_x, const Fp& sig_y) {
    Fp hx, hy;
    if (!hash_to_g1_with_scratch(scratch, msg_len, hx, hy)) return false;
    G2Affine pk{vk.x, vk.y};
    G2Affine g2;
    {
        G2 gen = g2_generator();
        g2 = {gen.X, gen.Y};
    }
    // e(H, pk) * e(-sig, g2) == 1
    return pairing_product_is_one(hx, hy, pk, sig_x, Fp::neg(sig_y), g2);
}

}  // namespace bls
