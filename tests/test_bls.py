"""BLS-over-BN254 host tests: field arithmetic cross-checked against Python
bignums, subgroup/bilinearity sanity, sign/verify + namespace separation
(the reference's signature tests, crypto/signature.rs:177-219)."""

import hashlib
import random

import pytest

from pushcdn_amd.crypto import bls
from pushcdn_amd.ops.build import build_core

P = 21888242871839275222246405745257275088696311157297823662689037894645226208583
R = 21888242871839275222246405745257275088548364400416034343698204186575808495617


@pytest.fixture(scope="module")
def core():
    return build_core()


def test_sha256_matches_hashlib(core):
    for data in [b"", b"abc", b"x" * 55, b"y" * 56, b"z" * 64, b"w" * 1000]:
        assert core._sha256(data) == hashlib.sha256(data).digest()


def test_fp_mul_against_python(core):
    rng = random.Random(42)
    for _ in range(50):
        a, b = rng.randrange(P), rng.randrange(P)
        got = int.from_bytes(
            core._fp_mul(a.to_bytes(32, "little"), b.to_bytes(32, "little")), "little"
        )
        assert got == a * b % P


def test_fp_inv_against_python(core):
    rng = random.Random(43)
    for _ in range(10):
        a = rng.randrange(1, P)
        got = int.from_bytes(core._fp_inv(a.to_bytes(32, "little")), "little")
        assert got == pow(a, P - 2, P)


def test_g1_scalar_mul_small(core):
    # 2G on alt_bn128 is a well-known value; verify on-curve + doubling law
    # via python: compute 2G with affine formulas.
    g = (1, 2)
    lam = (3 * g[0] * g[0]) * pow(2 * g[1], P - 2, P) % P
    x2 = (lam * lam - 2 * g[0]) % P
    y2 = (lam * (g[0] - x2) - g[1]) % P
    raw = core._g1_mul(2)
    gx = int.from_bytes(raw[:32], "little")
    gy = int.from_bytes(raw[32:], "little")
    assert (gx, gy) == (x2, y2)


def test_subgroup_orders(core):
    assert core._subgroup_ok()


def test_pairing_bilinearity(core):
    assert core._pairing_bilinear(7, 13)
    assert core._pairing_bilinear(1, 1)
    assert core._pairing_bilinear(123456789, 987654321)


def test_hash_to_g1_on_curve_and_deterministic(core):
    h1 = core._hash_to_g1("ns", b"message")
    h2 = core._hash_to_g1("ns", b"message")
    assert h1 == h2
    h3 = core._hash_to_g1("ns", b"other")
    assert h1 != h3
    x = int.from_bytes(h1[:32], "little")
    y = int.from_bytes(h1[32:], "little")
    assert y * y % P == (x**3 + 3) % P


def test_sign_verify_roundtrip():
    kp = bls.KeyPair.from_seed(0)
    sig = bls.sign(kp.private_key, bls.USER_MARSHAL_NAMESPACE, b"timestamp-bytes")
    assert len(sig) == 64 and len(kp.public_key) == 128
    assert bls.verify(kp.public_key, bls.USER_MARSHAL_NAMESPACE, b"timestamp-bytes", sig)


def test_namespace_separation():
    # reference signature.rs:177-219: same message, different namespace fails
    kp = bls.KeyPair.from_seed(7)
    sig = bls.sign(kp.private_key, bls.USER_MARSHAL_NAMESPACE, b"m")
    assert not bls.verify(kp.public_key, bls.BROKER_BROKER_NAMESPACE, b"m", sig)


def test_wrong_key_and_message_fail():
    kp1 = bls.KeyPair.from_seed(1)
    kp2 = bls.KeyPair.from_seed(2)
    sig = bls.sign(kp1.private_key, bls.USER_MARSHAL_NAMESPACE, b"m")
    assert not bls.verify(kp2.public_key, bls.USER_MARSHAL_NAMESPACE, b"m", sig)
    assert not bls.verify(kp1.public_key, bls.USER_MARSHAL_NAMESPACE, b"m2", sig)


def test_deterministic_keygen():
    assert bls.KeyPair.from_seed(5) == bls.KeyPair.from_seed(5)
    assert bls.KeyPair.from_seed(5) != bls.KeyPair.from_seed(6)


def test_malformed_inputs_rejected():
    kp = bls.KeyPair.from_seed(3)
    sig = bls.sign(kp.private_key, bls.USER_MARSHAL_NAMESPACE, b"m")
    assert not bls.verify(kp.public_key, bls.USER_MARSHAL_NAMESPACE, b"m", b"\x00" * 64)
    assert not bls.verify(kp.public_key, bls.USER_MARSHAL_NAMESPACE, b"m", b"\xff" * 64)
    assert not bls.verify(b"\x00" * 128, bls.USER_MARSHAL_NAMESPACE, b"m", sig)
    assert not bls.verify(b"junk", bls.USER_MARSHAL_NAMESPACE, b"m", sig)
    # bit-flipped signature
    bad = bytearray(sig)
    bad[5] ^= 1
    assert not bls.verify(kp.public_key, bls.USER_MARSHAL_NAMESPACE, b"m", bytes(bad))


def test_timestamp_helpers():
    kp = bls.KeyPair.from_seed(9)
    ts = 1_726_000_000
    sig = bls.sign_timestamp(kp.private_key, bls.USER_MARSHAL_NAMESPACE, ts)
    assert bls.verify_timestamp(kp.public_key, bls.USER_MARSHAL_NAMESPACE, ts, sig)
    assert not bls.verify_timestamp(kp.public_key, bls.USER_MARSHAL_NAMESPACE, ts + 1, sig)


def test_hard_exponentiation_chain_matches_generic(core):
    """The Fuentes-Castaneda x-chain must equal generic^c (c = the F-C
    multiple, verified symbolically in scripts/gen_bn254_constants.py lineage)."""
    for a, b in [(1, 1), (3, 5), (123456789, 987654321)]:
        assert core._hard_exp_chain_ok(a, b)


def test_fp12_fast_paths_consistent(core):
    """Sparse line multiplication (mul_by_034) and Granger-Scott cyclotomic
    squaring must agree with the dense/general implementations."""
    for a, b in [(3, 5), (7, 11), (123456, 654321)]:
        assert core._fp12_fastpath_ok(a, b)


def test_gpu_batch_verifier_host_fallback():
    """Below min_batch the micro-batching verifier answers from the host
    path (a K1 launch costs ~100 ms; tiny batches shouldn't pay it) — and
    bad signatures still fail."""
    import asyncio

    from pushcdn_amd.crypto import bls
    from pushcdn_amd.crypto.gpu_verify import GpuBatchVerifier

    async def go():
        v = GpuBatchVerifier(max_wait_s=0.01, min_batch=8)
        kp = bls.KeyPair.from_seed(5)
        sig = bls.sign(kp.private_key, "ns", b"msg")
        ok, bad = await asyncio.gather(
            v.verify(kp.public_key, "ns", b"msg", sig),
            v.verify(kp.public_key, "ns", b"other", sig),
        )
        assert ok is True and bad is False
        await v.close()

    asyncio.run(asyncio.wait_for(go(), 30))
