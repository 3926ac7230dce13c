"""BLS-over-BN254 signature scheme (host binding).

Capability mirror of the reference's scheme trait + jellyfish impl
(``cdn-proto/src/crypto/signature.rs:36-175``): namespace-prefixed signing,
deterministic keygen from a u64 seed (the reference broker's ``--key-seed``),
64 B G1 signatures, 128 B G2 verification keys.

The implementation is C++ (csrc/bls/, csrc/common/bn254*.h), shared
source-for-source with the K1 batched GPU verification kernel.
"""

from __future__ import annotations

from dataclasses import dataclass

# Namespaces (reference signature.rs:19-32)
USER_MARSHAL_NAMESPACE = "espresso-cdn-user-marshal-auth"
BROKER_BROKER_NAMESPACE = "espresso-cdn-broker-broker-auth"

_core = None


def _mod():
    global _core
    if _core is None:
        from ..ops.build import build_core

        _core = build_core()
    return _core


@dataclass(frozen=True)
class KeyPair:
    public_key: bytes   # 128 B serialized G2 verkey
    private_key: bytes  # 32 B Fr scalar

    @classmethod
    def from_seed(cls, seed: int) -> "KeyPair":
        sk, vk = _mod().keygen(seed & 0xFFFFFFFFFFFFFFFF)
        return cls(public_key=vk, private_key=sk)


def sign(private_key: bytes, namespace: str, message: bytes) -> bytes:
    return _mod().sign(private_key, namespace, message)


def verify(public_key: bytes, namespace: str, message: bytes, signature: bytes) -> bool:
    try:
        return _mod().verify(public_key, namespace, message, signature)
    except Exception:
        return False


def sign_timestamp(private_key: bytes, namespace: str, timestamp: int) -> bytes:
    """Sign a unix-seconds timestamp (LE u64 bytes), the auth-flow payload
    (reference auth/user.rs:42-58)."""
    return sign(private_key, namespace, timestamp.to_bytes(8, "little"))


def verify_timestamp(public_key: bytes, namespace: str, timestamp: int, signature: bytes) -> bool:
    return verify(public_key, namespace, timestamp.to_bytes(8, "little"), signature)
