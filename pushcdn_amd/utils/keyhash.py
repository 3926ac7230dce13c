"""64-bit FNV-1a — the routing hash for user public keys.

Must match the device implementation in csrc/hip/dataplane.hip bit-for-bit
(golden-tested in tests/test_gpu_kernels.py).

``seed`` XORs into the offset basis.  Brokers derive it from the cluster
private key (``derive_routing_seed``) so the routing hash is keyed: FNV-1a
alone is byte-invertible and an attacker could otherwise grind out a pubkey
whose hash collides with a victim's and siphon their Direct messages.  The
seed is shared cluster-wide (mesh ownership digests must agree across
brokers) but secret from users.  seed=0 == classic FNV-1a.
"""

from __future__ import annotations

import hashlib

_FNV_OFFSET = 0xCBF29CE484222325
_FNV_PRIME = 0x100000001B3
_MASK = (1 << 64) - 1


def fnv1a64(data: bytes, seed: int = 0) -> int:
    h = _FNV_OFFSET ^ (seed & _MASK)
    for b in data:
        h ^= b
        h = (h * _FNV_PRIME) & _MASK
    return h


def derive_routing_seed(cluster_private_key: bytes) -> int:
    """Cluster-shared secret seed for the routing hash, derived from the
    broker keypair every broker in the cluster already shares
    (reference broker.rs:285-288: peers must present the same keypair)."""
    d = hashlib.sha256(b"pushcdn-routing-hash-seed" + cluster_private_key).digest()
    return int.from_bytes(d[:8], "little")
