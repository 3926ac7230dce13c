"""64-bit FNV-1a — the routing hash for user public keys.

Must match the device implementation in csrc/hip/dataplane.hip bit-for-bit
(golden-tested in tests/test_gpu_kernels.py).
"""

from __future__ import annotations

_FNV_OFFSET = 0xCBF29CE484222325
_FNV_PRIME = 0x100000001B3
_MASK = (1 << 64) - 1


def fnv1a64(data: bytes) -> int:
    h = _FNV_OFFSET
    for b in data:
        h ^= b
        h = (h * _FNV_PRIME) & _MASK
    return h
