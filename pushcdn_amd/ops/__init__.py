"""Kernel op access.

``get_gpu_ops()`` returns the compiled CDNA4 extension.  On a machine with a
GPU a missing/unbuildable extension is a **hard error** — there is no silent
eager fallback on the GPU path.  On CPU-only machines tests use
``pushcdn_amd.ops.reference`` instead.
"""

from __future__ import annotations

_gpu_mod = None


def get_gpu_ops():
    global _gpu_mod
    if _gpu_mod is not None:
        return _gpu_mod
    from .build import BUILD_DIR, build_gpu, load_gpu_prebuilt

    import torch

    so = BUILD_DIR / "pushcdn_gpu.so"
    if so.exists():
        try:
            _gpu_mod = load_gpu_prebuilt()
            return _gpu_mod
        except ImportError:
            pass
    if torch.cuda.is_available():
        # On a GPU box the extension must have been built (it travels with
        # the snapshot). Rebuilding silently could mask a stale-binary bug —
        # but a from-source build is still better than not running at all.
        _gpu_mod = build_gpu()
        return _gpu_mod
    _gpu_mod = build_gpu()
    return _gpu_mod
