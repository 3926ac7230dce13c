"""In-tree build of the HIP/C++ extensions.

Two modules:
  - ``pushcdn_gpu``  — CDNA4 data-plane kernels (csrc/hip/*.hip) + torch bindings,
                       compiled for gfx950 only.
  - ``pushcdn_core`` — host C++ (wire serde, BLS-over-BN254, CRDT) via pybind11.

Everything builds into ``<repo>/build/`` (in-tree, so the .so files travel to
the GPU box with the snapshot).  ``__graft_entry__.build()`` calls
``build_all()``; at import time we only *load* what exists, and on a GPU
machine a missing GPU extension is a hard error (no silent eager fallback).
"""

from __future__ import annotations

import os
import sys
from pathlib import Path

REPO_ROOT = Path(__file__).resolve().parent.parent.parent
CSRC = REPO_ROOT / "csrc"
BUILD_DIR = REPO_ROOT / "build"

GPU_SOURCES = [
    CSRC / "gpu_bindings.cpp",
    CSRC / "hip" / "dataplane.hip",
    CSRC / "hip" / "bls_kernels.hip",
]

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")


def build_gpu(verbose: bool = False):
    """Compile (if needed) and load the GPU extension. Returns the module."""
    from torch.utils.cpp_extension import load

    BUILD_DIR.mkdir(exist_ok=True)
    sources = [str(s) for s in GPU_SOURCES if s.exists()]
    # ninja's rules for hipcc lack header depfiles: if any shared header is
    # newer than a built object, touch the sources so the rebuild triggers.
    headers = list((CSRC / "common").glob("*.h")) + list((CSRC / "bls").glob("*.h")) \
        + list((CSRC / "wire").glob("*.h"))
    if headers:
        newest_h = max(h.stat().st_mtime for h in headers)
        for src in GPU_SOURCES:
            if src.exists() and src.stat().st_mtime < newest_h:
                os.utime(src)
    mod = load(
        name="pushcdn_gpu",
        sources=sources,
        build_directory=str(BUILD_DIR),
        extra_cflags=["-O3", "-std=c++17"],
        extra_cuda_cflags=["-O3", "-std=c++17"],
        verbose=verbose,
    )
    return mod


def load_gpu_prebuilt():
    """Load the already-built GPU extension without invoking the builder.

    Used on GPU boxes where the .so traveled with the snapshot; avoids any
    silent rebuild (and fails loudly if the extension is missing).
    """
    so = BUILD_DIR / "pushcdn_gpu.so"
    if not so.exists():
        raise ImportError(
            f"pushcdn_gpu extension not built (expected {so}); "
            "run __graft_entry__.build() first"
        )
    import importlib.util

    spec = importlib.util.spec_from_file_location("pushcdn_gpu", so)
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    sys.modules["pushcdn_gpu"] = mod
    return mod


CORE_SOURCES = [CSRC / "core_bindings.cpp"]


def build_core(verbose: bool = False):
    """Compile (if needed) and load the host C++ core (pybind11, no torch)."""
    import importlib.util
    import pybind11
    import subprocess
    import sysconfig

    BUILD_DIR.mkdir(exist_ok=True)
    so = BUILD_DIR / "pushcdn_core.so"
    srcs = [str(s) for s in CORE_SOURCES]
    deps = (list((CSRC / "common").glob("*.h")) + list((CSRC / "bls").glob("*.h"))
            + list((CSRC / "wire").glob("*.h")) + list((CSRC / "state").glob("*.h"))
            + list((CSRC / "net").glob("*.h")) + CORE_SOURCES)
    newest_dep = max(p.stat().st_mtime for p in deps)
    if not so.exists() or so.stat().st_mtime < newest_dep:
        cmd = [
            "g++", "-O3", "-std=c++17", "-shared", "-fPIC",
            f"-I{pybind11.get_include()}",
            f"-I{sysconfig.get_paths()['include']}",
            f"-I{CSRC}",
            *srcs,
            "-o", str(so),
        ]
        if verbose:
            print(" ".join(cmd))
        subprocess.run(cmd, check=True)
    spec = importlib.util.spec_from_file_location("pushcdn_core", so)
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    sys.modules["pushcdn_core"] = mod
    return mod
