"""kubebrain_amd — MI355X-native KubeBrain MVCC hot path (product package).

Loads the in-tree HIP library (libkbslab.so, gfx950). There is NO CPU
fallback: importing works everywhere (the .so links the HIP runtime, which
loads without a GPU), but opening a store requires a GPU and fails loudly
otherwise (KB_ENOGPU).
"""
from __future__ import annotations

import ctypes
import os

_HERE = os.path.dirname(os.path.abspath(__file__))
LIB_PATH = os.path.join(_HERE, "libkbslab.so")


def build(force: bool = False) -> str:
    """Compile the HIP extension in-tree (hipcc --offload-arch=gfx950)."""
    import subprocess

    src_dir = os.path.join(_HERE, "csrc")
    srcs = [os.path.join(src_dir, f) for f in ("slab.hip", "store.cc", "cabi.cc",
                                               "comm.cc", "slab_dev.h", "store.h")]
    srcs.append(os.path.join(_HERE, "..", "include", "kb_slab.h"))
    stale = force or not os.path.exists(LIB_PATH) or any(
        os.path.getmtime(s) > os.path.getmtime(LIB_PATH) for s in srcs)
    if stale:
        subprocess.check_call(["make", "-C", src_dir])
    return LIB_PATH


def load_library() -> ctypes.CDLL:
    if not os.path.exists(LIB_PATH):
        build()
    return ctypes.CDLL(LIB_PATH)


def open_store(store_prefix: bytes = b"/registry", **kw):
    """Open a GPU-backed store (raises RuntimeError without a GPU)."""
    from .client import Store

    if not os.path.exists(LIB_PATH):
        build()
    try:
        return Store(LIB_PATH, "kb_", store_prefix=store_prefix, **kw)
    except RuntimeError as e:
        lib = ctypes.CDLL(LIB_PATH)
        buf = ctypes.create_string_buffer(512)
        code = lib.kb_last_error(buf, 512)
        raise RuntimeError(
            f"kubebrain_amd: kb_new failed (status {code}): "
            f"{buf.value.decode(errors='replace')}") from e
