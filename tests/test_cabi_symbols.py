"""CPU-side checks of the product library: it builds for gfx950, exports every
symbol include/kb_slab.h declares, and FAILS LOUDLY without a GPU (no CPU
fallback — DESIGN.md §1)."""
import ctypes
import os
import re

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _lib():
    import sys
    sys.path.insert(0, REPO)
    import kubebrain_amd
    return ctypes.CDLL(kubebrain_amd.build())


def header_symbols():
    hdr = open(os.path.join(REPO, "include", "kb_slab.h")).read()
    hdr = re.sub(r"/\*.*?\*/", "", hdr, flags=re.S)
    return sorted(set(re.findall(r"\b(kb_[a-z_0-9]+)\s*\(", hdr)) - {"kb_status"})


def test_library_exports_all_header_symbols():
    lib = _lib()
    syms = header_symbols()
    assert len(syms) >= 20
    missing = [s for s in syms if not hasattr(lib, s)]
    assert not missing, f"missing exports: {missing}"


def test_open_fails_loudly_without_gpu():
    import torch
    if torch.cuda.is_available():
        pytest.skip("GPU present; the loud-failure path is for CPU boxes")
    lib = _lib()
    lib.kb_new.restype = ctypes.c_void_p
    lib.kb_new.argtypes = [ctypes.c_char_p, ctypes.c_int, ctypes.c_longlong,
                           ctypes.c_int]
    h = lib.kb_new(b"/registry", 0, 0, 1)
    assert not h, "kb_new must fail without a GPU (no CPU fallback)"
    buf = ctypes.create_string_buffer(512)
    code = lib.kb_last_error(buf, 512)
    assert code == 14  # KB_ENOGPU
    assert b"no HIP device" in buf.value


def test_gfx950_code_object():
    """The shared library must carry a gfx950 code object (native path)."""
    import subprocess
    import kubebrain_amd
    path = kubebrain_amd.build()
    out = subprocess.run(["strings", "-a", path], capture_output=True,
                         text=True).stdout
    assert "gfx950" in out
