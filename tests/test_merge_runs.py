"""CPU tests for the cross-shard exchange's k-way merge + global limit cut
(kb_test_merge_runs -> comm.cc merge_runs), the host half of
kb_range_global. The device/RCCL half is covered by the gpu-marked
test_range_global parity test and the gloo shard test pins the same
algorithm end-to-end (tests/test_gloo_shard.py). Semantics mirrored:
receiver fork/merge scanner.go:269-300 + the limit+1 trick range.go:154-171.

No GPU needed: the symbol is pure host code in libkbslab.so.
"""
import ctypes as C
import os
import struct
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _lib():
    import kbclient  # builds/loads via the product client helpers

    from kubebrain_amd import build
    return C.CDLL(build())


def pack_run(kvs):
    out = []
    for key, val, rev in kvs:
        out.append(struct.pack("<QII", rev, len(key), len(val)))
        out.append(key)
        out.append(val)
    return b"".join(out)


def parse_list_wire(buf):
    (n,) = struct.unpack_from("<I", buf, 0)
    off = 4
    res = []
    for _ in range(n):
        (rev,) = struct.unpack_from("<Q", buf, off); off += 8
        (klen,) = struct.unpack_from("<I", buf, off); off += 4
        key = buf[off:off + klen]; off += klen
        (vlen,) = struct.unpack_from("<I", buf, off); off += 4
        val = buf[off:off + vlen]; off += vlen
        res.append((key, val, rev))
    return res


def merge(lib, runs, limit, cap=1 << 20):
    blobs = [pack_run(r) for r in runs]
    cat = b"".join(blobs)
    lens = (C.c_ulonglong * len(runs))(*[len(b) for b in blobs])
    out = C.create_string_buffer(cap)
    out_len = C.c_size_t()
    more = C.c_int()
    rc = lib.kb_test_merge_runs(cat, lens, C.c_int(len(runs)),
                                C.c_longlong(limit), out, C.c_size_t(cap),
                                C.byref(out_len), C.byref(more))
    return rc, parse_list_wire(out.raw[:out_len.value]), bool(more.value)


def test_merge_basics():
    lib = _lib()
    r0 = [(b"/a/1", b"v1", 10), (b"/a/4", b"v4", 11)]
    r1 = [(b"/a/2", b"v2", 12), (b"/a/5", b"v5", 13)]
    r2 = [(b"/a/3", b"v3", 14)]
    rc, kvs, more = merge(lib, [r0, r1, r2], 0)
    assert rc == 0 and not more
    assert [k for k, _, _ in kvs] == [b"/a/%d" % i for i in (1, 2, 3, 4, 5)]
    assert [r for _, _, r in kvs] == [10, 12, 14, 11, 13]


def test_merge_limit_cut_and_more():
    lib = _lib()
    # each shard contributes its first limit+1 winners (here limit=3)
    r0 = [(b"/a/%02d" % i, b"x", 100 + i) for i in (0, 3, 6, 9)]
    r1 = [(b"/a/%02d" % i, b"y", 200 + i) for i in (1, 4, 7, 10)]
    rc, kvs, more = merge(lib, [r0, r1], 3)
    assert rc == 0 and more
    assert [k for k, _, _ in kvs] == [b"/a/00", b"/a/01", b"/a/03"]
    # exactly limit winners in total -> no More
    rc, kvs, more = merge(lib, [r0[:2], r1[:1]], 3)
    assert rc == 0 and not more and len(kvs) == 3
    # fewer than limit -> no More
    rc, kvs, more = merge(lib, [r0[:1], r1[:1]], 3)
    assert rc == 0 and not more and len(kvs) == 2


def test_merge_empty_and_prefix_keys():
    lib = _lib()
    # empty shards + keys where one is a strict prefix of another (the
    # shorter key sorts first, like the slab's zero-padded column)
    r0 = [(b"/registry/pods/ns-1/a", b"v", 5)]
    r1 = []
    r2 = [(b"/registry/pods/ns-1/ab", b"w", 6)]
    rc, kvs, more = merge(lib, [r0, r1, r2], 0)
    assert rc == 0 and not more
    assert [k for k, _, _ in kvs] == [b"/registry/pods/ns-1/a",
                                      b"/registry/pods/ns-1/ab"]


def test_merge_overflow():
    lib = _lib()
    r0 = [(b"/k/%04d" % i, b"v" * 100, i + 1) for i in range(50)]
    rc, kvs, more = merge(lib, [r0], 0, cap=256)
    assert rc == 100  # KB_ENOBUF
