"""Larger-scale GPU parity: bulk-loaded keyspace, spot-checked against the
oracle with full-dump diffs after compaction (size-independent checks plus
byte-exact samples)."""
import ctypes
import os
import sys

import numpy as np
import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import parity

pytestmark = pytest.mark.gpu

parity.small_env()

VAL = 128
NNS = 100
PER = 500  # 50k keys


def bulk(store, keys, vals_blob):
    n = len(keys)
    klens = np.array([len(k) for k in keys], dtype=np.uint32).tobytes()
    vlens = np.full(n, VAL, dtype=np.uint32).tobytes()
    f = store._f("bulk_create")
    rc = f(ctypes.c_void_p(store.h), b"".join(keys), klens, vals_blob, vlens,
           ctypes.c_size_t(n))
    assert rc == 0


def test_bulk_scale_parity():
    d = parity.Dual()
    try:
        rng = np.random.default_rng(0x6B62)
        namespaces = [b"/registry/pods/ns-%04d" % i for i in range(NNS)]
        keys = [ns + b"/pod-%06d" % j for ns in namespaces for j in range(PER)]
        vals = rng.integers(0, 256, size=len(keys) * VAL, dtype=np.uint8).tobytes()
        bulk(d.o, keys, vals)
        bulk(d.p, keys, vals)
        assert d.o.current_rev() == d.p.current_rev()
        # extra revisions on a zipf subset
        zs = (rng.zipf(1.1, size=5000) - 1) % len(keys)
        for z in zs[:2000]:
            d.update(keys[int(z)], b"u" * VAL, 0)  # CAS fail path mostly
        base = 1000
        for z in zs[2000:]:
            k = keys[int(z)]
            rc, hr, kv = d.get(k, 0)
            d.update(k, b"w" * VAL, kv.revision if kv else 0)
        # tombstones
        for i in range(0, len(keys), 97):
            d.delete(keys[i], 0)
        # spot-check ranges at assorted revisions/limits
        cur = d.p.current_rev()
        qrng = np.random.default_rng(7)
        for _ in range(40):
            ns = namespaces[int(qrng.integers(NNS))]
            rev = int(qrng.choice([0, cur - 1, cur - 50, cur - 500, base + 100]))
            lim = int(qrng.choice([0, 1, 7, 100, 500, 600]))
            d.list(ns + b"/", ns + b"0", max(rev, 0), lim)
        # cross-namespace range with limit (tests ordered merge + More)
        d.list(b"/registry/pods/", b"/registry/pods0", 0, 1000)
        d.count(b"/registry/pods/", b"/registry/pods0")
        # gets
        for _ in range(100):
            k = keys[int(qrng.integers(len(keys)))]
            rev = int(qrng.choice([0, cur - 10, base + 50]))
            d.get(k, max(rev, 0))
        # compact and diff the complete store
        d.compact(cur - 100)
        d.diff_dump()
        d.list(b"/registry/pods/ns-0000/", b"/registry/pods/ns-00000", 0, 0)
        d.count(b"/registry/pods/", b"/registry/pods0")
        d.compact(0)
        d.diff_dump()
    finally:
        d.close()


def test_big_values():
    """Large values: heap, gather-arena overflow handling (chunk halving),
    point reads."""
    d = parity.Dual()
    try:
        import ctypes as C
        for st in (d.o, d.p):  # larger wire buffers for ~85MB of results
            st.BUF = 192 << 20
            st.buf = C.create_string_buffer(st.BUF)
        rng = np.random.default_rng(3)
        big = rng.integers(0, 256, size=2 << 20, dtype=np.uint8).tobytes()  # 2MB
        keys = [b"/registry/big/obj-%03d" % i for i in range(80)]
        for i, k in enumerate(keys):
            d.create(k, big[: (1 << 20) + i * 1024])
        # full list: 80 x ~1MB results exceed the 64MB test arena -> the
        # product must chunk-halve, not fail
        d.list(b"/registry/big/", b"/registry/big0", 0, 0)
        d.list(b"/registry/big/", b"/registry/big0", 0, 7)
        for i in (0, 17, 39):
            d.get(keys[i], 0)
        d.count(b"/registry/big/", b"/registry/big0")
        d.delete(keys[5], 0)
        d.get(keys[5], 0)
        d.compact(0)
        d.list(b"/registry/big/", b"/registry/big0", 0, 0)
    finally:
        d.close()
