"""ctypes client over the kb_* C-ABI (include/kb_slab.h).

The same class also drives the oracle's okb_* ABI from tests/kbclient.py —
the two ABIs are signature-identical by design so parity tests can diff them.
"""
from __future__ import annotations

import ctypes as C
import struct
from dataclasses import dataclass, field

# status codes (mirror oracle/oracle.h and include/kb_slab.h)
OK = 0
NOTFOUND = 1
CAS_FAILED = 2
UNCERTAIN = 3
COMPACTED = 4
INVALID_ARG = 5
UNSUPPORTED = 6
REV_DRIFT = 7
WATCH_LOW = 8
WATCH_EMPTY = 9
WATCH_DROPPED = 10
KEYTOOLONG = 11
BADKEY = 12
INTERNAL = 13
ENOGPU = 14
ENOBUF = 100


@dataclass
class Kv:
    key: bytes
    value: bytes
    revision: int


@dataclass
class Ev:
    type: int  # 0 CREATE, 1 PUT, 2 DELETE
    revision: int
    kv_revision: int
    key: bytes
    value: bytes


@dataclass
class WriteResp:
    status: int
    header_revision: int
    succeeded: bool
    kv: Kv | None = None


@dataclass
class RangeResp:
    status: int
    header_revision: int
    kvs: list = field(default_factory=list)
    more: bool = False


def _parse_kvs(buf: bytes):
    (n,) = struct.unpack_from("<I", buf, 0)
    off = 4
    kvs = []
    for _ in range(n):
        (rev,) = struct.unpack_from("<Q", buf, off); off += 8
        (klen,) = struct.unpack_from("<I", buf, off); off += 4
        key = buf[off:off + klen]; off += klen
        (vlen,) = struct.unpack_from("<I", buf, off); off += 4
        val = buf[off:off + vlen]; off += vlen
        kvs.append(Kv(key, val, rev))
    return kvs


def _parse_events(buf: bytes):
    (n,) = struct.unpack_from("<I", buf, 0)
    off = 4
    evs = []
    for _ in range(n):
        (t,) = struct.unpack_from("<i", buf, off); off += 4
        (rev, kvrev) = struct.unpack_from("<QQ", buf, off); off += 16
        (klen,) = struct.unpack_from("<I", buf, off); off += 4
        key = buf[off:off + klen]; off += klen
        (vlen,) = struct.unpack_from("<I", buf, off); off += 4
        val = buf[off:off + vlen]; off += vlen
        evs.append(Ev(t, rev, kvrev, key, val))
    return evs


class Store:
    """One API over both backends. prefix='okb_' (oracle) or 'kb_' (product)."""

    BUF = 64 << 20

    def __init__(self, so_path: str, prefix: str, store_prefix: bytes = b"/registry",
                 watch_cache_size: int = 0, events_ttl: int = 0, etcd_compat: bool = True,
                 **kw):
        self.lib = C.CDLL(so_path)
        self.p = prefix
        f = self._f("new")
        f.restype = C.c_void_p
        f.argtypes = [C.c_char_p, C.c_int, C.c_longlong, C.c_int]
        self.h = f(store_prefix, watch_cache_size, events_ttl, 1 if etcd_compat else 0)
        if not self.h:
            raise RuntimeError(f"{prefix}new failed")
        self.buf = C.create_string_buffer(self.BUF)

    def _f(self, name):
        return getattr(self.lib, self.p + name)

    def close(self):
        if self.h:
            f = self._f("free")
            f.argtypes = [C.c_void_p]
            f(self.h)
            self.h = None

    # -- ops --
    def create(self, key: bytes, val: bytes) -> WriteResp:
        hr = C.c_uint64(); succ = C.c_int()
        rc = self._f("create")(C.c_void_p(self.h), key, C.c_size_t(len(key)), val,
                               C.c_size_t(len(val)), C.byref(hr), C.byref(succ))
        return WriteResp(rc, hr.value, bool(succ.value))

    def _write(self, fn, key, prev_rev, val=None):
        hr = C.c_uint64(); succ = C.c_int(); has = C.c_int()
        vlen = C.c_size_t(); kvrev = C.c_uint64()
        if val is not None:
            rc = self._f(fn)(C.c_void_p(self.h), key, C.c_size_t(len(key)), val,
                             C.c_size_t(len(val)), C.c_uint64(prev_rev), C.byref(hr),
                             C.byref(succ), C.byref(has), self.buf, C.c_size_t(self.BUF),
                             C.byref(vlen), C.byref(kvrev))
        else:
            rc = self._f(fn)(C.c_void_p(self.h), key, C.c_size_t(len(key)),
                             C.c_uint64(prev_rev), C.byref(hr), C.byref(succ),
                             C.byref(has), self.buf, C.c_size_t(self.BUF),
                             C.byref(vlen), C.byref(kvrev))
        kv = Kv(key, self.buf[:vlen.value], kvrev.value) if has.value else None
        return WriteResp(rc, hr.value, bool(succ.value), kv)

    def update(self, key: bytes, val: bytes, prev_rev: int = 0) -> WriteResp:
        return self._write("update", key, prev_rev, val)

    def delete(self, key: bytes, prev_rev: int = 0) -> WriteResp:
        return self._write("delete", key, prev_rev)

    def get(self, key: bytes, rev: int = 0):
        hr = C.c_uint64(); has = C.c_int(); vlen = C.c_size_t(); mod = C.c_uint64()
        rc = self._f("get")(C.c_void_p(self.h), key, C.c_size_t(len(key)),
                            C.c_uint64(rev), C.byref(hr), C.byref(has), self.buf,
                            C.c_size_t(self.BUF), C.byref(vlen), C.byref(mod))
        kv = Kv(key, self.buf[:vlen.value], mod.value) if has.value else None
        return rc, hr.value, kv

    def list(self, start: bytes, end: bytes, rev: int = 0, limit: int = 0) -> RangeResp:
        out_len = C.c_size_t(); hr = C.c_uint64(); more = C.c_int()
        rc = self._f("list")(C.c_void_p(self.h), start, C.c_size_t(len(start)), end,
                             C.c_size_t(len(end)), C.c_uint64(rev), C.c_int64(limit),
                             self.buf, C.c_size_t(self.BUF), C.byref(out_len),
                             C.byref(hr), C.byref(more))
        if rc not in (OK,):
            return RangeResp(rc, hr.value)
        return RangeResp(rc, hr.value, _parse_kvs(self.buf[:out_len.value]),
                         bool(more.value))

    def count(self, start: bytes, end: bytes):
        hr = C.c_uint64(); cnt = C.c_uint64()
        rc = self._f("count")(C.c_void_p(self.h), start, C.c_size_t(len(start)), end,
                              C.c_size_t(len(end)), C.byref(hr), C.byref(cnt))
        return rc, hr.value, cnt.value

    def compact(self, rev: int = 0):
        out = C.c_uint64()
        rc = self._f("compact")(C.c_void_p(self.h), C.c_uint64(rev), C.byref(out))
        return rc, out.value

    def watch(self, prefix: bytes, rev: int = 0):
        st = C.c_int()
        f = self._f("watch")
        f.restype = C.c_longlong
        wid = f(C.c_void_p(self.h), prefix, C.c_size_t(len(prefix)), C.c_uint64(rev),
                C.byref(st))
        return st.value, wid

    def watch_poll(self, wid: int):
        out_len = C.c_size_t()
        rc = self._f("watch_poll")(C.c_void_p(self.h), C.c_longlong(wid), self.buf,
                                   C.c_size_t(self.BUF), C.byref(out_len))
        if rc != OK:
            return rc, []
        return rc, _parse_events(self.buf[:out_len.value])

    def watch_cancel(self, wid: int):
        self._f("watch_cancel")(C.c_void_p(self.h), C.c_longlong(wid))

    def stream(self, start: bytes, end: bytes, rev: int = 0):
        """ListByStream: list of batches (each a list of Kv); oracle and
        product expose different mechanics (batch-indexed vs handle), unified
        here."""
        batches = []
        if self.p == "okb_":
            idx = 0
            while True:
                out_len = C.c_size_t(); rrev = C.c_uint64()
                rc = self._f("stream_batch")(C.c_void_p(self.h), start,
                                             C.c_size_t(len(start)), end,
                                             C.c_size_t(len(end)), C.c_uint64(rev),
                                             C.c_uint64(idx), self.buf,
                                             C.c_size_t(self.BUF), C.byref(out_len),
                                             C.byref(rrev))
                if rc != OK:
                    return rc, batches
                kvs = _parse_kvs(self.buf[:out_len.value])
                if not kvs:
                    return OK, batches
                batches.append(kvs)
                idx += 1
        else:
            st = C.c_int(); rrev = C.c_uint64()
            f = self._f("stream_open")
            f.restype = C.c_longlong
            sid = f(C.c_void_p(self.h), start, C.c_size_t(len(start)), end,
                    C.c_size_t(len(end)), C.c_uint64(rev), C.byref(rrev),
                    C.byref(st))
            if st.value != OK:
                return st.value, batches
            while True:
                out_len = C.c_size_t()
                rc = self._f("stream_next")(C.c_void_p(self.h), C.c_longlong(sid),
                                            self.buf, C.c_size_t(self.BUF),
                                            C.byref(out_len))
                if rc != OK:
                    return rc, batches
                kvs = _parse_kvs(self.buf[:out_len.value])
                if not kvs:
                    return OK, batches
                batches.append(kvs)

    def partitions(self, start: bytes, end: bytes):
        out_len = C.c_size_t(); hr = C.c_uint64()
        rc = self._f("partitions")(C.c_void_p(self.h), start,
                                   C.c_size_t(len(start)), end,
                                   C.c_size_t(len(end)), self.buf,
                                   C.c_size_t(self.BUF), C.byref(out_len),
                                   C.byref(hr))
        assert rc == 0
        buf = self.buf[:out_len.value]
        (n,) = struct.unpack_from("<I", buf, 0)
        off = 4
        parts = []
        for _ in range(n):
            (ln,) = struct.unpack_from("<I", buf, off); off += 4
            parts.append(buf[off:off + ln]); off += ln
        return hr.value, parts

    def current_rev(self) -> int:
        f = self._f("current_rev")
        f.restype = C.c_uint64
        return f(C.c_void_p(self.h))

    def set_current_rev(self, rev: int):
        self._f("set_current_rev")(C.c_void_p(self.h), C.c_uint64(rev))

    def clock_advance(self, secs: int):
        self._f("clock_advance")(C.c_void_p(self.h), C.c_longlong(secs))

    def dump(self):
        """Full internal store: list[(internal_key, value)] sorted."""
        out_len = C.c_size_t(); n = C.c_uint64()
        rc = self._f("dump")(C.c_void_p(self.h), self.buf, C.c_size_t(self.BUF),
                             C.byref(out_len), C.byref(n))
        assert rc == 0, rc
        buf = self.buf[:out_len.value]
        (cnt,) = struct.unpack_from("<I", buf, 0)
        off = 4
        rows = []
        for _ in range(cnt):
            (klen,) = struct.unpack_from("<I", buf, off); off += 4
            k = buf[off:off + klen]; off += klen
            (vlen,) = struct.unpack_from("<I", buf, off); off += 4
            v = buf[off:off + vlen]; off += vlen
            rows.append((k, v))
        return rows

    def event_log(self):
        out_len = C.c_size_t()
        rc = self._f("event_log")(C.c_void_p(self.h), self.buf, C.c_size_t(self.BUF),
                                  C.byref(out_len))
        assert rc == 0, rc
        return _parse_events(self.buf[:out_len.value])

