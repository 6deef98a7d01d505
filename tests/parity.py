"""Dual-driver: replay one op sequence through the CPU oracle and the GPU
store, asserting bit-exact agreement on every response, the watch event
streams, and the final store dump."""
from __future__ import annotations

import os

import kbclient


def _fmt_kvs(kvs):
    return [(k.key, k.value, k.revision) for k in kvs]


class Dual:
    def __init__(self, store_prefix=b"/registry", watch_cache_size=0,
                 events_ttl=0, init_rev=1000):
        import kubebrain_amd

        self.o = kbclient.open_oracle(store_prefix=store_prefix,
                                      watch_cache_size=watch_cache_size,
                                      events_ttl=events_ttl)
        self.p = kubebrain_amd.open_store(store_prefix=store_prefix,
                                          watch_cache_size=watch_cache_size,
                                          events_ttl=events_ttl)
        if init_rev:
            self.o.set_current_rev(init_rev)
            self.p.set_current_rev(init_rev)
        self.watches = []  # (oracle_wid, product_wid)

    def close(self):
        self.o.close()
        self.p.close()

    def create(self, key, val):
        ro, rp = self.o.create(key, val), self.p.create(key, val)
        assert (ro.status, ro.header_revision, ro.succeeded) == \
               (rp.status, rp.header_revision, rp.succeeded), (key, ro, rp)
        return rp

    def update(self, key, val, prev=0):
        ro, rp = self.o.update(key, val, prev), self.p.update(key, val, prev)
        assert (ro.status, ro.header_revision, ro.succeeded, ro.kv) == \
               (rp.status, rp.header_revision, rp.succeeded, rp.kv), (key, ro, rp)
        return rp

    def delete(self, key, prev=0):
        ro, rp = self.o.delete(key, prev), self.p.delete(key, prev)
        assert (ro.status, ro.header_revision, ro.succeeded, ro.kv) == \
               (rp.status, rp.header_revision, rp.succeeded, rp.kv), (key, ro, rp)
        return rp

    def get(self, key, rev=0):
        go, gp = self.o.get(key, rev), self.p.get(key, rev)
        assert go == gp, (key, rev, go, gp)
        return gp

    def list(self, start, end, rev=0, limit=0):
        lo, lp = self.o.list(start, end, rev, limit), self.p.list(start, end, rev, limit)
        assert (lo.status, lo.header_revision, lo.more) == \
               (lp.status, lp.header_revision, lp.more), (start, end, rev, limit, lo, lp)
        assert _fmt_kvs(lo.kvs) == _fmt_kvs(lp.kvs), (start, end, rev, limit)
        return lp

    def count(self, start, end):
        co, cp = self.o.count(start, end), self.p.count(start, end)
        assert co == cp, (start, end, co, cp)
        return cp

    def compact(self, rev=0):
        co, cp = self.o.compact(rev), self.p.compact(rev)
        assert co == cp, (rev, co, cp)
        return cp

    def clock_advance(self, secs):
        self.o.clock_advance(secs)
        self.p.clock_advance(secs)

    def watch(self, prefix, rev=0):
        so, wo = self.o.watch(prefix, rev)
        sp, wp = self.p.watch(prefix, rev)
        assert so == sp, (prefix, rev, so, sp)
        if so == kbclient.OK:
            self.watches.append((wo, wp))
            return len(self.watches) - 1
        return None

    def poll(self, idx):
        wo, wp = self.watches[idx]
        so, eo = self.o.watch_poll(wo)
        sp, ep = self.p.watch_poll(wp)
        assert so == sp, (so, sp)
        assert eo == ep, (idx, eo[:5], ep[:5], len(eo), len(ep))
        return ep

    def poll_all(self):
        for i in range(len(self.watches)):
            self.poll(i)

    def stream(self, start, end, rev=0):
        so, bo = self.o.stream(start, end, rev)
        sp, bp = self.p.stream(start, end, rev)
        assert so == sp, (so, sp)
        assert len(bo) == len(bp), (len(bo), len(bp))
        for i, (x, y) in enumerate(zip(bo, bp)):
            assert _fmt_kvs(x) == _fmt_kvs(y), i
        return bp

    def partitions(self, start, end):
        po, pp = self.o.partitions(start, end), self.p.partitions(start, end)
        assert po == pp, (po, pp)
        return pp

    def diff_dump(self):
        do, dp = self.o.dump(), self.p.dump()
        assert len(do) == len(dp), (len(do), len(dp))
        for i, (a, b) in enumerate(zip(do, dp)):
            assert a == b, (i, a, b)

    def diff_event_log(self):
        assert self.o.event_log() == self.p.event_log()


def small_env():
    """Shrink device allocations for tests (read by kb_new via env)."""
    os.environ.setdefault("KB_MAX_ROWS", str(1 << 20))
    os.environ.setdefault("KB_HEAP_BYTES", str(256 << 20))
    os.environ.setdefault("KB_ARENA_BYTES", str(64 << 20))
    os.environ.setdefault("KB_FLUSH_ROWS", "512")  # exercise merges often
    os.environ.setdefault("KB_VALIDATE_DN", "1")   # verify async-merge row-count predictions
