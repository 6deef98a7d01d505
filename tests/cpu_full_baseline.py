#!/usr/bin/env python3
"""One-off FULL-config CPU baseline: the oracle (CPU
restatement, kind "port") loaded with the ENTIRE 10M-key / 1M-extra-rev
keyspace and timed on all host cores — no subsetting, no extrapolation.
Run on the GPU box (256 host cores) via gpurun; the default bench keeps the
bounded-sample baseline so it finishes in minutes, and cites this file's
committed result (profiles/cpu_full_baseline.json) as the full-config check.

Usage: python tests/cpu_full_baseline.py [--nns 2000] [--per-ns 5000]
"""
import argparse
import ctypes as C
import json
import os
import struct
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np

from kbclient import open_oracle

VAL_LEN = 512
LIMIT = 500


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--nns", type=int, default=2000)
    ap.add_argument("--per-ns", type=int, default=5000)
    ap.add_argument("--extra-revs", type=int, default=1000000)
    ap.add_argument("--queries", type=int, default=60000)
    args = ap.parse_args()
    rng = np.random.default_rng(0x6B62)
    o = open_oracle()
    o.set_current_rev(1000)
    namespaces = [b"/registry/pods/ns-%04d" % i for i in range(args.nns)]
    t0 = time.time()
    n_total = 0
    CH = 1 << 20
    keys = []
    for ns in namespaces:
        for j in range(args.per_ns):
            keys.append(ns + b"/pod-%06d" % j)
    for c0 in range(0, len(keys), CH):
        ck = keys[c0:c0 + CH]
        m = len(ck)
        vals = rng.integers(0, 256, size=m * VAL_LEN, dtype=np.uint8).tobytes()
        klens = (C.c_uint32 * m)(*[len(k) for k in ck])
        vlens = (C.c_uint32 * m)(*([VAL_LEN] * m))
        rc = o.lib.okb_bulk_create(C.c_void_p(o.h), b"".join(ck), klens, vals,
                                   vlens, C.c_size_t(m))
        assert rc == 0
        n_total += m
        print(f"loaded {n_total}", file=sys.stderr, flush=True)
    # extra revisions through the oracle's serial update path (sampled zipf)
    base = 1000
    revs = {k: base + i + 1 for i, k in enumerate(keys)}
    zs = (rng.zipf(1.1, size=args.extra_revs) - 1) % len(keys)
    vx = b"x" * VAL_LEN
    t_rev = time.time()
    for i in range(args.extra_revs):
        k = keys[int(zs[i])]
        r = o.update(k, vx, revs[k])
        if r.succeeded:
            revs[k] = r.header_revision
    build_s = time.time() - t0
    print(f"build {build_s:.0f}s (revs {time.time()-t_rev:.0f}s)",
          file=sys.stderr, flush=True)
    threads = os.cpu_count() or 1
    qs = []
    qrng = np.random.default_rng(0x6B62 + 7)
    for _ in range(args.queries):
        ns = namespaces[int(qrng.integers(len(namespaces)))]
        qs.append((ns + b"/", ns + b"0", 0, LIMIT))
    parts = []
    for s_, e_, rev, limit in qs:
        parts.append(struct.pack("<IIQQ", len(s_), len(e_), rev, limit))
        parts.append(s_)
        parts.append(e_)
    blob = b"".join(parts)
    total = C.c_ulonglong()
    secs = C.c_double()
    rc = o.lib.okb_bench_range(C.c_void_p(o.h), blob, C.c_size_t(len(qs)),
                               C.c_int(threads), C.byref(total), C.byref(secs))
    assert rc == 0
    range_ops_s = len(qs) / secs.value
    # serial txn rate
    t0 = time.time()
    ntx = 4000
    for i in range(ntx):
        k = keys[(i * 2503) % len(keys)]
        r = o.update(k, vx, revs[k])
        assert r.succeeded
        revs[k] = r.header_revision
    txn_s = time.time() - t0
    txn_ops_s = ntx / txn_s
    mix = 1.0 / (0.9 / range_ops_s + 0.1 / txn_ops_s)
    print(json.dumps({
        "what": "FULL-config CPU baseline (no subsetting): oracle port, "
                "whole 10M-key/1M-rev keyspace",
        "value": round(mix, 1), "unit": "ops/s", "cores": threads,
        "kind": "port",
        "n_keys": len(keys), "extra_revs": args.extra_revs,
        "queries": len(qs), "range_secs": round(secs.value, 2),
        "range_ops_per_sec": round(range_ops_s, 1),
        "txn_ops_per_sec": round(txn_ops_s, 1),
        "build_seconds": round(build_s, 1),
    }))


if __name__ == "__main__":
    main()
