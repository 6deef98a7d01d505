"""GPU micro-benchmark of per-op host+device costs (not a pytest; run via
gpurun to localize per-op latency, e.g. the delete path)."""
import ctypes
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

os.environ.setdefault("KB_MAX_ROWS", str(4 << 20))
os.environ.setdefault("KB_HEAP_BYTES", str(1 << 30))
os.environ.setdefault("KB_FLUSH_ROWS", "262144")
os.environ.setdefault("KB_EVENT_LOG", "0")

import kubebrain_amd


def perf(store):
    buf = ctypes.create_string_buffer(4096)
    store._f("perf_json")(ctypes.c_void_p(store.h), buf, ctypes.c_size_t(4096))
    return json.loads(buf.value.decode())


def main():
    s = kubebrain_amd.open_store()
    s.set_current_rev(1000)
    N = 20000
    t0 = time.time()
    revs = {}
    for i in range(N):
        k = b"/registry/pods/ns-%03d/p-%05d" % (i % 50, i)
        r = s.create(k, b"v" * 512)
        revs[k] = r.header_revision
    print(f"create x{N}: {time.time()-t0:.2f}s")
    # build a mid-size delta: updates via single-op path
    t0 = time.time()
    keys = list(revs)
    for i in range(3000):
        k = keys[i]
        r = s.update(k, b"u" * 512, revs[k])
        revs[k] = r.header_revision
    print(f"update x3000 (no reads): {time.time()-t0:.2f}s  perf={perf(s)}")
    # the suspect: deletes (each does get() -> syncReads -> GetBatch)
    for batch in range(4):
        t0 = time.time()
        p0 = perf(s)
        for i in range(200):
            k = keys[3000 + batch * 200 + i]
            r = s.delete(k, revs[k])
            assert r.succeeded
        dt = time.time() - t0
        p1 = perf(s)
        print(f"delete x200 batch{batch}: {dt:.2f}s ({dt/200*1e3:.2f}ms/op) "
              f"sync_s={p1['sync_s']-p0['sync_s']:.2f} "
              f"syncs={p1['syncs']-p0['syncs']} "
              f"merge_ms={p1['merge_ms']-p0['merge_ms']:.1f} "
              f"get_ms={p1['get_ms']-p0['get_ms']:.1f} "
              f"merges={p1['merges']-p0['merges']}")
    # gets alone
    t0 = time.time()
    for i in range(500):
        s.get(keys[i], 0)
    print(f"get x500: {time.time()-t0:.2f}s  perf={perf(s)}")
    s.close()


if __name__ == "__main__":
    main()
