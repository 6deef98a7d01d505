"""Runs bench.py's multi-rank path end-to-end on CPU (world_size 2, gloo)
with a mock device store, so the driver's 8-GPU scale run can't be the first
execution of the distributed logic (rank/env plumbing, sharding, collectives,
JSON emission)."""
import ctypes
import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


class MockStore:
    """Implements exactly the surface bench.py touches."""

    def __init__(self, **kw):
        self.rev = 0
        self.h = 1
        self.wids = 0

    def _f(self, name):
        return getattr(self, "f_" + name)

    # kbclient-style methods bench uses directly
    def set_current_rev(self, rev):
        self.rev = max(self.rev, rev)

    def current_rev(self):
        return self.rev

    def watch(self, prefix, rev=0):
        self.wids += 1
        return 0, self.wids

    def watch_poll(self, wid):
        return 0, []

    def watch_cancel(self, wid):
        pass

    def list(self, start, end, rev=0, limit=0):
        from kubebrain_amd.client import Kv, RangeResp
        n = min(limit if limit > 0 else 5, 5)
        kvs = [Kv(start + b"k%04d-r%d" % (i, os.environ.get("RANK", "0") != "0"),
                  b"v" * 8, 100 + i) for i in range(n)]
        return RangeResp(0, self.rev, kvs, False)

    def compact(self, rev=0):
        return 0, self.rev

    def close(self):
        pass

    # ctypes-style entry points
    def f_bulk_create(self, h, kblob, klens, vals, vlens, n):
        self.rev += n.value
        return 0

    def f_bench_txn(self, h, blob, n, out_ptr):
        for i in range(n.value):
            self.rev += 1
            out_ptr[i] = self.rev
        return 0

    def f_bench_del(self, h, blob, n, out_ptr):
        for i in range(n.value):
            self.rev += 1
            out_ptr[i] = self.rev
        return 0

    def f_bench_step(self, h, qblob, nq, tblob, ntx, d2h, out_ptr, total_ref,
                     secs_ref):
        for i in range(ntx.value):
            self.rev += 1
            out_ptr[i] = self.rev
        total_ref._obj.value = nq.value * 490
        secs_ref._obj.value = 0.001
        return 0

    def f_bench_range(self, h, blob, nq, d2h, total_ref, secs_ref):
        total_ref._obj.value = nq.value * 490
        secs_ref._obj.value = 0.001
        return 0

    def f_flush(self, h):
        return 0

    def f_sync(self, h):
        return 0

    def f_perf_reset(self, h):
        return 0

    def f_perf_json(self, h, buf, cap):
        buf.value = json.dumps({
            "scan_ms": 1.0, "gather_ms": 2.0, "merge_ms": 0.5, "get_ms": 0,
            "compact_ms": 0.1, "filter_ms": 0.1, "pack_d2h_ms": 0,
            "rows_scanned": 1000000, "bytes_gathered": 100000000,
            "winners": 100000, "slab_rows": 42, "heap_used": 1000,
            "delivered": 123, "sync_s": 0.0, "syncs": 1,
            "scan_launches": 10, "merges": 1, "compacts": 0,
            "filter_launches": 1, "filter_events": 10, "filter_watchers": 2,
        }).encode()
        return 0


def _worker(rank, world, port, outdir):
    os.environ.update({
        "RANK": str(rank), "WORLD_SIZE": str(world), "LOCAL_RANK": str(rank),
        "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port),
    })
    sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    import kubebrain_amd

    kubebrain_amd.open_store = lambda **kw: MockStore(**kw)
    import bench

    sys.argv = ["bench.py", "--steps", "3", "--warmup", "1", "--nns", "16",
                "--per-ns", "10", "--extra-revs", "50", "--watchers", "4",
                "--watch-events", "30", "--no-cpu-baseline"]
    import io
    import contextlib
    cap = io.StringIO()
    with contextlib.redirect_stdout(cap):
        bench.main()
    if rank == 0:
        out = cap.getvalue().strip()
        data = json.loads(out)
        assert data["n_gpus"] == world
        assert data["scaling"] == "weak"
        assert data["config"]["n_keys"] == 16 * world * 10  # weak scaling
        cs = data.get("cross_shard_range")
        assert cs and "error" not in cs, cs
        with open(os.path.join(outdir, "bench_mock.json"), "w") as f:
            f.write(out)


def test_bench_world2_gloo(tmp_path):
    import torch.multiprocessing as mp

    ctx = mp.get_context("spawn")
    world = 2
    procs = [ctx.Process(target=_worker, args=(r, world, 29773, str(tmp_path)))
             for r in range(world)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(240)
        assert p.exitcode == 0, p.exitcode
    data = json.loads(open(tmp_path / "bench_mock.json").read())
    assert data["value"] > 0
