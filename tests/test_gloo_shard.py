"""Multi-GPU path correctness on CPU: world_size-2 gloo processes, each
holding one shard (namespace-hash) of the keyspace in a CPU
oracle; cross-shard Range = per-shard scan + gather + k-way merge + global
limit cut, diffed against an unsharded oracle. This pins the distributed
algorithm the GPU bench uses (bench.py sharded mode / RCCL exchange)."""
import os
import struct
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))


def _shard_of(ns: bytes, world: int) -> int:
    # namespace-hash routing (stable across processes)
    import zlib
    return zlib.crc32(ns) % world


def merge_shard_results(per_shard, limit):
    """k-way merge of per-shard sorted winner lists + global limit+1/More cut
    (mirrors receiver fork/merge + range.go:154-171)."""
    import heapq
    merged = list(heapq.merge(*per_shard, key=lambda kv: kv[0]))
    more = False
    if limit > 0 and len(merged) > limit:
        more = True
        merged = merged[:limit]
    return merged, more


def _worker(rank, world, port):
    import torch.distributed as dist
    from kbclient import open_oracle

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)

    nns = 12
    per_ns = 25
    namespaces = [b"/registry/pods/ns-%02d" % i for i in range(nns)]
    # build my shard; ALSO build the full store on rank 0 for reference
    shard = open_oracle()
    shard.set_current_rev(1000)
    full = open_oracle() if rank == 0 else None
    if full:
        full.set_current_rev(1000)

    # deterministic interleaved global write order so revisions match:
    # writes route to the owner shard, every rank replays the global stream
    # and applies only its own (the single global TSO of DESIGN.md §3.4 is
    # modelled by set_current_rev before each applied write)
    rev = 1000
    for j in range(per_ns):
        for i, ns in enumerate(namespaces):
            key = ns + b"/pod-%03d" % j
            val = b"v-%02d-%03d" % (i, j)
            rev += 1
            if _shard_of(ns, world) == rank:
                shard.set_current_rev(rev - 1)
                r = shard.create(key, val)
                assert r.header_revision == rev, (r.header_revision, rev)
            shard.set_current_rev(rev)
            if full:
                r = full.create(key, val)
                assert r.header_revision == rev

    # a few deletes + updates through the same routed replay
    for i, ns in enumerate(namespaces[:4]):
        key = ns + b"/pod-000"
        rev += 1
        if _shard_of(ns, world) == rank:
            shard.set_current_rev(rev - 1)
            assert shard.delete(key, 0).succeeded
        shard.set_current_rev(rev)
        if full:
            assert full.delete(key, 0).succeeded

    import torch
    dist.barrier()

    # cross-shard Range queries (span all namespaces)
    for (start, end, qrev, limit) in [
        (b"/registry/pods/", b"/registry/pods0", 0, 0),
        (b"/registry/pods/", b"/registry/pods0", 0, 37),
        (b"/registry/pods/", b"/registry/pods0", 0, 300),
        (b"/registry/pods/ns-03", b"/registry/pods/ns-09", 0, 11),
        (b"/registry/pods/", b"/registry/pods0", rev - 40, 50),
    ]:
        r = shard.list(start, end, qrev, limit + (0 if limit == 0 else 1))
        mine = [(kv.key, kv.value, kv.revision) for kv in r.kvs]
        gathered = [None] * world
        dist.all_gather_object(gathered, mine)
        if rank == 0:
            merged, more = merge_shard_results(gathered, limit)
            ref = full.list(start, end, qrev, limit)
            refl = [(kv.key, kv.value, kv.revision) for kv in ref.kvs]
            assert merged == refl, (start, end, qrev, limit, len(merged), len(refl))
            assert more == ref.more, (limit, more, ref.more)
    dist.barrier()
    shard.close()
    if full:
        full.close()
    dist.destroy_process_group()


def test_sharded_range_merge_gloo():
    import torch.multiprocessing as mp

    port = 29771
    world = 2
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_worker, args=(r, world, port)) for r in range(world)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(180)
        assert p.exitcode == 0, p.exitcode
