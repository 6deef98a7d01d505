/* include/kb_slab.h — the drop-in C-ABI boundary of the MI355X-native
 * KubeBrain MVCC hot path (libkbslab.so).
 *
 * Every entry point cites the reference interface it replaces (paths into
 * /root/reference). A Go host binds this behind `storage.KvStorage` +
 * backend fast-path dispatch via cgo (see INTEGRATION.md for the binding
 * stub); error codes mirror the Go error taxonomy
 * (pkg/storage/interface.go:140-147, errors.go:20-70).
 *
 * Threading: all calls are thread-safe; writes are internally serialized
 * (single-writer, matching leader-only writes, pkg/server/etcd/kv.go:90-96).
 * Ownership: callers own all out buffers; the library copies in/out (the
 * reference's iterators copy too, pkg/storage/badger/iter.go:85-92).
 *
 * The library REQUIRES a HIP device (MI355X/gfx950): kb_new returns NULL and
 * kb_last_error() reports KB_ENOGPU when none is present. There is no CPU
 * fallback.
 */
#ifndef KB_SLAB_H
#define KB_SLAB_H

#include <stddef.h>
#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

/* status codes (== oracle/oracle.h codes; mirror storage/interface.go:140-147) */
enum kb_status {
  KB_OK = 0,
  KB_ENOTFOUND = 1,     /* storage.ErrKeyNotFound */
  KB_ECAS_FAILED = 2,   /* storage.ErrCASFailed */
  KB_EUNCERTAIN = 3,    /* storage.ErrUncertainResult (unreachable: local engine) */
  KB_ECOMPACTED = 4,    /* range revision < compact revision (scanner.go:617-621) */
  KB_EINVALID = 5,      /* invalid nil end / invalid range end (range.go:139-151) */
  KB_EUNSUPPORTED = 6,  /* ListByStream/GetPartitions: SURVEY.md §8f3, round-2+ */
  KB_EREV_DRIFT = 7,    /* backend.ErrRevisionDriftBack (backend.go:188) */
  KB_EWATCH_LOW = 8,    /* "cache event oldest revision ... newer" (watch.go:79-84) */
  KB_EWATCH_EMPTY = 9,  /* "empty cache event" (watch.go:67-71) */
  KB_EWATCH_DROPPED = 10, /* slow consumer dropped (watcherhub.go:84-94) */
  KB_EKEYTOOLONG = 11,  /* key > 96B: documented round-1 limit */
  KB_EBADKEY = 12,      /* key byte <= 0x24: the reference codec itself mis-orders
                           such keys (coder/normal.go:29-31) */
  KB_EINTERNAL = 13,
  KB_ENOGPU = 14,       /* no HIP device — the product path never falls back */
  KB_ENOBUF = 100       /* caller buffer too small; retry with a larger one */
};

typedef struct kb_store kb_store; /* one MVCC store on one GPU */

/* ---- lifecycle ----
 * Replaces backend.NewBackend wiring (pkg/backend/backend.go:145-186).
 * Capacity knobs via env: KB_MAX_ROWS (default 8M), KB_HEAP_BYTES (default
 * 2G), KB_FLUSH_ROWS (memtable flush threshold, default 65536), KB_DEVICE.
 */
kb_store* kb_new(const char* prefix, int watch_cache_size,
                 long long events_ttl_seconds, int enable_etcd_compatibility);
void kb_free(kb_store*);
int kb_last_error(char* msg, size_t cap); /* returns last kb_status, fills message */

/* ---- writes (leader txn protocol) ---- */
/* backend.Create (txn.go:33-77, creator/naive.go:48-105) */
int kb_create(kb_store*, const uint8_t* key, size_t klen, const uint8_t* val,
              size_t vlen, uint64_t* header_rev, int* succeeded);
/* backend.Update (txn.go:193-265); prev_rev==0 => create path */
int kb_update(kb_store*, const uint8_t* key, size_t klen, const uint8_t* val,
              size_t vlen, uint64_t prev_rev, uint64_t* header_rev, int* succeeded,
              int* has_kv, uint8_t* kv_val, size_t cap, size_t* kv_val_len,
              uint64_t* kv_rev);
/* backend.Delete (txn.go:79-190) */
int kb_delete(kb_store*, const uint8_t* key, size_t klen, uint64_t prev_rev,
              uint64_t* header_rev, int* succeeded, int* has_kv, uint8_t* kv_val,
              size_t cap, size_t* kv_val_len, uint64_t* kv_rev);

/* ---- reads (the GPU hot path) ---- */
/* backend.Get (range.go:34-121): MVCC point read at revision (0 = latest) */
int kb_get(kb_store*, const uint8_t* key, size_t klen, uint64_t rev,
           uint64_t* header_rev, int* has_kv, uint8_t* val, size_t cap,
           size_t* vlen, uint64_t* mod_rev);
/* backend.List (range.go:124-174): out = packed records
 * {u32 n; n × {u64 rev; u32 klen; key; u32 vlen; val}} (same wire as oracle) */
int kb_list(kb_store*, const uint8_t* start, size_t slen, const uint8_t* end,
            size_t elen, uint64_t rev, int64_t limit, uint8_t* out, size_t cap,
            size_t* out_len, uint64_t* header_rev, int* more);
/* backend.Count (range.go:177-205) */
int kb_count(kb_store*, const uint8_t* start, size_t slen, const uint8_t* end,
             size_t elen, uint64_t* header_rev, uint64_t* count);

/* ---- compaction (compact.go:31-127 + scanner.go:444-491,566-591) ---- */
int kb_compact(kb_store*, uint64_t rev, uint64_t* out_rev);

/* ---- watch (watch.go:37-159, ring.go, watcherhub.go) ---- */
long long kb_watch(kb_store*, const uint8_t* prefix, size_t plen, uint64_t rev,
                   int* status);
/* out = packed events {u32 n; n × {i32 type; u64 rev; u64 kv_rev; u32 klen;
 * key; u32 vlen; val}}. Returns KB_EWATCH_DROPPED once a slow consumer was
 * dropped (watcherhub.go:84-94: per-watcher buffer 10000). On KB_ENOBUF the
 * queue is left INTACT (*out_len = required size): retry with a larger
 * buffer and no event is lost. */
int kb_watch_poll(kb_store*, long long wid, uint8_t* out, size_t cap, size_t* out_len);
void kb_watch_cancel(kb_store*, long long wid);

/* ---- streaming full-range reads (backend.ListByStream range.go:247-256 +
 * scanner.RangeStream scanner.go:129-145; batches of 300, receiver.go:118-150)
 * and GetPartitions (range.go:208-245; single partition like badger.go:52-54,
 * sharding lives above this ABI). Stream results are pinned at read_rev; a
 * compaction past read_rev mid-stream returns KB_ECOMPACTED (the reference
 * holds an engine snapshot instead). ---- */
long long kb_stream_open(kb_store*, const uint8_t* start, size_t slen,
                         const uint8_t* end, size_t elen, uint64_t rev,
                         uint64_t* read_rev, int* status);
/* out = packed kvs like kb_list; an EMPTY batch is the end marker */
int kb_stream_next(kb_store*, long long sid, uint8_t* out, size_t cap,
                   size_t* out_len);
void kb_stream_close(kb_store*, long long sid);
int kb_partitions(kb_store*, const uint8_t* start, size_t slen,
                  const uint8_t* end, size_t elen, uint8_t* out, size_t cap,
                  size_t* out_len, uint64_t* header_rev);

/* ---- cross-shard exchange (RCCL over xGMI) ----
 * The key slab shards by namespace hash across the GPUs of one node, one
 * store per GPU. A Range that spans shards mirrors the reference's
 * multi-partition fork/merge (pkg/backend/scanner/scanner.go:269-300): each
 * shard scans locally, ONE collective exchanges the sorted winner runs
 * (allgather of counts, then payload padded to the max — NCCL/RCCL has no
 * allgatherv), and every rank k-way-merges with the global limit+1 cut.
 * The ncclUniqueId travels out-of-band on the caller's bootstrap channel
 * (cgo host RPC, torch.distributed gloo, ...). */
/* rank 0 generates the 128-byte ncclUniqueId */
int kb_comm_id(uint8_t* out, size_t cap, size_t* len);
/* collective: all ranks call with the same id; one store per GPU */
int kb_comm_init(kb_store*, const uint8_t* id, size_t id_len, int rank, int world);
int kb_comm_rank(kb_store*, int* rank, int* world);
void kb_comm_free(kb_store*);
/* cross-shard Range: COLLECTIVE (all ranks, same arguments); out = the
 * kb_list wire format with the globally merged result; degrades to the
 * local List when no communicator is initialized */
int kb_range_global(kb_store*, const uint8_t* start, size_t slen,
                    const uint8_t* end, size_t elen, uint64_t rev,
                    int64_t limit, uint8_t* out, size_t cap, size_t* out_len,
                    uint64_t* header_rev, int* more);
/* TEST-ONLY: the k-way merge + global limit cut on caller-supplied packed
 * runs (no GPU, no communicator) — lets CPU tests pin the exchange's merge
 * semantics; runs_cat = world runs back to back, lens their byte lengths */
int kb_test_merge_runs(const uint8_t* runs_cat, const unsigned long long* lens,
                       int world, long long limit, uint8_t* out, size_t cap,
                       size_t* out_len, int* more);

/* ---- revision (tso/tso.go:41-76) ---- */
unsigned long long kb_current_rev(kb_store*);
void kb_set_current_rev(kb_store*, unsigned long long rev); /* leader TSO init */

/* backend.Config.SkippedPrefixes (compact.go:108-127): comma-separated */
int kb_set_skipped_prefixes(kb_store*, const char* csv);
/* the encoded compact borders (golden-pinned by compact_test.go:36-79) */
int kb_compact_borders(kb_store*, uint8_t* out, size_t cap, size_t* out_len);

/* ---- test/ops hooks ---- */
void kb_clock_advance(kb_store*, long long seconds); /* TTL clock (scanner.go:147-177) */
int kb_flush(kb_store*); /* memtable -> HBM slab merge (normally automatic) */
/* full store dump (memtable flushed): {u32 n; n × {u32 klen; ikey; u32 vlen; val}}
 * sorted by internal key — byte-diffable against the oracle's okb_dump */
int kb_dump(kb_store*, uint8_t* out, size_t cap, size_t* out_len, uint64_t* n_rows);
int kb_event_log(kb_store*, uint8_t* out, size_t cap, size_t* out_len);

/* ---- bench support (the measured hot path; used by bench.py) ---- */
/* n Range queries in one device batch; returns total winners. mode bits:
 * 1 = d2h — results copied to pinned host memory, PIPELINED (the payload
 *     copy overlaps the next batch's kernels; call kb_sync before reading
 *     wall-clock);
 * 2 = keys_only — etcd3 RangeRequest.KeysOnly semantics: key + mod-revision
 *     per winner, no value bytes (an extension: the reference's etcd shim
 *     ignores the flag, server/etcd/kv.go:48-67).
 * mode 0 leaves results in the device arena (the `value` mode). */
int kb_bench_range(kb_store*, const uint8_t* qbuf, size_t nq, int mode,
                   unsigned long long* total_kvs, double* secs);
/* wait for in-flight pipelined D2H copies */
int kb_sync(kb_store*);
/* batched conditional updates (txn.go:249-265); tbuf = n × {u32 klen;
 * u64 prev_rev; u32 vlen; key; val}; out_revs[i] = new revision or 0 on CAS
 * failure */
int kb_bench_txn(kb_store*, const uint8_t* tbuf, size_t n, uint64_t* out_revs);
/* one bench step: Range batch launched async + txn batch overlapped on the
 * host while the kernels are in flight (results unchanged: kernels snapshot
 * device state at launch; writes stage host-side until the next sync).
 * mode bits as kb_bench_range, plus:
 * 4 = pipelined (excludes bit 1) — this step's Range batch stays in flight;
 *     `total` reports the PREVIOUS step's winners (first call reports 0) and
 *     kb_sync collects the final batch. Exact under MVCC: every Range reads
 *     at its fixed read_rev, so deferring collection by one step never
 *     changes results; txn CAS results (out_revs) are still returned
 *     in-step. */
int kb_bench_step(kb_store*, const uint8_t* qbuf, size_t nq,
                  const uint8_t* tbuf, size_t ntx, int mode, uint64_t* out_revs,
                  unsigned long long* total, double* secs);
/* batched deletes: dbuf = n × {u32 klen; u64 prev_rev; key} */
int kb_bench_del(kb_store*, const uint8_t* dbuf, size_t n, uint64_t* out_revs);
/* fast bulk insert == n serial Creates of fresh keys (see okb_bulk_create) */
int kb_bulk_create(kb_store*, const uint8_t* keys, const uint32_t* klens,
                   const uint8_t* vals, const uint32_t* vlens, size_t n);
/* perf counters as a JSON object (kernel-time totals from HIP events on the
 * store's stream, launch counts, rows scanned, bytes gathered) */
int kb_perf_json(kb_store*, char* out, size_t cap);
void kb_perf_reset(kb_store*);

#ifdef __cplusplus
}
#endif
#endif /* KB_SLAB_H */
