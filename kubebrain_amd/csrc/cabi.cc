// kubebrain_amd/csrc/cabi.cc — the exported C ABI (include/kb_slab.h) over the
// host store. Signatures intentionally mirror the oracle's okb_* ABI so the
// test harness drives both and diffs (tests/kbclient.py).

#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <string>
#include <vector>

#include "../../include/kb_slab.h"
#include "store.h"

using kbstore::Bytes;
using kbstore::Event;
using kbstore::Status;
using kbstore::Store;

namespace {
thread_local int g_last_status = 0;
thread_local std::string g_last_msg;

void set_err(int st, const std::string& msg) {
  g_last_status = st;
  g_last_msg = msg;
  static const bool dbg = getenv("KB_DEBUG") && atoi(getenv("KB_DEBUG")) != 0;
  if (dbg) fprintf(stderr, "[kb_slab] error %d: %s\n", st, msg.c_str());
}

struct Writer {
  uint8_t* out;
  size_t cap;
  size_t off = 0;
  bool overflow = false;
  void bytes(const void* p, size_t n) {
    if (!overflow && off + n <= cap)
      memcpy(out + off, p, n);
    else
      overflow = true;  // keep counting: off ends as the REQUIRED size,
                        // so every *out_len on KB_ENOBUF supports the
                        // grow-and-retry contract (kb_slab.h)
    off += n;
  }
  void u32(uint32_t v) { bytes(&v, 4); }
  void u64(uint64_t v) { bytes(&v, 8); }
  void i32(int32_t v) { bytes(&v, 4); }
  void str(const Bytes& s) { u32((uint32_t)s.size()); bytes(s.data(), s.size()); }
};

void writeEvents(Writer& w, const std::vector<Event>& evs) {
  w.u32((uint32_t)evs.size());
  for (auto& e : evs) {
    w.i32((int32_t)e.type);
    w.u64(e.revision);
    w.u64(e.kv_revision);
    w.str(e.kv_key);
    w.str(e.kv_value);
  }
}
}  // namespace

extern "C" {

kb_store* kb_new(const char* prefix, int watch_cache_size,
                 long long events_ttl_seconds, int enable_etcd_compatibility) {
  Store::Config cfg;
  cfg.prefix = prefix ? prefix : "/registry";
  if (watch_cache_size > 0) cfg.watch_cache_size = watch_cache_size;
  if (events_ttl_seconds > 0) cfg.events_ttl_seconds = events_ttl_seconds;
  cfg.enable_etcd_compatibility = enable_etcd_compatibility != 0;
  std::string err;
  Store* s = Store::Open(cfg, &err);
  if (!s) {
    set_err(KB_ENOGPU, err);
    return nullptr;
  }
  return (kb_store*)s;
}

void kb_free(kb_store* h) { delete (Store*)h; }

int kb_last_error(char* msg, size_t cap) {
  if (msg && cap) {
    size_t n = g_last_msg.size() < cap - 1 ? g_last_msg.size() : cap - 1;
    memcpy(msg, g_last_msg.data(), n);
    msg[n] = 0;
  }
  return g_last_status;
}

int kb_create(kb_store* h, const uint8_t* key, size_t klen, const uint8_t* val,
              size_t vlen, uint64_t* header_rev, int* succeeded) {
  Status st;
  auto r = ((Store*)h)->Create(Bytes((const char*)key, klen),
                               Bytes((const char*)val, vlen), &st);
  *header_rev = r.header_revision;
  *succeeded = r.succeeded;
  return st;
}

static int write_kv_out(const kbstore::WriteResponse& r, uint8_t* kv_val,
                        size_t cap, size_t* kv_val_len, uint64_t* kv_rev) {
  *kv_val_len = 0;
  *kv_rev = 0;
  if (r.has_kv) {
    if (r.kv.value.size() > cap) return KB_ENOBUF;
    memcpy(kv_val, r.kv.value.data(), r.kv.value.size());
    *kv_val_len = r.kv.value.size();
    *kv_rev = r.kv.revision;
  }
  return KB_OK;
}

int kb_update(kb_store* h, const uint8_t* key, size_t klen, const uint8_t* val,
              size_t vlen, uint64_t prev_rev, uint64_t* header_rev,
              int* succeeded, int* has_kv, uint8_t* kv_val, size_t cap,
              size_t* kv_val_len, uint64_t* kv_rev) {
  Status st;
  auto r = ((Store*)h)->Update(Bytes((const char*)key, klen),
                               Bytes((const char*)val, vlen), prev_rev, &st);
  *header_rev = r.header_revision;
  *succeeded = r.succeeded;
  *has_kv = r.has_kv;
  int rc = write_kv_out(r, kv_val, cap, kv_val_len, kv_rev);
  return rc != KB_OK ? rc : st;
}

int kb_delete(kb_store* h, const uint8_t* key, size_t klen, uint64_t prev_rev,
              uint64_t* header_rev, int* succeeded, int* has_kv, uint8_t* kv_val,
              size_t cap, size_t* kv_val_len, uint64_t* kv_rev) {
  Status st;
  auto r = ((Store*)h)->Delete(Bytes((const char*)key, klen), prev_rev, &st);
  *header_rev = r.header_revision;
  *succeeded = r.succeeded;
  *has_kv = r.has_kv;
  int rc = write_kv_out(r, kv_val, cap, kv_val_len, kv_rev);
  return rc != KB_OK ? rc : st;
}

int kb_get(kb_store* h, const uint8_t* key, size_t klen, uint64_t rev,
           uint64_t* header_rev, int* has_kv, uint8_t* val, size_t cap,
           size_t* vlen, uint64_t* mod_rev) {
  Status st;
  auto r = ((Store*)h)->Get(Bytes((const char*)key, klen), rev, &st);
  *header_rev = r.header_revision;
  *has_kv = r.has_kv;
  *vlen = 0;
  *mod_rev = 0;
  if (r.has_kv) {
    if (r.kv.value.size() > cap) return KB_ENOBUF;
    memcpy(val, r.kv.value.data(), r.kv.value.size());
    *vlen = r.kv.value.size();
    *mod_rev = r.kv.revision;
  }
  return st;
}

int kb_list(kb_store* h, const uint8_t* start, size_t slen, const uint8_t* end,
            size_t elen, uint64_t rev, int64_t limit, uint8_t* out, size_t cap,
            size_t* out_len, uint64_t* header_rev, int* more) {
  Status st;
  auto r = ((Store*)h)->List(Bytes((const char*)start, slen),
                             Bytes((const char*)end, elen), rev, limit, &st);
  *header_rev = r.header_revision;
  *more = r.more;
  Writer w{out, cap};
  w.u32((uint32_t)r.kvs.size());
  for (auto& kv : r.kvs) { w.u64(kv.revision); w.str(kv.key); w.str(kv.value); }
  *out_len = w.off;
  if (w.overflow) return KB_ENOBUF;
  return st;
}

int kb_count(kb_store* h, const uint8_t* start, size_t slen, const uint8_t* end,
             size_t elen, uint64_t* header_rev, uint64_t* count) {
  Status st;
  auto r = ((Store*)h)->Count(Bytes((const char*)start, slen),
                              Bytes((const char*)end, elen), &st);
  *header_rev = r.header_revision;
  *count = r.count;
  return st;
}

int kb_compact(kb_store* h, uint64_t rev, uint64_t* out_rev) {
  Status st;
  *out_rev = ((Store*)h)->Compact(rev, &st);
  return st;
}

long long kb_watch(kb_store* h, const uint8_t* prefix, size_t plen, uint64_t rev,
                   int* status) {
  Status st;
  long long wid = ((Store*)h)->Watch(Bytes((const char*)prefix, plen), rev, &st);
  *status = st;
  return wid;
}

int kb_watch_poll(kb_store* h, long long wid, uint8_t* out, size_t cap,
                  size_t* out_len) {
  // overflow-safe: the queue is only drained when the packed events fit, so
  // KB_ENOBUF can be retried with a larger buffer without losing events
  // (contiguous-revision delivery, backend.go:214-238). *out_len carries the
  // required size on KB_ENOBUF.
  Status st = ((Store*)h)->WatchPollWire(wid, out, cap, out_len);
  if (st == kbstore::NOBUF) return KB_ENOBUF;
  return st;
}

void kb_watch_cancel(kb_store* h, long long wid) { ((Store*)h)->WatchCancel(wid); }

long long kb_stream_open(kb_store* h, const uint8_t* start, size_t slen,
                         const uint8_t* end, size_t elen, uint64_t rev,
                         uint64_t* read_rev, int* status) {
  Status st;
  long long sid = ((Store*)h)->StreamOpen(Bytes((const char*)start, slen),
                                          Bytes((const char*)end, elen), rev,
                                          read_rev, &st);
  *status = st;
  return sid;
}

int kb_stream_next(kb_store* h, long long sid, uint8_t* out, size_t cap,
                   size_t* out_len) {
  std::vector<kbstore::KeyValue> kvs;
  Status st = ((Store*)h)->StreamNext(sid, &kvs);
  Writer w{out, cap};
  w.u32((uint32_t)kvs.size());
  for (auto& kv : kvs) { w.u64(kv.revision); w.str(kv.key); w.str(kv.value); }
  *out_len = w.off;
  if (w.overflow) return KB_ENOBUF;
  return st;
}

void kb_stream_close(kb_store* h, long long sid) { ((Store*)h)->StreamClose(sid); }

int kb_partitions(kb_store* h, const uint8_t* start, size_t slen,
                  const uint8_t* end, size_t elen, uint8_t* out, size_t cap,
                  size_t* out_len, uint64_t* header_rev) {
  auto parts = ((Store*)h)->GetPartitions(Bytes((const char*)start, slen),
                                          Bytes((const char*)end, elen),
                                          header_rev);
  Writer w{out, cap};
  w.u32((uint32_t)parts.size());
  for (auto& p : parts) w.str(p);
  *out_len = w.off;
  if (w.overflow) return KB_ENOBUF;
  return 0;
}

unsigned long long kb_current_rev(kb_store* h) {
  return ((Store*)h)->GetCurrentRevision();
}

void kb_set_current_rev(kb_store* h, unsigned long long rev) {
  ((Store*)h)->SetCurrentRevision(rev);
}

void kb_clock_advance(kb_store* h, long long seconds) {
  ((Store*)h)->ClockAdvance(seconds);
}

int kb_flush(kb_store* h) {
  std::string err;
  if (!((Store*)h)->Flush(&err)) {
    set_err(KB_EINTERNAL, err);
    return KB_EINTERNAL;
  }
  return KB_OK;
}

int kb_dump(kb_store* h, uint8_t* out, size_t cap, size_t* out_len,
            uint64_t* n_rows) {
  std::vector<std::pair<Bytes, Bytes>> rows;
  std::string err;
  if (!((Store*)h)->DumpStore(&rows, &err)) {
    set_err(KB_EINTERNAL, err);
    return KB_EINTERNAL;
  }
  Writer w{out, cap};
  w.u32((uint32_t)rows.size());
  for (auto& kv : rows) { w.str(kv.first); w.str(kv.second); }
  *out_len = w.off;
  *n_rows = rows.size();
  if (w.overflow) return KB_ENOBUF;
  return 0;
}

int kb_event_log(kb_store* h, uint8_t* out, size_t cap, size_t* out_len) {
  Writer w{out, cap};
  writeEvents(w, ((Store*)h)->EventLog());
  *out_len = w.off;
  if (w.overflow) return KB_ENOBUF;
  return 0;
}

int kb_bench_range(kb_store* h, const uint8_t* qbuf, size_t nq, int mode,
                   unsigned long long* total, double* secs) {
  std::string err;
  if (!((Store*)h)->BenchRange(qbuf, nq, mode, total, secs, &err)) {
    set_err(KB_EINTERNAL, err);
    return KB_EINTERNAL;
  }
  return 0;
}

int kb_sync(kb_store* h) {
  std::string err;
  if (!((Store*)h)->Sync(&err)) {
    set_err(KB_EINTERNAL, err);
    return KB_EINTERNAL;
  }
  return 0;
}

int kb_bench_txn(kb_store* h, const uint8_t* tbuf, size_t n, uint64_t* out_revs) {
  std::string err;
  if (!((Store*)h)->BenchTxn(tbuf, n, out_revs, &err)) {
    set_err(KB_EINTERNAL, err);
    return KB_EINTERNAL;
  }
  return 0;
}

int kb_bench_step(kb_store* h, const uint8_t* qbuf, size_t nq,
                  const uint8_t* tbuf, size_t ntx, int mode, uint64_t* out_revs,
                  unsigned long long* total, double* secs) {
  std::string err;
  if (!((Store*)h)->BenchStep(qbuf, nq, tbuf, ntx, mode, out_revs, total,
                              secs, &err)) {
    set_err(KB_EINTERNAL, err);
    return KB_EINTERNAL;
  }
  return 0;
}

int kb_bench_del(kb_store* h, const uint8_t* dbuf, size_t n, uint64_t* out_revs) {
  std::string err;
  if (!((Store*)h)->BenchDel(dbuf, n, out_revs, &err)) {
    set_err(KB_EINTERNAL, err);
    return KB_EINTERNAL;
  }
  return 0;
}

int kb_bulk_create(kb_store* h, const uint8_t* keys, const uint32_t* klens,
                   const uint8_t* vals, const uint32_t* vlens, size_t n) {
  std::string err;
  if (!((Store*)h)->BulkCreate(keys, klens, vals, vlens, n, &err)) {
    set_err(KB_EINTERNAL, err);
    return 1;
  }
  return 0;
}

int kb_set_skipped_prefixes(kb_store* h, const char* csv) {
  std::vector<Bytes> sp;
  Bytes cur;
  for (const char* p = csv;; ++p) {
    if (*p == ',' || *p == 0) {
      if (!cur.empty()) sp.push_back(cur);
      cur.clear();
      if (*p == 0) break;
    } else cur += *p;
  }
  ((Store*)h)->SetSkippedPrefixes(sp);
  return 0;
}

int kb_compact_borders(kb_store* h, uint8_t* out, size_t cap, size_t* out_len) {
  auto bs = ((Store*)h)->CompactBorders();
  Writer w{out, cap};
  w.u32((uint32_t)bs.size());
  for (auto& b : bs) w.str(b);
  *out_len = w.off;
  if (w.overflow) return KB_ENOBUF;
  return 0;
}

int kb_perf_json(kb_store* h, char* out, size_t cap) {
  std::string j = ((Store*)h)->PerfJson();
  if (j.size() + 1 > cap) return KB_ENOBUF;
  memcpy(out, j.data(), j.size() + 1);
  return 0;
}

void kb_perf_reset(kb_store* h) { ((Store*)h)->PerfReset(); }

}  // extern "C"
