// oracle/oracle_cabi.cc — TEST INFRASTRUCTURE ONLY (see oracle.h).
// ctypes-friendly C ABI over the oracle, used by tests/, smoke() and
// bench.py's cpu_baseline leg. Signatures intentionally mirror the product
// C-ABI (include/kb_slab.h) so one Python harness drives both and diffs.

#include <atomic>
#include <chrono>
#include <cstring>
#include <thread>
#include <vector>

#include "oracle.h"

using oracle::Backend;
using oracle::Bytes;
using oracle::Event;
using oracle::Status;

namespace {
constexpr int kEnoBuf = 100;

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

void writeKvs(Writer& w, const std::vector<oracle::KeyValue>& kvs) {
  w.u32((uint32_t)kvs.size());
  for (auto& kv : kvs) { w.u64(kv.revision); w.str(kv.key); w.str(kv.value); }
}

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

void* okb_new(const char* prefix, int watch_cache_size, long long events_ttl_seconds,
              int etcd_compat) {
  Backend::Config cfg;
  cfg.prefix = prefix;
  if (watch_cache_size > 0) cfg.watch_cache_size = watch_cache_size;
  if (events_ttl_seconds > 0) cfg.events_ttl_seconds = events_ttl_seconds;
  cfg.enable_etcd_compatibility = etcd_compat != 0;
  return new Backend(cfg);
}

void okb_free(void* h) { delete (Backend*)h; }

int okb_create(void* h, const uint8_t* key, size_t klen, const uint8_t* val,
               size_t vlen, uint64_t* header_rev, int* succeeded) {
  Status st;
  auto r = ((Backend*)h)->Create(Bytes((const char*)key, klen),
                                 Bytes((const char*)val, vlen), &st);
  *header_rev = r.header_revision;
  *succeeded = r.succeeded;
  return st;
}

int okb_update(void* h, const uint8_t* key, size_t klen, const uint8_t* val,
               size_t vlen, uint64_t prev_rev, uint64_t* header_rev, int* succeeded,
               int* has_kv, uint8_t* kv_val, size_t cap, size_t* kv_val_len,
               uint64_t* kv_rev) {
  Status st;
  auto r = ((Backend*)h)->Update(Bytes((const char*)key, klen),
                                 Bytes((const char*)val, vlen), prev_rev, &st);
  *header_rev = r.header_revision;
  *succeeded = r.succeeded;
  *has_kv = r.has_kv;
  *kv_val_len = 0; *kv_rev = 0;
  if (r.has_kv) {
    if (r.kv.value.size() > cap) return kEnoBuf;
    memcpy(kv_val, r.kv.value.data(), r.kv.value.size());
    *kv_val_len = r.kv.value.size();
    *kv_rev = r.kv.revision;
  }
  return st;
}

int okb_delete(void* h, const uint8_t* key, size_t klen, uint64_t prev_rev,
               uint64_t* header_rev, int* succeeded, int* has_kv, uint8_t* kv_val,
               size_t cap, size_t* kv_val_len, uint64_t* kv_rev) {
  Status st;
  auto r = ((Backend*)h)->Delete(Bytes((const char*)key, klen), prev_rev, &st);
  *header_rev = r.header_revision;
  *succeeded = r.succeeded;
  *has_kv = r.has_kv;
  *kv_val_len = 0; *kv_rev = 0;
  if (r.has_kv) {
    if (r.kv.value.size() > cap) return kEnoBuf;
    memcpy(kv_val, r.kv.value.data(), r.kv.value.size());
    *kv_val_len = r.kv.value.size();
    *kv_rev = r.kv.revision;
  }
  return st;
}

int okb_get(void* h, const uint8_t* key, size_t klen, uint64_t rev,
            uint64_t* header_rev, int* has_kv, uint8_t* val, size_t cap,
            size_t* vlen, uint64_t* mod_rev) {
  Status st;
  auto r = ((Backend*)h)->Get(Bytes((const char*)key, klen), rev, &st);
  *header_rev = r.header_revision;
  *has_kv = r.has_kv;
  *vlen = 0; *mod_rev = 0;
  if (r.has_kv) {
    if (r.kv.value.size() > cap) return kEnoBuf;
    memcpy(val, r.kv.value.data(), r.kv.value.size());
    *vlen = r.kv.value.size();
    *mod_rev = r.kv.revision;
  }
  return st;
}

int okb_list(void* h, const uint8_t* start, size_t slen, const uint8_t* end,
             size_t elen, uint64_t rev, int64_t limit, uint8_t* out, size_t cap,
             size_t* out_len, uint64_t* header_rev, int* more) {
  Status st;
  auto r = ((Backend*)h)->List(Bytes((const char*)start, slen),
                               Bytes((const char*)end, elen), rev, limit, &st);
  *header_rev = r.header_revision;
  *more = r.more;
  Writer w{out, cap};
  writeKvs(w, r.kvs);
  *out_len = w.off;
  if (w.overflow) return kEnoBuf;
  return st;
}

int okb_count(void* h, const uint8_t* start, size_t slen, const uint8_t* end,
              size_t elen, uint64_t* header_rev, uint64_t* count) {
  Status st;
  auto r = ((Backend*)h)->Count(Bytes((const char*)start, slen),
                                Bytes((const char*)end, elen), &st);
  *header_rev = r.header_revision;
  *count = r.count;
  return st;
}

int okb_compact(void* h, uint64_t rev, uint64_t* out_rev) {
  Status st;
  *out_rev = ((Backend*)h)->Compact(rev, &st);
  return st;
}

long long okb_watch(void* h, const uint8_t* prefix, size_t plen, uint64_t rev,
                    int* status) {
  Status st;
  int64_t wid = ((Backend*)h)->Watch(Bytes((const char*)prefix, plen), rev, &st);
  *status = st;
  return wid;
}

int okb_watch_poll(void* h, long long wid, uint8_t* out, size_t cap, size_t* out_len) {
  // overflow-safe like kb_watch_poll: KB_ENOBUF leaves the queue intact and
  // *out_len carries the required size
  Status st;
  size_t need = 0;
  auto evs = ((Backend*)h)->WatchPollLimited(wid, cap, &need, &st);
  if (st == oracle::NOBUF) { *out_len = need; return kEnoBuf; }
  Writer w{out, cap};
  writeEvents(w, evs);
  *out_len = w.off;
  if (w.overflow) return kEnoBuf;
  return st;
}

void okb_watch_cancel(void* h, long long wid) { ((Backend*)h)->WatchCancel(wid); }

unsigned long long okb_current_rev(void* h) { return ((Backend*)h)->GetCurrentRevision(); }
void okb_set_current_rev(void* h, unsigned long long rev) { ((Backend*)h)->SetCurrentRevision(rev); }
void okb_clock_advance(void* h, long long secs) { ((Backend*)h)->ClockAdvance(secs); }

// Dump the full internal store (sorted internal key -> value) for slab diffs.
int okb_dump(void* h, uint8_t* out, size_t cap, size_t* out_len, uint64_t* n_rows) {
  const auto& s = ((Backend*)h)->DumpStore();
  Writer w{out, cap};
  w.u32((uint32_t)s.size());
  for (auto& kv : s) { w.str(kv.first); w.str(kv.second); }
  *out_len = w.off;
  *n_rows = s.size();
  if (w.overflow) return kEnoBuf;
  return 0;
}

int okb_event_log(void* h, uint8_t* out, size_t cap, size_t* out_len) {
  Writer w{out, cap};
  writeEvents(w, ((Backend*)h)->EventLog());
  *out_len = w.off;
  if (w.overflow) return kEnoBuf;
  return 0;
}

// ListByStream: one call returns batch #idx (re-runs the scan; test-only
// oracle, simplicity over speed). rc: 0 with n>0; 0 with n==0 => end marker.
int okb_stream_batch(void* h, const uint8_t* start, size_t slen,
                     const uint8_t* end, size_t elen, uint64_t rev,
                     uint64_t batch_idx, uint8_t* out, size_t cap,
                     size_t* out_len, uint64_t* read_rev) {
  Status st;
  auto batches = ((Backend*)h)->ListByStream(Bytes((const char*)start, slen),
                                             Bytes((const char*)end, elen), rev,
                                             read_rev, &st);
  if (st != oracle::OK) return st;
  Writer w{out, cap};
  if (batch_idx < batches.size()) writeKvs(w, batches[batch_idx].kvs);
  else w.u32(0);
  *out_len = w.off;
  if (w.overflow) return kEnoBuf;
  return 0;
}

int okb_partitions(void* h, const uint8_t* start, size_t slen,
                   const uint8_t* end, size_t elen, uint8_t* out, size_t cap,
                   size_t* out_len, uint64_t* header_rev) {
  auto parts = ((Backend*)h)->GetPartitions(Bytes((const char*)start, slen),
                                            Bytes((const char*)end, elen),
                                            header_rev);
  Writer w{out, cap};
  w.u32((uint32_t)parts.size());
  for (auto& p : parts) w.str(p);
  *out_len = w.off;
  if (w.overflow) return kEnoBuf;
  return 0;
}

int okb_set_skipped_prefixes(void* h, const char* csv) {
  std::vector<Bytes> sp;
  Bytes cur;
  for (const char* p = csv; ; ++p) {
    if (*p == ',' || *p == 0) {
      if (!cur.empty()) sp.push_back(cur);
      cur.clear();
      if (*p == 0) break;
    } else cur += *p;
  }
  ((Backend*)h)->SetSkippedPrefixes(sp);
  return 0;
}

int okb_compact_borders(void* h, uint8_t* out, size_t cap, size_t* out_len) {
  auto bs = ((Backend*)h)->CompactBorders();
  Writer w{out, cap};
  w.u32((uint32_t)bs.size());
  for (auto& b : bs) w.str(b);
  *out_len = w.off;
  if (w.overflow) return kEnoBuf;
  return 0;
}

// ---- coder / ring / util helpers for the golden-vector tests ----
int okb_encode_key(const uint8_t* k, size_t klen, uint64_t rev, uint8_t* out,
                   size_t cap, size_t* olen) {
  Bytes ik = oracle::EncodeObjectKey(Bytes((const char*)k, klen), rev);
  if (ik.size() > cap) return kEnoBuf;
  memcpy(out, ik.data(), ik.size());
  *olen = ik.size();
  return 0;
}

int okb_decode_key(const uint8_t* ik, size_t iklen, uint8_t* ukey, size_t cap,
                   size_t* uklen, uint64_t* rev) {
  Bytes u;
  Status st = oracle::DecodeInternalKey(Bytes((const char*)ik, iklen), &u, rev);
  if (st != oracle::OK) return st;
  if (u.size() > cap) return kEnoBuf;
  memcpy(ukey, u.data(), u.size());
  *uklen = u.size();
  return 0;
}

int okb_parse_revision(const uint8_t* rb, size_t n, uint64_t* rev, int* tomb) {
  bool t;
  Status st = oracle::ParseRevision(Bytes((const char*)rb, n), rev, &t);
  *tomb = t;
  return st;
}

int okb_prefix_end(const uint8_t* p, size_t plen, uint8_t* out, size_t cap, size_t* olen) {
  Bytes e = oracle::PrefixEnd(Bytes((const char*)p, plen));
  if (e.size() > cap) return kEnoBuf;
  memcpy(out, e.data(), e.size());
  *olen = e.size();
  return 0;
}

// Standalone Ring check against ring_test.go:61-97 vectors.
int okb_ring_test(int capacity, const uint64_t* add_revs, size_t n, uint64_t find_rev,
                  int* empty, int* high, int* low, uint64_t* oldest, uint64_t* newest,
                  uint64_t* events_out, size_t events_cap, size_t* events_n) {
  oracle::Ring r(capacity);
  for (size_t i = 0; i < n; ++i) {
    Event e; e.type = Event::CREATE; e.revision = add_revs[i]; e.kv_revision = add_revs[i];
    r.Add(e);
  }
  auto ret = r.FindEvents(find_rev);
  *empty = ret.empty; *high = ret.high; *low = ret.low;
  *oldest = ret.empty ? 0 : ret.oldest.revision;
  *newest = ret.empty ? 0 : ret.newest.revision;
  if (ret.events.size() > events_cap) return kEnoBuf;
  for (size_t i = 0; i < ret.events.size(); ++i) events_out[i] = ret.events[i].revision;
  *events_n = ret.events.size();
  return 0;
}

// ---- bench support (bench.py cpu_baseline leg ONLY) ----

// Fast-path bulk insert producing exactly the state n serial Creates of
// distinct fresh keys would produce (creator/naive.go:96-101 + tso + events).
int okb_bulk_create(void* h, const uint8_t* keys, const uint32_t* klens,
                    const uint8_t* vals, const uint32_t* vlens, size_t n) {
  Backend* b = (Backend*)h;
  const uint8_t* kp = keys;
  const uint8_t* vp = vals;
  for (size_t i = 0; i < n; ++i) {
    uint64_t hr; int succ;
    okb_create(h, kp, klens[i], vp, vlens[i], &hr, &succ);
    if (!succ) return 1;
    kp += klens[i];
    vp += vlens[i];
  }
  (void)b;
  return 0;
}

// Timed multithreaded Range sweep (mirrors the reference's concurrent scan
// workers, scanner.go:264-286). Queries: packed (slen,elen:u32, rev:u64,
// limit:u64, start, end). Returns total winners via *total; wall seconds via
// *secs. Read-only (compact=false): safe to run concurrently.
int okb_bench_range(void* h, const uint8_t* qbuf, size_t nq, int threads,
                    unsigned long long* total, double* secs) {
  Backend* b = (Backend*)h;
  struct Q { Bytes s, e; uint64_t rev; int64_t limit; };
  std::vector<Q> qs;
  qs.reserve(nq);
  const uint8_t* p = qbuf;
  for (size_t i = 0; i < nq; ++i) {
    uint32_t slen, elen; uint64_t rev, limit;
    memcpy(&slen, p, 4); p += 4;
    memcpy(&elen, p, 4); p += 4;
    memcpy(&rev, p, 8); p += 8;
    memcpy(&limit, p, 8); p += 8;
    Q q; q.s.assign((const char*)p, slen); p += slen;
    q.e.assign((const char*)p, elen); p += elen;
    q.rev = rev; q.limit = (int64_t)limit;
    qs.push_back(std::move(q));
  }
  std::atomic<unsigned long long> tot{0};
  auto t0 = std::chrono::steady_clock::now();
  std::vector<std::thread> ts;
  int T = threads < 1 ? 1 : threads;
  for (int t = 0; t < T; ++t) {
    ts.emplace_back([&, t]() {
      unsigned long long local = 0;
      for (size_t i = t; i < qs.size(); i += T) {
        Status st;
        auto r = b->List(qs[i].s, qs[i].e, qs[i].rev, qs[i].limit, &st);
        local += r.kvs.size();
      }
      tot += local;
    });
  }
  for (auto& th : ts) th.join();
  auto t1 = std::chrono::steady_clock::now();
  *total = tot.load();
  *secs = std::chrono::duration<double>(t1 - t0).count();
  return 0;
}

}  // extern "C"
