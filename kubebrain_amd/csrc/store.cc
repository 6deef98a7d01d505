// kubebrain_amd/csrc/store.cc — host MVCC logic over the HBM slab.
// PRODUCT code (no oracle/ dependency, no CPU scan fallback: all range /
// point-read / compact / watch-filter compute runs on the GPU slab; the host
// only owns the write path and the bounded memtable).
// All semantics cites are into /root/reference.

#include "store.h"

#include <algorithm>
#include <chrono>
#include <cstring>

namespace kbstore {

using kbslab::DevGetQ;
using kbslab::DevRangeQ;
using kbslab::KEYW;
using kbslab::M_EVENTS;
using kbslab::M_FLAG9;
using kbslab::M_SAME_NEXT;
using kbslab::M_TOMB;

static const Bytes kTombstone = "tombstone";  // backend/util.go:28
static const Bytes kEvents = "/events/";      // backend/util.go:30

Bytes U64ToBytes(uint64_t v) {
  Bytes b(8, '\0');
  for (int i = 7; i >= 0; --i) { b[i] = (char)(v & 0xff); v >>= 8; }
  return b;
}

Bytes EncodeObjectKey(const Bytes& userKey, uint64_t rev) {
  // coder/normal.go:42-50
  Bytes key;
  key.reserve(4 + userKey.size() + 9);
  key.append("\x57\xfb\x80\x8b", 4);
  key += userKey;
  key += '$';
  key += U64ToBytes(rev);
  return key;
}

Bytes PrefixEnd(const Bytes& prefix) {  // util.go PrefixEnd
  Bytes end = prefix;
  for (int i = (int)end.size() - 1; i >= 0; --i) {
    if ((uint8_t)end[i] < 0xff) {
      end[i] = (char)((uint8_t)end[i] + 1);
      end.resize(i + 1);
      return end;
    }
  }
  return Bytes("\x00", 1);
}

static void pad96(const Bytes& k, uint8_t out[KEYW]) {
  memset(out, 0, KEYW);
  memcpy(out, k.data(), std::min(k.size(), (size_t)KEYW));
}

// fill a query bound: 96B padded prefix + tail into the per-batch qtails
// buffer when the bound is longer than the key column
static void setBound(uint8_t dst96[KEYW], uint32_t* klen, uint64_t* ko,
                     const Bytes& b, std::string* qtails) {
  pad96(b, dst96);
  *klen = (uint32_t)b.size();
  *ko = 0;
  if (b.size() > (size_t)KEYW) {
    *ko = qtails->size();
    qtails->append(b.data() + KEYW, b.size() - KEYW);
  }
}

static int64_t env_i64(const char* name, int64_t dflt) {
  const char* v = getenv(name);
  return v && *v ? atoll(v) : dflt;
}

Store* Store::Open(const Config& cfg_in, std::string* err) {
  Config cfg = cfg_in;
  cfg.max_rows = env_i64("KB_MAX_ROWS", cfg.max_rows);
  cfg.heap_bytes = env_i64("KB_HEAP_BYTES", cfg.heap_bytes);
  cfg.flush_rows = env_i64("KB_FLUSH_ROWS", cfg.flush_rows);
  cfg.device = (int)env_i64("KB_DEVICE", cfg.device);
  kbslab::Slab* slab = kbslab::Slab::Create(cfg.max_rows, cfg.heap_bytes,
                                            cfg.device, err);
  if (!slab) return nullptr;
  Store* s = new Store();
  s->cfg_ = cfg;
  s->slab_ = slab;
  s->ring_.init(cfg.watch_cache_size > 0 ? cfg.watch_cache_size : 200000);
  {
    std::string e2;
    if (!slab->EventRingInit(s->ring_.l, &e2)) {
      if (err) *err = e2;
      delete s;
      return nullptr;
    }
  }
  s->keep_event_log_ = env_i64("KB_EVENT_LOG", 1) != 0;  // tests: on; bench: off
  return s;
}

Store::~Store() { delete slab_; }

Status Store::validateKey(const Bytes& key) const {
  // keys > 96B spill their tail to the key-spill heap; the
  // reference codec imposes no length limit (coder/normal.go:42-50)
  if (key.size() > (size_t)kbslab::KB_MAX_KEY) return KEYTOOLONG;
  for (char c : key)
    if ((uint8_t)c <= 0x24) return BADKEY;  // coder/normal.go:29-31 constraint
  return OK;
}

// Range bounds additionally allow TRAILING 0x00 bytes: enc(k+"\x00"..., 0)
// and the 96B zero-padded (k,0) bound exclude exactly the same rows, so the
// etcd single-key idiom [k, k+"\0") keeps its meaning. Any other byte <= 0x24
// inside a bound would order differently against '$'-terminated internal keys
// (coder/normal.go:29-31) and is rejected loudly.
static Status validateBound(const Bytes& b) {
  if (b.size() > (size_t)kbslab::KB_MAX_KEY) return KEYTOOLONG;
  size_t end = b.size();
  while (end > 0 && b[end - 1] == '\x00') end--;
  for (size_t i = 0; i < end; ++i)
    if ((uint8_t)b[i] <= 0x24) return BADKEY;
  return OK;
}

uint64_t Store::GetCurrentRevision() { return committed_; }

void Store::SetCurrentRevision(uint64_t rev) {
  std::lock_guard<std::recursive_mutex> lk(mu_);
  committed_ = rev;  // tso.Commit (tso.go:60-72)
  if (dealt_ < rev) dealt_ = rev;
}

void Store::ClockAdvance(int64_t secs) { now_ += secs; }

uint64_t Store::deal(uint64_t prevRevision, Status* st) {
  // backend.go:190-206 over naiveTSO (tso.go:52-54)
  uint64_t rev = ++dealt_;
  if (prevRevision > 0 && rev < prevRevision) { *st = REV_DRIFT; return rev; }
  *st = OK;
  return rev;
}

uint64_t Store::mustDeal(uint64_t prevRevision) {  // txn.go:139-142
  Status st;
  return deal(prevRevision, &st);
}

uint64_t Store::stageSpill(const Bytes& key) {
  // tail of a key > 96B goes to the key-spill heap (absolute offset
  // pre-assigned, uploaded by syncReads like the value heap)
  if (key.size() <= (size_t)KEYW) return 0;
  if (spill_base_ < 0) spill_base_ = slab_->spill_used();
  uint64_t off = (uint64_t)(spill_base_ + (int64_t)spill_pending_.size());
  spill_pending_.append(key.data() + KEYW, key.size() - KEYW);
  spill_pending_.resize((spill_pending_.size() + 15) & ~15ull, '\0');
  return off;
}

void Store::putRow(const Bytes& key, uint64_t rev, const Bytes& val) {
  NewRow r;
  r.key = key;
  r.rev = rev;
  bool tomb = (val == kTombstone);
  bool ev = key.find(kEvents) != Bytes::npos;
  r.meta = kbslab::meta_make(tomb, false, ev, (uint32_t)key.size(),
                             (uint32_t)val.size());
  if (heap_base_ < 0) heap_base_ = slab_->heap_used();
  r.vo = (uint64_t)(heap_base_ + (int64_t)heap_pending_.size());
  r.ko = stageSpill(key);
  heap_pending_ += val;
  heap_pending_.resize((heap_pending_.size() + 15) & ~15ull, '\0');
  newrows_.push_back(std::move(r));
}

void Store::putRevRow(const Bytes& key, uint64_t objrev, bool flag9) {
  NewRow r;
  r.key = key;
  r.rev = 0;
  bool ev = key.find(kEvents) != Bytes::npos;
  r.meta = kbslab::meta_make(false, flag9, ev, (uint32_t)key.size(), flag9 ? 9 : 8);
  r.vo = objrev;
  r.ko = stageSpill(key);
  auto it = nr_revrow_.find(key);
  if (it != nr_revrow_.end()) {
    newrows_[it->second] = std::move(r);  // replace in place (unique key@0)
  } else {
    nr_revrow_[key] = newrows_.size();
    newrows_.push_back(std::move(r));
  }
  revIndex_[key] = RevEntry{objrev, flag9};
  if (flag9) tombstoned_.insert(key); else tombstoned_.erase(key);
  if (ev) events_keys_.insert(key);
}

bool Store::syncReads(std::string* err) {
  if (!fatal_.empty()) { if (err) *err = "store failed: " + fatal_; return false; }
  // an uncollected pipelined bench batch shares the range result buffers
  // with every other read path; drain it before any non-BenchStep work
  // (BenchStep defers its own collection deliberately)
  if (bench_pending_nq_ >= 0 && !in_bench_step_ &&
      !finishPendingBench(nullptr, err))
    return false;
  sync_n_++;
  auto t0 = std::chrono::steady_clock::now();
  if (!heap_pending_.empty()) {
    int64_t off = 0;
    if (!slab_->HeapAppend(heap_pending_.data(), (int64_t)heap_pending_.size(),
                           &off, err))
      return false;
    if (off != heap_base_) { if (err) *err = "heap offset drift"; return false; }
    heap_pending_.clear();
    heap_base_ = -1;
  }
  if (!spill_pending_.empty()) {
    int64_t off = 0;
    if (!slab_->SpillAppend(spill_pending_.data(),
                            (int64_t)spill_pending_.size(), &off, err))
      return false;
    if (off != spill_base_) { if (err) *err = "spill offset drift"; return false; }
    spill_pending_.clear();
    spill_base_ = -1;
  }
  if (!newrows_.empty()) {
    std::sort(newrows_.begin(), newrows_.end(),
              [](const NewRow& a, const NewRow& b) {
                int c = a.key.compare(b.key);
                if (c != 0) return c < 0;
                return a.rev < b.rev;
              });
    size_t m = newrows_.size();
    std::vector<uint8_t> keys(m * KEYW);
    std::vector<uint64_t> meta(m), rev(m), vo(m), ko(m);
    for (size_t i = 0; i < m; ++i) {
      pad96(newrows_[i].key, keys.data() + i * KEYW);
      meta[i] = newrows_[i].meta;
      rev[i] = newrows_[i].rev;
      vo[i] = newrows_[i].vo;
      ko[i] = newrows_[i].ko;
    }
    // delta-merge drops happen exactly where an uploaded rev-row's key
    // already has a rev-row in the delta run — tracked host-side, so the
    // merge can run fully async with an exact predicted row count
    if (slab_->delta_rows() + (int64_t)m > slab_->delta_capacity()) {
      if (!slab_->Fold(err)) return false;
      delta_revkeys_.clear();
    }
    int64_t drops = 0;
    for (size_t i = 0; i < m; ++i) {
      if (rev[i] != 0) continue;
      auto ins = delta_revkeys_.insert(newrows_[i].key);
      if (!ins.second) drops++;
    }
    int64_t predicted = slab_->delta_rows() + (int64_t)m - drops;
    if (!slab_->AppendRows(keys.data(), meta.data(), rev.data(), vo.data(),
                           ko.data(), (int64_t)m, err, predicted))
      return false;
    newrows_.clear();
    nr_revrow_.clear();
  }
  if (slab_->delta_rows() >= cfg_.flush_rows) {
    if (!slab_->Fold(err)) return false;
    delta_revkeys_.clear();
  }
  sync_s_ +=
      std::chrono::duration<double>(std::chrono::steady_clock::now() - t0).count();
  return true;
}

bool Store::foldLocked(std::string* err) {
  if (!syncReads(err)) return false;
  if (!slab_->Fold(err)) return false;
  delta_revkeys_.clear();  // the delta-run mirror must track every fold
  return true;
}

bool Store::Flush(std::string* err) {
  std::lock_guard<std::recursive_mutex> lk(mu_);
  return foldLocked(err);
}

// ---- events / watch -----------------------------------------------------

void Store::notify(const Bytes& key, const Bytes& val, uint64_t revision,
                   uint64_t prevRevision, bool valid, Event::Type type) {
  // txn.go:267-293 + serial collector (backend.go:208-270): writes are
  // serialized, so events commit in ascending revision order in place.
  if (revision == 0) return;
  committed_ = revision;
  if (dealt_ < revision) dealt_ = revision;
  if (!valid) return;
  Event e;
  e.type = type;
  e.revision = revision;
  e.kv_key = key;
  e.kv_value = val;
  e.kv_revision = type == Event::DELETE ? prevRevision : revision;
  if (keep_event_log_) event_log_.push_back(e);
  ring_.Add(e);  // backend.go:263 (ring keeps its own copy)
  pending_.push_back(std::move(e));
  if (pending_.size() >= 300) pumpEvents();  // eventBatchSize (backend.go:41)
}

void Store::pumpEvents() {
  if (pending_.empty()) return;
  const int64_t kBatch = 512;
  // seq of pending_[0]: the host ring and pending_ grow in lockstep (notify)
  const int64_t seq0 = ring_.e - (int64_t)pending_.size();
  std::string err;
  for (size_t b0 = 0; b0 < pending_.size(); b0 += kBatch) {
    int64_t e = std::min((size_t)kBatch, pending_.size() - b0);
    // push the batch into the DEVICE event ring unconditionally: catch-up
    // scans read the resident log even when no watcher is live right now
    std::vector<uint8_t> ekeys((size_t)e * KEYW);
    std::vector<uint64_t> erevs(e);
    for (int64_t j = 0; j < e; ++j) {
      const Event& ev = pending_[b0 + j];
      pad96(ev.kv_key, ekeys.data() + (size_t)j * KEYW);
      erevs[j] = ev.revision;
    }
    if (!slab_->EventRingPush(ekeys.data(), erevs.data(), e, seq0 + (int64_t)b0,
                              &err)) {
      fatal_ = err;
      pending_.clear();
      return;
    }
    bool any_live = false;
    for (auto& [id, w] : watchers_) if (!w.dropped) { any_live = true; break; }
    if (!any_live) continue;
    // fan-out: device ballot filter over the resident ring span; delivery =
    // one bitmap reference per watcher per batch (no per-event copies)
    std::vector<uint64_t> bitmap;
    int64_t W = 0;
    if (!slab_->WatchFilterRing(seq0 + (int64_t)b0, e, &bitmap, &W, &err)) {
      pending_.clear();
      return;
    }
    int64_t words = (e + 63) / 64;
    for (auto& [id, w] : watchers_) {
      if (w.dropped || w.slot < 0 || w.slot >= W) continue;
      PendRef pr{};
      pr.base = seq0 + (int64_t)b0;
      pr.count = (int32_t)e;
      int64_t cnt = 0;
      for (int64_t c = 0; c < words; ++c) {
        uint64_t bits = bitmap[(size_t)(w.slot * words + c)];
        pr.words[c] = bits;
        cnt += __builtin_popcountll(bits);
      }
      if (!cnt) continue;
      w.prefs.push_back(pr);
      w.pend_events += cnt;
      delivered_ += cnt;
      // slow consumer: pending refs about to leave the ring window
      if (!w.prefs.empty() &&
          ring_.e - w.prefs.front().base > (int64_t)ring_.l - kBatch) {
        w.dropped = true;
        w.prefs.clear();
        w.pend_events = 0;
        releaseSlot(w);
      }
    }
  }
  pending_.clear();
}

int64_t Store::StreamOpen(const Bytes& start, const Bytes& end,
                          uint64_t revision, uint64_t* read_rev, Status* st) {
  // range.go:247-256: rev defaults to current; the scan checks the compact
  // race once at open (scanner.go:594-626 via scan())
  std::lock_guard<std::recursive_mutex> lk(mu_);
  Status bv = validateBound(start);
  if (bv == OK) bv = validateBound(end);
  if (bv != OK) { *st = bv; return -1; }
  uint64_t rev = revision == 0 ? committed_ : revision;
  *read_rev = rev;
  Status cst = checkCompactRace(rev);
  if (cst != OK) { *st = cst; return -1; }
  StreamState ss{start, end, rev, false};
  int64_t sid = next_sid_++;
  streams_[sid] = std::move(ss);
  *st = OK;
  return sid;
}

Status Store::StreamNext(int64_t sid, std::vector<KeyValue>* kvs) {
  // rangeStreamBatch = 300 (scanner.go:45); batches carry More=true and the
  // end marker is an empty batch (receiver.go:118-150, scanner.go:131-143)
  std::lock_guard<std::recursive_mutex> lk(mu_);
  kvs->clear();
  auto it = streams_.find(sid);
  if (it == streams_.end()) return INVALID_ARG;
  StreamState& ss = it->second;
  if (ss.done) { streams_.erase(it); return OK; }
  std::string err;
  if (!syncReads(&err)) return INTERNAL;
  Status cst = checkCompactRace(ss.read_rev);
  if (cst != OK) { streams_.erase(it); return cst; }
  // assemble one 300-winner batch; chunked when the device winner arena
  // (max_winner_cap) is smaller than the batch. Continuation = exclusive
  // (key, rev=+inf) bound past the last chunk's final key (exact for
  // full-width keys; see List).
  const int64_t max_cap = slab_->max_winner_cap();
  std::vector<kbslab::RangeResult> outs;
  while ((int64_t)kvs->size() < 300) {
    DevRangeQ q{};
    std::string qtails;
    setBound(q.start, &q.start_klen, &q.start_ko, ss.frontier, &qtails);
    setBound(q.end, &q.end_klen, &q.end_ko, ss.end, &qtails);
    q.read_rev = ss.read_rev;
    q.start_rev = ss.started ? UINT64_MAX : 0;
    q.cap = std::min<int64_t>(300 - (int64_t)kvs->size(), max_cap);
    q.count_only = 0;
    if (!slab_->RangeBatch({q}, true, &outs, &err, qtails)) return INTERNAL;
    kbslab::RangeResult& r = outs[0];
    if (r.overflow) return NOBUF;
    for (auto& rec : r.recs) kvs->push_back(KeyValue{rec.key, rec.val, rec.rev});
    bool more = r.written >= q.cap || r.total > r.written;
    if (!r.recs.empty()) {
      ss.frontier = r.recs.back().key;
      ss.started = true;
    }
    if (!more || r.recs.empty()) { ss.done = true; break; }
  }
  if (kvs->empty()) streams_.erase(it);  // empty == end marker now
  return OK;
}

void Store::StreamClose(int64_t sid) {
  std::lock_guard<std::recursive_mutex> lk(mu_);
  streams_.erase(sid);
}

std::vector<Bytes> Store::GetPartitions(const Bytes& start, const Bytes& end,
                                        uint64_t* header_rev) {
  // range.go:208-245 over a single partition (badger.go:52-54):
  // PartitionKeys = [enc(start,0), enc(end,0)]
  std::lock_guard<std::recursive_mutex> lk(mu_);
  *header_rev = committed_;
  return {EncodeObjectKey(start, 0), EncodeObjectKey(end, 0)};
}

int64_t Store::Watch(const Bytes& prefix, uint64_t revision, Status* st) {
  // watch.go:37-99
  std::lock_guard<std::recursive_mutex> lk(mu_);
  pumpEvents();  // older events go only to older watchers
  Watcher w;
  w.prefix = prefix;
  if (revision == 0) {
    w.from_rev = 0;
  } else {
    // Ring.FindEvents (ring.go:84-118)
    if (ring_.e == 0) {
      if (revision > committed_) w.from_rev = revision;
      else { *st = WATCH_EMPTY; return -1; }
    } else {
      const Event& newest = ring_.arr[(ring_.e - 1) % ring_.l];
      const Event& oldest = ring_.arr[ring_.s % ring_.l];
      if (revision > newest.revision) {
        w.from_rev = revision;  // high
      } else if (revision < oldest.revision) {
        *st = WATCH_LOW;  // watch.go:79-84
        return -1;
      } else {
        int64_t n = ring_.e - ring_.s, lo = 0, hi = n;
        while (lo < hi) {
          int64_t mid = lo + (hi - lo) / 2;
          if (ring_.arr[(ring_.s + mid) % ring_.l].revision >= revision) hi = mid;
          else lo = mid + 1;
        }
        // catch-up (Ring.FindEvents, ring.go:84-118): filter the ring span
        // [s+lo, e) by prefix into bitmap refs. The span is resident in the
        // DEVICE event log (pushed at pump; pumpEvents ran above), so the
        // scan runs there; prefixes longer than the 96B filter column
        // compute exact bits host-side instead (the device test would be a
        // superset). Ring revisions ascend, so rev >= revision holds for
        // the whole span.
        int64_t base = ring_.s + lo, cnt = n - lo;
        int64_t total = 0;
        std::vector<uint64_t> words;
        if (cnt > 0 && prefix.size() <= (size_t)KEYW) {
          uint8_t p96c[KEYW];
          pad96(prefix, p96c);
          std::string err2;
          if (!slab_->WatchCatchup(p96c, (uint32_t)prefix.size(), revision,
                                   base, cnt, &words, &err2)) {
            *st = INTERNAL;
            return -1;
          }
        } else if (cnt > 0) {
          words.assign((size_t)((cnt + 63) / 64), 0);
          for (int64_t i = 0; i < cnt; ++i) {
            const Event& ev = ring_.arr[(base + i) % ring_.l];
            if (ev.kv_key.compare(0, prefix.size(), prefix) == 0)
              words[(size_t)(i >> 6)] |= 1ull << (i & 63);
          }
        }
        for (auto wd : words) total += __builtin_popcountll(wd);
        if (total > 0) {
          for (int64_t c0 = 0; c0 < cnt; c0 += 512) {
            PendRef pr{};
            pr.base = base + c0;
            pr.count = (int32_t)std::min<int64_t>(512, cnt - c0);
            int64_t got = 0;
            for (int64_t c = 0; c < 8 && (c0 >> 6) + c < (int64_t)words.size() &&
                                c * 64 < pr.count; ++c) {
              pr.words[c] = words[(size_t)((c0 >> 6) + c)];
              got += __builtin_popcountll(pr.words[c]);
            }
            if (got) {
              w.prefs.push_back(pr);
              w.pend_events += got;
            }
          }
        }
        // watch.go:91-95
        w.from_rev = total > 0 ? newest.revision + 1 : revision;
      }
    }
  }
  int64_t slot;
  if (!free_slots_.empty()) { slot = free_slots_.back(); free_slots_.pop_back(); }
  else slot = next_slot_++;
  w.slot = slot;
  uint8_t p96[KEYW];
  pad96(prefix, p96);
  std::string err;
  uint32_t plen96 = prefix.size() > (size_t)KEYW ? (uint32_t)KEYW
                                                 : (uint32_t)prefix.size();
  if (!slab_->WatcherSet(slot, p96, plen96, w.from_rev, &err)) {
    free_slots_.push_back(slot);  // the reserved slot must not leak
    *st = INTERNAL;
    return -1;
  }
  int64_t wid = next_wid_++;
  watchers_[wid] = std::move(w);
  *st = OK;
  return wid;
}

void Store::releaseSlot(Watcher& w) {
  // every drop path returns the device slot (slow-consumer drops included),
  // so the watcher table and k_watch_filter's per-event work stay bounded
  // under watcher churn
  if (w.slot < 0) return;
  slab_->WatcherClear(w.slot);
  free_slots_.push_back(w.slot);
  w.slot = -1;
}

std::vector<Event> Store::WatchPoll(int64_t wid, Status* st) {
  return WatchPollLimited(wid, SIZE_MAX, nullptr, st);
}

bool Store::materializeRefs(Watcher& w, std::vector<Event>* out) {
  for (const PendRef& pr : w.prefs) {
    if (pr.base < ring_.e - (int64_t)ring_.l)
      return false;  // ring overwrote the span: slow consumer fell behind
    for (int c = 0; c * 64 < pr.count; ++c) {
      uint64_t bits = pr.words[c];
      while (bits) {
        int j = __builtin_ctzll(bits);
        bits &= bits - 1;
        const Event& ev = ring_.arr[(pr.base + c * 64 + j) % ring_.l];
        // prefixes longer than the 96B device filter column: the bitmap is
        // a superset; apply the exact prefix here
        if (w.prefix.size() > (size_t)KEYW &&
            ev.kv_key.compare(0, w.prefix.size(), w.prefix) != 0)
          continue;
        out->push_back(ev);
      }
    }
  }
  return true;
}

// Drains the watcher's pending refs only if the serialized size (4 + per
// event 28+klen+vlen — the kb_watch_poll wire format) fits max_bytes;
// otherwise returns NOBUF with the refs INTACT so a retry with a larger
// buffer still sees every event (contiguous-revision delivery guarantee).
std::vector<Event> Store::WatchPollLimited(int64_t wid, size_t max_bytes,
                                           size_t* need_bytes, Status* st) {
  std::lock_guard<std::recursive_mutex> lk(mu_);
  pumpEvents();
  if (need_bytes) *need_bytes = 0;
  auto it = watchers_.find(wid);
  if (it == watchers_.end()) { *st = WATCH_DROPPED; return {}; }
  Watcher& w = it->second;
  std::vector<Event> out;
  if (!w.dropped && !materializeRefs(w, &out)) {
    w.dropped = true;  // refs went stale between pumps (slow consumer)
    w.prefs.clear();
    w.pend_events = 0;
  }
  if (w.dropped) {
    *st = WATCH_DROPPED;
    releaseSlot(w);  // no-op if pumpEvents already recycled it
    watchers_.erase(it);
    return {};
  }
  size_t need = 4;
  for (const Event& e : out)
    need += 28 + e.kv_key.size() + e.kv_value.size();
  if (need_bytes) *need_bytes = need;
  if (need > max_bytes) { *st = NOBUF; return {}; }
  w.prefs.clear();
  w.pend_events = 0;
  *st = OK;
  return out;
}

// wire-direct poll: serializes delivered events straight from the ring into
// the caller's buffer (the kb_watch_poll format: u32 count, then per event
// i32 type | u64 rev | u64 kv_rev | str key | str value) — no intermediate
// Event copies, and the NOBUF path sizes without copying. Same delivery
// contract as WatchPollLimited: refs drain only when everything fits.
Status Store::WatchPollWire(int64_t wid, uint8_t* out, size_t cap,
                            size_t* out_len) {
  std::lock_guard<std::recursive_mutex> lk(mu_);
  pumpEvents();
  *out_len = 0;
  auto it = watchers_.find(wid);
  if (it == watchers_.end()) return WATCH_DROPPED;
  Watcher& w = it->second;
  auto stale = [&]() {
    for (const PendRef& pr : w.prefs)
      if (pr.base < ring_.e - (int64_t)ring_.l) return true;
    return false;
  };
  if (!w.dropped && stale()) {
    w.dropped = true;
    w.prefs.clear();
    w.pend_events = 0;
  }
  if (w.dropped) {
    releaseSlot(w);
    watchers_.erase(it);
    return WATCH_DROPPED;
  }
  const bool longpfx = w.prefix.size() > (size_t)KEYW;
  // pass 1: exact count + wire size (no copies)
  size_t need = 4;
  uint32_t count = 0;
  auto each = [&](auto&& fn) {
    for (const PendRef& pr : w.prefs)
      for (int c = 0; c * 64 < pr.count; ++c) {
        uint64_t bits = pr.words[c];
        while (bits) {
          int j = __builtin_ctzll(bits);
          bits &= bits - 1;
          const Event& ev = ring_.arr[(pr.base + c * 64 + j) % ring_.l];
          if (longpfx &&
              ev.kv_key.compare(0, w.prefix.size(), w.prefix) != 0)
            continue;
          fn(ev);
        }
      }
  };
  each([&](const Event& ev) {
    need += 28 + ev.kv_key.size() + ev.kv_value.size();
    ++count;
  });
  *out_len = need;
  if (need > cap) return NOBUF;  // refs intact: retry with a larger buffer
  // pass 2: serialize straight from the ring
  uint8_t* p = out;
  auto put32 = [&](uint32_t v) { memcpy(p, &v, 4); p += 4; };
  auto put64 = [&](uint64_t v) { memcpy(p, &v, 8); p += 8; };
  put32(count);
  each([&](const Event& ev) {
    put32((uint32_t)ev.type);
    put64(ev.revision);
    put64(ev.kv_revision);
    put32((uint32_t)ev.kv_key.size());
    memcpy(p, ev.kv_key.data(), ev.kv_key.size());
    p += ev.kv_key.size();
    put32((uint32_t)ev.kv_value.size());
    memcpy(p, ev.kv_value.data(), ev.kv_value.size());
    p += ev.kv_value.size();
  });
  *out_len = (size_t)(p - out);
  w.prefs.clear();
  w.pend_events = 0;
  return OK;
}

void Store::WatchCancel(int64_t wid) {
  std::lock_guard<std::recursive_mutex> lk(mu_);
  auto it = watchers_.find(wid);
  if (it == watchers_.end()) return;
  releaseSlot(it->second);
  watchers_.erase(it);
}

// ---- txn protocol (txn.go, creator/naive.go) ----------------------------

Status Store::createInternal(const Bytes& key, const Bytes& value, uint64_t revision) {
  // creator/naive.go:48-105 against revIndex (== live revision-row contents)
  auto it = revIndex_.find(key);
  if (it == revIndex_.end()) {
    putRevRow(key, revision, false);
    putRow(key, revision, value);
    return OK;
  }
  uint64_t prevRevision = it->second.rev;
  bool isTombstone = it->second.tomb;
  if (isTombstone && prevRevision < revision) {  // naive.go:85-87
    putRevRow(key, revision, false);
    putRow(key, revision, value);
    return OK;
  }
  return CAS_FAILED;
}

WriteResponse Store::Create(const Bytes& key, const Bytes& value, Status* st) {
  // txn.go:33-77
  std::lock_guard<std::recursive_mutex> lk(mu_);
  WriteResponse resp;
  Status v = validateKey(key);
  if (v != OK) { *st = v; return resp; }
  ops_create_++;
  Status dst;
  uint64_t revision = deal(0, &dst);
  Status err = dst == OK ? createInternal(key, value, revision) : dst;
  notify(key, value, revision, 0, err == OK, Event::CREATE);
  if (newrows_.size() >= 16384) {
    std::string e_;
    if (!syncReads(&e_)) fatal_ = e_;  // surfaces on every later op
  }
  if (err == CAS_FAILED) {
    resp.header_revision = revision;
    *st = OK;
    return resp;
  } else if (err != OK) { *st = err; return resp; }
  resp.header_revision = revision;
  resp.succeeded = true;
  *st = OK;
  return resp;
}

Status Store::get(const Bytes& key, uint64_t revision, Bytes* val, uint64_t* modRev) {
  // range.go:82-121: largest object row (key,rev<=R); tombstone -> NOTFOUND.
  // GPU point read over base+delta runs (range.go:91-121 reverse-iter
  // semantics); pending writes are synced to the device delta run first.
  *modRev = 0;
  std::string err;
  if (!syncReads(&err)) return INTERNAL;
  uint64_t R = revision == 0 ? UINT64_MAX : revision;
  DevGetQ q{};
  std::string qtails;
  pad96(key, q.key);
  q.klen = (uint32_t)key.size();
  q.ko = 0;
  if (key.size() > (size_t)KEYW) {
    qtails.assign(key.data() + KEYW, key.size() - KEYW);
  }
  q.read_rev = R;
  std::vector<kbslab::GetResult> outs;
  if (!slab_->GetBatch({q}, &outs, &err, qtails)) return INTERNAL;
  if (!outs[0].found) return NOTFOUND;
  *modRev = outs[0].rev;
  if (outs[0].tomb) return NOTFOUND;
  *val = outs[0].val;
  return OK;
}

GetResponse Store::Get(const Bytes& key, uint64_t revision, Status* st) {
  // range.go:34-74
  std::lock_guard<std::recursive_mutex> lk(mu_);
  GetResponse resp;
  uint64_t curRev = committed_;
  Bytes val;
  uint64_t modRev = 0;
  Status err = get(key, revision, &val, &modRev);
  if (err == NOTFOUND) { resp.header_revision = curRev; *st = OK; return resp; }
  if (err != OK) { *st = err; return resp; }
  if (modRev > curRev) curRev = modRev;
  resp.header_revision = curRev;
  resp.has_kv = true;
  resp.kv = KeyValue{key, val, modRev};
  *st = OK;
  return resp;
}

WriteResponse Store::Update(const Bytes& key, const Bytes& value,
                            uint64_t prevRev, Status* st) {
  // txn.go:193-265
  std::lock_guard<std::recursive_mutex> lk(mu_);
  WriteResponse resp;
  Status v = validateKey(key);
  if (v != OK) { *st = v; return resp; }
  ops_update_++;
  uint64_t curRev = 0;
  Status err;
  if (prevRev == 0) {
    Status dst;
    curRev = deal(0, &dst);
    err = dst == OK ? createInternal(key, value, curRev) : dst;
    notify(key, value, curRev, prevRev, err == OK, Event::CREATE);
  } else {
    Status dst;
    uint64_t newRevision = deal(prevRev, &dst);
    if (dst != OK) { curRev = 0; err = dst; }
    else {
      // CAS(revKey, new8, old8) (txn.go:255-264; memkv/batch.go:72-92): a
      // 9B tombstone-flagged current value never equals the 8B expectation.
      auto it = revIndex_.find(key);
      if (it == revIndex_.end() || it->second.tomb || it->second.rev != prevRev) {
        err = CAS_FAILED;
      } else {
        putRevRow(key, newRevision, false);
        putRow(key, newRevision, value);
        err = OK;
      }
      curRev = newRevision;
    }
    notify(key, value, curRev, prevRev, err == OK, Event::PUT);
  }
  if (newrows_.size() >= 16384) {
    std::string e_;
    if (!syncReads(&e_)) fatal_ = e_;  // surfaces on every later op
  }
  resp.header_revision = curRev;
  resp.succeeded = (err == OK);
  if (err == CAS_FAILED) {
    Bytes val;
    uint64_t modRev = 0;
    Status getErr = get(key, 0, &val, &modRev);
    if (getErr != OK) {
      if (getErr == NOTFOUND) { *st = OK; return resp; }
      *st = getErr;
      return resp;
    }
    resp.header_revision = std::max(resp.header_revision, modRev);
    resp.has_kv = true;
    resp.kv = KeyValue{key, val, modRev};
    *st = OK;
    return resp;
  } else if (err != OK) { *st = err; return resp; }
  *st = OK;
  return resp;
}

WriteResponse Store::Delete(const Bytes& key, uint64_t prevRev, Status* st) {
  // txn.go:79-190
  std::lock_guard<std::recursive_mutex> lk(mu_);
  WriteResponse resp;
  Status v = validateKey(key);
  if (v != OK) { *st = v; return resp; }
  ops_delete_++;
  uint64_t expectedRevision = prevRev;
  Bytes oldVal;
  uint64_t modRevision = 0;
  Status err = get(key, 0, &oldVal, &modRevision);
  if (err != OK) {
    uint64_t rev = mustDeal(prevRev);  // txn.go:148-151
    notify(key, Bytes(), rev, 0, false, Event::DELETE);
    resp.header_revision = rev;
    if (err == NOTFOUND) { *st = OK; return resp; }
    *st = err;
    return resp;
  }
  Status dst;
  uint64_t newRevision = deal(prevRev, &dst);
  if (dst != OK) { *st = dst; return resp; }
  KeyValue old{key, oldVal, modRevision};
  if (expectedRevision > 0 && expectedRevision != modRevision) {
    err = CAS_FAILED;  // txn.go:162-166
  } else {
    if (expectedRevision == 0) expectedRevision = modRevision;
    if (newRevision <= modRevision) err = INTERNAL;  // txn.go:171-175
    else {
      // CAS(revKey, rev8+0x00, expected8) + Put(objKey, tombstone) (txn.go:177-186)
      auto it = revIndex_.find(key);
      if (it == revIndex_.end() || it->second.tomb || it->second.rev != expectedRevision) {
        err = CAS_FAILED;
      } else {
        putRevRow(key, newRevision, true);
        putRow(key, newRevision, kTombstone);
        err = OK;
      }
    }
  }
  notify(key, old.value, newRevision, old.revision, err == OK, Event::DELETE);
  if (newrows_.size() >= 16384) {
    std::string e_;
    if (!syncReads(&e_)) fatal_ = e_;  // surfaces on every later op
  }
  resp.header_revision = newRevision;
  resp.succeeded = (err == OK);
  if (err == CAS_FAILED) {
    Bytes val;
    uint64_t modRev = 0;
    Status getErr = get(key, 0, &val, &modRev);
    if (getErr != OK) {
      resp.has_kv = true;
      resp.kv = old;
      *st = OK;
      return resp;
    }
    resp.header_revision = std::max(resp.header_revision, modRev);
    resp.has_kv = true;
    resp.kv = KeyValue{key, val, modRev};
    *st = OK;
    return resp;
  } else if (err != OK) { *st = err; return resp; }
  resp.has_kv = true;
  resp.kv = old;
  *st = OK;
  return resp;
}

// ---- range --------------------------------------------------------------

Status Store::checkCompactRace(uint64_t revision) {
  // scanner.go:594-626 (read path)
  if (!compact_cell_set_) return OK;
  if (compact_cell_ > revision) return COMPACTED;
  return OK;
}

RangeResponse Store::List(const Bytes& start, const Bytes& end,
                          uint64_t revision, int64_t limit, Status* st) {
  // range.go:124-174 + scanner rangeWithLimit/worker.run on the GPU slab
  std::lock_guard<std::recursive_mutex> lk(mu_);
  RangeResponse resp;
  if (end.empty()) { *st = INVALID_ARG; return resp; }
  Status bv = validateBound(start);
  if (bv == OK) bv = validateBound(end);
  if (bv != OK) { *st = bv; return resp; }
  uint64_t reqRevision = revision;
  uint64_t curRevision = committed_;
  if (reqRevision == 0) reqRevision = curRevision;
  if (start >= end) { *st = INVALID_ARG; return resp; }
  int64_t lim = limit > 0 ? limit + 1 : 0;  // range.go:154-158
  Status cst = checkCompactRace(reqRevision);
  if (cst != OK) { *st = cst; return resp; }
  ops_range_++;
  std::string err;
  if (!syncReads(&err)) { *st = INTERNAL; return resp; }

  // GPU winners over base+delta (merged device-side), fetched in chunks.
  // Each chunk's cap is clamped to the device winner arena (max_winner_cap);
  // the frontier loop continues until `lim` is filled or the range is
  // exhausted, so limits larger than the arena stay exact. Continuation is an
  // exclusive (key, rev=+inf) bound (DevRangeQ.start_rev = UINT64_MAX):
  // the next chunk resumes strictly after every row of the last winner's key
  // — exact even for keys of the full 96B width.
  const int64_t max_cap = slab_->max_winner_cap();
  const bool tr = getenv("KB_TRACE") && *getenv("KB_TRACE");
  std::vector<kbslab::RangeResult> outs;
  std::vector<KeyValue> kvs;
  bool dev_more = true;
  Bytes dev_frontier = start;
  uint64_t frontier_rev = 0;  // first chunk: inclusive start-of-key
  auto need = [&]() { return lim <= 0 || (int64_t)kvs.size() < lim; };
  int64_t chunk_cap = 0;  // 0 = unbounded; halved on arena overflow
  while (need() && dev_more) {
    DevRangeQ q{};
    std::string qtails;
    setBound(q.start, &q.start_klen, &q.start_ko, dev_frontier, &qtails);
    setBound(q.end, &q.end_klen, &q.end_ko, end, &qtails);
    q.read_rev = reqRevision;
    q.start_rev = frontier_rev;
    q.cap = lim > 0 ? lim - (int64_t)kvs.size() : chunk_cap;
    if (chunk_cap > 0 && (q.cap <= 0 || q.cap > chunk_cap)) q.cap = chunk_cap;
    if (q.cap <= 0 || q.cap > max_cap) q.cap = max_cap;
    q.count_only = 0;
    if (!slab_->RangeBatch({q}, true, &outs, &err, qtails)) { *st = INTERNAL; return resp; }
    kbslab::RangeResult& r = outs[0];
    if (tr) fprintf(stderr, "[trace] List chunk cap=%lld written=%lld total=%lld recs=%zu kvs=%zu\n",
                    (long long)q.cap, (long long)r.written, (long long)r.total, r.recs.size(), kvs.size());
    if (r.overflow) {
      // results exceed the device arena: halve the chunk and continue (large
      // values); a single record larger than the arena is a real limit
      int64_t cur = q.cap > 0 ? q.cap : (r.written > 0 ? r.written : 4096);
      chunk_cap = cur / 2;
      if (chunk_cap < 1) { *st = NOBUF; return resp; }
      continue;
    }
    for (auto& rec : r.recs) {
      if (!need()) break;
      kvs.push_back(KeyValue{rec.key, rec.val, rec.rev});
    }
    // a chunk that filled its (clamped) cap, or saw more winners than it
    // materialized, may have more rows past the frontier
    dev_more = r.written >= q.cap || r.total > r.written;
    if (r.recs.empty()) break;  // no progress => exhausted
    dev_frontier = r.recs.back().key;
    frontier_rev = UINT64_MAX;  // strictly after the last winner's key
  }

  resp.header_revision = curRevision;
  if (lim > 0 && (int64_t)kvs.size() > limit) {  // range.go:168-171
    resp.more = true;
    kvs.resize(limit);
  }
  resp.kvs = std::move(kvs);
  *st = OK;
  return resp;
}

CountResponse Store::Count(const Bytes& start, const Bytes& end, Status* st) {
  // range.go:177-205
  std::lock_guard<std::recursive_mutex> lk(mu_);
  CountResponse resp;
  uint64_t rev = committed_;
  resp.header_revision = rev;
  if (!cfg_.enable_etcd_compatibility) { *st = OK; return resp; }
  Status bv = validateBound(start);
  if (bv == OK) bv = validateBound(end);
  if (bv != OK) { *st = bv; return resp; }
  Status cst = checkCompactRace(rev);
  if (cst != OK) { *st = cst; return resp; }
  std::string err;
  if (!syncReads(&err)) { *st = INTERNAL; return resp; }
  DevRangeQ q{};
  std::string qtails;
  setBound(q.start, &q.start_klen, &q.start_ko, start, &qtails);
  setBound(q.end, &q.end_klen, &q.end_ko, end, &qtails);
  q.read_rev = rev;
  q.cap = 0;
  q.count_only = 1;
  std::vector<kbslab::RangeResult> outs;
  if (!slab_->RangeBatch({q}, false, &outs, &err, qtails)) { *st = INTERNAL; return resp; }
  resp.count = (uint64_t)outs[0].total;
  *st = OK;
  return resp;
}

// ---- compaction ---------------------------------------------------------

uint64_t Store::getTimeoutRevision() {
  // scanner.go:147-177
  CompactRecord prev{0, 0};
  while (!compact_histories_.empty()) {
    CompactRecord head = compact_histories_.front();
    if (now_ - head.time < cfg_.events_ttl_seconds) break;
    compact_histories_.pop_front();
    prev = head;
  }
  return prev.revision;
}

std::vector<Bytes> Store::CompactBorders() const {
  // compact.go:108-127
  std::vector<Bytes> prefixes;
  prefixes.push_back(cfg_.prefix);
  for (auto& p : cfg_.skipped_prefixes) prefixes.push_back(p);
  std::vector<Bytes> borders;
  for (auto key : prefixes) {
    if (key.empty() || key.back() != '/') key += '/';
    borders.push_back(EncodeObjectKey(key, 0));
    borders.push_back(EncodeObjectKey(PrefixEnd(key), 0));
  }
  std::sort(borders.begin(), borders.end());
  return borders;
}

uint64_t Store::Compact(uint64_t revision, Status* st) {
  // compact.go:31-127 (retry-queue MinRevision stub == 0)
  std::lock_guard<std::recursive_mutex> lk(mu_);
  uint64_t curRevision = committed_;
  if (revision == 0 || revision > curRevision) revision = curRevision;
  // setCompactRecord (compact.go:70-105): a larger stored record skips the
  // cell update but NOT the border scans
  if (!(compact_cell_set_ && compact_cell_ > revision)) {
    compact_cell_set_ = true;
    compact_cell_ = revision;
  }
  std::string err;
  if (!foldLocked(&err)) { *st = INTERNAL; return revision; }
  pumpEvents();
  // borders (compact.go:108-127) — user-key-space bounds for the device
  // kernels (CompactBorders() above returns the encoded form for parity)
  std::vector<Bytes> prefixes;
  prefixes.push_back(cfg_.prefix);
  for (auto& p : cfg_.skipped_prefixes) prefixes.push_back(p);
  std::vector<Bytes> borders;
  for (auto key : prefixes) {
    if (key.empty() || key.back() != '/') key += '/';
    borders.push_back(key);
    borders.push_back(PrefixEnd(key));
  }
  std::sort(borders.begin(), borders.end());
  std::vector<std::pair<kbslab::Slab::Bound, kbslab::Slab::Bound>> bpairs;
  std::vector<uint64_t> timeout_revs;
  std::vector<std::pair<Bytes, Bytes>> pair_keys;
  for (size_t i = 0; i + 1 < borders.size(); i += 2) {
    // scanner.Compact: logCompactHistory + checkCompactRace(compact) Put +
    // per-scan timeout revision (scanner.go:147-198, 594-603) — one history
    // record is pushed AND one timeout revision popped PER pair, so the
    // values differ across pairs and each pair's scan must use its own
    compact_histories_.push_back(CompactRecord{revision, now_});
    compact_cell_set_ = true;
    compact_cell_ = revision;
    timeout_revs.push_back(getTimeoutRevision());
    kbslab::Slab::Bound lo{}, hi{};
    pad96(borders[i], lo.key);
    lo.rev = 0;
    pad96(borders[i + 1], hi.key);
    hi.rev = 0;
    bpairs.push_back({lo, hi});
    pair_keys.push_back({borders[i], borders[i + 1]});
  }
  if (!slab_->Compact(bpairs, revision, timeout_revs, &err)) {
    *st = INTERNAL;
    return revision;
  }
  // host mirror of deleted revision rows (flagged <= compactRev; TTL'd events)
  for (auto it = tombstoned_.begin(); it != tombstoned_.end();) {
    auto ri = revIndex_.find(*it);
    if (ri != revIndex_.end() && ri->second.tomb && ri->second.rev <= revision) {
      revIndex_.erase(ri);
      it = tombstoned_.erase(it);
    } else ++it;
  }
  // TTL'd /events/ keys, per pair with that pair's timeout revision
  for (auto it = events_keys_.begin(); it != events_keys_.end();) {
    bool dead = false;
    auto ri = revIndex_.find(*it);
    if (ri != revIndex_.end()) {
      for (size_t i = 0; i < pair_keys.size(); ++i) {
        if (timeout_revs[i] == 0) continue;
        if (*it >= pair_keys[i].first && *it < pair_keys[i].second &&
            ri->second.rev <= timeout_revs[i]) {
          dead = true;
          break;
        }
      }
    }
    if (dead) {
      tombstoned_.erase(*it);
      revIndex_.erase(ri);
      it = events_keys_.erase(it);
    } else ++it;
  }
  *st = OK;
  return revision;
}

// ---- dump / parity ------------------------------------------------------

bool Store::DumpStore(std::vector<std::pair<Bytes, Bytes>>* out, std::string* err) {
  std::lock_guard<std::recursive_mutex> lk(mu_);
  if (!foldLocked(err)) return false;
  std::vector<kbslab::DumpRow> rows;
  if (!slab_->Dump(&rows, err)) return false;
  out->clear();
  out->reserve(rows.size() + 1);
  for (auto& r : rows) {
    Bytes ik = EncodeObjectKey(r.key, r.rev);
    Bytes val;
    if (r.rev == 0) {
      val = U64ToBytes(r.vo);
      if (r.meta & M_FLAG9) val += '\x00';
    } else {
      val = r.val;
    }
    out->push_back({std::move(ik), std::move(val)});
  }
  if (compact_cell_set_) {
    Bytes ck = cfg_.prefix + "/compact_key";
    Bytes cv = U64ToBytes(compact_cell_);
    auto pos = std::lower_bound(out->begin(), out->end(), ck,
                                [](const std::pair<Bytes, Bytes>& a, const Bytes& b) {
                                  return a.first < b;
                                });
    out->insert(pos, {ck, cv});
  }
  return true;
}

// ---- bench --------------------------------------------------------------

bool Store::BulkCreate(const uint8_t* keys, const uint32_t* klens,
                       const uint8_t* vals, const uint32_t* vlens, size_t n,
                       std::string* err) {
  std::lock_guard<std::recursive_mutex> lk(mu_);
  const uint8_t* kp = keys;
  const uint8_t* vp = vals;
  for (size_t i = 0; i < n; ++i) {
    Bytes key((const char*)kp, klens[i]);
    Bytes val((const char*)vp, vlens[i]);
    kp += klens[i];
    vp += vlens[i];
    Status st;
    auto r = Create(key, val, &st);
    if (st != OK || !r.succeeded) {
      if (err) *err = "bulk create failed at " + std::to_string(i) + " key " + key;
      return false;
    }
  }
  return true;
}

// packed bench queries {u32 slen; u32 elen; u64 rev; u64 limit; start; end}
// straight into device query structs (no per-query allocations)
static void parseBenchQueries(const uint8_t* qbuf, size_t nq, uint64_t cur_rev,
                              int mode, std::vector<DevRangeQ>* qall,
                              std::vector<int64_t>* limits,
                              std::string* qtails) {
  qall->assign(nq, DevRangeQ{});
  limits->resize(nq);
  const int keys_only = (mode & 2) ? 1 : 0;
  const uint8_t* p = qbuf;
  for (size_t i = 0; i < nq; ++i) {
    uint32_t slen, elen;
    uint64_t rev, limit;
    memcpy(&slen, p, 4); p += 4;
    memcpy(&elen, p, 4); p += 4;
    memcpy(&rev, p, 8); p += 8;
    memcpy(&limit, p, 8); p += 8;
    DevRangeQ& q = (*qall)[i];
    setBound(q.start, &q.start_klen, &q.start_ko,
             Bytes((const char*)p, slen), qtails); p += slen;
    setBound(q.end, &q.end_klen, &q.end_ko,
             Bytes((const char*)p, elen), qtails); p += elen;
    q.read_rev = rev == 0 ? cur_rev : rev;
    q.start_rev = 0;
    q.cap = (int64_t)limit > 0 ? (int64_t)limit + 1 : 0;
    q.count_only = 0;
    q.keys_only = keys_only;
    (*limits)[i] = (int64_t)limit;
  }
}

bool Store::BenchRange(const uint8_t* qbuf, size_t nq, int mode,
                       unsigned long long* total, double* secs, std::string* err) {
  // the measured hot path: batched List semantics with inputs resident in HBM
  std::lock_guard<std::recursive_mutex> lk(mu_);
  if (!syncReads(err)) return false;
  std::vector<DevRangeQ> qall;
  std::vector<int64_t> limits;
  std::string qtails;
  parseBenchQueries(qbuf, nq, committed_, mode, &qall, &limits, &qtails);
  const bool d2h = (mode & 1) != 0;
  const int64_t kMax = 1024;
  unsigned long long tot = 0;
  auto t0 = std::chrono::steady_clock::now();
  std::vector<kbslab::RangeResult> outs;
  for (size_t b0 = 0; b0 < (size_t)nq; b0 += kMax) {
    size_t bn = std::min((size_t)kMax, (size_t)nq - b0);
    std::vector<DevRangeQ> dq(qall.begin() + b0, qall.begin() + b0 + bn);
    if (!slab_->RangeBatchEx(dq, d2h, /*parse=*/false, &outs, err, qtails)) return false;
    for (size_t j = 0; j < bn; ++j) {
      int64_t lim = limits[b0 + j];
      int64_t w = outs[j].written;
      tot += (unsigned long long)(lim > 0 && w > lim ? lim : w);
    }
  }
  *secs = std::chrono::duration<double>(std::chrono::steady_clock::now() - t0).count();
  *total = tot;
  ops_range_ += (int64_t)nq;
  return true;
}

bool Store::finishPendingBench(unsigned long long* total, std::string* err) {
  unsigned long long tot = 0;
  if (bench_pending_nq_ >= 0) {
    std::vector<kbslab::RangeResult> outs;
    int nq = bench_pending_nq_;
    bench_pending_nq_ = -1;
    if (!slab_->RangeBatchFinish(nq, false, false, &outs, err)) return false;
    for (int j = 0; j < nq; ++j) {
      int64_t lim = bench_pending_limits_[j];
      int64_t w = outs[j].written;
      tot += (unsigned long long)(lim > 0 && w > lim ? lim : w);
    }
  }
  if (total) *total = tot;
  return true;
}

bool Store::Sync(std::string* err) {
  std::lock_guard<std::recursive_mutex> lk(mu_);
  if (!finishPendingBench(nullptr, err)) return false;
  return slab_->DrainD2H(err);
}

bool Store::BenchStep(const uint8_t* qbuf, size_t nq, const uint8_t* tbuf,
                      size_t ntx, int mode, uint64_t* out_revs,
                      unsigned long long* total, double* secs,
                      std::string* err) {
  std::lock_guard<std::recursive_mutex> lk(mu_);
  const bool pipe = (mode & 4) != 0;
  const bool d2h = (mode & 1) != 0;
  if (pipe && d2h) { if (err) *err = "pipelined bench step excludes d2h"; return false; }
  in_bench_step_ = true;
  struct Reset { bool* f; ~Reset() { *f = false; } } _rst{&in_bench_step_};
  static const bool tr = getenv("KB_TRACE") && *getenv("KB_TRACE");
  auto lt = std::chrono::steady_clock::now();
  auto lap = [&](const char* what) {
    if (!tr) return;
    auto now = std::chrono::steady_clock::now();
    static int n = 0;
    if (++n % 397 < 8)
      fprintf(stderr, "[step] %s %.0f us\n", what,
              std::chrono::duration<double>(now - lt).count() * 1e6);
    lt = now;
  };
  if (!syncReads(err)) return false;
  lap("syncReads");
  std::vector<DevRangeQ> qall;
  std::vector<int64_t> limits;
  std::string qtails;
  parseBenchQueries(qbuf, nq, committed_, mode, &qall, &limits, &qtails);
  lap("parse");
  // pipelined (mode bit2): the previous step's range kernels ran while the
  // host applied its txns and parsed this step; collect them now, return
  // THEIR totals, and leave this step's batch in flight (range reads are
  // snapshot-exact at their read_rev, so deferring collection never changes
  // results)
  unsigned long long prev_tot = 0;
  if (pipe && !finishPendingBench(&prev_tot, err)) return false;
  lap("finish_prev");
  // txn ops parsed up front so the batched CAS lookup (f1) can launch BEFORE
  // the range batch: stream order runs the small lookup first, the host
  // applies the conditional-update protocol against the device results while
  // the range kernels are still in flight (both read the same pre-step
  // snapshot; writes stage host-side until the next syncReads)
  std::vector<BatchOp> tops(ntx);
  bool tuniq = ntx > 0 && ntx <= 1024;
  if (ntx > 0) {
    std::unordered_set<Bytes> seen;
    const uint8_t* p = tbuf;
    for (size_t i = 0; i < ntx; ++i) {
      uint32_t klen, vlen;
      uint64_t prev;
      memcpy(&klen, p, 4); p += 4;
      memcpy(&prev, p, 8); p += 8;
      memcpy(&vlen, p, 4); p += 4;
      tops[i].key.assign((const char*)p, klen); p += klen;
      tops[i].val.assign((const char*)p, vlen); p += vlen;
      tops[i].prev = prev;
      if (tuniq && (!seen.insert(tops[i].key).second ||
                    validateKey(tops[i].key) != OK))
        tuniq = false;
    }
  }
  auto t0 = std::chrono::steady_clock::now();
  if (tuniq) {
    std::vector<DevGetQ> gq(ntx);
    std::string gtails;
    for (size_t i = 0; i < ntx; ++i) {
      gq[i] = DevGetQ{};
      pad96(tops[i].key, gq[i].key);
      gq[i].klen = (uint32_t)tops[i].key.size();
      if (tops[i].key.size() > (size_t)KEYW) {
        gq[i].ko = gtails.size();
        gtails.append(tops[i].key.data() + KEYW, tops[i].key.size() - KEYW);
      }
      gq[i].read_rev = UINT64_MAX;
    }
    if (!slab_->GetBatchStart(gq, err, gtails)) return false;
  }
  lap("txn_parse_getstart");
  if (!slab_->RangeBatchStart(qall, err, qtails)) return false;
  lap("range_start");
  if (ntx > 0) {
    if (tuniq) {
      std::vector<kbslab::GetResult> cur;
      if (!slab_->GetBatchFinish((int)ntx, &cur, err)) return false;
      lap("get_finish");
      if (!applyTxnOps(tops.data(), ntx, cur, out_revs, err)) return false;
      lap("apply_txn");
    } else {
      // duplicate keys: serial reference protocol (overlapped, host-only)
      for (size_t i = 0; i < ntx; ++i) {
        Status st;
        auto r = Update(tops[i].key, tops[i].val, tops[i].prev, &st);
        if (st != OK) {
          if (err) *err = "bench txn failed st=" + std::to_string(st);
          return false;
        }
        out_revs[i] = r.succeeded ? r.header_revision : 0;
      }
    }
  }
  if (pipe) {  // leave this step's batch in flight; report the previous one
    bench_pending_nq_ = (int)nq;
    bench_pending_limits_ = std::move(limits);
    *secs = std::chrono::duration<double>(std::chrono::steady_clock::now() - t0).count();
    *total = prev_tot;
    ops_range_ += (int64_t)nq;
    return true;
  }
  std::vector<kbslab::RangeResult> outs;
  if (!slab_->RangeBatchFinish((int)nq, d2h, false, &outs, err)) return false;
  *secs = std::chrono::duration<double>(std::chrono::steady_clock::now() - t0).count();
  unsigned long long tot = 0;
  for (size_t j = 0; j < nq; ++j) {
    int64_t lim = limits[j];
    int64_t w = outs[j].written;
    tot += (unsigned long long)(lim > 0 && w > lim ? lim : w);
  }
  *total = tot;
  ops_range_ += (int64_t)nq;
  return true;
}

bool Store::BenchTxn(const uint8_t* tbuf, size_t n, uint64_t* out_revs,
                     std::string* err) {
  // f1, GPU-batched conditional updates: the CAS reads of the
  // reference's txn protocol (txn.go:249-265, creator/naive.go:48-105) run
  // as ONE batched device lookup against the slab's revision state
  // (base+delta runs, k_get2 meta-only at R=+inf ≡ the revision-row content
  // — rows and revision-rows are always staged together, so the latest
  // object row's (rev, tomb) IS the revIndex entry). The batched path never
  // reads the host revIndex_ map. Batches with duplicate keys fall back to
  // the serial protocol (within-batch CAS chains need sequencing).
  // Packed records: {u32 klen; u64 prev_rev; u32 vlen; key; val} x n.
  std::vector<BatchOp> ops(n);
  std::unordered_set<Bytes> seen;
  bool uniq = true;
  {
    const uint8_t* p = tbuf;
    for (size_t i = 0; i < n; ++i) {
      uint32_t klen, vlen;
      uint64_t prev;
      memcpy(&klen, p, 4); p += 4;
      memcpy(&prev, p, 8); p += 8;
      memcpy(&vlen, p, 4); p += 4;
      ops[i].key.assign((const char*)p, klen); p += klen;
      ops[i].val.assign((const char*)p, vlen); p += vlen;
      ops[i].prev = prev;
      if (uniq && !seen.insert(ops[i].key).second) uniq = false;
    }
  }
  std::lock_guard<std::recursive_mutex> lk(mu_);
  if (!uniq) {
    for (size_t i = 0; i < n; ++i) {
      Status st;
      auto r = Update(ops[i].key, ops[i].val, ops[i].prev, &st);
      if (st != OK) {
        if (err) *err = "bench txn failed st=" + std::to_string(st);
        return false;
      }
      out_revs[i] = r.succeeded ? r.header_revision : 0;
    }
    return true;
  }
  const size_t kChunk = 1024;  // device query batch bound (KB_MAX_Q)
  for (size_t c0 = 0; c0 < n; c0 += kChunk) {
    size_t cn = std::min(kChunk, n - c0);
    if (!syncReads(err)) return false;  // device revision state current
    std::vector<DevGetQ> qs(cn);
    std::string gtails;
    for (size_t i = 0; i < cn; ++i) {
      Status v = validateKey(ops[c0 + i].key);
      if (v != OK) { if (err) *err = "bench txn bad key"; return false; }
      qs[i] = DevGetQ{};
      pad96(ops[c0 + i].key, qs[i].key);
      qs[i].klen = (uint32_t)ops[c0 + i].key.size();
      if (ops[c0 + i].key.size() > (size_t)KEYW) {
        qs[i].ko = gtails.size();
        gtails.append(ops[c0 + i].key.data() + KEYW,
                      ops[c0 + i].key.size() - KEYW);
      }
      qs[i].read_rev = UINT64_MAX;
    }
    std::vector<kbslab::GetResult> cur;
    if (!slab_->GetBatchEx(qs, /*values=*/false, &cur, err, gtails)) return false;
    if (!applyTxnOps(&ops[c0], cn, cur, out_revs + c0, err)) return false;
  }
  return true;
}

// the host half of the batched conditional-update protocol: consumes the
// device CAS-lookup results in op order (revisions are consumed per op,
// success or not — txn.go:139-142)
bool Store::applyTxnOps(const BatchOp* ops, size_t cn,
                        const std::vector<kbslab::GetResult>& cur,
                        uint64_t* out_revs, std::string* err) {
  for (size_t i = 0; i < cn; ++i) {
    const BatchOp& op = ops[i];
    ops_update_++;
    Status dst;
    uint64_t newRev = deal(op.prev, &dst);
    if (dst != OK) {
      if (err) *err = "bench txn rev drift";
      return false;  // Update surfaces dst as *st (txn.go:139-142)
    }
    bool ok;
    if (op.prev == 0) {
      // create path (creator/naive.go:48-105) against device state
      ok = !cur[i].found || (cur[i].tomb && cur[i].rev < newRev);
    } else {
      // CAS(revKey, new8, old8): the flagged 9B value never matches
      ok = cur[i].found && !cur[i].tomb && cur[i].rev == op.prev;
    }
    if (ok) {
      putRevRow(op.key, newRev, false);
      putRow(op.key, newRev, op.val);
    }
    notify(op.key, op.val, newRev, op.prev, ok,
           op.prev == 0 ? Event::CREATE : Event::PUT);
    out_revs[i] = ok ? newRev : 0;
  }
  return true;
}

bool Store::BenchDel(const uint8_t* dbuf, size_t n, uint64_t* out_revs,
                     std::string* err) {
  // Batched deletes: ONE device lookup prefetches every key's current
  // (value, modRevision), then the reference's delete protocol
  // (txn.go:79-190) runs against the prefetched state. Unique keys per
  // batch; duplicates fall back to the serial path.
  // Packed: {u32 klen; u64 prev_rev; key} x n ; out_revs[i]=rev or 0.
  struct Op { Bytes key; uint64_t prev; };
  std::vector<Op> ops(n);
  std::unordered_set<Bytes> seen;
  bool uniq = true;
  {
    const uint8_t* p = dbuf;
    for (size_t i = 0; i < n; ++i) {
      uint32_t klen;
      uint64_t prev;
      memcpy(&klen, p, 4); p += 4;
      memcpy(&prev, p, 8); p += 8;
      ops[i].key.assign((const char*)p, klen); p += klen;
      ops[i].prev = prev;
      if (uniq && !seen.insert(ops[i].key).second) uniq = false;
    }
  }
  std::lock_guard<std::recursive_mutex> lk(mu_);
  if (!uniq) {
    for (size_t i = 0; i < n; ++i) {
      Status st;
      auto r = Delete(ops[i].key, ops[i].prev, &st);
      if (st != OK) { if (err) *err = "bench del st=" + std::to_string(st); return false; }
      out_revs[i] = r.succeeded ? r.header_revision : 0;
    }
    return true;
  }
  const size_t kChunk = 1024;
  for (size_t c0 = 0; c0 < n; c0 += kChunk) {
    size_t cn = std::min(kChunk, n - c0);
    if (!syncReads(err)) return false;
    std::vector<DevGetQ> qs(cn);
    std::string gtails;
    for (size_t i = 0; i < cn; ++i) {
      Status v = validateKey(ops[c0 + i].key);
      if (v != OK) { if (err) *err = "bench del bad key"; return false; }
      qs[i] = DevGetQ{};
      pad96(ops[c0 + i].key, qs[i].key);
      qs[i].klen = (uint32_t)ops[c0 + i].key.size();
      if (ops[c0 + i].key.size() > (size_t)KEYW) {
        qs[i].ko = gtails.size();
        gtails.append(ops[c0 + i].key.data() + KEYW,
                      ops[c0 + i].key.size() - KEYW);
      }
      qs[i].read_rev = UINT64_MAX;
    }
    std::vector<kbslab::GetResult> cur;
    if (!slab_->GetBatchEx(qs, /*values=*/true, &cur, err, gtails)) return false;
    for (size_t i = 0; i < cn; ++i) {
      const Op& op = ops[c0 + i];
      ops_delete_++;
      if (!cur[i].found || cur[i].tomb) {  // NOTFOUND leg (txn.go:148-151)
        uint64_t rev = mustDeal(op.prev);
        notify(op.key, Bytes(), rev, 0, false, Event::DELETE);
        out_revs[c0 + i] = 0;
        continue;
      }
      uint64_t modRevision = cur[i].rev;
      Status dst;
      uint64_t newRev = deal(op.prev, &dst);
      if (dst != OK) { if (err) *err = "bench del rev drift"; return false; }
      bool ok;
      if (op.prev > 0 && op.prev != modRevision) {
        ok = false;  // txn.go:162-166
      } else {
        // CAS(revKey, rev8+0x00, expected8) with expected == modRevision;
        // device state is consistent by construction (rows + rev-rows are
        // staged together), so the CAS succeeds (txn.go:177-186)
        ok = true;
        putRevRow(op.key, newRev, true);
        putRow(op.key, newRev, kTombstone);
      }
      notify(op.key, cur[i].val, newRev, modRevision, ok, Event::DELETE);
      out_revs[c0 + i] = ok ? newRev : 0;
    }
  }
  return true;
}

std::string Store::PerfJson() {
  const kbslab::Perf& p = slab_->perf;
  char buf[1536];
  snprintf(buf, sizeof(buf),
           "{\"scan_ms\":%.3f,\"gather_ms\":%.3f,\"get_ms\":%.3f,"
           "\"compact_ms\":%.3f,\"merge_ms\":%.3f,\"filter_ms\":%.3f,"
           "\"pack_d2h_ms\":%.3f,\"scan_launches\":%lld,\"rows_scanned\":%lld,"
           "\"bytes_gathered\":%lld,\"winners\":%lld,\"merges\":%lld,"
           "\"compacts\":%lld,\"filter_launches\":%lld,\"filter_events\":%lld,"
           "\"filter_watchers\":%lld,\"slab_rows\":%lld,\"heap_used\":%lld,"
           "\"delivered\":%lld,\"sync_s\":%.3f,\"syncs\":%lld,"
           "\"dbg\":[%.1f,%.1f,%.1f,%.1f,%.1f],"
           "\"ops\":{\"create\":%lld,\"update\":%lld,\"delete\":%lld,"
           "\"range\":%lld}}",
           p.scan_ms, p.gather_ms, p.get_ms, p.compact_ms, p.merge_ms,
           p.filter_ms, p.pack_d2h_ms, (long long)p.scan_launches,
           (long long)p.rows_scanned, (long long)p.bytes_gathered,
           (long long)p.winners, (long long)p.merges, (long long)p.compacts,
           (long long)p.filter_launches, (long long)p.filter_events,
           (long long)p.filter_watchers, (long long)slab_->rows(),
           (long long)slab_->heap_used(), (long long)delivered_,
           sync_s_, (long long)sync_n_,
           p.dbg_a, p.dbg_b, p.dbg_c, p.dbg_d, p.dbg_e,
           (long long)ops_create_,
           (long long)ops_update_, (long long)ops_delete_, (long long)ops_range_);
  return buf;
}

void Store::PerfReset() {
  slab_->perf = kbslab::Perf();
  ops_create_ = ops_update_ = ops_delete_ = ops_range_ = 0;
  sync_s_ = 0; sync_n_ = 0; delivered_ = 0;
}

}  // namespace kbstore
