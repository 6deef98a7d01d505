// kubebrain_amd/csrc/store.h — host side of the MI355X-native MVCC store.
// Mirrors the reference's backend.Backend surface (pkg/backend/backend.go:44-84)
// over the HBM slab (slab_dev.h). PRODUCT code — never touches oracle/.
//
// Write path: single-writer; revIndex (revision-row contents,
// the CAS target) + memtable (rows newer than the slab) on the host; reads
// merge GPU winners with the bounded memtable.
#pragma once

#include <cstdint>
#include <deque>
#include <map>
#include <memory>
#include <mutex>
#include <set>
#include <string>
#include <unordered_map>
#include <unordered_set>
#include <vector>

#include "slab_dev.h"

namespace kbstore {

using Bytes = std::string;

enum Status : int32_t {  // == include/kb_slab.h kb_status
  OK = 0, NOTFOUND = 1, CAS_FAILED = 2, UNCERTAIN = 3, COMPACTED = 4,
  INVALID_ARG = 5, UNSUPPORTED = 6, REV_DRIFT = 7, WATCH_LOW = 8,
  WATCH_EMPTY = 9, WATCH_DROPPED = 10, KEYTOOLONG = 11, BADKEY = 12,
  INTERNAL = 13, NOGPU = 14, NOBUF = 100,
};

struct Event {
  enum Type : int32_t { CREATE = 0, PUT = 1, DELETE = 2 };
  Type type;
  uint64_t revision;
  Bytes kv_key, kv_value;
  uint64_t kv_revision;
};

struct KeyValue { Bytes key, value; uint64_t revision = 0; };
struct GetResponse { uint64_t header_revision = 0; bool has_kv = false; KeyValue kv; };
struct RangeResponse { uint64_t header_revision = 0; std::vector<KeyValue> kvs; bool more = false; };
struct WriteResponse { uint64_t header_revision = 0; bool succeeded = false; bool has_kv = false; KeyValue kv; };
struct CountResponse { uint64_t header_revision = 0; uint64_t count = 0; };

class Store {
 public:
  struct Config {
    Bytes prefix = "/registry";
    std::vector<Bytes> skipped_prefixes;
    int watch_cache_size = 200000;      // historyCapacity (backend.go:39)
    bool enable_etcd_compatibility = true;
    int64_t events_ttl_seconds = 3600;  // eventsTTL (util.go:37)
    int64_t max_rows = 8 << 20;
    int64_t heap_bytes = 2ll << 30;
    int64_t flush_rows = 65536;
    int device = -1;
  };

  static Store* Open(const Config& cfg, std::string* err);
  ~Store();

  WriteResponse Create(const Bytes& key, const Bytes& value, Status* st);
  WriteResponse Update(const Bytes& key, const Bytes& value, uint64_t prevRev, Status* st);
  WriteResponse Delete(const Bytes& key, uint64_t prevRev, Status* st);
  GetResponse Get(const Bytes& key, uint64_t revision, Status* st);
  RangeResponse List(const Bytes& start, const Bytes& end, uint64_t revision,
                     int64_t limit, Status* st);
  CountResponse Count(const Bytes& start, const Bytes& end, Status* st);
  uint64_t Compact(uint64_t revision, Status* st);
  uint64_t GetCurrentRevision();
  void SetCurrentRevision(uint64_t rev);

  // ListByStream (range.go:247-256 + scanner.RangeStream, receiver.go:104-166):
  // handle-based streaming — batches of 300 winners per Next() at a pinned
  // readRev (the reference streams one snapshot; MVCC revision pinning gives
  // the same results unless a compaction intervenes mid-stream, which then
  // surfaces as KB_ECOMPACTED instead of stale rows).
  int64_t StreamOpen(const Bytes& start, const Bytes& end, uint64_t revision,
                     uint64_t* read_rev, Status* st);
  // fills kvs with the next batch (<=300); empty batch == end marker (stream
  // closes itself)
  Status StreamNext(int64_t sid, std::vector<KeyValue>* kvs);
  void StreamClose(int64_t sid);
  // GetPartitions (range.go:208-245; single partition like badger.go:52-54)
  std::vector<Bytes> GetPartitions(const Bytes& start, const Bytes& end,
                                   uint64_t* header_rev);

  int64_t Watch(const Bytes& prefix, uint64_t revision, Status* st);
  std::vector<Event> WatchPoll(int64_t wid, Status* st);
  // wire-direct poll (kb_watch_poll format), no intermediate Event copies
  Status WatchPollWire(int64_t wid, uint8_t* out, size_t cap, size_t* out_len);
  // non-destructive on overflow: if the serialized size exceeds max_bytes,
  // returns NOBUF (need_bytes = required size) with the queue intact
  std::vector<Event> WatchPollLimited(int64_t wid, size_t max_bytes,
                                      size_t* need_bytes, Status* st);
  void WatchCancel(int64_t wid);

  void ClockAdvance(int64_t secs);
  void SetSkippedPrefixes(const std::vector<Bytes>& sp) { cfg_.skipped_prefixes = sp; }
  std::vector<Bytes> CompactBorders() const;
  bool Flush(std::string* err);
  // sorted (internal key, value) pairs, byte-diffable vs the oracle dump
  bool DumpStore(std::vector<std::pair<Bytes, Bytes>>* out, std::string* err);
  const std::vector<Event>& EventLog() const { return event_log_; }

  // bench support
  bool BulkCreate(const uint8_t* keys, const uint32_t* klens, const uint8_t* vals,
                  const uint32_t* vlens, size_t n, std::string* err);
  // mode bits: 1 = d2h (pipelined payload copy to pinned host memory),
  // 2 = keys_only (etcd3 KeysOnly semantics: no value bytes gathered)
  bool BenchRange(const uint8_t* qbuf, size_t nq, int mode,
                  unsigned long long* total, double* secs, std::string* err);
  // batched conditional updates (txn.go:249-265 per op); out_revs[i] = new
  // revision on success, 0 on CAS failure
  bool BenchTxn(const uint8_t* tbuf, size_t n, uint64_t* out_revs,
                std::string* err);
  bool BenchDel(const uint8_t* dbuf, size_t n, uint64_t* out_revs,
                std::string* err);
  // one bench step: launch the range batch async, run the txn batch on the
  // host while the kernels are in flight, then collect
  bool BenchStep(const uint8_t* qbuf, size_t nq, const uint8_t* tbuf,
                 size_t ntx, int mode, uint64_t* out_revs,
                 unsigned long long* total, double* secs, std::string* err);
  // wait for in-flight pipelined D2H copies (call before reading wall-clock)
  bool Sync(std::string* err);
  std::string PerfJson();
  void PerfReset();

 private:
  Store() = default;
  // helpers (semantics cites in store.cc)
  Status validateKey(const Bytes& key) const;
  uint64_t deal(uint64_t prevRevision, Status* st);
  uint64_t mustDeal(uint64_t prevRevision);
  Status createInternal(const Bytes& key, const Bytes& value, uint64_t revision);
  void notify(const Bytes& key, const Bytes& val, uint64_t revision,
              uint64_t prevRevision, bool valid, Event::Type type);
  Status get(const Bytes& key, uint64_t revision, Bytes* val, uint64_t* modRev);
  Status checkCompactRace(uint64_t revision);
  void putRow(const Bytes& key, uint64_t rev, const Bytes& val);
  uint64_t stageSpill(const Bytes& key);  // stage a long key's tail
  void putRevRow(const Bytes& key, uint64_t objrev, bool flag9);
  void pumpEvents();  // fan-out pending events via the GPU filter
  struct BatchOp { Bytes key, val; uint64_t prev; };
  bool applyTxnOps(const BatchOp* ops, size_t cn,
                   const std::vector<kbslab::GetResult>& cur,
                   uint64_t* out_revs, std::string* err);
  struct Watcher;
  void releaseSlot(Watcher& w);  // recycle a device watcher slot (idempotent)
  // push pending values + new rows to the device (delta-run merge); folds the
  // delta into the base run past the fold threshold
  bool syncReads(std::string* err);
  bool foldLocked(std::string* err);  // syncReads + fold delta into base
  uint64_t getTimeoutRevision();

  Config cfg_;
  std::recursive_mutex mu_;
  kbslab::Slab* slab_ = nullptr;
  uint64_t committed_ = 0, dealt_ = 0;

  struct RevEntry { uint64_t rev; bool tomb; };
  std::unordered_map<Bytes, RevEntry> revIndex_;
  std::unordered_set<Bytes> tombstoned_;  // keys with flagged rev-rows
  std::unordered_set<Bytes> events_keys_; // keys containing "/events/"

  // rows written since the last device sync: values are
  // staged in heap_pending_ with pre-assigned absolute heap offsets
  struct NewRow { Bytes key; uint64_t rev, meta, vo, ko; };
  std::vector<NewRow> newrows_;
  std::unordered_map<Bytes, size_t> nr_revrow_;  // key -> index of rev-row
  Bytes heap_pending_;
  int64_t heap_base_ = -1;
  Bytes spill_pending_;   // key tails (> 96B) staged for the spill heap
  int64_t spill_base_ = -1;

  bool compact_cell_set_ = false;
  uint64_t compact_cell_ = 0;

  int64_t now_ = 0;
  struct CompactRecord { uint64_t revision; int64_t time; };
  std::deque<CompactRecord> compact_histories_;

  // watch (ring for catch-up, GPU filter for fan-out)
  struct Ring {
    int64_t s = 0, e = 0;
    int l = 0;
    std::vector<Event> arr;
    void init(int cap) { l = cap; arr.resize(cap); }
    void Add(const Event& ev) {
      arr[e % l] = ev;
      if (e == s + (int64_t)l) s++;
      e++;
    }
  } ring_;
  std::vector<Event> event_log_;
  std::vector<Event> pending_;  // not yet fanned out
  // delivery = bitmap references into the global event sequence (the host
  // ring materializes full events at poll time) — mirrors the reference,
  // which fans out shared batch POINTERS and filters/copies in the consumer
  // (watcherhub.go:78-100, watch.go:119-159)
  struct PendRef {
    int64_t base;        // global seq of bit 0
    int32_t count;       // events covered (<= 512)
    uint64_t words[8];
  };
  struct Watcher {
    int64_t slot;  // device slot; -1 once released (releaseSlot)
    Bytes prefix;
    uint64_t from_rev;
    std::deque<PendRef> prefs;
    int64_t pend_events = 0;
    bool dropped = false;
  };
  // materialize a watcher's pending refs from the host ring; false if any
  // ref was overwritten (slow consumer fell behind the ring window)
  bool materializeRefs(Watcher& w, std::vector<Event>* out);
  std::unordered_map<int64_t, Watcher> watchers_;
  std::vector<int64_t> free_slots_;
  int64_t next_slot_ = 0, next_wid_ = 1;
  // slow-consumer bound: a watcher whose pending refs span more than the
  // event ring window cannot be materialized and is dropped. The reference's
  // own bound varies between 10k batches and 3M events (watcherhub.go:30
  // sub-channel of 10000 batches x <=300 events); ours is the ring capacity
  // (watch_cache_size, default 200k events) — inside that envelope.

  // host-side perf
  int64_t ops_create_ = 0, ops_update_ = 0, ops_delete_ = 0, ops_range_ = 0;
  // one-step-deep bench pipeline (BenchStep mode bit2): the previous step's
  // range batch is finished at the NEXT step (or Sync), so its kernels
  // overlap the host's txn apply + next-step prep. Valid only between
  // consecutive BenchStep calls; Sync() drains it.
  int bench_pending_nq_ = -1;
  bool in_bench_step_ = false;
  std::vector<int64_t> bench_pending_limits_;
  bool finishPendingBench(unsigned long long* total, std::string* err);
  int64_t delivered_ = 0;  // watch events enqueued to watchers (fan-out)
  bool keep_event_log_ = true;
  double sync_s_ = 0;
  int64_t sync_n_ = 0;
  std::string fatal_;  // first unrecoverable device error (e.g. slab full)
  std::unordered_set<Bytes> delta_revkeys_;  // keys with rev-rows in the delta run
  struct StreamState {
    Bytes frontier, end;
    uint64_t read_rev;
    bool done;
    bool started = false;  // frontier is a continuation (exclusive key bound)
  };
  std::unordered_map<int64_t, StreamState> streams_;
  int64_t next_sid_ = 1;
};

Bytes EncodeObjectKey(const Bytes& userKey, uint64_t rev);  // coder/normal.go:42-50
Bytes PrefixEnd(const Bytes& prefix);                       // util.go
Bytes U64ToBytes(uint64_t v);

}  // namespace kbstore
