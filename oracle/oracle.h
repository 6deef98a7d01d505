// oracle/oracle.h
//
// TEST INFRASTRUCTURE ONLY.
// CPU restatement of kubewharf/kubebrain's MVCC hot-path semantics
// (reference tree: /root/reference). This is the parity oracle and the
// bench.py `cpu_baseline` leg. It must NOT be imported, linked, called or
// executed by anything outside tests/, __graft_entry__.smoke() and bench.py's
// cpu_baseline leg. The product path (kubebrain_amd/) never touches this code
// and fails loudly when its HIP extension is missing.
//
// Every function cites the reference file:line it restates. Parity is pinned
// by the golden vectors under tests/golden/ transcribed from the reference's
// own tests (backend_test.go, ring_test.go, coder/normal_test.go,
// compact_test.go, expire_test.go).

#pragma once

#include <cstdint>
#include <deque>
#include <map>
#include <memory>
#include <string>
#include <unordered_map>
#include <vector>

namespace oracle {

// ---- status codes (mirror pkg/storage/interface.go:140-147 + errors.go) ----
enum Status : int32_t {
  OK = 0,
  NOTFOUND = 1,        // storage.ErrKeyNotFound
  CAS_FAILED = 2,      // storage.ErrCASFailed
  UNCERTAIN = 3,       // storage.ErrUncertainResult (unused here; local engine)
  COMPACTED = 4,       // "revision %d less than compact revision %d" (scanner.go:617-621)
  INVALID_ARG = 5,     // nil end / invalid range end (range.go:139-151)
  UNSUPPORTED = 6,
  REV_DRIFT = 7,       // backend.go ErrRevisionDriftBack
  WATCH_LOW = 8,       // "cache event oldest revision ... newer than requested" (watch.go:78-84)
  WATCH_EMPTY = 9,     // "empty cache event" (watch.go:64-71)
  WATCH_DROPPED = 10,  // slow consumer dropped (watcherhub.go:84-94)
  KEYTOOLONG = 11,     // product-only; oracle has no width limit
  BADKEY = 12,
  INTERNAL = 13,
  NOBUF = 100,         // caller buffer too small (== KB_ENOBUF)
};

using Bytes = std::string;  // raw byte strings

// ---- coder (pkg/backend/coder/normal.go:42-70, rev.go:32-47) ----
Bytes EncodeObjectKey(const Bytes& userKey, uint64_t revision);  // normal.go:42-50
Bytes EncodeRevisionKey(const Bytes& userKey);                   // normal.go:53-55
// returns OK and fills userKey/revision, or INTERNAL on bad magic/split byte.
Status DecodeInternalKey(const Bytes& internalKey, Bytes* userKey, uint64_t* revision);  // normal.go:58-70
// rev value bytes: 8B => (rev,false); 9B => (rev,true); else error. rev.go:32-47
Status ParseRevision(const Bytes& revBytes, uint64_t* rev, bool* isTombstone);
Bytes U64ToBytes(uint64_t v);                 // util.go uint64ToBytes
Bytes PrefixEnd(const Bytes& prefix);         // util.go PrefixEnd

// ---- event (kubebrain-client proto semantics pinned by backend_test.go) ----
struct Event {
  enum Type : int32_t { CREATE = 0, PUT = 1, DELETE = 2 };  // proto.Event_EventType
  Type type;
  uint64_t revision;  // event revision
  Bytes kv_key;
  Bytes kv_value;     // DELETE: previous value (backend.go:240-249)
  uint64_t kv_revision;  // DELETE: PrevRevision; else revision (backend.go:240-256)
  bool operator==(const Event& o) const {
    return type == o.type && revision == o.revision && kv_key == o.kv_key &&
           kv_value == o.kv_value && kv_revision == o.kv_revision;
  }
};

struct KeyValue {
  Bytes key;
  Bytes value;
  uint64_t revision = 0;
};

// ---- Ring (pkg/backend/ring.go:24-118) ----
class Ring {
 public:
  explicit Ring(int l) : l_(l), arr_(l) {}
  void Add(const Event& e);          // ring.go:38-46
  void Reset() { s_ = e_ = 0; }      // ring.go:60-64
  struct FindRet {
    bool empty = false, high = false, low = false;
    Event newest, oldest;
    std::vector<Event> events;
  };
  FindRet FindEvents(uint64_t revision) const;  // ring.go:84-118
  int Size() const { return l_; }

 private:
  bool isEmpty() const { return e_ == 0; }
  int index(int64_t i) const { return (int)(i % (int64_t)l_); }
  int64_t s_ = 0, e_ = 0;
  int l_;
  std::vector<Event> arr_;
};

// ---- responses (field-for-field what backend_test.go asserts) ----
struct GetResponse { uint64_t header_revision = 0; bool has_kv = false; KeyValue kv; };
struct RangeResponse { uint64_t header_revision = 0; std::vector<KeyValue> kvs; bool more = false; };
struct WriteResponse {  // Create/Update/Delete responses share this shape
  uint64_t header_revision = 0;
  bool succeeded = false;
  bool has_kv = false;   // Update/Delete CAS-fail & Delete success carry a KV
  KeyValue kv;
};
struct CountResponse { uint64_t header_revision = 0; uint64_t count = 0; };

// ---- Backend: the oracle store ----
// Restates pkg/backend (backend.go, txn.go, range.go, compact.go, watch.go)
// over a memkv-semantics ordered map (pkg/storage/memkv). Single-threaded by
// design: the reference serializes writes (leader-only, kv.go:90-96) and the
// event collector (backend.go:208-270) is a single goroutine, so serial
// execution reproduces the committed behavior exactly.
class Backend {
 public:
  struct Config {
    Bytes prefix = "/registry";
    std::vector<Bytes> skipped_prefixes;       // backend.Config.SkippedPrefixes
    int watch_cache_size = 200000;             // historyCapacity backend.go:39
    bool enable_etcd_compatibility = true;     // gates Count (range.go:186-191)
    int64_t events_ttl_seconds = 3600;         // eventsTTL util.go:37
  };
  explicit Backend(const Config& cfg);

  // -- Backend interface (pkg/backend/backend.go:44-84) --
  WriteResponse Create(const Bytes& key, const Bytes& value, Status* st);   // txn.go:33-77
  WriteResponse Update(const Bytes& key, const Bytes& value, uint64_t prevRev, Status* st);  // txn.go:193-247
  WriteResponse Delete(const Bytes& key, uint64_t prevRev, Status* st);     // txn.go:79-190
  GetResponse Get(const Bytes& key, uint64_t revision, Status* st);         // range.go:34-74
  RangeResponse List(const Bytes& start, const Bytes& end, uint64_t revision,
                     int64_t limit, Status* st);                            // range.go:124-174
  CountResponse Count(const Bytes& start, const Bytes& end, Status* st);    // range.go:177-205
  uint64_t Compact(uint64_t revision, Status* st);   // compact.go:31-68; returns clamped rev
  uint64_t GetCurrentRevision() const { return committed_rev_; }  // tso.GetRevision
  void SetCurrentRevision(uint64_t rev);             // tso Init semantics

  // -- ListByStream (range.go:247-256 + scanner.RangeStream scanner.go:129-145,
  // receiver.go:104-166): batches of 300 (More=true, header=readRev) then an
  // end marker. Returned flattened here; the C ABI chunks it. --
  struct StreamBatch { std::vector<KeyValue> kvs; };
  std::vector<StreamBatch> ListByStream(const Bytes& start, const Bytes& end,
                                        uint64_t revision, uint64_t* read_rev,
                                        Status* st);
  // -- GetPartitions (range.go:208-245; badger single partition badger.go:52-54) --
  std::vector<Bytes> GetPartitions(const Bytes& start, const Bytes& end,
                                   uint64_t* header_rev);

  // -- watch (watch.go:37-159 + watcherhub) --
  // Registers a watcher; catch-up events are queued immediately per
  // watch.go:52-99. Returns watcher id or error in *st.
  int64_t Watch(const Bytes& prefix, uint64_t revision, Status* st);
  // Drain pending events for watcher (delivery batches flattened).
  std::vector<Event> WatchPoll(int64_t wid, Status* st);
  // non-destructive on overflow (same contract as kb_watch_poll): if the
  // serialized size exceeds max_bytes, returns NOBUF with the queue intact
  std::vector<Event> WatchPollLimited(int64_t wid, size_t max_bytes,
                                      size_t* need_bytes, Status* st);
  void WatchCancel(int64_t wid);

  // -- config/test hooks --
  void SetSkippedPrefixes(const std::vector<Bytes>& sp) { cfg_.skipped_prefixes = sp; }
  std::vector<Bytes> CompactBorders() const { return getCompactBorders(); }
  // Advance the TTL clock (scanner compactHistories, scanner.go:147-177).
  void ClockAdvance(int64_t seconds) { now_ += seconds; }
  // Dump the full internal store (sorted internal key -> value) for slab diff.
  const std::map<Bytes, Bytes>& DumpStore() const { return store_; }
  // All valid events ever emitted, in revision order (collector output).
  const std::vector<Event>& EventLog() const { return event_log_; }

 private:
  // storage ops on the memkv-semantics map
  bool storeGet(const Bytes& k, Bytes* v) const;
  // get (range.go:82-121): val/modRev of largest rev <= revision; tombstone -> NOTFOUND(modRev)
  Status get(const Bytes& key, uint64_t revision, Bytes* val, uint64_t* modRev) const;
  Status getInternalVal(const Bytes& key, uint64_t revision, Bytes* val, uint64_t* modRev) const;
  uint64_t deal(uint64_t prevRevision, Status* st);   // backend.go:190-206
  uint64_t mustDeal(uint64_t prevRevision);           // txn.go:139-142
  Status createInternal(const Bytes& key, const Bytes& value, uint64_t revision);  // creator/naive.go:48-105
  void notify(const Bytes& key, const Bytes& val, uint64_t revision,
              uint64_t prevRevision, bool valid, Event::Type type);  // txn.go:267-293 + collector
  Status checkCompactRace(uint64_t revision, bool compact);  // scanner.go:594-626
  Status setCompactRecord(uint64_t revision, bool* skip);    // compact.go:70-105
  uint64_t getTimeoutRevision();                             // scanner.go:147-177
  // the worker.run scan loop (scanner.go:389-516); limit<=0 means unlimited.
  // Returns winner count; appends winners to out (if non-null).
  int scanRange(const Bytes& start, const Bytes& end, uint64_t revision,
                int64_t limit, bool compact, uint64_t timeoutRevision,
                std::vector<KeyValue>* out);
  std::vector<Bytes> getCompactBorders() const;  // compact.go:108-127

  Config cfg_;
  Bytes compact_key_;  // getCompactKey(prefix) util.go:60-62
  std::map<Bytes, Bytes> store_;  // ordered internal keyspace (memkv semantics)
  uint64_t committed_rev_ = 0;    // naiveTSO.committedRevision
  uint64_t deal_rev_ = 0;         // naiveTSO.dealRevision
  Ring ring_;
  std::vector<Event> event_log_;
  int64_t now_ = 0;  // injected clock (seconds)
  struct CompactRecord { uint64_t revision; int64_t time; };
  std::deque<CompactRecord> compact_histories_;  // scanner compactRecordQueue
  struct Watcher {
    Bytes prefix;
    uint64_t from_rev;     // filterByRevision threshold (watch.go:152-158)
    std::vector<Event> pending;
    size_t log_pos;        // position in event_log_ already consumed
    bool dropped = false;
  };
  std::unordered_map<int64_t, Watcher> watchers_;
  int64_t next_wid_ = 1;
};

}  // namespace oracle
