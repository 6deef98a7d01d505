// kubebrain_amd/csrc/slab_dev.h — host-visible interface of the HBM slab
// engine (implemented in slab.hip, gfx950 kernels). PRODUCT code: no CPU
// fallback; creation fails without a HIP device.
//
// Data model: one sorted columnar run in HBM, ordered by
// (userKey, rev) == the reference's internal-key order
// (coder/normal.go:42-50), plus an append-only value heap.
#pragma once

#include <cstdint>
#include <string>
#include <utility>
#include <vector>

namespace kbslab {

constexpr int KEYW = 96;  // fixed-width zero-padded key column
// Keys LONGER than KEYW (SURVEY §7 hard part (a)): the column holds the
// first 96 bytes; the tail lives in an append-only key-spill heap addressed
// by the per-row `ko` column. Ordering: the zero-padded 96B prefix decides
// every compare except exact 96-byte prefix ties (impossible unless both
// keys are >= 96B, since key bytes are > 0x24 > 0x00), which compare tails.
constexpr int KB_MAX_KEY = 4096;  // validated bound (meta klen is 16-bit)

// meta word bits (per row)
constexpr uint64_t M_SAME_NEXT = 1ull << 0;  // next row has the same user key
constexpr uint64_t M_TOMB = 1ull << 1;       // value == "tombstone" (util.go:28)
constexpr uint64_t M_FLAG9 = 1ull << 2;      // 9-byte revision value (rev.go:26)
constexpr uint64_t M_EVENTS = 1ull << 3;     // key contains "/events/" (TTL)

#ifdef __HIPCC__
#define KB_HD __host__ __device__
#else
#define KB_HD
#endif
KB_HD inline uint64_t meta_make(bool tomb, bool flag9, bool events,
                                uint32_t klen, uint32_t vlen) {
  return (tomb ? M_TOMB : 0) | (flag9 ? M_FLAG9 : 0) | (events ? M_EVENTS : 0) |
         ((uint64_t)(klen & 0xffff) << 16) | ((uint64_t)vlen << 32);
}
KB_HD inline uint32_t meta_klen(uint64_t m) { return (uint32_t)((m >> 16) & 0xffff); }
KB_HD inline uint32_t meta_vlen(uint64_t m) { return (uint32_t)(m >> 32); }

#pragma pack(push, 1)
struct DevRangeQ {
  uint8_t start[KEYW];
  uint8_t end[KEYW];
  // true bound lengths; tails of bounds longer than KEYW live in the
  // per-batch query-tail buffer at start_ko/end_ko
  uint32_t start_klen, end_klen;
  uint64_t start_ko, end_ko;
  uint64_t read_rev;
  // scan begins at the first row with (key,rev) >= (start, start_rev).
  // 0 = inclusive start-of-key (the normal case); UINT64_MAX = strictly
  // after every row of `start` — the exclusive continuation bound chunked
  // List/Stream use (exact even for keys of the full KEYW width, where a
  // key-suffix trick would truncate)
  uint64_t start_rev;
  int64_t cap;        // winner-write cap per query (limit+1); <=0 => unbounded
  int32_t count_only;
  // etcd3 RangeRequest.KeysOnly semantics (an extension: the reference's etcd
  // shim ignores the flag, kv.go:48-67): gather key + mod-revision per winner,
  // no value bytes (header vlen = 0)
  int32_t keys_only;
};
struct DevGetQ {
  uint8_t key[KEYW];
  uint32_t klen, _pad;  // true key length; tail at ko when > KEYW
  uint64_t ko;
  uint64_t read_rev;  // UINT64_MAX for "latest"
};
#pragma pack(pop)

struct RangeResult {
  int64_t written = 0;   // winners materialized (<= cap)
  int64_t total = 0;     // winners seen before stop (exact when unbounded)
  int64_t bytes = 0;     // gathered record bytes for this query
  bool overflow = false; // per-query arena too small; retry with bigger qcap
  // parsed records (filled only in d2h mode): {rev, key, val}
  struct Rec { uint64_t rev; std::string key, val; };
  std::vector<Rec> recs;
};

struct GetResult {
  bool found = false;
  bool tomb = false;
  uint64_t rev = 0;
  uint32_t vlen = 0;  // value length (filled even in meta-only mode)
  std::string val;
};

// sorted delta (memtable flush) — vo entries already ABSOLUTE heap offsets
// (host adds the pre-merge heap_used base) or objRev for rev rows.
struct DeltaRows {
  std::vector<uint8_t> keys;  // m * KEYW (96B zero-padded prefixes)
  std::vector<uint64_t> meta, rev, vo;
  std::vector<uint64_t> ko;   // spill offsets for keys > KEYW (empty = none)
  std::vector<uint8_t> heap;  // new value bytes (4B-aligned records)
  int64_t m = 0;
};

struct DumpRow {
  std::string key;  // user key
  uint64_t rev;
  uint64_t meta;
  std::string val;  // object rows: heap bytes; rev rows: empty (vo=objRev)
  uint64_t vo;
};

struct Perf {
  double scan_ms = 0, gather_ms = 0, get_ms = 0, compact_ms = 0, merge_ms = 0,
         filter_ms = 0, pack_d2h_ms = 0;
  int64_t scan_launches = 0, gather_launches = 0, get_launches = 0,
          merges = 0, compacts = 0, filter_launches = 0;
  int64_t rows_scanned = 0, bytes_gathered = 0, winners = 0,
          filter_events = 0, filter_watchers = 0;
  double dbg_a = 0, dbg_b = 0, dbg_c = 0, dbg_d = 0, dbg_e = 0;
};



class Slab {
 public:
  // device < 0: use current device. Returns nullptr + *err on failure
  // (including "no HIP device": the product path fails loudly).
  static Slab* Create(int64_t max_rows, int64_t heap_cap, int device,
                      std::string* err);
  ~Slab();

  int64_t rows() const;
  int64_t delta_rows() const;
  int64_t delta_capacity() const;
  int64_t heap_used() const;
  // per-query winner cap of the device arena (KB_MAX_CAP); chunked List
  // clamps each chunk's cap to this and continues the frontier loop
  int64_t max_winner_cap() const;

  // merge sorted delta rows straight into the BASE run (GPU merge by ranks;
  // delta rev-rows REPLACE base rev-rows of the same key). Used by tests and
  // bulk paths; the hot write path uses AppendRows/Fold below.
  bool Merge(const DeltaRows& d, std::string* err);

  // append value bytes to the device heap; *off = absolute offset
  bool HeapAppend(const void* p, int64_t len, int64_t* off, std::string* err);
  // append key-tail bytes to the key-spill heap (keys > KEYW); *off absolute
  bool SpillAppend(const void* p, int64_t len, int64_t* off, std::string* err);
  int64_t spill_used() const;
  // merge sorted new rows (vo already absolute) into the DELTA run.
  // known_new_dn >= 0: the caller knows the exact post-merge row count
  // (drops = rev-row replacements it tracked), so the merge runs fully
  // async — no host sync; the next kernel on the stream queues behind it.
  // KB_VALIDATE_DN=1 re-checks the prediction (enabled by the test env).
  bool AppendRows(const uint8_t* keys, const uint64_t* meta, const uint64_t* rev,
                  const uint64_t* vo, const uint64_t* ko, int64_t m,
                  std::string* err, int64_t known_new_dn = -1);
  // fold the delta run into the base run; delta becomes empty
  bool Fold(std::string* err);

  // batched Range (the north-star kernel; scanner worker.run semantics,
  // scanner.go:389-516). d2h=false leaves records in the device arena
  // (bench "value" mode); d2h=true packs + copies + parses them.
  // qtails = concatenated tails of bounds longer than KEYW (may be empty)
  bool RangeBatch(const std::vector<DevRangeQ>& qs, bool d2h,
                  std::vector<RangeResult>* outs, std::string* err,
                  const std::string& qtails = std::string());
  // parse=false: D2H into pinned memory without materializing records
  bool RangeBatchEx(const std::vector<DevRangeQ>& qs, bool d2h, bool parse,
                    std::vector<RangeResult>* outs, std::string* err,
                    const std::string& qtails = std::string());
  // async split: Start launches the scan+gather without syncing, so host
  // work (e.g. the txn leg) overlaps the in-flight kernels; Finish collects.
  // d2h with parse=false runs PIPELINED: the payload copy lands in pinned
  // host memory on a copy stream, overlapping the next batch's kernels —
  // call DrainD2H() before reading wall-clock results.
  bool RangeBatchStart(const std::vector<DevRangeQ>& qs, std::string* err,
                       const std::string& qtails = std::string());
  bool RangeBatchFinish(int nq, bool d2h, bool parse,
                        std::vector<RangeResult>* outs, std::string* err);
  bool DrainD2H(std::string* err);

  // batched MVCC point read (range.go:91-121 reverse-iter semantics).
  // GetBatchEx(values=false) skips the value copy (meta-only: found/rev/
  // tomb) — the device-side revIndex lookup of the batched txn path.
  bool GetBatch(const std::vector<DevGetQ>& qs, std::vector<GetResult>* outs,
                std::string* err,
                const std::string& qtails = std::string());
  bool GetBatchEx(const std::vector<DevGetQ>& qs, bool values,
                  std::vector<GetResult>* outs, std::string* err,
                  const std::string& qtails = std::string());
  // async split: Start launches the lookup; Finish waits ONLY on the
  // lookup's completion event, so kernels launched on the stream AFTER
  // Start (e.g. the range batch) keep running while the host consumes the
  // results. values=false only (meta-only lookups).
  bool GetBatchStart(const std::vector<DevGetQ>& qs, std::string* err,
                     const std::string& qtails = std::string());
  bool GetBatchFinish(int nq, std::vector<GetResult>* outs, std::string* err);

  // compaction mark+sweep over encoded borders (compact.go:55-68 +
  // scanner.go:444-491, 566-591). Bounds are (key96, rev) pairs.
  // timeout_revs[i] is the TTL timeout revision of border pair i — the
  // reference pops one per scanner.Compact scan (scanner.go:147-177), so
  // with >=2 pairs the values differ per pair.
  struct Bound { uint8_t key[KEYW]; uint64_t rev; };
  bool Compact(const std::vector<std::pair<Bound, Bound>>& borders,
               uint64_t compact_rev, const std::vector<uint64_t>& timeout_revs,
               std::string* err);

  // full slab dump for parity diffs (debug; D2H of all columns + used heap)
  bool Dump(std::vector<DumpRow>* rows_out, std::string* err);

  // device-resident event log (north_star: GPU-resident event log): ring of
  // (key96, rev) columns pushed once per event batch; the fan-out filter and
  // catch-up scans read the resident ring. Values/full events stay host-side
  // for materialization at poll time.
  bool EventRingInit(int64_t cap, std::string* err);
  bool EventRingPush(const uint8_t* keys96, const uint64_t* revs,
                     int64_t count, int64_t base_seq, std::string* err);
  // fan-out: bitmap[w][ceil(count/64)] over all registered watchers for the
  // ring span [base_seq, base_seq+count)
  bool WatchFilterRing(int64_t base_seq, int64_t count,
                       std::vector<uint64_t>* bitmap, int64_t* n_watch_slots,
                       std::string* err);
  // catch-up: one prefix/from_rev filter over a resident ring span
  bool WatchCatchup(const uint8_t* pfx96, uint32_t plen, uint64_t from_rev,
                    int64_t base_seq, int64_t count,
                    std::vector<uint64_t>* words_out, std::string* err);
  bool WatcherSet(int64_t slot, const uint8_t* prefix, uint32_t plen,
                  uint64_t from_rev, std::string* err);  // slot grows table
  void WatcherClear(int64_t slot);

  Perf perf;

  struct Impl;
  Impl* p;
};

}  // namespace kbslab
