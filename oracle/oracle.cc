// oracle/oracle.cc — TEST INFRASTRUCTURE ONLY (see oracle.h header).
// CPU restatement of kubewharf/kubebrain's MVCC hot path. Reference cites are
// into /root/reference.

#include "oracle.h"

#include <algorithm>
#include <cstring>

namespace oracle {

static const Bytes kMagic("\x57\xfb\x80\x8b", 4);        // coder/normal.go:26
static const char kSplitByte = '$';                       // coder/normal.go:31
static const Bytes kTombstone = "tombstone";              // backend/util.go:28
static const Bytes kEvents = "/events/";                  // backend/util.go:30

Bytes U64ToBytes(uint64_t v) {  // util.go uint64ToBytes (big endian)
  Bytes b(8, '\0');
  for (int i = 7; i >= 0; --i) { b[i] = (char)(v & 0xff); v >>= 8; }
  return b;
}

static uint64_t BytesToU64(const Bytes& b) {  // binary.BigEndian.Uint64
  uint64_t v = 0;
  for (int i = 0; i < 8; ++i) v = (v << 8) | (uint8_t)b[i];
  return v;
}

Bytes EncodeObjectKey(const Bytes& userKey, uint64_t revision) {
  // coder/normal.go:42-50: {magic}{raw_key}{split_key}{revision:u64be}
  Bytes key;
  key.reserve(4 + userKey.size() + 1 + 8);
  key += kMagic;
  key += userKey;
  key += kSplitByte;
  key += U64ToBytes(revision);
  return key;
}

Bytes EncodeRevisionKey(const Bytes& userKey) {  // normal.go:53-55
  return EncodeObjectKey(userKey, 0);
}

Status DecodeInternalKey(const Bytes& ik, Bytes* userKey, uint64_t* revision) {
  // coder/normal.go:58-70
  if (ik.size() < 4 + 1 + 8 || ik.compare(0, 4, kMagic) != 0) return INTERNAL;
  if (ik[ik.size() - 9] != kSplitByte) return INTERNAL;
  *revision = BytesToU64(ik.substr(ik.size() - 8));
  *userKey = ik.substr(4, ik.size() - 4 - 9);
  return OK;
}

Status ParseRevision(const Bytes& rb, uint64_t* rev, bool* isTombstone) {
  // coder/rev.go:32-47
  if (rb.size() == 8) { *rev = BytesToU64(rb); *isTombstone = false; return OK; }
  if (rb.size() == 9) { *rev = BytesToU64(rb); *isTombstone = true; return OK; }
  return INTERNAL;
}

Bytes PrefixEnd(const Bytes& prefix) {  // backend/util.go PrefixEnd
  Bytes end = prefix;
  for (int i = (int)end.size() - 1; i >= 0; --i) {
    if ((uint8_t)end[i] < 0xff) {
      end[i] = (char)((uint8_t)end[i] + 1);
      end.resize(i + 1);
      return end;
    }
  }
  return Bytes("\x00", 1);  // noPrefixEnd
}

// ---- Ring (ring.go) ----
void Ring::Add(const Event& e) {  // ring.go:38-46
  arr_[index(e_)] = e;
  if (e_ == s_ + (int64_t)l_) s_++;
  e_++;
}

Ring::FindRet Ring::FindEvents(uint64_t revision) const {  // ring.go:84-118
  FindRet ret;
  if (isEmpty()) { ret.empty = true; return ret; }
  ret.newest = arr_[index(e_ - 1)];
  ret.oldest = arr_[index(s_)];
  if (revision > ret.newest.revision) { ret.high = true; return ret; }
  if (revision < ret.oldest.revision) { ret.low = true; return ret; }
  // sort.Search: least i in [0, e-s) with arr[idx(s+i)].Revision >= revision
  int64_t n = e_ - s_, lo = 0, hi = n;
  while (lo < hi) {
    int64_t mid = lo + (hi - lo) / 2;
    if (arr_[index(s_ + mid)].revision >= revision) hi = mid; else lo = mid + 1;
  }
  for (int64_t i = lo; i < n; ++i) ret.events.push_back(arr_[index(s_ + i)]);
  return ret;
}

// ---- Backend ----
Backend::Backend(const Config& cfg)
    : cfg_(cfg), ring_(cfg.watch_cache_size > 0 ? cfg.watch_cache_size : 200000) {
  compact_key_ = cfg_.prefix + "/" + "compact_key";  // util.go:60-62
}

void Backend::SetCurrentRevision(uint64_t rev) {
  // tso.Commit semantics (tso.go:60-72)
  committed_rev_ = rev;
  if (deal_rev_ < rev) deal_rev_ = rev;
}

bool Backend::storeGet(const Bytes& k, Bytes* v) const {
  auto it = store_.find(k);
  if (it == store_.end()) return false;
  *v = it->second;
  return true;
}

uint64_t Backend::deal(uint64_t prevRevision, Status* st) {
  // backend.go:190-206
  uint64_t rev = ++deal_rev_;  // tso.Deal (tso.go:52-54)
  if (prevRevision > 0 && rev < prevRevision) { *st = REV_DRIFT; return rev; }
  *st = OK;
  return rev;
}

uint64_t Backend::mustDeal(uint64_t prevRevision) {  // txn.go:139-142
  Status st;
  return deal(prevRevision, &st);
}

Status Backend::createInternal(const Bytes& key, const Bytes& value, uint64_t revision) {
  // creator/naive.go:48-105 over memkv batch semantics (memkv/batch.go:51-69)
  Bytes revisionKey = EncodeRevisionKey(key);
  Bytes objectKey = EncodeObjectKey(key, revision);
  Bytes revisionBytes = U64ToBytes(revision);

  Bytes oldRev;
  if (!storeGet(revisionKey, &oldRev)) {
    // create: PutIfNotExist(revKey) + Put(objKey) succeeds (naive.go:96-101)
    store_[revisionKey] = revisionBytes;
    store_[objectKey] = value;
    return OK;
  }
  // Conflict.Idx==0 carries the existing revision bytes (naive.go:61-64)
  uint64_t prevRevision; bool isTombstone;
  Status pst = ParseRevision(oldRev, &prevRevision, &isTombstone);
  if (pst != OK) return pst;
  if (isTombstone && prevRevision < revision) {
    // recreate over tombstone: CAS(revKey,new,old) + Put (naive.go:85-87,103-105)
    store_[revisionKey] = revisionBytes;
    store_[objectKey] = value;
    return OK;
  }
  return CAS_FAILED;
}

void Backend::notify(const Bytes& key, const Bytes& val, uint64_t revision,
                     uint64_t prevRevision, bool valid, Event::Type type) {
  // txn.go:267-293 (ring-buffer insert) + the serial equivalent of the
  // collector loop backend.go:208-270: writes are serialized here, so events
  // arrive in exactly ascending revision order and are committed in place.
  if (revision == 0) return;  // txn.go:269-273
  SetCurrentRevision(revision);  // backend.go:236 (also on invalid: 229-234)
  if (!valid) return;
  Event e;
  e.type = type;
  e.revision = revision;
  if (type == Event::DELETE) {  // backend.go:240-249
    e.kv_key = key; e.kv_value = val; e.kv_revision = prevRevision;
  } else {
    e.kv_key = key; e.kv_value = val; e.kv_revision = revision;
  }
  ring_.Add(e);          // backend.go:263 watchCache.Add
  event_log_.push_back(e);  // watchChan publish (backend.go:266-268)
}

WriteResponse Backend::Create(const Bytes& key, const Bytes& value, Status* st) {
  // txn.go:33-77 (+ b.create txn.go:64-77)
  WriteResponse resp;
  Status dst;
  uint64_t revision = deal(0, &dst);
  Status err = dst;
  if (err == OK) {
    // ttl handling: "/events/" keys get CreateWithTTL (txn.go:70-75); the
    // engine-level TTL is a no-op for memkv semantics — expiry is done by the
    // compaction TTL sweep (scanner.go:566-591).
    err = createInternal(key, value, revision);
  }
  notify(key, value, revision, 0, err == OK, Event::CREATE);
  if (err == CAS_FAILED) {
    resp.header_revision = revision;
    resp.succeeded = false;
    *st = OK;
    return resp;
  } else if (err != OK) {
    *st = err;
    return resp;
  }
  resp.header_revision = revision;
  resp.succeeded = true;
  *st = OK;
  return resp;
}

Status Backend::getInternalVal(const Bytes& key, uint64_t revision, Bytes* val,
                               uint64_t* modRev) const {
  // range.go:91-121: reverse iter [key@rev -> key@0), limit 1.
  // Reverse iter: start inclusive, end exclusive (badger/iter.go:50-63).
  uint64_t rev = revision == 0 ? UINT64_MAX : revision;
  Bytes startKey = EncodeObjectKey(key, rev);
  Bytes endKey = EncodeObjectKey(key, 0);
  auto it = store_.upper_bound(startKey);
  if (it == store_.begin()) return NOTFOUND;
  --it;
  if (!(it->first > endKey)) return NOTFOUND;  // io.EOF -> ErrKeyNotFound
  Bytes userKey; uint64_t r;
  if (DecodeInternalKey(it->first, &userKey, &r) != OK) return NOTFOUND;
  if (r == 0 || userKey != key) return NOTFOUND;  // range.go:112-118
  *val = it->second;
  *modRev = r;
  return OK;
}

Status Backend::get(const Bytes& key, uint64_t revision, Bytes* val, uint64_t* modRev) const {
  // range.go:82-89
  *modRev = 0;
  Status st = getInternalVal(key, revision, val, modRev);
  if (st == OK && *val == kTombstone) return NOTFOUND;  // tombstone -> not found (modRev kept)
  return st;
}

GetResponse Backend::Get(const Bytes& key, uint64_t revision, Status* st) {
  // range.go:34-74
  GetResponse resp;
  uint64_t curRev = committed_rev_;
  Bytes val; uint64_t modRev = 0;
  Status err = get(key, revision, &val, &modRev);
  if (err == NOTFOUND) {
    resp.header_revision = curRev;  // range.go:49-52
    *st = OK;
    return resp;
  } else if (err != OK) {
    *st = err;
    return resp;
  }
  if (modRev > curRev) curRev = modRev;  // range.go:59-61
  resp.header_revision = curRev;
  resp.has_kv = true;
  resp.kv = KeyValue{key, val, modRev};
  *st = OK;
  return resp;
}

WriteResponse Backend::Update(const Bytes& key, const Bytes& value,
                              uint64_t prevRev, Status* st) {
  // txn.go:193-247
  WriteResponse resp;
  uint64_t curRev = 0;
  Status err;
  if (prevRev == 0) {
    // create path (txn.go:216-218)
    Status dst;
    curRev = deal(0, &dst);
    err = dst == OK ? createInternal(key, value, curRev) : dst;
    notify(key, value, curRev, prevRev, err == OK, Event::CREATE);
  } else {
    // b.update (txn.go:249-265)
    Status dst;
    uint64_t newRevision = deal(prevRev, &dst);
    if (dst != OK) { curRev = 0; err = dst; }
    else {
      Bytes revisionKey = EncodeRevisionKey(key);
      Bytes oldRevisionBytes = U64ToBytes(prevRev);
      Bytes cur;
      // CAS(revKey, new, old): missing key or value mismatch => conflict
      // (memkv/batch.go:72-92; a 9B tombstone-flagged value never equals the
      // 8B expected bytes).
      if (!storeGet(revisionKey, &cur) || cur != oldRevisionBytes) {
        err = CAS_FAILED;
      } else {
        store_[revisionKey] = U64ToBytes(newRevision);
        store_[EncodeObjectKey(key, newRevision)] = value;
        err = OK;
      }
      curRev = newRevision;
    }
    notify(key, value, curRev, prevRev, err == OK, Event::PUT);
  }
  resp.header_revision = curRev;
  resp.succeeded = (err == OK);
  if (err == CAS_FAILED) {
    // txn.go:226-244: return latest value
    Bytes val; uint64_t modRev = 0;
    Status getErr = get(key, 0, &val, &modRev);
    if (getErr != OK) {
      if (getErr == NOTFOUND) { *st = OK; return resp; }  // Kv = nil
      *st = getErr;
      return resp;
    }
    resp.header_revision = std::max(resp.header_revision, modRev);
    resp.has_kv = true;
    resp.kv = KeyValue{key, val, modRev};
    *st = OK;
    return resp;
  } else if (err != OK) {
    *st = err;
    return resp;
  }
  *st = OK;
  return resp;
}

WriteResponse Backend::Delete(const Bytes& key, uint64_t prevRev, Status* st) {
  // txn.go:79-190
  WriteResponse resp;
  uint64_t expectedRevision = prevRev;
  Bytes oldVal; uint64_t modRevision = 0;
  Status err = get(key, 0, &oldVal, &modRevision);
  uint64_t rev;
  KeyValue old;  // KeyVal{} zero value on early error
  if (err != OK) {
    rev = mustDeal(prevRev);  // txn.go:148-151
    notify(key, Bytes(), rev, 0, false, Event::DELETE);
    resp.header_revision = rev;
    resp.succeeded = false;
    if (err == NOTFOUND) { *st = OK; return resp; }  // txn.go:101-103
    *st = err;
    return resp;
  }
  Status dst;
  uint64_t newRevision = deal(prevRev, &dst);
  if (dst != OK) { *st = dst; return resp; }  // txn.go:153-156 (rev=0, no notify)
  old = KeyValue{key, oldVal, modRevision};
  if (expectedRevision > 0 && expectedRevision != modRevision) {
    err = CAS_FAILED;  // txn.go:162-166
  } else {
    if (expectedRevision == 0) expectedRevision = modRevision;  // txn.go:167-169
    if (newRevision <= modRevision) {
      err = INTERNAL;  // txn.go:171-175 (plain error; unreachable serially)
    } else {
      // txn.go:177-186: CAS(revKey, rev:8B + 0x00, expected:8B) + Put(objKey, tombstone)
      Bytes revisionKey = EncodeRevisionKey(key);
      Bytes expectedBytes = U64ToBytes(expectedRevision);
      Bytes cur;
      if (!storeGet(revisionKey, &cur) || cur != expectedBytes) {
        err = CAS_FAILED;
      } else {
        store_[revisionKey] = U64ToBytes(newRevision) + Bytes("\x00", 1);
        store_[EncodeObjectKey(key, newRevision)] = kTombstone;
        err = OK;
      }
    }
  }
  rev = newRevision;
  notify(key, old.value, rev, old.revision, err == OK, Event::DELETE);  // txn.go:96
  resp.header_revision = rev;
  resp.succeeded = (err == OK);
  if (err == CAS_FAILED) {
    // txn.go:104-126
    Bytes val; uint64_t modRev = 0;
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
  } else if (err != OK) {
    *st = err;
    return resp;
  }
  resp.has_kv = true;
  resp.kv = old;  // txn.go:130-136
  *st = OK;
  return resp;
}

Status Backend::checkCompactRace(uint64_t revision, bool compact) {
  // scanner.go:594-626
  if (compact) {
    store_[compact_key_] = U64ToBytes(revision);  // batch.Put(CompactKey, rev)
    return OK;
  }
  Bytes val;
  if (!storeGet(compact_key_, &val)) return OK;  // ErrKeyNotFound -> nil
  uint64_t compactRevision = BytesToU64(val);
  if (compactRevision > revision) return COMPACTED;  // scanner.go:617-621
  return OK;
}

uint64_t Backend::getTimeoutRevision() {
  // scanner.go:147-177 (SupportTTL()==false for memkv semantics)
  CompactRecord prev{0, 0};
  while (!compact_histories_.empty()) {
    CompactRecord head = compact_histories_.front();
    int64_t interval = now_ - head.time;
    if (interval < cfg_.events_ttl_seconds) break;
    compact_histories_.pop_front();
    prev = head;
  }
  return prev.revision;
}

int Backend::scanRange(const Bytes& start, const Bytes& end, uint64_t revision,
                       int64_t limit, bool compact, uint64_t timeoutRevision,
                       std::vector<KeyValue>* out) {
  // worker.run (scanner.go:389-516). Snapshot-by-copy iteration like memkv
  // (memkv/iter.go:53-85) is needed only when compact=true (deletes mutate
  // the live map). The read path iterates live with the same early exit the
  // reference's worker has (receiver.needMore, scanner.go:416) — snapshotting
  // the whole range first would overstate the CPU cost of limited scans over
  // large namespaces (the timed cpu_baseline runs through here).
  std::vector<std::pair<Bytes, Bytes>> snapshot;
  if (compact) {
    for (auto it = store_.lower_bound(start); it != store_.end() && it->first < end; ++it)
      snapshot.emplace_back(it->first, it->second);
  }

  int count = 0;
  Bytes prevUserKey, prevValue;
  uint64_t prevRevision = 0;
  bool havePrev = false;  // distinguishes "" user key from none

  auto needMore = [&]() { return !(limit > 0 && out && (int64_t)out->size() >= limit); };

  bool stopped = false;
  // returns false to stop the scan (receiver.needMore, scanner.go:416)
  auto process = [&](const Bytes& ikey, const Bytes& value) -> bool {
    if (!needMore()) { stopped = true; return false; }
    Bytes curUserKey; uint64_t curRevision;
    if (DecodeInternalKey(ikey, &curUserKey, &curRevision) != OK) return true;  // scanner.go:435-439

    // compactIfExpired (scanner.go:444-447 -> 566-591): BEFORE the rev skip
    if (compact && timeoutRevision != 0 &&
        curUserKey.find(kEvents) != Bytes::npos) {
      if (curRevision == 0) {
        uint64_t rv = BytesToU64(value.substr(0, 8));
        if (rv <= timeoutRevision) { store_.erase(ikey); return true; }
      } else if (curRevision <= timeoutRevision) {
        store_.erase(ikey);
        return true;
      }
    }

    if (curRevision > revision) return true;  // scanner.go:451-453

    if (!havePrev || curUserKey != prevUserKey) {  // scanner.go:457-462
      if (prevRevision > 0 && prevValue != kTombstone) {
        if (out) out->push_back(KeyValue{prevUserKey, prevValue, prevRevision});
        count++;
      }
    } else {
      // multi-version: compact deletes the older version (scanner.go:464-469)
      if (compact && prevRevision > 0)
        store_.erase(EncodeObjectKey(prevUserKey, prevRevision));
    }
    // delete tombstone rows (scanner.go:471-475)
    if (compact && value == kTombstone) store_.erase(ikey);
    // delete 9B-flagged revision rows (scanner.go:477-491)
    if (compact && curRevision == 0 && value.size() == 9) {
      uint64_t objRev = BytesToU64(value.substr(0, 8));
      if (objRev > revision) return true;  // skip gc AND the prev update
      store_.erase(ikey);                  // DelCurrent
    }

    prevRevision = curRevision;
    prevUserKey = curUserKey;
    prevValue = value;
    havePrev = true;
    return true;
  };
  if (compact) {
    for (auto& kvp : snapshot)
      if (!process(kvp.first, kvp.second)) break;
  } else {
    for (auto it = store_.lower_bound(start);
         it != store_.end() && it->first < end; ++it)
      if (!process(it->first, it->second)) break;
  }

  // tail emit (scanner.go:503-509)
  if (!stopped && prevRevision > 0 && prevValue != kTombstone && needMore()) {
    if (out) out->push_back(KeyValue{prevUserKey, prevValue, prevRevision});
    count++;
  }
  return count;
}

RangeResponse Backend::List(const Bytes& start, const Bytes& end,
                            uint64_t revision, int64_t limit, Status* st) {
  // range.go:124-174
  RangeResponse resp;
  if (end.empty()) { *st = INVALID_ARG; return resp; }  // range.go:139-141
  uint64_t reqRevision = revision;
  uint64_t curRevision = committed_rev_;
  if (reqRevision == 0) reqRevision = curRevision;
  if (start >= end) { *st = INVALID_ARG; return resp; }  // range.go:149-151
  Bytes key = EncodeObjectKey(start, 0), rangeEnd = EncodeObjectKey(end, 0);
  int64_t lim = limit;
  if (lim > 0) lim++;  // range.go:154-158
  // scanner.Range (scanner.go:83-119): both limit paths checkCompactRace first
  Status cst = checkCompactRace(reqRevision, false);
  if (cst != OK) { *st = cst; return resp; }
  std::vector<KeyValue> kvs;
  scanRange(key, rangeEnd, reqRevision, lim, false, 0, &kvs);
  resp.header_revision = curRevision;
  if (lim > 0 && (int64_t)kvs.size() > limit) {  // range.go:168-171
    resp.more = true;
    kvs.resize(limit);
  }
  resp.kvs = std::move(kvs);
  *st = OK;
  return resp;
}

CountResponse Backend::Count(const Bytes& start, const Bytes& end, Status* st) {
  // range.go:177-205
  CountResponse resp;
  uint64_t rev = committed_rev_;
  resp.header_revision = rev;
  if (!cfg_.enable_etcd_compatibility) { resp.count = 0; *st = OK; return resp; }
  Bytes key = EncodeObjectKey(start, 0), rangeEnd = EncodeObjectKey(end, 0);
  Status cst = checkCompactRace(rev, false);
  if (cst != OK) { *st = cst; return resp; }
  resp.count = (uint64_t)scanRange(key, rangeEnd, rev, 0, false, 0, nullptr);
  *st = OK;
  return resp;
}

Status Backend::setCompactRecord(uint64_t revision, bool* skip) {
  // compact.go:70-105
  *skip = false;
  Bytes val;
  bool have = storeGet(compact_key_, &val);
  if (have && !val.empty()) {
    uint64_t compactRevision = BytesToU64(val);
    if (compactRevision > revision) {
      // "revision has already been compacted" -> return nil; note the caller
      // STILL runs the border scans with this smaller revision (compact.go:56-67)
      *skip = false;
      return OK;
    }
  }
  store_[compact_key_] = U64ToBytes(revision);  // CAS/PutIfNotExist, serial
  return OK;
}

std::vector<Bytes> Backend::getCompactBorders() const {
  // compact.go:108-127
  std::vector<Bytes> keyPrefixes;
  keyPrefixes.push_back(cfg_.prefix);
  for (auto& p : cfg_.skipped_prefixes) keyPrefixes.push_back(p);
  std::vector<Bytes> borders;
  for (auto key : keyPrefixes) {
    if (key.empty() || key.back() != '/') key += '/';
    borders.push_back(EncodeObjectKey(key, 0));
    borders.push_back(EncodeObjectKey(PrefixEnd(key), 0));
  }
  std::sort(borders.begin(), borders.end());
  return borders;
}

uint64_t Backend::Compact(uint64_t revision, Status* st) {
  // compact.go:31-68 (asyncFifoRetry.MinRevision()==0: no retry queue here —
  // SURVEY.md §2 "Retry queue OUT OF SCOPE", stub per compact.go:37-43)
  uint64_t curRevision = committed_rev_;
  if (revision == 0 || revision > curRevision) revision = curRevision;
  bool skip;
  Status sst = setCompactRecord(revision, &skip);
  if (sst != OK) { *st = sst; return revision; }
  auto borders = getCompactBorders();
  for (size_t i = 0; i + 1 < borders.size(); i += 2) {
    // scanner.Compact (scanner.go:195-198): logCompactHistory + scan(compact)
    compact_histories_.push_back(CompactRecord{revision, now_});
    // scan(): checkCompactRace(compact=true) Puts compact_key (scanner.go:597-603)
    checkCompactRace(revision, true);
    uint64_t timeoutRevision = getTimeoutRevision();
    scanRange(borders[i], borders[i + 1], revision, 0, true, timeoutRevision, nullptr);
  }
  *st = OK;
  return revision;
}

std::vector<Backend::StreamBatch> Backend::ListByStream(
    const Bytes& start, const Bytes& end, uint64_t revision, uint64_t* read_rev,
    Status* st) {
  // range.go:247-256: rev defaults to current; scanner.RangeStream streams the
  // full scan in batches of 300 (receiver.go:118-150)
  std::vector<StreamBatch> out;
  uint64_t rev = revision == 0 ? committed_rev_ : revision;
  *read_rev = rev;
  Status cst = checkCompactRace(rev, false);  // scanner.go:594-626 via scan()
  if (cst != OK) { *st = cst; return out; }
  std::vector<KeyValue> kvs;
  scanRange(EncodeObjectKey(start, 0), EncodeObjectKey(end, 0), rev, 0, false,
            0, &kvs);
  for (size_t i = 0; i < kvs.size(); i += 300) {
    StreamBatch b;
    for (size_t j = i; j < std::min(i + 300, kvs.size()); ++j)
      b.kvs.push_back(kvs[j]);
    out.push_back(std::move(b));
  }
  *st = OK;
  return out;
}

std::vector<Bytes> Backend::GetPartitions(const Bytes& start, const Bytes& end,
                                          uint64_t* header_rev) {
  // range.go:208-245 over the badger single partition (badger.go:52-54):
  // PartitionKeys = [enc(start,0), enc(end,0)]
  *header_rev = committed_rev_;
  return {EncodeObjectKey(start, 0), EncodeObjectKey(end, 0)};
}

int64_t Backend::Watch(const Bytes& prefix, uint64_t revision, Status* st) {
  // watch.go:37-99
  Watcher w;
  w.prefix = prefix;
  w.log_pos = event_log_.size();
  if (revision == 0) {
    w.from_rev = 0;  // processEvents(..., revision=0)
  } else {
    Ring::FindRet ret = ring_.FindEvents(revision);
    if (ret.empty) {
      if (revision > committed_rev_) {
        w.from_rev = revision;  // watch.go:62-66
      } else {
        *st = WATCH_EMPTY;  // watch.go:67-71
        return -1;
      }
    } else if (ret.high) {
      w.from_rev = revision;  // watch.go:74-77
    } else if (ret.low) {
      *st = WATCH_LOW;  // watch.go:79-84
      return -1;
    } else {
      // catch-up (watch.go:86-97)
      uint64_t rev = ret.newest.revision;
      std::vector<Event> events;
      for (auto& e : ret.events)
        if (e.kv_key.compare(0, prefix.size(), prefix) == 0) events.push_back(e);
      uint64_t lastRevision = revision;
      if (!events.empty()) {
        lastRevision = rev + 1;
        w.pending = std::move(events);
      }
      w.from_rev = lastRevision;
    }
  }
  int64_t wid = next_wid_++;
  watchers_[wid] = std::move(w);
  *st = OK;
  return wid;
}

std::vector<Event> Backend::WatchPoll(int64_t wid, Status* st) {
  return WatchPollLimited(wid, SIZE_MAX, nullptr, st);
}

// Same overflow contract as the product's kb_watch_poll: if the serialized
// size (4 + per event 28+klen+vlen) exceeds max_bytes, return NOBUF without
// consuming anything so a retry sees every event.
std::vector<Event> Backend::WatchPollLimited(int64_t wid, size_t max_bytes,
                                             size_t* need_bytes, Status* st) {
  if (need_bytes) *need_bytes = 0;
  auto it = watchers_.find(wid);
  if (it == watchers_.end()) { *st = WATCH_DROPPED; return {}; }
  Watcher& w = it->second;
  std::vector<Event> out;
  for (const Event& e : w.pending) out.push_back(e);
  // processEvents: filterByRevision then filterByPrefix (watch.go:119-133)
  size_t pos = w.log_pos;
  for (; pos < event_log_.size(); ++pos) {
    const Event& e = event_log_[pos];
    if (e.revision < w.from_rev) continue;
    if (e.kv_key.compare(0, w.prefix.size(), w.prefix) != 0) continue;
    out.push_back(e);
  }
  size_t need = 4;
  for (const Event& e : out) need += 28 + e.kv_key.size() + e.kv_value.size();
  if (need_bytes) *need_bytes = need;
  if (need > max_bytes) { *st = NOBUF; return {}; }
  w.pending.clear();
  w.log_pos = pos;
  *st = OK;
  return out;
}

void Backend::WatchCancel(int64_t wid) { watchers_.erase(wid); }

}  // namespace oracle
