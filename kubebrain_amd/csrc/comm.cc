// kubebrain_amd/csrc/comm.cc — cross-shard Range exchange over RCCL/xGMI
// (DESIGN.md §3.4). The key slab shards by namespace hash
// across the GPUs of one node; a Range that spans shards runs the scan
// kernel per shard and merges with ONE exchange step:
//
//   allgather(count, header_rev)  ->  allgather(payload, padded to max)
//   ->  k-way merge of the sorted per-shard runs + global limit cut
//
// mirroring the reference's receiver fork/merge
// (pkg/backend/scanner/scanner.go:269-300): each shard contributes its first
// limit+1 winners in key order; keys are disjoint across shards (hash
// sharding), so the global first limit+1 winners are a subset of the union
// and More = merged_total > limit, exactly the reference's limit+1 trick
// (range.go:154-171).
//
// NCCL/RCCL has no allgatherv; counts are exchanged first and the payload
// allgather is padded to the max — at ~300KB per shard the padding is noise
// against the 7x ~153 GB/s xGMI links. The ncclUniqueId travels out-of-band
// (the caller's bootstrap channel, e.g. torch.distributed gloo or the cgo
// host's own RPC).

#include <hip/hip_runtime.h>
#include <rccl/rccl.h>

#include <algorithm>
#include <cstring>
#include <map>
#include <mutex>
#include <string>
#include <vector>

#include "../../include/kb_slab.h"
#include "store.h"

using kbstore::Bytes;
using kbstore::KeyValue;
using kbstore::Status;
using kbstore::Store;

namespace {

struct Comm {
  ncclComm_t comm = nullptr;
  hipStream_t stream = nullptr;
  int rank = 0, world = 1;
  uint8_t* d_send = nullptr;
  int64_t send_cap = 0;
  uint8_t* d_recv = nullptr;
  int64_t recv_cap = 0;
  unsigned long long* d_meta = nullptr;      // [2]: bytes, header_rev
  unsigned long long* d_meta_all = nullptr;  // [2*world]

  ~Comm() {
    if (d_send) (void)hipFree(d_send);
    if (d_recv) (void)hipFree(d_recv);
    if (d_meta) (void)hipFree(d_meta);
    if (d_meta_all) (void)hipFree(d_meta_all);
    if (comm) (void)ncclCommDestroy(comm);
    if (stream) (void)hipStreamDestroy(stream);
  }
};

std::mutex g_mu;
std::map<kb_store*, Comm*> g_comms;

Comm* comm_of(kb_store* h) {
  std::lock_guard<std::mutex> lk(g_mu);
  auto it = g_comms.find(h);
  return it == g_comms.end() ? nullptr : it->second;
}

// tight wire record inside an exchanged run: {u64 rev; u32 klen; u32 vlen;
// key; val} — no padding (PCIe/xGMI bytes, not kernel loads)
void pack_run(const std::vector<KeyValue>& kvs, std::string* blob) {
  size_t need = 0;
  for (auto& kv : kvs) need += 16 + kv.key.size() + kv.value.size();
  blob->clear();
  blob->reserve(need);
  for (auto& kv : kvs) {
    uint64_t rev = kv.revision;
    uint32_t klen = (uint32_t)kv.key.size(), vlen = (uint32_t)kv.value.size();
    blob->append((const char*)&rev, 8);
    blob->append((const char*)&klen, 4);
    blob->append((const char*)&vlen, 4);
    blob->append(kv.key);
    blob->append(kv.value);
  }
}

struct RunView {
  const uint8_t* p;
  size_t len, off = 0;
  uint64_t rev = 0;
  const uint8_t* key = nullptr;
  uint32_t klen = 0, vlen = 0;
  const uint8_t* val = nullptr;
  bool next() {
    if (off + 16 > len) return false;
    memcpy(&rev, p + off, 8);
    memcpy(&klen, p + off + 8, 4);
    memcpy(&vlen, p + off + 12, 4);
    key = p + off + 16;
    val = key + klen;
    off += 16 + (size_t)klen + vlen;
    return off <= len;
  }
};

// k-way merge of W sorted runs + global limit cut; writes the kb_list wire
// format {u32 n; n x {u64 rev; u32 klen; key; u32 vlen; val}}. Returns the
// merged winner count BEFORE the cut (for More).
int64_t merge_runs(const uint8_t* const* runs, const size_t* lens, int world,
                   int64_t limit, uint8_t* out, size_t cap, size_t* out_len,
                   bool* overflow) {
  std::vector<RunView> v(world);
  std::vector<bool> has(world, false);
  for (int r = 0; r < world; ++r) {
    v[r].p = runs[r];
    v[r].len = lens[r];
    has[r] = v[r].next();
  }
  size_t off = 4;  // count patched at the end
  uint32_t n = 0;
  int64_t total = 0;
  *overflow = false;
  while (true) {
    int best = -1;
    for (int r = 0; r < world; ++r) {  // W <= 8: linear head scan beats a heap
      if (!has[r]) continue;
      if (best < 0) { best = r; continue; }
      int c = memcmp(v[r].key, v[best].key,
                     std::min(v[r].klen, v[best].klen));
      if (c < 0 || (c == 0 && v[r].klen < v[best].klen)) best = r;
    }
    if (best < 0) break;
    RunView& b = v[best];
    total++;
    if (limit <= 0 || total <= limit) {
      size_t need = 16 + (size_t)b.klen + b.vlen;
      if (off + need > cap) {
        *overflow = true;
      } else {
        // kb_list wire: u64 rev; u32 klen; key; u32 vlen; val
        memcpy(out + off, &b.rev, 8);
        memcpy(out + off + 8, &b.klen, 4);
        memcpy(out + off + 12, b.key, b.klen);
        memcpy(out + off + 12 + b.klen, &b.vlen, 4);
        memcpy(out + off + 16 + b.klen, b.val, b.vlen);
        off += need;
        n++;
      }
    }
    has[best] = b.next();
  }
  memcpy(out, &n, 4);
  *out_len = off;
  return total;
}

int nccl_ok(ncclResult_t r) { return r == ncclSuccess; }

}  // namespace

extern "C" {

/* rank 0 generates the ncclUniqueId; travels out-of-band to every rank */
int kb_comm_id(uint8_t* out, size_t cap, size_t* len) {
  *len = NCCL_UNIQUE_ID_BYTES;
  if (cap < NCCL_UNIQUE_ID_BYTES) return KB_ENOBUF;
  ncclUniqueId id;
  if (!nccl_ok(ncclGetUniqueId(&id))) return KB_EINTERNAL;
  memcpy(out, &id, NCCL_UNIQUE_ID_BYTES);
  return KB_OK;
}

/* collective: every rank of the node-wide shard group must call this with
 * the same id (one store per GPU) */
int kb_comm_init(kb_store* h, const uint8_t* id, size_t id_len, int rank,
                 int world) {
  if (!h || id_len != NCCL_UNIQUE_ID_BYTES || rank < 0 || rank >= world)
    return KB_EINVALID;
  {
    std::lock_guard<std::mutex> lk(g_mu);
    if (g_comms.count(h)) return KB_EINVALID;  // already initialized
  }
  Comm* c = new Comm();
  c->rank = rank;
  c->world = world;
  ncclUniqueId nid;
  memcpy(&nid, id, NCCL_UNIQUE_ID_BYTES);
  if (hipStreamCreate(&c->stream) != hipSuccess ||
      hipMalloc(&c->d_meta, 16) != hipSuccess ||
      hipMalloc(&c->d_meta_all, 16ll * world) != hipSuccess ||
      !nccl_ok(ncclCommInitRank(&c->comm, world, nid, rank))) {
    delete c;
    return KB_EINTERNAL;
  }
  std::lock_guard<std::mutex> lk(g_mu);
  g_comms[h] = c;
  return KB_OK;
}

int kb_comm_rank(kb_store* h, int* rank, int* world) {
  Comm* c = comm_of(h);
  if (!c) { *rank = 0; *world = 1; return KB_ENOTFOUND; }
  *rank = c->rank;
  *world = c->world;
  return KB_OK;
}

void kb_comm_free(kb_store* h) {
  Comm* c = nullptr;
  {
    std::lock_guard<std::mutex> lk(g_mu);
    auto it = g_comms.find(h);
    if (it != g_comms.end()) { c = it->second; g_comms.erase(it); }
  }
  delete c;
}

/* Cross-shard Range (configs[3]): local shard scan on the GPU slab, ONE
 * RCCL exchange over xGMI, k-way merge + global limit cut. COLLECTIVE:
 * every rank must call with the same bounds/rev/limit; every rank returns
 * the full merged result. Without kb_comm_init it degrades to the local
 * List (world=1). */
int kb_range_global(kb_store* h, const uint8_t* start, size_t slen,
                    const uint8_t* end, size_t elen, uint64_t rev,
                    int64_t limit, uint8_t* out, size_t cap, size_t* out_len,
                    uint64_t* header_rev, int* more) {
  Store* s = (Store*)h;
  Comm* c = comm_of(h);
  // local leg: first limit+1 winners of this shard (the merged global
  // first limit+1 is a subset of the union; range.go:154-171)
  Status st;
  auto r = s->List(Bytes((const char*)start, slen), Bytes((const char*)end, elen),
                   rev, limit > 0 ? limit + 1 : 0, &st);
  if (st != kbstore::OK) return st;
  std::string blob;
  pack_run(r.kvs, &blob);
  uint64_t hrev = r.header_revision;

  // no communicator: local-only degrade. (A world=1 communicator still runs
  // the full RCCL exchange below — that is how the collective path is
  // rehearsed on a single GPU.)
  if (!c) {
    const uint8_t* runs[1] = {(const uint8_t*)blob.data()};
    size_t lens[1] = {blob.size()};
    bool ovf = false;
    int64_t total = merge_runs(runs, lens, 1, limit, out, cap, out_len, &ovf);
    *header_rev = hrev;
    *more = limit > 0 && total > limit;
    return ovf ? KB_ENOBUF : KB_OK;
  }

  // exchange 1: (bytes, header_rev) per rank
  unsigned long long meta[2] = {(unsigned long long)blob.size(), hrev};
  if (hipMemcpyAsync(c->d_meta, meta, 16, hipMemcpyHostToDevice, c->stream) !=
      hipSuccess)
    return KB_EINTERNAL;
  if (!nccl_ok(ncclAllGather(c->d_meta, c->d_meta_all, 2, ncclUint64, c->comm,
                             c->stream)))
    return KB_EINTERNAL;
  std::vector<unsigned long long> meta_all(2ull * c->world);
  if (hipMemcpyAsync(meta_all.data(), c->d_meta_all, 16ll * c->world,
                     hipMemcpyDeviceToHost, c->stream) != hipSuccess ||
      hipStreamSynchronize(c->stream) != hipSuccess)
    return KB_EINTERNAL;
  int64_t mx = 0;
  for (int rr = 0; rr < c->world; ++rr) {
    mx = std::max(mx, (int64_t)meta_all[2 * rr]);
    hrev = std::max(hrev, (uint64_t)meta_all[2 * rr + 1]);
  }
  mx = (mx + 15) & ~15ll;
  // exchange 2: payload allgather padded to mx
  if (mx > 0) {
    if (mx > c->send_cap) {
      if (c->d_send) (void)hipFree(c->d_send);
      if (hipMalloc(&c->d_send, mx + mx / 2) != hipSuccess) return KB_EINTERNAL;
      c->send_cap = mx + mx / 2;
    }
    if (mx * c->world > c->recv_cap) {
      if (c->d_recv) (void)hipFree(c->d_recv);
      if (hipMalloc(&c->d_recv, (mx + mx / 2) * c->world) != hipSuccess)
        return KB_EINTERNAL;
      c->recv_cap = (mx + mx / 2) * c->world;
    }
    if (!blob.empty() &&
        hipMemcpyAsync(c->d_send, blob.data(), blob.size(),
                       hipMemcpyHostToDevice, c->stream) != hipSuccess)
      return KB_EINTERNAL;
    if (!nccl_ok(ncclAllGather(c->d_send, c->d_recv, mx, ncclUint8, c->comm,
                               c->stream)))
      return KB_EINTERNAL;
  }
  std::vector<std::string> host_runs(c->world);
  std::vector<const uint8_t*> runps(c->world);
  std::vector<size_t> lens(c->world);
  for (int rr = 0; rr < c->world; ++rr) {
    size_t bytes = (size_t)meta_all[2 * rr];
    host_runs[rr].resize(bytes);
    if (bytes &&
        hipMemcpyAsync(host_runs[rr].data(), c->d_recv + (int64_t)rr * mx,
                       bytes, hipMemcpyDeviceToHost, c->stream) != hipSuccess)
      return KB_EINTERNAL;
    runps[rr] = (const uint8_t*)host_runs[rr].data();
    lens[rr] = bytes;
  }
  if (hipStreamSynchronize(c->stream) != hipSuccess) return KB_EINTERNAL;
  bool ovf = false;
  int64_t total =
      merge_runs(runps.data(), lens.data(), c->world, limit, out, cap, out_len, &ovf);
  *header_rev = hrev;
  *more = limit > 0 && total > limit;
  return ovf ? KB_ENOBUF : KB_OK;
}

/* TEST-ONLY: drive the k-way merge + global limit cut without a GPU or a
 * communicator (pins the exchange's merge semantics on CPU; the gloo shard
 * test pins the same algorithm end-to-end). runs = concatenated wire runs,
 * lens[world] their byte lengths. */
int kb_test_merge_runs(const uint8_t* runs_cat, const unsigned long long* lens,
                       int world, long long limit, uint8_t* out, size_t cap,
                       size_t* out_len, int* more) {
  std::vector<const uint8_t*> runps(world);
  std::vector<size_t> ls(world);
  size_t off = 0;
  for (int r = 0; r < world; ++r) {
    runps[r] = runs_cat + off;
    ls[r] = (size_t)lens[r];
    off += ls[r];
  }
  bool ovf = false;
  int64_t total =
      merge_runs(runps.data(), ls.data(), world, limit, out, cap, out_len, &ovf);
  *more = limit > 0 && total > limit;
  return ovf ? KB_ENOBUF : KB_OK;
}

}  // extern "C"
