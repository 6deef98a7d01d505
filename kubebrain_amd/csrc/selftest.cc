// kubebrain_amd/csrc/selftest.cc — step-by-step device-engine exerciser used
// to localize GPU faults (run with AMD_SERIALIZE_KERNEL=3). Not part of the
// product API.
#include <algorithm>
#include <cstdio>
#include <cstring>
#include <string>
#include <vector>

#include "slab_dev.h"

using namespace kbslab;

static void pad(const std::string& k, uint8_t out[KEYW]) {
  memset(out, 0, KEYW);
  memcpy(out, k.data(), k.size());
}

#define STEP(name, expr)                                        \
  do {                                                          \
    std::string err;                                            \
    bool ok = (expr);                                           \
    printf("STEP %-24s : %s %s\n", name, ok ? "OK" : "FAIL",    \
           err.c_str());                                        \
    fflush(stdout);                                             \
    if (!ok) return 1;                                          \
  } while (0)

int main() {
  std::string err;
  setenv("KB_MAX_Q", "64", 0);
  setenv("KB_ARENA_BYTES", "16777216", 0);
  Slab* s = Slab::Create(1 << 16, 32 << 20, -1, &err);
  printf("STEP create                  : %s %s\n", s ? "OK" : "FAIL", err.c_str());
  fflush(stdout);
  if (!s) return 1;

  // empty-range scan
  {
    DevRangeQ q{};
    pad("/registry/pods/", q.start);
    pad("/registry/pods0", q.end);
    q.read_rev = 100;
    q.cap = 10;
    std::vector<RangeResult> outs;
    STEP("scan_empty", s->RangeBatch({q}, true, &outs, &err));
    printf("  written=%lld total=%lld\n", (long long)outs[0].written,
           (long long)outs[0].total);
  }

  // merge 12 rows: 4 keys x (rev row + obj rows)
  {
    DeltaRows d;
    struct R { std::string k; uint64_t rev; std::string v; bool tomb; bool f9; uint64_t objrev; };
    std::vector<R> rows = {
        {"/registry/events/x", 0, "", false, false, 7},
        {"/registry/events/x", 7, "ve7", false, false, 0},
        {"/registry/pods/a", 0, "", false, false, 3},
        {"/registry/pods/a", 2, "va2", false, false, 0},
        {"/registry/pods/a", 3, "va3", false, false, 0},
        {"/registry/pods/b", 0, "", false, true, 5},
        {"/registry/pods/b", 4, "vb4", false, false, 0},
        {"/registry/pods/b", 5, "tombstone", true, false, 0},
        {"/registry/pods/c", 0, "", false, false, 6},
        {"/registry/pods/c", 6, "vc6", false, false, 0},
    };
    d.m = (int64_t)rows.size();
    d.keys.resize(rows.size() * KEYW);
    for (size_t i = 0; i < rows.size(); ++i) {
      pad(rows[i].k, d.keys.data() + i * KEYW);
      bool ev = rows[i].k.find("/events/") != std::string::npos;
      if (rows[i].rev == 0) {
        d.meta.push_back(meta_make(false, rows[i].f9, ev, (uint32_t)rows[i].k.size(),
                                   rows[i].f9 ? 9 : 8));
        d.rev.push_back(0);
        d.vo.push_back(rows[i].objrev);
      } else {
        d.meta.push_back(meta_make(rows[i].tomb, false, ev,
                                   (uint32_t)rows[i].k.size(),
                                   (uint32_t)rows[i].v.size()));
        d.rev.push_back(rows[i].rev);
        d.vo.push_back((uint64_t)d.heap.size());
        d.heap.insert(d.heap.end(), rows[i].v.begin(), rows[i].v.end());
        d.heap.resize((d.heap.size() + 15) & ~15ull);
      }
    }
    STEP("merge_initial", s->Merge(d, &err));
    printf("  rows=%lld heap=%lld\n", (long long)s->rows(), (long long)s->heap_used());
  }

  // scan over merged rows at rev=10: winners = a@3, c@6 (b tombstoned), ev@7
  {
    DevRangeQ q{};
    pad("/registry/", q.start);
    pad("/registry0", q.end);
    q.read_rev = 10;
    q.cap = 10;
    std::vector<RangeResult> outs;
    STEP("scan_small", s->RangeBatch({q}, true, &outs, &err));
    printf("  written=%lld total=%lld:", (long long)outs[0].written,
           (long long)outs[0].total);
    for (auto& r : outs[0].recs)
      printf(" (%s@%llu=%s)", r.key.c_str(), (unsigned long long)r.rev, r.val.c_str());
    printf("\n");
    if (outs[0].total != 3) { printf("BAD winner count\n"); return 1; }
  }

  // scan at rev=2: only a@2
  {
    DevRangeQ q{};
    pad("/registry/pods/", q.start);
    pad("/registry/pods0", q.end);
    q.read_rev = 2;
    q.cap = 10;
    std::vector<RangeResult> outs;
    STEP("scan_rev2", s->RangeBatch({q}, true, &outs, &err));
    if (outs[0].total != 1 || outs[0].recs[0].val != "va2") {
      printf("BAD rev2 result total=%lld\n", (long long)outs[0].total);
      return 1;
    }
  }

  // point get
  {
    DevGetQ g{};
    pad("/registry/pods/b", g.key);
    g.read_rev = UINT64_MAX;
    std::vector<GetResult> outs;
    STEP("get_batch", s->GetBatch({g}, &outs, &err));
    printf("  found=%d tomb=%d rev=%llu val=%s\n", outs[0].found, outs[0].tomb,
           (unsigned long long)outs[0].rev, outs[0].val.c_str());
    if (!outs[0].found || !outs[0].tomb || outs[0].rev != 5) return 1;
  }

  // watch: device event ring + ballot filter + catch-up scan
  {
    STEP("watcher_set", s->WatcherSet(0, (const uint8_t*)"/registry/pods/", 15, 1, &err));
    STEP("ring_init", s->EventRingInit(1024, &err));
    std::vector<uint8_t> ekeys(3 * KEYW, 0);
    pad("/registry/pods/a", ekeys.data());
    pad("/registry/cm/x", ekeys.data() + KEYW);
    pad("/registry/pods/b", ekeys.data() + 2 * KEYW);
    std::vector<uint64_t> erevs = {7, 8, 9};
    STEP("ring_push", s->EventRingPush(ekeys.data(), erevs.data(), 3, 0, &err));
    std::vector<uint64_t> bm;
    int64_t W = 0;
    STEP("watch_filter", s->WatchFilterRing(0, 3, &bm, &W, &err));
    printf("  W=%lld bm0=%llx\n", (long long)W, (unsigned long long)bm[0]);
    if ((bm[0] & 7) != 5) { printf("BAD filter bitmap\n"); return 1; }
    std::vector<uint64_t> cw;
    uint8_t p96[KEYW] = {0};
    memcpy(p96, "/registry/pods/", 15);
    STEP("watch_catchup", s->WatchCatchup(p96, 15, 8, 0, 3, &cw, &err));
    if ((cw[0] & 7) != 4) { printf("BAD catchup bitmap %llx\n",
                                   (unsigned long long)cw[0]); return 1; }
  }

  // second merge (newer revisions + rev-row replacement)
  {
    DeltaRows d;
    d.m = 2;
    d.keys.resize(2 * KEYW);
    pad("/registry/pods/a", d.keys.data());
    d.meta.push_back(meta_make(false, false, false, 16, 8));
    d.rev.push_back(0);
    d.vo.push_back(9);  // rev-row replacement: a -> 9
    pad("/registry/pods/a", d.keys.data() + KEYW);
    d.meta.push_back(meta_make(false, false, false, 16, 3));
    d.rev.push_back(9);
    d.vo.push_back((uint64_t)s->heap_used());
    d.heap = {'v', 'a', '9', 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0};
    STEP("merge_replace", s->Merge(d, &err));
    std::vector<DumpRow> rows;
    STEP("dump", s->Dump(&rows, &err));
    printf("  rows=%zu:", rows.size());
    for (auto& r : rows)
      printf(" %s@%llu", r.key.c_str(), (unsigned long long)r.rev);
    printf("\n");
    if (rows.size() != 11) { printf("BAD merged dump size\n"); return 1; }
  }

  // compact at rev 6, no TTL
  {
    std::vector<std::pair<Slab::Bound, Slab::Bound>> borders(1);
    pad("/registry/", borders[0].first.key);
    borders[0].first.rev = 0;
    pad("/registry0", borders[0].second.key);
    borders[0].second.rev = 0;
    STEP("compact", s->Compact(borders, 6, {0}, &err));
    std::vector<DumpRow> rows;
    STEP("dump2", s->Dump(&rows, &err));
    printf("  rows=%zu:", rows.size());
    for (auto& r : rows)
      printf(" %s@%llu", r.key.c_str(), (unsigned long long)r.rev);
    printf("\n");
  }

  // bigger randomized merge + scan consistency (exercise multi-tile scan)
  {
    DeltaRows d;
    const int N = 5000;
    d.m = N;
    d.keys.resize((size_t)N * KEYW);
    char buf[64];
    for (int i = 0; i < N; ++i) {
      snprintf(buf, sizeof(buf), "/registry/zz/ns-%03d/obj-%05d", i % 7, i);
      std::string k(buf);
      pad(k, d.keys.data() + (size_t)i * KEYW);
      if (i % 2 == 0) {
        d.meta.push_back(meta_make(false, false, false, (uint32_t)k.size(), 8));
        d.rev.push_back(0);
        d.vo.push_back(100 + i);
      } else {
        d.meta.push_back(meta_make(false, false, false, (uint32_t)k.size(), 16));
        d.rev.push_back(100 + i);
        d.vo.push_back((uint64_t)(s->heap_used() + d.heap.size()));
        d.heap.insert(d.heap.end(), 16, (uint8_t)('A' + i % 26));
      }
    }
    // keys must be sorted: generated sorted? ns-%03d varies i%7 -> NOT sorted.
    // sort rows by (key, rev)
    std::vector<int> idx(N);
    for (int i = 0; i < N; ++i) idx[i] = i;
    std::sort(idx.begin(), idx.end(), [&](int a, int b) {
      int c = memcmp(d.keys.data() + (size_t)a * KEYW,
                     d.keys.data() + (size_t)b * KEYW, KEYW);
      if (c != 0) return c < 0;
      return d.rev[a] < d.rev[b];
    });
    DeltaRows ds;
    ds.m = N;
    ds.keys.resize(d.keys.size());
    ds.heap = d.heap;
    for (int j = 0; j < N; ++j) {
      int i = idx[j];
      memcpy(ds.keys.data() + (size_t)j * KEYW, d.keys.data() + (size_t)i * KEYW, KEYW);
      ds.meta.push_back(d.meta[i]);
      ds.rev.push_back(d.rev[i]);
      ds.vo.push_back(d.vo[i]);
    }
    STEP("merge_5k", s->Merge(ds, &err));
    DevRangeQ q{};
    pad("/registry/zz/", q.start);
    pad("/registry/zz0", q.end);
    q.read_rev = 1000000;
    q.cap = 0;  // unbounded
    std::vector<RangeResult> outs;
    STEP("scan_5k", s->RangeBatch({q}, false, &outs, &err));
    printf("  total=%lld (expect 2500)\n", (long long)outs[0].total);
    if (outs[0].total != 2500) return 1;
  }

  printf("STEP destroy ...\n");
  fflush(stdout);
  delete s;
  printf("SELFTEST PASS\n");
  return 0;
}
