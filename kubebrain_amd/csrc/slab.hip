// kubebrain_amd/csrc/slab.hip — MI355X (gfx950) HBM slab engine.
//
// All device kernels of the MVCC hot path. HBM-bound
// integer/byte work by design (north_star: indexing/scan, no MFMA):
// wavefront = 64 lanes, __ballot is 64-bit, coalesced 16B-per-lane loads
// where layout permits. Semantics cites are into /root/reference.

#include <hip/hip_runtime.h>

#include <chrono>
#include <cstdio>
#include <cstring>

#include "slab_dev.h"

namespace kbslab {

#define HIP_CHECK(expr)                                                    \
  do {                                                                     \
    hipError_t _e = (expr);                                                \
    if (_e != hipSuccess) {                                                \
      if (err) *err = std::string(#expr) + ": " + hipGetErrorString(_e);   \
      return false;                                                        \
    }                                                                      \
  } while (0)

#define HIP_CHECK_NULL(expr)                                               \
  do {                                                                     \
    hipError_t _e = (expr);                                                \
    if (_e != hipSuccess) {                                                \
      if (err) *err = std::string(#expr) + ": " + hipGetErrorString(_e);   \
      return nullptr;                                                      \
    }                                                                      \
  } while (0)

static inline int64_t ceil_div(int64_t a, int64_t b) { return (a + b - 1) / b; }

// ---------------------------------------------------------------- device ---

// byte-lexicographic compare of two 96B zero-padded keys (== internal-key
// order for keys with bytes > '$'; coder/normal.go:29-31).
__device__ __forceinline__ int keycmp96(const uint8_t* a, const uint8_t* b) {
  const uint64_t* A = (const uint64_t*)a;
  const uint64_t* B = (const uint64_t*)b;
#pragma unroll
  for (int i = 0; i < KEYW / 8; ++i) {
    uint64_t x = A[i], y = B[i];
    if (x != y) {
      x = __builtin_bswap64(x);
      y = __builtin_bswap64(y);
      return x < y ? -1 : 1;
    }
  }
  return 0;
}

// Full-key compare with spill tails (keys > KEYW): the 96B
// zero-padded prefix decides every compare except exact 96-byte prefix ties
// — only possible when BOTH keys are >= 96B (key bytes are > 0x24 > 0x00) —
// which compare their tails from the spill heaps. The tail loop runs on the
// rare tie path only; the hot predicate never touches it.
__device__ __forceinline__ int keycmp_ext(
    const uint8_t* a96, uint32_t alen, uint64_t ako, const uint8_t* aspill,
    const uint8_t* b96, uint32_t blen, uint64_t bko, const uint8_t* bspill) {
  int c = keycmp96(a96, b96);
  if (c || (alen <= (uint32_t)KEYW && blen <= (uint32_t)KEYW)) return c;
  uint32_t at = alen > (uint32_t)KEYW ? alen - KEYW : 0;
  uint32_t bt = blen > (uint32_t)KEYW ? blen - KEYW : 0;
  const uint8_t* ap = aspill + ako;
  const uint8_t* bp = bspill + bko;
  uint32_t m = at < bt ? at : bt;
  for (uint32_t i = 0; i < m; ++i)
    if (ap[i] != bp[i]) return ap[i] < bp[i] ? -1 : 1;
  return at == bt ? 0 : (at < bt ? -1 : 1);
}

// one sorted run's columns (base or delta); spill is shared store-wide
struct Run {
  const uint8_t* keys;
  const uint64_t* meta;
  const uint64_t* rev;
  const uint64_t* vo;
  const uint64_t* ko;
};

// a query-side key: 96B padded prefix + optional tail in `tails`
struct QKey {
  const uint8_t* k96;
  uint32_t len;
  uint64_t ko;
  const uint8_t* tails;
};

__device__ __forceinline__ QKey row_qk(const Run& r, const uint8_t* spill,
                                       int64_t i) {
  return QKey{r.keys + i * KEYW, meta_klen(r.meta[i]), r.ko[i], spill};
}

__device__ __forceinline__ int rowcmp_q(const Run& r, const uint8_t* spill,
                                        int64_t i, const QKey& q) {
  return keycmp_ext(r.keys + i * KEYW, meta_klen(r.meta[i]), r.ko[i], spill,
                    q.k96, q.len, q.ko, q.tails);
}

// first row in [lo,hi) with (key,rev) >= (q,qrev)
__device__ int64_t d_lb_range(const Run& r, const uint8_t* spill, int64_t lo,
                              int64_t hi, const QKey& q, uint64_t qrev) {
  while (lo < hi) {
    int64_t mid = (lo + hi) >> 1;
    int c = rowcmp_q(r, spill, mid, q);
    if (c < 0 || (c == 0 && r.rev[mid] < qrev)) lo = mid + 1; else hi = mid;
  }
  return lo;
}

// wave-cooperative 64-ary lower bound: every round all 64 lanes probe one
// pivot each (64 independent loads in ONE memory round trip), the ballot of
// "pivot < q" collapses the range 64x. Depth log64(n): a 21M-row slab costs
// ~5 round trips where the serial binary search pays ~25 dependent ones.
// All lanes of the wave must call; the result is wave-uniform.
__device__ int64_t d_lb_wave(const Run& r, const uint8_t* spill, int64_t n,
                             const QKey& q, uint64_t qrev) {
  const int lane = threadIdx.x & 63;
  int64_t lo = 0, hi = n;  // answer in [lo, hi]
  while (hi - lo > 64) {
    int64_t step = (hi - lo) >> 6;  // >= 1
    int64_t p = lo + (int64_t)(lane + 1) * step;
    bool lt = false;  // keys[p] < q (monotone non-increasing in p)
    if (p < hi) {
      int c = rowcmp_q(r, spill, p, q);
      lt = c < 0 || (c == 0 && r.rev[p] < qrev);
    }
    int k = __popcll(__ballot(lt));
    int64_t nlo = lo + (int64_t)k * step;
    int64_t nhi = k == 64 ? hi : min(lo + (int64_t)(k + 1) * step, hi);
    lo = nlo;
    hi = nhi;
  }
  int64_t p = lo + lane;
  bool ge = false;  // first in-range row with keys[p] >= q
  if (p < hi) {
    int c = rowcmp_q(r, spill, p, q);
    ge = !(c < 0 || (c == 0 && r.rev[p] < qrev));
  }
  uint64_t b = __ballot(ge);
  return b ? lo + (__ffsll((unsigned long long)b) - 1) : hi;
}

// adjacent-row full-key equality (same spill both sides)
__device__ __forceinline__ bool rows_same_key(const uint8_t* keys,
                                              const uint64_t* meta,
                                              const uint64_t* ko,
                                              const uint8_t* spill, int64_t i,
                                              int64_t j) {
  const uint64_t* a = (const uint64_t*)(keys + i * KEYW);
  const uint64_t* b = (const uint64_t*)(keys + j * KEYW);
#pragma unroll
  for (int k = 0; k < KEYW / 8; ++k)
    if (a[k] != b[k]) return false;
  uint32_t al = meta_klen(meta[i]), bl = meta_klen(meta[j]);
  if (al <= (uint32_t)KEYW && bl <= (uint32_t)KEYW) return true;
  if (al != bl) return false;
  const uint8_t* ap = spill + ko[i];
  const uint8_t* bp = spill + ko[j];
  for (uint32_t t = 0; t < al - (uint32_t)KEYW; ++t)
    if (ap[t] != bp[t]) return false;
  return true;
}

// ---- same_next: adjacent-key equality bits over one run ----
__global__ void k_same_next(const uint8_t* __restrict__ keys,
                            uint64_t* __restrict__ meta,
                            const uint64_t* __restrict__ ko,
                            const uint8_t* __restrict__ spill, int64_t n) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  uint64_t m = meta[i];
  bool same = i + 1 < n && rows_same_key(keys, meta, ko, spill, i, i + 1);
  meta[i] = same ? (m | M_SAME_NEXT) : (m & ~M_SAME_NEXT);
}

constexpr uint64_t ROW_TAG_DELTA = 1ull << 63;  // winner row lives in the delta run
constexpr uint64_t ROW_MASK = ROW_TAG_DELTA - 1;

// ---- two-run variant: base run + sorted delta run merged at scan time ---
// The delta run holds rows strictly newer than the base run's rows of the
// same key, so the global winner of a key is the delta
// winner when the delta has any row of the key with rev<=R, else the base
// winner — base winners are suppressed by a delta probe, and the two ordered
// winner lists are merged by rank (keys never collide across lists).

// scan block width: runtime-tunable (KB_SCAN_T in {256,512,1024}); the
// kernel reads blockDim.x, only the LDS wave-count array is sized for the max
constexpr int SCAN_T_MAX = 1024;

// shadow-revision column (base run only): shadow[i] = smallest delta-run
// revision (>=1) of base row i's key, UINT64_MAX = the delta holds no row of
// that key. Maintained by k_shadow_mark at delta-insert time and reset to
// MAX whenever the base is rebuilt (fold/compact empty the delta). Turns the
// scan's winner-suppression probe — a per-winner binary search over the
// delta — into one coalesced 8B stream read: base winner i is suppressed
// iff shadow[i] != MAX && shadow[i] <= R (a strictly-newer delta row of the
// key exists at or below the read revision).
__global__ void k_shadow_mark(Run b, int64_t n, Run dnew, int64_t m,
                              const uint8_t* __restrict__ spill,
                              uint64_t* __restrict__ shadow) {
  int64_t j = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (j >= m) return;
  uint64_t rv = dnew.rev[j];
  if (rv == 0) return;  // rev-row holes never suppress
  QKey k = row_qk(dnew, spill, j);
  for (int64_t i = d_lb_range(b, spill, 0, n, k, 1); i < n; ++i) {
    if (rowcmp_q(b, spill, i, k) != 0) break;
    atomicMin((unsigned long long*)&shadow[i], (unsigned long long)rv);
  }
}

// one run's winner scan (ordered append); returns written, *total = seen.
// Two-pass per tile: (1) each wave reads a CONTIGUOUS chunk of rows with
// fully-coalesced 8B meta/rev streams, recording winner flags in a register
// bitmask -- no LDS, no syncs inside the pass; (2) after one barrier
// publishes the per-wave totals, a register-only ballot replay scatters the
// winner row indices at their ordered offsets. Two __syncthreads per tile
// of up to NW*4096 rows (the old design paid two per 1024 rows).
__device__ int64_t scan_run_winners(
    const Run& run, const uint8_t* spill, int64_t lo, int64_t hi, uint64_t R,
    int64_t cap, uint64_t* out, int64_t out_cap, uint64_t tagbit,
    const uint64_t* shadow,  // shadow-revision column (null => no suppression)
    int64_t* total_out, int64_t* scanned_accum, int* wave_cnt,
    unsigned long long* dbg) {  // KB_SCAN_DBG phase cycles (null in prod)
  const uint64_t* __restrict__ rev = run.rev;
  const uint64_t* __restrict__ meta = run.meta;
  const int lane = threadIdx.x & 63, w = threadIdx.x >> 6;
  const int NW = blockDim.x >> 6;
  const int64_t span = hi - lo;
  if (span <= 0) { *total_out = 0; return 0; }
  // per-wave chunk: cover the span in one tile when it fits (<= NW*4096
  // rows); for capped queries over much larger spans, tile by a winner-
  // density heuristic so the cap early-exit still bounds the read volume
  int64_t target = span;
  if (cap != INT64_MAX && 6 * cap < span) target = 6 * cap;
  int64_t C = (target + NW - 1) / NW;
  C = (C + 63) & ~63ll;
  if (C > 4096) C = 4096;
  const int64_t TILE = (int64_t)NW * C;
  int64_t scanned = 0;
  int64_t cnt = 0;  // winners before this tile (uniform across the block)
  const bool dbg0 = dbg && threadIdx.x == 0;
  unsigned long long tprev = dbg0 ? wall_clock64() : 0;
  for (int64_t t = lo; t < hi; t += TILE) {
    const int64_t wbase = t + (int64_t)w * C;
    int rounds = 0;
    if (wbase < hi) rounds = (int)((min(C, hi - wbase) + 63) >> 6);
    // pass 1a: streaming winner predicate — rev/meta 8B streams only, no
    // probes and no divergent chains, so the loads pipeline across rounds.
    // same_next set => row i+1 exists, shares the key, and (rows of one key
    // being contiguous and its key < qend) lies below hi, so the rev[i+1]
    // load is gated by i+1<hi alone, independent of the meta bits.
    // pass 1: streaming winner predicate — rev/meta/shadow 8B streams only,
    // no probes and no divergent chains, so the loads pipeline across
    // rounds. same_next set => row i+1 exists, shares the key, and (rows of
    // one key being contiguous and its key < qend) lies below hi, so the
    // rev[i+1] load is gated by i+1<hi alone, independent of the meta bits.
    // Suppression is the shadow-revision stream: a base winner loses iff a
    // strictly-newer delta row of its key exists at or below R.
    uint64_t flags = 0;
    // full rounds carry no bounds check (every lane in range), so the
    // rev/meta/shadow loads have no control dependence and pipeline across
    // the unrolled iterations; only the ragged tail round is guarded
    const int full = (int)min((int64_t)rounds, (hi - wbase) >> 6);
#pragma unroll 4
    for (int r = 0; r < full; ++r) {
      int64_t i = wbase + ((int64_t)r << 6) + lane;
      uint64_t rv = rev[i], m = meta[i];
      uint64_t rvn = (i + 1 < hi) ? rev[i + 1] : 0;
      uint64_t sh = shadow ? shadow[i] : UINT64_MAX;
      bool win = false;
      if (rv > 0 && rv <= R && !(m & M_TOMB) &&
          !(sh != UINT64_MAX && sh <= R))
        win = !(m & M_SAME_NEXT) || rvn > R;
      if (win) flags |= 1ull << r;
    }
    for (int r = full; r < rounds; ++r) {
      int64_t i = wbase + ((int64_t)r << 6) + lane;
      bool win = false;
      if (i < hi) {
        uint64_t rv = rev[i], m = meta[i];
        uint64_t rvn = (i + 1 < hi) ? rev[i + 1] : 0;
        uint64_t sh = shadow ? shadow[i] : UINT64_MAX;
        if (rv > 0 && rv <= R && !(m & M_TOMB) &&
            !(sh != UINT64_MAX && sh <= R))
          win = !(m & M_SAME_NEXT) || rvn > R;
      }
      if (win) flags |= 1ull << r;
    }
    if (dbg0) { unsigned long long t1 = wall_clock64(); atomicAdd(&dbg[1], t1 - tprev); tprev = t1; }
    uint32_t wcnt = 0;
    for (int r = 0; r < rounds; ++r)
      wcnt += (uint32_t)__popcll(__ballot((flags >> r) & 1));
    if (dbg0) { unsigned long long t2 = wall_clock64(); atomicAdd(&dbg[2], t2 - tprev); tprev = t2; }
    if (lane == 0) wave_cnt[w] = (int)wcnt;
    __syncthreads();
    int64_t waveoff = 0, tile_total = 0;
    for (int k = 0; k < NW; ++k) {
      if (k < w) waveoff += wave_cnt[k];
      tile_total += wave_cnt[k];
    }
    // pass 2: ordered scatter by ballot replay (no memory re-reads)
    if (out && wcnt) {
      uint32_t done = 0;
      for (int r = 0; r < rounds; ++r) {
        uint64_t b = __ballot((flags >> r) & 1);
        if ((flags >> r) & 1) {
          int64_t idx = cnt + waveoff + done +
                        (int64_t)__popcll(b & ((1ull << lane) - 1));
          if (idx < cap && idx < out_cap)
            out[idx] = (uint64_t)(wbase + ((int64_t)r << 6) + lane) | tagbit;
        }
        done += (uint32_t)__popcll(b);
      }
    }
    if (dbg0) { unsigned long long t3 = wall_clock64(); atomicAdd(&dbg[3], t3 - tprev); tprev = t3; }
    cnt += tile_total;
    scanned += min(TILE, hi - t);
    if (cnt >= cap) break;
    __syncthreads();  // wave_cnt reused next tile
  }
  *total_out = cnt;
  if (scanned_accum) *scanned_accum += scanned;
  int64_t written = cnt < cap ? cnt : cap;
  if (written > out_cap) written = out_cap;
  return out ? written : 0;
}

// lower_bound over a winner list via key indirection (no cross-list ties)
__device__ int64_t d_lb_winlist(const uint64_t* rows, int64_t n, const Run& b,
                                const Run& d, const uint8_t* spill,
                                const QKey& q) {
  int64_t lo = 0, hi = n;
  while (lo < hi) {
    int64_t mid = (lo + hi) >> 1;
    uint64_t rt = rows[mid];
    const Run& rr = (rt & ROW_TAG_DELTA) ? d : b;
    if (rowcmp_q(rr, spill, (int64_t)(rt & ROW_MASK), q) < 0) lo = mid + 1;
    else hi = mid;
  }
  return lo;
}

// bounds pre-pass: one 256-thread block per query, its 4 waves resolving the
// 4 run bounds concurrently via the wave-cooperative 64-ary search. As a
// standalone launch all 4*nq searches overlap chip-wide; inside the scan
// kernel they sat on every block's critical path (~7 us serial per block).
__global__ void k_range_bounds(Run b, int64_t n, Run d, int64_t dn,
                               const uint8_t* __restrict__ spill,
                               const uint8_t* __restrict__ qtails,
                               const DevRangeQ* __restrict__ qs, int nq,
                               int64_t* __restrict__ bounds_g /*[4*nq]*/) {
  int q = blockIdx.x;
  if (q >= nq) return;
  const DevRangeQ& Q = qs[q];
  const QKey qstart{Q.start, Q.start_klen, Q.start_ko, qtails};
  const QKey qend{Q.end, Q.end_klen, Q.end_ko, qtails};
  int w = threadIdx.x >> 6;
  int64_t r = -1;
  if (w == 0) r = d_lb_wave(b, spill, n, qstart, Q.start_rev);
  else if (w == 1) r = d_lb_wave(b, spill, n, qend, 0);
  else if (w == 2) r = d_lb_wave(d, spill, dn, qstart, Q.start_rev);
  else if (w == 3) r = d_lb_wave(d, spill, dn, qend, 0);
  if (w < 4 && (threadIdx.x & 63) == 0) bounds_g[(int64_t)q * 4 + w] = r;
}

__global__ void k_range_scan2(
    Run b, int64_t n, Run d, int64_t dn, const uint8_t* __restrict__ spill,
    const uint8_t* __restrict__ qtails, const DevRangeQ* __restrict__ qs,
    int nq, int64_t max_cap,
    uint64_t* __restrict__ rows_b, uint64_t* __restrict__ rows_d,
    uint64_t* __restrict__ rows_m, int64_t* __restrict__ found_out,
    int64_t* __restrict__ total_out, unsigned long long* __restrict__ scanned_out,
    const int64_t* __restrict__ bounds_g,  // from k_range_bounds
    const uint64_t* __restrict__ shadow,  // base shadow-revision column
    unsigned long long* dbg) {  // KB_SCAN_DBG phase cycles (null in prod)
  int q = blockIdx.x;
  if (q >= nq) return;
  __shared__ int wave_cnt[SCAN_T_MAX / 64];
  const DevRangeQ& Q = qs[q];
  const bool dbg0 = dbg && threadIdx.x == 0;
  unsigned long long tk0 = dbg0 ? wall_clock64() : 0;
  // bounds resolved by the k_range_bounds pre-pass
  const int64_t lo_s = bounds_g[(int64_t)q * 4 + 0];
  const int64_t hi_s = bounds_g[(int64_t)q * 4 + 1];
  const int64_t dlo_s = bounds_g[(int64_t)q * 4 + 2];
  const int64_t dhi_s = bounds_g[(int64_t)q * 4 + 3];
  if (dbg0) { unsigned long long tkb = wall_clock64(); atomicAdd(&dbg[0], tkb - tk0); }
  const int64_t cap = Q.cap > 0 ? Q.cap : INT64_MAX;
  int64_t scanned = 0;
  int64_t dtotal = 0, btotal = 0;
  uint64_t* outd = Q.count_only ? nullptr : rows_d + (int64_t)q * max_cap;
  uint64_t* outb = Q.count_only ? nullptr : rows_b + (int64_t)q * max_cap;
  int64_t nB = scan_run_winners(d, spill, dlo_s, dhi_s, Q.read_rev,
                                cap, outd, max_cap, ROW_TAG_DELTA, nullptr,
                                &dtotal, &scanned, wave_cnt, dbg);
  __syncthreads();  // wave_cnt handoff between the two runs
  int64_t nA = scan_run_winners(b, spill, lo_s, hi_s, Q.read_rev, cap,
                                outb, max_cap, 0, dn ? shadow : nullptr,
                                &btotal, &scanned, wave_cnt, dbg);
  __syncthreads();  // winner lists complete before the merge reads them
  unsigned long long tkm = dbg0 ? wall_clock64() : 0;
  // merge by rank into rows_m (keys are disjoint across the two lists)
  int64_t cap_m = nA + nB;
  if (cap_m > cap) cap_m = cap;
  if (cap_m > max_cap) cap_m = max_cap;
  if (!Q.count_only) {
    uint64_t* outm = rows_m + (int64_t)q * max_cap;
    for (int64_t j = threadIdx.x; j < nA + nB; j += blockDim.x) {
      uint64_t rt;
      int64_t pos;
      if (j < nA) {
        rt = outb[j];
        QKey k = row_qk(b, spill, (int64_t)(rt & ROW_MASK));
        pos = j + d_lb_winlist(outd, nB, b, d, spill, k);
      } else {
        rt = outd[j - nA];
        QKey k = row_qk(d, spill, (int64_t)(rt & ROW_MASK));
        pos = (j - nA) + d_lb_winlist(outb, nA, b, d, spill, k);
      }
      if (pos < cap_m) outm[pos] = rt;
    }
  }
  if (dbg0) {
    unsigned long long tke = wall_clock64();
    atomicAdd(&dbg[4], tke - tkm);
    atomicAdd(&dbg[5], tke - tk0);
    atomicMax(&dbg[6], tke - tk0);  // slowest block (tail skew diagnostic)
    atomicAdd(&dbg[7], 1ull);
  }
  if (threadIdx.x == 0) {
    total_out[q] = btotal + dtotal;
    found_out[q] = Q.count_only ? 0 : cap_m;
    atomicAdd(scanned_out, (unsigned long long)scanned);
  }
}

// two-run point read: the delta run wins when it has any row of the key <= R
__global__ void k_get2(Run b, Run d, int64_t n, int64_t dn,
                       const uint8_t* __restrict__ spill,
                       const uint8_t* __restrict__ qtails,
                       const uint8_t* __restrict__ heap,
                       const DevGetQ* __restrict__ qs, int nq, int copy_vals,
                       uint8_t* __restrict__ out, int64_t slot,
                       uint64_t* __restrict__ orev, uint64_t* __restrict__ ometa,
                       int32_t* __restrict__ ofound, int32_t* __restrict__ oovf) {
  int q = blockIdx.x * (blockDim.x / 64) + (threadIdx.x >> 6);
  int lane = threadIdx.x & 63;
  if (q >= nq) return;
  const DevGetQ& Q = qs[q];
  const QKey qk{Q.key, Q.klen, Q.ko, qtails};
  int64_t row = -1;
  int isdelta = 0;
  // upper_bound(key,R) == lower_bound(key,R+1); R+1 saturates safely (real
  // revisions are far below UINT64_MAX). Wave-cooperative searches: the
  // whole wave runs each 64-ary probe round in one memory round trip.
  uint64_t nr = Q.read_rev == UINT64_MAX ? UINT64_MAX : Q.read_rev + 1;
  if (dn > 0) {
    int64_t ub = d_lb_wave(d, spill, dn, qk, nr);
    if (ub > 0) {
      int64_t c = ub - 1;
      if (d.rev[c] >= 1 && rowcmp_q(d, spill, c, qk) == 0) {
        row = c;
        isdelta = 1;
      }
    }
  }
  if (row < 0) {
    int64_t ub = d_lb_wave(b, spill, n, qk, nr);
    if (ub > 0) {
      int64_t c = ub - 1;
      if (b.rev[c] >= 1 && rowcmp_q(b, spill, c, qk) == 0) row = c;
    }
  }
  if (row < 0) {
    if (lane == 0) { ofound[q] = 0; oovf[q] = 0; }
    return;
  }
  const uint64_t* meta = isdelta ? d.meta : b.meta;
  const uint64_t* rev = isdelta ? d.rev : b.rev;
  const uint64_t* vo = isdelta ? d.vo : b.vo;
  uint64_t m = meta[row];
  uint32_t vlen = meta_vlen(m);
  if (lane == 0) {
    ofound[q] = 1;
    orev[q] = rev[row];
    ometa[q] = m;
    oovf[q] = (copy_vals && vlen > slot) ? 1 : 0;
  }
  if (!copy_vals || vlen > slot) return;
  const uint8_t* vs = heap + vo[row];
  uint8_t* vd = out + (int64_t)q * slot;
  uint32_t w16 = vlen >> 4;
  for (uint32_t b = lane; b < w16; b += 64)
    ((uint4*)vd)[b] = ((const uint4*)vs)[b];
  if (lane == 0)
    for (uint32_t b = w16 * 16; b < vlen; ++b) vd[b] = vs[b];
}

// ---- gather: winners -> packed records in the per-query device arena ----
// record: u64 rev | u32 klen | u32 vlen | key (pad8) | val (pad8)
__device__ __forceinline__ int64_t rec_bytes(uint32_t klen, uint32_t vlen) {
  // 16B-aligned record: header | key pad16 | value pad16 (uint4 copies)
  return 16 + ((klen + 15) & ~15u) + ((vlen + 15) & ~15u);
}

__global__ void k_gather(Run b, Run d,
                         const uint8_t* __restrict__ heap,
                         const uint64_t* __restrict__ rows_out, int64_t max_cap,
                         const int64_t* __restrict__ found_out,
                         const DevRangeQ* __restrict__ qs, int nq,
                         uint8_t* __restrict__ gbuf, int64_t qcap,
                         int64_t* __restrict__ offs,  // [q*max_cap+j] scratch
                         int64_t* __restrict__ gbytes_out,
                         int32_t* __restrict__ overflow,
                         unsigned long long* __restrict__ bytes_out) {
  int q = blockIdx.x;
  if (q >= nq) return;
  int64_t nwin = found_out[q];
  const bool konly = qs[q].keys_only != 0;
  const uint64_t* rows = rows_out + (int64_t)q * max_cap;
  int64_t* qoffs = offs + (int64_t)q * max_cap;
  __shared__ int64_t lds[256];
  __shared__ int64_t base_s;
  if (threadIdx.x == 0) base_s = 0;
  __syncthreads();
  // phase A: exclusive offsets by chunked block scan
  for (int64_t c0 = 0; c0 < nwin; c0 += blockDim.x) {
    int64_t j = c0 + threadIdx.x;
    int64_t sz = 0;
    if (j < nwin) {
      uint64_t rt = rows[j];
      uint64_t m = (rt & ROW_TAG_DELTA ? d.meta : b.meta)[rt & ROW_MASK];
      sz = rec_bytes(meta_klen(m), konly ? 0 : meta_vlen(m));
    }
    lds[threadIdx.x] = sz;
    __syncthreads();
    for (int off = 1; off < 256; off <<= 1) {
      int64_t add = threadIdx.x >= (unsigned)off ? lds[threadIdx.x - off] : 0;
      __syncthreads();
      lds[threadIdx.x] += add;
      __syncthreads();
    }
    if (j < nwin) qoffs[j] = base_s + lds[threadIdx.x] - sz;
    __syncthreads();
    if (threadIdx.x == 0) base_s += lds[255];
    __syncthreads();
  }
  if (threadIdx.x == 0) {
    gbytes_out[q] = base_s;
    overflow[q] = base_s > qcap ? 1 : 0;
    atomicAdd(bytes_out, (unsigned long long)(base_s <= qcap ? base_s : 0));
  }
  // record copies happen in k_gather_copy (wider blocks, better latency
  // hiding); this kernel only computes offsets/overflow
}

// copy winners' records into the arena; launched with 512 threads per query
__global__ void k_gather_copy(Run rb, Run rd,
                              const uint8_t* __restrict__ spill,
                              const uint8_t* __restrict__ heap,
                              const uint64_t* __restrict__ rows_out,
                              int64_t max_cap,
                              const int64_t* __restrict__ found_out,
                              const DevRangeQ* __restrict__ qs, int nq,
                              uint8_t* __restrict__ gbuf, int64_t qcap,
                              const int64_t* __restrict__ offs,
                              const int32_t* __restrict__ overflow,
                              int gwl) {  // log2(record-group lanes), KB_GATHER_GW
  int q = blockIdx.x;
  if (q >= nq || overflow[q]) return;
  int64_t nwin = found_out[q];
  const bool konly = qs[q].keys_only != 0;
  const uint64_t* rows = rows_out + (int64_t)q * max_cap;
  const int64_t* qoffs = offs + (int64_t)q * max_cap;
  // record groups of 2^gwl lanes — independent load chains across records
  const int gw = 1 << gwl, gpw = 64 >> gwl;  // lanes per group, groups per wave
  int lane = threadIdx.x & 63, w = threadIdx.x >> 6;
  int grp = lane >> gwl, gl = lane & (gw - 1);
  uint8_t* qb = gbuf + (int64_t)q * qcap;
  int64_t stride = (blockDim.x / 64) * gpw;
  for (int64_t j = (int64_t)w * gpw + grp; j < nwin; j += stride) {
    uint64_t rt = rows[j];
    bool isd = (rt & ROW_TAG_DELTA) != 0;
    int64_t row = (int64_t)(rt & ROW_MASK);
    const Run& r = isd ? rd : rb;
    uint64_t m = r.meta[row];
    uint32_t klen = meta_klen(m), vlen = konly ? 0 : meta_vlen(m);
    uint8_t* dst = qb + qoffs[j];
    if (gl == 0) {
      *(uint64_t*)dst = r.rev[row];
      ((uint32_t*)dst)[2] = klen;
      ((uint32_t*)dst)[3] = vlen;
    }
    const uint8_t* ks = r.keys + row * KEYW;
    uint8_t* kd = dst + 16;  // 8-aligned (records are 16B-aligned)
    uint32_t kin = klen > (uint32_t)KEYW ? (uint32_t)KEYW : klen;
    // column rows are 96B-strided => 8-aligned; copy whole u64s (may round
    // up to 7B into the record's 16B key padding — always in-bounds)
    for (uint32_t b = gl; b < (kin + 7) / 8; b += gw)
      ((uint64_t*)kd)[b] = ((const uint64_t*)ks)[b];
    if (klen > (uint32_t)KEYW) {  // spill tail (keys > 96B)
      const uint8_t* ts = spill + r.ko[row];
      for (uint32_t t = gl; t < klen - (uint32_t)KEYW; t += gw)
        kd[KEYW + t] = ts[t];
    }
    if (konly) continue;
    const uint8_t* vs = heap + r.vo[row];
    uint8_t* vd = dst + 16 + ((klen + 15) & ~15u);
    uint32_t w16 = vlen >> 4;
    for (uint32_t b = gl; b < w16; b += gw)
      ((uint4*)vd)[b] = ((const uint4*)vs)[b];
    if (gl == 0)
      for (uint32_t b = w16 * 16; b < vlen; ++b) vd[b] = vs[b];
  }
}

// wave-per-record gather (KB_GATHER_MODE=2): grid (nq, cap/4), 4 records per
// 256-thread block, one 64-lane wave each — a 512B value is ONE uint4 round
// over 32 lanes and the 96B key one u64 round, with no serial record chain
// inside a lane group. Idle blocks (j >= found) retire immediately.
__global__ void k_gather_copy2(Run rb, Run rd,
                               const uint8_t* __restrict__ spill,
                               const uint8_t* __restrict__ heap,
                               const uint64_t* __restrict__ rows_out,
                               int64_t max_cap,
                               const int64_t* __restrict__ found_out,
                               const DevRangeQ* __restrict__ qs, int nq,
                               uint8_t* __restrict__ gbuf, int64_t qcap,
                               const int64_t* __restrict__ offs,
                               const int32_t* __restrict__ overflow) {
  int q = blockIdx.x;
  if (q >= nq || overflow[q]) return;
  int64_t j = (int64_t)blockIdx.y * (blockDim.x >> 6) + (threadIdx.x >> 6);
  if (j >= found_out[q]) return;
  const bool konly = qs[q].keys_only != 0;
  const int lane = threadIdx.x & 63;
  uint64_t rt = rows_out[(int64_t)q * max_cap + j];
  bool isd = (rt & ROW_TAG_DELTA) != 0;
  int64_t row = (int64_t)(rt & ROW_MASK);
  const Run& r = isd ? rd : rb;
  uint64_t m = r.meta[row];
  uint32_t klen = meta_klen(m), vlen = konly ? 0 : meta_vlen(m);
  uint8_t* dst = gbuf + (int64_t)q * qcap + offs[(int64_t)q * max_cap + j];
  if (lane == 0) {
    *(uint64_t*)dst = r.rev[row];
    ((uint32_t*)dst)[2] = klen;
    ((uint32_t*)dst)[3] = vlen;
  }
  const uint8_t* ks = r.keys + row * KEYW;
  uint8_t* kd = dst + 16;  // records are 16B-aligned
  uint32_t kin = klen > (uint32_t)KEYW ? (uint32_t)KEYW : klen;
  for (uint32_t b = lane; b < (kin + 7) / 8; b += 64)
    ((uint64_t*)kd)[b] = ((const uint64_t*)ks)[b];
  if (klen > (uint32_t)KEYW) {  // spill tail (keys > 96B)
    const uint8_t* ts = spill + r.ko[row];
    for (uint32_t t = lane; t < klen - (uint32_t)KEYW; t += 64)
      kd[KEYW + t] = ts[t];
  }
  if (konly) return;
  const uint8_t* vs = heap + r.vo[row];
  uint8_t* vd = dst + 16 + ((klen + 15) & ~15u);
  uint32_t w16 = vlen >> 4;
  for (uint32_t b = lane; b < w16; b += 64)
    ((uint4*)vd)[b] = ((const uint4*)vs)[b];
  if (lane == 0)
    for (uint32_t b = w16 * 16; b < vlen; ++b) vd[b] = vs[b];
}

// pack used per-query regions contiguously for one D2H
__global__ void k_pack(const uint8_t* __restrict__ gbuf, int64_t qcap,
                       const int64_t* __restrict__ gbytes,
                       const int64_t* __restrict__ goffs,
                       const int32_t* __restrict__ overflow,
                       uint8_t* __restrict__ out, int nq) {
  int q = blockIdx.x;
  if (q >= nq) return;
  if (overflow[q]) return;  // nothing was gathered for this query
  int64_t bytes = gbytes[q];
  const uint4* src = (const uint4*)(gbuf + (int64_t)q * qcap);
  uint4* dst = (uint4*)(out + goffs[q]);
  int64_t words = bytes / 16;  // records are 16B-aligned (goffs too)
  for (int64_t w = threadIdx.x; w < words; w += blockDim.x) dst[w] = src[w];
}

// ---- generic exclusive scan (u64), 256-wide blocks ----------------------
__global__ void k_block_scan(const uint64_t* __restrict__ in,
                             uint64_t* __restrict__ out,
                             uint64_t* __restrict__ sums, int64_t n) {
  __shared__ uint64_t lds[256];
  int64_t base = (int64_t)blockIdx.x * 256;
  int t = threadIdx.x;
  uint64_t v = base + t < n ? in[base + t] : 0;
  lds[t] = v;
  __syncthreads();
  for (int off = 1; off < 256; off <<= 1) {
    uint64_t add = t >= off ? lds[t - off] : 0;
    __syncthreads();
    lds[t] += add;
    __syncthreads();
  }
  if (base + t < n) out[base + t] = lds[t] - v;
  if (t == 255) sums[blockIdx.x] = lds[255];
}

__global__ void k_add_offsets(uint64_t* __restrict__ out,
                              const uint64_t* __restrict__ block_offs, int64_t n) {
  int64_t i = (int64_t)blockIdx.x * 256 + threadIdx.x;
  if (i < n) out[i] += block_offs[blockIdx.x];
}

// ---- compaction (compact.go + scanner.go:444-491, 566-591) --------------
__global__ void k_merge_newn(const uint64_t* __restrict__ dropx, int64_t n,
                             int64_t m, int64_t* __restrict__ out_n) {
  if (threadIdx.x == 0 && blockIdx.x == 0) *out_n = n - (int64_t)dropx[n] + m;
}

__global__ void k_fill_u64(uint64_t* a, uint64_t v, int64_t n) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) a[i] = v;
}

__global__ void k_find_bounds(Run a, int64_t n, const uint8_t* spill,
                              const uint8_t* bkeys, const uint64_t* brevs, int nb,
                              int64_t* out) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  // borders are <= KEYW (validated host-side); len=KEYW orders a border
  // before any longer key sharing its full 96B prefix, which is exactly
  // lower-bound semantics
  if (i < nb)
    out[i] = d_lb_range(a, spill, 0, n,
                        QKey{bkeys + (int64_t)i * KEYW, (uint32_t)KEYW, 0, nullptr},
                        brevs[i]);
}

__global__ void k_compact_mark(const uint64_t* __restrict__ meta,
                               const uint64_t* __restrict__ rev,
                               const uint64_t* __restrict__ vo, int64_t lo,
                               int64_t hi, uint64_t compact_rev,
                               uint64_t timeout_rev, uint64_t* __restrict__ keep) {
  int64_t i = lo + (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= hi) return;
  uint64_t r = rev[i], m = meta[i];
  // TTL expiry of "/events/" keys, checked BEFORE the rev skip (scanner.go:444-447)
  if (timeout_rev != 0 && (m & M_EVENTS)) {
    if (r == 0) {
      if (vo[i] <= timeout_rev) { keep[i] = 0; return; }  // scanner.go:577-581
    } else if (r <= timeout_rev) { keep[i] = 0; return; } // scanner.go:583-586
  }
  if (r > compact_rev) return;  // skipped rows survive (scanner.go:451-453)
  if (r == 0) {
    // 9B-flagged revision rows die unless objRev > compactRev (scanner.go:477-491)
    if ((m & M_FLAG9) && vo[i] <= compact_rev) keep[i] = 0;
    return;
  }
  if (m & M_TOMB) { keep[i] = 0; return; }  // scanner.go:471-475
  // superseded version: a newer same-key row with rev<=compactRev (scanner.go:464-469)
  if ((m & M_SAME_NEXT) && rev[i + 1] <= compact_rev) keep[i] = 0;
}

__global__ void k_heap_sizes(const uint64_t* __restrict__ keep,
                             const uint64_t* __restrict__ rev,
                             const uint64_t* __restrict__ meta,
                             uint64_t* __restrict__ sz, int64_t n) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  sz[i] = (keep[i] && rev[i] > 0) ? ((meta_vlen(meta[i]) + 15) & ~15ull) : 0;
}

__global__ void k_compact_scatter(const uint8_t* __restrict__ keysA,
                                  const uint64_t* __restrict__ metaA,
                                  const uint64_t* __restrict__ revA,
                                  const uint64_t* __restrict__ voA,
                                  const uint64_t* __restrict__ koA,
                                  const uint64_t* __restrict__ keep,
                                  const uint64_t* __restrict__ nidx,
                                  const uint64_t* __restrict__ heap_off,
                                  const uint64_t* __restrict__ spill_off,
                                  uint8_t* __restrict__ keysB,
                                  uint64_t* __restrict__ metaB,
                                  uint64_t* __restrict__ revB,
                                  uint64_t* __restrict__ voB,
                                  uint64_t* __restrict__ koB, int64_t n) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n || !keep[i]) return;
  int64_t j = (int64_t)nidx[i];
  revB[j] = revA[i];
  metaB[j] = metaA[i];
  voB[j] = revA[i] == 0 ? voA[i] : heap_off[i];
  koB[j] = meta_klen(metaA[i]) > (uint32_t)KEYW ? spill_off[i] : 0;
  const uint64_t* ks = (const uint64_t*)(keysA + i * KEYW);
  uint64_t* kd = (uint64_t*)(keysB + j * KEYW);
#pragma unroll
  for (int k = 0; k < KEYW / 8; ++k) kd[k] = ks[k];
}

// key-spill compaction (keys > KEYW): surviving tails are packed into the
// fresh spill heap, mirroring the value-heap sweep
__global__ void k_spill_sizes(const uint64_t* __restrict__ keep,
                              const uint64_t* __restrict__ meta,
                              uint64_t* __restrict__ sz, int64_t n) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  uint32_t kl = meta_klen(meta[i]);
  sz[i] = (keep[i] && kl > (uint32_t)KEYW) ? (((uint64_t)kl - KEYW + 15) & ~15ull) : 0;
}

__global__ void k_spill_scatter(const uint64_t* __restrict__ keep,
                                const uint64_t* __restrict__ meta,
                                const uint64_t* __restrict__ koA,
                                const uint64_t* __restrict__ spill_off,
                                const uint8_t* __restrict__ spillA,
                                uint8_t* __restrict__ spillB, int64_t n) {
  int64_t i = (int64_t)blockIdx.x * (blockDim.x / 64) + (threadIdx.x >> 6);
  int lane = threadIdx.x & 63;
  if (i >= n || !keep[i]) return;
  uint32_t kl = meta_klen(meta[i]);
  if (kl <= (uint32_t)KEYW) return;
  uint32_t t = kl - (uint32_t)KEYW;
  const uint8_t* src = spillA + koA[i];
  uint8_t* dst = spillB + spill_off[i];
  for (uint32_t x = lane; x < t; x += 64) dst[x] = src[x];
}

__global__ void k_heap_scatter(const uint64_t* __restrict__ keep,
                               const uint64_t* __restrict__ rev,
                               const uint64_t* __restrict__ meta,
                               const uint64_t* __restrict__ voA,
                               const uint64_t* __restrict__ heap_off,
                               const uint8_t* __restrict__ heapA,
                               uint8_t* __restrict__ heapB, int64_t n) {
  int64_t i = (int64_t)blockIdx.x * (blockDim.x / 64) + (threadIdx.x >> 6);
  int lane = threadIdx.x & 63;
  if (i >= n || !keep[i] || rev[i] == 0) return;
  uint64_t bytes = (meta_vlen(meta[i]) + 15) & ~15ull;
  const uint4* src = (const uint4*)(heapA + voA[i]);
  uint4* dst = (uint4*)(heapB + heap_off[i]);
  for (uint64_t w = lane; w < bytes / 16; w += 64) dst[w] = src[w];
}

// ---- merge (memtable flush): delta ranks + scatter ----------------------
// inverted small-insert merge (m <= 1024 new rows into an n-row run): ONE
// block searches each NEW row into the big run — m*log2(n) work instead of
// k_merge_rank's n*log2(m) — writes the new rows at their final slots, and
// leaves a difference array v[0..n+1] whose prefix sum is each base row's
// slot shift (insertions +1 at ub_j, drops -1 after lb_j). drop[] marks
// replaced base rows ((key,rev)-equal memtable revision-rows). new_n -> s_d.
__global__ void k_merge_insert_inv(
    Run a, int64_t n, Run dnew, int64_t m, const uint8_t* __restrict__ spill,
    uint64_t* __restrict__ v /*[n+2] zeroed*/,
    uint64_t* __restrict__ drop /*[n] zeroed*/,
    uint64_t* __restrict__ outpos /*[m] delta dest slots (for the fixup)*/,
    uint8_t* keysB, uint64_t* metaB, uint64_t* revB, uint64_t* voB,
    uint64_t* koB, int64_t* __restrict__ newn_out) {
  __shared__ uint16_t ldup[1024];
  int64_t j = threadIdx.x;
  bool has = j < m;
  int64_t lb = n;
  bool dup = false;
  if (has) {
    QKey k = row_qk(dnew, spill, j);
    lb = d_lb_range(a, spill, 0, n, k, dnew.rev[j]);
    dup = lb < n && a.rev[lb] == dnew.rev[j] &&
          rowcmp_q(a, spill, lb, k) == 0;
  }
  ldup[threadIdx.x] = dup ? 1 : 0;
  __syncthreads();
  // exclusive prefix of dup flags over the block (m <= 1024)
  int64_t dupx = 0, total = 0;
  for (int t = 0; t < (int)m; ++t) {
    if (t < (int)j) dupx += ldup[t];
    total += ldup[t];
  }
  if (threadIdx.x == 0 && newn_out) *newn_out = n + m - total;
  if (!has) return;
  if (dup) {
    drop[lb] = 1;
    atomicAdd(&v[lb + 1], (uint64_t)-1ll);
  }
  atomicAdd(&v[lb + (dup ? 1 : 0)], 1ull);  // ub_j = lb + dup
  int64_t out = lb - dupx + j;
  outpos[j] = (uint64_t)out;
  revB[out] = dnew.rev[j];
  metaB[out] = dnew.meta[j];
  voB[out] = dnew.vo[j];
  koB[out] = dnew.ko[j];
  const uint4* ks = (const uint4*)(dnew.keys + j * KEYW);
  uint4* kd = (uint4*)(keysB + out * KEYW);
#pragma unroll
  for (int t = 0; t < KEYW / 16; ++t) kd[t] = ks[t];
}

// multi-block inverted merge rank (any m into n>0 rows): each NEW row
// binary-searches the big run once — m*log2(n) probes instead of the
// forward path's n*log2(m) — and leaves the same difference array v as the
// single-block variant. dup flags go to dup[] (scanned separately to give
// each new row its drops-before count).
__global__ void k_merge_rank_inv_big(Run a, int64_t n, Run dnew, int64_t m,
                                     const uint8_t* __restrict__ spill,
                                     uint64_t* __restrict__ v /*[n+2] zeroed*/,
                                     uint64_t* __restrict__ drop /*[n] zeroed*/,
                                     uint64_t* __restrict__ dup /*[m+1] zeroed*/,
                                     uint64_t* __restrict__ lbs /*[m]*/) {
  int64_t j = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (j >= m) return;
  QKey k = row_qk(dnew, spill, j);
  int64_t lb = d_lb_range(a, spill, 0, n, k, dnew.rev[j]);
  bool isdup = lb < n && a.rev[lb] == dnew.rev[j] &&
               rowcmp_q(a, spill, lb, k) == 0;
  lbs[j] = (uint64_t)lb;
  if (isdup) {
    dup[j] = 1;
    drop[lb] = 1;
    atomicAdd(&v[lb + 1], (uint64_t)-1ll);
  }
  atomicAdd(&v[lb + (isdup ? 1 : 0)], 1ull);
}

// delta scatter for the big inverted path: slot = lb_j + j - dupx[j];
// rewrites lbs[j] with the final slot for the same_next fixup
__global__ void k_merge_scatter_delta2(
    Run dnew, int64_t m, const uint64_t* __restrict__ dupx /*excl scan*/,
    uint64_t* __restrict__ lbs, uint8_t* keysB, uint64_t* metaB,
    uint64_t* revB, uint64_t* voB, uint64_t* koB) {
  int64_t j = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (j >= m) return;
  int64_t out = (int64_t)lbs[j] + j - (int64_t)dupx[j];
  lbs[j] = (uint64_t)out;
  revB[out] = dnew.rev[j];
  metaB[out] = dnew.meta[j];
  voB[out] = dnew.vo[j];
  koB[out] = dnew.ko[j];
  const uint4* ks = (const uint4*)(dnew.keys + j * KEYW);
  uint4* kd = (uint4*)(keysB + out * KEYW);
#pragma unroll
  for (int t = 0; t < KEYW / 16; ++t) kd[t] = ks[t];
}

__global__ void k_set_newn(const uint64_t* __restrict__ dupx, int64_t n,
                           int64_t m, int64_t* __restrict__ newn_out) {
  if (threadIdx.x == 0 && blockIdx.x == 0)
    *newn_out = n + m - (int64_t)dupx[m];
}

// base-row scatter for the inverted path: slot = i + excl_scan(v)[i+1]
__global__ void k_merge_scatter_base2(
    Run a, const uint64_t* __restrict__ sv /*excl scan of v, [n+2]*/,
    const uint64_t* __restrict__ drop, uint8_t* keysB, uint64_t* metaB,
    uint64_t* revB, uint64_t* voB, uint64_t* koB, int64_t n) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n || drop[i]) return;
  int64_t j = i + (int64_t)sv[i + 1];
  revB[j] = a.rev[i];
  metaB[j] = a.meta[i];
  voB[j] = a.vo[i];
  koB[j] = a.ko[i];
  const uint4* ks = (const uint4*)(a.keys + i * KEYW);
  uint4* kd = (uint4*)(keysB + j * KEYW);
#pragma unroll
  for (int t = 0; t < KEYW / 16; ++t) kd[t] = ks[t];
}

// same_next fixup after an inverted insert: only rows adjacent to an
// inserted slot can change (relative order of surviving base rows is
// preserved and their meta bits travel with them); recompute at out_j-1 and
// out_j for every inserted row
__global__ void k_same_next_fixup(uint8_t* __restrict__ keys,
                                  uint64_t* __restrict__ meta,
                                  const uint64_t* __restrict__ ko,
                                  const uint8_t* __restrict__ spill,
                                  int64_t new_n,
                                  const uint64_t* __restrict__ outpos,
                                  int64_t m) {
  int64_t t = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (t >= 2 * m) return;
  int64_t i = (int64_t)outpos[t >> 1] - (t & 1);
  if (i < 0 || i >= new_n) return;
  uint64_t mm = meta[i];
  bool same = i + 1 < new_n && rows_same_key(keys, meta, ko, spill, i, i + 1);
  meta[i] = same ? (mm | M_SAME_NEXT) : (mm & ~M_SAME_NEXT);
}

__global__ void k_merge_scatter_delta(
    Run a, int64_t n, Run dnew, int64_t m, const uint8_t* __restrict__ spill,
    const uint64_t* __restrict__ dropx,
    uint8_t* keysB, uint64_t* metaB, uint64_t* revB, uint64_t* voB,
    uint64_t* koB) {
  int64_t j = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (j >= m) return;
  QKey k = row_qk(dnew, spill, j);
  int64_t lb = d_lb_range(a, spill, 0, n, k, dnew.rev[j]);
  int64_t out = lb - (int64_t)dropx[lb] + j;
  revB[out] = dnew.rev[j];
  metaB[out] = dnew.meta[j];
  voB[out] = dnew.vo[j];
  koB[out] = dnew.ko[j];
  const uint64_t* ks = (const uint64_t*)(dnew.keys + j * KEYW);
  uint64_t* kd = (uint64_t*)(keysB + out * KEYW);
#pragma unroll
  for (int kk = 0; kk < KEYW / 8; ++kk) kd[kk] = ks[kk];
}

// single-workgroup merge for small runs (n+m <= 8192): rank, drop-scan,
// scatter and same_next in ONE launch — the multi-kernel path costs ~10
// launches plus two host syncs, which dominates small per-step delta merges.
__global__ void k_merge_small(Run src, int64_t n, Run dnew, int64_t m,
                              const uint8_t* __restrict__ spill,
                              uint8_t* __restrict__ okeys,
                              uint64_t* __restrict__ ometa,
                              uint64_t* __restrict__ orev,
                              uint64_t* __restrict__ ovo,
                              uint64_t* __restrict__ oko,
                              int64_t* __restrict__ out_n) {
  __shared__ uint16_t dropx[8193];  // exclusive drop counts (n <= 8192)
  __shared__ uint16_t partial[257];
  int t = threadIdx.x;
  const int T = blockDim.x;
  // per-thread chunk of src rows: drop flags + partial sums
  int chunk = (int)((n + T - 1) / T);
  int i0 = t * chunk;
  int i1 = min((int64_t)i0 + chunk, n);
  uint16_t cnt = 0;
  for (int i = i0; i < i1; ++i) {
    QKey k = row_qk(src, spill, i);
    int64_t lb = d_lb_range(dnew, spill, 0, m, k, src.rev[i]);
    bool drop = lb < m && dnew.rev[lb] == src.rev[i] &&
                rowcmp_q(dnew, spill, lb, k) == 0;
    dropx[i] = drop ? 1 : 0;
    cnt += drop;
  }
  partial[t] = cnt;
  __syncthreads();
  if (t == 0) {
    uint16_t acc = 0;
    for (int k = 0; k < T; ++k) { uint16_t v = partial[k]; partial[k] = acc; acc += v; }
    partial[T] = acc;
  }
  __syncthreads();
  // exclusive scan within chunk
  {
    uint16_t acc = partial[t];
    for (int i = i0; i < i1; ++i) { uint16_t v = dropx[i]; dropx[i] = acc; acc += v; }
  }
  __syncthreads();
  int64_t dropped = partial[T];
  int64_t new_n = n - dropped + m;
  if (t == 0) *out_n = new_n;
  // scatter src rows
  for (int64_t i = t; i < n; i += T) {
    QKey k = row_qk(src, spill, i);
    int64_t lb = d_lb_range(dnew, spill, 0, m, k, src.rev[i]);
    bool drop = lb < m && dnew.rev[lb] == src.rev[i] &&
                rowcmp_q(dnew, spill, lb, k) == 0;
    if (drop) continue;
    int64_t j = i - (int64_t)dropx[i] + lb;
    orev[j] = src.rev[i];
    ometa[j] = src.meta[i];
    ovo[j] = src.vo[i];
    oko[j] = src.ko[i];
    const uint64_t* ks = (const uint64_t*)(src.keys + i * KEYW);
    uint64_t* kd = (uint64_t*)(okeys + j * KEYW);
#pragma unroll
    for (int k2 = 0; k2 < KEYW / 8; ++k2) kd[k2] = ks[k2];
  }
  // scatter new rows
  for (int64_t j = t; j < m; j += T) {
    QKey k = row_qk(dnew, spill, j);
    int64_t lb = d_lb_range(src, spill, 0, n, k, dnew.rev[j]);
    int64_t dpx = lb < n ? dropx[lb] : (int64_t)dropped;
    int64_t o = lb - dpx + j;
    orev[o] = dnew.rev[j];
    ometa[o] = dnew.meta[j];
    ovo[o] = dnew.vo[j];
    oko[o] = dnew.ko[j];
    const uint64_t* ks = (const uint64_t*)(dnew.keys + j * KEYW);
    uint64_t* kd = (uint64_t*)(okeys + o * KEYW);
#pragma unroll
    for (int k2 = 0; k2 < KEYW / 8; ++k2) kd[k2] = ks[k2];
  }
  __syncthreads();
  // same_next over the merged rows (full-key equality incl. spill tails)
  for (int64_t i = t; i < new_n; i += T) {
    bool same = i + 1 < new_n &&
                rows_same_key(okeys, ometa, oko, spill, i, i + 1);
    ometa[i] = same ? (ometa[i] | M_SAME_NEXT) : (ometa[i] & ~M_SAME_NEXT);
  }
}

// ---- watch fan-out over the DEVICE-RESIDENT event log -------------------
// (watch.go:119-159 per-watcher predicate; north_star: "GPU-resident event
// log"). The log is a ring of (key96, rev) columns in HBM, pushed once per
// event batch; the fan-out filter stages each batch through LDS and tests
// all watchers against it; catch-up (Ring.FindEvents, ring.go:84-118) is a
// thread-per-event scan over the resident ring. Values/full events stay in
// the host ring for materialization at poll time — matching the reference,
// which fans out shared batch POINTERS and filters in the consumer
// (watcherhub.go:78-100, watch.go:119-159).

// one 64-thread block per (64-watcher slice x 64-event chunk): at 10k
// watchers / 512 events that is 157x8 = 1256 single-wave blocks (the old
// one-thread-per-watcher shape launched only W/256 = 40 blocks and left the
// chip idle). Each block stages its event chunk (key96+rev, 6.5 KB) in LDS;
// each thread (= one watcher) tests the 64 events and emits exactly one
// bitmap word — no atomics, no cross-wave syncs, no ballot needed.
__global__ void k_watch_filter2(const uint8_t* __restrict__ er_keys,
                                const uint64_t* __restrict__ er_rev,
                                int64_t ring_cap, int64_t base_seq,
                                int64_t count,
                                const uint8_t* __restrict__ wpfx,
                                const uint32_t* __restrict__ wplen,
                                const uint64_t* __restrict__ wfrom,
                                const uint32_t* __restrict__ wlive, int64_t W,
                                uint64_t* __restrict__ bitmap, int64_t words) {
  __shared__ uint64_t lkeys[64 * (KEYW / 8)];
  __shared__ uint64_t lrev[64];
  const int64_t w = (int64_t)blockIdx.x * 64 + threadIdx.x;
  const int64_t c0 = (int64_t)blockIdx.y << 6;
  const int64_t cn = min((int64_t)64, count - c0);
  // stage the chunk: consecutive threads load consecutive u64s (ring slots
  // may wrap, so the slot is recomputed per element)
  for (int64_t v = threadIdx.x; v < cn * (KEYW / 8); v += 64) {
    int64_t j = v / (KEYW / 8), k = v % (KEYW / 8);
    int64_t slot = (base_seq + c0 + j) % ring_cap;
    lkeys[v] = ((const uint64_t*)(er_keys + slot * KEYW))[k];
  }
  if (threadIdx.x < cn) {
    int64_t slot = (base_seq + c0 + threadIdx.x) % ring_cap;
    lrev[threadIdx.x] = er_rev[slot];
  }
  __syncthreads();
  if (w >= W) return;
  uint64_t acc = 0;
  if (wlive[w]) {
    const uint8_t* pfx = wpfx + w * KEYW;
    const uint32_t plen = wplen[w];
    const uint64_t from = wfrom[w];
    for (int64_t j = 0; j < cn; ++j) {
      bool ok = lrev[j] >= from;  // filterByRevision (watch.go:152-158)
      if (ok) {                   // filterByPrefix (watch.go:139-149)
        const uint8_t* k = (const uint8_t*)(lkeys + j * (KEYW / 8));
        uint32_t b = 0;
        for (; b + 8 <= plen; b += 8)
          if (*(const uint64_t*)(pfx + b) != *(const uint64_t*)(k + b)) break;
        if (b + 8 <= plen) ok = false;
        else
          for (; b < plen && ok; ++b)
            if (pfx[b] != k[b]) ok = false;
      }
      if (ok) acc |= 1ull << j;
    }
  }
  bitmap[w * words + blockIdx.y] = acc;
}

// catch-up: ONE watcher's prefix over a resident ring span, thread-per-event
__global__ void k_watch_catchup(const uint8_t* __restrict__ er_keys,
                                const uint64_t* __restrict__ er_rev,
                                int64_t ring_cap, int64_t base_seq,
                                int64_t count, const uint8_t* __restrict__ pfx,
                                uint32_t plen, uint64_t from_rev,
                                uint64_t* __restrict__ words) {
  int64_t e = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int lane = threadIdx.x & 63;
  bool ok = false;
  if (e < count) {
    int64_t slot = (base_seq + e) % ring_cap;
    ok = er_rev[slot] >= from_rev;
    if (ok) {
      const uint8_t* k = er_keys + slot * KEYW;
      uint32_t b = 0;
      for (; b + 8 <= plen; b += 8)
        if (*(const uint64_t*)(pfx + b) != *(const uint64_t*)(k + b)) break;
      if (b + 8 <= plen) ok = false;
      else
        for (; b < plen && ok; ++b)
          if (pfx[b] != k[b]) ok = false;
    }
  }
  uint64_t b = __ballot(ok);
  if (lane == 0 && e < count) words[e >> 6] = b;
}

// ------------------------------------------------------------------ Impl ---

struct Slab::Impl {
  int device = 0;
  hipStream_t stream = nullptr;
  int64_t max_rows = 0, heap_cap = 0;
  int64_t n = 0, heap_used_ = 0;

  struct Col {
    uint8_t* keys = nullptr;
    uint64_t *meta = nullptr, *rev = nullptr, *vo = nullptr, *ko = nullptr;
    Run run() const { return Run{keys, meta, rev, vo, ko}; }
  };
  Col A, B;
  Col DA, DB;            // delta-run ping-pong
  int64_t dn = 0;        // delta-run rows
  int64_t delta_cap = 0; // rows per delta buffer
  uint8_t *heapA = nullptr, *heapB = nullptr;
  // key-spill heap (tails of keys > KEYW), shared by base+delta; compacted
  // with the rows in Compact
  uint8_t *spillA = nullptr, *spillB = nullptr;
  int64_t spill_cap = 0, spill_used_ = 0;
  // per-batch query-tail buffer (bounds/get keys > KEYW)
  uint8_t* d_qtails = nullptr;
  int64_t qt_cap = 0;
  bool ensure_qtails(const std::string& qt, std::string* err) {
    if ((int64_t)qt.size() > qt_cap) {
      if (d_qtails) (void)hipFree(d_qtails);
      int64_t cap = (int64_t)qt.size() * 2 + 4096;
      HIP_CHECK(hipMalloc(&d_qtails, cap));
      qt_cap = cap;
    }
    if (!qt.empty())
      HIP_CHECK(hipMemcpyAsync(d_qtails, qt.data(), qt.size(),
                               hipMemcpyHostToDevice, stream));
    return true;
  }

  // scan scratch (u64, shared across ops)
  uint64_t* shadow = nullptr;  // [max_rows] base shadow-revision column
  uint64_t* d_outpos = nullptr;  // [1024] inverted-insert dest slots
  // big inverted merge scratch: dup flags / their scan / new-row lower
  // bounds (then final slots), all m-sized, grown on demand
  uint64_t *d_mdup = nullptr, *d_mdupx = nullptr, *d_mlbs = nullptr;
  uint8_t* h_uploads[2] = {nullptr, nullptr};  // pinned delta-upload mirrors
  hipEvent_t ev_up[2] = {nullptr, nullptr};
  int up_idx = 0;
  DevRangeQ* h_qs[2] = {nullptr, nullptr};  // pinned query-upload mirrors
  hipEvent_t ev_qs[2] = {nullptr, nullptr};
  int qs_idx = 0;
  // pinned double-buffered staging for heap/spill appends: the old pageable
  // copies needed a FULL stream sync per call (in the pipelined bench that
  // waited out the previous step's scan+gather kernels)
  uint8_t* h_blob[2] = {nullptr, nullptr};
  int64_t h_blob_cap[2] = {0, 0};
  hipEvent_t ev_blob[2] = {nullptr, nullptr};
  int blob_idx = 0;
  uint8_t* stage_blob(const void* src, int64_t len, std::string* err) {
    int slot = blob_idx ^= 1;
    if (!ev_blob[slot]) {
      hipError_t _e = hipEventCreate(&ev_blob[slot]);
      if (_e != hipSuccess) { if (err) *err = hipGetErrorString(_e); return nullptr; }
    }
    (void)hipEventSynchronize(ev_blob[slot]);
    if (len > h_blob_cap[slot]) {
      if (h_blob[slot]) (void)hipHostFree(h_blob[slot]);
      int64_t cap = len + len / 2 + 4096;
      hipError_t _e = hipHostMalloc(&h_blob[slot], cap);
      if (_e != hipSuccess) { if (err) *err = hipGetErrorString(_e); h_blob[slot] = nullptr; h_blob_cap[slot] = 0; return nullptr; }
      h_blob_cap[slot] = cap;
    }
    memcpy(h_blob[slot], src, len);
    return h_blob[slot];
  }
  int64_t mergebuf_cap = 0;
  bool ensure_mergebuf(int64_t m, std::string* err) {
    if (m + 1 <= mergebuf_cap) return true;
    int64_t cap = m + m / 2 + 64;
    for (void* q : {(void*)d_mdup, (void*)d_mdupx, (void*)d_mlbs})
      if (q) (void)hipFree(q);
    HIP_CHECK(hipMalloc(&d_mdup, cap * 8));
    HIP_CHECK(hipMalloc(&d_mdupx, cap * 8));
    HIP_CHECK(hipMalloc(&d_mlbs, cap * 8));
    mergebuf_cap = cap;
    return true;
  }
  uint64_t *s_a = nullptr, *s_b = nullptr, *s_c = nullptr, *s_d = nullptr,
           *s_e = nullptr;  // max_rows+2
  uint64_t *lv1 = nullptr, *lv1o = nullptr, *lv2 = nullptr, *lv2o = nullptr,
           *lv3 = nullptr, *lv3o = nullptr;

  // range/get scratch
  int max_q = 1024;
  int scan_t = 1024;            // KB_SCAN_T: threads per scan block
  int gather_gwl = 4;           // KB_GATHER_GW: log2 lanes per record group
  int gather_t = 512;           // KB_GATHER_T: threads per gather_copy block
  int gather_mode = 1;          // KB_GATHER_MODE: 1=record groups, 2=wave/record
  int64_t max_cap = 4352;       // winners per query cap (>= limit+1 for etcd's 500)
  int64_t arena_bytes = 384ll << 20;
  DevRangeQ* d_qs = nullptr;
  DevGetQ* d_gq = nullptr;
  uint64_t* d_rows = nullptr;   // max_q*max_cap (base winners)
  uint64_t* d_rows2 = nullptr;  // delta winners
  uint64_t* d_rowsm = nullptr;  // merged winners
  int64_t* d_bounds4 = nullptr;  // [4*max_q] k_range_bounds results
  int64_t* d_offs = nullptr;    // max_q*max_cap
  int64_t *d_found = nullptr, *d_total = nullptr, *d_gbytes = nullptr;
  int32_t *d_ovf = nullptr, *d_found32 = nullptr;
  uint64_t *d_orev = nullptr, *d_ometa = nullptr;
  unsigned long long *d_scanned = nullptr, *d_bytes = nullptr;
  uint8_t* h_resmeta = nullptr;  // pinned mirror of the resmeta block
  int64_t resmeta_bytes = 0;
  uint8_t* d_gbuf = nullptr;    // arena
  uint8_t* d_pack = nullptr;    // arena
  int64_t* d_goffs = nullptr;   // max_q+1
  int64_t* d_bounds = nullptr;  // compact bounds
  uint8_t* d_bkeys = nullptr;
  uint64_t* d_brevs = nullptr;

  // delta upload scratch (grown on demand)
  uint8_t* d_dkeys = nullptr;
  uint64_t *d_dmeta = nullptr, *d_drev = nullptr, *d_dvo = nullptr,
           *d_dko = nullptr;
  int64_t upload_cap = 0;

  // watcher table
  int64_t wcap = 0;
  uint8_t* d_wpfx = nullptr;
  uint32_t *d_wplen = nullptr, *d_wlive = nullptr;
  uint64_t* d_wfrom = nullptr;
  // device-resident event log ring (keys+revs; values stay host-side)
  uint8_t* er_keys = nullptr;
  uint64_t* er_rev = nullptr;
  int64_t er_cap = 0;
  uint64_t* d_bitmap = nullptr;
  int64_t bitmap_cap = 0;

  // KB_SCAN_DBG=1: per-phase wall_clock64 cycle sums from k_range_scan2
  // ([0]=bounds [1]=pass1a [2]=pass1b [3]=scatter [4]=merge [5]=block total)
  unsigned long long* d_dbg = nullptr;
  double wall_khz = 0;

  hipEvent_t ev0 = nullptr, ev1 = nullptr, ev2 = nullptr, ev3 = nullptr;
  hipEvent_t ev_g0 = nullptr, ev_g1 = nullptr;  // async get lookup
  uint64_t* h_gmeta = nullptr;   // pinned [2*max_q]: orev | ometa
  int32_t* h_gfound = nullptr;   // pinned [max_q]

  // host staging (pinned for fast D2H)
  uint8_t* h_pack = nullptr;
  int64_t h_pack_cap = 0;
  bool ensure_hpack(int64_t need, std::string* err) {
    if (need <= h_pack_cap) return true;
    if (h_pack) (void)hipHostFree(h_pack);
    int64_t cap = need + need / 2;
    HIP_CHECK(hipHostMalloc(&h_pack, cap));
    h_pack_cap = cap;
    return true;
  }

  // pipelined payload D2H (bench d2h mode): ping-pong pack arenas; each
  // batch's payload copy runs on the copy stream and overlaps the NEXT
  // batch's kernels on the compute stream (the PCIe copy and the scan use
  // different engines). Slot reuse waits on that slot's prior copy.
  hipStream_t cstream = nullptr;
  uint8_t* d_packs[2] = {nullptr, nullptr};  // [0] aliases d_pack
  uint8_t* h_packs[2] = {nullptr, nullptr};
  int64_t h_packs_cap[2] = {0, 0};
  hipEvent_t ev_pk[2] = {nullptr, nullptr}, ev_cp[2] = {nullptr, nullptr};
  bool cp_busy[2] = {false, false};
  int pk_idx = 0;
  double* pack_ms_acc = nullptr;  // -> perf.pack_d2h_ms (set by Slab)

  bool ensure_pipe(int idx, int64_t need, std::string* err) {
    if (!cstream) {
      HIP_CHECK(hipStreamCreate(&cstream));
      HIP_CHECK(hipEventCreate(&ev_pk[0]));
      HIP_CHECK(hipEventCreate(&ev_pk[1]));
      HIP_CHECK(hipEventCreate(&ev_cp[0]));
      HIP_CHECK(hipEventCreate(&ev_cp[1]));
      d_packs[0] = d_pack;
    }
    if (idx == 1 && !d_packs[1]) HIP_CHECK(hipMalloc(&d_packs[1], arena_bytes));
    if (need > h_packs_cap[idx]) {
      if (h_packs[idx]) (void)hipHostFree(h_packs[idx]);
      int64_t cap = need + need / 2;
      HIP_CHECK(hipHostMalloc(&h_packs[idx], cap));
      h_packs_cap[idx] = cap;
    }
    return true;
  }

  // wait for slot idx's in-flight copy (accumulating its event time)
  void wait_slot(int idx) {
    if (!cp_busy[idx]) return;
    (void)hipEventSynchronize(ev_cp[idx]);
    float ms = 0;
    if (hipEventElapsedTime(&ms, ev_pk[idx], ev_cp[idx]) == hipSuccess &&
        pack_ms_acc)
      *pack_ms_acc += ms;
    cp_busy[idx] = false;
  }

  ~Impl() {
    wait_slot(0);
    wait_slot(1);
    if (d_packs[1]) (void)hipFree(d_packs[1]);
    for (int i = 0; i < 2; ++i) if (h_packs[i]) (void)hipHostFree(h_packs[i]);
    for (hipEvent_t e : {ev_pk[0], ev_pk[1], ev_cp[0], ev_cp[1]})
      if (e) (void)hipEventDestroy(e);
    if (cstream) (void)hipStreamDestroy(cstream);
    for (void* p : {(void*)A.keys, (void*)A.meta, (void*)A.rev, (void*)A.vo,
                    (void*)B.keys, (void*)B.meta, (void*)B.rev, (void*)B.vo,
                    (void*)DA.keys, (void*)DA.meta, (void*)DA.rev, (void*)DA.vo,
                    (void*)DB.keys, (void*)DB.meta, (void*)DB.rev, (void*)DB.vo,
                    (void*)d_rows2, (void*)d_rowsm,
                    (void*)heapA, (void*)heapB, (void*)s_a, (void*)s_b,
                    (void*)s_c, (void*)s_d, (void*)s_e, (void*)lv1, (void*)lv1o, (void*)lv2,
                    (void*)lv2o, (void*)lv3, (void*)lv3o, (void*)d_qs,
                    (void*)d_gq, (void*)d_rows, (void*)d_offs,
                    (void*)d_found,  // base of the resmeta block
                    (void*)d_found32, (void*)d_orev, (void*)d_ometa,
                    (void*)shadow, (void*)d_outpos, (void*)d_mdup, (void*)d_mdupx,
                    (void*)d_mlbs, (void*)d_gbuf,
                    (void*)d_pack, (void*)d_goffs, (void*)d_bounds,
                    (void*)d_bkeys, (void*)d_brevs, (void*)d_dkeys,
                    (void*)d_dmeta, (void*)d_drev, (void*)d_dvo, (void*)d_dko,
                    (void*)spillA, (void*)spillB, (void*)d_qtails,
                    (void*)A.ko, (void*)B.ko, (void*)DA.ko, (void*)DB.ko,
                    (void*)d_wpfx,
                    (void*)d_wplen, (void*)d_wlive, (void*)d_wfrom,
                    (void*)er_keys, (void*)er_rev, (void*)d_bitmap}) {
      if (p) (void)hipFree(p);
    }
    if (h_pack) (void)hipHostFree(h_pack);
    if (h_resmeta) (void)hipHostFree(h_resmeta);
    for (int i = 0; i < 2; ++i) {
      if (h_uploads[i]) (void)hipHostFree(h_uploads[i]);
      if (ev_up[i]) (void)hipEventDestroy(ev_up[i]);
      if (h_qs[i]) (void)hipHostFree(h_qs[i]);
      if (ev_qs[i]) (void)hipEventDestroy(ev_qs[i]);
      if (h_blob[i]) (void)hipHostFree(h_blob[i]);
      if (ev_blob[i]) (void)hipEventDestroy(ev_blob[i]);
    }
    if (h_gmeta) (void)hipHostFree(h_gmeta);
    if (h_gfound) (void)hipHostFree(h_gfound);
    if (ev_g0) (void)hipEventDestroy(ev_g0);
    if (ev_g1) (void)hipEventDestroy(ev_g1);
    if (ev0) (void)hipEventDestroy(ev0);
    if (ev1) (void)hipEventDestroy(ev1);
    if (ev2) (void)hipEventDestroy(ev2);
    if (ev3) (void)hipEventDestroy(ev3);
    if (stream) (void)hipStreamDestroy(stream);
  }

  // exclusive scan of d_in[0..cnt) into d_out, total via small D2H
  bool scan(const uint64_t* d_in, uint64_t* d_out, int64_t cnt, uint64_t* total,
            std::string* err) {
    int64_t nb1 = ceil_div(cnt, 256);
    hipLaunchKernelGGL(k_block_scan, dim3((uint32_t)nb1), dim3(256), 0, stream,
                       d_in, d_out, lv1, cnt);
    if (nb1 > 1) {
      int64_t nb2 = ceil_div(nb1, 256);
      hipLaunchKernelGGL(k_block_scan, dim3((uint32_t)nb2), dim3(256), 0, stream,
                         lv1, lv1o, lv2, nb1);
      if (nb2 > 1) {
        int64_t nb3 = ceil_div(nb2, 256);
        hipLaunchKernelGGL(k_block_scan, dim3((uint32_t)nb3), dim3(256), 0,
                           stream, lv2, lv2o, lv3, nb2);
        if (nb3 > 1) {
          // 4th level covers up to 256^4 = 4.3G rows (nb3 <= 256 by max_rows)
          if (nb3 > 256) { if (err) *err = "scan: size too large"; return false; }
          hipLaunchKernelGGL(k_block_scan, dim3(1), dim3(256), 0, stream, lv3,
                             lv3o, lv3 + 256, nb3);
          hipLaunchKernelGGL(k_add_offsets, dim3((uint32_t)nb3), dim3(256), 0,
                             stream, lv2o, lv3o, nb2);
        }
        hipLaunchKernelGGL(k_add_offsets, dim3((uint32_t)nb2), dim3(256), 0,
                           stream, lv1o, lv2o, nb1);
      }
      hipLaunchKernelGGL(k_add_offsets, dim3((uint32_t)nb1), dim3(256), 0,
                         stream, d_out, lv1o, cnt);
    }
    if (total) {
      uint64_t last_out = 0, last_in = 0;
      HIP_CHECK(hipMemcpyAsync(&last_out, d_out + (cnt - 1), 8,
                               hipMemcpyDeviceToHost, stream));
      HIP_CHECK(hipMemcpyAsync(&last_in, const_cast<uint64_t*>(d_in) + (cnt - 1),
                               8, hipMemcpyDeviceToHost, stream));
      HIP_CHECK(hipStreamSynchronize(stream));
      *total = last_out + last_in;
    }
    return true;
  }

  bool ensure_delta(int64_t m, std::string* err) {
    if (m <= upload_cap) return true;
    int64_t cap = m + m / 2 + 1024;
    // one device block + one pinned mirror: per-step uploads become a single
    // host memcpy + ONE async H2D (five pageable copies each stalled the
    // host on the staging pool)
    if (d_dkeys) (void)hipFree(d_dkeys);
    for (int i = 0; i < 2; ++i) {
      if (h_uploads[i]) (void)hipHostFree(h_uploads[i]);
      HIP_CHECK(hipHostMalloc(&h_uploads[i], cap * (KEYW + 32)));
      if (!ev_up[i]) HIP_CHECK(hipEventCreate(&ev_up[i]));
    }
    int64_t bytes = cap * (KEYW + 32);
    HIP_CHECK(hipMalloc(&d_dkeys, bytes));
    d_dmeta = (uint64_t*)(d_dkeys + cap * KEYW);
    d_drev = d_dmeta + cap;
    d_dvo = d_drev + cap;
    d_dko = d_dvo + cap;
    upload_cap = cap;
    return true;
  }

  // generic two-run merge: src (n rows) + newer (m rows, device arrays,
  // rev-rows replace) -> dst; recomputes same_next. dst must hold n+m rows.
  bool mergeRuns(const Col& src, int64_t n, const Run& dnew,
                 int64_t m, Col& dst, int64_t* out_n, std::string* err,
                 bool device_newn_ok = false) {
    if (n + m <= 2048) {  // single-launch path only where one CU wins
      hipLaunchKernelGGL(k_merge_small, dim3(1), dim3(256), 0, stream,
                         src.run(), n, dnew, m, spillA,
                         dst.keys, dst.meta, dst.rev, dst.vo, dst.ko,
                         (int64_t*)s_d);
      if (device_newn_ok) return true;  // caller knows new_n; stay async
      int64_t nn = 0;
      HIP_CHECK(hipMemcpyAsync(&nn, s_d, 8, hipMemcpyDeviceToHost, stream));
      HIP_CHECK(hipStreamSynchronize(stream));
      *out_n = nn;
      return true;
    }
    if (m > 0 && m <= 1024 && n > 0) {
      // inverted small-insert path (the per-step txn merge): m searches into
      // the n-row run, not n searches into the m new rows
      if (!d_outpos) HIP_CHECK(hipMalloc(&d_outpos, 1024 * 8));
      HIP_CHECK(hipMemsetAsync(s_a, 0, (n + 2) * 8, stream));
      HIP_CHECK(hipMemsetAsync(s_b, 0, n * 8, stream));
      hipLaunchKernelGGL(k_merge_insert_inv, dim3(1), dim3(1024), 0, stream,
                         src.run(), n, dnew, m, spillA, s_a, s_b, d_outpos,
                         dst.keys, dst.meta, dst.rev, dst.vo, dst.ko,
                         (int64_t*)s_d);
      if (!scan(s_a, s_c, n + 2, nullptr, err)) return false;  // no host sync
      hipLaunchKernelGGL(k_merge_scatter_base2, dim3((uint32_t)ceil_div(n, 256)),
                         dim3(256), 0, stream, src.run(), s_c, s_b, dst.keys,
                         dst.meta, dst.rev, dst.vo, dst.ko, n);
      int64_t new_n;
      if (device_newn_ok) {
        new_n = *out_n;
      } else {
        new_n = 0;
        HIP_CHECK(hipMemcpyAsync(&new_n, s_d, 8, hipMemcpyDeviceToHost, stream));
        HIP_CHECK(hipStreamSynchronize(stream));
      }
      hipLaunchKernelGGL(k_same_next_fixup, dim3((uint32_t)ceil_div(2 * m, 256)),
                         dim3(256), 0, stream, dst.keys, dst.meta, dst.ko,
                         spillA, new_n, d_outpos, m);
      *out_n = new_n;
      return true;
    }
    if (n > 0 && m > 0) {
      // big inverted path (folds, bulk appends): m*log2(n) probes + two
      // prefix sums + streaming scatters (the forward n*log2(m) rank path
      // paid ~400M random probes for a 20.7M x 512k fold and is gone).
      if (!ensure_mergebuf(m, err)) return false;
      HIP_CHECK(hipMemsetAsync(s_a, 0, (n + 2) * 8, stream));
      HIP_CHECK(hipMemsetAsync(s_b, 0, n * 8, stream));
      HIP_CHECK(hipMemsetAsync(d_mdup, 0, (m + 1) * 8, stream));
      hipLaunchKernelGGL(k_merge_rank_inv_big, dim3((uint32_t)ceil_div(m, 256)),
                         dim3(256), 0, stream, src.run(), n, dnew, m, spillA,
                         s_a, s_b, d_mdup, d_mlbs);
      if (!scan(d_mdup, d_mdupx, m + 1, nullptr, err)) return false;
      if (!scan(s_a, s_c, n + 2, nullptr, err)) return false;
      hipLaunchKernelGGL(k_merge_scatter_delta2, dim3((uint32_t)ceil_div(m, 256)),
                         dim3(256), 0, stream, dnew, m, d_mdupx, d_mlbs,
                         dst.keys, dst.meta, dst.rev, dst.vo, dst.ko);
      hipLaunchKernelGGL(k_merge_scatter_base2, dim3((uint32_t)ceil_div(n, 256)),
                         dim3(256), 0, stream, src.run(), s_c, s_b, dst.keys,
                         dst.meta, dst.rev, dst.vo, dst.ko, n);
      hipLaunchKernelGGL(k_set_newn, dim3(1), dim3(64), 0, stream, d_mdupx, n,
                         m, (int64_t*)s_d);
      int64_t new_n;
      if (device_newn_ok) {
        new_n = *out_n;
      } else {
        new_n = 0;
        HIP_CHECK(hipMemcpyAsync(&new_n, s_d, 8, hipMemcpyDeviceToHost, stream));
        HIP_CHECK(hipStreamSynchronize(stream));
      }
      hipLaunchKernelGGL(k_same_next_fixup, dim3((uint32_t)ceil_div(2 * m, 256)),
                         dim3(256), 0, stream, dst.keys, dst.meta, dst.ko,
                         spillA, new_n, d_mlbs, m);
      *out_n = new_n;
      return true;
    }
    // n == 0 here (every caller guards m > 0; n > 0 took the inverted path)
    HIP_CHECK(hipMemsetAsync(s_c, 0, 8, stream));
    if (m > 0) {
      int64_t mb = ceil_div(m, 256);
      hipLaunchKernelGGL(k_merge_scatter_delta, dim3((uint32_t)mb), dim3(256),
                         0, stream, src.run(), n, dnew, m, spillA,
                         s_c, dst.keys, dst.meta, dst.rev, dst.vo, dst.ko);
    }
    // new_n on device (s_d[0]) to avoid an extra host round trip; same_next
    // must cover the full possible extent (n+m), which is safe: rows beyond
    // new_n are untouched dst scratch read-only here... they ARE read, so
    // clamp instead: compute new_n first, then same_next sized by it.
    hipLaunchKernelGGL(k_merge_newn, dim3(1), dim3(64), 0, stream, s_c, n, m,
                       (int64_t*)s_d);
    int64_t new_n;
    if (device_newn_ok) {
      // caller tracks new_n exactly; same_next sized by the upper bound is
      // safe only if it matches — use the caller-known count via *out_n
      new_n = *out_n;
    } else {
      new_n = 0;
      HIP_CHECK(hipMemcpyAsync(&new_n, s_d, 8, hipMemcpyDeviceToHost, stream));
      HIP_CHECK(hipStreamSynchronize(stream));
    }
    if (new_n > 0) {
      int64_t sb = ceil_div(new_n, 256);
      hipLaunchKernelGGL(k_same_next, dim3((uint32_t)sb), dim3(256), 0, stream,
                         dst.keys, dst.meta, dst.ko, spillA, new_n);
    }
    *out_n = new_n;
    return true;
  }
};

static int64_t env_i64(const char* name, int64_t dflt) {
  const char* v = getenv(name);
  return v && *v ? atoll(v) : dflt;
}

static int kb_trace() { static int t = env_i64("KB_TRACE", 0) ? 1 : 0; return t; }

Slab* Slab::Create(int64_t max_rows, int64_t heap_cap, int device,
                   std::string* err) {
  int cnt = 0;
  hipError_t e = hipGetDeviceCount(&cnt);
  if (e != hipSuccess || cnt == 0) {
    if (err) *err = "no HIP device (KB_ENOGPU): the product path has no CPU fallback";
    return nullptr;
  }
  if (device >= 0) HIP_CHECK_NULL(hipSetDevice(device));
  Slab* s = new Slab();
  s->p = new Impl();
  Impl* I = s->p;
  I->device = device < 0 ? 0 : device;
  I->max_rows = max_rows;
  I->heap_cap = heap_cap;
  I->max_q = (int)env_i64("KB_MAX_Q", 1024);
  I->scan_t = (int)env_i64("KB_SCAN_T", 1024);
  I->gather_t = (int)env_i64("KB_GATHER_T", 512);
  I->gather_mode = (int)env_i64("KB_GATHER_MODE", 1);
  if (I->gather_t < 64) I->gather_t = 64;
  if (I->gather_t > 1024) I->gather_t = 1024;
  I->gather_t &= ~63;
  {
    int64_t gwv = env_i64("KB_GATHER_GW", 16);
    int l = 0;
    while ((1 << l) < gwv && l < 6) ++l;
    I->gather_gwl = l;
  }
  // the scan prologue resolves 4 bounds with waves 0-3: >= 256 threads
  if (I->scan_t < 256) I->scan_t = 256;
  if (I->scan_t > 1024) I->scan_t = 1024;
  I->scan_t &= ~63;
  I->max_cap = env_i64("KB_MAX_CAP", 4352);
  I->arena_bytes = env_i64("KB_ARENA_BYTES", 384ll << 20);
  std::string lerr;
  if (!err) err = &lerr;
  auto fail = [&](const char*) { delete s; return (Slab*)nullptr; };
  HIP_CHECK_NULL(hipStreamCreate(&I->stream));
  HIP_CHECK_NULL(hipEventCreate(&I->ev0));
  HIP_CHECK_NULL(hipEventCreate(&I->ev1));
  HIP_CHECK_NULL(hipEventCreate(&I->ev2));
  HIP_CHECK_NULL(hipEventCreate(&I->ev3));
  HIP_CHECK_NULL(hipEventCreate(&I->ev_g0));
  HIP_CHECK_NULL(hipEventCreate(&I->ev_g1));
  HIP_CHECK_NULL(hipHostMalloc(&I->h_gmeta, (int64_t)I->max_q * 16));
  HIP_CHECK_NULL(hipHostMalloc(&I->h_gfound, (int64_t)I->max_q * 4));
  for (Impl::Col* c : {&I->A, &I->B}) {
    HIP_CHECK_NULL(hipMalloc(&c->keys, max_rows * KEYW));
    HIP_CHECK_NULL(hipMalloc(&c->meta, max_rows * 8));
    HIP_CHECK_NULL(hipMalloc(&c->rev, max_rows * 8));
    HIP_CHECK_NULL(hipMalloc(&c->vo, max_rows * 8));
    HIP_CHECK_NULL(hipMalloc(&c->ko, max_rows * 8));
  }
  if (env_i64("KB_SCAN_DBG", 0)) {
    HIP_CHECK_NULL(hipMalloc(&I->d_dbg, 8 * 8));
    HIP_CHECK_NULL(hipMemset(I->d_dbg, 0, 8 * 8));
    int khz = 0;
    (void)hipDeviceGetAttribute(&khz, hipDeviceAttributeWallClockRate, I->device);
    I->wall_khz = khz > 0 ? (double)khz : 100000.0;
  }
  I->spill_cap = env_i64("KB_SPILL_BYTES", 256ll << 20);
  HIP_CHECK_NULL(hipMalloc(&I->spillA, I->spill_cap));
  HIP_CHECK_NULL(hipMalloc(&I->spillB, I->spill_cap));
  I->delta_cap = env_i64("KB_DELTA_CAP", 1 << 16);
  if (I->delta_cap > max_rows) I->delta_cap = max_rows;
  for (Impl::Col* c : {&I->DA, &I->DB}) {
    HIP_CHECK_NULL(hipMalloc(&c->keys, I->delta_cap * KEYW));
    HIP_CHECK_NULL(hipMalloc(&c->meta, I->delta_cap * 8));
    HIP_CHECK_NULL(hipMalloc(&c->rev, I->delta_cap * 8));
    HIP_CHECK_NULL(hipMalloc(&c->vo, I->delta_cap * 8));
    HIP_CHECK_NULL(hipMalloc(&c->ko, I->delta_cap * 8));
  }
  HIP_CHECK_NULL(hipMalloc(&I->heapA, heap_cap));
  HIP_CHECK_NULL(hipMalloc(&I->heapB, heap_cap));
  HIP_CHECK_NULL(hipMalloc(&I->s_a, (max_rows + 2) * 8));
  HIP_CHECK_NULL(hipMalloc(&I->s_b, (max_rows + 2) * 8));
  HIP_CHECK_NULL(hipMalloc(&I->s_c, (max_rows + 2) * 8));
  HIP_CHECK_NULL(hipMalloc(&I->s_d, (max_rows + 2) * 8));
  HIP_CHECK_NULL(hipMalloc(&I->s_e, (max_rows + 2) * 8));
  int64_t nb1 = ceil_div(max_rows + 2, 256) + 1;
  int64_t nb2 = ceil_div(nb1, 256) + 1;
  HIP_CHECK_NULL(hipMalloc(&I->lv1, nb1 * 8));
  HIP_CHECK_NULL(hipMalloc(&I->lv1o, nb1 * 8));
  HIP_CHECK_NULL(hipMalloc(&I->lv2, nb2 * 8));
  HIP_CHECK_NULL(hipMalloc(&I->lv2o, nb2 * 8));
  HIP_CHECK_NULL(hipMalloc(&I->lv3, 512 * 8));  // +256 spill slot for level 4
  HIP_CHECK_NULL(hipMalloc(&I->lv3o, 256 * 8));
  HIP_CHECK_NULL(hipMalloc(&I->d_qs, sizeof(DevRangeQ) * I->max_q));
  HIP_CHECK_NULL(hipMalloc(&I->d_gq, sizeof(DevGetQ) * I->max_q));
  HIP_CHECK_NULL(hipMalloc(&I->d_rows, (int64_t)I->max_q * I->max_cap * 8));
  HIP_CHECK_NULL(hipMalloc(&I->d_rows2, (int64_t)I->max_q * I->max_cap * 8));
  HIP_CHECK_NULL(hipMalloc(&I->d_rowsm, (int64_t)I->max_q * I->max_cap * 8));
  HIP_CHECK_NULL(hipMalloc(&I->d_bounds4, (int64_t)I->max_q * 4 * 8));
  HIP_CHECK_NULL(hipMalloc(&I->shadow, max_rows * 8));
  HIP_CHECK_NULL(hipMemset(I->shadow, 0xFF, max_rows * 8));
  HIP_CHECK_NULL(hipMalloc(&I->d_offs, (int64_t)I->max_q * I->max_cap * 8));
  {
    // found/total/gbytes/ovf/scanned/bytes live in ONE allocation so
    // RangeBatchFinish reads them all back in ONE D2H (6 small copies cost
    // ~20 us/step of copy-queue time otherwise)
    uint8_t* base = nullptr;
    I->resmeta_bytes = I->max_q * 8 * 3 + I->max_q * 4 + 16;
    HIP_CHECK_NULL(hipMalloc(&base, I->resmeta_bytes));
    I->d_found = (int64_t*)base;
    I->d_total = (int64_t*)(base + I->max_q * 8);
    I->d_gbytes = (int64_t*)(base + I->max_q * 16);
    I->d_ovf = (int32_t*)(base + I->max_q * 24);
    I->d_scanned = (unsigned long long*)(base + I->max_q * 24 + I->max_q * 4);
    I->d_bytes = I->d_scanned + 1;
    HIP_CHECK_NULL(hipHostMalloc(&I->h_resmeta, I->resmeta_bytes));
  }
  HIP_CHECK_NULL(hipMalloc(&I->d_found32, I->max_q * 4));
  HIP_CHECK_NULL(hipMalloc(&I->d_orev, I->max_q * 8));
  HIP_CHECK_NULL(hipMalloc(&I->d_ometa, I->max_q * 8));
  HIP_CHECK_NULL(hipMalloc(&I->d_gbuf, I->arena_bytes));
  HIP_CHECK_NULL(hipMalloc(&I->d_pack, I->arena_bytes));
  HIP_CHECK_NULL(hipMalloc(&I->d_goffs, (I->max_q + 1) * 8));
  HIP_CHECK_NULL(hipMalloc(&I->d_bounds, 256 * 8));
  HIP_CHECK_NULL(hipMalloc(&I->d_bkeys, 256 * KEYW));
  HIP_CHECK_NULL(hipMalloc(&I->d_brevs, 256 * 8));
  (void)fail;
  return s;
}

Slab::~Slab() { delete p; }
int64_t Slab::rows() const { return p->n; }
int64_t Slab::delta_rows() const { return p->dn; }
int64_t Slab::delta_capacity() const { return p->delta_cap; }
int64_t Slab::heap_used() const { return p->heap_used_; }
int64_t Slab::max_winner_cap() const { return p->max_cap; }

int64_t Slab::spill_used() const { return p->spill_used_; }

bool Slab::SpillAppend(const void* src, int64_t len, int64_t* off,
                       std::string* err) {
  Impl* I = p;
  if (len == 0) { *off = I->spill_used_; return true; }
  if (I->spill_used_ + len > I->spill_cap) {
    if (err) *err = "key-spill heap full (KB_SPILL_BYTES)";
    return false;
  }
  if (len > (8 << 20)) {
    HIP_CHECK(hipMemcpyAsync(I->spillA + I->spill_used_, src, len,
                             hipMemcpyHostToDevice, I->stream));
    HIP_CHECK(hipMemcpyAsync(I->spillB + I->spill_used_, src, len,
                             hipMemcpyHostToDevice, I->stream));
    HIP_CHECK(hipStreamSynchronize(I->stream));
    *off = I->spill_used_;
    I->spill_used_ += len;
    return true;
  }
  uint8_t* h = I->stage_blob(src, len, err);
  if (!h) return false;
  HIP_CHECK(hipMemcpyAsync(I->spillA + I->spill_used_, h, len,
                           hipMemcpyHostToDevice, I->stream));
  HIP_CHECK(hipMemcpyAsync(I->spillB + I->spill_used_, h, len,
                           hipMemcpyHostToDevice, I->stream));
  HIP_CHECK(hipEventRecord(I->ev_blob[I->blob_idx], I->stream));
  *off = I->spill_used_;
  I->spill_used_ += len;
  return true;
}

bool Slab::HeapAppend(const void* src, int64_t len, int64_t* off, std::string* err) {
  Impl* I = p;
  if (len == 0) { *off = I->heap_used_; return true; }
  if (I->heap_used_ + len > I->heap_cap) {
    if (err) *err = "heap full (KB_HEAP_BYTES)";
    return false;
  }
  if (len > (8 << 20)) {  // bulk loads: pageable copy + sync beats GB-scale pinned staging
    HIP_CHECK(hipMemcpyAsync(I->heapA + I->heap_used_, src, len,
                             hipMemcpyHostToDevice, I->stream));
    HIP_CHECK(hipMemcpyAsync(I->heapB + I->heap_used_, src, len,
                             hipMemcpyHostToDevice, I->stream));
    HIP_CHECK(hipStreamSynchronize(I->stream));
    *off = I->heap_used_;
    I->heap_used_ += len;
    return true;
  }
  uint8_t* h = I->stage_blob(src, len, err);
  if (!h) return false;
  HIP_CHECK(hipMemcpyAsync(I->heapA + I->heap_used_, h, len,
                           hipMemcpyHostToDevice, I->stream));
  HIP_CHECK(hipMemcpyAsync(I->heapB + I->heap_used_, h, len,
                           hipMemcpyHostToDevice, I->stream));
  HIP_CHECK(hipEventRecord(I->ev_blob[I->blob_idx], I->stream));
  *off = I->heap_used_;
  I->heap_used_ += len;
  return true;
}

bool Slab::AppendRows(const uint8_t* keys, const uint64_t* meta,
                      const uint64_t* rev, const uint64_t* vo,
                      const uint64_t* ko, int64_t m,
                      std::string* err, int64_t known_new_dn) {
  Impl* I = p;
  if (m == 0) return true;
  if (I->dn + m > I->delta_cap) {
    // fold first to make room (prediction no longer valid: go sync)
    if (!Fold(err)) return false;
    known_new_dn = -1;
    if (m > I->delta_cap) { if (err) *err = "append larger than delta cap"; return false; }
  }
  static const bool validate = getenv("KB_VALIDATE_DN") &&
                               atoi(getenv("KB_VALIDATE_DN")) != 0;
  bool async = known_new_dn >= 0 && !validate;
  if (!async) HIP_CHECK(hipEventRecord(I->ev0, I->stream));
  if (!I->ensure_delta(m, err)) return false;
  {
    // double-buffered pinned mirror: wait only for THIS slot's previous
    // upload (two steps back — long since done), never the whole stream
    int slot = I->up_idx ^= 1;
    HIP_CHECK(hipEventSynchronize(I->ev_up[slot]));
    int64_t cap = I->upload_cap;
    uint8_t* h = I->h_uploads[slot];
    memcpy(h, keys, m * KEYW);
    memcpy(h + cap * KEYW, meta, m * 8);
    memcpy(h + cap * KEYW + cap * 8, rev, m * 8);
    memcpy(h + cap * KEYW + cap * 16, vo, m * 8);
    memcpy(h + cap * KEYW + cap * 24, ko, m * 8);
    // pinned source => truly async enqueues (no staging-pool stall)
    HIP_CHECK(hipMemcpyAsync(I->d_dkeys, h, m * KEYW, hipMemcpyHostToDevice, I->stream));
    HIP_CHECK(hipMemcpyAsync(I->d_dmeta, h + cap * KEYW, m * 8, hipMemcpyHostToDevice, I->stream));
    HIP_CHECK(hipMemcpyAsync(I->d_drev, h + cap * KEYW + cap * 8, m * 8, hipMemcpyHostToDevice, I->stream));
    HIP_CHECK(hipMemcpyAsync(I->d_dvo, h + cap * KEYW + cap * 16, m * 8, hipMemcpyHostToDevice, I->stream));
    HIP_CHECK(hipMemcpyAsync(I->d_dko, h + cap * KEYW + cap * 24, m * 8, hipMemcpyHostToDevice, I->stream));
    HIP_CHECK(hipEventRecord(I->ev_up[slot], I->stream));
  }
  int64_t new_dn = async ? known_new_dn : 0;  // async: mergeRuns reads it
  if (kb_trace()) fprintf(stderr, "[trace] AppendRows n=%lld m=%lld async=%d\n",
                          (long long)I->dn, (long long)m, (int)async);
  Run up{I->d_dkeys, I->d_dmeta, I->d_drev, I->d_dvo, I->d_dko};
  // shadow-mark the base rows of every inserted key BEFORE the merge
  // kernels queue (same stream => ordering is irrelevant, both read only
  // the base run and the upload)
  if (I->n > 0)
    hipLaunchKernelGGL(k_shadow_mark, dim3((uint32_t)ceil_div(m, 256)),
                       dim3(256), 0, I->stream, I->A.run(), I->n, up, m,
                       I->spillA, I->shadow);
  if (!I->mergeRuns(I->DA, I->dn, up, m, I->DB, &new_dn, err,
                    /*device_newn_ok=*/async))
    return false;
  if (async) {
    // kernels queue on the stream; subsequent reads queue behind them.
    // NOTE: the host `keys/meta/rev/vo` buffers were consumed by the H2D
    // copies above, which HIP stages synchronously for pageable memory.
    perf.merges++;
    std::swap(I->DA, I->DB);
    I->dn = known_new_dn;
    return true;
  }
  HIP_CHECK(hipEventRecord(I->ev1, I->stream));
  HIP_CHECK(hipStreamSynchronize(I->stream));
  float ms = 0;
  (void)hipEventElapsedTime(&ms, I->ev0, I->ev1);
  perf.merge_ms += ms;
  perf.merges++;
  if (known_new_dn >= 0 && new_dn != known_new_dn) {
    if (err) *err = "delta row-count prediction mismatch: " +
                    std::to_string(known_new_dn) + " vs " + std::to_string(new_dn);
    return false;
  }
  std::swap(I->DA, I->DB);
  I->dn = new_dn;
  return true;
}

bool Slab::Fold(std::string* err) {
  Impl* I = p;
  if (I->dn == 0) return true;
  if (I->n + I->dn > I->max_rows) { if (err) *err = "slab full (KB_MAX_ROWS)"; return false; }
  HIP_CHECK(hipEventRecord(I->ev2, I->stream));
  int64_t new_n = 0;
  if (!I->mergeRuns(I->A, I->n, I->DA.run(), I->dn, I->B, &new_n, err))
    return false;
  HIP_CHECK(hipEventRecord(I->ev3, I->stream));
  HIP_CHECK(hipStreamSynchronize(I->stream));
  float ms = 0;
  (void)hipEventElapsedTime(&ms, I->ev2, I->ev3);
  perf.merge_ms += ms;
  perf.merges++;
  std::swap(I->A, I->B);
  I->n = new_n;
  I->dn = 0;
  // base rebuilt + delta emptied => no key has a delta row
  HIP_CHECK(hipMemsetAsync(I->shadow, 0xFF, I->n * 8, I->stream));
  HIP_CHECK(hipStreamSynchronize(I->stream));
  return true;
}

bool Slab::Merge(const DeltaRows& d, std::string* err) {
  Impl* I = p;
  if (d.m == 0 && d.heap.empty()) return true;
  if (!Fold(err)) return false;  // keep run ordering invariants
  if (I->n + d.m > I->max_rows) { if (err) *err = "slab full (KB_MAX_ROWS)"; return false; }
  if (I->heap_used_ + (int64_t)d.heap.size() > I->heap_cap) {
    if (err) *err = "heap full (KB_HEAP_BYTES)";
    return false;
  }
  HIP_CHECK(hipEventRecord(I->ev0, I->stream));
  if (!d.heap.empty()) {
    HIP_CHECK(hipMemcpyAsync(I->heapA + I->heap_used_, d.heap.data(),
                             d.heap.size(), hipMemcpyHostToDevice, I->stream));
    HIP_CHECK(hipMemcpyAsync(I->heapB + I->heap_used_, d.heap.data(),
                             d.heap.size(), hipMemcpyHostToDevice, I->stream));
    I->heap_used_ += (int64_t)d.heap.size();
  }
  if (d.m > 0) {
    if (!I->ensure_delta(d.m, err)) return false;
    HIP_CHECK(hipMemcpyAsync(I->d_dkeys, d.keys.data(), d.m * KEYW,
                             hipMemcpyHostToDevice, I->stream));
    HIP_CHECK(hipMemcpyAsync(I->d_dmeta, d.meta.data(), d.m * 8,
                             hipMemcpyHostToDevice, I->stream));
    HIP_CHECK(hipMemcpyAsync(I->d_drev, d.rev.data(), d.m * 8,
                             hipMemcpyHostToDevice, I->stream));
    HIP_CHECK(hipMemcpyAsync(I->d_dvo, d.vo.data(), d.m * 8,
                             hipMemcpyHostToDevice, I->stream));
    if ((int64_t)d.ko.size() == d.m) {
      HIP_CHECK(hipMemcpyAsync(I->d_dko, d.ko.data(), d.m * 8,
                               hipMemcpyHostToDevice, I->stream));
    } else {
      HIP_CHECK(hipMemsetAsync(I->d_dko, 0, d.m * 8, I->stream));
    }
    int64_t new_n = 0;
    Run up{I->d_dkeys, I->d_dmeta, I->d_drev, I->d_dvo, I->d_dko};
    if (!I->mergeRuns(I->A, I->n, up, d.m, I->B, &new_n, err))
      return false;
    std::swap(I->A, I->B);
    I->n = new_n;
    // base rebuilt, delta empty (folded above) => clear shadows
    HIP_CHECK(hipMemsetAsync(I->shadow, 0xFF, I->n * 8, I->stream));
  }
  HIP_CHECK(hipEventRecord(I->ev1, I->stream));
  HIP_CHECK(hipStreamSynchronize(I->stream));
  float ms = 0;
  (void)hipEventElapsedTime(&ms, I->ev0, I->ev1);
  perf.merge_ms += ms;
  perf.merges++;
  return true;
}

bool Slab::RangeBatch(const std::vector<DevRangeQ>& qs, bool d2h,
                      std::vector<RangeResult>* outs, std::string* err,
                      const std::string& qtails) {
  return RangeBatchEx(qs, d2h, true, outs, err, qtails);
}

bool Slab::RangeBatchStart(const std::vector<DevRangeQ>& qs, std::string* err,
                           const std::string& qtails) {
  Impl* I = p;
  int nq = (int)qs.size();
  if (kb_trace()) fprintf(stderr, "[trace] RBStart nq=%d cap0=%lld\n", nq, nq?(long long)qs[0].cap:-1);
  if (nq == 0) return true;
  if (nq > I->max_q) { if (err) *err = "too many queries per batch (KB_MAX_Q)"; return false; }
  int64_t qcap = I->arena_bytes / nq;
  qcap &= ~15ll;
  if (!I->ensure_qtails(qtails, err)) return false;
  {
    // pinned double-buffered query upload (a 900-query batch is ~200 KB;
    // pageable hipMemcpyAsync stalls the host on the staging pool)
    if (!I->h_qs[0]) {
      for (int i = 0; i < 2; ++i) {
        HIP_CHECK(hipHostMalloc(&I->h_qs[i], sizeof(DevRangeQ) * I->max_q));
        HIP_CHECK(hipEventCreate(&I->ev_qs[i]));
      }
    }
    int slot = I->qs_idx ^= 1;
    HIP_CHECK(hipEventSynchronize(I->ev_qs[slot]));
    memcpy(I->h_qs[slot], qs.data(), sizeof(DevRangeQ) * nq);
    HIP_CHECK(hipMemcpyAsync(I->d_qs, I->h_qs[slot], sizeof(DevRangeQ) * nq,
                             hipMemcpyHostToDevice, I->stream));
    HIP_CHECK(hipEventRecord(I->ev_qs[slot], I->stream));
  }
  HIP_CHECK(hipMemsetAsync(I->d_scanned, 0, 8, I->stream));
  HIP_CHECK(hipMemsetAsync(I->d_bytes, 0, 8, I->stream));
  HIP_CHECK(hipEventRecord(I->ev0, I->stream));
  hipLaunchKernelGGL(k_range_bounds, dim3(nq), dim3(256), 0, I->stream,
                     I->A.run(), I->n, I->DA.run(), I->dn, I->spillA,
                     I->d_qtails, I->d_qs, nq, I->d_bounds4);
  hipLaunchKernelGGL(k_range_scan2, dim3(nq), dim3((uint32_t)I->scan_t), 0, I->stream,
                     I->A.run(), I->n, I->DA.run(), I->dn, I->spillA,
                     I->d_qtails, I->d_qs, nq,
                     I->max_cap, I->d_rows, I->d_rows2, I->d_rowsm, I->d_found,
                     I->d_total, I->d_scanned, I->d_bounds4, I->shadow,
                     I->d_dbg);
  HIP_CHECK(hipEventRecord(I->ev1, I->stream));
  hipLaunchKernelGGL(k_gather, dim3(nq), dim3(256), 0, I->stream, I->A.run(),
                     I->DA.run(), I->heapA, I->d_rowsm, I->max_cap,
                     I->d_found, I->d_qs, nq, I->d_gbuf, qcap, I->d_offs,
                     I->d_gbytes, I->d_ovf, I->d_bytes);
  if (I->gather_mode == 2) {
    int64_t maxw = 0;
    for (const DevRangeQ& q : qs) {
      int64_t c = q.cap > 0 && q.cap < I->max_cap ? q.cap : I->max_cap;
      if (c > maxw) maxw = c;
    }
    hipLaunchKernelGGL(k_gather_copy2,
                       dim3(nq, (uint32_t)ceil_div(maxw, 4)), dim3(256), 0,
                       I->stream, I->A.run(), I->DA.run(), I->spillA, I->heapA,
                       I->d_rowsm, I->max_cap, I->d_found, I->d_qs, nq,
                       I->d_gbuf, qcap, I->d_offs, I->d_ovf);
  } else {
    hipLaunchKernelGGL(k_gather_copy, dim3(nq), dim3((uint32_t)I->gather_t), 0, I->stream,
                       I->A.run(), I->DA.run(), I->spillA, I->heapA, I->d_rowsm,
                       I->max_cap, I->d_found, I->d_qs, nq, I->d_gbuf, qcap,
                       I->d_offs, I->d_ovf, I->gather_gwl);
  }
  HIP_CHECK(hipEventRecord(I->ev2, I->stream));
  return true;
}

bool Slab::RangeBatchEx(const std::vector<DevRangeQ>& qs, bool d2h, bool parse,
                        std::vector<RangeResult>* outs, std::string* err,
                        const std::string& qtails) {
  if (!RangeBatchStart(qs, err, qtails)) return false;
  return RangeBatchFinish((int)qs.size(), d2h, parse, outs, err);
}

bool Slab::RangeBatchFinish(int nq, bool d2h, bool parse,
                            std::vector<RangeResult>* outs, std::string* err) {
  Impl* I = p;
  outs->assign(nq, RangeResult());
  if (nq == 0) return true;
  int64_t qcap = I->arena_bytes / nq;
  qcap &= ~15ll;
  // small result metadata: d_found/d_total/d_gbytes are CONTIGUOUS slices of
  // one allocation (see Create), so one D2H covers them; ovf+counters ride
  // two more copies
  // found/total/gbytes/ovf/scanned/bytes share one device block: one D2H
  // into the pinned mirror instead of six small copies on the store stream
  std::vector<int64_t> found(nq), total(nq), gbytes(nq);
  std::vector<int32_t> ovf(nq);
  unsigned long long scanned = 0, bytes = 0;
  {
    if (kb_trace()) {
      fprintf(stderr, "[trace] RBFinish nq=%d readback %lld B h=%p d=%p\n", nq, (long long)I->resmeta_bytes, (void*)I->h_resmeta, (void*)I->d_found);
      hipPointerAttribute_t pa{};
      hipError_t pe = hipPointerGetAttributes(&pa, I->d_found);
      fprintf(stderr, "[trace] dev attr rc=%d type=%d\n", (int)pe, (int)pa.type);
      pe = hipPointerGetAttributes(&pa, I->h_resmeta);
      fprintf(stderr, "[trace] host attr rc=%d type=%d\n", (int)pe, (int)pa.type);
    }
    if (env_i64("KB_SYNC_READBACK", 0)) {
      HIP_CHECK(hipStreamSynchronize(I->stream));
      if (kb_trace()) fprintf(stderr, "[trace] f0 pre-synced\n");
      HIP_CHECK(hipMemcpy(I->h_resmeta, I->d_found, I->resmeta_bytes,
                          hipMemcpyDeviceToHost));
    } else {
      HIP_CHECK(hipMemcpyAsync(I->h_resmeta, I->d_found, I->resmeta_bytes,
                               hipMemcpyDeviceToHost, I->stream));
      if (kb_trace()) fprintf(stderr, "[trace] f1 copy enq\n");
      HIP_CHECK(hipStreamSynchronize(I->stream));
    }
    if (kb_trace()) fprintf(stderr, "[trace] f2 synced\n");
    memcpy(found.data(), I->h_resmeta, nq * 8);
    memcpy(total.data(), I->h_resmeta + I->max_q * 8, nq * 8);
    memcpy(gbytes.data(), I->h_resmeta + I->max_q * 16, nq * 8);
    memcpy(ovf.data(), I->h_resmeta + I->max_q * 24, nq * 4);
    memcpy(&scanned, I->h_resmeta + I->max_q * 28, 8);
    memcpy(&bytes, I->h_resmeta + I->max_q * 28 + 8, 8);
    if (kb_trace()) fprintf(stderr, "[trace] f3 memcpys done w0=%lld ovf0=%d\n", (long long)found[0], (int)ovf[0]);
  }
  float ms = 0;
  (void)hipEventElapsedTime(&ms, I->ev0, I->ev1);
  perf.scan_ms += ms;
  (void)hipEventElapsedTime(&ms, I->ev1, I->ev2);
  perf.gather_ms += ms;
  if (I->d_dbg) {  // KB_SCAN_DBG: block-serial ms per phase, summed over blocks
    unsigned long long c[8] = {0};
    HIP_CHECK(hipMemcpy(c, I->d_dbg, 8 * 8, hipMemcpyDeviceToHost));
    HIP_CHECK(hipMemset(I->d_dbg, 0, 8 * 8));
    perf.dbg_a += (double)c[0] / I->wall_khz;
    perf.dbg_b += (double)c[1] / I->wall_khz;
    perf.dbg_c += (double)c[2] / I->wall_khz;
    perf.dbg_d += (double)c[3] / I->wall_khz;
    perf.dbg_e += (double)c[4] / I->wall_khz;
    fprintf(stderr,
            "[scan_dbg] blocks=%llu total_ms=%.2f max_block_us=%.1f avg_block_us=%.1f\n",
            c[7], (double)c[5] / I->wall_khz,
            c[7] ? (double)c[6] / I->wall_khz * 1e3 : 0.0,
            c[7] ? (double)c[5] / c[7] / I->wall_khz * 1e3 : 0.0);
  }
  perf.scan_launches++;
  perf.gather_launches++;
  perf.rows_scanned += (int64_t)scanned;
  perf.bytes_gathered += (int64_t)bytes;
  for (int q = 0; q < nq; ++q) {
    RangeResult& r = (*outs)[q];
    r.written = found[q];
    r.total = total[q];
    r.bytes = gbytes[q];
    r.overflow = ovf[q] != 0;
    perf.winners += found[q];
  }
  if (!d2h) return true;
  std::vector<int64_t> goffs(nq + 1);
  int64_t acc = 0;
  for (int q = 0; q < nq; ++q) { goffs[q] = acc; acc += ovf[q] ? 0 : gbytes[q]; }
  goffs[nq] = acc;
  if (!parse) {
    // raw bench mode: PIPELINED pack + D2H. The payload copy runs on the
    // copy stream and overlaps the next batch's kernels; slot reuse (every
    // second batch) waits on that slot's previous copy. DrainD2H() collects
    // the tail.
    I->pack_ms_acc = &perf.pack_d2h_ms;
    int idx = I->pk_idx;
    I->pk_idx ^= 1;
    if (!I->ensure_pipe(idx, acc, err)) return false;
    I->wait_slot(idx);
    HIP_CHECK(hipMemcpyAsync(I->d_goffs, goffs.data(), (nq + 1) * 8,
                             hipMemcpyHostToDevice, I->stream));
    hipLaunchKernelGGL(k_pack, dim3(nq), dim3(256), 0, I->stream, I->d_gbuf,
                       qcap, I->d_gbytes, I->d_goffs, I->d_ovf, I->d_packs[idx],
                       nq);
    HIP_CHECK(hipEventRecord(I->ev_pk[idx], I->stream));
    HIP_CHECK(hipStreamWaitEvent(I->cstream, I->ev_pk[idx], 0));
    if (acc > 0)
      HIP_CHECK(hipMemcpyAsync(I->h_packs[idx], I->d_packs[idx], acc,
                               hipMemcpyDeviceToHost, I->cstream));
    HIP_CHECK(hipEventRecord(I->ev_cp[idx], I->cstream));
    I->cp_busy[idx] = true;
    return true;
  }
  // parse mode (List/Stream): synchronous pack + one D2H + parse
  if (kb_trace()) fprintf(stderr, "[trace] parse acc=%lld w0=%lld\n", (long long)acc, (long long)(*outs)[0].written);
  I->wait_slot(0);  // d_pack aliases pipeline slot 0
  HIP_CHECK(hipEventRecord(I->ev0, I->stream));
  HIP_CHECK(hipMemcpyAsync(I->d_goffs, goffs.data(), (nq + 1) * 8,
                           hipMemcpyHostToDevice, I->stream));
  hipLaunchKernelGGL(k_pack, dim3(nq), dim3(256), 0, I->stream, I->d_gbuf, qcap,
                     I->d_gbytes, I->d_goffs, I->d_ovf, I->d_pack, nq);
  if (!I->ensure_hpack(acc, err)) return false;
  if (acc > 0)
    HIP_CHECK(hipMemcpyAsync(I->h_pack, I->d_pack, acc,
                             hipMemcpyDeviceToHost, I->stream));
  HIP_CHECK(hipEventRecord(I->ev1, I->stream));
  HIP_CHECK(hipStreamSynchronize(I->stream));
  (void)hipEventElapsedTime(&ms, I->ev0, I->ev1);
  perf.pack_d2h_ms += ms;
  const uint8_t* hp_end = I->h_pack + acc;
  for (int q = 0; q < nq; ++q) {
    RangeResult& r = (*outs)[q];
    if (r.overflow) continue;
    const uint8_t* pp = I->h_pack + goffs[q];
    if (r.written < 0 || r.written > I->max_cap) {
      fprintf(stderr, "[parse_bug] q=%d written=%lld total=%lld gbytes=%lld\n",
              q, (long long)r.written, (long long)r.total, (long long)r.bytes);
      if (err) *err = "corrupt range result counters";
      return false;
    }
    r.recs.reserve(r.written);
    for (int64_t j = 0; j < r.written; ++j) {
      uint64_t rv;
      uint32_t klen, vlen;
      memcpy(&rv, pp, 8);
      memcpy(&klen, pp + 8, 4);
      memcpy(&vlen, pp + 12, 4);
      if (klen > 8192 || pp + 16 + ((klen + 15) & ~15u) + vlen > hp_end) {
        fprintf(stderr,
                "[parse_bug] q=%d j=%lld/%lld rv=%llu klen=%u vlen=%u "
                "goff=%lld acc=%lld gbytes=%lld\n",
                q, (long long)j, (long long)r.written,
                (unsigned long long)rv, klen, vlen, (long long)goffs[q],
                (long long)acc, (long long)r.bytes);
        if (err) *err = "corrupt packed record";
        return false;
      }
      RangeResult::Rec rec;
      rec.rev = rv;
      rec.key.assign((const char*)pp + 16, klen);
      rec.val.assign((const char*)pp + 16 + ((klen + 15) & ~15u), vlen);
      r.recs.push_back(std::move(rec));
      pp += 16 + ((klen + 15) & ~15u) + ((vlen + 15) & ~15u);
    }
  }
  if (kb_trace()) fprintf(stderr, "[trace] parse done\n");
  return true;
}

bool Slab::DrainD2H(std::string* err) {
  p->pack_ms_acc = &perf.pack_d2h_ms;
  p->wait_slot(0);
  p->wait_slot(1);
  return true;
}

bool Slab::GetBatch(const std::vector<DevGetQ>& qs, std::vector<GetResult>* outs,
                    std::string* err, const std::string& qtails) {
  return GetBatchEx(qs, true, outs, err, qtails);
}

bool Slab::GetBatchStart(const std::vector<DevGetQ>& qs, std::string* err,
                         const std::string& qtails) {
  Impl* I = p;
  int nq = (int)qs.size();
  if (nq == 0) return true;
  if (nq > I->max_q) { if (err) *err = "too many gets per batch"; return false; }
  if (!I->ensure_qtails(qtails, err)) return false;
  HIP_CHECK(hipMemcpyAsync(I->d_gq, qs.data(), sizeof(DevGetQ) * nq,
                           hipMemcpyHostToDevice, I->stream));
  HIP_CHECK(hipEventRecord(I->ev_g0, I->stream));
  int blocks = (int)ceil_div(nq, 4);
  hipLaunchKernelGGL(k_get2, dim3(blocks), dim3(256), 0, I->stream,
                     I->A.run(), I->DA.run(), I->n, I->dn, I->spillA,
                     I->d_qtails, I->heapA, I->d_gq, nq, 0,
                     I->d_gbuf, 0, I->d_orev, I->d_ometa, I->d_found32,
                     I->d_ovf);
  HIP_CHECK(hipMemcpyAsync(I->h_gmeta, I->d_orev, nq * 8, hipMemcpyDeviceToHost, I->stream));
  HIP_CHECK(hipMemcpyAsync(I->h_gmeta + I->max_q, I->d_ometa, nq * 8, hipMemcpyDeviceToHost, I->stream));
  HIP_CHECK(hipMemcpyAsync(I->h_gfound, I->d_found32, nq * 4, hipMemcpyDeviceToHost, I->stream));
  HIP_CHECK(hipEventRecord(I->ev_g1, I->stream));
  return true;
}

bool Slab::GetBatchFinish(int nq, std::vector<GetResult>* outs, std::string* err) {
  Impl* I = p;
  outs->assign(nq, GetResult());
  if (nq == 0) return true;
  HIP_CHECK(hipEventSynchronize(I->ev_g1));
  float ms = 0;
  (void)hipEventElapsedTime(&ms, I->ev_g0, I->ev_g1);
  perf.get_ms += ms;
  perf.get_launches++;
  for (int q = 0; q < nq; ++q) {
    GetResult& g = (*outs)[q];
    g.found = I->h_gfound[q] != 0;
    if (!g.found) continue;
    uint64_t m = I->h_gmeta[I->max_q + q];
    g.rev = I->h_gmeta[q];
    g.tomb = (m & M_TOMB) != 0;
    g.vlen = meta_vlen(m);
  }
  return true;
}

bool Slab::GetBatchEx(const std::vector<DevGetQ>& qs, bool values,
                      std::vector<GetResult>* outs, std::string* err,
                      const std::string& qtails) {
  Impl* I = p;
  auto tt0 = std::chrono::steady_clock::now();
  auto lap = [&](double* acc) {
    auto t = std::chrono::steady_clock::now();
    *acc += std::chrono::duration<double>(t - tt0).count() * 1e3;
    tt0 = t;
  };
  int nq = (int)qs.size();
  outs->assign(nq, GetResult());
  if (nq == 0) return true;
  if (nq > I->max_q) { if (err) *err = "too many gets per batch"; return false; }
  if (!I->ensure_qtails(qtails, err)) return false;
  int64_t slot = I->arena_bytes / nq;
  slot &= ~15ll;
  lap(&perf.dbg_a);
  HIP_CHECK(hipMemcpyAsync(I->d_gq, qs.data(), sizeof(DevGetQ) * nq,
                           hipMemcpyHostToDevice, I->stream));
  lap(&perf.dbg_b);
  HIP_CHECK(hipEventRecord(I->ev0, I->stream));
  int blocks = (int)ceil_div(nq, 4);
  hipLaunchKernelGGL(k_get2, dim3(blocks), dim3(256), 0, I->stream,
                     I->A.run(), I->DA.run(), I->n, I->dn, I->spillA,
                     I->d_qtails, I->heapA, I->d_gq, nq, values ? 1 : 0,
                     I->d_gbuf, slot, I->d_orev, I->d_ometa, I->d_found32,
                     I->d_ovf);
  HIP_CHECK(hipEventRecord(I->ev1, I->stream));
  std::vector<uint64_t> orev(nq), ometa(nq);
  std::vector<int32_t> ofound(nq), oovf(nq);
  HIP_CHECK(hipMemcpyAsync(orev.data(), I->d_orev, nq * 8, hipMemcpyDeviceToHost, I->stream));
  HIP_CHECK(hipMemcpyAsync(ometa.data(), I->d_ometa, nq * 8, hipMemcpyDeviceToHost, I->stream));
  HIP_CHECK(hipMemcpyAsync(ofound.data(), I->d_found32, nq * 4, hipMemcpyDeviceToHost, I->stream));
  HIP_CHECK(hipMemcpyAsync(oovf.data(), I->d_ovf, nq * 4, hipMemcpyDeviceToHost, I->stream));
  lap(&perf.dbg_c);
  HIP_CHECK(hipStreamSynchronize(I->stream));
  lap(&perf.dbg_d);
  float ms = 0;
  (void)hipEventElapsedTime(&ms, I->ev0, I->ev1);
  perf.get_ms += ms;
  perf.get_launches++;
  for (int q = 0; q < nq; ++q) {
    GetResult& g = (*outs)[q];
    g.found = ofound[q] != 0;
    if (!g.found) continue;
    if (oovf[q]) { if (err) *err = "get value exceeds arena slot"; return false; }
    g.rev = orev[q];
    g.tomb = (ometa[q] & M_TOMB) != 0;
    uint32_t vlen = meta_vlen(ometa[q]);
    if (!values) { g.vlen = vlen; continue; }
    g.vlen = vlen;
    g.val.resize(vlen);
    if (vlen)
      HIP_CHECK(hipMemcpy(g.val.data(), I->d_gbuf + (int64_t)q * slot, vlen,
                          hipMemcpyDeviceToHost));
  }
  lap(&perf.dbg_e);
  return true;
}

bool Slab::Compact(const std::vector<std::pair<Bound, Bound>>& borders,
                   uint64_t compact_rev, const std::vector<uint64_t>& timeout_revs,
                   std::string* err) {
  if (timeout_revs.size() != borders.size()) {
    if (err) *err = "Compact: timeout_revs must be per border pair";
    return false;
  }
  Impl* I = p;
  if (!Fold(err)) return false;  // compaction sweeps the single base run
  int64_t n = I->n;
  if (n == 0) return true;
  HIP_CHECK(hipEventRecord(I->ev0, I->stream));
  // keep := 1
  hipLaunchKernelGGL(k_fill_u64, dim3((uint32_t)ceil_div(n, 256)), dim3(256), 0,
                     I->stream, I->s_a, 1ull, n);
  // resolve border row ranges
  int nb = (int)borders.size() * 2;
  if (nb > 256) { if (err) *err = "too many compact borders"; return false; }
  std::vector<uint8_t> bkeys((size_t)nb * KEYW);
  std::vector<uint64_t> brevs(nb);
  for (size_t i = 0; i < borders.size(); ++i) {
    memcpy(bkeys.data() + (2 * i) * KEYW, borders[i].first.key, KEYW);
    brevs[2 * i] = borders[i].first.rev;
    memcpy(bkeys.data() + (2 * i + 1) * KEYW, borders[i].second.key, KEYW);
    brevs[2 * i + 1] = borders[i].second.rev;
  }
  HIP_CHECK(hipMemcpyAsync(I->d_bkeys, bkeys.data(), bkeys.size(),
                           hipMemcpyHostToDevice, I->stream));
  HIP_CHECK(hipMemcpyAsync(I->d_brevs, brevs.data(), nb * 8,
                           hipMemcpyHostToDevice, I->stream));
  hipLaunchKernelGGL(k_find_bounds, dim3(1), dim3(256), 0, I->stream,
                     I->A.run(), n, I->spillA, I->d_bkeys, I->d_brevs, nb,
                     I->d_bounds);
  std::vector<int64_t> hb(nb);
  HIP_CHECK(hipMemcpyAsync(hb.data(), I->d_bounds, nb * 8, hipMemcpyDeviceToHost,
                           I->stream));
  HIP_CHECK(hipStreamSynchronize(I->stream));
  for (int i = 0; i < nb; i += 2) {
    int64_t lo = hb[i], hi = hb[i + 1];
    if (hi <= lo) continue;
    hipLaunchKernelGGL(k_compact_mark, dim3((uint32_t)ceil_div(hi - lo, 256)),
                       dim3(256), 0, I->stream, I->A.meta, I->A.rev, I->A.vo, lo,
                       hi, compact_rev, timeout_revs[i / 2], I->s_a);
  }
  // new row index
  uint64_t kept = 0;
  if (!I->scan(I->s_a, I->s_b, n, &kept, err)) return false;
  // heap offsets
  hipLaunchKernelGGL(k_heap_sizes, dim3((uint32_t)ceil_div(n, 256)), dim3(256),
                     0, I->stream, I->s_a, I->A.rev, I->A.meta, I->s_c, n);
  uint64_t new_heap = 0;
  if (!I->scan(I->s_c, I->s_d, n, &new_heap, err)) return false;
  // key-spill offsets (tails of surviving keys > KEYW)
  hipLaunchKernelGGL(k_spill_sizes, dim3((uint32_t)ceil_div(n, 256)), dim3(256),
                     0, I->stream, I->s_a, I->A.meta, I->s_c, n);
  uint64_t new_spill = 0;
  if (!I->scan(I->s_c, I->s_e, n, &new_spill, err)) return false;
  hipLaunchKernelGGL(k_compact_scatter, dim3((uint32_t)ceil_div(n, 256)),
                     dim3(256), 0, I->stream, I->A.keys, I->A.meta, I->A.rev,
                     I->A.vo, I->A.ko, I->s_a, I->s_b, I->s_d, I->s_e,
                     I->B.keys, I->B.meta, I->B.rev, I->B.vo, I->B.ko, n);
  hipLaunchKernelGGL(k_heap_scatter, dim3((uint32_t)ceil_div(n, 4)), dim3(256),
                     0, I->stream, I->s_a, I->A.rev, I->A.meta, I->A.vo, I->s_d,
                     I->heapA, I->heapB, n);
  if (new_spill > 0)
    hipLaunchKernelGGL(k_spill_scatter, dim3((uint32_t)ceil_div(n, 4)),
                       dim3(256), 0, I->stream, I->s_a, I->A.meta, I->A.ko,
                       I->s_e, I->spillA, I->spillB, n);
  int64_t sb = ceil_div((int64_t)kept, 256);
  if (kept > 0)
    hipLaunchKernelGGL(k_same_next, dim3((uint32_t)sb), dim3(256), 0, I->stream,
                       I->B.keys, I->B.meta, I->B.ko, I->spillB, (int64_t)kept);
  HIP_CHECK(hipEventRecord(I->ev1, I->stream));
  HIP_CHECK(hipStreamSynchronize(I->stream));
  float ms = 0;
  (void)hipEventElapsedTime(&ms, I->ev0, I->ev1);
  perf.compact_ms += ms;
  perf.compacts++;
  std::swap(I->A, I->B);
  std::swap(I->heapA, I->heapB);
  std::swap(I->spillA, I->spillB);
  I->n = (int64_t)kept;
  // compaction folds the delta first and rewrites the base => clear shadows
  if (I->n > 0)
    HIP_CHECK(hipMemsetAsync(I->shadow, 0xFF, I->n * 8, I->stream));
  I->heap_used_ = (int64_t)new_heap;
  I->spill_used_ = (int64_t)new_spill;
  HIP_CHECK(hipStreamSynchronize(I->stream));
  return true;
}

bool Slab::Dump(std::vector<DumpRow>* rows_out, std::string* err) {
  Impl* I = p;
  if (!Fold(err)) return false;
  rows_out->clear();
  int64_t n = I->n;
  if (n == 0) return true;
  std::vector<uint8_t> keys((size_t)n * KEYW);
  std::vector<uint64_t> meta(n), rev(n), vo(n), ko(n);
  std::vector<uint8_t> heap(I->heap_used_);
  std::vector<uint8_t> spill(I->spill_used_);
  HIP_CHECK(hipMemcpyAsync(keys.data(), I->A.keys, keys.size(), hipMemcpyDeviceToHost, I->stream));
  HIP_CHECK(hipMemcpyAsync(meta.data(), I->A.meta, n * 8, hipMemcpyDeviceToHost, I->stream));
  HIP_CHECK(hipMemcpyAsync(rev.data(), I->A.rev, n * 8, hipMemcpyDeviceToHost, I->stream));
  HIP_CHECK(hipMemcpyAsync(vo.data(), I->A.vo, n * 8, hipMemcpyDeviceToHost, I->stream));
  HIP_CHECK(hipMemcpyAsync(ko.data(), I->A.ko, n * 8, hipMemcpyDeviceToHost, I->stream));
  if (I->heap_used_)
    HIP_CHECK(hipMemcpyAsync(heap.data(), I->heapA, I->heap_used_, hipMemcpyDeviceToHost, I->stream));
  if (I->spill_used_)
    HIP_CHECK(hipMemcpyAsync(spill.data(), I->spillA, I->spill_used_, hipMemcpyDeviceToHost, I->stream));
  HIP_CHECK(hipStreamSynchronize(I->stream));
  rows_out->reserve(n);
  for (int64_t i = 0; i < n; ++i) {
    DumpRow r;
    uint32_t klen = meta_klen(meta[i]);
    uint32_t kin = klen > (uint32_t)KEYW ? (uint32_t)KEYW : klen;
    r.key.assign((const char*)keys.data() + i * KEYW, kin);
    if (klen > (uint32_t)KEYW)  // spill tail (keys > 96B)
      r.key.append((const char*)spill.data() + ko[i], klen - KEYW);
    r.rev = rev[i];
    r.meta = meta[i];
    r.vo = vo[i];
    if (rev[i] > 0) r.val.assign((const char*)heap.data() + vo[i], meta_vlen(meta[i]));
    rows_out->push_back(std::move(r));
  }
  return true;
}

bool Slab::WatcherSet(int64_t slot, const uint8_t* prefix, uint32_t plen,
                      uint64_t from_rev, std::string* err) {
  Impl* I = p;
  if (slot >= I->wcap) {
    int64_t cap = std::max<int64_t>(1024, slot + slot / 2 + 1);
    uint8_t* npfx; uint32_t *nplen, *nlive; uint64_t* nfrom;
    HIP_CHECK(hipMalloc(&npfx, cap * KEYW));
    HIP_CHECK(hipMalloc(&nplen, cap * 4));
    HIP_CHECK(hipMalloc(&nlive, cap * 4));
    HIP_CHECK(hipMalloc(&nfrom, cap * 8));
    HIP_CHECK(hipMemsetAsync(nlive, 0, cap * 4, I->stream));
    if (I->wcap > 0) {
      HIP_CHECK(hipMemcpyAsync(npfx, I->d_wpfx, I->wcap * KEYW, hipMemcpyDeviceToDevice, I->stream));
      HIP_CHECK(hipMemcpyAsync(nplen, I->d_wplen, I->wcap * 4, hipMemcpyDeviceToDevice, I->stream));
      HIP_CHECK(hipMemcpyAsync(nlive, I->d_wlive, I->wcap * 4, hipMemcpyDeviceToDevice, I->stream));
      HIP_CHECK(hipMemcpyAsync(nfrom, I->d_wfrom, I->wcap * 8, hipMemcpyDeviceToDevice, I->stream));
    }
    HIP_CHECK(hipStreamSynchronize(I->stream));
    for (void* q : {(void*)I->d_wpfx, (void*)I->d_wplen, (void*)I->d_wlive, (void*)I->d_wfrom})
      if (q) (void)hipFree(q);
    I->d_wpfx = npfx; I->d_wplen = nplen; I->d_wlive = nlive; I->d_wfrom = nfrom;
    I->wcap = cap;
  }
  uint8_t buf[KEYW] = {0};
  memcpy(buf, prefix, plen > KEYW ? KEYW : plen);
  uint32_t one = 1;
  HIP_CHECK(hipMemcpyAsync(I->d_wpfx + slot * KEYW, buf, KEYW, hipMemcpyHostToDevice, I->stream));
  HIP_CHECK(hipMemcpyAsync(I->d_wplen + slot, &plen, 4, hipMemcpyHostToDevice, I->stream));
  HIP_CHECK(hipMemcpyAsync(I->d_wfrom + slot, &from_rev, 8, hipMemcpyHostToDevice, I->stream));
  HIP_CHECK(hipMemcpyAsync(I->d_wlive + slot, &one, 4, hipMemcpyHostToDevice, I->stream));
  HIP_CHECK(hipStreamSynchronize(I->stream));
  return true;
}

void Slab::WatcherClear(int64_t slot) {
  Impl* I = p;
  if (slot >= I->wcap) return;
  uint32_t zero = 0;
  (void)hipMemcpyAsync(I->d_wlive + slot, &zero, 4, hipMemcpyHostToDevice, I->stream);
  (void)hipStreamSynchronize(I->stream);
}

bool Slab::EventRingInit(int64_t cap, std::string* err) {
  Impl* I = p;
  if (I->er_keys) return true;
  I->er_cap = cap;
  HIP_CHECK(hipMalloc(&I->er_keys, cap * KEYW));
  HIP_CHECK(hipMalloc(&I->er_rev, cap * 8));
  return true;
}

bool Slab::EventRingPush(const uint8_t* keys96, const uint64_t* revs,
                         int64_t count, int64_t base_seq, std::string* err) {
  Impl* I = p;
  if (!I->er_keys) { if (err) *err = "event ring not initialized"; return false; }
  // contiguous seqs; the ring may wrap once per push
  int64_t s0 = base_seq % I->er_cap;
  int64_t first = std::min(count, I->er_cap - s0);
  HIP_CHECK(hipMemcpyAsync(I->er_keys + s0 * KEYW, keys96, first * KEYW,
                           hipMemcpyHostToDevice, I->stream));
  HIP_CHECK(hipMemcpyAsync(I->er_rev + s0, revs, first * 8,
                           hipMemcpyHostToDevice, I->stream));
  if (first < count) {
    HIP_CHECK(hipMemcpyAsync(I->er_keys, keys96 + first * KEYW,
                             (count - first) * KEYW, hipMemcpyHostToDevice,
                             I->stream));
    HIP_CHECK(hipMemcpyAsync(I->er_rev, revs + first, (count - first) * 8,
                             hipMemcpyHostToDevice, I->stream));
  }
  HIP_CHECK(hipStreamSynchronize(I->stream));
  return true;
}

bool Slab::WatchFilterRing(int64_t base_seq, int64_t count,
                           std::vector<uint64_t>* bitmap,
                           int64_t* n_watch_slots, std::string* err) {
  Impl* I = p;
  int64_t W = I->wcap;
  *n_watch_slots = W;
  bitmap->clear();
  if (W == 0 || count == 0) return true;
  if (count > 512) { if (err) *err = "filter batch > 512"; return false; }
  int64_t words = ceil_div(count, 64);
  if (W * words > I->bitmap_cap) {
    int64_t cap = W * words * 2;
    if (I->d_bitmap) (void)hipFree(I->d_bitmap);
    HIP_CHECK(hipMalloc(&I->d_bitmap, cap * 8));
    I->bitmap_cap = cap;
  }
  HIP_CHECK(hipEventRecord(I->ev0, I->stream));
  hipLaunchKernelGGL(k_watch_filter2,
                     dim3((uint32_t)ceil_div(W, 64), (uint32_t)words),
                     dim3(64), 0, I->stream, I->er_keys, I->er_rev, I->er_cap,
                     base_seq, count, I->d_wpfx, I->d_wplen, I->d_wfrom,
                     I->d_wlive, W, I->d_bitmap, words);
  HIP_CHECK(hipEventRecord(I->ev1, I->stream));
  bitmap->resize(W * words);
  HIP_CHECK(hipMemcpyAsync(bitmap->data(), I->d_bitmap, W * words * 8,
                           hipMemcpyDeviceToHost, I->stream));
  HIP_CHECK(hipStreamSynchronize(I->stream));
  float ms = 0;
  (void)hipEventElapsedTime(&ms, I->ev0, I->ev1);
  perf.filter_ms += ms;
  perf.filter_launches++;
  perf.filter_events += count;
  perf.filter_watchers += W;
  return true;
}

bool Slab::WatchCatchup(const uint8_t* pfx96, uint32_t plen, uint64_t from_rev,
                        int64_t base_seq, int64_t count,
                        std::vector<uint64_t>* words_out, std::string* err) {
  Impl* I = p;
  words_out->clear();
  if (count <= 0) return true;
  int64_t words = ceil_div(count, 64);
  if (words > I->bitmap_cap) {
    int64_t cap = words * 2;
    if (I->d_bitmap) (void)hipFree(I->d_bitmap);
    HIP_CHECK(hipMalloc(&I->d_bitmap, cap * 8));
    I->bitmap_cap = cap;
  }
  uint8_t pfx[KEYW] = {0};
  memcpy(pfx, pfx96, plen > (uint32_t)KEYW ? KEYW : plen);
  HIP_CHECK(hipMemcpyAsync(I->d_bkeys, pfx, KEYW, hipMemcpyHostToDevice,
                           I->stream));
  HIP_CHECK(hipEventRecord(I->ev0, I->stream));
  hipLaunchKernelGGL(k_watch_catchup, dim3((uint32_t)ceil_div(count, 256)),
                     dim3(256), 0, I->stream, I->er_keys, I->er_rev, I->er_cap,
                     base_seq, count, I->d_bkeys,
                     plen > (uint32_t)KEYW ? (uint32_t)KEYW : plen, from_rev,
                     I->d_bitmap);
  HIP_CHECK(hipEventRecord(I->ev1, I->stream));
  words_out->resize(words);
  HIP_CHECK(hipMemcpyAsync(words_out->data(), I->d_bitmap, words * 8,
                           hipMemcpyDeviceToHost, I->stream));
  HIP_CHECK(hipStreamSynchronize(I->stream));
  float ms = 0;
  (void)hipEventElapsedTime(&ms, I->ev0, I->ev1);
  perf.filter_ms += ms;
  perf.filter_launches++;
  perf.filter_events += count;
  perf.filter_watchers += 1;
  return true;
}

}  // namespace kbslab
