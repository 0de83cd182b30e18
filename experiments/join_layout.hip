/*
 * join_layout.hip — isolates the lds_join slowdown seen when bucket input
 * moved from compact offsets to slack regions (b*capB + sizes[b]).
 * Same kernel copy, same bucket contents, five layouts:
 *   compact (offsets), slack capB=1080 (product), 1024 (pow2 stride),
 *   1160 (odd line count), 2048 (extreme gaps).
 * Diagnostic only.
 */
#include "../distributed_join_amd/csrc/dj_rng.h"

#include <hip/hip_runtime.h>

#include <cstdint>
#include <cstdio>
#include <cstdlib>
#include <vector>

#define CHECK(c)                                                      \
  do {                                                                \
    hipError_t e = (c);                                               \
    if (e != hipSuccess) {                                            \
      printf("HIP error %s at %d\n", hipGetErrorString(e), __LINE__); \
      exit(1);                                                        \
    }                                                                 \
  } while (0)

constexpr int THREADS = 1024;
constexpr int SLOTS = 2048;
constexpr int STAGE = 1024;
constexpr int64_t EMPTY = -1;
constexpr int ROWCAP = 1536;

template <int SL = SLOTS, int ST = STAGE, int TH = THREADS>
__global__ __launch_bounds__(TH) void join_kernel(
  const longlong2* __restrict__ lrows, const int64_t* __restrict__ loff,
  const uint32_t* __restrict__ lsizes, int64_t capBl, const longlong2* __restrict__ rrows,
  const int64_t* __restrict__ roff, const uint32_t* __restrict__ rsizes, int64_t capBr, int B,
  int64_t* __restrict__ out0, int64_t* __restrict__ out1, int64_t* __restrict__ out2,
  int64_t* __restrict__ out3, int64_t cap, unsigned long long* counter)
{
  extern __shared__ __attribute__((aligned(16))) char smem[];
  longlong2* tbl = (longlong2*)smem;
  int64_t* stage = (int64_t*)(tbl + SL);
  long long* base_sh = (long long*)(stage + 4 * ST);
  uint32_t* cur_sh = (uint32_t*)(base_sh + 1);
  const uint32_t smask = SL - 1;
  constexpr int S = ST;

  for (int b = blockIdx.x; b < B; b += gridDim.x) {
    const int64_t l0 = lsizes ? (int64_t)b * capBl : loff[b];
    const int64_t l1 = lsizes ? l0 + lsizes[b] : loff[b + 1];
    const int64_t r0 = rsizes ? (int64_t)b * capBr : roff[b];
    const int64_t r1 = rsizes ? r0 + rsizes[b] : roff[b + 1];
    const int64_t lnb = l1 - l0;
    if (lnb == 0 || r1 == r0 || lnb > SL * 3 / 4) continue;
    for (int s = threadIdx.x; s < SL; s += blockDim.x) tbl[s].x = EMPTY;
    if (threadIdx.x == 0) *cur_sh = 0;
    __syncthreads();
    for (int64_t i = l0 + threadIdx.x; i < l1; i += blockDim.x) {
      longlong2 row = lrows[i];
      uint32_t slot = (uint32_t)dj_mix64((uint64_t)row.x) & smask;
      for (;;) {
        unsigned long long old = atomicCAS((unsigned long long*)&tbl[slot].x,
                                           (unsigned long long)EMPTY,
                                           (unsigned long long)row.x);
        if (old == (unsigned long long)EMPTY) break;
        slot = (slot + 1) & smask;
      }
      tbl[slot].y = row.y;
    }
    __syncthreads();
    for (int64_t j = r0 + threadIdx.x; j < r1; j += blockDim.x) {
      longlong2 prow = rrows[j];
      uint32_t slot = (uint32_t)dj_mix64((uint64_t)prow.x) & smask;
      for (;;) {
        longlong2 e = tbl[slot];
        if (e.x == EMPTY) break;
        if (e.x == prow.x) {
          uint32_t pos = atomicAdd(cur_sh, 1u);
          if (pos < (uint32_t)S) {
            stage[0 * S + pos] = prow.x;
            stage[1 * S + pos] = e.y;
            stage[2 * S + pos] = prow.x;
            stage[3 * S + pos] = prow.y;
          } else {
            long long idx = (long long)atomicAdd(counter, 1ull);
            if (idx < cap) {
              out0[idx] = prow.x;
              out1[idx] = e.y;
              out2[idx] = prow.x;
              out3[idx] = prow.y;
            }
          }
        }
        slot = (slot + 1) & smask;
      }
    }
    __syncthreads();
    const uint32_t total = min(*cur_sh, (uint32_t)S);
    if (threadIdx.x == 0 && total)
      *base_sh = (long long)atomicAdd(counter, (unsigned long long)total);
    __syncthreads();
    if (total) {
      const long long base = *base_sh;
      for (uint32_t i = threadIdx.x; i < total; i += blockDim.x) {
        long long idx = base + (long long)i;
        if (idx < cap) {
          out0[idx] = stage[0 * S + i];
          out1[idx] = stage[1 * S + i];
          out2[idx] = stage[2 * S + i];
          out3[idx] = stage[3 * S + i];
        }
      }
    }
    __syncthreads();
  }
}

/* pipelined variant: build/probe rows of the NEXT phase are prefetched
 * while the current phase computes, and the table clear is folded into the
 * flush phase — 4 barriers per bucket instead of 5 and global-load latency
 * hidden behind LDS work. */
__global__ __launch_bounds__(THREADS) void join_kernel_v2(
  const longlong2* __restrict__ lrows, const int64_t* __restrict__ loff,
  const longlong2* __restrict__ rrows, const int64_t* __restrict__ roff, int B,
  int64_t* __restrict__ out0, int64_t* __restrict__ out1, int64_t* __restrict__ out2,
  int64_t* __restrict__ out3, int64_t cap, unsigned long long* counter)
{
  extern __shared__ __attribute__((aligned(16))) char smem[];
  longlong2* tbl = (longlong2*)smem;
  int64_t* stage = (int64_t*)(tbl + SLOTS);
  long long* base_sh = (long long*)(stage + 4 * STAGE);
  uint32_t* cur_sh = (uint32_t*)(base_sh + 1);
  const uint32_t smask = SLOTS - 1;
  constexpr int S = STAGE;
  const int tid = threadIdx.x;

  for (int s = tid; s < SLOTS; s += blockDim.x) tbl[s].x = EMPTY;
  if (tid == 0) *cur_sh = 0;

  /* prefetch first bucket's build row */
  int b = blockIdx.x;
  int64_t l0 = b < B ? loff[b] : 0, l1 = b < B ? loff[b + 1] : 0;
  longlong2 pb;
  if (b < B && l0 + tid < l1) pb = lrows[l0 + tid];
  __syncthreads();

  for (; b < B; b += gridDim.x) {
    const int64_t r0 = roff[b], r1 = roff[b + 1];
    const int64_t lnb = l1 - l0;
    const int64_t rnb = r1 - r0;
    /* build: first round from the prefetched row, remainder (rare) loaded */
    if (lnb > 0 && rnb > 0 && lnb <= ROWCAP) {
      for (int64_t i = l0 + tid; i < l1; i += blockDim.x) {
        longlong2 row = (i == l0 + tid) ? pb : lrows[i];
        uint32_t slot = (uint32_t)dj_mix64((uint64_t)row.x) & smask;
        for (;;) {
          unsigned long long old = atomicCAS((unsigned long long*)&tbl[slot].x,
                                             (unsigned long long)EMPTY,
                                             (unsigned long long)row.x);
          if (old == (unsigned long long)EMPTY) break;
          slot = (slot + 1) & smask;
        }
        tbl[slot].y = row.y;
      }
    }
    /* prefetch this bucket's probe row while waiting on the build barrier */
    longlong2 pr;
    if (rnb > 0 && r0 + tid < r1) pr = rrows[r0 + tid];
    __syncthreads();
    const bool live = lnb > 0 && rnb > 0 && lnb <= ROWCAP;
    if (live) {
      for (int64_t j = r0 + tid; j < r1; j += blockDim.x) {
        longlong2 prow = (j == r0 + tid) ? pr : rrows[j];
        uint32_t slot = (uint32_t)dj_mix64((uint64_t)prow.x) & smask;
        for (;;) {
          longlong2 e = tbl[slot];
          if (e.x == EMPTY) break;
          if (e.x == prow.x) {
            uint32_t pos = atomicAdd(cur_sh, 1u);
            if (pos < (uint32_t)S) {
              stage[0 * S + pos] = prow.x;
              stage[1 * S + pos] = e.y;
              stage[2 * S + pos] = prow.x;
              stage[3 * S + pos] = prow.y;
            } else {
              long long idx = (long long)atomicAdd(counter, 1ull);
              if (idx < cap) {
                out0[idx] = prow.x;
                out1[idx] = e.y;
                out2[idx] = prow.x;
                out3[idx] = prow.y;
              }
            }
          }
          slot = (slot + 1) & smask;
        }
      }
    }
    /* prefetch next bucket's build row while waiting on the probe barrier */
    const int nb = b + gridDim.x;
    int64_t nl0 = nb < B ? loff[nb] : 0, nl1 = nb < B ? loff[nb + 1] : 0;
    if (nb < B && nl0 + tid < nl1) pb = lrows[nl0 + tid];
    __syncthreads();
    const uint32_t total = min(*cur_sh, (uint32_t)S);
    if (tid == 0 && total)
      *base_sh = (long long)atomicAdd(counter, (unsigned long long)total);
    __syncthreads();
    /* flush + clear the table + reset cur for the next bucket */
    if (total) {
      const long long base = *base_sh;
      for (uint32_t i = tid; i < total; i += blockDim.x) {
        long long idx = base + (long long)i;
        if (idx < cap) {
          out0[idx] = stage[0 * S + i];
          out1[idx] = stage[1 * S + i];
          out2[idx] = stage[2 * S + i];
          out3[idx] = stage[3 * S + i];
        }
      }
    }
    for (int s = tid; s < SLOTS; s += blockDim.x) tbl[s].x = EMPTY;
    if (tid == 0) *cur_sh = 0;
    l0 = nl0;
    l1 = nl1;
    __syncthreads();
  }
}

/* K consecutive buckets share one LDS table: a probe key can only equal a
 * build key of its own bucket (equal keys hash to the same bucket), so the
 * merged table is correct by construction; rows of K buckets are contiguous
 * (coalesced full-wave loads) and the per-bucket barrier overhead drops Kx. */
template <int K, int SLOTS2, int STAGE2>
__global__ __launch_bounds__(THREADS) void join_kernel_multi(
  const longlong2* __restrict__ lrows, const int64_t* __restrict__ loff,
  const longlong2* __restrict__ rrows, const int64_t* __restrict__ roff, int B,
  int64_t* __restrict__ out0, int64_t* __restrict__ out1, int64_t* __restrict__ out2,
  int64_t* __restrict__ out3, int64_t cap, unsigned long long* counter)
{
  extern __shared__ __attribute__((aligned(16))) char smem[];
  longlong2* tbl = (longlong2*)smem;
  int64_t* stage = (int64_t*)(tbl + SLOTS2);
  long long* base_sh = (long long*)(stage + 4 * STAGE2);
  uint32_t* cur_sh = (uint32_t*)(base_sh + 1);
  const uint32_t smask = SLOTS2 - 1;
  const int tid = threadIdx.x;
  const int nsuper = (B + K - 1) / K;

  for (int sb = blockIdx.x; sb < nsuper; sb += gridDim.x) {
    const int b0 = sb * K;
    const int bK = min(b0 + K, B);
    const int64_t l0 = loff[b0], l1 = loff[bK];
    const int64_t r0 = roff[b0], r1 = roff[bK];
    if (l1 - l0 == 0 || r1 == r0 || l1 - l0 > (int64_t)(SLOTS2 * 3 / 4)) continue;
    for (int s2 = tid; s2 < SLOTS2; s2 += blockDim.x) tbl[s2].x = EMPTY;
    if (tid == 0) *cur_sh = 0;
    __syncthreads();
    for (int64_t i = l0 + tid; i < l1; i += blockDim.x) {
      longlong2 row = lrows[i];
      uint32_t slot = (uint32_t)dj_mix64((uint64_t)row.x) & smask;
      for (;;) {
        unsigned long long old = atomicCAS((unsigned long long*)&tbl[slot].x,
                                           (unsigned long long)EMPTY,
                                           (unsigned long long)row.x);
        if (old == (unsigned long long)EMPTY) break;
        slot = (slot + 1) & smask;
      }
      tbl[slot].y = row.y;
    }
    __syncthreads();
    for (int64_t j = r0 + tid; j < r1; j += blockDim.x) {
      longlong2 prow = rrows[j];
      uint32_t slot = (uint32_t)dj_mix64((uint64_t)prow.x) & smask;
      for (;;) {
        longlong2 e = tbl[slot];
        if (e.x == EMPTY) break;
        if (e.x == prow.x) {
          uint32_t pos = atomicAdd(cur_sh, 1u);
          if (pos < (uint32_t)STAGE2) {
            stage[0 * STAGE2 + pos] = prow.x;
            stage[1 * STAGE2 + pos] = e.y;
            stage[2 * STAGE2 + pos] = prow.x;
            stage[3 * STAGE2 + pos] = prow.y;
          } else {
            long long idx = (long long)atomicAdd(counter, 1ull);
            if (idx < cap) {
              out0[idx] = prow.x;
              out1[idx] = e.y;
              out2[idx] = prow.x;
              out3[idx] = prow.y;
            }
          }
        }
        slot = (slot + 1) & smask;
      }
    }
    __syncthreads();
    const uint32_t total = min(*cur_sh, (uint32_t)STAGE2);
    if (tid == 0 && total)
      *base_sh = (long long)atomicAdd(counter, (unsigned long long)total);
    __syncthreads();
    if (total) {
      const long long base = *base_sh;
      for (uint32_t i = tid; i < total; i += blockDim.x) {
        long long idx = base + (long long)i;
        if (idx < cap) {
          out0[idx] = stage[0 * STAGE2 + i];
          out1[idx] = stage[1 * STAGE2 + i];
          out2[idx] = stage[2 * STAGE2 + i];
          out3[idx] = stage[3 * STAGE2 + i];
        }
      }
    }
    __syncthreads();
  }
}

/* cumulative phase ablation of the CURRENT single-pass staged join:
 * 1=init, 2=+build, 3=+probe walk (matches counted, discarded),
 * 4=+stage append, 5=full (reserve + flush). Successive differences give
 * the marginal cost of each phase. */
template <int LEVEL>
__global__ __launch_bounds__(THREADS) void join_ablate(
  const longlong2* __restrict__ lrows, const int64_t* __restrict__ loff,
  const longlong2* __restrict__ rrows, const int64_t* __restrict__ roff, int B,
  int64_t* __restrict__ out0, int64_t* __restrict__ out1, int64_t* __restrict__ out2,
  int64_t* __restrict__ out3, int64_t cap, unsigned long long* counter, int* __restrict__ sink)
{
  extern __shared__ __attribute__((aligned(16))) char smem[];
  longlong2* tbl = (longlong2*)smem;
  int64_t* stage = (int64_t*)(tbl + SLOTS);
  long long* base_sh = (long long*)(stage + 4 * STAGE);
  uint32_t* cur_sh = (uint32_t*)(base_sh + 1);
  const uint32_t smask = SLOTS - 1;
  constexpr int S = STAGE;
  uint32_t acc = 0;
  for (int b = blockIdx.x; b < B; b += gridDim.x) {
    const int64_t l0 = loff[b], l1 = loff[b + 1];
    const int64_t r0 = roff[b], r1 = roff[b + 1];
    const int64_t lnb = l1 - l0;
    if (lnb == 0 || r1 == r0 || lnb > ROWCAP) continue;
    for (int s = threadIdx.x; s < SLOTS; s += blockDim.x) tbl[s].x = EMPTY;
    if (threadIdx.x == 0) *cur_sh = 0;
    __syncthreads();
    if (LEVEL >= 2) {
      for (int64_t i = l0 + threadIdx.x; i < l1; i += blockDim.x) {
        longlong2 row = lrows[i];
        uint32_t slot = (uint32_t)dj_mix64((uint64_t)row.x) & smask;
        for (;;) {
          unsigned long long old = atomicCAS((unsigned long long*)&tbl[slot].x,
                                             (unsigned long long)EMPTY,
                                             (unsigned long long)row.x);
          if (old == (unsigned long long)EMPTY) break;
          slot = (slot + 1) & smask;
        }
        tbl[slot].y = row.y;
      }
    }
    __syncthreads();
    if (LEVEL >= 3) {
      for (int64_t j = r0 + threadIdx.x; j < r1; j += blockDim.x) {
        longlong2 prow = rrows[j];
        uint32_t slot = (uint32_t)dj_mix64((uint64_t)prow.x) & smask;
        for (;;) {
          longlong2 e = tbl[slot];
          if (e.x == EMPTY) break;
          if (e.x == prow.x) {
            if (LEVEL >= 4) {
              uint32_t pos = atomicAdd(cur_sh, 1u);
              if (pos < (uint32_t)S) {
                stage[0 * S + pos] = prow.x;
                stage[1 * S + pos] = e.y;
                stage[2 * S + pos] = prow.x;
                stage[3 * S + pos] = prow.y;
              }
            } else {
              acc += (uint32_t)e.y;
            }
          }
          slot = (slot + 1) & smask;
        }
      }
    }
    __syncthreads();
    if (LEVEL >= 5) {
      const uint32_t total = min(*cur_sh, (uint32_t)S);
      if (threadIdx.x == 0 && total)
        *base_sh = (long long)atomicAdd(counter, (unsigned long long)total);
      __syncthreads();
      if (total) {
        const long long base = *base_sh;
        for (uint32_t i = threadIdx.x; i < total; i += blockDim.x) {
          long long idx = base + (long long)i;
          if (idx < cap) {
            out0[idx] = stage[0 * S + i];
            out1[idx] = stage[1 * S + i];
            out2[idx] = stage[2 * S + i];
            out3[idx] = stage[3 * S + i];
          }
        }
      }
    } else if (LEVEL >= 4 && threadIdx.x == 0) {
      acc += *cur_sh;
    }
    __syncthreads();
  }
  if (acc == 0xFFFFFFFFu) *sink = 1;  // keep acc live
}

/* full join with a FLAT flush: all threads cover total*4 column writes
 * (the per-column loop leaves 3/4 of lanes idle at ~229 staged rows) */
__global__ __launch_bounds__(THREADS) void join_kernel_flat(
  const longlong2* __restrict__ lrows, const int64_t* __restrict__ loff,
  const longlong2* __restrict__ rrows, const int64_t* __restrict__ roff, int B,
  int64_t* __restrict__ out0, int64_t* __restrict__ out1, int64_t* __restrict__ out2,
  int64_t* __restrict__ out3, int64_t cap, unsigned long long* counter)
{
  extern __shared__ __attribute__((aligned(16))) char smem[];
  longlong2* tbl = (longlong2*)smem;
  int64_t* stage = (int64_t*)(tbl + SLOTS);
  long long* base_sh = (long long*)(stage + 4 * STAGE);
  uint32_t* cur_sh = (uint32_t*)(base_sh + 1);
  const uint32_t smask = SLOTS - 1;
  constexpr int S = STAGE;
  for (int b = blockIdx.x; b < B; b += gridDim.x) {
    const int64_t l0 = loff[b], l1 = loff[b + 1];
    const int64_t r0 = roff[b], r1 = roff[b + 1];
    const int64_t lnb = l1 - l0;
    if (lnb == 0 || r1 == r0 || lnb > ROWCAP) continue;
    for (int s = threadIdx.x; s < SLOTS; s += blockDim.x) tbl[s].x = EMPTY;
    if (threadIdx.x == 0) *cur_sh = 0;
    __syncthreads();
    for (int64_t i = l0 + threadIdx.x; i < l1; i += blockDim.x) {
      longlong2 row = lrows[i];
      uint32_t slot = (uint32_t)dj_mix64((uint64_t)row.x) & smask;
      for (;;) {
        unsigned long long old = atomicCAS((unsigned long long*)&tbl[slot].x,
                                           (unsigned long long)EMPTY,
                                           (unsigned long long)row.x);
        if (old == (unsigned long long)EMPTY) break;
        slot = (slot + 1) & smask;
      }
      tbl[slot].y = row.y;
    }
    __syncthreads();
    for (int64_t j = r0 + threadIdx.x; j < r1; j += blockDim.x) {
      longlong2 prow = rrows[j];
      uint32_t slot = (uint32_t)dj_mix64((uint64_t)prow.x) & smask;
      for (;;) {
        longlong2 e = tbl[slot];
        if (e.x == EMPTY) break;
        if (e.x == prow.x) {
          uint32_t pos = atomicAdd(cur_sh, 1u);
          if (pos < (uint32_t)S) {
            stage[0 * S + pos] = prow.x;
            stage[1 * S + pos] = e.y;
            stage[2 * S + pos] = prow.x;
            stage[3 * S + pos] = prow.y;
          } else {
            long long idx = (long long)atomicAdd(counter, 1ull);
            if (idx < cap) {
              out0[idx] = prow.x;
              out1[idx] = e.y;
              out2[idx] = prow.x;
              out3[idx] = prow.y;
            }
          }
        }
        slot = (slot + 1) & smask;
      }
    }
    __syncthreads();
    const uint32_t total = min(*cur_sh, (uint32_t)S);
    if (threadIdx.x == 0 && total)
      *base_sh = (long long)atomicAdd(counter, (unsigned long long)total);
    __syncthreads();
    if (total) {
      const long long base = *base_sh;
      for (uint32_t i = threadIdx.x; i < 4u * total; i += blockDim.x) {
        uint32_t c = i / total, r = i - c * total;
        long long idx = base + (long long)r;
        if (idx < cap) {
          int64_t v = stage[(size_t)c * S + r];
          (c == 0 ? out0 : c == 1 ? out1 : c == 2 ? out2 : out3)[idx] = v;
        }
      }
    }
    __syncthreads();
  }
}

/* fill bucket b with sizes[b] rows: build keys b*4096+i, probe keys
 * b*4096+i for i<30% else non-matching */
__global__ void fill_kernel(longlong2* rows, const int64_t* starts,
                            const uint32_t* sizes, int B, int probe)
{
  int b = blockIdx.x;
  int nrows = sizes[b];
  int match = (int)(nrows * 0.3);
  for (int i = threadIdx.x; i < nrows; i += blockDim.x) {
    int64_t k = (int64_t)b * 4096 + (probe && i >= match ? 2048 + i : i);
    rows[starts[b] + i] = {k, (int64_t)i};
  }
}

static double run_case(const char* name, int B, int nrows, int match, int64_t capB, int reps,
                       bool arena_carve, bool dirty, bool poisson)
{
  /* layout: capB==0 => compact; capB==-1 => compact + pipelined kernel */
  int64_t stride = capB > 0 ? capB : nrows;
  int64_t total = (int64_t)B * stride;
  longlong2 *lrows, *rrows;
  int64_t *loff, *starts;
  uint32_t* sizes;
  int64_t *o0, *o1, *o2, *o3;
  unsigned long long* counter;
  int64_t cap = (int64_t)B * match + 1024;
  char* arena = nullptr;
  if (arena_carve) {
    size_t need = 2 * (size_t)total * 16 + (size_t)(B + 1) * 8 + (size_t)B * 12 +
                  4 * (size_t)cap * 8 + 4096;
    CHECK(hipMalloc(&arena, need));
    char* pp = arena;
    auto take = [&](size_t b) { void* r = pp; pp += (b + 255) & ~(size_t)255; return r; };
    lrows = (longlong2*)take((size_t)total * 16);
    rrows = (longlong2*)take((size_t)total * 16);
    loff = (int64_t*)take((size_t)(B + 1) * 8);
    starts = (int64_t*)take((size_t)B * 8);
    sizes = (uint32_t*)take((size_t)B * 4);
    o0 = (int64_t*)take((size_t)cap * 8);
    o1 = (int64_t*)take((size_t)cap * 8);
    o2 = (int64_t*)take((size_t)cap * 8);
    o3 = (int64_t*)take((size_t)cap * 8);
    counter = (unsigned long long*)take(8);
  } else {
    CHECK(hipMalloc(&lrows, total * 16));
    CHECK(hipMalloc(&rrows, total * 16));
    CHECK(hipMalloc(&loff, (B + 1) * 8));
    CHECK(hipMalloc(&starts, B * 8));
    CHECK(hipMalloc(&sizes, B * 4));
    CHECK(hipMalloc(&o0, cap * 8));
    CHECK(hipMalloc(&o1, cap * 8));
    CHECK(hipMalloc(&o2, cap * 8));
    CHECK(hipMalloc(&o3, cap * 8));
    CHECK(hipMalloc(&counter, 8));
  }
  std::vector<int64_t> h_off(B + 1), h_starts(B);
  std::vector<uint32_t> h_sizes(B);
  for (int b = 0; b < B; b++)
    h_sizes[b] = poisson ? (uint32_t)(nrows - 84 + (int)(dj_mix64((uint64_t)b) % 169))
                         : (uint32_t)nrows;  // ~uniform +-84 (~3 sigma-ish spread)
  for (int b = 0; b <= B; b++) h_off[b] = (int64_t)b * stride;
  for (int b = 0; b < B; b++) h_starts[b] = (int64_t)b * stride;
  CHECK(hipMemcpy(loff, h_off.data(), (B + 1) * 8, hipMemcpyHostToDevice));
  CHECK(hipMemcpy(starts, h_starts.data(), B * 8, hipMemcpyHostToDevice));
  CHECK(hipMemcpy(sizes, h_sizes.data(), B * 4, hipMemcpyHostToDevice));
  fill_kernel<<<B, 256>>>(lrows, starts, sizes, B, 0);
  fill_kernel<<<B, 256>>>(rrows, starts, sizes, B, 1);
  CHECK(hipDeviceSynchronize());

  size_t lds = SLOTS * 16 + 4 * STAGE * 8 + 16;
  int grid = B < 8192 ? B : 8192;
  hipEvent_t e0, e1;
  CHECK(hipEventCreate(&e0));
  CHECK(hipEventCreate(&e1));
  double ms_sum = 0;
  unsigned long long nout = 0;
  for (int rep = 0; rep < reps; rep++) {
    CHECK(hipMemset(counter, 0, 8));
    if (dirty) {  // re-write inputs so the join starts with a dirty L2
      fill_kernel<<<B, 256>>>(lrows, starts, sizes, B, 0);
      fill_kernel<<<B, 256>>>(rrows, starts, sizes, B, 1);
    }
    CHECK(hipEventRecord(e0));
    if (capB == -5) {  /* bigger buckets: 4096-slot table, 1 block/CU */
      size_t lds5 = 4096 * 16 + 4 * 1024 * 8 + 16;
      join_kernel<4096, 1024, 1024><<<grid, 1024, lds5>>>(lrows, loff, nullptr, 0, rrows, loff,
                                                          nullptr, 0, B, o0, o1, o2, o3, cap,
                                                          counter);
    } else if (capB == -6) {  /* smaller buckets: 1024-slot, 512 thr, 4 blocks/CU */
      size_t lds6 = 1024 * 16 + 4 * 512 * 8 + 16;
      join_kernel<1024, 512, 512><<<grid, 512, lds6>>>(lrows, loff, nullptr, 0, rrows, loff,
                                                       nullptr, 0, B, o0, o1, o2, o3, cap,
                                                       counter);
    } else if (capB == -7) {
      join_kernel_flat<<<grid, THREADS, lds>>>(lrows, loff, rrows, loff, B, o0, o1, o2, o3,
                                               cap, counter);
    } else if (capB == -1)
      join_kernel_v2<<<grid, THREADS, lds>>>(lrows, loff, rrows, loff, B, o0, o1, o2, o3, cap,
                                             counter);
    else if (capB == -2) {
      size_t lds2 = 4096 * 16 + 4 * 384 * 8 + 16;
      join_kernel_multi<2, 4096, 384><<<grid, THREADS, lds2>>>(lrows, loff, rrows, loff, B, o0,
                                                               o1, o2, o3, cap, counter);
    } else if (capB == -4) {
      size_t lds4 = 8192 * 16 + 4 * 768 * 8 + 16;
      join_kernel_multi<4, 8192, 768><<<grid, THREADS, lds4>>>(lrows, loff, rrows, loff, B, o0,
                                                               o1, o2, o3, cap, counter);
    }
    else if (capB)
      join_kernel<><<<grid, THREADS, lds>>>(lrows, nullptr, sizes, capB, rrows, nullptr, sizes,
                                            capB, B, o0, o1, o2, o3, cap, counter);
    else
      join_kernel<><<<grid, THREADS, lds>>>(lrows, loff, nullptr, 0, rrows, loff, nullptr, 0,
                                            B, o0, o1, o2, o3, cap, counter);
    CHECK(hipEventRecord(e1));
    CHECK(hipEventSynchronize(e1));
    float ms;
    CHECK(hipEventElapsedTime(&ms, e0, e1));
    if (rep) ms_sum += ms;
    CHECK(hipMemcpy(&nout, counter, 8, hipMemcpyDeviceToHost));
  }
  CHECK(hipGetLastError());
  double avg = ms_sum / (reps - 1);
  printf("%-34s capB=%5lld  %8.3f ms  (out=%llu)\n", name, (long long)capB, avg,
         (unsigned long long)nout);
  if (arena_carve) {
    CHECK(hipFree(arena));
  } else {
    CHECK(hipFree(lrows)); CHECK(hipFree(rrows)); CHECK(hipFree(loff));
    CHECK(hipFree(starts)); CHECK(hipFree(sizes)); CHECK(hipFree(o0)); CHECK(hipFree(o1));
    CHECK(hipFree(o2)); CHECK(hipFree(o3)); CHECK(hipFree(counter));
  }
  CHECK(hipEventDestroy(e0)); CHECK(hipEventDestroy(e1));
  return avg;
}

int main(int argc, char** argv)
{
  int B = argc > 1 ? atoi(argv[1]) : 131072;
  int nrows = argc > 2 ? atoi(argv[2]) : 762;
  int match = (int)(nrows * 0.3);
  int reps = 4;
  if (argc > 3 && atoi(argv[3]) == 1) {
    /* cumulative phase ablation on bench-shaped buckets */
    int64_t stride = nrows, total = (int64_t)B * stride;
    longlong2 *lrows, *rrows; int64_t *loff, *starts; uint32_t* szs;
    int64_t *o0, *o1, *o2, *o3; unsigned long long* counter; int* sink;
    int64_t cap = (int64_t)B * match + 1024;
    CHECK(hipMalloc(&lrows, total * 16)); CHECK(hipMalloc(&rrows, total * 16));
    CHECK(hipMalloc(&loff, (B + 1) * 8)); CHECK(hipMalloc(&starts, B * 8));
    CHECK(hipMalloc(&szs, B * 4));
    CHECK(hipMalloc(&o0, cap * 8)); CHECK(hipMalloc(&o1, cap * 8));
    CHECK(hipMalloc(&o2, cap * 8)); CHECK(hipMalloc(&o3, cap * 8));
    CHECK(hipMalloc(&counter, 8)); CHECK(hipMalloc(&sink, 4));
    std::vector<int64_t> h_off(B + 1), h_st(B); std::vector<uint32_t> h_sz(B, (uint32_t)nrows);
    for (int b = 0; b <= B; b++) h_off[b] = (int64_t)b * stride;
    for (int b = 0; b < B; b++) h_st[b] = (int64_t)b * stride;
    CHECK(hipMemcpy(loff, h_off.data(), (B + 1) * 8, hipMemcpyHostToDevice));
    CHECK(hipMemcpy(starts, h_st.data(), B * 8, hipMemcpyHostToDevice));
    CHECK(hipMemcpy(szs, h_sz.data(), B * 4, hipMemcpyHostToDevice));
    fill_kernel<<<B, 256>>>(lrows, starts, szs, B, 0);
    fill_kernel<<<B, 256>>>(rrows, starts, szs, B, 1);
    CHECK(hipDeviceSynchronize());
    size_t lds = SLOTS * 16 + 4 * STAGE * 8 + 16;
    int grid = B < 8192 ? B : 8192;
    hipEvent_t e0, e1; CHECK(hipEventCreate(&e0)); CHECK(hipEventCreate(&e1));
    auto run = [&](auto kern, const char* name) {
      double ms_sum = 0;
      for (int rep = 0; rep < reps; rep++) {
        CHECK(hipMemset(counter, 0, 8));
        CHECK(hipEventRecord(e0));
        kern<<<grid, THREADS, lds>>>(lrows, loff, rrows, loff, B, o0, o1, o2, o3, cap,
                                     counter, sink);
        CHECK(hipEventRecord(e1)); CHECK(hipEventSynchronize(e1));
        float ms; CHECK(hipEventElapsedTime(&ms, e0, e1));
        if (rep) ms_sum += ms;
      }
      CHECK(hipGetLastError());
      printf("%-28s %7.3f ms\n", name, ms_sum / (reps - 1));
    };
    run(join_ablate<1>, "init only");
    run(join_ablate<2>, "+build");
    run(join_ablate<3>, "+probe walk (no stage)");
    run(join_ablate<4>, "+stage append (no flush)");
    run(join_ablate<5>, "full (reserve+flush)");
    return 0;
  }
  run_case("base B=131072 r=762", B, nrows, match, 0, reps, false, true, false);
  run_case("flat flush", B, nrows, match, -7, reps, false, true, false);
  run_case("big B/2 r=1526 1blk/CU", B / 2, nrows * 2, (int)(nrows * 2 * 0.3), -5, reps,
           false, true, false);
  run_case("small 2B r=381 512thr", B * 2, nrows / 2, (int)(nrows / 2 * 0.3), -6, reps,
           false, true, false);
  return 0;
}
