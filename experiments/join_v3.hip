/*
 * join_v3.hip — second round-2 sweep: software-pipelined staged scatters
 * (next tile's global loads issue BEFORE the current tile's flush, hiding
 * HBM load latency under the store burst — pass A runs 1 block/CU, so
 * barriered phases can't overlap across blocks), full-occupancy wave join
 * (4 x 512-thread blocks/CU = 32 waves), watermark flushes, wave-aggregated
 * emits.
 *
 * Timed + checksum-verified against the product kernels, as join_v2.hip.
 * Build: hipcc --offload-arch=gfx950 -O3 join_v3.hip \
 *          ../distributed_join_amd/csrc/dj_kernels.hip -o join_v3
 * Diagnostic only.
 */
#include "../distributed_join_amd/csrc/dj_kernels.hpp"
#include "../distributed_join_amd/csrc/dj_rng.h"

#include <hip/hip_runtime.h>

#include <algorithm>
#include <cmath>
#include <cstdint>
#include <cstdio>
#include <cstdlib>
#include <functional>
#include <type_traits>
#include <vector>

#define CHECK(c)                                                      \
  do {                                                                \
    hipError_t e = (c);                                               \
    if (e != hipSuccess) {                                            \
      printf("HIP error %s at %d\n", hipGetErrorString(e), __LINE__); \
      exit(1);                                                        \
    }                                                                 \
  } while (0)

constexpr int64_t EMPTY = -1;

__global__ void checksum_kernel(const int64_t* o0, const int64_t* o1, const int64_t* o2,
                                const int64_t* o3, int64_t n, unsigned long long* acc)
{
  unsigned long long local = 0;
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride)
    local += dj_mix64((uint64_t)o0[i]) * 3 + dj_mix64((uint64_t)o1[i]) * 5 +
             dj_mix64((uint64_t)o2[i]) * 7 + dj_mix64((uint64_t)o3[i]);
  for (int off = 32; off; off >>= 1) local += __shfl_down(local, off);
  if ((threadIdx.x & 63) == 0) atomicAdd(acc, local);
}

/* wave shfl exclusive scan of hist[0..P) -> base[0..P); partials[16] LDS */
__device__ __forceinline__ void wave_excl_scan(const uint32_t* hist, uint32_t* base,
                                               uint32_t* partials, int P)
{
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  uint32_t v = (tid < P) ? hist[tid] : 0;
  uint32_t incl = v;
#pragma unroll
  for (int off = 1; off < 64; off <<= 1) {
    uint32_t up = __shfl_up(incl, off);
    if (lane >= off) incl += up;
  }
  if (lane == 63 && wid < (P + 63) / 64) partials[wid] = incl;
  __syncthreads();
  if (wid == 0) {
    const int nw = (P + 63) / 64;
    uint32_t pv = (lane < nw) ? partials[lane] : 0;
    uint32_t pincl = pv;
#pragma unroll
    for (int off = 1; off < 64; off <<= 1) {
      uint32_t up = __shfl_up(pincl, off);
      if (lane >= off) pincl += up;
    }
    if (lane < nw) partials[lane] = pincl - pv;
  }
  __syncthreads();
  if (tid < P) base[tid] = incl - v + partials[wid];
}

__device__ __forceinline__ uint32_t groupA2(int64_t key, int P)
{
  return (uint32_t)(dj_mix64((uint64_t)key) >> 40) & (uint32_t)(P - 1);
}
__device__ __forceinline__ uint32_t subF2(int64_t key, int F)
{
  uint64_t m = dj_mix64((uint64_t)key);
  if (F <= 256) return (uint32_t)(m >> 32) & (uint32_t)(F - 1);
  uint32_t lo = (uint32_t)(m >> 32) & 255u;
  uint32_t hi = (uint32_t)(m >> 50) & (uint32_t)((F >> 8) - 1);
  return lo | (hi << 8);
}

/* ------------------- pipelined pass-A slack scatter (A-pipe) ------------- */
/* Structure per tile: scan -> claim -> stage -> [issue NEXT tile loads] ->
 * flush -> rank(next). The next tile's 8 nontemporal loads per lane are in
 * flight while the flush's global stores drain. */
constexpr int ATILE = 8192;
constexpr int ATHREADS = 1024;
constexpr int AVPT = ATILE / ATHREADS; /* 8 */

__global__ __launch_bounds__(ATHREADS) void slackA_pipe_kernel(
  const int64_t* __restrict__ keys, const int64_t* __restrict__ pay, int64_t n, int P,
  int64_t capA, uint32_t* __restrict__ gcursor, int* __restrict__ any_overflow,
  longlong2* __restrict__ out_pairs)
{
  extern __shared__ __attribute__((aligned(16))) char smem[];
  longlong2* tbuf = (longlong2*)smem;
  uint32_t* hist = (uint32_t*)(tbuf + ATILE);
  uint32_t* base = hist + P;
  uint32_t* gcur = base + P;
  uint32_t* glim = gcur + P;
  uint32_t* partials = glim + P; /* 16 */
  __shared__ int s_ovf;
  const int tid = threadIdx.x;
  if (tid == 0) s_ovf = 0;
  if (tid < P) hist[tid] = 0;
  const int64_t chunk = (n + gridDim.x - 1) / gridDim.x;
  const int64_t start = (int64_t)blockIdx.x * chunk;
  const int64_t end = min(start + chunk, n);
  if (start >= end) return;
  __syncthreads();

  longlong2 r[AVPT];
  uint32_t g[AVPT], rank[AVPT];
  /* preload + rank tile 0 */
#pragma unroll
  for (int v = 0; v < AVPT; v++) {
    int64_t i = start + (int64_t)v * ATHREADS + tid;
    if (i < end) {
      r[v].x = __builtin_nontemporal_load(&keys[i]);
      r[v].y = pay ? __builtin_nontemporal_load(&pay[i]) : i;
      g[v] = groupA2(r[v].x, P);
      rank[v] = atomicAdd(&hist[g[v]], 1u);
    }
  }
  __syncthreads();

  for (int64_t t0 = start; t0 < end; t0 += ATILE) {
    const int count = (int)min((int64_t)ATILE, end - t0);
    const int64_t t1 = t0 + ATILE;
    wave_excl_scan(hist, base, partials, P);
    if (tid < P) {
      const int64_t seg0 = (int64_t)tid * capA;
      gcur[tid] = (uint32_t)seg0 + atomicAdd(&gcursor[tid], hist[tid]);
      glim[tid] = (uint32_t)(seg0 + capA);
    }
    __syncthreads();
#pragma unroll
    for (int v = 0; v < AVPT; v++) {
      int64_t i = t0 + (int64_t)v * ATHREADS + tid;
      if (i < end) tbuf[base[g[v]] + rank[v]] = r[v];
    }
    __syncthreads();
    if (tid < P) hist[tid] = 0; /* free after the claim; rank(next) comes after flush sync */
    /* issue next tile's loads BEFORE the flush */
    longlong2 r2[AVPT];
#pragma unroll
    for (int v = 0; v < AVPT; v++) {
      int64_t i = t1 + (int64_t)v * ATHREADS + tid;
      if (i < end) {
        r2[v].x = __builtin_nontemporal_load(&keys[i]);
        r2[v].y = pay ? __builtin_nontemporal_load(&pay[i]) : i;
      }
    }
    /* flush tile t */
    for (int pos = tid; pos < count; pos += blockDim.x) {
      longlong2 row = tbuf[pos];
      uint32_t gg = groupA2(row.x, P);
      uint32_t dst = gcur[gg] + (uint32_t)(pos - base[gg]);
      if (dst < glim[gg]) {
        __builtin_nontemporal_store(row.x, &out_pairs[dst].x);
        __builtin_nontemporal_store(row.y, &out_pairs[dst].y);
      } else {
        s_ovf = 1;
      }
    }
    __syncthreads();
    /* rank tile t+1 (its loads landed during the flush) */
#pragma unroll
    for (int v = 0; v < AVPT; v++) {
      int64_t i = t1 + (int64_t)v * ATHREADS + tid;
      if (i < end) {
        r[v] = r2[v];
        g[v] = groupA2(r[v].x, P);
        rank[v] = atomicAdd(&hist[g[v]], 1u);
      }
    }
    __syncthreads();
  }
  if (tid == 0 && s_ovf) atomicOr(any_overflow, 2);
}

/* -------- pipelined pass-B over slack-A segments (B-pipe) ---------------- */
/* block = pass-A group a; COUNTED: keep the seghist sweep and compact
 * contiguous output (the product convention); !COUNTED: slack analytic
 * starts b*capB + LDS cursors + lens out, no count sweep. Both pipeline the
 * staged span exactly as A-pipe. */
constexpr int BTILE = 4096;
constexpr int BVPT = BTILE / ATHREADS; /* 4 */

template <bool COUNTED>
__global__ __launch_bounds__(ATHREADS) void subpartB_pipe_kernel(
  const longlong2* __restrict__ in_pairs, const uint32_t* __restrict__ seg_len, int64_t capA,
  int B, int F, int64_t capB, const int64_t* __restrict__ segout /* PA+1, COUNTED only */,
  longlong2* __restrict__ out_pairs, int64_t* __restrict__ bucket_offsets /* B+1 */,
  uint32_t* __restrict__ lens, int* __restrict__ any_overflow)
{
  extern __shared__ __attribute__((aligned(16))) char smem[];
  longlong2* tbuf = (longlong2*)smem;
  uint32_t* hist = (uint32_t*)(tbuf + BTILE);
  uint32_t* base = hist + F;
  uint32_t* gcur = base + F;
  uint32_t* seghist = gcur + F;           /* COUNTED only */
  uint32_t* partials = seghist + F;       /* 16 */
  __shared__ int s_ovf;
  const int tid = threadIdx.x;
  const int a = blockIdx.x;
  const int64_t s0 = (int64_t)a * capA;
  const int64_t s1 = s0 + seg_len[a];
  if (tid == 0) s_ovf = 0;
  if (COUNTED) {
    if (tid < F) seghist[tid] = 0;
    __syncthreads();
    for (int64_t i = s0 + tid; i < s1; i += blockDim.x)
      atomicAdd(&seghist[subF2(__builtin_nontemporal_load(&in_pairs[i].x), F)], 1u);
    __syncthreads();
    if (tid == 0) {
      uint32_t acc = 0;
      const int64_t ob = segout[a];
      for (int j = 0; j < F; j++) {
        uint32_t c = seghist[j];
        gcur[j] = (uint32_t)(ob + acc);
        bucket_offsets[(size_t)a * F + j] = ob + acc;
        acc += c;
      }
      if (a == gridDim.x - 1) bucket_offsets[B] = segout[gridDim.x];
    }
  } else {
    for (int j = tid; j < F; j += blockDim.x)
      gcur[j] = (uint32_t)(((int64_t)a * F + j) * capB);
  }
  if (tid < F) hist[tid] = 0;
  __syncthreads();
  if (s0 >= s1) {
    if (!COUNTED)
      for (int j = tid; j < F; j += blockDim.x) lens[(size_t)a * F + j] = 0;
    return;
  }

  longlong2 r[BVPT];
  uint32_t g[BVPT], rank[BVPT];
#pragma unroll
  for (int v = 0; v < BVPT; v++) {
    int64_t i = s0 + (int64_t)v * ATHREADS + tid;
    if (i < s1) {
      r[v].x = __builtin_nontemporal_load(&in_pairs[i].x);
      r[v].y = __builtin_nontemporal_load(&in_pairs[i].y);
      g[v] = subF2(r[v].x, F);
      rank[v] = atomicAdd(&hist[g[v]], 1u);
    }
  }
  __syncthreads();

  for (int64_t t0 = s0; t0 < s1; t0 += BTILE) {
    const int count = (int)min((int64_t)BTILE, s1 - t0);
    const int64_t t1 = t0 + BTILE;
    wave_excl_scan(hist, base, partials, F);
    __syncthreads();
#pragma unroll
    for (int v = 0; v < BVPT; v++) {
      int64_t i = t0 + (int64_t)v * ATHREADS + tid;
      if (i < s1) tbuf[base[g[v]] + rank[v]] = r[v];
    }
    __syncthreads();
    longlong2 r2[BVPT];
#pragma unroll
    for (int v = 0; v < BVPT; v++) {
      int64_t i = t1 + (int64_t)v * ATHREADS + tid;
      if (i < s1) {
        r2[v].x = __builtin_nontemporal_load(&in_pairs[i].x);
        r2[v].y = __builtin_nontemporal_load(&in_pairs[i].y);
      }
    }
    for (int pos = tid; pos < count; pos += blockDim.x) {
      longlong2 row = tbuf[pos];
      uint32_t gg = subF2(row.x, F);
      uint32_t dst = gcur[gg] + (uint32_t)(pos - base[gg]);
      if (COUNTED) {
        out_pairs[dst] = row;
      } else {
        uint32_t limit = (uint32_t)(((int64_t)a * F + gg) * capB + capB);
        if (dst < limit)
          out_pairs[dst] = row;
        else
          s_ovf = 1;
      }
    }
    __syncthreads();
    if (tid < F) gcur[tid] += hist[tid];
    if (tid < F) hist[tid] = 0;
    __syncthreads();
#pragma unroll
    for (int v = 0; v < BVPT; v++) {
      int64_t i = t1 + (int64_t)v * ATHREADS + tid;
      if (i < s1) {
        r[v] = r2[v];
        g[v] = subF2(r[v].x, F);
        rank[v] = atomicAdd(&hist[g[v]], 1u);
      }
    }
    __syncthreads();
  }
  if (!COUNTED) {
    __syncthreads();
    for (int j = tid; j < F; j += blockDim.x) {
      int64_t b = (int64_t)a * F + j;
      uint32_t len = gcur[j] - (uint32_t)(b * capB);
      lens[b] = len > (uint32_t)capB ? (uint32_t)capB : len;
    }
    if (tid == 0 && s_ovf) atomicOr(any_overflow, 2);
  }
}

/* ------------------- wave join, occupancy-focused shapes ----------------- */
/* As join_v2's wave_join_kernel, plus: watermark flush (sync every round,
 * flush only when cur >= WATER or done), wave-aggregated emit kept. */
template <int WAVES, int SLOTS, int STAGE, bool SLACK>
__global__ __launch_bounds__(WAVES * 64)
  void wave_join2_kernel(const longlong2* __restrict__ lrows, const int64_t* __restrict__ loff,
                         const uint32_t* __restrict__ llen, int64_t capL,
                         const longlong2* __restrict__ rrows, const int64_t* __restrict__ roff,
                         const uint32_t* __restrict__ rlen, int64_t capR, int64_t B,
                         int64_t* __restrict__ out0, int64_t* __restrict__ out1,
                         int64_t* __restrict__ out2, int64_t* __restrict__ out3, int64_t cap,
                         unsigned long long* counter, uint32_t* __restrict__ overflow_flags,
                         int* __restrict__ any_overflow)
{
  constexpr int WATER = STAGE - STAGE / 4;
  extern __shared__ __attribute__((aligned(16))) char smem[];
  longlong2* tbl = ((longlong2*)smem) + (threadIdx.x >> 6) * SLOTS;
  int64_t* stage = (int64_t*)(((longlong2*)smem) + WAVES * SLOTS);
  uint32_t* cur_sh = (uint32_t*)(stage + 4 * STAGE);
  long long* base_sh = (long long*)(cur_sh + 2);
  const int lane = threadIdx.x & 63;
  const uint64_t lt = lane ? (~0ull >> (64 - lane)) : 0ull;
  const uint32_t smask = SLOTS - 1;
  if (threadIdx.x == 0) *cur_sh = 0;
  __syncthreads();

  const int64_t bstride = (int64_t)gridDim.x * WAVES;
  for (int64_t bbase = (int64_t)blockIdx.x * WAVES;; bbase += bstride) {
    const bool done = bbase >= B;
    if (!done) {
      const int64_t b = bbase + (threadIdx.x >> 6);
      bool active = b < B;
      int64_t l0 = 0, l1 = 0, r0 = 0, r1 = 0;
      if (active) {
        if (SLACK) {
          l0 = b * capL;
          l1 = l0 + llen[b];
          r0 = b * capR;
          r1 = r0 + rlen[b];
        } else {
          l0 = loff[b];
          l1 = loff[b + 1];
          r0 = roff[b];
          r1 = roff[b + 1];
        }
      }
      const int64_t lnb = l1 - l0;
      if (active && (lnb == 0 || r1 == r0)) active = false;
      if (active && lnb > SLOTS * 3 / 4) {
        if (lane == 0) {
          overflow_flags[b] = 1;
          atomicOr(any_overflow, 1);
        }
        active = false;
      }
      if (active) {
        for (int s = lane; s < SLOTS; s += 64) tbl[s].x = EMPTY;
        for (int64_t i = l0 + lane; i < l1; i += 64) {
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
        for (int64_t j0 = r0; j0 < r1; j0 += 64) {
          const int64_t j = j0 + lane;
          const bool rowv = j < r1;
          longlong2 prow;
          prow.x = 0;
          prow.y = 0;
          uint32_t slot = 0;
          if (rowv) {
            prow = rrows[j];
            slot = (uint32_t)dj_mix64((uint64_t)prow.x) & smask;
          }
          bool walking = rowv;
          for (;;) {
            int64_t mval = 0;
            bool have = false;
            if (walking) {
              longlong2 e = tbl[slot];
              if (e.x == EMPTY) {
                walking = false;
              } else {
                slot = (slot + 1) & smask;
                if (e.x == prow.x) {
                  mval = e.y;
                  have = true;
                }
              }
            }
            uint64_t m = __ballot(have);
            if (m) {
              const int leader = (int)(__ffsll((unsigned long long)m) - 1);
              uint32_t basep = 0;
              if (lane == leader) basep = atomicAdd(cur_sh, (uint32_t)__popcll(m));
              basep = __shfl(basep, leader);
              if (have) {
                uint32_t pos = basep + (uint32_t)__popcll(m & lt);
                if (pos < (uint32_t)STAGE) {
                  stage[0 * STAGE + pos] = prow.x;
                  stage[1 * STAGE + pos] = mval;
                  stage[2 * STAGE + pos] = prow.x;
                  stage[3 * STAGE + pos] = prow.y;
                } else {
                  long long idx = (long long)atomicAdd(counter, 1ull);
                  if (idx < cap) {
                    out0[idx] = prow.x;
                    out1[idx] = mval;
                    out2[idx] = prow.x;
                    out3[idx] = prow.y;
                  }
                }
              }
            }
            if (__ballot(walking) == 0) break;
          }
        }
      }
    }
    __syncthreads();
    const uint32_t cur = *cur_sh;
    if (cur >= (uint32_t)WATER || (done && cur)) {
      const uint32_t total = min(cur, (uint32_t)STAGE);
      if (threadIdx.x == 0)
        *base_sh = (long long)atomicAdd(counter, (unsigned long long)total);
      __syncthreads();
      const long long gbase = *base_sh;
      for (uint32_t i = threadIdx.x; i < total; i += blockDim.x) {
        long long idx = gbase + (long long)i;
        if (idx < cap) {
          out0[idx] = stage[0 * STAGE + i];
          out1[idx] = stage[1 * STAGE + i];
          out2[idx] = stage[2 * STAGE + i];
          out3[idx] = stage[3 * STAGE + i];
        }
      }
      __syncthreads();
      if (threadIdx.x == 0) *cur_sh = 0;
      __syncthreads();
    }
    if (done) break;
  }
}

/* ------------- J-k: K-bucket flush + wave-aggregated emit + watermark ---- */
template <int SLOTS2, int STAGE>
__global__ __launch_bounds__(1024) void lds_join_kw_kernel(
  const longlong2* __restrict__ lrows, const int64_t* __restrict__ loff,
  const uint32_t* __restrict__ llen, int64_t capL, const longlong2* __restrict__ rrows,
  const int64_t* __restrict__ roff, const uint32_t* __restrict__ rlen, int64_t capR, int B,
  int64_t* __restrict__ out0, int64_t* __restrict__ out1, int64_t* __restrict__ out2,
  int64_t* __restrict__ out3, int64_t cap, unsigned long long* counter,
  uint32_t* __restrict__ overflow_flags, int* __restrict__ any_overflow)
{
  constexpr int WATER = STAGE - STAGE / 4;
  extern __shared__ __attribute__((aligned(16))) char smem[];
  longlong2* tbl = (longlong2*)smem;
  int64_t* stage = (int64_t*)(tbl + SLOTS2);
  long long* base_sh = (long long*)(stage + 4 * STAGE);
  uint32_t* cur_sh = (uint32_t*)(base_sh + 1);
  const uint32_t smask = SLOTS2 - 1;
  const int lane = threadIdx.x & 63;
  const uint64_t lt = lane ? (~0ull >> (64 - lane)) : 0ull;
  const bool slack = llen != nullptr;
  if (threadIdx.x == 0) *cur_sh = 0;
  __syncthreads();
  for (int bb = blockIdx.x;; bb += gridDim.x) {
    const bool done = bb >= B;
    if (!done) {
      const int b = bb;
      int64_t l0, l1, r0, r1;
      if (slack) {
        l0 = (int64_t)b * capL;
        l1 = l0 + llen[b];
        r0 = (int64_t)b * capR;
        r1 = r0 + rlen[b];
      } else {
        l0 = loff[b];
        l1 = loff[b + 1];
        r0 = roff[b];
        r1 = roff[b + 1];
      }
      const int64_t lnb = l1 - l0;
      bool active = !(lnb == 0 || r1 == r0);
      if (active && lnb > SLOTS2 * 3 / 4) {
        if (threadIdx.x == 0) {
          overflow_flags[b] = 1;
          atomicOr(any_overflow, 1);
        }
        active = false;
      }
      if (active) {
        for (int s = threadIdx.x; s < SLOTS2; s += blockDim.x) tbl[s].x = EMPTY;
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
        for (int64_t j0 = r0; j0 < r1; j0 += blockDim.x) {
          const int64_t j = j0 + threadIdx.x;
          const bool rowv = j < r1;
          longlong2 prow;
          prow.x = 0;
          prow.y = 0;
          uint32_t slot = 0;
          if (rowv) {
            prow = rrows[j];
            slot = (uint32_t)dj_mix64((uint64_t)prow.x) & smask;
          }
          bool walking = rowv;
          for (;;) {
            int64_t mval = 0;
            bool have = false;
            if (walking) {
              longlong2 e = tbl[slot];
              if (e.x == EMPTY) {
                walking = false;
              } else {
                slot = (slot + 1) & smask;
                if (e.x == prow.x) {
                  mval = e.y;
                  have = true;
                }
              }
            }
            uint64_t m = __ballot(have);
            if (m) {
              const int leader = (int)(__ffsll((unsigned long long)m) - 1);
              uint32_t basep = 0;
              if (lane == leader) basep = atomicAdd(cur_sh, (uint32_t)__popcll(m));
              basep = __shfl(basep, leader);
              if (have) {
                uint32_t pos = basep + (uint32_t)__popcll(m & lt);
                if (pos < (uint32_t)STAGE) {
                  stage[0 * STAGE + pos] = prow.x;
                  stage[1 * STAGE + pos] = mval;
                  stage[2 * STAGE + pos] = prow.x;
                  stage[3 * STAGE + pos] = prow.y;
                } else {
                  long long idx = (long long)atomicAdd(counter, 1ull);
                  if (idx < cap) {
                    out0[idx] = prow.x;
                    out1[idx] = mval;
                    out2[idx] = prow.x;
                    out3[idx] = prow.y;
                  }
                }
              }
            }
            if (__ballot(walking) == 0) break;
          }
        }
      }
    }
    __syncthreads();
    const uint32_t cur = *cur_sh;
    if (cur >= (uint32_t)WATER || (done && cur)) {
      const uint32_t total = min(cur, (uint32_t)STAGE);
      if (threadIdx.x == 0)
        *base_sh = (long long)atomicAdd(counter, (unsigned long long)total);
      __syncthreads();
      const long long gbase = *base_sh;
      for (uint32_t i = threadIdx.x; i < total; i += blockDim.x) {
        long long idx = gbase + (long long)i;
        if (idx < cap) {
          out0[idx] = stage[0 * STAGE + i];
          out1[idx] = stage[1 * STAGE + i];
          out2[idx] = stage[2 * STAGE + i];
          out3[idx] = stage[3 * STAGE + i];
        }
      }
      __syncthreads();
      if (threadIdx.x == 0) *cur_sh = 0;
      __syncthreads();
    }
    if (done) break;
  }
}

/* ------------------------------------------------------------------ main */

struct DBuf {
  void* p{nullptr};
  DBuf() = default;
  explicit DBuf(size_t bytes) { CHECK(hipMalloc(&p, bytes)); }
  ~DBuf()
  {
    if (p) (void)hipFree(p);
  }
  DBuf(const DBuf&) = delete;
  DBuf(DBuf&& o) : p(o.p) { o.p = nullptr; }
  DBuf& operator=(DBuf&& o)
  {
    if (p) (void)hipFree(p);
    p = o.p;
    o.p = nullptr;
    return *this;
  }
  int64_t* i64() const { return (int64_t*)p; }
  uint32_t* u32() const { return (uint32_t*)p; }
};

static float time_body(int reps, const std::function<void()>& fn)
{
  hipEvent_t e0, e1;
  CHECK(hipEventCreate(&e0));
  CHECK(hipEventCreate(&e1));
  float best = 1e30f;
  for (int i = 0; i < reps; i++) {
    CHECK(hipEventRecord(e0));
    fn();
    CHECK(hipEventRecord(e1));
    CHECK(hipEventSynchronize(e1));
    float ms;
    CHECK(hipEventElapsedTime(&ms, e0, e1));
    if (ms < best) best = ms;
  }
  CHECK(hipEventDestroy(e0));
  CHECK(hipEventDestroy(e1));
  return best;
}

int main(int argc, char** argv)
{
  int64_t n = 100'000'000;
  if (argc > 1) n = atoll(argv[1]);
  const int64_t rand_max = 2 * n;
  printf("join_v3: n=%lld rows per table, sel 0.3\n", (long long)n);

  DBuf bk(n * 8), bp(n * 8), pk(n * 8), pp(n * 8);
  dj::generate_build(bk.i64(), bp.i64(), n, rand_max, DJ_DEFAULT_SEED, true, 0, n, 0);
  dj::generate_probe(pk.i64(), pp.i64(), n, rand_max, 0.3, DJ_DEFAULT_SEED, 0, n, 0);
  CHECK(hipDeviceSynchronize());

  const int64_t cap = n / 2;
  DBuf o0(cap * 8), o1(cap * 8), o2(cap * 8), o3(cap * 8);
  DBuf counter(8), anyovf(8), err(8), acc(8);

  unsigned long long ref_count = 0, ref_sum = 0;
  auto verify = [&](const char* name, float ms) {
    unsigned long long c, s;
    CHECK(hipMemcpy(&c, counter.p, 8, hipMemcpyDeviceToHost));
    CHECK(hipMemset(acc.p, 0, 8));
    hipLaunchKernelGGL(checksum_kernel, dim3(2048), dim3(256), 0, 0, o0.i64(), o1.i64(),
                       o2.i64(), o3.i64(), (int64_t)c, (unsigned long long*)acc.p);
    CHECK(hipMemcpy(&s, acc.p, 8, hipMemcpyDeviceToHost));
    if (!ref_count) {
      ref_count = c;
      ref_sum = s;
    }
    printf("%-34s %.3f ms  count=%llu  %s\n", name, ms, c,
           (c == ref_count && s == ref_sum) ? "OK" : "MISMATCH");
  };

  /* reference output via product path at B=131072 (also the baseline time) */
  {
    const int B = dj::bucket_count_for(n, n);
    const int PA = dj::bucket_groups_for(B);
    const int64_t capA = dj::slack_capA(n, PA);
    DBuf tmp((size_t)PA * capA * 16);
    DBuf counts((size_t)dj::kBucketBlocks * PA * 4), totals((size_t)PA * 4);
    DBuf segoff((size_t)(PA + 1) * 8);
    DBuf lpairs((size_t)n * 16), rpairs((size_t)n * 16);
    DBuf loff((size_t)(B + 1) * 8), roff((size_t)(B + 1) * 8);
    DBuf flags((size_t)B * 4);
    CHECK(hipMemset(anyovf.p, 0, 4));
    float tp = time_body(3, [&] {
      dj::bucket_partition2(bk.i64(), bp.i64(), n, B, (longlong2*)tmp.p, counts.u32(),
                            totals.u32(), segoff.i64(), loff.i64(), (longlong2*)lpairs.p,
                            (int*)anyovf.p, 0);
    });
    dj::bucket_partition2(pk.i64(), pp.i64(), n, B, (longlong2*)tmp.p, counts.u32(),
                          totals.u32(), segoff.i64(), roff.i64(), (longlong2*)rpairs.p,
                          (int*)anyovf.p, 0);
    CHECK(hipDeviceSynchronize());
    printf("baseline bucket_partition2 B=%d: %.3f ms/table\n", B, tp);
    float t = time_body(3, [&] {
      CHECK(hipMemsetAsync(counter.p, 0, 8, 0));
      CHECK(hipMemsetAsync(anyovf.p, 0, 4, 0));
      CHECK(hipMemsetAsync(err.p, 0, 4, 0));
      dj::lds_join((longlong2*)lpairs.p, loff.i64(), (longlong2*)rpairs.p, roff.i64(), B,
                   2048, o0.i64(), o1.i64(), o2.i64(), o3.i64(), cap, counter.i64(),
                   flags.u32(), (int*)anyovf.p, (int*)err.p, 0);
    });
    verify("BASE lds_join B=131072", t);
    /* J-kw on compact layout: watermark flush + wave-agg emit */
    auto run_kw = [&](auto slots_tag, auto stage_tag, const char* nm) {
      constexpr int SL = decltype(slots_tag)::value;
      constexpr int ST = decltype(stage_tag)::value;
      float tk = time_body(3, [&] {
        CHECK(hipMemsetAsync(counter.p, 0, 8, 0));
        CHECK(hipMemsetAsync(anyovf.p, 0, 4, 0));
        size_t lds = (size_t)SL * 16 + 4 * (size_t)ST * 8 + 16;
        int grid = B < 8192 ? B : 8192;
        hipLaunchKernelGGL((lds_join_kw_kernel<SL, ST>), dim3(grid), dim3(1024), lds, 0,
                           (longlong2*)lpairs.p, loff.i64(), nullptr, 0, (longlong2*)rpairs.p,
                           roff.i64(), nullptr, 0, B, o0.i64(), o1.i64(), o2.i64(), o3.i64(),
                           cap, (unsigned long long*)counter.p, flags.u32(), (int*)anyovf.p);
        CHECK(hipGetLastError());
      });
      verify(nm, tk);
    };
    run_kw(std::integral_constant<int, 2048>{}, std::integral_constant<int, 1024>{},
           "J-kw S2048 ST1024 B=131072");
    run_kw(std::integral_constant<int, 2048>{}, std::integral_constant<int, 1536>{},
           "J-kw S2048 ST1536 B=131072");
  }

  /* pipelined pass A (PA=512 and 1024) */
  for (int PA : {512, 1024}) {
    const int64_t capA = dj::slack_capA(n, PA);
    DBuf tmpA((size_t)PA * capA * 16), cursA((size_t)PA * 4);
    size_t ldsA = (size_t)ATILE * 16 + 4 * (size_t)PA * 4 + 64;
    CHECK(hipMemset(anyovf.p, 0, 4));
    float tA = time_body(3, [&] {
      CHECK(hipMemsetAsync(cursA.p, 0, (size_t)PA * 4, 0));
      hipLaunchKernelGGL(slackA_pipe_kernel, dim3(dj::kBucketBlocks), dim3(ATHREADS), ldsA, 0,
                         bk.i64(), bp.i64(), n, PA, capA, cursA.u32(), (int*)anyovf.p,
                         (longlong2*)tmpA.p);
      CHECK(hipGetLastError());
    });
    int ovf;
    CHECK(hipMemcpy(&ovf, anyovf.p, 4, hipMemcpyDeviceToHost));
    printf("A-pipe PA=%d: %.3f ms/table (ovf=%d)\n", PA, tA, ovf);

    /* pipelined pass B on this A output */
    for (int F : {256, 512, 1024}) {
      const int64_t B = (int64_t)PA * F;
      if (B < 131072 || B > 1048576) continue;
      const int64_t lambda = n / B;
      int64_t capB = lambda + (int64_t)(7.0 * sqrt((double)lambda)) + 8;
      capB = (capB + 7) & ~7ll;
      /* COUNTED (compact) variant */
      {
        DBuf outp((size_t)n * 16), boff((size_t)(B + 1) * 8), segout((size_t)(PA + 1) * 8);
        /* segout = exclusive scan of seg lens (on host, tiny) */
        std::vector<uint32_t> lensA(PA);
        CHECK(hipMemcpy(lensA.data(), cursA.p, (size_t)PA * 4, hipMemcpyDeviceToHost));
        std::vector<int64_t> so(PA + 1, 0);
        for (int i = 0; i < PA; i++)
          so[i + 1] = so[i] + std::min<int64_t>(lensA[i], capA);
        CHECK(hipMemcpy(segout.p, so.data(), (size_t)(PA + 1) * 8, hipMemcpyHostToDevice));
        size_t ldsB = (size_t)BTILE * 16 + 4 * (size_t)F * 4 + 64;
        float tB = time_body(3, [&] {
          hipLaunchKernelGGL((subpartB_pipe_kernel<true>), dim3(PA), dim3(ATHREADS), ldsB, 0,
                             (longlong2*)tmpA.p, cursA.u32(), capA, (int)B, F, 0,
                             segout.i64(), (longlong2*)outp.p, boff.i64(), nullptr,
                             (int*)anyovf.p);
          CHECK(hipGetLastError());
        });
        printf("B-pipe-cnt PA=%d F=%d: %.3f ms/table\n", PA, F, tB);
      }
      /* slack variant */
      {
        DBuf outp((size_t)B * capB * 16), lens((size_t)B * 4);
        size_t ldsB = (size_t)BTILE * 16 + 4 * (size_t)F * 4 + 64;
        float tB = time_body(3, [&] {
          hipLaunchKernelGGL((subpartB_pipe_kernel<false>), dim3(PA), dim3(ATHREADS), ldsB, 0,
                             (longlong2*)tmpA.p, cursA.u32(), capA, (int)B, F, capB, nullptr,
                             (longlong2*)outp.p, nullptr, lens.u32(), (int*)anyovf.p);
          CHECK(hipGetLastError());
        });
        printf("B-pipe-slk PA=%d F=%d capB=%lld: %.3f ms/table\n", PA, F, (long long)capB,
               tB);
      }
    }
  }

  /* full new pipeline at B=1M: A-pipe + B-pipe-slk both tables, then wave
   * joins (and J-kw slack at B=131072 via F=256... F=256 needs PA=512) */
  {
    const int PA = 1024, F = 1024;
    const int64_t B = (int64_t)PA * F;
    const int64_t capA = dj::slack_capA(n, PA);
    const int64_t lambda = n / B;
    int64_t capB = lambda + (int64_t)(7.0 * sqrt((double)lambda)) + 8;
    capB = (capB + 7) & ~7ll;
    DBuf tmpA((size_t)PA * capA * 16), cursA((size_t)PA * 4);
    DBuf lpairs((size_t)B * capB * 16), rpairs((size_t)B * capB * 16);
    DBuf llen((size_t)B * 4), rlen((size_t)B * 4);
    DBuf flags((size_t)B * 4);
    size_t ldsA = (size_t)ATILE * 16 + 4 * (size_t)PA * 4 + 64;
    size_t ldsB = (size_t)BTILE * 16 + 4 * (size_t)F * 4 + 64;
    CHECK(hipMemset(anyovf.p, 0, 4));
    auto passAB = [&](const int64_t* keys, const int64_t* pay, longlong2* out,
                      uint32_t* lens) {
      CHECK(hipMemsetAsync(cursA.p, 0, (size_t)PA * 4, 0));
      hipLaunchKernelGGL(slackA_pipe_kernel, dim3(dj::kBucketBlocks), dim3(ATHREADS), ldsA, 0,
                         keys, pay, n, PA, capA, cursA.u32(), (int*)anyovf.p,
                         (longlong2*)tmpA.p);
      CHECK(hipGetLastError());
      hipLaunchKernelGGL((subpartB_pipe_kernel<false>), dim3(PA), dim3(ATHREADS), ldsB, 0,
                         (longlong2*)tmpA.p, cursA.u32(), capA, (int)B, F, capB, nullptr, out,
                         nullptr, lens, (int*)anyovf.p);
      CHECK(hipGetLastError());
    };
    float tfull = time_body(2, [&] {
      passAB(bk.i64(), bp.i64(), (longlong2*)lpairs.p, llen.u32());
      passAB(pk.i64(), pp.i64(), (longlong2*)rpairs.p, rlen.u32());
    });
    int ovf;
    CHECK(hipMemcpy(&ovf, anyovf.p, 4, hipMemcpyDeviceToHost));
    printf("pipe A+B both tables B=%lld: %.3f ms (ovf=%d)\n", (long long)B, tfull, ovf);

    auto run_wave = [&](auto wtag, auto stag, auto ttag, const char* nm) {
      constexpr int W = decltype(wtag)::value;
      constexpr int SL = decltype(stag)::value;
      constexpr int ST = decltype(ttag)::value;
      float t = time_body(3, [&] {
        CHECK(hipMemsetAsync(counter.p, 0, 8, 0));
        CHECK(hipMemsetAsync(anyovf.p, 0, 4, 0));
        size_t lds = (size_t)W * SL * 16 + (size_t)ST * 32 + 64;
        int64_t bblocks = (B + W - 1) / W;
        int grid = (int)(bblocks < 8192 ? bblocks : 8192);
        hipLaunchKernelGGL((wave_join2_kernel<W, SL, ST, true>), dim3(grid), dim3(W * 64),
                           lds, 0, (longlong2*)lpairs.p, nullptr, llen.u32(), capB,
                           (longlong2*)rpairs.p, nullptr, rlen.u32(), capB, B, o0.i64(),
                           o1.i64(), o2.i64(), o3.i64(), cap,
                           (unsigned long long*)counter.p, flags.u32(), (int*)anyovf.p);
        CHECK(hipGetLastError());
      });
      verify(nm, t);
    };
    run_wave(std::integral_constant<int, 8>{}, std::integral_constant<int, 256>{},
             std::integral_constant<int, 256>{}, "J-wave2 W8 S256 ST256 (4blk/CU)");
    run_wave(std::integral_constant<int, 8>{}, std::integral_constant<int, 256>{},
             std::integral_constant<int, 512>{}, "J-wave2 W8 S256 ST512 (3blk/CU)");
    run_wave(std::integral_constant<int, 4>{}, std::integral_constant<int, 256>{},
             std::integral_constant<int, 256>{}, "J-wave2 W4 S256 ST256 (6blk/CU)");
  }

  printf("done\n");
  return 0;
}
