/*
 * join_v2.hip — round-2 candidate kernels for the bucketed local join,
 * timed and checksum-verified against the round-1 product kernels on
 * bench-shaped data (default 100M x 100M, selectivity 0.3).
 *
 * Candidates (motivated by profiles/r01_ablation.txt: the r1 join's binding
 * costs are per-bucket serialization ~6.7us/bucket and the per-bucket global
 * reserve atomic; the r1 pass B's cost is the seghist count sweep that
 * fetches full 16 B lines for 8 B keys):
 *   J-wave  : wave-per-bucket join — one 64-lane wave owns one small bucket
 *             (256-slot LDS table), no block-wide syncs between buckets,
 *             block-shared output stage flushed every FE rounds (global
 *             counter atomics cut ~10x).
 *   J-k4    : round-1 kernel shape, but the stage accumulates K=4 sequential
 *             buckets per flush (cuts reserve atomics 4x; minimal change).
 *   A-ws    : pass-A slack scatter with wave-level (shfl) scans replacing the
 *             Hillis-Steele LDS scan (20 -> ~7 __syncthreads per tile).
 *   B-slack : pass B with NO count sweep — per-sub-bucket slack segments
 *             (start = b*capB analytic, LDS cursors, lengths out), wave-scan
 *             staging. Join variants read (capB, len[]) slack layout.
 *
 * Build: hipcc --offload-arch=gfx950 -O3 join_v2.hip \
 *          ../distributed_join_amd/csrc/dj_kernels.hip -o join_v2
 * Diagnostic only (not linked into the product library).
 */
#include "../distributed_join_amd/csrc/dj_kernels.hpp"
#include "../distributed_join_amd/csrc/dj_rng.h"

#include <hip/hip_runtime.h>

#include <cmath>
#include <cstdint>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <functional>
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

/* ------------------------------------------------------------ checksum -- */

__global__ void checksum_kernel(const int64_t* o0, const int64_t* o1, const int64_t* o2,
                                const int64_t* o3, int64_t n, unsigned long long* acc)
{
  unsigned long long local = 0;
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride)
    local += dj_mix64((uint64_t)o0[i]) * 3 + dj_mix64((uint64_t)o1[i]) * 5 +
             dj_mix64((uint64_t)o2[i]) * 7 + dj_mix64((uint64_t)o3[i]);
  /* wave-reduce then one atomic per wave */
  for (int off = 32; off; off >>= 1) local += __shfl_down(local, off);
  if ((threadIdx.x & 63) == 0) atomicAdd(acc, local);
}

/* -------------------------------------------- wave-per-bucket join ------ */
/* One wave owns one bucket: private SLOTS-slot LDS table (cap = 3/4),
 * build + probe with no block-wide sync (DS ops from one wave execute in
 * order). Matches append to a block-shared LDS stage (wave-aggregated
 * ballot + one ds-atomic per ballot round); the stage flushes to the global
 * output every FE rounds with ONE global counter atomic. */
template <int WAVES, int SLOTS, int STAGE, bool SLACK>
__global__ __launch_bounds__(WAVES * 64) void wave_join_kernel(
  const longlong2* __restrict__ lrows, const int64_t* __restrict__ loff,
  const uint32_t* __restrict__ llen, int64_t capL, const longlong2* __restrict__ rrows,
  const int64_t* __restrict__ roff, const uint32_t* __restrict__ rlen, int64_t capR,
  int64_t B, int FE, int64_t* __restrict__ out0, int64_t* __restrict__ out1,
  int64_t* __restrict__ out2, int64_t* __restrict__ out3, int64_t cap,
  unsigned long long* counter, uint32_t* __restrict__ overflow_flags,
  int* __restrict__ any_overflow)
{
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
  int64_t bbase = (int64_t)blockIdx.x * WAVES;
  int round = 0;
  for (;;) {
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
        /* build (wave-local; DS in-order within the wave) */
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
        /* probe + wave-aggregated stage append */
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
      bbase += bstride;
      round++;
    }
    if (round >= FE || done) {
      __syncthreads();
      const uint32_t total = min(*cur_sh, (uint32_t)STAGE);
      if (threadIdx.x == 0 && total)
        *base_sh = (long long)atomicAdd(counter, (unsigned long long)total);
      __syncthreads();
      if (total) {
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
      }
      __syncthreads();
      if (threadIdx.x == 0) *cur_sh = 0;
      __syncthreads();
      round = 0;
    }
    if (done) break;
  }
}

template <int WAVES, int SLOTS, int STAGE, bool SLACK>
float run_wave_join(const longlong2* lrows, const int64_t* loff, const uint32_t* llen,
                    int64_t capL, const longlong2* rrows, const int64_t* roff,
                    const uint32_t* rlen, int64_t capR, int64_t B, int FE, int64_t* o0,
                    int64_t* o1, int64_t* o2, int64_t* o3, int64_t cap,
                    unsigned long long* counter, uint32_t* flags, int* anyovf, int reps,
                    unsigned long long* h_count, unsigned long long* h_sum,
                    unsigned long long* d_acc)
{
  size_t lds = (size_t)WAVES * SLOTS * 16 + 4 * (size_t)STAGE * 8 + 32;
  int64_t bblocks = (B + WAVES - 1) / WAVES;
  int grid = (int)(bblocks < 8192 ? bblocks : 8192);
  hipEvent_t e0, e1;
  CHECK(hipEventCreate(&e0));
  CHECK(hipEventCreate(&e1));
  float best = 1e30f;
  for (int it = 0; it < reps; it++) {
    CHECK(hipMemsetAsync(counter, 0, 8, 0));
    CHECK(hipMemsetAsync(anyovf, 0, 4, 0));
    CHECK(hipEventRecord(e0));
    hipLaunchKernelGGL((wave_join_kernel<WAVES, SLOTS, STAGE, SLACK>), dim3(grid),
                       dim3(WAVES * 64), lds, 0, lrows, loff, llen, capL, rrows, roff, rlen,
                       capR, B, FE, o0, o1, o2, o3, cap, counter, flags, anyovf);
    CHECK(hipGetLastError());
    CHECK(hipEventRecord(e1));
    CHECK(hipEventSynchronize(e1));
    float ms;
    CHECK(hipEventElapsedTime(&ms, e0, e1));
    if (ms < best) best = ms;
  }
  unsigned long long cnt;
  CHECK(hipMemcpy(&cnt, counter, 8, hipMemcpyDeviceToHost));
  int ovf;
  CHECK(hipMemcpy(&ovf, anyovf, 4, hipMemcpyDeviceToHost));
  CHECK(hipMemset(d_acc, 0, 8));
  hipLaunchKernelGGL(checksum_kernel, dim3(2048), dim3(256), 0, 0, o0, o1, o2, o3,
                     (int64_t)cnt, d_acc);
  CHECK(hipMemcpy(h_sum, d_acc, 8, hipMemcpyDeviceToHost));
  *h_count = cnt;
  if (ovf) printf("    (any_overflow=%d!)\n", ovf);
  CHECK(hipEventDestroy(e0));
  CHECK(hipEventDestroy(e1));
  return best;
}

/* ------------------------------- K-bucket accumulated flush (J-k4) ------ */
template <int SLOTS2, int KBUK>
__global__ __launch_bounds__(1024) void lds_join_k_kernel(
  const longlong2* __restrict__ lrows, const int64_t* __restrict__ loff,
  const longlong2* __restrict__ rrows, const int64_t* __restrict__ roff, int B,
  int64_t* __restrict__ out0, int64_t* __restrict__ out1, int64_t* __restrict__ out2,
  int64_t* __restrict__ out3, int64_t cap, unsigned long long* counter,
  uint32_t* __restrict__ overflow_flags, int* __restrict__ any_overflow)
{
  constexpr int S = 1024; /* stage rows */
  extern __shared__ __attribute__((aligned(16))) char smem[];
  longlong2* tbl = (longlong2*)smem;
  int64_t* stage = (int64_t*)(tbl + SLOTS2);
  long long* base_sh = (long long*)(stage + 4 * S);
  uint32_t* cur_sh = (uint32_t*)(base_sh + 1);
  const uint32_t smask = SLOTS2 - 1;
  if (threadIdx.x == 0) *cur_sh = 0;
  __syncthreads();
  for (int bb = blockIdx.x * KBUK; bb < B; bb += gridDim.x * KBUK) {
    for (int k = 0; k < KBUK; k++) {
      const int b = bb + k;
      if (b >= B) break;
      const int64_t l0 = loff[b], l1 = loff[b + 1];
      const int64_t r0 = roff[b], r1 = roff[b + 1];
      const int64_t lnb = l1 - l0;
      if (lnb == 0 || r1 == r0) continue;
      if (lnb > SLOTS2 * 3 / 4) {
        if (threadIdx.x == 0) {
          overflow_flags[b] = 1;
          atomicOr(any_overflow, 1);
        }
        continue;
      }
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
    }
    /* flush once per K buckets */
    const uint32_t total = min(*cur_sh, (uint32_t)S);
    if (threadIdx.x == 0 && total)
      *base_sh = (long long)atomicAdd(counter, (unsigned long long)total);
    __syncthreads();
    if (total) {
      const long long gbase = *base_sh;
      for (uint32_t i = threadIdx.x; i < total; i += blockDim.x) {
        long long idx = gbase + (long long)i;
        if (idx < cap) {
          out0[idx] = stage[0 * S + i];
          out1[idx] = stage[1 * S + i];
          out2[idx] = stage[2 * S + i];
          out3[idx] = stage[3 * S + i];
        }
      }
    }
    __syncthreads();
    if (threadIdx.x == 0) *cur_sh = 0;
    __syncthreads();
  }
}

/* ------------------------------------------ wave-scan staged scatter ---- */
/* shfl-based exclusive scan of hist[0..P) into base[0..P): per-wave
 * inclusive shfl_up scan + one cross-wave partial scan; 2 syncs total
 * (Hillis-Steele over P=512/1024 costs 18-20). */
template <int P_MAX>
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
    if (lane < nw) partials[lane] = pincl - pv; /* exclusive */
  }
  __syncthreads();
  if (tid < P) base[tid] = incl - v + partials[wid];
}

/* pass-A slack scatter with wave scans (copy of bucket_scatter_slack_kernel,
 * scan replaced; P <= 1024) */
constexpr int SLACK_TILE2 = 8192;
__global__ __launch_bounds__(1024) void slackA_ws_kernel(
  const int64_t* __restrict__ keys, const int64_t* __restrict__ pay, int64_t n, int P,
  int64_t capA, uint32_t* __restrict__ gcursor, int* __restrict__ any_overflow,
  longlong2* __restrict__ out_pairs)
{
  extern __shared__ __attribute__((aligned(16))) char smem[];
  longlong2* tbuf = (longlong2*)smem;
  uint32_t* hist = (uint32_t*)(tbuf + SLACK_TILE2);
  uint32_t* base = hist + P;
  uint32_t* gcur = base + P;
  uint32_t* glim = gcur + P;
  uint32_t* partials = glim + P; /* 16 */
  __shared__ int s_ovf;
  const int tid = threadIdx.x;
  if (tid == 0) s_ovf = 0;
  const int64_t chunk = (n + gridDim.x - 1) / gridDim.x;
  const int64_t start = (int64_t)blockIdx.x * chunk;
  const int64_t end = min(start + chunk, n);
  constexpr int VPT = SLACK_TILE2 / 1024;
  for (int64_t t0 = start; t0 < end; t0 += SLACK_TILE2) {
    const int count = (int)min((int64_t)SLACK_TILE2, end - t0);
    if (tid < P) hist[tid] = 0;
    __syncthreads();
    longlong2 r[VPT];
    uint32_t g[VPT], rank[VPT];
#pragma unroll
    for (int v = 0; v < VPT; v++) {
      int64_t i = t0 + (int64_t)v * blockDim.x + tid;
      if (i < end) {
        r[v].x = __builtin_nontemporal_load(&keys[i]);
        r[v].y = pay ? __builtin_nontemporal_load(&pay[i]) : i;
        g[v] = (uint32_t)(dj_mix64((uint64_t)r[v].x) >> 40) & (uint32_t)(P - 1);
        rank[v] = atomicAdd(&hist[g[v]], 1u);
      }
    }
    __syncthreads();
    wave_excl_scan<1024>(hist, base, partials, P);
    /* claim this tile's run in each group's slack segment */
    if (tid < P) {
      const int64_t seg0 = (int64_t)tid * capA;
      gcur[tid] = (uint32_t)seg0 + atomicAdd(&gcursor[tid], hist[tid]);
      glim[tid] = (uint32_t)(seg0 + capA);
    }
    __syncthreads();
#pragma unroll
    for (int v = 0; v < VPT; v++) {
      int64_t i = t0 + (int64_t)v * blockDim.x + tid;
      if (i < end) tbuf[base[g[v]] + rank[v]] = r[v];
    }
    __syncthreads();
    for (int pos = tid; pos < count; pos += blockDim.x) {
      longlong2 row = tbuf[pos];
      uint32_t gg = (uint32_t)(dj_mix64((uint64_t)row.x) >> 40) & (uint32_t)(P - 1);
      uint32_t dst = gcur[gg] + (uint32_t)(pos - base[gg]);
      if (dst < glim[gg])
        out_pairs[dst] = row;
      else
        s_ovf = 1;
    }
    __syncthreads();
  }
  __syncthreads();
  if (tid == 0 && s_ovf) atomicOr(any_overflow, 2);
}

/* runtime-F sub-bucket bits, identical to dj_kernels.hip subF_of */
__device__ __forceinline__ uint32_t subF2(int64_t key, int F)
{
  uint64_t m = dj_mix64((uint64_t)key);
  if (F <= 256) return (uint32_t)(m >> 32) & (uint32_t)(F - 1);
  uint32_t lo = (uint32_t)(m >> 32) & 255u;
  uint32_t hi = (uint32_t)(m >> 50) & (uint32_t)((F >> 8) - 1);
  return lo | (hi << 8);
}

/* pass-B SLACK: block = pass-A group a; F sub-buckets at analytic slack
 * starts (b*capB); LDS cursors (only this block writes group a); no count
 * sweep; lengths written at block end; wave scans. */
constexpr int TILE2 = 4096;
__global__ __launch_bounds__(1024) void subpart_slack2_kernel(
  const longlong2* __restrict__ in_pairs, const uint32_t* __restrict__ seg_len, int64_t capA,
  int F, int64_t capB, longlong2* __restrict__ out_pairs, uint32_t* __restrict__ lens,
  int* __restrict__ any_overflow)
{
  extern __shared__ __attribute__((aligned(16))) char smem[];
  longlong2* tbuf = (longlong2*)smem;
  uint32_t* hist = (uint32_t*)(tbuf + TILE2);
  uint32_t* base = hist + F;
  uint32_t* gcur = base + F;
  uint32_t* partials = gcur + F; /* 16 */
  __shared__ int s_ovf;
  const int tid = threadIdx.x;
  const int a = blockIdx.x;
  const int64_t s0 = (int64_t)a * capA;
  const int64_t s1 = s0 + seg_len[a];
  if (tid == 0) s_ovf = 0;
  for (int j = tid; j < F; j += blockDim.x)
    gcur[j] = (uint32_t)(((int64_t)a * F + j) * capB);
  __syncthreads();
  constexpr int VPT = TILE2 / 1024;
  for (int64_t t0 = s0; t0 < s1; t0 += TILE2) {
    const int count = (int)min((int64_t)TILE2, s1 - t0);
    if (tid < F) hist[tid] = 0;
    __syncthreads();
    longlong2 r[VPT];
    uint32_t g[VPT], rank[VPT];
#pragma unroll
    for (int v = 0; v < VPT; v++) {
      int64_t i = t0 + (int64_t)v * blockDim.x + tid;
      if (i < s1) {
        r[v].x = __builtin_nontemporal_load(&in_pairs[i].x);
        r[v].y = __builtin_nontemporal_load(&in_pairs[i].y);
        g[v] = subF2(r[v].x, F);
        rank[v] = atomicAdd(&hist[g[v]], 1u);
      }
    }
    __syncthreads();
    wave_excl_scan<1024>(hist, base, partials, F);
    __syncthreads();
#pragma unroll
    for (int v = 0; v < VPT; v++) {
      int64_t i = t0 + (int64_t)v * blockDim.x + tid;
      if (i < s1) tbuf[base[g[v]] + rank[v]] = r[v];
    }
    __syncthreads();
    for (int pos = tid; pos < count; pos += blockDim.x) {
      longlong2 row = tbuf[pos];
      uint32_t gg = subF2(row.x, F);
      uint32_t dst = gcur[gg] + (uint32_t)(pos - base[gg]);
      uint32_t limit = (uint32_t)(((int64_t)a * F + gg) * capB + capB);
      if (dst < limit)
        out_pairs[dst] = row;
      else
        s_ovf = 1;
    }
    __syncthreads();
    if (tid < F) gcur[tid] += hist[tid];
    __syncthreads();
  }
  __syncthreads();
  for (int j = tid; j < F; j += blockDim.x) {
    int64_t b = (int64_t)a * F + j;
    uint32_t len = gcur[j] - (uint32_t)(b * capB);
    lens[b] = len > (uint32_t)capB ? (uint32_t)capB : len;
  }
  if (tid == 0 && s_ovf) atomicOr(any_overflow, 2);
}

/* compact pass B with wave scans (drop-in shape of bucket_subpart_slack):
 * count sweep kept, but scans via shfl — isolates the scan-cost question */

/* ------------------------------------------------------------------ main */

struct DBuf {
  void* p{nullptr};
  DBuf() = default;
  explicit DBuf(size_t bytes) { CHECK(hipMalloc(&p, bytes)); }
  ~DBuf()
  {
    if (p) hipFree(p);
  }
  DBuf(const DBuf&) = delete;
  DBuf(DBuf&& o) : p(o.p) { o.p = nullptr; }
  DBuf& operator=(DBuf&& o)
  {
    if (p) hipFree(p);
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
  printf("join_v2: n=%lld rows per table, sel 0.3\n", (long long)n);

  DBuf bk(n * 8), bp(n * 8), pk(n * 8), pp(n * 8);
  dj::generate_build(bk.i64(), bp.i64(), n, rand_max, DJ_DEFAULT_SEED, true, 0, n, 0);
  dj::generate_probe(pk.i64(), pp.i64(), n, rand_max, 0.3, DJ_DEFAULT_SEED, 0, n, 0);
  CHECK(hipDeviceSynchronize());

  const int64_t cap = n / 2;
  DBuf o0(cap * 8), o1(cap * 8), o2(cap * 8), o3(cap * 8);
  DBuf counter(8), anyovf(8), err(8), acc(8);

  /* ---------- reference: product path at its default B ---------- */
  unsigned long long ref_count = 0, ref_sum = 0;

  auto partition_to = [&](int B, DBuf& lpairs, DBuf& loff, DBuf& rpairs, DBuf& roff,
                          float* tA_out) {
    const int PA = dj::bucket_groups_for(B);
    const int64_t capA = dj::slack_capA(n, PA);
    DBuf tmp((size_t)PA * capA * 16);
    DBuf counts((size_t)dj::kBucketBlocks * PA * 4), totals((size_t)PA * 4);
    DBuf segoff((size_t)(PA + 1) * 8);
    lpairs = DBuf((size_t)n * 16);
    rpairs = DBuf((size_t)n * 16);
    loff = DBuf((size_t)(B + 1) * 8);
    roff = DBuf((size_t)(B + 1) * 8);
    CHECK(hipMemset(anyovf.p, 0, 4));
    float t = time_body(3, [&] {
      dj::bucket_partition2(bk.i64(), bp.i64(), n, B, (longlong2*)tmp.p, counts.u32(),
                            totals.u32(), segoff.i64(), loff.i64(), (longlong2*)lpairs.p,
                            (int*)anyovf.p, 0);
    });
    float t2 = time_body(3, [&] {
      dj::bucket_partition2(pk.i64(), pp.i64(), n, B, (longlong2*)tmp.p, counts.u32(),
                            totals.u32(), segoff.i64(), roff.i64(), (longlong2*)rpairs.p,
                            (int*)anyovf.p, 0);
    });
    int ovf;
    CHECK(hipMemcpy(&ovf, anyovf.p, 4, hipMemcpyDeviceToHost));
    printf("  partition B=%d (PA=%d F=%d): left %.3f ms, right %.3f ms (ovf=%d)\n", B, PA,
           B / PA, t, t2, ovf);
    if (tA_out) *tA_out = t + t2;
    return PA;
  };

  {
    const int B = dj::bucket_count_for(n, n);
    DBuf lpairs, loff, rpairs, roff;
    float tpart;
    partition_to(B, lpairs, loff, rpairs, roff, &tpart);
    DBuf flags((size_t)B * 4);
    float t = time_body(3, [&] {
      CHECK(hipMemsetAsync(counter.p, 0, 8, 0));
      CHECK(hipMemsetAsync(anyovf.p, 0, 4, 0));
      CHECK(hipMemsetAsync(err.p, 0, 4, 0));
      dj::lds_join((longlong2*)lpairs.p, loff.i64(), (longlong2*)rpairs.p, roff.i64(), B,
                   2048, o0.i64(), o1.i64(), o2.i64(), o3.i64(), cap, counter.i64(),
                   flags.u32(), (int*)anyovf.p, (int*)err.p, 0);
    });
    CHECK(hipMemcpy(&ref_count, counter.p, 8, hipMemcpyDeviceToHost));
    CHECK(hipMemset(acc.p, 0, 8));
    hipLaunchKernelGGL(checksum_kernel, dim3(2048), dim3(256), 0, 0, o0.i64(), o1.i64(),
                       o2.i64(), o3.i64(), (int64_t)ref_count, (unsigned long long*)acc.p);
    CHECK(hipMemcpy(&ref_sum, acc.p, 8, hipMemcpyDeviceToHost));
    printf("BASE  lds_join B=%d: %.3f ms  count=%llu sum=%016llx\n", B, t, ref_count,
           ref_sum);

    /* J-k4 on same layout */
    for (int K : {2, 4, 8}) {
      float tk = time_body(3, [&] {
        CHECK(hipMemsetAsync(counter.p, 0, 8, 0));
        CHECK(hipMemsetAsync(anyovf.p, 0, 4, 0));
        size_t lds = 2048 * 16 + 4 * 1024 * 8 + 16;
        int grid = (B / K) < 8192 ? (B / K) : 8192;
        if (K == 2)
          hipLaunchKernelGGL((lds_join_k_kernel<2048, 2>), dim3(grid), dim3(1024), lds, 0,
                             (longlong2*)lpairs.p, loff.i64(), (longlong2*)rpairs.p,
                             roff.i64(), B, o0.i64(), o1.i64(), o2.i64(), o3.i64(), cap,
                             (unsigned long long*)counter.p, flags.u32(), (int*)anyovf.p);
        else if (K == 4)
          hipLaunchKernelGGL((lds_join_k_kernel<2048, 4>), dim3(grid), dim3(1024), lds, 0,
                             (longlong2*)lpairs.p, loff.i64(), (longlong2*)rpairs.p,
                             roff.i64(), B, o0.i64(), o1.i64(), o2.i64(), o3.i64(), cap,
                             (unsigned long long*)counter.p, flags.u32(), (int*)anyovf.p);
        else
          hipLaunchKernelGGL((lds_join_k_kernel<2048, 8>), dim3(grid), dim3(1024), lds, 0,
                             (longlong2*)lpairs.p, loff.i64(), (longlong2*)rpairs.p,
                             roff.i64(), B, o0.i64(), o1.i64(), o2.i64(), o3.i64(), cap,
                             (unsigned long long*)counter.p, flags.u32(), (int*)anyovf.p);
        CHECK(hipGetLastError());
      });
      unsigned long long c, s;
      CHECK(hipMemcpy(&c, counter.p, 8, hipMemcpyDeviceToHost));
      CHECK(hipMemset(acc.p, 0, 8));
      hipLaunchKernelGGL(checksum_kernel, dim3(2048), dim3(256), 0, 0, o0.i64(), o1.i64(),
                         o2.i64(), o3.i64(), (int64_t)c, (unsigned long long*)acc.p);
      CHECK(hipMemcpy(&s, acc.p, 8, hipMemcpyDeviceToHost));
      printf("J-k%d  lds_join_k B=%d: %.3f ms  count=%llu sum=%016llx  %s\n", K, B, tk, c, s,
             (c == ref_count && s == ref_sum) ? "OK" : "MISMATCH");
    }
  }

  /* ---------- wave-per-bucket join on compact layouts ---------- */
  for (int B : {524288, 1048576}) {
    DBuf lpairs, loff, rpairs, roff;
    float tpart;
    partition_to(B, lpairs, loff, rpairs, roff, &tpart);
    DBuf flags((size_t)B * 4);
    unsigned long long c, s;
    float t;
    if (B == 524288) {
      t = run_wave_join<8, 512, 1024, false>((longlong2*)lpairs.p, loff.i64(), nullptr, 0,
                                             (longlong2*)rpairs.p, roff.i64(), nullptr, 0, B,
                                             2, o0.i64(), o1.i64(), o2.i64(), o3.i64(), cap,
                                             (unsigned long long*)counter.p, flags.u32(),
                                             (int*)anyovf.p, 3, &c, &s,
                                             (unsigned long long*)acc.p);
      printf("J-wave W8 S512 ST1024 B=%d: %.3f ms  count=%llu sum=%016llx  %s\n", B, t, c, s,
             (c == ref_count && s == ref_sum) ? "OK" : "MISMATCH");
      t = run_wave_join<8, 512, 512, false>((longlong2*)lpairs.p, loff.i64(), nullptr, 0,
                                            (longlong2*)rpairs.p, roff.i64(), nullptr, 0, B,
                                            1, o0.i64(), o1.i64(), o2.i64(), o3.i64(), cap,
                                            (unsigned long long*)counter.p, flags.u32(),
                                            (int*)anyovf.p, 3, &c, &s,
                                            (unsigned long long*)acc.p);
      printf("J-wave W8 S512 ST512  B=%d: %.3f ms  count=%llu sum=%016llx  %s\n", B, t, c, s,
             (c == ref_count && s == ref_sum) ? "OK" : "MISMATCH");
    } else {
      t = run_wave_join<8, 256, 1024, false>((longlong2*)lpairs.p, loff.i64(), nullptr, 0,
                                             (longlong2*)rpairs.p, roff.i64(), nullptr, 0, B,
                                             3, o0.i64(), o1.i64(), o2.i64(), o3.i64(), cap,
                                             (unsigned long long*)counter.p, flags.u32(),
                                             (int*)anyovf.p, 3, &c, &s,
                                             (unsigned long long*)acc.p);
      printf("J-wave W8 S256 ST1024 B=%d: %.3f ms  count=%llu sum=%016llx  %s\n", B, t, c, s,
             (c == ref_count && s == ref_sum) ? "OK" : "MISMATCH");
      t = run_wave_join<8, 256, 512, false>((longlong2*)lpairs.p, loff.i64(), nullptr, 0,
                                            (longlong2*)rpairs.p, roff.i64(), nullptr, 0, B,
                                            2, o0.i64(), o1.i64(), o2.i64(), o3.i64(), cap,
                                            (unsigned long long*)counter.p, flags.u32(),
                                            (int*)anyovf.p, 3, &c, &s,
                                            (unsigned long long*)acc.p);
      printf("J-wave W8 S256 ST512  B=%d: %.3f ms  count=%llu sum=%016llx  %s\n", B, t, c, s,
             (c == ref_count && s == ref_sum) ? "OK" : "MISMATCH");
      t = run_wave_join<16, 256, 2048, false>((longlong2*)lpairs.p, loff.i64(), nullptr, 0,
                                              (longlong2*)rpairs.p, roff.i64(), nullptr, 0,
                                              B, 4, o0.i64(), o1.i64(), o2.i64(), o3.i64(),
                                              cap, (unsigned long long*)counter.p,
                                              flags.u32(), (int*)anyovf.p, 3, &c, &s,
                                              (unsigned long long*)acc.p);
      printf("J-wave W16 S256 ST2048 B=%d: %.3f ms  count=%llu sum=%016llx  %s\n", B, t, c,
             s, (c == ref_count && s == ref_sum) ? "OK" : "MISMATCH");
      t = run_wave_join<12, 256, 1024, false>((longlong2*)lpairs.p, loff.i64(), nullptr, 0,
                                              (longlong2*)rpairs.p, roff.i64(), nullptr, 0,
                                              B, 2, o0.i64(), o1.i64(), o2.i64(), o3.i64(),
                                              cap, (unsigned long long*)counter.p,
                                              flags.u32(), (int*)anyovf.p, 3, &c, &s,
                                              (unsigned long long*)acc.p);
      printf("J-wave W12 S256 ST1024 B=%d: %.3f ms  count=%llu sum=%016llx  %s\n", B, t, c,
             s, (c == ref_count && s == ref_sum) ? "OK" : "MISMATCH");
    }
  }

  /* ---------- pass-A wave-scan + pass-B slack pipeline ---------- */
  for (int B : {524288, 1048576}) {
    const int PA = dj::bucket_groups_for(B);
    const int F = B / PA;
    const int64_t capA = dj::slack_capA(n, PA);
    const int64_t lambda = n / B;
    /* capB ~ lambda + 7*sqrt(lambda), rounded up to 8 */
    int64_t capB = lambda + (int64_t)(7.0 * __builtin_sqrt((double)lambda)) + 8;
    capB = (capB + 7) & ~7ll;
    DBuf tmpA((size_t)PA * capA * 16), cursA((size_t)PA * 4);
    DBuf lpairs((size_t)B * capB * 16), rpairs((size_t)B * capB * 16);
    DBuf llen((size_t)B * 4), rlen((size_t)B * 4);
    DBuf flags((size_t)B * 4);
    size_t ldsA = (size_t)SLACK_TILE2 * 16 + 4 * (size_t)PA * 4 + 64;
    size_t ldsB = (size_t)TILE2 * 16 + 3 * (size_t)F * 4 + 64;

    auto passAB = [&](const int64_t* keys, const int64_t* pay, longlong2* out, uint32_t* lens) {
      CHECK(hipMemsetAsync(cursA.p, 0, (size_t)PA * 4, 0));
      hipLaunchKernelGGL(slackA_ws_kernel, dim3(dj::kBucketBlocks), dim3(1024), ldsA, 0, keys,
                         pay, n, PA, capA, cursA.u32(), (int*)anyovf.p, (longlong2*)tmpA.p);
      CHECK(hipGetLastError());
      hipLaunchKernelGGL(subpart_slack2_kernel, dim3(PA), dim3(1024), ldsB, 0,
                         (longlong2*)tmpA.p, cursA.u32(), capA, F, capB, out, lens,
                         (int*)anyovf.p);
      CHECK(hipGetLastError());
    };
    CHECK(hipMemset(anyovf.p, 0, 4));
    /* time pass A alone (ws) */
    float tA = time_body(3, [&] {
      CHECK(hipMemsetAsync(cursA.p, 0, (size_t)PA * 4, 0));
      hipLaunchKernelGGL(slackA_ws_kernel, dim3(dj::kBucketBlocks), dim3(1024), ldsA, 0,
                         bk.i64(), bp.i64(), n, PA, capA, cursA.u32(), (int*)anyovf.p,
                         (longlong2*)tmpA.p);
      CHECK(hipGetLastError());
    });
    float tB = time_body(3, [&] {
      hipLaunchKernelGGL(subpart_slack2_kernel, dim3(PA), dim3(1024), ldsB, 0,
                         (longlong2*)tmpA.p, cursA.u32(), capA, F, capB, (longlong2*)lpairs.p,
                         llen.u32(), (int*)anyovf.p);
      CHECK(hipGetLastError());
    });
    int ovf;
    CHECK(hipMemcpy(&ovf, anyovf.p, 4, hipMemcpyDeviceToHost));
    printf("A-ws  PA=%d: %.3f ms/table   B-slack F=%d capB=%lld: %.3f ms/table (ovf=%d)\n",
           PA, tA, F, (long long)capB, tB, ovf);
    /* full slack pipeline -> wave join on slack layout */
    CHECK(hipMemset(anyovf.p, 0, 4));
    passAB(bk.i64(), bp.i64(), (longlong2*)lpairs.p, llen.u32());
    passAB(pk.i64(), pp.i64(), (longlong2*)rpairs.p, rlen.u32());
    CHECK(hipDeviceSynchronize());
    CHECK(hipMemcpy(&ovf, anyovf.p, 4, hipMemcpyDeviceToHost));
    if (ovf) printf("  slack pipeline overflow=%d (results invalid)\n", ovf);
    unsigned long long c, s;
    float t;
    if (B == 524288) {
      t = run_wave_join<8, 512, 1024, true>((longlong2*)lpairs.p, nullptr, llen.u32(), capB,
                                            (longlong2*)rpairs.p, nullptr, rlen.u32(), capB,
                                            B, 2, o0.i64(), o1.i64(), o2.i64(), o3.i64(),
                                            cap, (unsigned long long*)counter.p, flags.u32(),
                                            (int*)anyovf.p, 3, &c, &s,
                                            (unsigned long long*)acc.p);
      printf("J-wave-slack W8 S512 B=%d: %.3f ms  count=%llu sum=%016llx  %s\n", B, t, c, s,
             (c == ref_count && s == ref_sum) ? "OK" : "MISMATCH");
    } else {
      t = run_wave_join<8, 256, 1024, true>((longlong2*)lpairs.p, nullptr, llen.u32(), capB,
                                            (longlong2*)rpairs.p, nullptr, rlen.u32(), capB,
                                            B, 3, o0.i64(), o1.i64(), o2.i64(), o3.i64(),
                                            cap, (unsigned long long*)counter.p, flags.u32(),
                                            (int*)anyovf.p, 3, &c, &s,
                                            (unsigned long long*)acc.p);
      printf("J-wave-slack W8 S256 B=%d: %.3f ms  count=%llu sum=%016llx  %s\n", B, t, c, s,
             (c == ref_count && s == ref_sum) ? "OK" : "MISMATCH");
    }
  }

  printf("done\n");
  return 0;
}
