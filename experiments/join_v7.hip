/*
 * join_v7.hip — round-2 sweep #4: squeeze pass B and the join toward the
 * streaming ceiling, and measure that ceiling (read+write copy, not just
 * stream_read) so the partition kernels have an honest target.
 *  - copy16 nt-load/plain-store and nt/nt: the r+w ceiling for 16 B rows.
 *  - pass-B slack variants: glim precompute (kill the per-row 64-bit
 *    multiply in the flush bound check), BTILE 8192 (VPT 8, 1 blk/CU).
 *  - join KBUK sweep (4 = adopted, 8, 16) and nontemporal bucket reads.
 * Build: hipcc --offload-arch=gfx950 -O3 join_v7.hip \
 *          ../distributed_join_amd/csrc/dj_kernels.hip -o join_v7
 * Diagnostic only.
 */
#include "../distributed_join_amd/csrc/dj_kernels.hpp"
#include "../distributed_join_amd/csrc/dj_rng.h"

#include <hip/hip_runtime.h>

#include <cstdint>
#include <cstdio>
#include <cstdlib>
#include <functional>

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

/* ------------------------- r+w streaming ceiling ------------------------- */
template <bool NTSTORE>
__global__ __launch_bounds__(1024) void copy16_kernel(const longlong2* __restrict__ in,
                                                      longlong2* __restrict__ out, int64_t n)
{
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    longlong2 v;
    v.x = __builtin_nontemporal_load(&in[i].x);
    v.y = __builtin_nontemporal_load(&in[i].y);
    if (NTSTORE) {
      __builtin_nontemporal_store(v.x, &out[i].x);
      __builtin_nontemporal_store(v.y, &out[i].y);
    } else {
      out[i] = v;
    }
  }
}

/* ---------------- pass-B slack variants (see dj_kernels.hip) ------------- */
__device__ __forceinline__ uint32_t subF2(int64_t key, int F)
{
  uint64_t m = dj_mix64((uint64_t)key);
  if (F <= 256) return (uint32_t)(m >> 32) & (uint32_t)(F - 1);
  uint32_t lo = (uint32_t)(m >> 32) & 255u;
  uint32_t hi = (uint32_t)(m >> 50) & (uint32_t)((F >> 8) - 1);
  return lo | (hi << 8);
}

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

template <int TILE, bool GLIM>
__global__ __launch_bounds__(1024) void passB_kernel(const longlong2* __restrict__ in_pairs,
                                                     const uint32_t* __restrict__ seg_len,
                                                     int64_t capA, int F, int64_t capB,
                                                     longlong2* __restrict__ out_pairs,
                                                     uint32_t* __restrict__ lens,
                                                     int* __restrict__ any_overflow)
{
  constexpr int VPT = TILE / 1024;
  extern __shared__ __attribute__((aligned(16))) char smem[];
  longlong2* tbuf = (longlong2*)smem;
  uint32_t* hist = (uint32_t*)(tbuf + TILE);
  uint32_t* base = hist + F;
  uint32_t* gcur = base + F;
  uint32_t* glim = gcur + F;
  uint32_t* partials = glim + F;
  __shared__ int s_ovf;
  const int tid = threadIdx.x;
  const int a = blockIdx.x;
  const int64_t s0 = (int64_t)a * capA;
  const int64_t s1 = s0 + seg_len[a];
  if (tid == 0) s_ovf = 0;
  for (int j = tid; j < F; j += blockDim.x) {
    int64_t b0 = ((int64_t)a * F + j) * capB;
    gcur[j] = (uint32_t)b0;
    if (GLIM) glim[j] = (uint32_t)(b0 + capB);
  }
  if (tid < F) hist[tid] = 0;
  __syncthreads();
  if (s0 >= s1) {
    for (int j = tid; j < F; j += blockDim.x) lens[(size_t)a * F + j] = 0;
    return;
  }
  longlong2 r[VPT];
  uint32_t g[VPT], rank[VPT];
#pragma unroll
  for (int v = 0; v < VPT; v++) {
    int64_t i = s0 + (int64_t)v * 1024 + tid;
    if (i < s1) {
      r[v].x = __builtin_nontemporal_load(&in_pairs[i].x);
      r[v].y = __builtin_nontemporal_load(&in_pairs[i].y);
      g[v] = subF2(r[v].x, F);
      rank[v] = atomicAdd(&hist[g[v]], 1u);
    }
  }
  __syncthreads();
  for (int64_t t0 = s0; t0 < s1; t0 += TILE) {
    const int count = (int)min((int64_t)TILE, s1 - t0);
    const int64_t t1 = t0 + TILE;
    wave_excl_scan(hist, base, partials, F);
    __syncthreads();
#pragma unroll
    for (int v = 0; v < VPT; v++) {
      int64_t i = t0 + (int64_t)v * 1024 + tid;
      if (i < s1) tbuf[base[g[v]] + rank[v]] = r[v];
    }
    __syncthreads();
    longlong2 r2[VPT];
#pragma unroll
    for (int v = 0; v < VPT; v++) {
      int64_t i = t1 + (int64_t)v * 1024 + tid;
      if (i < s1) {
        r2[v].x = __builtin_nontemporal_load(&in_pairs[i].x);
        r2[v].y = __builtin_nontemporal_load(&in_pairs[i].y);
      }
    }
    for (int pos = tid; pos < count; pos += blockDim.x) {
      longlong2 row = tbuf[pos];
      uint32_t gg = subF2(row.x, F);
      uint32_t dst = gcur[gg] + (uint32_t)(pos - base[gg]);
      uint32_t limit = GLIM ? glim[gg] : (uint32_t)(((int64_t)a * F + gg) * capB + capB);
      if (dst < limit)
        out_pairs[dst] = row;
      else
        s_ovf = 1;
    }
    __syncthreads();
    if (tid < F) gcur[tid] += hist[tid];
    if (tid < F) hist[tid] = 0;
    __syncthreads();
#pragma unroll
    for (int v = 0; v < VPT; v++) {
      int64_t i = t1 + (int64_t)v * 1024 + tid;
      if (i < s1) {
        r[v] = r2[v];
        g[v] = subF2(r[v].x, F);
        rank[v] = atomicAdd(&hist[g[v]], 1u);
      }
    }
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

__device__ __noinline__ void epilogue_flush(int64_t* stage, long long* base_sh,
                                            uint32_t* cur_sh, int64_t* out0, int64_t* out1,
                                            int64_t* out2, int64_t* out3, int64_t cap,
                                            unsigned long long* counter)
{
  constexpr int S = 1024;
  __syncthreads();
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
}

/* ------------- join KBUK sweep + optional nontemporal reads --------------
 * The signature now carries the product kernel's unused compact-layout
 * pointer args (kernarg-layout parity); EPI adds its leftover-stage epilogue
 * flush — bisecting which source difference reproduces the 2.37 vs 1.41 ms
 * product/clone gap. */
/* EPI: 0 = none (leaky), 1 = product epilogue, 2 = epilogue without the
 * leading barrier (diagnostic), 3 = spill-style epilogue (per-thread global
 * atomic, no reserve), 4 = noinline-function epilogue,
 * 5 = kflush exact (scan + goto + k!=kflush — current product structure),
 * 6 = scan only (kflush computed but flush cond stays k<KBUK-1),
 * 7 = k!=kflush with kflush=KBUK-1 (no scan, no goto),
 * 8 = goto structure with constant flush slot (no scan) */
template <int SLOTS2, int KBUK, bool NT, int EPI = 0>
__global__ __launch_bounds__(1024) void join_kn_kernel(
  const longlong2* __restrict__ lrows, const int64_t* __restrict__ loff,
  const uint32_t* __restrict__ llen, int64_t capL, const longlong2* __restrict__ rrows,
  const int64_t* __restrict__ roff, const uint32_t* __restrict__ rlen, int64_t capR,
  int B, int64_t* __restrict__ out0, int64_t* __restrict__ out1, int64_t* __restrict__ out2,
  int64_t* __restrict__ out3, int64_t cap, unsigned long long* counter,
  uint32_t* __restrict__ overflow_flags, int* __restrict__ any_overflow,
  int* __restrict__ error)
{
  constexpr int S = 1024;
  constexpr int WATER = S - S / 4;
  extern __shared__ __attribute__((aligned(16))) char smem[];
  longlong2* tbl = (longlong2*)smem;
  int64_t* stage = (int64_t*)(smem + (size_t)SLOTS2 * 16);
  long long* base_sh = (long long*)(stage + 4 * S);
  uint32_t* cur_sh = (uint32_t*)(base_sh + 1);
  const uint32_t smask = SLOTS2 - 1;
  if (threadIdx.x == 0) *cur_sh = 0;
  __syncthreads();
  for (int bb = blockIdx.x * KBUK; bb < B; bb += gridDim.x * KBUK) {
    int kflush = KBUK - 1;
    if (EPI == 5 || EPI == 6) { /* scan for the last valid bucket */
      const int kend = (B - bb < KBUK) ? (B - bb) : KBUK;
      kflush = 0;
      for (int k = kend - 1; k > 0; k--) {
        const int64_t lnb = llen[bb + k], rnb = rlen[bb + k];
        if (lnb > 0 && rnb > 0 && lnb <= SLOTS2 * 3 / 4) {
          kflush = k;
          break;
        }
      }
      if (EPI == 6) kflush = KBUK - 1; /* scan result discarded */
    }
    for (int k = 0; k < KBUK; k++) {
      const int b = bb + k;
      if (b >= B) break;
      const int64_t l0 = (int64_t)b * capL;
      const int64_t l1 = l0 + llen[b];
      const int64_t r0 = (int64_t)b * capR;
      const int64_t r1 = r0 + rlen[b];
      const int64_t lnb = l1 - l0;
      if (lnb == 0 || r1 == r0) {
        if (EPI == 5 || EPI == 7 || EPI == 8) {
          if (k != kflush) continue;
          goto flush;
        }
        continue;
      }
      if (lnb > SLOTS2 * 3 / 4) {
        if (threadIdx.x == 0) {
          overflow_flags[b] = 1;
          atomicOr(any_overflow, 1);
        }
        if (EPI == 5 || EPI == 7 || EPI == 8) {
          if (k != kflush) continue;
          goto flush;
        }
        continue;
      }
      for (int s = threadIdx.x; s < SLOTS2; s += blockDim.x) tbl[s].x = EMPTY;
      __syncthreads();
      for (int64_t i = l0 + threadIdx.x; i < l1; i += blockDim.x) {
        longlong2 row;
        if (NT) {
          row.x = __builtin_nontemporal_load(&lrows[i].x);
          row.y = __builtin_nontemporal_load(&lrows[i].y);
        } else
          row = lrows[i];
        if (row.x == EMPTY) {
          *error = 1;
          continue;
        }
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
        longlong2 prow;
        if (NT) {
          prow.x = __builtin_nontemporal_load(&rrows[j].x);
          prow.y = __builtin_nontemporal_load(&rrows[j].y);
        } else
          prow = rrows[j];
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
    flush:  // label used by EPI 5/7/8 empty-bucket paths (mirrors product)
      __syncthreads();
      if ((EPI == 5 ? (k != kflush) : (k < KBUK - 1)) && *cur_sh < (uint32_t)WATER)
        continue;
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
  if (EPI == 1 || EPI == 2) {
    if (EPI == 1) __syncthreads();
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
  } else if (EPI == 3) {
    __syncthreads();
    const uint32_t total = min(*cur_sh, (uint32_t)S);
    for (uint32_t i = threadIdx.x; i < total; i += blockDim.x) {
      long long idx = (long long)atomicAdd(counter, 1ull);
      if (idx < cap) {
        out0[idx] = stage[0 * S + i];
        out1[idx] = stage[1 * S + i];
        out2[idx] = stage[2 * S + i];
        out3[idx] = stage[3 * S + i];
      }
    }
  } else if (EPI == 4) {
    epilogue_flush(stage, base_sh, cur_sh, out0, out1, out2, out3, cap, counter);
  }
}

struct DBuf {
  void* p{nullptr};
  explicit DBuf(size_t bytes) { CHECK(hipMalloc(&p, bytes)); }
  ~DBuf()
  {
    if (p) (void)hipFree(p);
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
  printf("join_v7: n=%lld rows/table, sel 0.3\n", (long long)n);

  DBuf bk(n * 8), bp(n * 8), pk(n * 8), pp(n * 8);
  dj::generate_build(bk.i64(), bp.i64(), n, rand_max, DJ_DEFAULT_SEED, true, 0, n, 0);
  dj::generate_probe(pk.i64(), pp.i64(), n, rand_max, 0.3, DJ_DEFAULT_SEED, 0, n, 0);
  CHECK(hipDeviceSynchronize());

  const int B = dj::bucket_count_for(n, n);
  const int PA = dj::bucket_groups_for(B);
  const int F = B / PA;
  const int64_t capA = dj::slack_capA(n, PA);
  const int64_t capB = dj::slack_capB(n, B);
  printf("B=%d PA=%d F=%d capA=%lld capB=%lld\n", B, PA, F, (long long)capA, (long long)capB);

  const int64_t cap = n / 2;
  DBuf o0(cap * 8), o1(cap * 8), o2(cap * 8), o3(cap * 8);
  DBuf counter(8), anyovf(8), err(8), acc(8);
  DBuf lpairs((size_t)B * capB * 16), rpairs((size_t)B * capB * 16);
  DBuf tmpL((size_t)PA * capA * 16), tmpR((size_t)PA * capA * 16);
  DBuf curL((size_t)PA * 4), curR((size_t)PA * 4);
  DBuf llen((size_t)B * 4), rlen((size_t)B * 4), flags((size_t)B * 4);

  /* --------------------- streaming r+w ceiling ---------------------- */
  {
    float t = time_body(5, [&] {
      hipLaunchKernelGGL(copy16_kernel<false>, dim3(4096), dim3(1024), 0, 0,
                         (longlong2*)tmpL.p, (longlong2*)tmpR.p, n);
    });
    printf("copy16 ntload/plainstore : %.3f ms  %.1f GB/s\n", t, 32.0 * n / t / 1e6);
    t = time_body(5, [&] {
      hipLaunchKernelGGL(copy16_kernel<true>, dim3(4096), dim3(1024), 0, 0,
                         (longlong2*)tmpL.p, (longlong2*)tmpR.p, n);
    });
    printf("copy16 ntload/ntstore    : %.3f ms  %.1f GB/s\n", t, 32.0 * n / t / 1e6);
  }

  /* run the product slack partition once: tmp/cursors hold pass-A output */
  CHECK(hipMemset(anyovf.p, 0, 4));
  dj::bucket_partition2_slack(bk.i64(), bp.i64(), n, B, (longlong2*)tmpL.p, curL.u32(), capB,
                              (longlong2*)lpairs.p, llen.u32(), (int*)anyovf.p, 0);
  dj::bucket_partition2_slack(pk.i64(), pp.i64(), n, B, (longlong2*)tmpR.p, curR.u32(), capB,
                              (longlong2*)rpairs.p, rlen.u32(), (int*)anyovf.p, 0);
  CHECK(hipDeviceSynchronize());
  int hovf = 0;
  CHECK(hipMemcpy(&hovf, anyovf.p, 4, hipMemcpyDeviceToHost));
  printf("partition ovf=%d\n", hovf);

  /* --------------------- pass-B variants (left table) ---------------------- */
  auto runB = [&](const char* nm, const std::function<void()>& fn) {
    float t = time_body(3, [&] {
      CHECK(hipMemsetAsync(anyovf.p, 0, 4, 0));
      fn();
    });
    printf("%-36s %.3f ms/table\n", nm, t);
  };
  size_t ldsB4 = (size_t)4096 * 16 + 5 * (size_t)F * 4 + 64;
  size_t ldsB8 = (size_t)8192 * 16 + 5 * (size_t)F * 4 + 64;
  runB("passB T4096 glim=0 (adopted)", [&] {
    hipLaunchKernelGGL((passB_kernel<4096, false>), dim3(PA), dim3(1024), ldsB4, 0,
                       (longlong2*)tmpL.p, curL.u32(), capA, F, capB, (longlong2*)lpairs.p,
                       llen.u32(), (int*)anyovf.p);
    CHECK(hipGetLastError());
  });
  runB("passB T4096 glim=1", [&] {
    hipLaunchKernelGGL((passB_kernel<4096, true>), dim3(PA), dim3(1024), ldsB4, 0,
                       (longlong2*)tmpL.p, curL.u32(), capA, F, capB, (longlong2*)lpairs.p,
                       llen.u32(), (int*)anyovf.p);
    CHECK(hipGetLastError());
  });
  runB("passB T8192 glim=1 (1blk/CU)", [&] {
    hipLaunchKernelGGL((passB_kernel<8192, true>), dim3(PA), dim3(1024), ldsB8, 0,
                       (longlong2*)tmpL.p, curL.u32(), capA, F, capB, (longlong2*)lpairs.p,
                       llen.u32(), (int*)anyovf.p);
    CHECK(hipGetLastError());
  });
  /* restore canonical bucket contents for the join sweep (both tables) */
  CHECK(hipMemset(anyovf.p, 0, 4));
  hipLaunchKernelGGL((passB_kernel<4096, true>), dim3(PA), dim3(1024), ldsB4, 0,
                     (longlong2*)tmpL.p, curL.u32(), capA, F, capB, (longlong2*)lpairs.p,
                     llen.u32(), (int*)anyovf.p);
  hipLaunchKernelGGL((passB_kernel<4096, true>), dim3(PA), dim3(1024), ldsB4, 0,
                     (longlong2*)tmpR.p, curR.u32(), capA, F, capB, (longlong2*)rpairs.p,
                     rlen.u32(), (int*)anyovf.p);
  CHECK(hipDeviceSynchronize());

  /* --------------------------- join sweep ---------------------------- */
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
    printf("%-36s %.3f ms  count=%llu  %s\n", name, ms, c,
           (c == ref_count && s == ref_sum) ? "OK" : "MISMATCH");
  };
  auto reset_join = [&] {
    CHECK(hipMemsetAsync(counter.p, 0, 8, 0));
    CHECK(hipMemsetAsync(err.p, 0, 4, 0));
    CHECK(hipMemsetAsync(flags.p, 0, (size_t)B * 4, 0));
  };
  verify("product lds_join_slack (fixed)", time_body(3, [&] {
           reset_join();
           dj::lds_join_slack((longlong2*)lpairs.p, llen.u32(), capB, (longlong2*)rpairs.p,
                              rlen.u32(), capB, B, 2048, o0.i64(), o1.i64(), o2.i64(),
                              o3.i64(), cap, counter.i64(), flags.u32(), (int*)anyovf.p,
                              (int*)err.p, 0);
         }));
  auto runJ = [&](auto ktag, auto nttag, auto epitag, const char* nm) {
    constexpr int KB = decltype(ktag)::value;
    constexpr bool NT = decltype(nttag)::value;
    constexpr int EPI = decltype(epitag)::value;
    verify(nm, time_body(3, [&] {
             reset_join();
             size_t lds = (size_t)2048 * 16 + 4 * 1024 * 8 + 16;
             int64_t groups = ((int64_t)B + KB - 1) / KB;
             int grid = (int)(groups < 8192 ? groups : 8192);
             hipLaunchKernelGGL((join_kn_kernel<2048, KB, NT, EPI>), dim3(grid), dim3(1024),
                                lds, 0, (longlong2*)lpairs.p, (const int64_t*)nullptr,
                                llen.u32(), capB, (longlong2*)rpairs.p,
                                (const int64_t*)nullptr, rlen.u32(), capB, B, o0.i64(),
                                o1.i64(), o2.i64(), o3.i64(), cap,
                                (unsigned long long*)counter.p, flags.u32(), (int*)anyovf.p,
                                (int*)err.p);
             CHECK(hipGetLastError());
           }));
  };
  auto F_ = std::integral_constant<bool, false>{};
  auto K4 = std::integral_constant<int, 4>{};
  runJ(K4, F_, std::integral_constant<int, 0>{}, "J K4 baseline (fast/leaky)");
  runJ(K4, F_, std::integral_constant<int, 5>{}, "J K4 kflush exact (product)");
  runJ(K4, F_, std::integral_constant<int, 6>{}, "J K4 scan only");
  runJ(K4, F_, std::integral_constant<int, 7>{}, "J K4 goto only");
  printf("done\n");
  return 0;
}
