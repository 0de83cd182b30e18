/*
 * ablate_join.hip — phase ablation of the bucket pipeline's two hot kernels
 * on bench-shaped data (100M rows, B=65536), to attribute their time:
 *   subpart: count sweep | offsets | staged scatter (load+rank / stage / flush)
 *   lds_join: table init | build | probe-count | probe-write | out writes
 * Variants toggle phases via template flags; wall time per kernel printed.
 * Diagnostic only (not linked into the product library).
 */
#include "../distributed_join_amd/csrc/dj_rng.h"

#include <hip/hip_runtime.h>

#include <cstdint>
#include <cstdio>

#define CHECK(c)                                                      \
  do {                                                                \
    hipError_t e = (c);                                               \
    if (e != hipSuccess) {                                            \
      printf("HIP error %s at %d\n", hipGetErrorString(e), __LINE__); \
      return 1;                                                       \
    }                                                                 \
  } while (0)

constexpr int THREADS = 1024;
constexpr int SUB = 256;
constexpr int TILE = 4096;

__device__ __forceinline__ uint32_t subB_of(int64_t k)
{
  return (uint32_t)(dj_mix64((uint64_t)k) >> 32) & 255u;
}

/* subpart ablation: COUNT (seghist sweep), RANK (load+hist atomic), STAGE
 * (tbuf writes), FLUSH (global writes) */
template <bool COUNT, bool RANK, bool STAGE, bool FLUSH>
__global__ __launch_bounds__(THREADS) void subpart_ablate(
  const longlong2* __restrict__ in, const int64_t* __restrict__ segoff,
  longlong2* __restrict__ out, int* sink)
{
  extern __shared__ __attribute__((aligned(16))) char smem[];
  longlong2* tbuf = (longlong2*)smem;
  uint32_t* hist = (uint32_t*)(tbuf + TILE);
  uint32_t* base = hist + SUB;
  uint32_t* gcur = base + SUB;
  uint32_t* seghist = gcur + SUB;
  const int tid = threadIdx.x;
  const int64_t s0 = segoff[blockIdx.x], s1 = segoff[blockIdx.x + 1];
  if (tid < SUB) seghist[tid] = 0;
  __syncthreads();
  if (COUNT) {
    for (int64_t i = s0 + tid; i < s1; i += blockDim.x)
      atomicAdd(&seghist[subB_of(__builtin_nontemporal_load(&in[i].x))], 1u);
  }
  __syncthreads();
  if (tid == 0) {
    uint32_t acc = 0;
    for (int j = 0; j < SUB; j++) {
      gcur[j] = (uint32_t)s0 + acc;
      acc += seghist[j];
    }
  }
  __syncthreads();
  if (!RANK) {
    if (tid == 0 && s1 > s0) *sink = (int)seghist[0];
    return;
  }
  constexpr int VPT = TILE / THREADS;
  for (int64_t t0 = s0; t0 < s1; t0 += TILE) {
    const int count = (int)min((int64_t)TILE, s1 - t0);
    if (tid < SUB) hist[tid] = 0;
    __syncthreads();
    longlong2 r[VPT];
    uint32_t g[VPT], rank[VPT];
    int nv = 0;
    for (int v = 0; v < VPT; v++) {
      int64_t i = t0 + (int64_t)v * blockDim.x + tid;
      if (i < s1) {
        r[v].x = __builtin_nontemporal_load(&in[i].x);
        r[v].y = __builtin_nontemporal_load(&in[i].y);
        g[v] = subB_of(r[v].x);
        rank[v] = atomicAdd(&hist[g[v]], 1u);
        nv = v + 1;
      }
    }
    __syncthreads();
    if (tid < SUB) base[tid] = hist[tid];
    __syncthreads();
    for (int off = 1; off < SUB; off <<= 1) {
      uint32_t add = (tid < SUB && tid >= off) ? base[tid - off] : 0;
      __syncthreads();
      if (tid < SUB) base[tid] += add;
      __syncthreads();
    }
    if (tid < SUB) base[tid] -= hist[tid];
    __syncthreads();
    if (STAGE) {
      for (int v = 0; v < nv; v++) tbuf[base[g[v]] + rank[v]] = r[v];
    }
    __syncthreads();
    if (FLUSH) {
      for (int pos = tid; pos < count; pos += blockDim.x) {
        longlong2 row = tbuf[pos];
        uint32_t gg = subB_of(row.x);
        out[gcur[gg] + (pos - base[gg])] = row;
      }
    } else if (STAGE && tid == 0) {
      if (tbuf[count - 1].x == 42424242) *sink = 1;
    }
    __syncthreads();
    if (tid < SUB) gcur[tid] += hist[tid];
    __syncthreads();
  }
}

/* lds_join ablation: INIT, BUILD, PROBE1 (count walk), PROBE2+WRITE */
template <bool INIT, bool BUILD, bool PROBE1, bool PROBE2>
__global__ __launch_bounds__(THREADS) void join_ablate(
  const longlong2* __restrict__ lrows, const int64_t* __restrict__ loff,
  const longlong2* __restrict__ rrows, const int64_t* __restrict__ roff, int B,
  int64_t* __restrict__ out0, int64_t* __restrict__ out1, int64_t* __restrict__ out2,
  int64_t* __restrict__ out3, int64_t cap, unsigned long long* counter)
{
  extern __shared__ __attribute__((aligned(16))) char smem[];
  longlong2* tbl = (longlong2*)smem;
  long long* base_sh = (long long*)(smem + 4096 * sizeof(longlong2));
  uint32_t* total_sh = (uint32_t*)(base_sh + 1);
  uint32_t* cur_sh = total_sh + 1;
  const uint32_t smask = 4095;
  for (int b = blockIdx.x; b < B; b += gridDim.x) {
    const int64_t l0 = loff[b], l1 = loff[b + 1];
    const int64_t r0 = roff[b], r1 = roff[b + 1];
    if (l1 - l0 == 0 || r1 == r0 || l1 - l0 > 3072) continue;
    if (INIT) {
      for (int s = threadIdx.x; s < 4096; s += blockDim.x) tbl[s].x = -1;
    }
    if (threadIdx.x == 0) {
      *total_sh = 0;
      *cur_sh = 0;
    }
    __syncthreads();
    if (BUILD) {
      for (int64_t i = l0 + threadIdx.x; i < l1; i += blockDim.x) {
        longlong2 row = lrows[i];
        uint32_t slot = (uint32_t)dj_mix64((uint64_t)row.x) & smask;
        for (;;) {
          unsigned long long old =
            atomicCAS((unsigned long long*)&tbl[slot].x, (unsigned long long)(-1ll),
                      (unsigned long long)row.x);
          if (old == (unsigned long long)(-1ll)) break;
          slot = (slot + 1) & smask;
        }
        tbl[slot].y = row.y;
      }
    }
    __syncthreads();
    uint32_t my = 0;
    if (PROBE1) {
      for (int64_t j = r0 + threadIdx.x; j < r1; j += blockDim.x) {
        int64_t key = rrows[j].x;
        uint32_t slot = (uint32_t)dj_mix64((uint64_t)key) & smask;
        for (;;) {
          longlong2 e = tbl[slot];
          if (e.x == -1) break;
          if (e.x == key) my++;
          slot = (slot + 1) & smask;
        }
      }
    }
    if (my) atomicAdd(total_sh, my);
    __syncthreads();
    if (threadIdx.x == 0 && *total_sh)
      *base_sh = (long long)atomicAdd(counter, (unsigned long long)*total_sh);
    __syncthreads();
    if (PROBE2 && *total_sh) {
      const long long base = *base_sh;
      uint32_t w = my ? atomicAdd(cur_sh, my) : 0;
      if (my) {
        for (int64_t j = r0 + threadIdx.x; j < r1; j += blockDim.x) {
          longlong2 prow = rrows[j];
          uint32_t slot = (uint32_t)dj_mix64((uint64_t)prow.x) & smask;
          for (;;) {
            longlong2 e = tbl[slot];
            if (e.x == -1) break;
            if (e.x == prow.x) {
              long long idx = base + (long long)w;
              if (idx < cap) {
                out0[idx] = prow.x;
                out1[idx] = e.y;
                out2[idx] = prow.x;
                out3[idx] = prow.y;
              }
              w++;
            }
            slot = (slot + 1) & smask;
          }
        }
      }
    }
    __syncthreads();
  }
}

__global__ void gen_pairs(longlong2* p, int64_t n, uint64_t stream)
{
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    p[i].x = (int64_t)(dj_hash64(1234, stream, (uint64_t)i) % (200000001ull));
    p[i].y = i;
  }
}

template <typename F>
static double bench_ms(F launch, int iters)
{
  hipEvent_t a, b;
  hipEventCreate(&a);
  hipEventCreate(&b);
  launch();
  hipDeviceSynchronize();
  hipEventRecord(a);
  for (int i = 0; i < iters; i++) launch();
  hipEventRecord(b);
  hipEventSynchronize(b);
  float ms;
  hipEventElapsedTime(&ms, a, b);
  hipEventDestroy(a);
  hipEventDestroy(b);
  return ms / iters;
}

int main()
{
  setbuf(stdout, NULL);
  printf("start\n");
  const int64_t n = 100'000'000;
  const int PA = 256, B = 65536;
  longlong2 *in, *out;
  int64_t* segoff;
  int* sink;
  CHECK(hipMalloc(&in, n * 16));
  CHECK(hipMalloc(&out, n * 16));
  CHECK(hipMalloc(&segoff, (PA + 1) * 8));
  CHECK(hipMalloc(&sink, 4));
  printf("allocs done\n");
  hipLaunchKernelGGL(gen_pairs, dim3(2048), dim3(256), 0, 0, in, n, 7);
  {
    /* equal segments for the ablation (uniform hash -> near-equal anyway) */
    int64_t* h = new int64_t[PA + 1];
    for (int i = 0; i <= PA; i++) h[i] = n * i / PA;
    CHECK(hipMemcpy(segoff, h, (PA + 1) * 8, hipMemcpyHostToDevice));
  }
  printf("segoff done\n");
  size_t lds = TILE * 16 + 4 * SUB * 4;
  double t;
#define RUN(c, r, s, f, name)                                                            \
  t = bench_ms(                                                                          \
    [&] {                                                                                \
      hipLaunchKernelGGL((subpart_ablate<c, r, s, f>), dim3(PA), dim3(THREADS), lds, 0,  \
                         in, segoff, out, sink);                                         \
    },                                                                                   \
    5);                                                                                  \
  printf("subpart %-22s %.3f ms\n", name, t);
  RUN(true, false, false, false, "count-only")
  RUN(true, true, false, false, "count+rank")
  RUN(true, true, true, false, "count+rank+stage")
  RUN(true, true, true, true, "full")
#undef RUN

  /* join ablation over bucketed data: reuse out as "bucketed" input (the
   * real pipeline's distribution); offsets B+1 from uniform split */
  printf("subpart done\n");
  int64_t* boff;
  printf("A\n");
  CHECK(hipMalloc(&boff, (B + 1) * 8));
  printf("B\n");
  {
    int64_t* h = new int64_t[B + 1];
    for (int i = 0; i <= B; i++) h[i] = n * (int64_t)i / B;
    CHECK(hipMemcpy(boff, h, (B + 1) * 8, hipMemcpyHostToDevice));
  }
  printf("C\n");
  int64_t *o0, *o1, *o2, *o3;
  unsigned long long* ctr;
  int64_t cap = n + (n >> 3);
  CHECK(hipMalloc(&o0, cap * 8));
  CHECK(hipMalloc(&o1, cap * 8));
  CHECK(hipMalloc(&o2, cap * 8));
  CHECK(hipMalloc(&o3, cap * 8));
  CHECK(hipMalloc(&ctr, 8));
  printf("D\n");
  size_t jlds = 4096 * 16 + 16;
#define JRUN(i, bl, p1, p2, name)                                                      \
  CHECK(hipMemset(ctr, 0, 8));                                                         \
  t = bench_ms(                                                                        \
    [&] {                                                                              \
      (void)hipMemsetAsync(ctr, 0, 8);                                                 \
      hipLaunchKernelGGL((join_ablate<i, bl, p1, p2>), dim3(8192), dim3(THREADS),      \
                         jlds, 0, out, boff, out, boff, B, o0, o1, o2, o3, cap, ctr);  \
    },                                                                                 \
    5);                                                                                \
  printf("join %-26s %.3f ms\n", name, t);
  JRUN(true, false, false, false, "init-only")
  JRUN(true, true, false, false, "init+build")
  JRUN(true, true, true, false, "init+build+probecount")
  JRUN(true, true, true, true, "full (self-join)")
#undef JRUN
  return 0;
}
