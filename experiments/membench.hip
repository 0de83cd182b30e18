/*
 * membench.hip — microbenchmark establishing MI355X ceilings relevant to the
 * join hot path: streaming read, random 16 B gather from a large table,
 * global atomicCAS scatter, and single-counter atomicAdd. Results inform
 * DESIGN.md's per-kernel roofline choices. Not part of the product library.
 *
 * Build: hipcc --offload-arch=gfx950 -O3 membench.hip -o membench
 */
#include <hip/hip_runtime.h>

#include <cstdint>
#include <cstdio>

#define CHECK(c)                                                     \
  do {                                                               \
    hipError_t e = (c);                                              \
    if (e != hipSuccess) {                                           \
      printf("HIP error %s at %d\n", hipGetErrorString(e), __LINE__); \
      return 1;                                                      \
    }                                                                \
  } while (0)

__device__ __forceinline__ uint64_t mix64(uint64_t x)
{
  x += 0x9E3779B97F4A7C15ULL;
  x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ULL;
  x = (x ^ (x >> 27)) * 0x94D049BB133111EBULL;
  return x ^ (x >> 31);
}

__global__ void stream_read(const int64_t* __restrict__ a, int64_t n, int64_t* out)
{
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  int64_t acc = 0;
  for (; i < n; i += stride) acc += a[i];
  if (acc == 42) *out = acc;
}

__global__ void random_gather16(const longlong2* __restrict__ tbl, uint64_t mask, int64_t nops,
                                int64_t* out)
{
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  int64_t acc = 0;
  for (; i < nops; i += stride) {
    longlong2 v = tbl[mix64((uint64_t)i) & mask];
    acc += v.x + v.y;
  }
  if (acc == 42) *out = acc;
}

/* two dependent random reads per op (like a probe walk) */
__global__ void random_gather16_dep2(const longlong2* __restrict__ tbl, uint64_t mask,
                                     int64_t nops, int64_t* out)
{
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  int64_t acc = 0;
  for (; i < nops; i += stride) {
    longlong2 v = tbl[mix64((uint64_t)i) & mask];
    acc += v.x;
    longlong2 w = tbl[(mix64((uint64_t)i) + 1 + (uint64_t)(v.x & 1)) & mask];
    acc += w.y;
  }
  if (acc == 42) *out = acc;
}

__global__ void atomic_cas_scatter(int64_t* tbl, uint64_t mask, int64_t nops)
{
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < nops; i += stride) {
    uint64_t slot = mix64((uint64_t)i) & mask;
    atomicCAS((unsigned long long*)&tbl[slot * 2], ~0ull, (unsigned long long)i);
  }
}

__global__ void atomic_add_scatter(int64_t* tbl, uint64_t mask, int64_t nops)
{
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < nops; i += stride) {
    uint64_t slot = mix64((uint64_t)i) & mask;
    atomicAdd((unsigned long long*)&tbl[slot * 2], 1ull);
  }
}

__global__ void atomic_add_single(unsigned long long* counter, int64_t nops)
{
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < nops; i += stride) atomicAdd(counter, 1ull);
}

__global__ void atomic_add_single_waveagg(unsigned long long* counter, int64_t nops)
{
  const int lane = threadIdx.x & 63;
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < nops; i += stride) {
    uint64_t m = __ballot(1);
    unsigned long long base = 0;
    int leader = __ffsll((unsigned long long)m) - 1;
    if (lane == leader) base = atomicAdd(counter, (unsigned long long)__popcll(m));
    base = __shfl(base, leader);
    (void)base;
  }
}

template <typename F>
static double bench(F launch, int iters)
{
  hipEvent_t a, b;
  hipEventCreate(&a);
  hipEventCreate(&b);
  launch();  // warm
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
  const int64_t TBL_BYTES = 4LL << 30;  // 4 GiB table, past L3
  const uint64_t nslots = TBL_BYTES / 16;
  const uint64_t mask = nslots - 1;
  const int64_t nops = 100'000'000;
  longlong2* tbl;
  int64_t* out;
  CHECK(hipMalloc(&tbl, TBL_BYTES));
  CHECK(hipMalloc(&out, 8));
  CHECK(hipMemset(tbl, 0x11, TBL_BYTES));
  dim3 g(2048), t(256);

  double ms;
  ms = bench([&] { hipLaunchKernelGGL(stream_read, g, t, 0, 0, (int64_t*)tbl, TBL_BYTES / 8, out); }, 5);
  printf("stream_read        : %.3f ms  %.1f GB/s\n", ms, TBL_BYTES / ms / 1e6);

  ms = bench([&] { hipLaunchKernelGGL(random_gather16, g, t, 0, 0, tbl, mask, nops, out); }, 5);
  printf("random_gather16    : %.3f ms  %.1f M ops/s  (alg %.1f GB/s, line %.1f GB/s)\n", ms,
         nops / ms / 1e3, nops * 16.0 / ms / 1e6, nops * 128.0 / ms / 1e6);

  ms = bench([&] { hipLaunchKernelGGL(random_gather16_dep2, g, t, 0, 0, tbl, mask, nops, out); }, 5);
  printf("random_gather_dep2 : %.3f ms  %.1f M rows/s\n", ms, nops / ms / 1e3);

  ms = bench([&] { hipLaunchKernelGGL(atomic_cas_scatter, g, t, 0, 0, (int64_t*)tbl, mask, nops); }, 5);
  printf("atomic_cas_scatter : %.3f ms  %.1f M ops/s\n", ms, nops / ms / 1e3);

  ms = bench([&] { hipLaunchKernelGGL(atomic_add_scatter, g, t, 0, 0, (int64_t*)tbl, mask, nops); }, 5);
  printf("atomic_add_scatter : %.3f ms  %.1f M ops/s\n", ms, nops / ms / 1e3);

  unsigned long long* ctr;
  CHECK(hipMalloc(&ctr, 8));
  CHECK(hipMemset(ctr, 0, 8));
  ms = bench([&] { hipLaunchKernelGGL(atomic_add_single, g, t, 0, 0, ctr, nops / 10); }, 3);
  printf("atomic_add_1ctr    : %.3f ms  %.1f M ops/s\n", ms, nops / 10 / ms / 1e3);
  ms = bench([&] { hipLaunchKernelGGL(atomic_add_single_waveagg, g, t, 0, 0, ctr, nops / 10); }, 3);
  printf("atomic_add_1ctr_wag: %.3f ms  %.1f M ops/s\n", ms, nops / 10 / ms / 1e3);
  return 0;
}
