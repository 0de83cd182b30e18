/*
 * scatter_sweep.hip — staging-tile-size sweep for the two bucket-partition
 * passes on the bench shape (100M rows, P=512 then 256-way sub-split).
 *
 * Hypothesis (PMC r01: 31 GB touched vs 16 GB algorithmic => ~1.9x write
 * amplification): flush runs average TILE/P rows; at TILE=4096, P=512 that
 * is 8 rows = 128 B unaligned => ~2 lines touched per line of payload.
 * Bigger tiles lengthen runs (8192/512 = 16 rows => 1.5x) at the cost of
 * occupancy (128 KB LDS => 1 block/CU).  Also measures an atomic-cursor
 * pass-A variant that needs no separate count+scan pass (slack-preallocated
 * group segments; usable by the non-stable N=1 local partition only).
 * Diagnostic only (not linked into the product library).
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
constexpr int PA = 512;   // pass-A groups
constexpr int SUB = 256;  // pass-B fanout
constexpr int BLOCKS = 512;

__device__ __forceinline__ uint32_t groupA_of(int64_t k)
{
  return (uint32_t)(dj_mix64((uint64_t)k) & (uint64_t)(PA - 1));
}
__device__ __forceinline__ uint32_t subB_of(int64_t k)
{
  return (uint32_t)(dj_mix64((uint64_t)k) >> 32) & (uint32_t)(SUB - 1);
}

/* exclusive scan of P per-thread values (tid<P holds base[tid]) via wave
 * shfl scans + one cross-wave round: 2 barriers instead of 2*log2(P) */
__device__ __forceinline__ void wave_excl_scan(uint32_t* base, uint32_t* wsum, int P, int tid)
{
  const int lane = tid & 63, wid = tid >> 6;
  const int nw = P >> 6;
  uint32_t v = (tid < P) ? base[tid] : 0;
  uint32_t x = v;
  for (int d = 1; d < 64; d <<= 1) {
    uint32_t y = __shfl_up(x, d);
    if (lane >= d) x += y;
  }
  if (tid < P && lane == 63) wsum[wid] = x;
  __syncthreads();
  if (wid == 0) {
    uint32_t w = (lane < nw) ? wsum[lane] : 0;
    uint32_t xx = w;
    for (int d = 1; d < 64; d <<= 1) {
      uint32_t y = __shfl_up(xx, d);
      if (lane >= d) xx += y;
    }
    if (lane < nw) wsum[lane] = xx - w;  // exclusive wave bases
  }
  __syncthreads();
  if (tid < P) base[tid] = x - v + wsum[wid];
}

__global__ void gen_kernel(int64_t* keys, int64_t* pay, int64_t n)
{
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  keys[i] = dj_build_key((uint64_t)i, (uint64_t)n, (uint64_t)(2 * n), 1234);
  pay[i] = i;
}

/* per-block per-group counts for pass A (product bucket_count equivalent) */
__global__ __launch_bounds__(THREADS) void countA_kernel(const int64_t* __restrict__ keys,
                                                         int64_t n, uint32_t* __restrict__ counts)
{
  __shared__ uint32_t hist[PA];
  if (threadIdx.x < PA) hist[threadIdx.x] = 0;
  __syncthreads();
  const int64_t chunk = (n + gridDim.x - 1) / gridDim.x;
  const int64_t s = (int64_t)blockIdx.x * chunk, e = min(s + chunk, n);
  for (int64_t i = s + threadIdx.x; i < e; i += blockDim.x)
    atomicAdd(&hist[groupA_of(__builtin_nontemporal_load(&keys[i]))], 1u);
  __syncthreads();
  if (threadIdx.x < PA) counts[(size_t)blockIdx.x * PA + threadIdx.x] = hist[threadIdx.x];
}

/* pass A scatter, templated staging tile.  ATOMIC_CUR: per-tile global
 * atomicAdd cursor per group into slack segments (g*cap), no counts/segoff. */
template <int TILE, bool ATOMIC_CUR, bool WAVESCAN = false>
__global__ __launch_bounds__(THREADS) void scatterA_kernel(
  const int64_t* __restrict__ keys, const int64_t* __restrict__ pay, int64_t n,
  const uint32_t* __restrict__ counts, const int64_t* __restrict__ segoff,
  unsigned long long* __restrict__ gcursor, int64_t cap, longlong2* __restrict__ out)
{
  extern __shared__ __attribute__((aligned(16))) char smem[];
  longlong2* tbuf = (longlong2*)smem;
  uint32_t* hist = (uint32_t*)(tbuf + TILE);
  uint32_t* base = hist + PA;
  uint32_t* gcur = base + PA;
  __shared__ uint32_t wsum[16];
  const int tid = threadIdx.x;
  if (!ATOMIC_CUR && tid < PA)
    gcur[tid] = (uint32_t)segoff[tid] + counts[(size_t)blockIdx.x * PA + tid];
  __syncthreads();
  const int64_t chunk = (n + gridDim.x - 1) / gridDim.x;
  const int64_t start = (int64_t)blockIdx.x * chunk, end = min(start + chunk, n);
  constexpr int VPT = TILE / THREADS;
  for (int64_t t0 = start; t0 < end; t0 += TILE) {
    const int count = (int)min((int64_t)TILE, end - t0);
    if (tid < PA) hist[tid] = 0;
    __syncthreads();
    longlong2 r[VPT];
    uint32_t g[VPT], rank[VPT];
    int nv = 0;
    for (int v = 0; v < VPT; v++) {
      int64_t i = t0 + (int64_t)v * blockDim.x + tid;
      if (i < end) {
        r[v].x = __builtin_nontemporal_load(&keys[i]);
        r[v].y = __builtin_nontemporal_load(&pay[i]);
        g[v] = groupA_of(r[v].x);
        rank[v] = atomicAdd(&hist[g[v]], 1u);
        nv = v + 1;
      }
    }
    __syncthreads();
    if (tid < PA) base[tid] = hist[tid];
    __syncthreads();
    if (WAVESCAN) {
      wave_excl_scan(base, wsum, PA, tid);
    } else {
      for (int off = 1; off < PA; off <<= 1) {
        uint32_t add = (tid < PA && tid >= off) ? base[tid - off] : 0;
        __syncthreads();
        if (tid < PA) base[tid] += add;
        __syncthreads();
      }
      if (tid < PA) base[tid] -= hist[tid];
    }
    __syncthreads();
    if (ATOMIC_CUR && tid < PA)
      gcur[tid] = (uint32_t)((int64_t)tid * cap) +
                  (uint32_t)atomicAdd(&gcursor[tid], (unsigned long long)hist[tid]);
    __syncthreads();
    for (int v = 0; v < nv; v++) tbuf[base[g[v]] + rank[v]] = r[v];
    __syncthreads();
    for (int pos = tid; pos < count; pos += blockDim.x) {
      longlong2 row = tbuf[pos];
      uint32_t gg = groupA_of(row.x);
      out[gcur[gg] + (pos - base[gg])] = row;
    }
    __syncthreads();
    if (!ATOMIC_CUR && tid < PA) gcur[tid] += hist[tid];
    __syncthreads();
  }
}

/* pass B: one block per pass-A group segment, SUB-way sub-split */
template <int TILE, bool WAVESCAN = false>
__global__ __launch_bounds__(THREADS) void scatterB_kernel(const longlong2* __restrict__ in,
                                                           const int64_t* __restrict__ segoff,
                                                           longlong2* __restrict__ out)
{
  extern __shared__ __attribute__((aligned(16))) char smem[];
  longlong2* tbuf = (longlong2*)smem;
  uint32_t* hist = (uint32_t*)(tbuf + TILE);
  uint32_t* base = hist + SUB;
  uint32_t* gcur = base + SUB;
  uint32_t* seghist = gcur + SUB;
  __shared__ uint32_t wsum[16];
  const int tid = threadIdx.x;
  const int64_t s0 = segoff[blockIdx.x], s1 = segoff[blockIdx.x + 1];
  if (tid < SUB) seghist[tid] = 0;
  __syncthreads();
  for (int64_t i = s0 + tid; i < s1; i += blockDim.x)
    atomicAdd(&seghist[subB_of(__builtin_nontemporal_load(&in[i].x))], 1u);
  __syncthreads();
  if (tid == 0) {
    uint32_t acc = 0;
    for (int j = 0; j < SUB; j++) {
      gcur[j] = (uint32_t)s0 + acc;
      acc += seghist[j];
    }
  }
  __syncthreads();
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
    if (WAVESCAN) {
      wave_excl_scan(base, wsum, SUB, tid);
    } else {
      for (int off = 1; off < SUB; off <<= 1) {
        uint32_t add = (tid < SUB && tid >= off) ? base[tid - off] : 0;
        __syncthreads();
        if (tid < SUB) base[tid] += add;
        __syncthreads();
      }
      if (tid < SUB) base[tid] -= hist[tid];
    }
    __syncthreads();
    for (int v = 0; v < nv; v++) tbuf[base[g[v]] + rank[v]] = r[v];
    __syncthreads();
    for (int pos = tid; pos < count; pos += blockDim.x) {
      longlong2 row = tbuf[pos];
      uint32_t gg = subB_of(row.x);
      out[gcur[gg] + (pos - base[gg])] = row;
    }
    __syncthreads();
    if (tid < SUB) gcur[tid] += hist[tid];
    __syncthreads();
  }
}

template <int TILE>
void run_variant(const int64_t* keys, const int64_t* pay, int64_t n, uint32_t* counts,
                 int64_t* segoff_a, unsigned long long* gcursor, longlong2* mid,
                 longlong2* out, int reps)
{
  size_t ldsA = TILE * sizeof(longlong2) + 3 * PA * sizeof(uint32_t);
  size_t ldsB = TILE * sizeof(longlong2) + 4 * SUB * sizeof(uint32_t);

  /* counts + host scan -> per-block cursors + segment offsets */
  countA_kernel<<<BLOCKS, THREADS>>>(keys, n, counts);
  std::vector<uint32_t> h_counts((size_t)BLOCKS * PA);
  CHECK(hipMemcpy(h_counts.data(), counts, h_counts.size() * 4, hipMemcpyDeviceToHost));
  std::vector<int64_t> h_seg(PA + 1, 0);
  std::vector<uint32_t> h_blockoff((size_t)BLOCKS * PA);
  for (int g = 0; g < PA; g++) {
    int64_t tot = 0;
    for (int b = 0; b < BLOCKS; b++) {
      h_blockoff[(size_t)b * PA + g] = (uint32_t)tot;
      tot += h_counts[(size_t)b * PA + g];
    }
    h_seg[g + 1] = h_seg[g] + tot;
  }
  CHECK(hipMemcpy(counts, h_blockoff.data(), h_blockoff.size() * 4, hipMemcpyHostToDevice));
  CHECK(hipMemcpy(segoff_a, h_seg.data(), (PA + 1) * 8, hipMemcpyHostToDevice));

  hipEvent_t e0, e1;
  CHECK(hipEventCreate(&e0));
  CHECK(hipEventCreate(&e1));
  float msA = 0, msB = 0, msAat = 0;
  const int64_t cap = n / PA + n / PA / 16 + 1024;  // ~6% slack
  float msAw = 0, msBw = 0;
  for (int rep = 0; rep < reps; rep++) {
    float ms;
    CHECK(hipMemset(gcursor, 0, PA * 8));
    CHECK(hipEventRecord(e0));
    scatterA_kernel<TILE, true>
      <<<BLOCKS, THREADS, ldsA>>>(keys, pay, n, counts, segoff_a, gcursor, cap, mid);
    CHECK(hipEventRecord(e1));
    CHECK(hipEventSynchronize(e1));
    CHECK(hipEventElapsedTime(&ms, e0, e1));
    if (rep) msAat += ms;

    CHECK(hipMemset(gcursor, 0, PA * 8));
    CHECK(hipEventRecord(e0));
    scatterA_kernel<TILE, true, true>
      <<<BLOCKS, THREADS, ldsA>>>(keys, pay, n, counts, segoff_a, gcursor, cap, mid);
    CHECK(hipEventRecord(e1));
    CHECK(hipEventSynchronize(e1));
    CHECK(hipEventElapsedTime(&ms, e0, e1));
    if (rep) msAw += ms;

    CHECK(hipEventRecord(e0));
    scatterB_kernel<TILE><<<PA, THREADS, ldsB>>>(mid, segoff_a, out);
    CHECK(hipEventRecord(e1));
    CHECK(hipEventSynchronize(e1));
    CHECK(hipEventElapsedTime(&ms, e0, e1));
    if (rep) msB += ms;

    CHECK(hipEventRecord(e0));
    scatterB_kernel<TILE, true><<<PA, THREADS, ldsB>>>(mid, segoff_a, out);
    CHECK(hipEventRecord(e1));
    CHECK(hipEventSynchronize(e1));
    CHECK(hipEventElapsedTime(&ms, e0, e1));
    if (rep) msBw += ms;
  }
  CHECK(hipGetLastError());
  int r = reps - 1;
  printf("TILE %5d | A-atomic %7.3f -> wavescan %7.3f ms | B %7.3f -> wavescan %7.3f ms\n",
         TILE, msAat / r, msAw / r, msB / r, msBw / r);
  CHECK(hipEventDestroy(e0));
  CHECK(hipEventDestroy(e1));
}

int main(int argc, char** argv)
{
  int64_t n = argc > 1 ? atoll(argv[1]) : 100000000;
  int reps = argc > 2 ? atoi(argv[2]) : 4;
  int64_t *keys, *pay, *segoff_a;
  longlong2 *mid, *out;
  uint32_t* counts;
  unsigned long long* gcursor;
  CHECK(hipMalloc(&keys, n * 8));
  CHECK(hipMalloc(&pay, n * 8));
  CHECK(hipMalloc(&mid, (n + PA * (n / PA / 8 + 1024)) * 16));
  CHECK(hipMalloc(&out, (n + PA * (n / PA / 8 + 1024)) * 16));
  CHECK(hipMalloc(&counts, (size_t)BLOCKS * PA * 4));
  CHECK(hipMalloc(&segoff_a, (PA + 1) * 8));
  CHECK(hipMalloc(&gcursor, PA * 8));
  gen_kernel<<<(int)((n + 255) / 256), 256>>>(keys, pay, n);
  CHECK(hipDeviceSynchronize());
  run_variant<4096>(keys, pay, n, counts, segoff_a, gcursor, mid, out, reps);
  run_variant<6144>(keys, pay, n, counts, segoff_a, gcursor, mid, out, reps);
  run_variant<8192>(keys, pay, n, counts, segoff_a, gcursor, mid, out, reps);
  return 0;
}
