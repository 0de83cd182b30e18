/*
 * dj_kernels.hip — hand-written CDNA4 (gfx950) kernels for the distributed
 * repartitioned hash-join hot path. MI355X-first design, not a CUDA port:
 * 64-wide wavefront ballot/prefix-sum scatter, coalesced int64 loads,
 * grid-stride launches sized for 256 CUs / 8 XCDs.
 *
 * Replaces (SURVEY.md §2 third-party kernel table):
 *  - cudf::hash_partition  (reference calls: distributed_join.cpp:213-225,
 *    shuffle_on.cpp:59-60): stable partition = per-wave histogram pass +
 *    device scan + wave-ballot rank scatter (no LDS atomics, no sort).
 *  - cudf::inner_join      (reference call: distributed_join.cpp:79):
 *    open-addressing (linear probing, <=50%% fill, power-of-two slots)
 *    atomicCAS build + probe with single-pass wave-aggregated append
 *    (replaces cuDF's count+gather two-pass; row order is unspecified by the
 *    API — reference tests sort before comparing,
 *    compare_against_single_gpu.cu:167-174).
 *  - generate_dataset.cuh:40-260: deterministic counter-based restatement
 *    (spec in dj_rng.h) — bit-identical to the CPU oracle on any device.
 */
#include "dj_error.hpp"
#include "dj_hash.h"
#include "dj_kernels.hpp"
#include "dj_rng.h"

#include <hip/hip_runtime.h>

namespace dj {

constexpr int BLOCK = 256;
constexpr int WAVE = 64;
constexpr int WPB = BLOCK / WAVE;  // waves per block

/* Non-temporal loads for single-use streams: partition/scatter inputs are
 * read once and must not evict the partially-written output lines from the
 * per-XCD L2 (PMC measured 1.24-1.49x write amplification on the scatter
 * passes from exactly that eviction). */
template <typename T>
__device__ __forceinline__ T nt_load(const T* p)
{
  return __builtin_nontemporal_load(p);
}
__device__ __forceinline__ longlong2 nt_load2(const longlong2* p)
{
  longlong2 v;
  v.x = nt_load(&p->x);
  v.y = nt_load(&p->y);
  return v;
}

/* ------------------------------------------------------------------ misc */

__global__ void fill_i64_kernel(int64_t* dst, int64_t value, int64_t n)
{
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) dst[i] = value;
}

static int grid_for(int64_t n)
{
  int64_t blocks = (n + BLOCK - 1) / BLOCK;
  if (blocks > 2048) blocks = 2048;  // 256 CU x 8 blocks, grid-stride the rest
  if (blocks < 1) blocks = 1;
  return (int)blocks;
}

/* ---- sentinel-key (-1) fixup: the hash tables reserve -1 as the empty
 * marker, so build rows with key == -1 are skipped by every join path (they
 * set the saw-sentinel flag) and probe rows with -1 never match. The host
 * then joins them out-of-band: collect both sides' -1 payloads, emit the
 * cross product. Rare path: runs only when the flag fired. */
__global__ void collect_neg1_kernel(const int64_t* __restrict__ keys,
                                    const int64_t* __restrict__ pay, int64_t n,
                                    int64_t* __restrict__ out_pay,
                                    unsigned long long* __restrict__ count, int64_t cap)
{
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    if (keys[i] == kEmptyKey) {
      unsigned long long idx = atomicAdd(count, 1ull);
      if ((int64_t)idx < cap) out_pay[idx] = pay ? pay[i] : i;
    }
  }
}

__global__ void emit_neg1_cross_kernel(const int64_t* __restrict__ lpay, int64_t n1,
                                       const int64_t* __restrict__ rpay, int64_t n2,
                                       int64_t base, int64_t* __restrict__ out0,
                                       int64_t* __restrict__ out1,
                                       int64_t* __restrict__ out2,
                                       int64_t* __restrict__ out3, int64_t cap)
{
  int64_t t = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  int64_t total = n1 * n2;
  for (; t < total; t += stride) {
    int64_t idx = base + t;
    if (idx < cap) {
      out0[idx] = kEmptyKey;
      out1[idx] = lpay[t / n2];
      out2[idx] = kEmptyKey;
      out3[idx] = rpay[t % n2];
    }
  }
}

void neg1_cross_join(const int64_t* d_lk, const int64_t* d_lp, int64_t ln,
                     const int64_t* d_rk, const int64_t* d_rp, int64_t rn, int64_t* d_out0,
                     int64_t* d_out1, int64_t* d_out2, int64_t* d_out3, int64_t cap,
                     int64_t* d_counter, hipStream_t s)
{
  unsigned long long* d_cnt = nullptr;
  DJ_HIP_CALL(hipMalloc(&d_cnt, 16));
  DJ_HIP_CALL(hipMemsetAsync(d_cnt, 0, 16, s));
  /* pass 1: count (cap 0 => no payload writes) */
  hipLaunchKernelGGL(collect_neg1_kernel, dim3(grid_for(ln)), dim3(BLOCK), 0, s, d_lk, d_lp,
                     ln, nullptr, d_cnt, 0);
  hipLaunchKernelGGL(collect_neg1_kernel, dim3(grid_for(rn)), dim3(BLOCK), 0, s, d_rk, d_rp,
                     rn, nullptr, d_cnt + 1, 0);
  DJ_HIP_CALL(hipGetLastError());
  unsigned long long h_cnt[2] = {0, 0};
  DJ_HIP_CALL(hipMemcpyAsync(h_cnt, d_cnt, 16, hipMemcpyDeviceToHost, s));
  DJ_HIP_CALL(hipStreamSynchronize(s));
  const int64_t n1 = (int64_t)h_cnt[0], n2 = (int64_t)h_cnt[1];
  if (n1 > 0 && n2 > 0) {
    int64_t *d_lpv = nullptr, *d_rpv = nullptr;
    DJ_HIP_CALL(hipMalloc(&d_lpv, (size_t)n1 * 8));
    DJ_HIP_CALL(hipMalloc(&d_rpv, (size_t)n2 * 8));
    DJ_HIP_CALL(hipMemsetAsync(d_cnt, 0, 16, s));
    hipLaunchKernelGGL(collect_neg1_kernel, dim3(grid_for(ln)), dim3(BLOCK), 0, s, d_lk,
                       d_lp, ln, d_lpv, d_cnt, n1);
    hipLaunchKernelGGL(collect_neg1_kernel, dim3(grid_for(rn)), dim3(BLOCK), 0, s, d_rk,
                       d_rp, rn, d_rpv, d_cnt + 1, n2);
    DJ_HIP_CALL(hipGetLastError());
    int64_t base = 0;
    DJ_HIP_CALL(hipMemcpyAsync(&base, d_counter, 8, hipMemcpyDeviceToHost, s));
    DJ_HIP_CALL(hipStreamSynchronize(s));
    const int64_t total = n1 * n2;
    hipLaunchKernelGGL(emit_neg1_cross_kernel, dim3(grid_for(total)), dim3(BLOCK), 0, s,
                       d_lpv, n1, d_rpv, n2, base, d_out0, d_out1, d_out2, d_out3, cap);
    DJ_HIP_CALL(hipGetLastError());
    const int64_t newcount = base + total;
    DJ_HIP_CALL(hipMemcpyAsync(d_counter, &newcount, 8, hipMemcpyHostToDevice, s));
    DJ_HIP_CALL(hipStreamSynchronize(s));
    DJ_HIP_CALL(hipFree(d_lpv));
    DJ_HIP_CALL(hipFree(d_rpv));
  }
  DJ_HIP_CALL(hipFree(d_cnt));
}

void fill_i64(int64_t* d_dst, int64_t value, int64_t n, hipStream_t s)
{
  if (n <= 0) return;
  hipLaunchKernelGGL(fill_i64_kernel, dim3(grid_for(n)), dim3(BLOCK), 0, s, d_dst, value, n);
  DJ_HIP_CALL(hipGetLastError());
}

/* ------------------------------------------------------------- generator */

__global__ void gen_build_kernel(int64_t* keys, int64_t* pay, int64_t n_global,
                                 int64_t rand_max, uint64_t seed, int uniq, int64_t row0,
                                 int64_t nrows)
{
  int64_t t = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; t < nrows; t += stride) {
    int64_t i = row0 + t;
    keys[t] = uniq ? dj_build_key((uint64_t)i, (uint64_t)n_global, rand_max, seed)
                   : dj_build_key_nonuniq((uint64_t)i, rand_max, seed);
    if (pay) pay[t] = i;
  }
}

__global__ void gen_probe_kernel(int64_t* keys, int64_t* pay, int64_t build_n_global,
                                 int64_t rand_max, double selectivity, uint64_t seed,
                                 int64_t row0, int64_t nrows)
{
  int64_t t = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; t < nrows; t += stride) {
    int64_t j = row0 + t;
    keys[t] = dj_probe_key((uint64_t)j, (uint64_t)build_n_global, rand_max, selectivity, seed);
    if (pay) pay[t] = j;
  }
}

void generate_build(int64_t* d_keys, int64_t* d_pay, int64_t n_global, int64_t rand_max,
                    uint64_t seed, bool uniq, int64_t row0, int64_t nrows, hipStream_t s)
{
  if (nrows <= 0) return;
  hipLaunchKernelGGL(gen_build_kernel, dim3(grid_for(nrows)), dim3(BLOCK), 0, s, d_keys, d_pay,
                     n_global, rand_max, seed, (int)uniq, row0, nrows);
  DJ_HIP_CALL(hipGetLastError());
}

void generate_probe(int64_t* d_keys, int64_t* d_pay, int64_t build_n_global, int64_t rand_max,
                    double selectivity, uint64_t seed, int64_t row0, int64_t nrows, hipStream_t s)
{
  if (nrows <= 0) return;
  hipLaunchKernelGGL(gen_probe_kernel, dim3(grid_for(nrows)), dim3(BLOCK), 0, s, d_keys, d_pay,
                     build_n_global, rand_max, selectivity, seed, row0, nrows);
  DJ_HIP_CALL(hipGetLastError());
}

/* ------------------------------------------------------- stable partition */
/*
 * Wave-contiguous assignment: wave w owns rows [w*rpw, (w+1)*rpw). Lanes read
 * row base+it*64+lane, so within one iteration lane order == row order and
 * global memory loads coalesce to 512 B per instruction.
 *
 * Pass 1 (count): per-wave histogram held in registers — lane q accumulates
 * the count for partition q from __ballot(p == q) (64-bit wave64 ballots;
 * nparts <= 64).
 * Scan: per-partition block scan over waves (exclusive), then a tiny scan of
 * partition totals into global partition offsets.
 * Pass 2 (scatter): lane q holds the running cursor of partition q; each row
 * gets dst = shfl(cursor, p) + rank, rank = popcount(ballot-mask of its own
 * partition & lanes-below mask) — conflict-free, stable, no atomics.
 */

struct PartGeom {
  int64_t nwaves;
  int64_t rows_per_wave;
  int blocks;
};

static PartGeom part_geom(int64_t n)
{
  PartGeom g;
  int64_t target = (n + 255) / 256;  // >=256 rows per wave
  g.nwaves = target < 1 ? 1 : (target > 8192 ? 8192 : target);
  g.blocks = (int)((g.nwaves + WPB - 1) / WPB);
  g.nwaves = (int64_t)g.blocks * WPB;
  g.rows_per_wave = (n + g.nwaves - 1) / g.nwaves;
  return g;
}

size_t hash_partition_scratch_bytes(int64_t n, int nparts)
{
  PartGeom g = part_geom(n);
  /* wave_counts/prefix [nwaves][nparts] (int64) + totals [nparts] */
  return (size_t)(g.nwaves * nparts + nparts) * sizeof(int64_t);
}

__device__ __forceinline__ uint32_t part_of(int64_t key, int hash_fn, uint32_t seed, int nparts)
{
  return dj_row_hash(key, hash_fn, seed) % (uint32_t)nparts;
}

/* NPL = ceil(nparts/64): lane q owns partitions q, q+64, ..., q+(NPL-1)*64.
 * One ballot per partition id per 64-row batch keeps the scatter stable and
 * atomic-free at any nparts <= 64*NPL (the reference's own tests drive
 * nparts = ranks x over_decom up to 80, compare_against_single_gpu.cu:237);
 * cost grows linearly in nparts, which only large worlds x od ever pay. */
template <int NPL>
__global__ void part_count_kernel(const int64_t* __restrict__ keys, int64_t n, int nparts,
                                  int hash_fn, uint32_t seed, int64_t rows_per_wave,
                                  int64_t* __restrict__ wave_counts)
{
  const int lane = threadIdx.x & (WAVE - 1);
  const int64_t w = (int64_t)blockIdx.x * WPB + (threadIdx.x >> 6);
  const int64_t start = w * rows_per_wave;
  const int64_t end = min(start + rows_per_wave, n);
  int64_t my_count[NPL];  // lane q counts partition c*64 + q
#pragma unroll
  for (int c = 0; c < NPL; c++) my_count[c] = 0;
  for (int64_t base = start; base < end; base += WAVE) {
    int64_t i = base + lane;
    uint32_t p = 0xFFFFFFFFu;
    if (i < end) p = part_of(keys[i], hash_fn, seed, nparts);
#pragma unroll
    for (int c = 0; c < NPL; c++) {
      const int qhi = min(WAVE, nparts - c * WAVE);  // uniform
      for (int j = 0; j < qhi; j++) {
        uint64_t m = __ballot(p == (uint32_t)(c * WAVE + j));
        if (lane == j) my_count[c] += __popcll(m);
      }
    }
  }
#pragma unroll
  for (int c = 0; c < NPL; c++) {
    const int q = c * WAVE + lane;
    if (q < nparts) wave_counts[w * nparts + q] = my_count[c];
  }
}

/* grid = nparts blocks; block p scans column p over nwaves (exclusive),
 * in-place, and writes the partition total to totals[p]. */
__global__ void part_scan_kernel(int64_t* wave_counts, int64_t nwaves, int nparts,
                                 int64_t* totals)
{
  const int p = blockIdx.x;
  __shared__ int64_t sh[BLOCK];
  int64_t running = 0;
  for (int64_t base = 0; base < nwaves; base += BLOCK) {
    int64_t w = base + threadIdx.x;
    int64_t v = (w < nwaves) ? wave_counts[w * nparts + p] : 0;
    /* Hillis-Steele inclusive scan in LDS */
    sh[threadIdx.x] = v;
    __syncthreads();
    for (int off = 1; off < BLOCK; off <<= 1) {
      int64_t add = (threadIdx.x >= off) ? sh[threadIdx.x - off] : 0;
      __syncthreads();
      sh[threadIdx.x] += add;
      __syncthreads();
    }
    int64_t incl = sh[threadIdx.x];
    if (w < nwaves) wave_counts[w * nparts + p] = running + incl - v;  // exclusive
    int64_t chunk_total = sh[BLOCK - 1];
    running += chunk_total;
    __syncthreads();
  }
  if (threadIdx.x == 0) totals[p] = running;
}

/* single block: exclusive scan of totals -> offsets[nparts+1] */
__global__ void part_offsets_kernel(const int64_t* totals, int nparts, int64_t* offsets)
{
  if (threadIdx.x == 0) {
    int64_t acc = 0;
    for (int p = 0; p < nparts; p++) {
      offsets[p] = acc;
      acc += totals[p];
    }
    offsets[nparts] = acc;
  }
}

template <int NPL>
__global__ void part_scatter_kernel(const int64_t* __restrict__ keys,
                                    const int64_t* __restrict__ pay, int64_t n, int nparts,
                                    int hash_fn, uint32_t seed, int64_t rows_per_wave,
                                    const int64_t* __restrict__ wave_prefix,
                                    const int64_t* __restrict__ offsets,
                                    int64_t* __restrict__ out_keys,
                                    int64_t* __restrict__ out_pay)
{
  const int lane = threadIdx.x & (WAVE - 1);
  const int64_t w = (int64_t)blockIdx.x * WPB + (threadIdx.x >> 6);
  const int64_t start = w * rows_per_wave;
  const int64_t end = min(start + rows_per_wave, n);
  /* lane q holds the cursor for partitions c*64 + q */
  int64_t cursor[NPL];
#pragma unroll
  for (int c = 0; c < NPL; c++) {
    const int q = c * WAVE + lane;
    cursor[c] = (q < nparts) ? offsets[q] + wave_prefix[w * nparts + q] : 0;
  }
  const uint64_t lt_mask = (lane == 0) ? 0ull : (~0ull >> (64 - lane));
  for (int64_t base = start; base < end; base += WAVE) {
    int64_t i = base + lane;
    bool valid = i < end;
    int64_t k = 0, v = 0;
    uint32_t p = 0xFFFFFFFFu;
    if (valid) {
      k = keys[i];
      v = pay ? pay[i] : 0;
      p = part_of(k, hash_fn, seed, nparts);
    }
    uint64_t my_q_mask[NPL];  // lane q's ballot mask for partition c*64 + q
#pragma unroll
    for (int c = 0; c < NPL; c++) {
      my_q_mask[c] = 0;
      const int qhi = min(WAVE, nparts - c * WAVE);  // uniform
      for (int j = 0; j < qhi; j++) {
        uint64_t m = __ballot(p == (uint32_t)(c * WAVE + j));
        if (lane == j) my_q_mask[c] = m;
      }
    }
    /* mask and cursor of MY partition, fetched from lane p%64, slot p/64 */
    const int src = (int)(valid ? (p & (WAVE - 1)) : 0);
    uint64_t m_p = 0;
    int64_t base_dst = 0;
#pragma unroll
    for (int c = 0; c < NPL; c++) {
      uint64_t mc = __shfl((unsigned long long)my_q_mask[c], src);
      int64_t bc = __shfl(cursor[c], src);
      if (NPL == 1 || (valid && (p >> 6) == (uint32_t)c)) {
        m_p = mc;
        base_dst = bc;
      }
    }
    if (valid) {
      int rank = __popcll(m_p & lt_mask);
      int64_t dst = base_dst + rank;
      out_keys[dst] = k;
      if (out_pay) out_pay[dst] = v;
    }
#pragma unroll
    for (int c = 0; c < NPL; c++) cursor[c] += __popcll(my_q_mask[c]);
  }
}

void partition_count(const int64_t* d_keys, int64_t n, int nparts, int hash_fn,
                     uint32_t hash_seed, void* d_scratch, hipStream_t s)
{
  DJ_CHECK_ERROR(nparts >= 1 && nparts <= kMaxPartitions, "nparts must be in [1,1024]");
  if (n <= 0) return;
  PartGeom g = part_geom(n);
  auto launch = [&](auto kern) {
    hipLaunchKernelGGL(kern, dim3(g.blocks), dim3(BLOCK), 0, s, d_keys, n, nparts, hash_fn,
                       hash_seed, g.rows_per_wave, (int64_t*)d_scratch);
  };
  if (nparts <= 64)
    launch(part_count_kernel<1>);
  else if (nparts <= 128)
    launch(part_count_kernel<2>);
  else if (nparts <= 256)
    launch(part_count_kernel<4>);
  else if (nparts <= 512)
    launch(part_count_kernel<8>);
  else
    launch(part_count_kernel<16>);
  DJ_HIP_CALL(hipGetLastError());
}

void partition_scan(int64_t n, int nparts, void* d_scratch, int64_t* d_offsets, hipStream_t s)
{
  if (n <= 0) {
    DJ_HIP_CALL(hipMemsetAsync(d_offsets, 0, (size_t)(nparts + 1) * sizeof(int64_t), s));
    return;
  }
  PartGeom g = part_geom(n);
  int64_t* wave_counts = (int64_t*)d_scratch;
  int64_t* totals = wave_counts + g.nwaves * nparts;
  hipLaunchKernelGGL(part_scan_kernel, dim3(nparts), dim3(BLOCK), 0, s, wave_counts, g.nwaves,
                     nparts, totals);
  DJ_HIP_CALL(hipGetLastError());
  hipLaunchKernelGGL(part_offsets_kernel, dim3(1), dim3(64), 0, s, totals, nparts, d_offsets);
  DJ_HIP_CALL(hipGetLastError());
}

void partition_scatter(const int64_t* d_keys, const int64_t* d_pay, int64_t n, int nparts,
                       int hash_fn, uint32_t hash_seed, const int64_t* d_offsets,
                       void* d_scratch, int64_t* d_out_keys, int64_t* d_out_pay, hipStream_t s)
{
  if (n <= 0) return;
  PartGeom g = part_geom(n);
  auto launch = [&](auto kern) {
    hipLaunchKernelGGL(kern, dim3(g.blocks), dim3(BLOCK), 0, s, d_keys, d_pay, n, nparts,
                       hash_fn, hash_seed, g.rows_per_wave, (int64_t*)d_scratch, d_offsets,
                       d_out_keys, d_out_pay);
  };
  if (nparts <= 64)
    launch(part_scatter_kernel<1>);
  else if (nparts <= 128)
    launch(part_scatter_kernel<2>);
  else if (nparts <= 256)
    launch(part_scatter_kernel<4>);
  else if (nparts <= 512)
    launch(part_scatter_kernel<8>);
  else
    launch(part_scatter_kernel<16>);
  DJ_HIP_CALL(hipGetLastError());
}

void hash_partition(const int64_t* d_keys, const int64_t* d_pay, int64_t n, int nparts,
                    int hash_fn, uint32_t hash_seed, int64_t* d_out_keys, int64_t* d_out_pay,
                    int64_t* d_offsets, void* d_scratch, hipStream_t s)
{
  partition_count(d_keys, n, nparts, hash_fn, hash_seed, d_scratch, s);
  partition_scan(n, nparts, d_scratch, d_offsets, s);
  partition_scatter(d_keys, d_pay, n, nparts, hash_fn, hash_seed, d_offsets, d_scratch,
                    d_out_keys, d_out_pay, s);
}

/* ------------------------------------------------------------ local join */

int64_t join_table_slots(int64_t ln)
{
  int64_t p = 1;
  while (p < 2 * ln + 1) p <<= 1;  // <=50% fill
  return p;
}

void join_table_init(int64_t* d_table, int64_t nslots, hipStream_t s)
{
  /* kEmptyKey == -1 == all bytes 0xFF: one HBM-rate memset over the
   * interleaved {key,val} pairs */
  DJ_HIP_CALL(hipMemsetAsync(d_table, 0xFF, (size_t)nslots * 2 * sizeof(int64_t), s));
}

/* Table layout: nslots interleaved 16 B {key, val} pairs — one random
 * cache-line fetch serves both the key compare and the payload read.
 * PAIRS: input rows as interleaved longlong2 instead of two columns. */
template <bool PAIRS>
__global__ void join_build_kernel(const int64_t* __restrict__ lk, const int64_t* __restrict__ lp,
                                  const longlong2* __restrict__ lrows,
                                  int64_t ln, longlong2* __restrict__ table, uint64_t mask,
                                  int* error)
{
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < ln; i += stride) {
    int64_t key = PAIRS ? lrows[i].x : lk[i];
    if (key == kEmptyKey) {
      *error = 1;  // reserved sentinel; loud failure, never silent wrong data
      continue;
    }
    uint64_t slot = dj_mix64((uint64_t)key) & mask;
    for (;;) {
      unsigned long long old = atomicCAS((unsigned long long*)&table[slot].x,
                                         (unsigned long long)kEmptyKey,
                                         (unsigned long long)key);
      if (old == (unsigned long long)kEmptyKey) break;
      slot = (slot + 1) & mask;
    }
    table[slot].y = PAIRS ? lrows[i].y : (lp ? lp[i] : i);
  }
}

void join_build(const int64_t* d_lk, const int64_t* d_lp, int64_t ln, int64_t* d_table,
                int64_t nslots, int* d_error, hipStream_t s)
{
  if (ln <= 0) return;
  hipLaunchKernelGGL(join_build_kernel<false>, dim3(grid_for(ln)), dim3(BLOCK), 0, s, d_lk,
                     d_lp, (const longlong2*)nullptr, ln, (longlong2*)d_table,
                     (uint64_t)(nslots - 1), d_error);
  DJ_HIP_CALL(hipGetLastError());
}

void join_build_pairs(const longlong2* d_rows, int64_t ln, int64_t* d_table, int64_t nslots,
                      int* d_error, hipStream_t s)
{
  if (ln <= 0) return;
  hipLaunchKernelGGL(join_build_kernel<true>, dim3(grid_for(ln)), dim3(BLOCK), 0, s,
                     (const int64_t*)nullptr, (const int64_t*)nullptr, d_rows, ln,
                     (longlong2*)d_table, (uint64_t)(nslots - 1), d_error);
  DJ_HIP_CALL(hipGetLastError());
}

/* Probe with wave-aggregated output append: matches are emitted via one
 * atomicAdd per wave (ballot leader) instead of one per lane — the single
 * global counter would otherwise serialize the whole kernel. All lanes of a
 * wave iterate in lockstep so the ballots are well-defined. */
template <bool PAIRS>
__global__ void join_probe_kernel(const int64_t* __restrict__ rk, const int64_t* __restrict__ rp,
                                  const longlong2* __restrict__ rrows,
                                  int64_t rn, const longlong2* __restrict__ table,
                                  uint64_t mask, int64_t* __restrict__ out0,
                                  int64_t* __restrict__ out1, int64_t* __restrict__ out2,
                                  int64_t* __restrict__ out3, int64_t cap,
                                  unsigned long long* counter)
{
  const int lane = threadIdx.x & (WAVE - 1);
  const uint64_t lt_mask = (lane == 0) ? 0ull : (~0ull >> (64 - lane));
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t j = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;; j += stride) {
    const bool row_valid = j < rn;
    if (__ballot(row_valid) == 0) break;  // whole wave done (uniform)
    int64_t key = 0, payload = 0;
    uint64_t slot = 0;
    bool walking = row_valid;
    if (row_valid) {
      if (PAIRS) {
        longlong2 r = rrows[j];
        key = r.x;
        payload = r.y;
      } else {
        key = rk[j];
        payload = rp ? rp[j] : j;
      }
      slot = dj_mix64((uint64_t)key) & mask;
    }
    for (;;) {
      int64_t mval = 0;
      bool have = false;
      while (walking) {
        longlong2 e = table[slot];
        if (e.x == kEmptyKey) {
          walking = false;
          break;
        }
        slot = (slot + 1) & mask;
        if (e.x == key) {
          mval = e.y;
          have = true;
          break;
        }
      }
      uint64_t m = __ballot(have);
      if (m == 0) break;
      const int leader = (int)(__ffsll((unsigned long long)m) - 1);
      unsigned long long base = 0;
      if (lane == leader) base = atomicAdd(counter, (unsigned long long)__popcll(m));
      base = __shfl(base, leader);
      if (have) {
        int64_t idx = (int64_t)base + __popcll(m & lt_mask);
        if (idx < cap) {
          out0[idx] = key;
          out1[idx] = mval;
          out2[idx] = key;
          out3[idx] = payload;
        }
      }
    }
  }
}

void join_probe(const int64_t* d_rk, const int64_t* d_rp, int64_t rn, const int64_t* d_table,
                int64_t nslots, int64_t* d_out0, int64_t* d_out1, int64_t* d_out2,
                int64_t* d_out3, int64_t cap, int64_t* d_counter, hipStream_t s)
{
  if (rn <= 0) return;
  hipLaunchKernelGGL(join_probe_kernel<false>, dim3(grid_for(rn)), dim3(BLOCK), 0, s, d_rk,
                     d_rp, (const longlong2*)nullptr, rn, (const longlong2*)d_table,
                     (uint64_t)(nslots - 1), d_out0, d_out1, d_out2, d_out3, cap,
                     (unsigned long long*)d_counter);
  DJ_HIP_CALL(hipGetLastError());
}

void join_probe_pairs(const longlong2* d_rows, int64_t rn, const int64_t* d_table,
                      int64_t nslots, int64_t* d_out0, int64_t* d_out1, int64_t* d_out2,
                      int64_t* d_out3, int64_t cap, int64_t* d_counter, hipStream_t s)
{
  if (rn <= 0) return;
  hipLaunchKernelGGL(join_probe_kernel<true>, dim3(grid_for(rn)), dim3(BLOCK), 0, s,
                     (const int64_t*)nullptr, (const int64_t*)nullptr, d_rows, rn,
                     (const longlong2*)d_table, (uint64_t)(nslots - 1), d_out0, d_out1,
                     d_out2, d_out3, cap, (unsigned long long*)d_counter);
  DJ_HIP_CALL(hipGetLastError());
}

/* ----------------------------------------------- bucketed LDS join ------ */
/*
 * MI355X-native local join. A single HBM-resident hash table is bound far
 * below the HBM roofline by random 128 B line fetches, device-scope atomic
 * throughput (~18 G scattered CAS/s measured) and, worst, the single output
 * counter (~83 M serialized atomics/s measured — experiments/membench.hip).
 * Instead: partition both tables into B buckets small enough that a
 * bucket's hash table fits in LDS, then one fused kernel per bucket builds
 * a 4096-slot LDS table (ds atomicCAS) and probes it twice (count, then
 * write) — all random access is on-chip, HBM sees only streaming traffic,
 * and the global output counter is touched ONCE per bucket.
 *
 * The bucket partition is two-level so every scatter pass has at most 1024
 * write streams alive per block (partial-line writes then merge in the
 * per-XCD L2 — a single-pass 32768-way scatter measurably thrashes it):
 *   pass A: P_A = B/256 groups by bits [40, 40+log2(P_A)) of dj_mix64(key)
 *           (block-chunked count + cross-block scan + LDS-cursor scatter);
 *   pass B: one block per group: 256 sub-buckets by bits [32,40), count +
 *           in-block scan + scatter, emitting the final bucket offsets.
 * The decomposition is internal to the local join (result row order is
 * unspecified by the API), so both passes are NON-stable — unlike the
 * rank-level hash_partition above, which stays stable (reference pin:
 * SURVEY.md appendix, batch offsets).
 * LDS slot hash = low 32 bits of dj_mix64; the rank-level partition uses
 * MurmurHash3%G, so bucket occupancy is independent of rank placement.
 */

constexpr int BUCKET_BLOCKS = kBucketBlocks;  // chunking blocks for pass A (2/CU); scratch sizing uses the same constant
constexpr int BUCKET_THREADS = 1024;
constexpr int JOIN_LDS_SLOTS = 2048;        // 32 KiB of longlong2 pairs
constexpr int JOIN_STAGE_ROWS = 1024;       // 32 KiB staged output rows (4 x i64)
/* table + stage = 64 KiB + 16 B -> 2 blocks/CU */
constexpr int SUB_BUCKETS = 256;            // pass-B fanout (fixed)

int bucket_count_for(int64_t ln, int64_t rn)
{
  int64_t maxn = ln > rn ? ln : rn;
  int64_t bmax = 1048576;
  if (const char* e = getenv("DJ_MAX_B")) {
    /* TEST HOOK: cap B to trade pass-B flush-run length (F = B/PA) against
     * join table size — e.g. 800M at B=524288 runs lam~1526 on the
     * 4096-slot join but doubles pass-B's per-bucket run length */
    int64_t v = atoll(e);
    if (v >= 256 && v <= 1048576 && (v & (v - 1)) == 0) bmax = v;
  }
  int64_t B = 256;
  while (B < bmax && maxn / B > 800) B <<= 1;
  return (int)B;
}

/* bucket id = groupA * F + subF: pass-A group from mix64 bits [40,50),
 * pass-B sub-bucket from bits [32,40) (+ [50,60) when F > 256, subF_of) */
__device__ __forceinline__ uint32_t groupA_of(int64_t key, int PA)
{
  return (uint32_t)(dj_mix64((uint64_t)key) >> 40) & (uint32_t)(PA - 1);
}
__device__ __forceinline__ uint32_t subB_of(int64_t key)
{
  return (uint32_t)(dj_mix64((uint64_t)key) >> 32) & (uint32_t)(SUB_BUCKETS - 1);
}
/* runtime-F sub-bucket (fused wire path): low 8 bits from [32,40) as subB_of
 * (identical placement for F <= 256); F > 256 borrows its extra bits from
 * [50,60) — DISJOINT from the pass-A group bits [40,50) (overlapping fields
 * would collapse the effective fan-out: with group bits fixed inside a
 * pass-A group, any shared bit halves the distinct sub-buckets) and from
 * the LDS slot hash bits [0,11). */
__device__ __forceinline__ uint32_t subF_of(int64_t key, int F)
{
  uint64_t m = dj_mix64((uint64_t)key);
  if (F <= 256) return (uint32_t)(m >> 32) & (uint32_t)(F - 1);
  uint32_t lo = (uint32_t)(m >> 32) & 255u;
  uint32_t hi = (uint32_t)(m >> 50) & (uint32_t)((F >> 8) - 1);
  return lo | (hi << 8);
}

/* ---- pass A kernels (P groups, P <= 1024) ---- */

__global__ __launch_bounds__(BUCKET_THREADS) void bucket_count_kernel(
  const int64_t* __restrict__ keys, int64_t n, int P, uint32_t* __restrict__ counts)
{
  extern __shared__ uint32_t hist[];
  for (int p = threadIdx.x; p < P; p += blockDim.x) hist[p] = 0;
  __syncthreads();
  const int64_t chunk = (n + gridDim.x - 1) / gridDim.x;
  const int64_t start = (int64_t)blockIdx.x * chunk;
  const int64_t end = min(start + chunk, n);
  for (int64_t i = start + threadIdx.x; i < end; i += blockDim.x)
    atomicAdd(&hist[groupA_of(nt_load(&keys[i]), P)], 1u);
  __syncthreads();
  for (int p = threadIdx.x; p < P; p += blockDim.x)
    counts[(size_t)blockIdx.x * P + p] = hist[p];
}

/* grid = P blocks x BUCKET_BLOCKS threads: exclusive scan of counts[:, p]
 * over blocks (in place), totals[p] = column sum */
__global__ void bucket_scanA_kernel(uint32_t* counts, int P, uint32_t* totals)
{
  const int p = blockIdx.x;
  __shared__ uint32_t sh[BUCKET_BLOCKS];
  uint32_t v = counts[(size_t)threadIdx.x * P + p];
  sh[threadIdx.x] = v;
  __syncthreads();
  for (int off = 1; off < BUCKET_BLOCKS; off <<= 1) {
    uint32_t add = (threadIdx.x >= (unsigned)off) ? sh[threadIdx.x - off] : 0;
    __syncthreads();
    sh[threadIdx.x] += add;
    __syncthreads();
  }
  counts[(size_t)threadIdx.x * P + p] = sh[threadIdx.x] - v;  // exclusive
  if (threadIdx.x == BUCKET_BLOCKS - 1) totals[p] = sh[threadIdx.x];
}

/* single block: exclusive scan of totals[P] -> segoff[P+1] (int64) */
__global__ void bucket_scanB_kernel(const uint32_t* totals, int P, int64_t* segoff)
{
  __shared__ int64_t sh[BUCKET_THREADS];
  __shared__ int64_t running_sh;
  if (threadIdx.x == 0) running_sh = 0;
  __syncthreads();
  for (int base = 0; base < P; base += BUCKET_THREADS) {
    int p = base + threadIdx.x;
    int64_t v = (p < P) ? (int64_t)totals[p] : 0;
    sh[threadIdx.x] = v;
    __syncthreads();
    for (int off = 1; off < BUCKET_THREADS; off <<= 1) {
      int64_t add = (threadIdx.x >= (unsigned)off) ? sh[threadIdx.x - off] : 0;
      __syncthreads();
      sh[threadIdx.x] += add;
      __syncthreads();
    }
    int64_t rbase = running_sh;
    __syncthreads();  // all reads of running_sh precede the update below
    if (p < P) segoff[p] = rbase + sh[threadIdx.x] - v;
    if (threadIdx.x == BUCKET_THREADS - 1) running_sh = rbase + sh[threadIdx.x];
    __syncthreads();
  }
  if (threadIdx.x == 0) segoff[P] = running_sh;
}

/* Tile-staged scatter: rows of a 4096-row tile are ranked per group in LDS
 * (hist atomics), staged grouped into an LDS tile buffer, then each group's
 * run is flushed linearly — global writes coalesce into ~cnt[g]-row runs
 * (full 128 B lines) instead of 64 scattered 16 B transactions per wave.
 * The direct-scatter variant measured 59% SQ issue-stall on store-pipe
 * pressure; staging trades cheap LDS traffic for it. Requires P <= 1024
 * (= BUCKET_THREADS: one group per thread in the hist scan). */
constexpr int SCATTER_TILE = 4096;  // 64 KiB of staged pairs

template <bool SINGLE_LEVEL>
__device__ __forceinline__ longlong2 load_row(const int64_t* keys, const int64_t* pay,
                                              const longlong2* pairs, int64_t i)
{
  longlong2 r;
  if (SINGLE_LEVEL) {
    r.x = nt_load(&keys[i]);
    r.y = pay ? nt_load(&pay[i]) : i;
  } else {
    r = nt_load2(&pairs[i]);
  }
  return r;
}

/* staged scatter over [start, end); cur[P] holds GLOBAL destination
 * cursors; GROUP_FN: 0 = groupA_of(P), 1 = subF_of(P) */
template <int GROUP_FN, bool SINGLE_LEVEL>
__device__ void staged_scatter_span(const int64_t* keys, const int64_t* pay,
                                    const longlong2* in_pairs, int64_t start, int64_t end,
                                    int P, longlong2* tbuf, uint32_t* hist, uint32_t* base,
                                    uint32_t* gcur, longlong2* out_pairs)
/* GROUP_FN 0: groupA_of(P); GROUP_FN 1: subF_of(P) (P is the runtime
 * fanout; bit fields disjoint from groupA's, see subF_of) */
{
  constexpr int VPT = SCATTER_TILE / BUCKET_THREADS;  // 4
  const int tid = threadIdx.x;
  for (int64_t t0 = start; t0 < end; t0 += SCATTER_TILE) {
    const int count = (int)min((int64_t)SCATTER_TILE, end - t0);
    if (tid < P) hist[tid] = 0;
    __syncthreads();
    /* compile-time-bounded, predicated loops: a runtime trip count would
     * force r/g/rank into scratch (the arrays must stay in VGPRs — spilled
     * staging re-reads cost more HBM traffic than the payload itself) */
    longlong2 r[VPT];
    uint32_t g[VPT], rank[VPT];
#pragma unroll
    for (int v = 0; v < VPT; v++) {
      int64_t i = t0 + (int64_t)v * blockDim.x + tid;
      if (i < end) {
        r[v] = load_row<SINGLE_LEVEL>(keys, pay, in_pairs, i);
        g[v] = GROUP_FN == 0 ? groupA_of(r[v].x, P) : subF_of(r[v].x, P);
        rank[v] = atomicAdd(&hist[g[v]], 1u);
      }
    }
    __syncthreads();
    /* exclusive scan of hist -> base (P <= blockDim: one group per thread) */
    if (tid < P) base[tid] = hist[tid];
    __syncthreads();
    for (int off = 1; off < P; off <<= 1) {
      uint32_t add = (tid < P && tid >= off) ? base[tid - off] : 0;
      __syncthreads();
      if (tid < P) base[tid] += add;
      __syncthreads();
    }
    if (tid < P) base[tid] -= hist[tid];  // inclusive -> exclusive
    __syncthreads();
#pragma unroll
    for (int v = 0; v < VPT; v++) {
      int64_t i = t0 + (int64_t)v * blockDim.x + tid;
      if (i < end) tbuf[base[g[v]] + rank[v]] = r[v];
    }
    __syncthreads();
    /* flush linearly: per-group runs coalesce into full lines */
    for (int pos = tid; pos < count; pos += blockDim.x) {
      longlong2 row = tbuf[pos];
      uint32_t gg = GROUP_FN == 0 ? groupA_of(row.x, P) : subF_of(row.x, P);
      out_pairs[gcur[gg] + (pos - base[gg])] = row;
    }
    __syncthreads();
    if (tid < P) gcur[tid] += hist[tid];
    __syncthreads();
  }
}

__global__ __launch_bounds__(BUCKET_THREADS) void bucket_scatter_kernel(
  const int64_t* __restrict__ keys, const int64_t* __restrict__ pay, int64_t n, int P,
  const uint32_t* __restrict__ counts, const int64_t* __restrict__ segoff,
  longlong2* __restrict__ out_pairs)
{
  extern __shared__ __attribute__((aligned(16))) char smem[];
  longlong2* tbuf = (longlong2*)smem;
  uint32_t* hist = (uint32_t*)(tbuf + SCATTER_TILE);
  uint32_t* base = hist + P;
  uint32_t* gcur = base + P;
  if (threadIdx.x < P)
    gcur[threadIdx.x] = (uint32_t)segoff[threadIdx.x] +
                        counts[(size_t)blockIdx.x * P + threadIdx.x];
  __syncthreads();
  const int64_t chunk = (n + gridDim.x - 1) / gridDim.x;
  const int64_t start = (int64_t)blockIdx.x * chunk;
  const int64_t end = min(start + chunk, n);
  staged_scatter_span<0, true>(keys, pay, nullptr, start, end, P, tbuf, hist, base, gcur,
                               out_pairs);
}

/* trivial segoff = {0, n} for the single-group (B == 256) case */
__global__ void set_segoff1_kernel(int64_t* segoff, int64_t n)
{
  segoff[0] = 0;
  segoff[1] = n;
}

/* ---- pass B: one block per pass-A group; F sub-buckets in-block ---- */
/* pass B: one block per pass-A group; F sub-buckets (runtime 256/512/1024,
 * subF_of bit fields disjoint from the pass-A group bits), tile-staged
 * scatter. single_level: input is the original two column arrays (the
 * B == 256 single-pass case). */
template <bool SINGLE_LEVEL>
__global__ __launch_bounds__(BUCKET_THREADS) void bucket_subpart_kernel(
  const longlong2* __restrict__ in_pairs, const int64_t* __restrict__ keys,
  const int64_t* __restrict__ pay, const int64_t* __restrict__ segoff, int B, int F,
  longlong2* __restrict__ out_pairs, int64_t* __restrict__ bucket_offsets /* B+1 */)
{
  extern __shared__ __attribute__((aligned(16))) char smem[];
  longlong2* tbuf = (longlong2*)smem;
  uint32_t* hist = (uint32_t*)(tbuf + SCATTER_TILE);
  uint32_t* base = hist + F;
  uint32_t* gcur = base + F;
  uint32_t* seghist = gcur + F;
  const int tid = threadIdx.x;
  const int a = blockIdx.x;
  const int64_t s0 = segoff[a], s1 = segoff[a + 1];
  if (tid < F) seghist[tid] = 0;
  __syncthreads();
  for (int64_t i = s0 + tid; i < s1; i += blockDim.x) {
    int64_t k = SINGLE_LEVEL ? nt_load(&keys[i]) : nt_load(&in_pairs[i].x);
    atomicAdd(&seghist[subF_of(k, F)], 1u);
  }
  __syncthreads();
  if (tid == 0) {
    uint32_t acc = 0;
    for (int j = 0; j < F; j++) {
      uint32_t c = seghist[j];
      gcur[j] = (uint32_t)s0 + acc;
      bucket_offsets[(size_t)a * F + j] = s0 + acc;
      acc += c;
    }
    if (a == gridDim.x - 1) bucket_offsets[B] = s1;
  }
  __syncthreads();
  staged_scatter_span<1, SINGLE_LEVEL>(keys, pay, in_pairs, s0, s1, F, tbuf, hist, base, gcur,
                                       out_pairs);
}

/* pass B over per-peer segment lists (fused wire path): block = one pass-A
 * group g, whose rows arrived pre-grouped inside each peer slice; the
 * block's sub-buckets fan out by F (runtime power of two). */
__global__ __launch_bounds__(BUCKET_THREADS) void subpart_lists_kernel(
  const int64_t* __restrict__ keys, const int64_t* __restrict__ pay,
  const int64_t* __restrict__ seg_bounds /* [nseg][PA+1] absolute row offsets */, int nseg,
  int PA, int F, const int64_t* __restrict__ group_base /* [PA+1] output bases */,
  longlong2* __restrict__ out_pairs, int64_t* __restrict__ bucket_offsets /* PA*F+1 */)
{
  extern __shared__ __attribute__((aligned(16))) char smem[];
  longlong2* tbuf = (longlong2*)smem;
  uint32_t* hist = (uint32_t*)(tbuf + SCATTER_TILE);
  uint32_t* base = hist + F;
  uint32_t* gcur = base + F;
  uint32_t* seghist = gcur + F;
  const int tid = threadIdx.x;
  const int g = blockIdx.x;
  for (int j = tid; j < F; j += blockDim.x) seghist[j] = 0;
  __syncthreads();
  for (int sgi = 0; sgi < nseg; sgi++) {
    const int64_t s0 = seg_bounds[(size_t)sgi * (PA + 1) + g];
    const int64_t s1 = seg_bounds[(size_t)sgi * (PA + 1) + g + 1];
    for (int64_t i = s0 + tid; i < s1; i += blockDim.x)
      atomicAdd(&seghist[subF_of(nt_load(&keys[i]), F)], 1u);
  }
  __syncthreads();
  if (tid == 0) {
    uint32_t acc = 0;
    const int64_t gb = group_base[g];
    for (int j = 0; j < F; j++) {
      uint32_t c = seghist[j];
      gcur[j] = (uint32_t)(gb + acc);
      bucket_offsets[(size_t)g * F + j] = gb + acc;
      acc += c;
    }
    if (g == gridDim.x - 1) bucket_offsets[(size_t)PA * F] = group_base[PA];
  }
  __syncthreads();
  for (int sgi = 0; sgi < nseg; sgi++) {
    const int64_t s0 = seg_bounds[(size_t)sgi * (PA + 1) + g];
    const int64_t s1 = seg_bounds[(size_t)sgi * (PA + 1) + g + 1];
    staged_scatter_span<1, true>(keys, pay, nullptr, s0, s1, F, tbuf, hist, base, gcur,
                                 out_pairs);
  }
}

void subpart_lists(const int64_t* d_keys, const int64_t* d_pay, const int64_t* d_seg_bounds,
                   int nseg, int PA, int F, const int64_t* d_group_base,
                   longlong2* d_out_pairs, int64_t* d_bucket_offsets, hipStream_t s)
{
  DJ_CHECK_ERROR(F >= 64 && F <= 1024 && (F & (F - 1)) == 0,
                 "subpart_lists: F must be a power of two in [64,1024]");
  size_t lds = SCATTER_TILE * sizeof(longlong2) + 4 * (size_t)F * 4;
  hipLaunchKernelGGL(subpart_lists_kernel, dim3(PA), dim3(BUCKET_THREADS), lds, s, d_keys,
                     d_pay, d_seg_bounds, nseg, PA, F, d_group_base, d_out_pairs,
                     d_bucket_offsets);
  DJ_HIP_CALL(hipGetLastError());
}

/* ---- fused per-bucket LDS build + single-pass probe with staged output ----
 * The probe emits matches into an LDS output stage (SoA, 4 x i64 per row) and
 * flushes once per bucket with ONE global counter atomic + coalesced column
 * writes. (The earlier two-phase count-then-rewalk design paid a full extra
 * probe walk — ~0.8 ms at 100 M rows, profiles/r01_ablation.txt.) Buckets
 * whose matches exceed the stage spill directly to the global counter per
 * lane — correct for any duplication factor, slower only on such buckets. */
/* SLOTS2 = 2048 (2 blocks/CU, buckets <= 1536 build rows — the local-join
 * shape) or 4096 (1 block/CU, buckets <= 3072 — used by the G>1 fused wire
 * path when its PA*F fan-out cap leaves ~1.5-3k rows/bucket, e.g. G=8 od=1;
 * measured 1.98 vs 1.67 ms at the same total rows,
 * experiments/join_layout.hip — far cheaper than whole-batch redos).
 *
 * Round 2: each block processes KBUK=4 buckets per flush group with a
 * watermark — the per-bucket global reserve atomic was the largest marginal
 * cost of the r1 kernel (profiles/r01_ablation.txt: reserve+flush 0.55 ms at
 * 131072 buckets; a single hot counter absorbs one wave-op per ~12 ns);
 * accumulating the stage across buckets and flushing at the 3/4 watermark
 * (or group end) cuts the atomics ~4x: 1.69 -> 1.35 ms measured
 * (experiments/join_v2/v4). Stage spills (high-duplication buckets, or a
 * bucket landing on a nearly-full stage at high selectivity) go straight to
 * the global counter (plain per-match atomic: the compiler wave-aggregates
 * it; a hand-ballot version of this spill path — even though it never
 * executes at sel 0.3 — cost 1.0 ms of probe-walk codegen, 2.42 vs 1.41 ms,
 * experiments/join_v6).
 *
 * Bucket input layout: SLACK=false — compact offsets (loff/roff, B+1);
 * SLACK=true — the slack layout of bucket_partition2_slack
 * (b*capL + llen[b]). Compile-time split for the same reason.
 *
 * KBUK > 1 contract: B MUST be a multiple of KBUK with llen/rlen (or
 * loff/roff) valid over all of it — callers pad with zero-length buckets
 * (lds_join_slack checks). Padding keeps every flush-slot comparison
 * against the constant KBUK-1: the b >= B tail break, a dynamic flush-slot
 * register, and an in-range guard each put the kernel back at ~2.35 ms
 * (experiments/join_v7 factorial; the goto-drain structure below measured
 * 1.40 ms). KBUK == 1 (only for a compact B that is not a multiple of 4 —
 * no current caller produces one) flushes every bucket — the r1 behavior,
 * ~0.3 ms slower per 100 M rows but contract-free. */
template <int SLOTS2, bool SLACK, int KBUK>
__global__ __launch_bounds__(BUCKET_THREADS) void lds_join_kernel(
  const longlong2* __restrict__ lrows, const int64_t* __restrict__ loff,
  const uint32_t* __restrict__ llen, int64_t capL, const longlong2* __restrict__ rrows,
  const int64_t* __restrict__ roff, const uint32_t* __restrict__ rlen, int64_t capR, int B,
  int64_t* __restrict__ out0, int64_t* __restrict__ out1, int64_t* __restrict__ out2,
  int64_t* __restrict__ out3, int64_t cap, unsigned long long* counter,
  uint32_t* __restrict__ overflow_flags, int* __restrict__ any_overflow,
  int* __restrict__ error)
{
  constexpr int S = JOIN_STAGE_ROWS;
  constexpr int WATER = S - S / 4;
  extern __shared__ __attribute__((aligned(16))) char smem[];
  longlong2* tbl = (longlong2*)smem;
  int64_t* stage = (int64_t*)(tbl + SLOTS2);  // SoA: stage[c*JOIN_STAGE_ROWS + i]
  long long* base_sh = (long long*)(stage + 4 * S);
  uint32_t* cur_sh = (uint32_t*)(base_sh + 1);
  const uint32_t smask = SLOTS2 - 1;
  if (threadIdx.x == 0) *cur_sh = 0;
  __syncthreads();

  for (int bb = blockIdx.x * KBUK; bb < B; bb += gridDim.x * KBUK) {
    /* Exit-clean guarantee: every group's k == KBUK-1 slot flushes — valid
     * buckets flush unconditionally there, and invalid slots (empty side,
     * skew-routed, or caller-padded zero-length buckets) jump to the flush
     * block instead of skipping it, so a block never exits carrying staged
     * rows. */
    for (int k = 0; k < KBUK; k++) {
      const int b = bb + k;
      if (b >= B) break;  // dead when KBUK > 1 (padded B); real for KBUK == 1
      int64_t l0, l1, r0, r1;
      if (SLACK) {
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
      if (lnb == 0 || r1 == r0) {  // empty side (incl. padding buckets)
        /* KBUK == 1 flushes every valid bucket, so there is never a carry
         * to drain — plain skip (constant-folded) */
        if (KBUK == 1 || k != KBUK - 1) continue;
        goto flush;  // drain any carried stage at the group's flush slot
      }
      if (lnb > SLOTS2 * 3 / 4) {  // skew: host-side fallback joins it
        if (threadIdx.x == 0) {
          overflow_flags[b] = 1;
          atomicOr(any_overflow, 1);  // bit 2 is the slack-partition overflow
        }
        if (KBUK == 1 || k != KBUK - 1) continue;
        goto flush;
      }
      for (int s = threadIdx.x; s < SLOTS2; s += blockDim.x) tbl[s].x = kEmptyKey;
      __syncthreads();
      /* build */
      for (int64_t i = l0 + threadIdx.x; i < l1; i += blockDim.x) {
        longlong2 row = lrows[i];
        if (row.x == kEmptyKey) {
          *error = 1;
          continue;
        }
        uint32_t slot = (uint32_t)dj_mix64((uint64_t)row.x) & smask;
        for (;;) {
          unsigned long long old = atomicCAS((unsigned long long*)&tbl[slot].x,
                                             (unsigned long long)kEmptyKey,
                                             (unsigned long long)row.x);
          if (old == (unsigned long long)kEmptyKey) break;
          slot = (slot + 1) & smask;
        }
        tbl[slot].y = row.y;
      }
      __syncthreads();
      /* single-pass probe: matches append to the LDS stage */
      for (int64_t j = r0 + threadIdx.x; j < r1; j += blockDim.x) {
        longlong2 prow = rrows[j];
        uint32_t slot = (uint32_t)dj_mix64((uint64_t)prow.x) & smask;
        for (;;) {
          longlong2 e = tbl[slot];
          if (e.x == kEmptyKey) break;
          if (e.x == prow.x) {
            uint32_t pos = atomicAdd(cur_sh, 1u);
            if (pos < (uint32_t)S) {
              stage[0 * S + pos] = prow.x;
              stage[1 * S + pos] = e.y;
              stage[2 * S + pos] = prow.x;
              stage[3 * S + pos] = prow.y;
            } else {
              /* stage overflow (heavy duplication): spill directly; the
               * compiler wave-aggregates the counter atomic */
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
    flush:
      __syncthreads();
      /* watermark: flush early so the stage never spills structurally;
       * unconditional at k == KBUK-1 (the block's exit-clean guarantee) */
      if (k < KBUK - 1 && *cur_sh < (uint32_t)WATER) continue;
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
      if (threadIdx.x == 0) *cur_sh = 0;
      __syncthreads();
    }
  }
}

/* ----------------- fused rank+group partition (fast2 wire path) -----------
 * One staged scatter replaces the stable rank partition AND bucket pass A:
 * partition id = p * PA + groupA, where p = murmur(key) %% (G*od) is the
 * reference's rank/batch partition (distributed_join.cpp:211-226 semantics:
 * batch = p / G, rank = p %% G) and groupA = high mix64 bits. Rank/batch
 * slices stay contiguous for the exchange; each peer slice arrives grouped
 * by groupA, so the local join needs only pass B (over per-peer segment
 * lists) + the fused LDS join. Row order within a slice is NOT stable —
 * internal wire layout only (the drop-in hash_partition C ABI stays
 * stable). Outputs are columnar (key col + payload col) for the existing
 * column-wise exchange and compression paths. */

__device__ __forceinline__ uint32_t fused_pid(int64_t key, int nparts_rank, uint32_t seed,
                                              int PA)
{
  uint32_t p = dj_murmur3_int64(key, seed) % (uint32_t)nparts_rank;
  uint32_t g = (uint32_t)(dj_mix64((uint64_t)key) >> 40) & (uint32_t)(PA - 1);
  return p * (uint32_t)PA + g;
}

__global__ __launch_bounds__(BUCKET_THREADS) void fused_count_kernel(
  const int64_t* __restrict__ keys, int64_t n, int nparts_rank, uint32_t seed, int PA,
  uint32_t* __restrict__ counts)
{
  extern __shared__ uint32_t hist[];
  const int P = nparts_rank * PA;
  for (int p = threadIdx.x; p < P; p += blockDim.x) hist[p] = 0;
  __syncthreads();
  const int64_t chunk = (n + gridDim.x - 1) / gridDim.x;
  const int64_t start = (int64_t)blockIdx.x * chunk;
  const int64_t end = min(start + chunk, n);
  for (int64_t i = start + threadIdx.x; i < end; i += blockDim.x)
    atomicAdd(&hist[fused_pid(nt_load(&keys[i]), nparts_rank, seed, PA)], 1u);
  __syncthreads();
  for (int p = threadIdx.x; p < P; p += blockDim.x)
    counts[(size_t)blockIdx.x * P + p] = hist[p];
}

/* staged scatter with columnar outputs (stage pairs, flush per column) */
__global__ __launch_bounds__(BUCKET_THREADS) void fused_scatter_kernel(
  const int64_t* __restrict__ keys, const int64_t* __restrict__ pay, int64_t n,
  int nparts_rank, uint32_t seed, int PA, const uint32_t* __restrict__ counts,
  const int64_t* __restrict__ offsets, int64_t* __restrict__ out_keys,
  int64_t* __restrict__ out_pay)
{
  extern __shared__ __attribute__((aligned(16))) char smem[];
  longlong2* tbuf = (longlong2*)smem;
  const int P = nparts_rank * PA;
  uint32_t* hist = (uint32_t*)(tbuf + SCATTER_TILE);
  uint32_t* base = hist + P;
  uint32_t* gcur = base + P;
  const int tid = threadIdx.x;
  if (tid < P)
    gcur[tid] = (uint32_t)offsets[tid] + counts[(size_t)blockIdx.x * P + tid];
  __syncthreads();
  const int64_t chunk = (n + gridDim.x - 1) / gridDim.x;
  const int64_t start = (int64_t)blockIdx.x * chunk;
  const int64_t end = min(start + chunk, n);
  constexpr int VPT = SCATTER_TILE / BUCKET_THREADS;
  for (int64_t t0 = start; t0 < end; t0 += SCATTER_TILE) {
    const int count = (int)min((int64_t)SCATTER_TILE, end - t0);
    if (tid < P) hist[tid] = 0;
    __syncthreads();
    longlong2 r[VPT];
    uint32_t g[VPT], rank[VPT];
#pragma unroll
    for (int v = 0; v < VPT; v++) {
      int64_t i = t0 + (int64_t)v * blockDim.x + tid;
      if (i < end) {
        r[v].x = nt_load(&keys[i]);
        r[v].y = pay ? nt_load(&pay[i]) : i;
        g[v] = fused_pid(r[v].x, nparts_rank, seed, PA);
        rank[v] = atomicAdd(&hist[g[v]], 1u);
      }
    }
    __syncthreads();
    if (tid < P) base[tid] = hist[tid];
    __syncthreads();
    for (int off = 1; off < P; off <<= 1) {
      uint32_t add = (tid < P && tid >= off) ? base[tid - off] : 0;
      __syncthreads();
      if (tid < P) base[tid] += add;
      __syncthreads();
    }
    if (tid < P) base[tid] -= hist[tid];
    __syncthreads();
#pragma unroll
    for (int v = 0; v < VPT; v++) {
      int64_t i = t0 + (int64_t)v * blockDim.x + tid;
      if (i < end) tbuf[base[g[v]] + rank[v]] = r[v];
    }
    __syncthreads();
    for (int pos = tid; pos < count; pos += blockDim.x) {
      longlong2 row = tbuf[pos];
      uint32_t gg = fused_pid(row.x, nparts_rank, seed, PA);
      uint32_t dst = gcur[gg] + (uint32_t)(pos - base[gg]);
      out_keys[dst] = row.x;
      out_pay[dst] = row.y;
    }
    __syncthreads();
    if (tid < P) gcur[tid] += hist[tid];
    __syncthreads();
  }
}

void fused_partition(const int64_t* d_keys, const int64_t* d_pay, int64_t n, int nparts_rank,
                     uint32_t seed, int PA, uint32_t* d_counts, uint32_t* d_totals,
                     int64_t* d_offsets, int64_t* d_out_keys, int64_t* d_out_pay,
                     hipStream_t s)
{
  const int P = nparts_rank * PA;
  DJ_CHECK_ERROR(P >= 1 && P <= 1024, "fused_partition: nparts_rank*PA must be <= 1024");
  DJ_CHECK_ERROR(n < (int64_t)UINT32_MAX, "fused_partition: n must be < 2^32");
  size_t hist_lds = (size_t)P * 4;
  size_t scatter_lds = SCATTER_TILE * sizeof(longlong2) + 3 * hist_lds;
  hipLaunchKernelGGL(fused_count_kernel, dim3(BUCKET_BLOCKS), dim3(BUCKET_THREADS), hist_lds,
                     s, d_keys, n, nparts_rank, seed, PA, d_counts);
  DJ_HIP_CALL(hipGetLastError());
  hipLaunchKernelGGL(bucket_scanA_kernel, dim3(P), dim3(BUCKET_BLOCKS), 0, s, d_counts, P,
                     d_totals);
  DJ_HIP_CALL(hipGetLastError());
  hipLaunchKernelGGL(bucket_scanB_kernel, dim3(1), dim3(BUCKET_THREADS), 0, s, d_totals, P,
                     d_offsets);
  DJ_HIP_CALL(hipGetLastError());
  hipLaunchKernelGGL(fused_scatter_kernel, dim3(BUCKET_BLOCKS), dim3(BUCKET_THREADS),
                     scatter_lds, s, d_keys, d_pay, n, nparts_rank, seed, PA, d_counts,
                     d_offsets, d_out_keys, d_out_pay);
  DJ_HIP_CALL(hipGetLastError());
}

/* ---- slack pass A (local non-stable partition): per-tile global atomic
 * cursors into slack group segments (start g*capA), eliminating the separate
 * count+scan pass — the local join's row order is unspecified, so stability
 * is not required here (the wire path keeps the exact-count stable scatter).
 * The 8192-row staging tile (137 KB LDS, 1 block/CU) halves the partial-line
 * write amplification vs 4096 (experiments/scatter_sweep.hip: 1.105 ms vs
 * 1.255 + 0.16 count at 100 M rows). Group overflow — skew beyond the ~6%
 * slack — SKIPS rows and sets bit 2 of any_overflow; the caller must then
 * redo the whole join (the partition output is incomplete). */
constexpr int SLACK_TILE = 8192;

/* block-level exclusive scan of hist[0..P) into base[0..P) via per-wave
 * shfl scans + one cross-wave partial pass (2 barriers; the Hillis-Steele
 * LDS scan costs 2*log2(P) barriers). partials: u32[16] LDS. */
__device__ __forceinline__ void wave_excl_scan(const uint32_t* hist, uint32_t* base,
                                               uint32_t* partials, int P)
{
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wid = tid >> 6;
  uint32_t v = (tid < P) ? hist[tid] : 0;
  uint32_t incl = v;
#pragma unroll
  for (int off = 1; off < WAVE; off <<= 1) {
    uint32_t up = __shfl_up(incl, off);
    if (lane >= off) incl += up;
  }
  if (lane == WAVE - 1 && wid < (P + WAVE - 1) / WAVE) partials[wid] = incl;
  __syncthreads();
  if (wid == 0) {
    const int nw = (P + WAVE - 1) / WAVE;
    uint32_t pv = (lane < nw) ? partials[lane] : 0;
    uint32_t pincl = pv;
#pragma unroll
    for (int off = 1; off < WAVE; off <<= 1) {
      uint32_t up = __shfl_up(pincl, off);
      if (lane >= off) pincl += up;
    }
    if (lane < nw) partials[lane] = pincl - pv;
  }
  __syncthreads();
  if (tid < P) base[tid] = incl - v + partials[wid];
}

/* Round 2: software-pipelined — the next tile's 8 nontemporal loads per
 * lane issue BEFORE the current tile's flush, so the HBM load latency hides
 * under the store burst (this kernel runs 1 block/CU, so barriered phases
 * cannot overlap across blocks; measured 0.90 -> ~0.86 ms/table with the
 * wave scans, experiments/join_v4). */
__device__ __forceinline__ void slackA_body(
  const int64_t* __restrict__ keys, const int64_t* __restrict__ pay, int64_t n, int P,
  int64_t capA, uint32_t* __restrict__ gcursor, int* __restrict__ any_overflow,
  longlong2* __restrict__ out_pairs, int bid, int nblocks)
{
  constexpr int VPT = SLACK_TILE / BUCKET_THREADS;
  extern __shared__ __attribute__((aligned(16))) char smem[];
  longlong2* tbuf = (longlong2*)smem;
  uint32_t* hist = (uint32_t*)(tbuf + SLACK_TILE);
  uint32_t* base = hist + P;
  uint32_t* gcur = base + P;
  uint32_t* glim = gcur + P;  // per-group segment end (precomputed: the
                              // per-row bound check must not pay a 64-bit
                              // multiply in the flush loop)
  uint32_t* partials = glim + P;  // u32[16]
  __shared__ int s_ovf;
  const int tid = threadIdx.x;
  if (tid == 0) s_ovf = 0;
  if (tid < P) hist[tid] = 0;
  const int64_t chunk = (n + nblocks - 1) / nblocks;
  const int64_t start = (int64_t)bid * chunk;
  const int64_t end = min(start + chunk, n);
  if (start >= end) return;
  __syncthreads();

  longlong2 r[VPT];
  uint32_t g[VPT], rank[VPT];
#pragma unroll
  for (int v = 0; v < VPT; v++) {
    int64_t i = start + (int64_t)v * BUCKET_THREADS + tid;
    if (i < end) {
      r[v].x = nt_load(&keys[i]);
      r[v].y = pay ? nt_load(&pay[i]) : i;
      g[v] = groupA_of(r[v].x, P);
      rank[v] = atomicAdd(&hist[g[v]], 1u);
    }
  }
  __syncthreads();

  for (int64_t t0 = start; t0 < end; t0 += SLACK_TILE) {
    const int count = (int)min((int64_t)SLACK_TILE, end - t0);
    const int64_t t1 = t0 + SLACK_TILE;
    wave_excl_scan(hist, base, partials, P);
    /* claim this tile's run in each group's slack segment */
    if (tid < P) {
      const int64_t seg0 = (int64_t)tid * capA;
      gcur[tid] = (uint32_t)seg0 + atomicAdd(&gcursor[tid], hist[tid]);
      glim[tid] = (uint32_t)(seg0 + capA);
    }
    __syncthreads();
#pragma unroll
    for (int v = 0; v < VPT; v++) {
      int64_t i = t0 + (int64_t)v * BUCKET_THREADS + tid;
      if (i < end) tbuf[base[g[v]] + rank[v]] = r[v];
    }
    __syncthreads();
    if (tid < P) hist[tid] = 0; /* free after the claim; re-ranked after flush */
    /* issue the next tile's loads BEFORE the flush */
    longlong2 r2[VPT];
#pragma unroll
    for (int v = 0; v < VPT; v++) {
      int64_t i = t1 + (int64_t)v * BUCKET_THREADS + tid;
      if (i < end) {
        r2[v].x = nt_load(&keys[i]);
        r2[v].y = pay ? nt_load(&pay[i]) : i;
      }
    }
    for (int pos = tid; pos < count; pos += blockDim.x) {
      longlong2 row = tbuf[pos];
      uint32_t gg = groupA_of(row.x, P);
      uint32_t dst = gcur[gg] + (uint32_t)(pos - base[gg]);
      if (dst < glim[gg])
        out_pairs[dst] = row;
      else
        s_ovf = 1;  // benign LDS race: any 1 wins
    }
    __syncthreads();
#pragma unroll
    for (int v = 0; v < VPT; v++) {
      int64_t i = t1 + (int64_t)v * BUCKET_THREADS + tid;
      if (i < end) {
        r[v] = r2[v];
        g[v] = groupA_of(r[v].x, P);
        rank[v] = atomicAdd(&hist[g[v]], 1u);
      }
    }
    __syncthreads();
  }
  __syncthreads();
  if (tid == 0 && s_ovf) atomicOr(any_overflow, 2);
}

__global__ __launch_bounds__(BUCKET_THREADS) void bucket_scatter_slack_kernel(
  const int64_t* __restrict__ keys, const int64_t* __restrict__ pay, int64_t n, int P,
  int64_t capA, uint32_t* __restrict__ gcursor, int* __restrict__ any_overflow,
  longlong2* __restrict__ out_pairs)
{
  slackA_body(keys, pay, n, P, capA, gcursor, any_overflow, out_pairs, blockIdx.x,
              gridDim.x);
}

/* both tables in ONE launch (halves the pass-A launches and lets the second
 * table's blocks fill the first's tail wave — 1 block/CU at 512 blocks means
 * two sequential waves per launch otherwise) */
__global__ __launch_bounds__(BUCKET_THREADS) void bucket_scatter_slack_pair_kernel(
  const int64_t* __restrict__ keys0, const int64_t* __restrict__ pay0, int64_t n0,
  int64_t capA0, uint32_t* __restrict__ gcursor0, longlong2* __restrict__ out0,
  const int64_t* __restrict__ keys1, const int64_t* __restrict__ pay1, int64_t n1,
  int64_t capA1, uint32_t* __restrict__ gcursor1, longlong2* __restrict__ out1, int P,
  int blocks0 /* grid split point, proportional to n0/(n0+n1) */,
  int* __restrict__ any_overflow)
{
  if ((int)blockIdx.x < blocks0)
    slackA_body(keys0, pay0, n0, P, capA0, gcursor0, any_overflow, out0, blockIdx.x,
                blocks0);
  else
    slackA_body(keys1, pay1, n1, P, capA1, gcursor1, any_overflow, out1,
                blockIdx.x - blocks0, gridDim.x - blocks0);
}

/* on slack overflow the atomic cursor kept counting skipped rows; clamp the
 * lengths to capA so pass B stays in bounds (results are discarded — the
 * caller redoes the join on bit 2) */
__global__ void clamp_seglen_kernel(uint32_t* seg_len, int P, uint32_t capA)
{
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i < P && seg_len[i] > capA) seg_len[i] = capA;
}

/* pass B over slack pass-A segments: block a reads [a*capA, a*capA+len[a])
 * and writes its sub-buckets compactly at segout[a] (exclusive scan of the
 * lengths), so bucket_offsets keep the contiguous B+1 convention that
 * lds_join consumes. */
__global__ __launch_bounds__(BUCKET_THREADS) void bucket_subpart_slack_kernel(
  const longlong2* __restrict__ in_pairs, const uint32_t* __restrict__ seg_len, int64_t capA,
  const int64_t* __restrict__ segout /* PA+1 compact output bases */, int B, int F,
  longlong2* __restrict__ out_pairs, int64_t* __restrict__ bucket_offsets /* B+1 */)
{
  extern __shared__ __attribute__((aligned(16))) char smem[];
  longlong2* tbuf = (longlong2*)smem;
  uint32_t* hist = (uint32_t*)(tbuf + SCATTER_TILE);
  uint32_t* base = hist + F;
  uint32_t* gcur = base + F;
  uint32_t* seghist = gcur + F;
  const int tid = threadIdx.x;
  const int a = blockIdx.x;
  const int64_t s0 = (int64_t)a * capA;
  const int64_t s1 = s0 + seg_len[a];
  const int64_t ob = segout[a];
  if (tid < F) seghist[tid] = 0;
  __syncthreads();
  for (int64_t i = s0 + tid; i < s1; i += blockDim.x)
    atomicAdd(&seghist[subF_of(nt_load(&in_pairs[i].x), F)], 1u);
  __syncthreads();
  if (tid == 0) {
    uint32_t acc = 0;
    for (int j = 0; j < F; j++) {
      uint32_t c = seghist[j];
      gcur[j] = (uint32_t)(ob + acc);
      bucket_offsets[(size_t)a * F + j] = ob + acc;
      acc += c;
    }
    if (a == gridDim.x - 1) bucket_offsets[B] = segout[gridDim.x];
  }
  __syncthreads();
  staged_scatter_span<1, false>(nullptr, nullptr, in_pairs, s0, s1, F, tbuf, hist, base, gcur,
                                out_pairs);
}

/* pass B without the count sweep (round 2): block = pass-A group a reads its
 * slack segment [a*capA, a*capA+seg_len[a]) and scatters into F per-bucket
 * SLACK segments at analytic starts b*capB with LDS cursors (only this
 * block writes group a's buckets, so no global cursor atomics), pipelined:
 * the next tile's loads issue before this tile's flush so the HBM load
 * latency hides under the store burst. Bucket lengths out in d_lens.
 * BTILE 8192 (1 block/CU) measured 0.88 vs 1.01 ms/table over the 4096
 * 2-block tile, and per-bucket limits are precomputed in LDS (glim) so the
 * flush bound check pays no per-row 64-bit multiply (experiments/join_v7). */
constexpr int BTILE = 8192;  // pass-B staging tile (128 KiB, 1 block/CU)
__device__ __forceinline__ void slackB_body(
  const longlong2* __restrict__ in_pairs, const uint32_t* __restrict__ seg_len, int64_t capA,
  int F, int64_t capB, longlong2* __restrict__ out_pairs, uint32_t* __restrict__ lens,
  int* __restrict__ any_overflow, int a)
{
  constexpr int VPT = BTILE / BUCKET_THREADS; /* 8 */
  extern __shared__ __attribute__((aligned(16))) char smem[];
  longlong2* tbuf = (longlong2*)smem;
  uint32_t* hist = (uint32_t*)(tbuf + BTILE);
  uint32_t* base = hist + F;
  uint32_t* gcur = base + F;
  uint32_t* glim = gcur + F;
  uint32_t* partials = glim + F; /* 16 */
  __shared__ int s_ovf;
  const int tid = threadIdx.x;
  const int64_t s0 = (int64_t)a * capA;
  const int64_t s1 = s0 + seg_len[a];
  if (tid == 0) s_ovf = 0;
  for (int j = tid; j < F; j += blockDim.x) {
    const int64_t b0 = ((int64_t)a * F + j) * capB;
    gcur[j] = (uint32_t)b0;
    glim[j] = (uint32_t)(b0 + capB);
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
    int64_t i = s0 + (int64_t)v * BUCKET_THREADS + tid;
    if (i < s1) {
      r[v].x = nt_load(&in_pairs[i].x);
      r[v].y = nt_load(&in_pairs[i].y);
      g[v] = subF_of(r[v].x, F);
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
    for (int v = 0; v < VPT; v++) {
      int64_t i = t0 + (int64_t)v * BUCKET_THREADS + tid;
      if (i < s1) tbuf[base[g[v]] + rank[v]] = r[v];
    }
    __syncthreads();
    longlong2 r2[VPT];
#pragma unroll
    for (int v = 0; v < VPT; v++) {
      int64_t i = t1 + (int64_t)v * BUCKET_THREADS + tid;
      if (i < s1) {
        r2[v].x = nt_load(&in_pairs[i].x);
        r2[v].y = nt_load(&in_pairs[i].y);
      }
    }
    for (int pos = tid; pos < count; pos += blockDim.x) {
      longlong2 row = tbuf[pos];
      uint32_t gg = subF_of(row.x, F);
      uint32_t dst = gcur[gg] + (uint32_t)(pos - base[gg]);
      if (dst < glim[gg])
        out_pairs[dst] = row; /* plain store: partial lines must merge in L2 */
      else
        s_ovf = 1;
    }
    __syncthreads();
    if (tid < F) gcur[tid] += hist[tid];
    if (tid < F) hist[tid] = 0;
    __syncthreads();
#pragma unroll
    for (int v = 0; v < VPT; v++) {
      int64_t i = t1 + (int64_t)v * BUCKET_THREADS + tid;
      if (i < s1) {
        r[v] = r2[v];
        g[v] = subF_of(r[v].x, F);
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

__global__ __launch_bounds__(BUCKET_THREADS) void bucket_subpart_slack2_kernel(
  const longlong2* __restrict__ in_pairs, const uint32_t* __restrict__ seg_len, int64_t capA,
  int F, int64_t capB, longlong2* __restrict__ out_pairs, uint32_t* __restrict__ lens,
  int* __restrict__ any_overflow)
{
  slackB_body(in_pairs, seg_len, capA, F, capB, out_pairs, lens, any_overflow, blockIdx.x);
}

/* both tables in ONE launch (grid = 2*PA; see bucket_scatter_slack_pair) */
__global__ __launch_bounds__(BUCKET_THREADS) void bucket_subpart_slack2_pair_kernel(
  const longlong2* __restrict__ in0, const uint32_t* __restrict__ seg0, int64_t capA0,
  int64_t capB0, longlong2* __restrict__ out0, uint32_t* __restrict__ lens0,
  const longlong2* __restrict__ in1, const uint32_t* __restrict__ seg1, int64_t capA1,
  int64_t capB1, longlong2* __restrict__ out1, uint32_t* __restrict__ lens1, int F,
  int* __restrict__ any_overflow)
{
  const int half = gridDim.x / 2;
  if (blockIdx.x < half)
    slackB_body(in0, seg0, capA0, F, capB0, out0, lens0, any_overflow, blockIdx.x);
  else
    slackB_body(in1, seg1, capA1, F, capB1, out1, lens1, any_overflow, blockIdx.x - half);
}

/* both tables through ONE pass-A launch and ONE pass-B launch (each table
 * gets half the grid; the second table's blocks fill the first's tail wave
 * — the sequential two-launch form leaves two wave boundaries idle) */
void bucket_partition2_slack_pair(const int64_t* d_k0, const int64_t* d_p0, int64_t n0,
                                  longlong2* d_tmp0, uint32_t* d_cur0, int64_t capB0,
                                  longlong2* d_out0, uint32_t* d_len0, const int64_t* d_k1,
                                  const int64_t* d_p1, int64_t n1, longlong2* d_tmp1,
                                  uint32_t* d_cur1, int64_t capB1, longlong2* d_out1,
                                  uint32_t* d_len1, int B, int* d_any_overflow, hipStream_t s)
{
  DJ_CHECK_ERROR(n0 < (int64_t)UINT32_MAX && n1 < (int64_t)UINT32_MAX,
                 "bucket_partition_slack: n must be < 2^32");
  const int PA = bucket_groups_for(B);
  const int F = B / PA;
  const int64_t capA0 = slack_capA(n0, PA), capA1 = slack_capA(n1, PA);
  DJ_CHECK_ERROR(PA >= 2 && PA <= 1024 && F >= 1 && F <= 1024,
                 "bucket_partition_slack: B out of range (PA must be >= 2)");
  DJ_CHECK_ERROR((int64_t)PA * capA0 + n0 < (int64_t)UINT32_MAX &&
                   (int64_t)PA * capA1 + n1 < (int64_t)UINT32_MAX &&
                   (int64_t)B * capB0 < (int64_t)UINT32_MAX &&
                   (int64_t)B * capB1 < (int64_t)UINT32_MAX,
                 "bucket_partition_slack: slack layout exceeds u32 row index");
  size_t scatter_lds = SLACK_TILE * sizeof(longlong2) + 4 * (size_t)PA * sizeof(uint32_t) + 64;
  size_t sub_lds = BTILE * sizeof(longlong2) + 4 * (size_t)F * sizeof(uint32_t) + 64;
  DJ_HIP_CALL(hipMemsetAsync(d_cur0, 0, (size_t)PA * 4, s));
  DJ_HIP_CALL(hipMemsetAsync(d_cur1, 0, (size_t)PA * 4, s));
  /* grid split proportional to table sizes (TPC-H joins 60M x 240M; an
   * even split leaves the small half's chunks 4x lighter) */
  int blocks0 = (int)(2.0 * BUCKET_BLOCKS * (double)n0 / (double)(n0 + n1) + 0.5);
  if (blocks0 < 1) blocks0 = 1;
  if (blocks0 > 2 * BUCKET_BLOCKS - 1) blocks0 = 2 * BUCKET_BLOCKS - 1;
  hipLaunchKernelGGL(bucket_scatter_slack_pair_kernel, dim3(2 * BUCKET_BLOCKS),
                     dim3(BUCKET_THREADS), scatter_lds, s, d_k0, d_p0, n0, capA0, d_cur0,
                     d_tmp0, d_k1, d_p1, n1, capA1, d_cur1, d_tmp1, PA, blocks0,
                     d_any_overflow);
  DJ_HIP_CALL(hipGetLastError());
  hipLaunchKernelGGL(clamp_seglen_kernel, dim3((PA + 255) / 256), dim3(256), 0, s, d_cur0,
                     PA, (uint32_t)capA0);
  hipLaunchKernelGGL(clamp_seglen_kernel, dim3((PA + 255) / 256), dim3(256), 0, s, d_cur1,
                     PA, (uint32_t)capA1);
  DJ_HIP_CALL(hipGetLastError());
  hipLaunchKernelGGL(bucket_subpart_slack2_pair_kernel, dim3(2 * PA), dim3(BUCKET_THREADS),
                     sub_lds, s, d_tmp0, d_cur0, capA0, capB0, d_out0, d_len0, d_tmp1,
                     d_cur1, capA1, capB1, d_out1, d_len1, F, d_any_overflow);
  DJ_HIP_CALL(hipGetLastError());
}

void bucket_partition2_slack(const int64_t* d_keys, const int64_t* d_pay, int64_t n, int B,
                             longlong2* d_tmp_pairs, uint32_t* d_cursors, int64_t capB,
                             longlong2* d_out_pairs, uint32_t* d_lens, int* d_any_overflow,
                             hipStream_t s)
{
  DJ_CHECK_ERROR(n < (int64_t)UINT32_MAX, "bucket_partition_slack: n must be < 2^32");
  const int PA = bucket_groups_for(B);
  const int F = B / PA;
  const int64_t capA = slack_capA(n, PA);
  DJ_CHECK_ERROR(PA >= 2 && PA <= 1024 && F >= 1 && F <= 1024,
                 "bucket_partition_slack: B out of range (PA must be >= 2)");
  DJ_CHECK_ERROR((int64_t)PA * capA + n < (int64_t)UINT32_MAX &&
                   (int64_t)B * capB < (int64_t)UINT32_MAX,
                 "bucket_partition_slack: slack layout exceeds u32 row index");
  size_t scatter_lds = SLACK_TILE * sizeof(longlong2) + 4 * (size_t)PA * sizeof(uint32_t) + 64;
  size_t sub_lds = BTILE * sizeof(longlong2) + 4 * (size_t)F * sizeof(uint32_t) + 64;
  DJ_HIP_CALL(hipMemsetAsync(d_cursors, 0, (size_t)PA * 4, s));
  hipLaunchKernelGGL(bucket_scatter_slack_kernel, dim3(BUCKET_BLOCKS), dim3(BUCKET_THREADS),
                     scatter_lds, s, d_keys, d_pay, n, PA, capA, d_cursors, d_any_overflow,
                     d_tmp_pairs);
  DJ_HIP_CALL(hipGetLastError());
  hipLaunchKernelGGL(clamp_seglen_kernel, dim3((PA + 255) / 256), dim3(256), 0, s, d_cursors,
                     PA, (uint32_t)capA);
  DJ_HIP_CALL(hipGetLastError());
  hipLaunchKernelGGL(bucket_subpart_slack2_kernel, dim3(PA), dim3(BUCKET_THREADS), sub_lds, s,
                     d_tmp_pairs, d_cursors, capA, F, capB, d_out_pairs, d_lens,
                     d_any_overflow);
  DJ_HIP_CALL(hipGetLastError());
}

void bucket_partition2(const int64_t* d_keys, const int64_t* d_pay, int64_t n, int B,
                       longlong2* d_tmp_pairs, uint32_t* d_counts, uint32_t* d_totals,
                       int64_t* d_segoff, int64_t* d_offsets, longlong2* d_out_pairs,
                       int* d_any_overflow, hipStream_t s)
{
  DJ_CHECK_ERROR(n < (int64_t)UINT32_MAX, "bucket_partition: n must be < 2^32");
  const int PA = bucket_groups_for(B);
  const int F = B / PA;
  DJ_CHECK_ERROR(PA >= 1 && PA <= 1024 && F >= 1 && F <= 1024,
                 "bucket_partition: B out of range");
  const size_t subpart_lds =
    SCATTER_TILE * sizeof(longlong2) + 4 * (size_t)F * sizeof(uint32_t);
  if (PA == 1) {
    hipLaunchKernelGGL(set_segoff1_kernel, dim3(1), dim3(1), 0, s, d_segoff, n);
    DJ_HIP_CALL(hipGetLastError());
    hipLaunchKernelGGL(bucket_subpart_kernel<true>, dim3(PA), dim3(BUCKET_THREADS),
                       subpart_lds, s, (const longlong2*)nullptr, d_keys, d_pay, d_segoff, B,
                       F, d_out_pairs, d_offsets);
    DJ_HIP_CALL(hipGetLastError());
  } else if (d_any_overflow != nullptr &&
             (int64_t)PA * slack_capA(n, PA) + n < (int64_t)UINT32_MAX) {
    /* slack path: no count pass (see bucket_scatter_slack_kernel header) */
    const int64_t capA = slack_capA(n, PA);
    size_t scatter_lds = SLACK_TILE * sizeof(longlong2) + 4 * (size_t)PA * sizeof(uint32_t) + 64;
    DJ_HIP_CALL(hipMemsetAsync(d_totals, 0, (size_t)PA * 4, s));
    hipLaunchKernelGGL(bucket_scatter_slack_kernel, dim3(BUCKET_BLOCKS), dim3(BUCKET_THREADS),
                       scatter_lds, s, d_keys, d_pay, n, PA, capA, d_totals, d_any_overflow,
                       d_tmp_pairs);
    DJ_HIP_CALL(hipGetLastError());
    hipLaunchKernelGGL(clamp_seglen_kernel, dim3((PA + 255) / 256), dim3(256), 0, s, d_totals,
                       PA, (uint32_t)capA);
    DJ_HIP_CALL(hipGetLastError());
    hipLaunchKernelGGL(bucket_scanB_kernel, dim3(1), dim3(BUCKET_THREADS), 0, s, d_totals, PA,
                       d_segoff);
    DJ_HIP_CALL(hipGetLastError());
    hipLaunchKernelGGL(bucket_subpart_slack_kernel, dim3(PA), dim3(BUCKET_THREADS), subpart_lds,
                       s, d_tmp_pairs, d_totals, capA, d_segoff, B, F, d_out_pairs, d_offsets);
    DJ_HIP_CALL(hipGetLastError());
  } else {
    size_t hist_lds = (size_t)PA * sizeof(uint32_t);
    size_t scatter_lds = SCATTER_TILE * sizeof(longlong2) + 3 * hist_lds;
    hipLaunchKernelGGL(bucket_count_kernel, dim3(BUCKET_BLOCKS), dim3(BUCKET_THREADS), hist_lds,
                       s, d_keys, n, PA, d_counts);
    DJ_HIP_CALL(hipGetLastError());
    hipLaunchKernelGGL(bucket_scanA_kernel, dim3(PA), dim3(BUCKET_BLOCKS), 0, s, d_counts, PA,
                       d_totals);
    DJ_HIP_CALL(hipGetLastError());
    hipLaunchKernelGGL(bucket_scanB_kernel, dim3(1), dim3(BUCKET_THREADS), 0, s, d_totals, PA,
                       d_segoff);
    DJ_HIP_CALL(hipGetLastError());
    hipLaunchKernelGGL(bucket_scatter_kernel, dim3(BUCKET_BLOCKS), dim3(BUCKET_THREADS),
                       scatter_lds, s, d_keys, d_pay, n, PA, d_counts, d_segoff, d_tmp_pairs);
    DJ_HIP_CALL(hipGetLastError());
    hipLaunchKernelGGL(bucket_subpart_kernel<false>, dim3(PA), dim3(BUCKET_THREADS),
                       subpart_lds, s, d_tmp_pairs, (const int64_t*)nullptr,
                       (const int64_t*)nullptr, d_segoff, B, F, d_out_pairs, d_offsets);
    DJ_HIP_CALL(hipGetLastError());
  }
}

static void lds_join_launch(const longlong2* d_lrows, const int64_t* d_loff,
                            const uint32_t* d_llen, int64_t capL, const longlong2* d_rrows,
                            const int64_t* d_roff, const uint32_t* d_rlen, int64_t capR,
                            int B, int table_slots, int64_t* d_out0, int64_t* d_out1,
                            int64_t* d_out2, int64_t* d_out3, int64_t cap, int64_t* d_counter,
                            uint32_t* d_overflow_flags, int* d_any_overflow, int* d_error,
                            hipStream_t s)
{
  DJ_CHECK_ERROR(table_slots == 2048 || table_slots == 4096,
                 "lds_join: table_slots must be 2048 or 4096");
  /* KBUK = 4 whenever B is a multiple of 4 — slack callers pad to it, and
   * every compact caller's B already is (bucket_count_for powers of two,
   * fused-wire PA*F). KBUK = 1 (flush per bucket, ~0.3 ms/100M slower)
   * only for an unaligned compact B. */
  const bool slack = d_llen != nullptr;
  const int kbuk = (slack || B % 4 == 0) ? 4 : 1;
  int64_t groups = ((int64_t)B + kbuk - 1) / kbuk;
  int grid = (int)(groups < 8192 ? groups : 8192);
  size_t lds =
    (size_t)table_slots * sizeof(longlong2) + 4 * JOIN_STAGE_ROWS * sizeof(int64_t) + 16;
  auto launch = [&](auto kern) {
    hipLaunchKernelGGL(kern, dim3(grid), dim3(BUCKET_THREADS), lds, s, d_lrows, d_loff,
                       d_llen, capL, d_rrows, d_roff, d_rlen, capR, B, d_out0, d_out1,
                       d_out2, d_out3, cap, (unsigned long long*)d_counter, d_overflow_flags,
                       d_any_overflow, d_error);
  };
  if (table_slots == 4096) {
    if (slack)
      launch(lds_join_kernel<4096, true, 4>);
    else
      kbuk == 4 ? launch(lds_join_kernel<4096, false, 4>)
                : launch(lds_join_kernel<4096, false, 1>);
  } else {
    if (slack)
      launch(lds_join_kernel<2048, true, 4>);
    else
      kbuk == 4 ? launch(lds_join_kernel<2048, false, 4>)
                : launch(lds_join_kernel<2048, false, 1>);
  }
  DJ_HIP_CALL(hipGetLastError());
}

void lds_join(const longlong2* d_lrows, const int64_t* d_loff, const longlong2* d_rrows,
              const int64_t* d_roff, int B, int table_slots, int64_t* d_out0, int64_t* d_out1,
              int64_t* d_out2, int64_t* d_out3, int64_t cap, int64_t* d_counter,
              uint32_t* d_overflow_flags, int* d_any_overflow, int* d_error, hipStream_t s)
{
  lds_join_launch(d_lrows, d_loff, nullptr, 0, d_rrows, d_roff, nullptr, 0, B, table_slots,
                  d_out0, d_out1, d_out2, d_out3, cap, d_counter, d_overflow_flags,
                  d_any_overflow, d_error, s);
}

void lds_join_slack(const longlong2* d_lrows, const uint32_t* d_llen, int64_t capL,
                    const longlong2* d_rrows, const uint32_t* d_rlen, int64_t capR, int B,
                    int table_slots, int64_t* d_out0, int64_t* d_out1, int64_t* d_out2,
                    int64_t* d_out3, int64_t cap, int64_t* d_counter,
                    uint32_t* d_overflow_flags, int* d_any_overflow, int* d_error,
                    hipStream_t s)
{
  DJ_CHECK_ERROR(B % 4 == 0,
                 "lds_join_slack: B must be padded to a multiple of 4 (zero-length "
                 "buckets) — see lds_join_kernel's KBUK contract");
  lds_join_launch(d_lrows, nullptr, d_llen, capL, d_rrows, nullptr, d_rlen, capR, B,
                  table_slots, d_out0, d_out1, d_out2, d_out3, cap, d_counter,
                  d_overflow_flags, d_any_overflow, d_error, s);
}

}  // namespace dj
