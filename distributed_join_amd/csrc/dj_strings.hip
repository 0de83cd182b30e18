/*
 * dj_strings.hip — gfx950 kernels for the strings-column path of the
 * distributed join (BASELINE config 4). MI355X-native equivalents of the
 * reference's thrust-based helpers (SURVEY.md §2 row "String column
 * support"):
 *   - sizes_from_offsets  <- calculate_string_sizes_from_offsets
 *                            (strings_column.cu:81-109, adjacent difference)
 *   - offsets_from_sizes  <- calculate_string_offsets_from_sizes
 *                            (strings_column.cu:111-131, inclusive scan with
 *                            offset[0] = 0 — sizes, not offsets, go on the
 *                            wire; receiver rebuilds offsets)
 *   - gather_sizes/gather_strings <- thrust::gather of per-partition char
 *                            offsets (strings_column.cu:39-79) generalized to
 *                            a row permutation (used by partition & join
 *                            output assembly)
 *   - make_test_string_sizes/fill_test_strings <- the deterministic string
 *                            payload of test/string_payload.cu:50-94
 *                            (len = k%7+1, char = 'a'+k%26)
 * String offsets are int32 (cudf convention; chars per column < 2^31 — the
 * reference shares this bound).
 */
#include "dj_error.hpp"
#include "dj_kernels.hpp"

#include <hip/hip_runtime.h>

namespace dj {

namespace {
constexpr int SBLOCK = 256;
constexpr int SVPT = 8;  // elements per thread in the scan partials pass
constexpr int CHUNK = SBLOCK * SVPT;

int sgrid(int64_t n)
{
  int64_t b = (n + SBLOCK - 1) / SBLOCK;
  if (b > 2048) b = 2048;
  if (b < 1) b = 1;
  return (int)b;
}
}  // namespace

__global__ void sizes_from_offsets_kernel(const int32_t* __restrict__ offsets, int64_t n,
                                          int32_t* __restrict__ sizes)
{
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) sizes[i] = offsets[i + 1] - offsets[i];
}

__global__ void sum_sizes_i64_kernel(const int32_t* __restrict__ sizes, int64_t n,
                                     int64_t* __restrict__ total)
{
  __shared__ int64_t red[256];
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  int64_t acc = 0;
  for (; i < n; i += stride) acc += sizes[i];
  red[threadIdx.x] = acc;
  __syncthreads();
  for (int off = 128; off > 0; off >>= 1) {
    if (threadIdx.x < (unsigned)off) red[threadIdx.x] += red[threadIdx.x + off];
    __syncthreads();
  }
  if (threadIdx.x == 0) atomicAdd((unsigned long long*)total, (unsigned long long)red[0]);
}

void sum_sizes_i64(const int32_t* d_sizes, int64_t n, int64_t* d_total, hipStream_t s)
{
  int64_t b = (n + 255) / 256;
  if (b > 1024) b = 1024;
  if (b < 1) b = 1;
  hipLaunchKernelGGL(sum_sizes_i64_kernel, dim3((uint32_t)b), dim3(256), 0, s, d_sizes, n,
                     d_total);
  DJ_HIP_CALL(hipGetLastError());
}

void sizes_from_offsets(const int32_t* d_offsets, int64_t n, int32_t* d_sizes, hipStream_t s)
{
  if (n <= 0) return;
  hipLaunchKernelGGL(sizes_from_offsets_kernel, dim3(sgrid(n)), dim3(SBLOCK), 0, s, d_offsets,
                     n, d_sizes);
  DJ_HIP_CALL(hipGetLastError());
}

/* ---- exclusive scan of int32 sizes -> int32 offsets[n+1], offsets[0]=0 ----
 * Tile-coalesced: a block owns a 4096-element tile; loads/stores are
 * lane-contiguous, the per-thread serial work happens in LDS. (The first
 * version gave each THREAD a serial chunk — lane-adjacent addresses 8 KB
 * apart, 64 lines per load instruction: the scan alone cost 5 ms of the
 * TPC-H step, gpurun_out/r2_tpch_prof3.) Whole-column totals stay < 2^31
 * (the int32 offsets cap, enforced upstream), so in-tile prefixes fit i32. */

constexpr int SCAN_T = 4096;             // elements per block tile (16 KiB LDS)
constexpr int SSEG = SCAN_T / SBLOCK;    // serial segment per thread (16)

__global__ void scan_tile_partials_kernel(const int32_t* __restrict__ sizes, int64_t n,
                                          int64_t ntiles, int64_t* __restrict__ partials)
{
  __shared__ int64_t red[SBLOCK];
  for (int64_t c = blockIdx.x; c < ntiles; c += gridDim.x) {
    const int64_t base = c * SCAN_T;
    int64_t acc = 0;
#pragma unroll
    for (int k = 0; k < SSEG; k++) {
      int64_t i = base + (int64_t)k * SBLOCK + threadIdx.x;
      if (i < n) acc += sizes[i];
    }
    red[threadIdx.x] = acc;
    __syncthreads();
    for (int off = SBLOCK / 2; off > 0; off >>= 1) {
      if (threadIdx.x < (unsigned)off) red[threadIdx.x] += red[threadIdx.x + off];
      __syncthreads();
    }
    if (threadIdx.x == 0) partials[c] = red[0];
    __syncthreads();
  }
}

__global__ void scan_tile_finalize_kernel(const int32_t* __restrict__ sizes, int64_t n,
                                          int64_t ntiles,
                                          const int64_t* __restrict__ partials,
                                          int32_t* __restrict__ offsets)
{
  __shared__ int32_t tile[SCAN_T];
  __shared__ int64_t segsum[SBLOCK + 1];
  for (int64_t c = blockIdx.x; c < ntiles; c += gridDim.x) {
    const int64_t base = c * SCAN_T;
#pragma unroll
    for (int k = 0; k < SSEG; k++) {
      int64_t i = base + (int64_t)k * SBLOCK + threadIdx.x;
      tile[(int64_t)k * SBLOCK + threadIdx.x] = (i < n) ? sizes[i] : 0;
    }
    __syncthreads();
    /* in-place exclusive scan of this thread's contiguous LDS segment */
    int32_t acc = 0;
#pragma unroll
    for (int v = 0; v < SSEG; v++) {
      int32_t x = tile[threadIdx.x * SSEG + v];
      tile[threadIdx.x * SSEG + v] = acc;
      acc += x;
    }
    segsum[threadIdx.x] = acc;
    __syncthreads();
    /* exclusive scan of the 256 segment sums (Hillis-Steele in LDS) */
    for (int off = 1; off < SBLOCK; off <<= 1) {
      int64_t add = (threadIdx.x >= (unsigned)off) ? segsum[threadIdx.x - off] : 0;
      __syncthreads();
      segsum[threadIdx.x] += add;
      __syncthreads();
    }
    if (threadIdx.x == 0) segsum[SBLOCK] = segsum[SBLOCK - 1];  // tile total
    __syncthreads();
    const int64_t tbase = partials[c];
#pragma unroll
    for (int k = 0; k < SSEG; k++) {
      int64_t i = base + (int64_t)k * SBLOCK + threadIdx.x;
      if (i < n) {
        int j = (int)((int64_t)k * SBLOCK + threadIdx.x);
        int64_t seg_excl = (j >= SSEG) ? segsum[j / SSEG - 1] : 0;
        offsets[i] = (int32_t)(tbase + seg_excl + tile[j]);
      }
    }
    if (threadIdx.x == 0 && base + SCAN_T >= n)
      offsets[n] = (int32_t)(tbase + segsum[SBLOCK]);
    __syncthreads();
  }
}

/* single block: exclusive scan of partials in place */
__global__ void scan_partials_exclusive_kernel(int64_t* partials, int64_t nchunks)
{
  __shared__ int64_t sh[1024];
  __shared__ int64_t running_sh;
  if (threadIdx.x == 0) running_sh = 0;
  __syncthreads();
  for (int64_t base = 0; base < nchunks; base += 1024) {
    int64_t c = base + threadIdx.x;
    int64_t v = (c < nchunks) ? partials[c] : 0;
    sh[threadIdx.x] = v;
    __syncthreads();
    for (int off = 1; off < 1024; off <<= 1) {
      int64_t add = (threadIdx.x >= (unsigned)off) ? sh[threadIdx.x - off] : 0;
      __syncthreads();
      sh[threadIdx.x] += add;
      __syncthreads();
    }
    int64_t rbase = running_sh;
    __syncthreads();
    if (c < nchunks) partials[c] = rbase + sh[threadIdx.x] - v;
    if (threadIdx.x == 1023) running_sh = rbase + sh[threadIdx.x];
    __syncthreads();
  }
}

size_t offsets_from_sizes_scratch_bytes(int64_t n)
{
  int64_t nchunks = (n + CHUNK - 1) / CHUNK;
  return (size_t)(nchunks > 0 ? nchunks : 1) * sizeof(int64_t);
}

void offsets_from_sizes(const int32_t* d_sizes, int64_t n, int32_t* d_offsets, void* d_scratch,
                        hipStream_t s)
{
  if (n <= 0) {
    DJ_HIP_CALL(hipMemsetAsync(d_offsets, 0, sizeof(int32_t), s));
    return;
  }
  int64_t* partials = (int64_t*)d_scratch;  // scratch sized by CHUNK < SCAN_T
  int64_t ntiles = (n + SCAN_T - 1) / SCAN_T;
  int tgrid = (int)(ntiles < 2048 ? ntiles : 2048);
  hipLaunchKernelGGL(scan_tile_partials_kernel, dim3(tgrid), dim3(SBLOCK), 0, s, d_sizes, n,
                     ntiles, partials);
  DJ_HIP_CALL(hipGetLastError());
  hipLaunchKernelGGL(scan_partials_exclusive_kernel, dim3(1), dim3(1024), 0, s, partials,
                     ntiles);
  DJ_HIP_CALL(hipGetLastError());
  hipLaunchKernelGGL(scan_tile_finalize_kernel, dim3(tgrid), dim3(SBLOCK), 0, s, d_sizes, n,
                     ntiles, partials, d_offsets);
  DJ_HIP_CALL(hipGetLastError());
}

/* ---- permutation gathers ---- */

__global__ void gather_sizes_kernel(const int32_t* __restrict__ src_off,
                                    const int64_t* __restrict__ idx, int64_t n,
                                    int32_t* __restrict__ sizes)
{
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    int64_t j = idx[i];
    sizes[i] = src_off[j + 1] - src_off[j];
  }
}

void gather_sizes(const int32_t* d_src_off, const int64_t* d_idx, int64_t n, int32_t* d_sizes,
                  hipStream_t s)
{
  if (n <= 0) return;
  hipLaunchKernelGGL(gather_sizes_kernel, dim3(sgrid(n)), dim3(SBLOCK), 0, s, d_src_off, d_idx,
                     n, d_sizes);
  DJ_HIP_CALL(hipGetLastError());
}

/* sizes AND source char starts in one pass over the random offsets (the
 * 8 B line serves both); gather_chars_from_starts then streams the staged
 * starts sequentially instead of re-walking idx + src_off at random —
 * cuts the chars gather's line traffic roughly in half at TPC-H scale */
__global__ void gather_sizes_starts_kernel(const int32_t* __restrict__ src_off,
                                           const int64_t* __restrict__ idx, int64_t n,
                                           int32_t* __restrict__ sizes,
                                           int32_t* __restrict__ starts)
{
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    int64_t j = idx[i];
    int32_t s0 = src_off[j];
    sizes[i] = src_off[j + 1] - s0;
    starts[i] = s0;
  }
}

void gather_sizes_starts(const int32_t* d_src_off, const int64_t* d_idx, int64_t n,
                         int32_t* d_sizes, int32_t* d_starts, hipStream_t s)
{
  if (n <= 0) return;
  hipLaunchKernelGGL(gather_sizes_starts_kernel, dim3(sgrid(n)), dim3(SBLOCK), 0, s,
                     d_src_off, d_idx, n, d_sizes, d_starts);
  DJ_HIP_CALL(hipGetLastError());
}

/* per row: load the 1..3 aligned u64 words covering [s0, s0+len) once and
 * extract bytes from registers — a byte-loop re-loads the same line len
 * times (len ≈ 8 per row at TPC-H/config-4 string sizes, so this quarters
 * the load instructions; the byte STORES stay — adjacent rows share dst
 * words, so wide stores would race on the shared bytes). Safe to read the
 * full aligned window: hipMalloc allocations are >= 4 KiB aligned, so the
 * 8 B-aligned words of any valid [s0, s0+len) range stay in-allocation. */
__global__ void gather_chars_from_starts_kernel(const uint8_t* __restrict__ src_chars,
                                                const int32_t* __restrict__ starts, int64_t n,
                                                const int32_t* __restrict__ dst_off,
                                                uint8_t* __restrict__ dst_chars)
{
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    const int64_t s0 = (int64_t)starts[i];
    const int64_t d0 = dst_off[i];
    const int32_t len = dst_off[i + 1] - (int32_t)d0;
    const uint64_t* words = (const uint64_t*)(src_chars + (s0 & ~(int64_t)7));
    int off = (int)(s0 & 7);  // first byte's position in words[0]
    uint64_t w = words[0];
    int wi = 0;
    for (int32_t k = 0; k < len; k++) {
      const int b = off + k;
      if ((b >> 3) != wi) {
        wi = b >> 3;
        w = words[wi];
      }
      dst_chars[d0 + k] = (uint8_t)(w >> ((b & 7) * 8));
    }
  }
}

void gather_chars_from_starts(const uint8_t* d_src_chars, const int32_t* d_starts, int64_t n,
                              const int32_t* d_dst_off, uint8_t* d_dst_chars, hipStream_t s)
{
  if (n <= 0) return;
  hipLaunchKernelGGL(gather_chars_from_starts_kernel, dim3(sgrid(n)), dim3(SBLOCK), 0, s,
                     d_src_chars, d_starts, n, d_dst_off, d_dst_chars);
  DJ_HIP_CALL(hipGetLastError());
}

/* ---- deterministic test/bench string payload (string_payload.cu:50-94) ---- */

__global__ void test_string_sizes_kernel(const int64_t* __restrict__ keys, int64_t n,
                                         int32_t* __restrict__ sizes)
{
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) sizes[i] = (int32_t)(keys[i] % 7 + 1);
}

__global__ void fill_test_strings_kernel(const int64_t* __restrict__ keys, int64_t n,
                                         const int32_t* __restrict__ offsets,
                                         uint8_t* __restrict__ chars)
{
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    int32_t o0 = offsets[i], o1 = offsets[i + 1];
    uint8_t ch = (uint8_t)('a' + keys[i] % 26);
    for (int32_t k = o0; k < o1; k++) chars[k] = ch;
  }
}

void make_test_string_sizes(const int64_t* d_keys, int64_t n, int32_t* d_sizes, hipStream_t s)
{
  if (n <= 0) return;
  hipLaunchKernelGGL(test_string_sizes_kernel, dim3(sgrid(n)), dim3(SBLOCK), 0, s, d_keys, n,
                     d_sizes);
  DJ_HIP_CALL(hipGetLastError());
}

void fill_test_strings(const int64_t* d_keys, int64_t n, const int32_t* d_offsets,
                       uint8_t* d_chars, hipStream_t s)
{
  if (n <= 0) return;
  hipLaunchKernelGGL(fill_test_strings_kernel, dim3(sgrid(n)), dim3(SBLOCK), 0, s, d_keys, n,
                     d_offsets, d_chars);
  DJ_HIP_CALL(hipGetLastError());
}

}  // namespace dj
