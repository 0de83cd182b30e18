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

/* ---- exclusive scan of int32 sizes -> int32 offsets[n+1], offsets[0]=0 ---- */

__global__ void scan_partials_kernel(const int32_t* __restrict__ sizes, int64_t n,
                                     int64_t nchunks, int64_t* __restrict__ partials)
{
  int64_t c = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; c < nchunks; c += stride) {
    int64_t start = c * CHUNK;
    int64_t end = min(start + (int64_t)CHUNK, n);
    int64_t acc = 0;
    for (int64_t i = start; i < end; i++) acc += sizes[i];
    partials[c] = acc;
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

__global__ void scan_finalize_kernel(const int32_t* __restrict__ sizes, int64_t n,
                                     int64_t nchunks, const int64_t* __restrict__ partials,
                                     int32_t* __restrict__ offsets)
{
  int64_t c = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; c < nchunks; c += stride) {
    int64_t start = c * CHUNK;
    int64_t end = min(start + (int64_t)CHUNK, n);
    int64_t acc = partials[c];
    for (int64_t i = start; i < end; i++) {
      offsets[i] = (int32_t)acc;
      acc += sizes[i];
    }
    if (end == n) offsets[n] = (int32_t)acc;
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
  int64_t nchunks = (n + CHUNK - 1) / CHUNK;
  int64_t* partials = (int64_t*)d_scratch;
  hipLaunchKernelGGL(scan_partials_kernel, dim3(sgrid(nchunks)), dim3(SBLOCK), 0, s, d_sizes, n,
                     nchunks, partials);
  DJ_HIP_CALL(hipGetLastError());
  hipLaunchKernelGGL(scan_partials_exclusive_kernel, dim3(1), dim3(1024), 0, s, partials,
                     nchunks);
  DJ_HIP_CALL(hipGetLastError());
  hipLaunchKernelGGL(scan_finalize_kernel, dim3(sgrid(nchunks)), dim3(SBLOCK), 0, s, d_sizes, n,
                     nchunks, partials, d_offsets);
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

__global__ void gather_chars_kernel(const int32_t* __restrict__ src_off,
                                    const uint8_t* __restrict__ src_chars,
                                    const int64_t* __restrict__ idx, int64_t n,
                                    const int32_t* __restrict__ dst_off,
                                    uint8_t* __restrict__ dst_chars)
{
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    int64_t j = idx[i];
    int32_t s0 = src_off[j], s1 = src_off[j + 1];
    int32_t d0 = dst_off[i];
    for (int32_t k = 0; k < s1 - s0; k++) dst_chars[d0 + k] = src_chars[s0 + k];
  }
}

void gather_chars(const int32_t* d_src_off, const uint8_t* d_src_chars, const int64_t* d_idx,
                  int64_t n, const int32_t* d_dst_off, uint8_t* d_dst_chars, hipStream_t s)
{
  if (n <= 0) return;
  hipLaunchKernelGGL(gather_chars_kernel, dim3(sgrid(n)), dim3(SBLOCK), 0, s, d_src_off,
                     d_src_chars, d_idx, n, d_dst_off, d_dst_chars);
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
