/*
 * join_v6.hip — round-2 regression isolation: the integrated product
 * lds_join_slack measured 2.45 ms inside bench.py but the identical-shape
 * experiment kernel (join_v4 J-k4w slack) measured 1.38 ms standalone.
 * This binary times BOTH kernels on the SAME product-produced slack layout,
 * each (a) quiescent (partition long finished) and (b) in-context
 * (immediately after both tables' slack partitions on the same stream,
 * segment-timed with events between enqueues) — separating codegen
 * differences from L2-writeback / phase-attribution effects.
 * Build: hipcc --offload-arch=gfx950 -O3 join_v6.hip \
 *          ../distributed_join_amd/csrc/dj_kernels.hip -o join_v6
 * Diagnostic only.
 */
#include "../distributed_join_amd/csrc/dj_kernels.hpp"
#include "../distributed_join_amd/csrc/dj_rng.h"

#include <hip/hip_runtime.h>

#include <cmath>
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

/* ---- join_v4's J-k4w slack kernel, verbatim shape (the 1.38 ms result) -- */
template <int SLOTS2, int KBUK>
__global__ __launch_bounds__(1024) void join_k4w_kernel(
  const longlong2* __restrict__ lrows, const uint32_t* __restrict__ llen, int64_t capL,
  const longlong2* __restrict__ rrows, const uint32_t* __restrict__ rlen, int64_t capR,
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
    for (int k = 0; k < KBUK; k++) {
      const int b = bb + k;
      if (b >= B) break;
      const int64_t l0 = (int64_t)b * capL;
      const int64_t l1 = l0 + llen[b];
      const int64_t r0 = (int64_t)b * capR;
      const int64_t r1 = r0 + rlen[b];
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
      if (k < KBUK - 1 && *cur_sh < (uint32_t)WATER) continue;
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
  printf("join_v6: n=%lld rows/table, sel 0.3 (product-kernel isolation)\n", (long long)n);

  DBuf bk(n * 8), bp(n * 8), pk(n * 8), pp(n * 8);
  dj::generate_build(bk.i64(), bp.i64(), n, rand_max, DJ_DEFAULT_SEED, true, 0, n, 0);
  dj::generate_probe(pk.i64(), pp.i64(), n, rand_max, 0.3, DJ_DEFAULT_SEED, 0, n, 0);
  CHECK(hipDeviceSynchronize());

  const int B = dj::bucket_count_for(n, n);
  const int PA = dj::bucket_groups_for(B);
  const int64_t capA = dj::slack_capA(n, PA);
  const int64_t capB = dj::slack_capB(n, B);
  printf("B=%d PA=%d capA=%lld capB=%lld\n", B, PA, (long long)capA, (long long)capB);

  const int64_t cap = n / 2;
  DBuf o0(cap * 8), o1(cap * 8), o2(cap * 8), o3(cap * 8);
  DBuf counter(8), anyovf(8), err(8), acc(8);
  DBuf lpairs((size_t)B * capB * 16), rpairs((size_t)B * capB * 16);
  DBuf tmp((size_t)PA * capA * 16);
  DBuf cursors((size_t)PA * 4);
  DBuf llen((size_t)B * 4), rlen((size_t)B * 4), flags((size_t)B * 4);

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
    printf("%-40s %.3f ms  count=%llu  %s\n", name, ms, c,
           (c == ref_count && s == ref_sum) ? "OK" : "MISMATCH");
  };

  auto partition_both = [&] {
    CHECK(hipMemsetAsync(anyovf.p, 0, 4, 0));
    dj::bucket_partition2_slack(bk.i64(), bp.i64(), n, B, (longlong2*)tmp.p, cursors.u32(),
                                capB, (longlong2*)lpairs.p, llen.u32(), (int*)anyovf.p, 0);
    dj::bucket_partition2_slack(pk.i64(), pp.i64(), n, B, (longlong2*)tmp.p, cursors.u32(),
                                capB, (longlong2*)rpairs.p, rlen.u32(), (int*)anyovf.p, 0);
  };
  auto reset_join = [&] {
    CHECK(hipMemsetAsync(counter.p, 0, 8, 0));
    CHECK(hipMemsetAsync(err.p, 0, 4, 0));
    CHECK(hipMemsetAsync(flags.p, 0, (size_t)B * 4, 0));
  };
  auto join_product = [&] {
    dj::lds_join_slack((longlong2*)lpairs.p, llen.u32(), capB, (longlong2*)rpairs.p,
                       rlen.u32(), capB, B, 2048, o0.i64(), o1.i64(), o2.i64(), o3.i64(),
                       cap, counter.i64(), flags.u32(), (int*)anyovf.p, (int*)err.p, 0);
  };
  auto join_k4w = [&] {
    size_t lds = (size_t)2048 * 16 + 4 * 1024 * 8 + 16;
    int grid = (B / 4) < 8192 ? (B / 4) : 8192;
    hipLaunchKernelGGL((join_k4w_kernel<2048, 4>), dim3(grid), dim3(1024), lds, 0,
                       (longlong2*)lpairs.p, llen.u32(), capB, (longlong2*)rpairs.p,
                       rlen.u32(), capB, B, o0.i64(), o1.i64(), o2.i64(), o3.i64(), cap,
                       (unsigned long long*)counter.p, flags.u32(), (int*)anyovf.p,
                       (int*)err.p);
    CHECK(hipGetLastError());
  };

  partition_both();
  CHECK(hipDeviceSynchronize());

  /* 1. quiescent joins (experiment conditions) */
  verify("product lds_join_slack quiescent", time_body(3, [&] {
           reset_join();
           join_product();
         }));
  verify("v4 join_k4w quiescent", time_body(3, [&] {
           reset_join();
           join_k4w();
         }));

  /* 2. in-context: full step, segment-timed (bench conditions) */
  hipEvent_t ev[3];
  for (auto& evi : ev) CHECK(hipEventCreate(&evi));
  auto full_step = [&](const std::function<void()>& join, const char* name) {
    float bpart = 1e30f, bjoin = 1e30f, btot = 1e30f;
    for (int i = 0; i < 3; i++) {
      reset_join();
      CHECK(hipEventRecord(ev[0]));
      partition_both();
      CHECK(hipEventRecord(ev[1]));
      join();
      CHECK(hipEventRecord(ev[2]));
      CHECK(hipEventSynchronize(ev[2]));
      float p, j;
      CHECK(hipEventElapsedTime(&p, ev[0], ev[1]));
      CHECK(hipEventElapsedTime(&j, ev[1], ev[2]));
      if (p + j < btot) {
        btot = p + j;
        bpart = p;
        bjoin = j;
      }
    }
    printf("%-40s part=%.3f join=%.3f total=%.3f ms\n", name, bpart, bjoin, btot);
    verify(name, bjoin);
  };
  full_step(join_product, "FULL STEP product (part+join)");
  full_step(join_k4w, "FULL STEP v4-k4w (part+join)");

  /* 3. in-context but with an L2 flush (4 GB dummy read) between partition
   * and join — if the quiescent/in-context gap is dirty-L2 writeback, the
   * flush absorbs it and the join segment returns to quiescent time */
  {
    DBuf dummy((size_t)4 << 30);
    CHECK(hipMemset(dummy.p, 1, (size_t)4 << 30));
    CHECK(hipDeviceSynchronize());
    float bflush = 1e30f, bjoin = 1e30f;
    for (int i = 0; i < 3; i++) {
      reset_join();
      CHECK(hipEventRecord(ev[0]));
      partition_both();
      CHECK(hipMemset(acc.p, 0, 8));
      hipLaunchKernelGGL(checksum_kernel, dim3(8192), dim3(256), 0, 0, dummy.i64(),
                         dummy.i64(), dummy.i64(), dummy.i64(), (int64_t)(1 << 27),
                         (unsigned long long*)acc.p);
      CHECK(hipEventRecord(ev[1]));
      join_product();
      CHECK(hipEventRecord(ev[2]));
      CHECK(hipEventSynchronize(ev[2]));
      float f, j;
      CHECK(hipEventElapsedTime(&f, ev[0], ev[1]));
      CHECK(hipEventElapsedTime(&j, ev[1], ev[2]));
      if (j < bjoin) {
        bjoin = j;
        bflush = f;
      }
    }
    printf("%-40s part+flush=%.3f join=%.3f ms\n", "product join after L2 flush", bflush,
           bjoin);
  }
  printf("done\n");
  return 0;
}
