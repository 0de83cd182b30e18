/*
 * dj_rng.h — deterministic, device-independent synthetic input generator spec.
 *
 * Replaces the reference generator's curand/occupancy scheme
 * (/root/reference/generate_dataset/generate_dataset.cuh:40-260), whose output
 * depends on the GPU model (grid size is occupancy-derived, lottery scan uses
 * atomicCAS races). This restatement keeps the reference's DISTRIBUTIONS
 * (generate_dataset.cuh:137-162 doc comment):
 *   - build keys: build_n values drawn from [0, rand_max]; when
 *     uniq_build_tbl_keys, each value appears exactly once (the "lottery").
 *   - probe keys: with probability `selectivity` a key present in the build
 *     table (uniform over build rows), otherwise a key from
 *     [0, rand_max] \ {build keys} (uniform over the complement).
 *   - payload = global row index (src/generate_table.cuh:39-57).
 * but makes every row a pure function of (seed, global row index), so the CPU
 * oracle, the numpy harness, and the gfx950 HIP kernels produce identical
 * bytes on any machine.
 *
 * Mechanism: a 4-round Feistel permutation `dj_perm` over [0, L) with
 * cycle-walking (L = rand_max + 1). build_key(i) = dj_perm(i) for i < build_n
 * gives unique pseudo-random keys; the complement of the build key set is
 * exactly {dj_perm(i) : build_n <= i < L}, so non-matching probe keys are
 * drawn as dj_perm(build_n + u % (L - build_n)) — guaranteed absent from the
 * build table, uniform over the complement, no lottery array needed.
 *
 * This header is plain C99 and compiles under gcc, g++ and hipcc (functions
 * are marked __host__ __device__ when compiled as HIP).
 */
#ifndef DJ_RNG_H
#define DJ_RNG_H

#include <stdint.h>

#if defined(__HIPCC__) || defined(__HIP_DEVICE_COMPILE__)
#define DJ_HD __host__ __device__ static inline
#else
#define DJ_HD static inline
#endif

/* splitmix64 finalizer — the stateless 64-bit mixer everything derives from */
DJ_HD uint64_t dj_mix64(uint64_t x)
{
  x += 0x9E3779B97F4A7C15ULL;
  x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ULL;
  x = (x ^ (x >> 27)) * 0x94D049BB133111EBULL;
  return x ^ (x >> 31);
}

/* keyed counter hash: h(seed, stream, i) */
DJ_HD uint64_t dj_hash64(uint64_t seed, uint64_t stream, uint64_t i)
{
  return dj_mix64(seed ^ dj_mix64(stream ^ dj_mix64(i)));
}

/* uniform double in [0,1) from the top 53 bits */
DJ_HD double dj_u01(uint64_t h) { return (double)(h >> 11) * (1.0 / 9007199254740992.0); }

/* number of bits needed to cover [0, L), rounded up to an even count >= 2 */
DJ_HD int dj_perm_bits(uint64_t L)
{
  int k = 1;
  while (k < 63 && ((1ULL << k) < L)) k++;
  if (k & 1) k++;
  return k;
}

/*
 * dj_perm: bijective pseudo-random permutation of [0, L), keyed by `seed`.
 * 4-round balanced Feistel over 2^k >= L with cycle-walking back into [0, L).
 * Expected walk length < 2 iterations for any L.
 */
DJ_HD uint64_t dj_perm(uint64_t i, uint64_t L, uint64_t seed)
{
  const int k        = dj_perm_bits(L);
  const int half     = k / 2;
  const uint64_t hm  = (1ULL << half) - 1;
  uint64_t x         = i;
  do {
    uint64_t l = x >> half;
    uint64_t r = x & hm;
    for (int round = 0; round < 4; round++) {
      uint64_t f = dj_hash64(seed, (uint64_t)(0xF00D + round), r) & hm;
      uint64_t nl = r;
      uint64_t nr = l ^ f;
      l = nl;
      r = nr;
    }
    x = (l << half) | r;
  } while (x >= L);
  return x;
}

/*
 * Build table row: key + payload for global row index i (0 <= i < build_n).
 * Unique keys in [0, rand_max]; payload = i.
 * (Non-unique build keys — the reference's uniq_build_tbl_keys=false mode —
 *  use stream 3 plain uniform draws; see dj_build_key_nonuniq.)
 */
DJ_HD int64_t dj_build_key(uint64_t i, uint64_t build_n, int64_t rand_max, uint64_t seed)
{
  (void)build_n;
  return (int64_t)dj_perm(i, (uint64_t)rand_max + 1, seed);
}

DJ_HD int64_t dj_build_key_nonuniq(uint64_t i, int64_t rand_max, uint64_t seed)
{
  return (int64_t)(dj_hash64(seed, 3, i) % ((uint64_t)rand_max + 1));
}

/*
 * Probe table row key for global row index j.
 * With prob. `selectivity`: key = build_key(u1 % build_n)  (present in build).
 * Otherwise: key = dj_perm(build_n + u2 % (L - build_n))    (absent from build).
 * Requires rand_max >= build_n (so the complement is non-empty).
 */
DJ_HD int64_t dj_probe_key(uint64_t j,
                           uint64_t build_n,
                           int64_t rand_max,
                           double selectivity,
                           uint64_t seed)
{
  const uint64_t L = (uint64_t)rand_max + 1;
  double u         = dj_u01(dj_hash64(seed, 10, j));
  if (u < selectivity) {
    uint64_t idx = dj_hash64(seed, 11, j) % build_n;
    return (int64_t)dj_perm(idx, L, seed);
  } else {
    uint64_t idx = build_n + dj_hash64(seed, 12, j) % (L - build_n);
    return (int64_t)dj_perm(idx, L, seed);
  }
}

/* default seed follows the reference benchmark (generate_dataset.cuh:44) */
#define DJ_DEFAULT_SEED 1234ULL

#endif /* DJ_RNG_H */
