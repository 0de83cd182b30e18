/*
 * dj_kernels.hpp — host-side launchers for the gfx950 HIP kernels of the
 * distributed repartitioned hash join hot path.
 *
 * These replace the third-party cuDF 0.19 kernels the reference calls on its
 * hot path (SURVEY.md §2 third-party kernel table):
 *   - hash_partition  <- cudf::hash_partition (distributed_join.cpp:213-225,
 *                        shuffle_on.cpp:59-60)
 *   - build/probe     <- cudf::inner_join (distributed_join.cpp:79)
 *   - generate_*      <- generate_dataset.cuh:40-260 (restated deterministic)
 *
 * All launchers are stream-ordered on the given hipStream_t and operate on
 * raw device pointers (columnar int64 arrays). No torch types.
 */
#pragma once

#include <hip/hip_runtime.h>

#include <cstdint>

namespace dj {

/* int64 value marking an empty hash-table slot; build rows carrying it are
 * skipped in-table and joined out-of-band (neg1_cross_join below). Set via
 * 0xFF memset. */
constexpr int64_t kEmptyKey = -1;

/* nparts = world_size x over_decom; the stable wave-ballot partition covers
 * any nparts <= 1024 (lane q owns partitions q, q+64, ... — cost linear in
 * nparts). The reference's own tests drive up to 80 (8 ranks x od 10,
 * compare_against_single_gpu.cu:237-268). */
constexpr int kMaxPartitions = 1024;

/* ----- synthetic inputs (deterministic, spec in dj_rng.h) ----- */
void generate_build(int64_t* d_keys, int64_t* d_pay, int64_t n_global, int64_t rand_max,
                    uint64_t seed, bool uniq, int64_t row0, int64_t nrows, hipStream_t s);
void generate_probe(int64_t* d_keys, int64_t* d_pay, int64_t build_n_global, int64_t rand_max,
                    double selectivity, uint64_t seed, int64_t row0, int64_t nrows, hipStream_t s);

/* ----- stable hash partition (histogram + scan + wave-ballot scatter) ----- */
size_t hash_partition_scratch_bytes(int64_t n, int nparts);
/* sub-steps, so the C ABI can time each kernel separately */
void partition_count(const int64_t* d_keys, int64_t n, int nparts, int hash_fn,
                     uint32_t hash_seed, void* d_scratch, hipStream_t s);
void partition_scan(int64_t n, int nparts, void* d_scratch, int64_t* d_offsets, hipStream_t s);
void partition_scatter(const int64_t* d_keys, const int64_t* d_pay, int64_t n, int nparts,
                       int hash_fn, uint32_t hash_seed, const int64_t* d_offsets,
                       void* d_scratch, int64_t* d_out_keys, int64_t* d_out_pay,
                       hipStream_t s);
/* d_offsets: device array of nparts+1 int64 partition offsets (offsets[0]=0).
 * Stable: rows keep input order inside each partition. nparts <= 1024. */
void hash_partition(const int64_t* d_keys, const int64_t* d_pay, int64_t n, int nparts,
                    int hash_fn, uint32_t hash_seed, int64_t* d_out_keys, int64_t* d_out_pay,
                    int64_t* d_offsets, void* d_scratch, hipStream_t s);

/* ----- local inner join: open-addressing build + probe-append ----- */
/* Number of table slots for ln build rows (power of two, <=50% fill).
 * The table buffer holds nslots interleaved 16 B {key,val} pairs
 * (2*nslots int64). */
int64_t join_table_slots(int64_t ln);
/* Initialize table pairs to kEmptyKey (async memset over 16*nslots bytes). */
void join_table_init(int64_t* d_table, int64_t nslots, hipStream_t s);
/* Insert build rows. d_error (device int32) set to 1 if any key==kEmptyKey. */
void join_build(const int64_t* d_lk, const int64_t* d_lp, int64_t ln, int64_t* d_table,
                int64_t nslots, int* d_error, hipStream_t s);
/* Probe rows; append matches (lkey, lpay, rkey, rpay) to the 4 output
 * columns at positions drawn from d_counter (device int64, caller-zeroed),
 * one wave-aggregated atomic per emit round. Rows beyond `cap` are counted
 * but not written (caller re-runs bigger). */
void join_probe(const int64_t* d_rk, const int64_t* d_rp, int64_t rn, const int64_t* d_table,
                int64_t nslots, int64_t* d_out0, int64_t* d_out1, int64_t* d_out2,
                int64_t* d_out3, int64_t cap, int64_t* d_counter, hipStream_t s);

/* ----- bucketed LDS join (the product local-join path; see dj_kernels.hip
 * "bucketed LDS join" comment block) ----- */
constexpr int kBucketBlocks = 512;
constexpr int kSubBuckets = 256;
constexpr int kJoinBucketRowCap = 1536;  // 75% of the 2048-slot LDS table
int bucket_count_for(int64_t ln, int64_t rn);
/* decomposition B = PA x F: PA pass-A groups (<=1024, one per thread in the
 * scans), F pass-B sub-buckets (256 up to B=262144 — the historical shape —
 * then 512/1024, bit fields disjoint per subF_of) */
inline int bucket_groups_for(int B)
{
  int PA = B / 256;
  if (PA < 1) PA = 1;
  if (PA > 1024) PA = 1024;
  return PA;
}
/* pass-A slack segment capacity (rows per group): mean + ~6% + 1024 covers
 * hash-uniform inputs w.h.p.; skewed inputs overflow -> bit 2 of
 * any_overflow -> the caller redoes the join exactly */
inline int64_t slack_capA(int64_t n, int PA)
{
  const int64_t m = n / PA;
  return m + m / 16 + 1024;
}
/* per-bucket slack capacity for the count-free pass B (mean + 16 sqrt(mean),
 * rounded to 8). 7 sigma covers Poisson (unique-ish keys) but duplicate keys
 * inflate bucket variance by ~(1 + multiplicity): TPC-H lineitem (m ~ 4)
 * overflowed 7 sigma EVERY step, wasting the slack attempt before the exact
 * compact retry. 16 sqrt(lam) covers multiplicity up to ~8 for ~25% more
 * slack memory; beyond that bit 2 of any_overflow still routes to the
 * compact retry (same contract as slack_capA). */
inline int64_t slack_capB(int64_t n, int B)
{
  const int64_t lam = n / B < 1 ? 1 : n / B;
  int64_t s = 1;
  while (s * s < lam) s++;  // ceil(sqrt(lam))
  int64_t cap = lam + 16 * s + 16;
  return (cap + 7) & ~(int64_t)7;
}
/* Two-level non-stable partition into B buckets (B = PA*F per
 * bucket_groups_for; up to 1024x1024 = 1M buckets for ~800M-row tables) of
 * interleaved 16 B {key,payload} pairs. d_tmp_pairs: pass-A staging —
 * longlong2[PA * slack_capA(n, PA)] when d_any_overflow is non-null (the
 * slack path: no count pass; group skew beyond the slack sets BIT 2 of
 * *d_any_overflow and the partition output is INCOMPLETE — caller must redo
 * the whole join), else longlong2[n] (exact two-pass path). d_counts:
 * u32[kBucketBlocks*PA]; d_totals: u32[PA]; d_segoff: int64[PA+1];
 * d_offsets: int64[B+1]. */
void bucket_partition2(const int64_t* d_keys, const int64_t* d_pay, int64_t n, int B,
                       longlong2* d_tmp_pairs, uint32_t* d_counts, uint32_t* d_totals,
                       int64_t* d_segoff, int64_t* d_offsets, longlong2* d_out_pairs,
                       int* d_any_overflow, hipStream_t s);
/* Count-free two-level slack partition (round 2, the product local-join
 * path): pass A as bucket_partition2's slack path, then pass B scatters each
 * group's rows into per-bucket SLACK segments at analytic starts
 * b*capB (capB = slack_capB(n, B)) with LDS cursors — no seghist sweep (the
 * r1 pass B fetched full 16 B lines just to histogram 8 B keys: 3.28 GB PMC
 * fetch vs 2.4 GB algorithmic). Bucket b holds d_lens[b] rows at
 * d_out_pairs[b*capB]. Overflow of any slack segment sets bit 2 of
 * *d_any_overflow (partition output incomplete -> caller redoes exactly).
 * Requires PA >= 2, n < 2^32, PA*slack_capA(n,PA)+... and B*capB < 2^32
 * (checked; errors loudly otherwise). d_tmp_pairs: longlong2[PA*capA];
 * d_cursors: u32[PA]. */
void bucket_partition2_slack(const int64_t* d_keys, const int64_t* d_pay, int64_t n, int B,
                             longlong2* d_tmp_pairs, uint32_t* d_cursors, int64_t capB,
                             longlong2* d_out_pairs, uint32_t* d_lens, int* d_any_overflow,
                             hipStream_t s);
/* both tables through one pass-A and one pass-B launch (tail-wave fill);
 * needs a private tmp/cursor set per table */
void bucket_partition2_slack_pair(const int64_t* d_k0, const int64_t* d_p0, int64_t n0,
                                  longlong2* d_tmp0, uint32_t* d_cur0, int64_t capB0,
                                  longlong2* d_out0, uint32_t* d_len0, const int64_t* d_k1,
                                  const int64_t* d_p1, int64_t n1, longlong2* d_tmp1,
                                  uint32_t* d_cur1, int64_t capB1, longlong2* d_out1,
                                  uint32_t* d_len1, int B, int* d_any_overflow,
                                  hipStream_t s);
/* Fused per-bucket LDS build+probe over bucketed pair tables. table_slots:
 * 2048 (2 blocks/CU, bucket cap 1536 build rows = kJoinBucketRowCap) or
 * 4096 (1 block/CU, cap 3072 — for the fused wire path when the PA*F
 * fan-out cap leaves big buckets). Buckets whose build side exceeds the cap
 * set overflow_flags[b]/any_overflow and are skipped (host runs the
 * global-table path on them / redoes the batch). */
void lds_join(const longlong2* d_lrows, const int64_t* d_loff, const longlong2* d_rrows,
              const int64_t* d_roff, int B, int table_slots, int64_t* d_out0, int64_t* d_out1,
              int64_t* d_out2, int64_t* d_out3, int64_t cap, int64_t* d_counter,
              uint32_t* d_overflow_flags, int* d_any_overflow, int* d_error, hipStream_t s);
/* Same fused join over the slack bucket layout of bucket_partition2_slack:
 * bucket b = d_lrows[b*capL .. +d_llen[b]) x d_rrows[b*capR .. +d_rlen[b]).
 * Contract: B must be a multiple of 4 (the in-kernel flush-group size) —
 * pad d_llen/d_rlen with zero-length buckets; errors loudly otherwise. */
void lds_join_slack(const longlong2* d_lrows, const uint32_t* d_llen, int64_t capL,
                    const longlong2* d_rrows, const uint32_t* d_rlen, int64_t capR, int B,
                    int table_slots, int64_t* d_out0, int64_t* d_out1, int64_t* d_out2,
                    int64_t* d_out3, int64_t cap, int64_t* d_counter,
                    uint32_t* d_overflow_flags, int* d_any_overflow, int* d_error,
                    hipStream_t s);
/* Global-table build/probe over interleaved pair inputs (skew fallback). */
void join_build_pairs(const longlong2* d_rows, int64_t ln, int64_t* d_table, int64_t nslots,
                      int* d_error, hipStream_t s);
void join_probe_pairs(const longlong2* d_rows, int64_t rn, const int64_t* d_table,
                      int64_t nslots, int64_t* d_out0, int64_t* d_out1, int64_t* d_out2,
                      int64_t* d_out3, int64_t cap, int64_t* d_counter, hipStream_t s);

/* ----- strings-column kernels (dj_strings.hip; see its header comment for
 * the reference helpers each replaces) ----- */
void sizes_from_offsets(const int32_t* d_offsets, int64_t n, int32_t* d_sizes, hipStream_t s);
/* true 64-bit total of int32 sizes -> d_total (one i64; caller zeroes) —
 * overflow guard for the int32 offsets convention */
void sum_sizes_i64(const int32_t* d_sizes, int64_t n, int64_t* d_total, hipStream_t s);
size_t offsets_from_sizes_scratch_bytes(int64_t n);
void offsets_from_sizes(const int32_t* d_sizes, int64_t n, int32_t* d_offsets, void* d_scratch,
                        hipStream_t s);
void gather_sizes_starts(const int32_t* d_src_off, const int64_t* d_idx, int64_t n,
                         int32_t* d_sizes, int32_t* d_starts, hipStream_t s);
void gather_chars_from_starts(const uint8_t* d_src_chars, const int32_t* d_starts, int64_t n,
                              const int32_t* d_dst_off, uint8_t* d_dst_chars, hipStream_t s);
void gather_sizes(const int32_t* d_src_off, const int64_t* d_idx, int64_t n, int32_t* d_sizes,
                  hipStream_t s);
void make_test_string_sizes(const int64_t* d_keys, int64_t n, int32_t* d_sizes, hipStream_t s);
void fill_test_strings(const int64_t* d_keys, int64_t n, const int32_t* d_offsets,
                       uint8_t* d_chars, hipStream_t s);

/* ----- fused rank+group partition + segment-list pass B (fast2 wire path;
 * see dj_kernels.hip "fused rank+group partition" block) ----- */
void fused_partition(const int64_t* d_keys, const int64_t* d_pay, int64_t n, int nparts_rank,
                     uint32_t seed, int PA, uint32_t* d_counts, uint32_t* d_totals,
                     int64_t* d_offsets, int64_t* d_out_keys, int64_t* d_out_pay,
                     hipStream_t s);
void subpart_lists(const int64_t* d_keys, const int64_t* d_pay, const int64_t* d_seg_bounds,
                   int nseg, int PA, int F, const int64_t* d_group_base,
                   longlong2* d_out_pairs, int64_t* d_bucket_offsets, hipStream_t s);

/* ----- cascaded codec (dj_compress.hip; wire format in its header) -----
 * 32-byte slice header. scheme bit0 = delta, bit1 = RLE (values subslice
 * and run-lengths subslice follow, each bitpacked; nruns/len_bits valid
 * iff RLE). bits == 0xFFFF means stored raw (scheme ignored). */
struct CompSliceHeader {
  uint32_t bits;      // value bit width (0xFFFF = raw)
  uint32_t scheme;    // bit0 delta, bit1 RLE
  uint64_t count;     // original element count
  uint64_t nruns;     // RLE only
  uint32_t len_bits;  // RLE only: run-length bit width
  uint32_t reserved;
};
size_t compress_bound(int64_t count, int elem_size);
size_t compress_scratch_bytes(int64_t count);
/* num_rles > 0 requires d_scratch (compress_scratch_bytes(count)); the
 * packed-vs-raw decision happens on-device, fully stream-ordered. The
 * final 32 B header at d_out carries everything needed for
 * compressed_size_from_header after one sync. */
void compress_slice_async(const void* d_in, int64_t count, int elem_size, int num_rles,
                          int num_deltas, int use_bp, uint8_t* d_out, void* d_scratch,
                          hipStream_t s);
size_t compressed_size_from_header(const CompSliceHeader& h, int elem_size);
void decompress_slice_async(const uint8_t* d_comp, const CompSliceHeader& h, int elem_size,
                            void* d_out, void* d_scratch, hipStream_t s);

/* ----- small utilities ----- */
void fill_i64(int64_t* d_dst, int64_t value, int64_t n, hipStream_t s);
/* out-of-band join of rows whose key equals the reserved empty sentinel
 * (-1): every table path skips them (setting the saw-sentinel flag); this
 * collects both sides' -1 payloads and appends their cross product to the
 * output, advancing *d_counter. Host-synchronous; rare path. */
void neg1_cross_join(const int64_t* d_lk, const int64_t* d_lp, int64_t ln,
                     const int64_t* d_rk, const int64_t* d_rp, int64_t rn, int64_t* d_out0,
                     int64_t* d_out1, int64_t* d_out2, int64_t* d_out3, int64_t cap,
                     int64_t* d_counter, hipStream_t s);

}  // namespace dj
