/*
 * distributed_join.h — C ABI of the MI355X-native distributed repartitioned
 * hash join (libdistjoin.so).
 *
 * This is the drop-in boundary for the hot path of rapidsai/distributed-join
 * (reference at /root/reference). Each entry point names the reference
 * interface it replaces (file:line). The C++ surface mirroring the
 * reference's exact signatures (distributed_inner_join / shuffle_on /
 * Communicator) lives in include/distributed_join.hpp; this C ABI is the
 * additive layer the Python/ctypes measurement harness binds (SURVEY.md
 * §8b: "a thin C ABI ... exported for the Python/ctypes measurement
 * harness — additive, not replacing the C++ surface").
 *
 * Conventions: plain pointers + sizes, no torch/cudf types. Pointers named
 * d_* are DEVICE pointers (HIP), h_* are host pointers. All compute calls
 * are stream-ordered on the library's internal HIP stream; dj_sync()
 * synchronizes it. Errors print and exit(1), mirroring the reference's
 * error.hpp:22-99 contract.
 */
#ifndef DISTRIBUTED_JOIN_H
#define DISTRIBUTED_JOIN_H

#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

/* ---------------- device & memory management (replaces RMM pool usage,
 * reference src/setup.cpp:51-106; pool semantics via hipMallocAsync pool) */
int dj_device_count(void);
void dj_set_device(int device);
void* dj_dmalloc(int64_t bytes);
void dj_dfree(void* ptr);
void dj_memcpy_h2d(void* d_dst, const void* h_src, int64_t bytes);
void dj_memcpy_d2h(void* h_dst, const void* d_src, int64_t bytes);
void dj_memcpy_d2d(void* d_dst, const void* d_src, int64_t bytes);
void dj_sync(void);

/* ---------------- deterministic synthetic inputs (replaces
 * generate_dataset.cuh:40-260 / generate_table.cuh:39-57; spec: dj_rng.h).
 * Writes rows [row0, row0+nrows) of the GLOBAL table into device arrays. */
void dj_generate_build(int64_t* d_keys, int64_t* d_pay, int64_t n_global, int64_t rand_max,
                       uint64_t seed, int uniq, int64_t row0, int64_t nrows);
void dj_generate_probe(int64_t* d_keys, int64_t* d_pay, int64_t build_n_global,
                       int64_t rand_max, double selectivity, uint64_t seed, int64_t row0,
                       int64_t nrows);

/* ---------------- stable hash partition (replaces cudf::hash_partition as
 * called at distributed_join.cpp:213-225 and shuffle_on.cpp:59-60).
 * hash_fn: 0 = MURMUR3, 1 = IDENTITY (dj_hash.h spec). nparts in [1,64].
 * h_offsets: HOST int64[nparts+1] partition offsets (synchronizes). */
int64_t dj_partition_scratch_bytes(int64_t n, int nparts);
void dj_hash_partition(const int64_t* d_keys, const int64_t* d_pay, int64_t n, int nparts,
                       int hash_fn, uint32_t hash_seed, int64_t* d_out_keys,
                       int64_t* d_out_pay, int64_t* h_offsets, void* d_scratch);

/* ---------------- local inner join (replaces cudf::inner_join as called at
 * distributed_join.cpp:71-83). Open-addressing table of nslots =
 * dj_join_table_slots(ln) interleaved 16 B {key,val} pairs; caller
 * allocates d_table (2*nslots int64), d_error (1 int32, zeroed), d_counter
 * (1 int64, zeroed). Output columns are (lkey, lpay, rkey, rpay) — left
 * columns then right columns with the join key duplicated, row order
 * unspecified (reference pin: compare_against_single_gpu.cu:163-174).
 * Build keys equal to -1 (the empty-slot sentinel) are a loud error. */
int64_t dj_join_table_slots(int64_t ln);
void dj_join_table_init(int64_t* d_table, int64_t nslots);
void dj_join_build(const int64_t* d_lk, const int64_t* d_lp, int64_t ln,
                   int64_t* d_table, int64_t nslots, int* d_error);
void dj_join_probe(const int64_t* d_rk, const int64_t* d_rp, int64_t rn,
                   const int64_t* d_table, int64_t nslots,
                   int64_t* d_out0, int64_t* d_out1, int64_t* d_out2, int64_t* d_out3,
                   int64_t cap, int64_t* d_counter);
int64_t dj_read_counter_i64(const int64_t* d_counter);
int dj_read_error_i32(const int* d_error);

/* ---------------- bucketed LDS local join (the PRODUCT local-join path;
 * same drop-in semantics as above, different engine: both tables are
 * bucket-partitioned so each bucket's hash table lives in LDS — no
 * HBM-resident table, all hot random access on-chip). Buckets whose build
 * side exceeds the LDS capacity (data skew / heavy duplicates) fall back to
 * the global-table path transparently. d_counter/d_error as above. */
int64_t dj_bucket_join_scratch_bytes(int64_t ln, int64_t rn);
void dj_bucket_local_join(const int64_t* d_lk, const int64_t* d_lp, int64_t ln,
                          const int64_t* d_rk, const int64_t* d_rp, int64_t rn,
                          int64_t* d_out0, int64_t* d_out1, int64_t* d_out2,
                          int64_t* d_out3, int64_t cap, int64_t* d_counter, int* d_error,
                          void* d_scratch);

/* Convenience one-call local join over the bucketed path (allocates its own
 * scratch; for tests and smoke, not the bench timed region). Returns the
 * match count; writes at most cap rows. */
int64_t dj_local_inner_join(const int64_t* d_lk, const int64_t* d_lp, int64_t ln,
                            const int64_t* d_rk, const int64_t* d_rp, int64_t rn,
                            int64_t* d_out0, int64_t* d_out1, int64_t* d_out2,
                            int64_t* d_out3, int64_t cap);
/* Same over the global-table build/probe path (the fallback engine),
 * exposed so tests can cross-check both engines. */
int64_t dj_local_inner_join_global(const int64_t* d_lk, const int64_t* d_lp, int64_t ln,
                                   const int64_t* d_rk, const int64_t* d_rp, int64_t rn,
                                   int64_t* d_out0, int64_t* d_out1, int64_t* d_out2,
                                   int64_t* d_out3, int64_t cap);

/* ---------------- phase timing (hipEvent pairs around every kernel launch;
 * replaces the reference's report_timing wall-clock prints,
 * distributed_join.cpp:120-130,235-240) */
enum dj_phase {
  DJ_PHASE_GENERATE = 0,
  DJ_PHASE_PART_COUNT = 1,
  DJ_PHASE_PART_SCAN = 2,
  DJ_PHASE_PART_SCATTER = 3,
  DJ_PHASE_TABLE_INIT = 4,
  DJ_PHASE_BUILD = 5,
  DJ_PHASE_PROBE = 6,
  DJ_PHASE_COMM = 7,
  DJ_PHASE_CONCAT = 8,
  DJ_PHASE_BUCKET_COUNT = 9,
  DJ_PHASE_BUCKET_SCAN = 10,
  DJ_PHASE_BUCKET_SCATTER = 11,
  DJ_PHASE_JOIN_FUSED = 12,
  DJ_PHASE_COUNT_ = 13
};
void dj_timing_enable(int on);
void dj_timing_reset(void);
/* total milliseconds and number of launches recorded for a phase
 * (synchronizes the stream) */
double dj_timing_total_ms(int phase);
int64_t dj_timing_launches(int phase);

/* ---------------- RCCL communicator over xGMI (replaces the reference's
 * UCX/NCCL Communicator, communicator.hpp:31-90 + communicator.cpp:783-869;
 * bootstrap id is exchanged by the launcher, replacing MPI_Bcast of
 * ncclUniqueId at communicator.cpp:799-817). */
int dj_rccl_unique_id_bytes(void);
void dj_rccl_get_unique_id(void* h_id_bytes);
void dj_comm_init(int rank, int size, const void* h_id_bytes);
void dj_comm_finalize(void);
int dj_comm_rank(void);
int dj_comm_size(void);
/* grouped peer-slice exchange: for each peer p, send send_counts[p] int64
 * elements from d_send + send_offsets[p], receive recv_counts[p] into
 * d_recv + recv_offsets[p] (counts/offsets are HOST arrays, elements).
 * Implements send/recv_data_by_offset + all_to_all_comm
 * (all_to_all_comm.cpp:126-189,307-478) as one ncclGroupStart/End of
 * ncclSend/ncclRecv on the comm stream — no staging copies (RCCL takes
 * device pointers directly; the reference's 256 B staging at
 * communicator.cpp:820-869 is unnecessary over xGMI). */
void dj_all_to_all_i64(const int64_t* d_send, const int64_t* h_send_offsets,
                       int64_t* d_recv, const int64_t* h_recv_offsets);
/* exchange per-peer row counts (host arrays of size comm_size):
 * replaces communicate_sizes (all_to_all_comm.cpp:54-100) */
void dj_exchange_sizes(const int64_t* h_send_counts, int64_t* h_recv_counts);
/* world-1 RCCL init + grouped self send/recv of n bytes through the
 * RCCLCommunicator path (the same calls the N>1 exchange makes per peer);
 * returns 0 on success, 1 on payload mismatch */
int dj_rccl_selftest(int64_t n);

/* cascaded codec roundtrip (test hook): compress count elements of
 * elem_size (4/8) from d_in with {num_rles, num_deltas, use_bp} cascaded
 * passes, decompress into d_out, return the wire size in bytes */
int64_t dj_compress_roundtrip(const void* d_in, int64_t count, int elem_size, int num_rles,
                              int num_deltas, int use_bp, void* d_out);

#ifdef __cplusplus
}
#endif

#endif /* DISTRIBUTED_JOIN_H */
