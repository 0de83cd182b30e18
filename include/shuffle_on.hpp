/*
 * shuffle_on.hpp — hash shuffle, mirroring the reference's signatures
 * (reference: src/shuffle_on.hpp:44-64) over our re-declared cudf types.
 * Placement spec: hash % group_size per dj_hash.h (MurmurHash3_32 of the
 * int64 key bytes / identity truncation) — the reference pins identity-hash
 * placement only (test_shuffle_on.cpp:78-83); MurmurHash3 placement is
 * pinned to our documented spec.
 */
#pragma once

#include "all_to_all_comm.hpp"
#include "communicator.hpp"
#include "compression.hpp"
#include "dj_cudf_types.hpp"

#include <memory>
#include <vector>

std::unique_ptr<cudf::table> shuffle_on(cudf::table_view const& input,
                                        std::vector<cudf::size_type> const& on_columns,
                                        CommunicationGroup comm_group,
                                        Communicator* communicator,
                                        std::vector<ColumnCompressionOptions> compression_options,
                                        cudf::hash_id hash_function = cudf::hash_id::HASH_MURMUR3,
                                        uint32_t hash_seed          = cudf::DEFAULT_HASH_SEED,
                                        bool report_timing          = false,
                                        void* preallocated_pinned_buffer = nullptr);

/* all-ranks, stride-1 convenience overload (shuffle_on.hpp:57-64) */
std::unique_ptr<cudf::table> shuffle_on(cudf::table_view const& input,
                                        std::vector<cudf::size_type> const& on_columns,
                                        Communicator* communicator,
                                        std::vector<ColumnCompressionOptions> compression_options,
                                        cudf::hash_id hash_function = cudf::hash_id::HASH_MURMUR3,
                                        uint32_t hash_seed          = cudf::DEFAULT_HASH_SEED,
                                        bool report_timing          = false,
                                        void* preallocated_pinned_buffer = nullptr);
