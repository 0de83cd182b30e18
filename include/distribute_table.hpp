/*
 * distribute_table.hpp — root->workers scatter and workers->root gather,
 * mirroring the reference (src/distribute_table.hpp:36-49; fixed-width
 * columns only, like the reference's distribute_table.cpp:115-248). Used by
 * the parity tests, as in the reference's test suite.
 */
#pragma once

#include "communicator.hpp"
#include "dj_cudf_types.hpp"

#include <memory>

/** Distribute a table from the root rank to all ranks (collective).
 * `global_table` is significant only on rank 0. Returns each rank's local
 * slice (contiguous row ranges, root included). */
std::unique_ptr<cudf::table> distribute_table(cudf::table_view global_table,
                                              Communicator* communicator);

/** Merge tables from all ranks onto the root (collective). Returns the
 * merged table on rank 0, nullptr elsewhere. */
std::unique_ptr<cudf::table> collect_tables(cudf::table_view table, Communicator* communicator);
