/*
 * compression.hpp — mirror of the reference's compression-options surface
 * (reference: src/compression.hpp:42-58, 253-361). Executable here:
 * CompressionMethod::none and cascaded with {0|1 delta passes} + bitpack
 * (our own wire format, dj_compress.hip — the nvcomp format is
 * parity-unpinned); RLE passes and lz4 raise
 * std::runtime_error at use with a clear message.
 */
#pragma once

#include "communicator.hpp"
#include "dj_cudf_types.hpp"

#include <stdexcept>
#include <vector>

enum class CompressionMethod { none, cascaded, lz4 };

/* our POD stand-in for nvcomp's nvcompCascadedFormatOpts */
struct nvcompCascadedFormatOpts {
  int num_RLEs{0};
  int num_deltas{0};
  int use_bp{0};
};

/* mirrors reference compression.hpp:44-58 */
struct ColumnCompressionOptions {
  CompressionMethod compression_method;
  nvcompCascadedFormatOpts cascaded_format;
  std::vector<ColumnCompressionOptions> children_compression_options;

  ColumnCompressionOptions(CompressionMethod compression_method     = CompressionMethod::none,
                           nvcompCascadedFormatOpts cascaded_format = {},
                           std::vector<ColumnCompressionOptions> children_compression_options = {})
    : compression_method(compression_method),
      cascaded_format(cascaded_format),
      children_compression_options(children_compression_options)
  {
  }
};

/* mirrors generate_compression_options_distributed (compression.cpp:97-150):
 * per-column options for a table. compression=false -> `none` for every
 * column; compression=true -> bitpack cascaded (fixed policy; use
 * generate_auto_select_compression_options for sampling-based choice). */
std::vector<ColumnCompressionOptions> generate_compression_options_distributed(
  cudf::table_view input, bool compression);

/* mirrors generate_none_compression_options (compression.hpp:313): `none`
 * for every column (STRING columns get two `none` children) */
std::vector<ColumnCompressionOptions> generate_none_compression_options(
  cudf::table_view input_table);

/* mirrors generate_auto_select_compression_options (compression.hpp:302,
 * compression.cpp:36-69): per-column sampling selection over the executable
 * cascaded schemes ({0|1 deltas} + bitpack); STRING columns select on the
 * offsets child and never compress chars. Data-dependent, not parity-pinned
 * (the reference sampled through nvcomp's CascadedSelector). */
std::vector<ColumnCompressionOptions> generate_auto_select_compression_options(
  cudf::table_view input_table);

/* mirror broadcast_compression_options (compression.hpp:328,345): rank 0's
 * choices distributed to every rank — over the registered Communicator
 * (the reference used MPI_Bcast on MPI_COMM_WORLD). Non-root ranks may pass
 * empty/default options. */
ColumnCompressionOptions broadcast_compression_options(cudf::column_view input_column,
                                                       ColumnCompressionOptions input_options);
std::vector<ColumnCompressionOptions> broadcast_compression_options(
  cudf::table_view input_table, std::vector<ColumnCompressionOptions> input_options);
/* additive overloads taking the Communicator explicitly (for user-supplied
 * transports; the 2-argument forms use the registered default) */
ColumnCompressionOptions broadcast_compression_options(cudf::column_view input_column,
                                                       ColumnCompressionOptions input_options,
                                                       Communicator* comm);
std::vector<ColumnCompressionOptions> broadcast_compression_options(
  cudf::table_view input_table, std::vector<ColumnCompressionOptions> input_options,
  Communicator* comm);
