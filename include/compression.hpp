/*
 * compression.hpp — signature-level mirror of the reference's compression
 * options (reference: src/compression.hpp:42-58). Per the hot-path scope
 * (nvcomp layer replaced by plain RCCL over xGMI;
 * CompressionMethod::none kept in signatures for drop-in), only
 * CompressionMethod::none is executable in this build; requesting cascaded
 * or lz4 raises std::runtime_error at use. A HIP cascaded codec is the
 * ranked-next item.
 */
#pragma once

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
 * per-column options for a table. With compression=false returns `none` for
 * every column; compression=true is not implemented in this build. */
std::vector<ColumnCompressionOptions> generate_compression_options_distributed(
  cudf::table_view input, bool compression);
