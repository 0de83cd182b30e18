/*
 * all_to_all_comm.hpp — the reference's all-to-all plan/execution layer
 * (reference: src/all_to_all_comm.hpp:36-362), MI355X-native.
 *
 * Kept: CommunicationGroup (grid/stride rank subsets, hpp:72-113),
 * communicate_sizes (hpp:115-135), AllToAllCommBuffer plan records
 * (hpp:139-183), append/exec/postprocess entry points, and the
 * AllToAllCommunicator class (hpp:274-362) with the same collective
 * discipline (every call collective over the group's ranks;
 * launch_communication blocks the host — hpp:331).
 *
 * Changed vs reference: no MPI (size exchange goes through the Communicator
 * on a small device staging buffer), no nvcomp (CompressionMethod::none
 * only — see compression.hpp), no UCX registration. The exchange itself is
 * grouped per-peer send/recv of contiguous device slices over RCCL/xGMI
 * (all_to_all_comm.cpp:126-189 pattern).
 */
#pragma once

#include "communicator.hpp"
#include "compression.hpp"
#include "dj_cudf_types.hpp"

#include <cstdint>
#include <memory>
#include <vector>

enum COMM_TAGS { placeholder_tag, exchange_size_tag };

/* mirrors reference all_to_all_comm.hpp:72-113 (grid/stride subsets of the
 * world; grid_size must divide world size, stride must divide grid_size).
 * The 2-arg form reads the current rank from default_communicator(),
 * replacing the reference's MPI_COMM_WORLD global. */
class CommunicationGroup {
 public:
  CommunicationGroup(int grid_size, int stride = 1);
  CommunicationGroup(int grid_size, int stride, int mpi_rank);

  int size() const { return grid_size / stride; }
  int get_global_rank(int local_idx) const { return group_start + local_idx * stride; }
  int get_local_idx() const { return (mpi_rank - group_start) / stride; }

 private:
  int mpi_rank;
  int group_start;
  int grid_size;
  int stride;
};

/* mirrors all_to_all_comm.hpp:115-135; host-side counts, exchanged off the
 * hot data path (reference used MPI host buffers, all_to_all_comm.cpp:54-100;
 * ours stages through a small device buffer on the communicator) */
void communicate_sizes(std::vector<int64_t> const& send_offset,
                       std::vector<int64_t>& recv_offset,
                       CommunicationGroup comm_group,
                       Communicator* communicator);

void communicate_sizes(std::vector<cudf::size_type> const& send_offset,
                       std::vector<int64_t>& recv_offset,
                       CommunicationGroup comm_group,
                       Communicator* communicator);

void warmup_all_to_all(Communicator* communicator);

/* mirrors all_to_all_comm.hpp:139-183 (compression buffers unused while only
 * CompressionMethod::none is supported) */
struct AllToAllCommBuffer {
  const void* send_buffer;
  void* recv_buffer;
  std::vector<int64_t> send_offsets;
  std::vector<int64_t> recv_offsets;
  cudf::data_type dtype;
  CompressionMethod compression_method;
  nvcompCascadedFormatOpts cascaded_format;

  AllToAllCommBuffer(const void* send_buffer,
                     void* recv_buffer,
                     std::vector<int64_t> send_offsets,
                     std::vector<int64_t> recv_offsets,
                     cudf::data_type dtype,
                     CompressionMethod compression_method,
                     nvcompCascadedFormatOpts cascaded_format)
    : send_buffer(send_buffer),
      recv_buffer(recv_buffer),
      send_offsets(std::move(send_offsets)),
      recv_offsets(std::move(recv_offsets)),
      dtype(dtype),
      compression_method(compression_method),
      cascaded_format(cascaded_format)
  {
  }
};

/* fixed-width variant (all_to_all_comm.hpp:229-234) */
void append_to_all_to_all_comm_buffers(cudf::table_view input,
                                       cudf::mutable_table_view output,
                                       std::vector<cudf::size_type> const& send_offsets,
                                       std::vector<int64_t> const& recv_offsets,
                                       std::vector<AllToAllCommBuffer>& all_to_all_comm_buffers,
                                       std::vector<ColumnCompressionOptions> compression_options);

/* mirrors all_to_all_comm.hpp:253-258; nonblocking when the communicator
 * groups by batch — enclose in communicator->start()/stop() */
void all_to_all_comm(std::vector<AllToAllCommBuffer>& all_to_all_comm_buffers,
                     CommunicationGroup comm_group,
                     Communicator* communicator,
                     bool include_current_rank        = true,
                     bool report_timing               = false,
                     void* preallocated_pinned_buffer = nullptr);

/* mirrors all_to_all_comm.hpp:265-269 */
void postprocess_all_to_all_comm(std::vector<AllToAllCommBuffer>& all_to_all_comm_buffers,
                                 CommunicationGroup comm_group,
                                 Communicator* communicator,
                                 bool include_current_rank = true,
                                 bool report_timing        = false);

/* mirrors all_to_all_comm.hpp:274-362 */
class AllToAllCommunicator {
 public:
  AllToAllCommunicator(cudf::table_view input_table,
                       std::vector<cudf::size_type> offsets,
                       CommunicationGroup comm_group,
                       Communicator* communicator,
                       std::vector<ColumnCompressionOptions> compression_options,
                       bool explicit_copy_to_current_rank = false);

  AllToAllCommunicator(cudf::table_view input_table,
                       std::vector<cudf::size_type> offsets,
                       Communicator* communicator,
                       std::vector<ColumnCompressionOptions> compression_options,
                       bool explicit_copy_to_current_rank = false);

  AllToAllCommunicator(const AllToAllCommunicator&) = delete;
  AllToAllCommunicator& operator=(const AllToAllCommunicator&) = delete;
  AllToAllCommunicator(AllToAllCommunicator&&)                 = default;

  /** Allocate the received table (synchronous). */
  std::unique_ptr<cudf::table> allocate_communicated_table();

  /** Collective; blocks the host until the exchange completes. */
  void launch_communication(cudf::mutable_table_view communicated_table,
                            bool report_timing               = false,
                            void* preallocated_pinned_buffer = nullptr);

 private:
  cudf::table_view input_table;
  CommunicationGroup comm_group;
  Communicator* communicator;
  bool explicit_copy_to_current_rank;
  std::vector<cudf::size_type> send_offsets;
  std::vector<int64_t> recv_offsets;
  std::vector<ColumnCompressionOptions> compression_options;
  /* strings machinery (reference all_to_all_comm.hpp:349-360: per string
   * column the char-offset boundaries per peer, plus device buffers of row
   * sizes to send / received — sizes, not offsets, go on the wire). Opaque
   * here; defined in dj_cpp_api.hip. */
  struct StringsState;
  std::shared_ptr<StringsState> strings;
  friend struct AllToAllCommunicatorAccess;
};
