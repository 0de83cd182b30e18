/*
 * dj_cudf_types.hpp — OUR minimal re-declarations of the cuDF types the
 * reference's public API surface uses (cudf::table, cudf::table_view,
 * cudf::column, cudf::size_type, cudf::data_type, cudf::hash_id).
 *
 * This is NOT NVIDIA cuDF. Per the drop-in contract ("the
 * reference's exact C++ signatures, our own headers re-declaring the types
 * cuDF provided"), code written against the reference's
 * distributed_inner_join / shuffle_on / Communicator interfaces compiles and
 * links against these headers unchanged. Only the subset of the cuDF API the
 * reference's interface touches is provided; column data lives in HIP device
 * memory (MI355X HBM3E). Do not link this together with real libcudf.
 *
 * v1 supports fixed-width INT32/INT64 columns (the BASELINE configs 1-3
 * types); STRING is declared for the config-4 path.
 */
#pragma once

#include <cstdint>
#include <memory>
#include <stdexcept>
#include <vector>

namespace cudf {

using size_type = int32_t;

enum class type_id : int32_t {
  EMPTY = 0,
  INT8,
  INT32,
  INT64,
  STRING,
  /* chrono types (reference dtype coverage, compare_against_single_gpu.cu:
   * 237-268): joins operate on the integer representation — DAYS is 4-byte,
   * the rest 8-byte, matching cuDF's reps */
  TIMESTAMP_DAYS,
  TIMESTAMP_SECONDS,
  TIMESTAMP_MILLISECONDS,
  TIMESTAMP_MICROSECONDS,
  TIMESTAMP_NANOSECONDS,
  DURATION_DAYS,
  DURATION_SECONDS,
  DURATION_MILLISECONDS,
  DURATION_MICROSECONDS,
  DURATION_NANOSECONDS,
};

struct data_type {
  data_type() = default;
  explicit constexpr data_type(type_id id) : _id(id) {}
  constexpr type_id id() const { return _id; }
  bool operator==(data_type o) const { return _id == o._id; }
  bool operator!=(data_type o) const { return _id != o._id; }

 private:
  type_id _id{type_id::EMPTY};
};

inline constexpr size_type size_of(data_type t)
{
  switch (t.id()) {
    case type_id::INT8: return 1;
    case type_id::INT32:
    case type_id::TIMESTAMP_DAYS:
    case type_id::DURATION_DAYS: return 4;
    case type_id::INT64:
    case type_id::TIMESTAMP_SECONDS:
    case type_id::TIMESTAMP_MILLISECONDS:
    case type_id::TIMESTAMP_MICROSECONDS:
    case type_id::TIMESTAMP_NANOSECONDS:
    case type_id::DURATION_SECONDS:
    case type_id::DURATION_MILLISECONDS:
    case type_id::DURATION_MICROSECONDS:
    case type_id::DURATION_NANOSECONDS: return 8;
    default: return 0;  // STRING handled via children
  }
}

/* fixed-width 4-/8-byte classification used by the join/partition engine */
inline constexpr bool is_rep_int32(data_type t) { return size_of(t) == 4; }
inline constexpr bool is_rep_int64(data_type t) { return size_of(t) == 8; }

/* mirrors cudf::hash_id as used by shuffle_on.hpp:49 */
enum class hash_id : int32_t { HASH_IDENTITY = 0, HASH_MURMUR3 = 1 };
constexpr uint32_t DEFAULT_HASH_SEED = 0;

class column_view {
 public:
  column_view() = default;
  column_view(data_type type, size_type size, const void* data)
    : _type(type), _size(size), _data(data)
  {
  }
  /* STRING column: `data` = int32 offsets (size+1 entries), plus a chars
   * byte buffer — mirroring cuDF's offsets/chars children, which the
   * reference's strings path relies on (strings_column.cu:39-145).
   * child(0) = offsets view, child(1) = chars view. */
  column_view(data_type type, size_type size, const void* offsets, const void* chars,
              int64_t chars_bytes)
    : _type(type), _size(size), _data(offsets), _chars(chars), _chars_size(chars_bytes)
  {
  }
  data_type type() const { return _type; }
  size_type size() const { return _size; }
  template <typename T>
  const T* head() const
  {
    return static_cast<const T*>(_data);
  }
  const void* chars() const { return _chars; }
  int64_t chars_size() const { return _chars_size; }
  column_view child(int i) const
  {
    if (i == 0) return column_view(data_type(type_id::INT32), _size + 1, _data);
    return column_view(data_type(type_id::INT8), (size_type)_chars_size, _chars);
  }
  template <typename T>
  const T* begin() const
  {
    return head<T>();
  }

 private:
  data_type _type{};
  size_type _size{0};
  const void* _data{nullptr};
  const void* _chars{nullptr};
  int64_t _chars_size{0};
};

class mutable_column_view {
 public:
  mutable_column_view() = default;
  mutable_column_view(data_type type, size_type size, void* data)
    : _type(type), _size(size), _data(data)
  {
  }
  mutable_column_view(data_type type, size_type size, void* offsets, void* chars,
                      int64_t chars_bytes)
    : _type(type), _size(size), _data(offsets), _chars(chars), _chars_size(chars_bytes)
  {
  }
  data_type type() const { return _type; }
  size_type size() const { return _size; }
  template <typename T>
  T* head() const
  {
    return static_cast<T*>(_data);
  }
  void* chars() const { return _chars; }
  int64_t chars_size() const { return _chars_size; }
  operator column_view() const
  {
    return column_view(_type, _size, _data, _chars, _chars_size);
  }

 private:
  data_type _type{};
  size_type _size{0};
  void* _data{nullptr};
  void* _chars{nullptr};
  int64_t _chars_size{0};
};

/* owning device column (HIP device memory; allocation in dj_cpp_api.hip) */
class column {
 public:
  column(data_type type, size_type size);            // allocates device memory
  column(data_type type, size_type size, void* adopt_device_ptr);
  /* STRING column: allocates int32[size+1] offsets + chars_bytes chars */
  column(size_type size, int64_t chars_bytes);
  column(const column&) = delete;
  column& operator=(const column&) = delete;
  column(column&& o) noexcept;
  column& operator=(column&& o) noexcept;
  ~column();

  data_type type() const { return _type; }
  size_type size() const { return _size; }
  void* head() { return _data; }
  const void* head() const { return _data; }
  void* chars() { return _chars; }
  const void* chars() const { return _chars; }
  int64_t chars_size() const { return _chars_size; }
  column_view view() const { return column_view(_type, _size, _data, _chars, _chars_size); }
  mutable_column_view mutable_view()
  {
    return mutable_column_view(_type, _size, _data, _chars, _chars_size);
  }

 private:
  data_type _type{};
  size_type _size{0};
  void* _data{nullptr};
  void* _chars{nullptr};
  int64_t _chars_size{0};
};

class table_view {
 public:
  table_view() = default;
  explicit table_view(std::vector<column_view> cols) : _cols(std::move(cols)) {}
  size_type num_columns() const { return (size_type)_cols.size(); }
  size_type num_rows() const { return _cols.empty() ? 0 : _cols[0].size(); }
  column_view column(size_type i) const { return _cols.at(i); }

 private:
  std::vector<column_view> _cols;
};

class mutable_table_view {
 public:
  mutable_table_view() = default;
  explicit mutable_table_view(std::vector<mutable_column_view> cols) : _cols(std::move(cols)) {}
  size_type num_columns() const { return (size_type)_cols.size(); }
  size_type num_rows() const { return _cols.empty() ? 0 : _cols[0].size(); }
  mutable_column_view column(size_type i) const { return _cols.at(i); }
  operator table_view() const
  {
    std::vector<column_view> v(_cols.begin(), _cols.end());
    return table_view(std::move(v));
  }

 private:
  std::vector<mutable_column_view> _cols;
};

class table {
 public:
  explicit table(std::vector<std::unique_ptr<column>> cols) : _cols(std::move(cols)) {}
  size_type num_columns() const { return (size_type)_cols.size(); }
  size_type num_rows() const { return _cols.empty() ? 0 : _cols[0]->size(); }
  column& get_column(size_type i) { return *_cols.at(i); }
  const column& get_column(size_type i) const { return *_cols.at(i); }
  table_view view() const
  {
    std::vector<column_view> v;
    for (auto& c : _cols) v.push_back(c->view());
    return table_view(std::move(v));
  }
  mutable_table_view mutable_view()
  {
    std::vector<mutable_column_view> v;
    for (auto& c : _cols) v.push_back(c->mutable_view());
    return mutable_table_view(std::move(v));
  }

 private:
  std::vector<std::unique_ptr<column>> _cols;
};

}  // namespace cudf
