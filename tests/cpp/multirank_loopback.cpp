/*
 * multirank_loopback.cpp — multi-rank exercise of the C++ drop-in
 * orchestration on ONE GPU: G ranks run as host threads, each with its own
 * `LoopbackCommunicator` (a user-style implementation of the public
 * Communicator interface — include/communicator.hpp — exchanging device
 * buffers through an in-process mailbox with D2D copies).
 *
 * Validates, truly multi-rank: the rank-level partition slices, size
 * exchange, per-peer offsets, over-decomposition batching and result
 * assembly of distributed_inner_join — everything except RCCL itself.
 * The concatenated 2-rank result must equal the 1-rank result on the same
 * global inputs (order-insensitive checksum + row count), which in turn is
 * parity-tested against the CPU oracle elsewhere.
 *
 * Build/run: see tests/test_gpu_cpp_api.py::test_multirank_loopback.
 */
#include "all_to_all_comm.hpp"
#include "communicator.hpp"
#include "compression.hpp"
#include "distribute_table.hpp"
#include "distributed_join.hpp"
#include "shuffle_on.hpp"

#include "../../distributed_join_amd/csrc/dj_rng.h"

#include <hip/hip_runtime.h>

#include <condition_variable>
#include <cstdio>
#include <cstring>
#include <map>
#include <mutex>
#include <thread>
#include <vector>

#define CHECK(c)                                                      \
  do {                                                                \
    hipError_t e = (c);                                               \
    if (e != hipSuccess) {                                            \
      printf("HIP error %s at %d\n", hipGetErrorString(e), __LINE__); \
      exit(1);                                                        \
    }                                                                 \
  } while (0)

/* ---------------- in-process mailbox transport ---------------- */

struct Mailbox {
  struct Msg {
    const void* src;
    size_t bytes;
    bool consumed{false};
  };
  std::mutex m;
  std::condition_variable cv;
  /* key: (src_rank, dst_rank, seq of that pair) */
  std::map<std::tuple<int, int, long>, Msg> msgs;
  std::map<std::pair<int, int>, long> send_seq, recv_seq;

  void post(int src, int dst, const void* buf, size_t bytes)
  {
    std::lock_guard<std::mutex> g(m);
    long seq = send_seq[{src, dst}]++;
    msgs[{src, dst, seq}] = Msg{buf, bytes, false};
    cv.notify_all();
  }
  /* blocks until the matching send arrives, then D2D-copies it */
  void fetch(int src, int dst, void* out, size_t bytes)
  {
    std::unique_lock<std::mutex> g(m);
    long seq = recv_seq[{src, dst}]++;
    cv.wait(g, [&] { return msgs.count({src, dst, seq}) != 0; });
    Msg& msg = msgs[{src, dst, seq}];
    if (msg.bytes != bytes) {
      printf("loopback size mismatch %zu vs %zu\n", msg.bytes, bytes);
      exit(1);
    }
    CHECK(hipMemcpy(out, msg.src, bytes, hipMemcpyDeviceToDevice));
    msg.consumed = true;
    cv.notify_all();
  }
  void wait_all_consumed(int src)
  {
    std::unique_lock<std::mutex> g(m);
    cv.wait(g, [&] {
      for (auto& kv : msgs)
        if (std::get<0>(kv.first) == src && !kv.second.consumed) return false;
      return true;
    });
  }
};

/* user-style Communicator implementation against the public header */
class LoopbackCommunicator : public Communicator {
 public:
  LoopbackCommunicator(int rank, int size, Mailbox* mb) : mb_(mb)
  {
    mpi_rank = rank;
    mpi_size = size;
    current_device = 0;
  }
  void initialize() override {}
  void start() override {}
  void stop() override
  {
    /* block until every posted send was consumed and device work drained —
     * the blocking contract of Communicator::stop */
    mb_->wait_all_consumed(mpi_rank);
    CHECK(hipDeviceSynchronize());
  }
  void send(const void* buf, int64_t count, int element_size, int dest) override
  {
    CHECK(hipDeviceSynchronize());  // payload must be ready before posting
    mb_->post(mpi_rank, dest, buf, (size_t)count * element_size);
  }
  void recv(void* buf, int64_t count, int element_size, int source) override
  {
    mb_->fetch(source, mpi_rank, buf, (size_t)count * element_size);
  }
  void finalize() override {}
  bool group_by_batch() override { return true; }

 private:
  Mailbox* mb_;
};

/* ---------------- helpers ---------------- */

extern "C" void dj_gen_test_strings(const int64_t* d_keys, int64_t n, void** out_offsets,
                                    void** out_chars, int64_t* out_chars_bytes);
extern "C" void dj_dfree(void* p);

static std::unique_ptr<cudf::table> make_rank_tables(int64_t n_global, int64_t rows,
                                                     int64_t row0, bool build)
{
  std::vector<int64_t> keys(rows), pay(rows);
  for (int64_t t = 0; t < rows; t++) {
    int64_t i = row0 + t;
    keys[t] = build ? dj_build_key((uint64_t)i, (uint64_t)n_global, 2 * n_global, 1234)
                    : dj_probe_key((uint64_t)i, (uint64_t)n_global, 2 * n_global, 0.3, 1234);
    pay[t] = i;
  }
  auto kcol = std::make_unique<cudf::column>(cudf::data_type(cudf::type_id::INT64),
                                             (cudf::size_type)rows);
  auto pcol = std::make_unique<cudf::column>(cudf::data_type(cudf::type_id::INT64),
                                             (cudf::size_type)rows);
  CHECK(hipMemcpy(kcol->head(), keys.data(), rows * 8, hipMemcpyHostToDevice));
  CHECK(hipMemcpy(pcol->head(), pay.data(), rows * 8, hipMemcpyHostToDevice));
  std::vector<std::unique_ptr<cudf::column>> cols;
  cols.push_back(std::move(kcol));
  cols.push_back(std::move(pcol));
  return std::make_unique<cudf::table>(std::move(cols));
}

/* order-insensitive checksum: sum of mix64 over row tuples */
static void checksum(const cudf::table& t, uint64_t* out_sum, int64_t* out_rows)
{
  int64_t n = t.num_rows();
  std::vector<int64_t> c0(n), c1(n), c2(n), c3(n);
  CHECK(hipMemcpy(c0.data(), t.get_column(0).head(), n * 8, hipMemcpyDeviceToHost));
  CHECK(hipMemcpy(c1.data(), t.get_column(1).head(), n * 8, hipMemcpyDeviceToHost));
  CHECK(hipMemcpy(c2.data(), t.get_column(2).head(), n * 8, hipMemcpyDeviceToHost));
  CHECK(hipMemcpy(c3.data(), t.get_column(3).head(), n * 8, hipMemcpyDeviceToHost));
  uint64_t s = 0;
  for (int64_t i = 0; i < n; i++)
    s += dj_mix64((uint64_t)c0[i] ^ dj_mix64((uint64_t)c1[i] ^
                                             dj_mix64((uint64_t)c2[i] ^ (uint64_t)c3[i])));
  *out_sum = s;
  *out_rows = n;
}

/* (key int64, payload STRING) table for a rank slice — the config-4 shape
 * (string_payload.cu:50-74 semantics via the library's test-string
 * generator: len = k%%7+1, char = 'a'+k%%26) */
struct StrTable {
  std::unique_ptr<cudf::column> kcol;
  void* offsets{nullptr};
  void* chars{nullptr};
  int64_t chars_bytes{0};
  cudf::table_view view() const
  {
    return cudf::table_view(
      {kcol->view(), cudf::column_view(cudf::data_type(cudf::type_id::STRING), kcol->size(),
                                       offsets, chars, chars_bytes)});
  }
  StrTable() = default;
  StrTable(StrTable&& o) noexcept
    : kcol(std::move(o.kcol)), offsets(o.offsets), chars(o.chars), chars_bytes(o.chars_bytes)
  {
    o.offsets = o.chars = nullptr;
  }
  StrTable(const StrTable&) = delete;
  ~StrTable()
  {
    if (offsets) dj_dfree(offsets);
    if (chars) dj_dfree(chars);
  }
};

static StrTable make_rank_strtable(int64_t n_global, int64_t rows, int64_t row0, bool build)
{
  std::vector<int64_t> keys(rows);
  for (int64_t t = 0; t < rows; t++) {
    int64_t i = row0 + t;
    keys[t] = build ? dj_build_key((uint64_t)i, (uint64_t)n_global, 2 * n_global, 1234)
                    : dj_probe_key((uint64_t)i, (uint64_t)n_global, 2 * n_global, 0.3, 1234);
  }
  StrTable st;
  st.kcol = std::make_unique<cudf::column>(cudf::data_type(cudf::type_id::INT64),
                                           (cudf::size_type)rows);
  CHECK(hipMemcpy(st.kcol->head(), keys.data(), rows * 8, hipMemcpyHostToDevice));
  dj_gen_test_strings((const int64_t*)st.kcol->head(), rows, &st.offsets, &st.chars,
                      &st.chars_bytes);
  return st;
}

/* order-insensitive checksum over (int64, STRING, int64, STRING) rows */
static void checksum_str(const cudf::table& t, uint64_t* out_sum, int64_t* out_rows)
{
  int64_t n = t.num_rows();
  std::vector<int64_t> c0(n), c2(n);
  CHECK(hipMemcpy(c0.data(), t.get_column(0).head(), n * 8, hipMemcpyDeviceToHost));
  CHECK(hipMemcpy(c2.data(), t.get_column(2).head(), n * 8, hipMemcpyDeviceToHost));
  auto str_col = [&](int c, std::vector<int32_t>& off, std::vector<uint8_t>& ch) {
    const cudf::column& col = t.get_column(c);
    off.resize((size_t)n + 1);
    CHECK(hipMemcpy(off.data(), col.head(), (n + 1) * 4, hipMemcpyDeviceToHost));
    ch.resize((size_t)col.chars_size());
    if (col.chars_size())
      CHECK(hipMemcpy(ch.data(), col.chars(), col.chars_size(), hipMemcpyDeviceToHost));
  };
  std::vector<int32_t> off1, off3;
  std::vector<uint8_t> ch1, ch3;
  str_col(1, off1, ch1);
  str_col(3, off3, ch3);
  auto fold = [](const std::vector<int32_t>& off, const std::vector<uint8_t>& ch, int64_t i) {
    uint64_t h = 1469598103934665603ull;
    for (int32_t p = off[(size_t)i]; p < off[(size_t)i + 1]; p++)
      h = (h ^ ch[(size_t)p]) * 1099511628211ull;
    return h;
  };
  uint64_t s = 0;
  for (int64_t i = 0; i < n; i++)
    s += dj_mix64((uint64_t)c0[i] ^ dj_mix64(fold(off1, ch1, i)) ^
                  dj_mix64((uint64_t)c2[i] ^ dj_mix64(fold(off3, ch3, i) + 1)));
  *out_sum = s;
  *out_rows = n;
}

int main(int argc, char** argv)
{
  const int G = argc > 1 ? atoi(argv[1]) : 2;
  const int over_decom = argc > 2 ? atoi(argv[2]) : 2;
  const int64_t n_global = argc > 3 ? atoll(argv[3]) : 400000;
  CHECK(hipSetDevice(0));

  /* reference: single-rank result on the concatenated inputs */
  uint64_t want_sum;
  int64_t want_rows;
  {
    auto left = make_rank_tables(n_global, n_global, 0, true);
    auto right = make_rank_tables(n_global, n_global, 0, false);
    Mailbox mb;
    LoopbackCommunicator comm(0, 1, &mb);
    auto opts = generate_compression_options_distributed(left->view(), false);
    auto res = distributed_inner_join(left->view(), right->view(), {0}, {0}, &comm, opts,
                                      opts, 1, false, nullptr, 1);
    checksum(*res, &want_sum, &want_rows);
  }

  /* G ranks as threads over rank slices */
  Mailbox mb;
  std::vector<uint64_t> sums(G);
  std::vector<int64_t> rows(G);
  std::vector<std::thread> threads;
  for (int r = 0; r < G; r++) {
    threads.emplace_back([&, r] {
      CHECK(hipSetDevice(0));
      int64_t per = n_global / G;
      auto left = make_rank_tables(n_global, per, r * per, true);
      auto right = make_rank_tables(n_global, per, r * per, false);
      LoopbackCommunicator comm(r, G, &mb);
      /* alternate per-rank-invariant variants across runs of this binary via
       * argv: nvlink_domain_size and compression are exercised by the python
       * wrapper invoking multiple configurations */
      const int nvl = (argc > 4) ? atoi(argv[4]) : 1;
      const bool compress = (argc > 5) && atoi(argv[5]) != 0;
      /* sampling selector on rank 0 + broadcast over the loopback transport
       * (reference compression.cpp:36-130 flow); every rank must end up
       * with rank 0's choice */
      auto sel = (r == 0) ? generate_auto_select_compression_options(left->view())
                          : std::vector<ColumnCompressionOptions>{};
      auto bopts = broadcast_compression_options(left->view(), sel, &comm);
      if (bopts.size() != 2 ||
          bopts[0].compression_method != CompressionMethod::cascaded ||
          bopts[0].cascaded_format.use_bp != 1) {
        printf("BCAST OPTS WRONG rank %d\n", r);
        exit(1);
      }
      auto opts = compress ? bopts
                           : generate_compression_options_distributed(left->view(), compress);
      auto res = distributed_inner_join(left->view(), right->view(), {0}, {0}, &comm, opts,
                                        opts, over_decom, false, nullptr, nvl);
      checksum(*res, &sums[r], &rows[r]);
    });
  }
  for (auto& t : threads) t.join();

  uint64_t got_sum = 0;
  int64_t got_rows = 0;
  for (int r = 0; r < G; r++) {
    got_sum += sums[r];
    got_rows += rows[r];
  }

  /* shuffle_on (identity hash): every received key must be ≡ rank (mod G) —
   * the reference's placement pin (test_shuffle_on.cpp:78-83) — and
   * distribute_table/collect_tables must round-trip the global table. */
  {
    Mailbox mb2;
    std::vector<int> shuffle_ok(G, 0), roundtrip_ok(G, 0);
    std::vector<std::thread> th2;
    const int64_t n = 100000;
    for (int r = 0; r < G; r++) {
      th2.emplace_back([&, r] {
        CHECK(hipSetDevice(0));
        LoopbackCommunicator comm(r, G, &mb2);
        /* per-rank input: arbitrary int64 keys */
        auto t = make_rank_tables(n, n / G, r * (n / G), false);
        auto opts = generate_compression_options_distributed(t->view(), false);
        auto shuffled = shuffle_on(t->view(), {0}, &comm, opts,
                                   cudf::hash_id::HASH_IDENTITY, 0);
        int64_t m = shuffled->num_rows();
        std::vector<int64_t> keys(m);
        CHECK(hipMemcpy(keys.data(), shuffled->get_column(0).head(), m * 8,
                        hipMemcpyDeviceToHost));
        bool ok = true;
        for (int64_t i = 0; i < m; i++)
          ok &= ((uint32_t)((uint64_t)keys[i] & 0xFFFFFFFFu) % (uint32_t)G == (uint32_t)r);
        shuffle_ok[r] = ok ? 1 : 0;

        /* distribute/collect round trip (global table significant on root) */
        std::unique_ptr<cudf::table> global;
        cudf::table_view gview;
        if (r == 0) {
          global = make_rank_tables(n, n, 0, true);
          gview = global->view();
        }
        auto local = distribute_table(gview, &comm);
        auto merged = collect_tables(local->view(), &comm);
        if (r == 0) {
          std::vector<int64_t> a(n), b(n);
          CHECK(hipMemcpy(a.data(), global->get_column(0).head(), n * 8,
                          hipMemcpyDeviceToHost));
          CHECK(hipMemcpy(b.data(), merged->get_column(0).head(), n * 8,
                          hipMemcpyDeviceToHost));
          roundtrip_ok[0] = (merged->num_rows() == n && a == b) ? 1 : 0;
        } else {
          roundtrip_ok[r] = (merged == nullptr) ? 1 : 0;
        }
      });
    }
    for (auto& t : th2) t.join();
    for (int r = 0; r < G; r++) {
      if (!shuffle_ok[r]) {
        printf("SHUFFLE PLACEMENT MISMATCH rank %d\n", r);
        return 1;
      }
      if (!roundtrip_ok[r]) {
        printf("DISTRIBUTE/COLLECT MISMATCH rank %d\n", r);
        return 1;
      }
    }
    printf("shuffle identity placement + distribute/collect roundtrip OK\n");
  }
  /* strings payload multirank (config 4 shape): sizes on the wire,
   * receiver-side offset rebuild (strings_column.cu:39-145 semantics),
   * compressed sizes wire when argv[5] asks for compression */
  {
    const int64_t ns = 200000;
    const int nvl = (argc > 4) ? atoi(argv[4]) : 1;
    const bool compress = (argc > 5) && atoi(argv[5]) != 0;
    uint64_t want_s;
    int64_t want_r;
    {
      Mailbox mb1;
      LoopbackCommunicator c1(0, 1, &mb1);
      auto l = make_rank_strtable(ns, ns, 0, true);
      auto r = make_rank_strtable(ns, ns, 0, false);
      auto o = generate_compression_options_distributed(l.view(), compress);
      auto res = distributed_inner_join(l.view(), r.view(), {0}, {0}, &c1, o, o, 1, false,
                                        nullptr, 1);
      checksum_str(*res, &want_s, &want_r);
    }
    Mailbox mb2;
    std::vector<uint64_t> ssums(G);
    std::vector<int64_t> srows(G);
    std::vector<std::thread> th;
    for (int r = 0; r < G; r++) {
      th.emplace_back([&, r] {
        CHECK(hipSetDevice(0));
        int64_t per = ns / G;
        auto l = make_rank_strtable(ns, per, r * per, true);
        auto rt = make_rank_strtable(ns, per, r * per, false);
        LoopbackCommunicator comm(r, G, &mb2);
        auto o = generate_compression_options_distributed(l.view(), compress);
        auto res = distributed_inner_join(l.view(), rt.view(), {0}, {0}, &comm, o, o,
                                          over_decom, false, nullptr, nvl);
        checksum_str(*res, &ssums[r], &srows[r]);
      });
    }
    for (auto& t : th) t.join();
    uint64_t gs = 0;
    int64_t gr = 0;
    for (int r = 0; r < G; r++) {
      gs += ssums[r];
      gr += srows[r];
    }
    if (gs != want_s || gr != want_r) {
      printf("STRINGS MULTIRANK MISMATCH (%lld vs %lld rows)\n", (long long)gr,
             (long long)want_r);
      return 1;
    }
    printf("strings multirank OK (%lld rows)\n", (long long)gr);
  }

  /* composite (2-column) join keys: fused-hash placement both sides,
   * collision filter in the local join — the multi-key shuffle + local-join
   * route at G ranks must equal the single-rank result */
  {
    const int64_t nm = 120000;
    auto make_mk = [&](int64_t rows, int64_t row0) {
      std::vector<int64_t> k0(rows), k1(rows), pay(rows);
      for (int64_t t = 0; t < rows; t++) {
        int64_t i = row0 + t;
        k0[t] = (int64_t)(dj_mix64((uint64_t)i) % 700);
        k1[t] = (int64_t)(dj_mix64((uint64_t)i + 77) % 700);
        pay[t] = i;
      }
      std::vector<std::unique_ptr<cudf::column>> cols;
      for (auto* v : {&k0, &k1, &pay}) {
        auto c = std::make_unique<cudf::column>(cudf::data_type(cudf::type_id::INT64),
                                                (cudf::size_type)rows);
        CHECK(hipMemcpy(c->head(), v->data(), rows * 8, hipMemcpyHostToDevice));
        cols.push_back(std::move(c));
      }
      return std::make_unique<cudf::table>(std::move(cols));
    };
    auto checksum6 = [&](const cudf::table& t, uint64_t* out_sum, int64_t* out_rows) {
      int64_t n = t.num_rows();
      std::vector<std::vector<int64_t>> c(6, std::vector<int64_t>(n));
      for (int j = 0; j < 6; j++)
        CHECK(hipMemcpy(c[j].data(), t.get_column(j).head(), n * 8, hipMemcpyDeviceToHost));
      uint64_t s = 0;
      for (int64_t i = 0; i < n; i++) {
        uint64_t h = 0;
        for (int j = 0; j < 6; j++) h = dj_mix64(h ^ (uint64_t)c[j][i]);
        s += h;
      }
      *out_sum = s;
      *out_rows = n;
    };
    uint64_t want_s;
    int64_t want_r;
    {
      Mailbox mb1;
      LoopbackCommunicator c1(0, 1, &mb1);
      auto l = make_mk(nm, 0);
      auto r = make_mk(nm, 1000000);
      auto o = generate_compression_options_distributed(l->view(), false);
      auto res = distributed_inner_join(l->view(), r->view(), {0, 1}, {0, 1}, &c1, o, o, 1,
                                        false, nullptr, 1);
      checksum6(*res, &want_s, &want_r);
    }
    Mailbox mb3;
    std::vector<uint64_t> msums(G);
    std::vector<int64_t> mrows(G);
    std::vector<std::thread> th;
    for (int r = 0; r < G; r++) {
      th.emplace_back([&, r] {
        CHECK(hipSetDevice(0));
        int64_t per = nm / G;
        auto l = make_mk(per, (int64_t)r * per);
        auto rt = make_mk(per, 1000000 + (int64_t)r * per);
        LoopbackCommunicator comm(r, G, &mb3);
        auto o = generate_compression_options_distributed(l->view(), false);
        auto res = distributed_inner_join(l->view(), rt->view(), {0, 1}, {0, 1}, &comm, o, o,
                                          1, false, nullptr, G);
        checksum6(*res, &msums[r], &mrows[r]);
      });
    }
    for (auto& t : th) t.join();
    uint64_t gs = 0;
    int64_t gr = 0;
    for (int r = 0; r < G; r++) {
      gs += msums[r];
      gr += mrows[r];
    }
    if (G > 1 && nm % G == 0) {
      if (gs != want_s || gr != want_r) {
        printf("MULTIKEY MULTIRANK MISMATCH (%lld vs %lld rows)\n", (long long)gr,
               (long long)want_r);
        return 1;
      }
      printf("multikey multirank OK (%lld rows)\n", (long long)gr);
    }
  }

  printf("single-rank: rows=%lld sum=%llx\n", (long long)want_rows,
         (unsigned long long)want_sum);
  printf("%d-rank(od=%d): rows=%lld sum=%llx\n", G, over_decom, (long long)got_rows,
         (unsigned long long)got_sum);
  if (got_rows != want_rows || got_sum != want_sum) {
    printf("MULTIRANK MISMATCH\n");
    return 1;
  }
  printf("MULTIRANK OK\n");
  return 0;
}
