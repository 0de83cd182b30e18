"""GPU tests of the C++ drop-in surface (distributed_inner_join / shuffle_on
/ AllToAllCommunicator path) through its C ABI, single-rank. The multi-rank
path is the same code with RCCL peers (covered by the gloo plan tests on CPU
and the driver's multi-GPU bench at round end).

Reference semantics checked: result equals the oracle's global join on
identical inputs (compare_against_single_gpu.cu pattern), including
over-decomposition batching (distributed_join.cpp:244-329) which must not
change the result.
"""
import numpy as np
import os
import pytest

import oracle

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def dj():
    import distributed_join_amd as dj
    dj.require_gpu()
    return dj


@pytest.fixture(scope="module")
def comm(dj):
    c = dj.CppCommunicator(0, 1)
    yield c
    c.destroy()


@pytest.mark.parametrize("over_decom", [1, 4])
def test_cpp_distributed_join_single_rank(dj, comm, over_decom):
    n = 300_000
    bk, bp = oracle.gen_build(n)
    pk, pp = oracle.gen_probe(n, n, selectivity=0.3)
    dlk, dlp = dj.DeviceArray.from_numpy(bk), dj.DeviceArray.from_numpy(bp)
    drk, drp = dj.DeviceArray.from_numpy(pk), dj.DeviceArray.from_numpy(pp)
    got = dj.cpp_distributed_inner_join(comm, dlk, dlp, n, drk, drp, n,
                                        over_decom=over_decom)
    assert len(got) == 4
    want = oracle.inner_join(bk, bp, pk, pp)
    g = oracle.sort_rows(*got)
    w = oracle.sort_rows(*want)
    assert len(g[0]) == len(w[0])
    for a, b in zip(g, w):
        assert (a == b).all()


def test_cpp_distributed_join_analytical(dj, comm):
    # the reference's own KAT through the C++ drop-in path
    size = 30_000
    lk = np.arange(size, dtype=np.int64) * 3
    lp = np.arange(size, dtype=np.int64)
    rk = np.arange(size, dtype=np.int64) * 5
    rp = np.arange(size, dtype=np.int64)
    dlk, dlp = dj.DeviceArray.from_numpy(lk), dj.DeviceArray.from_numpy(lp)
    drk, drp = dj.DeviceArray.from_numpy(rk), dj.DeviceArray.from_numpy(rp)
    c0, c1, c2, c3 = dj.cpp_distributed_inner_join(comm, dlk, dlp, size, drk, drp, size)
    assert len(c0) == size // 5
    assert (c0 % 15 == 0).all() and (c1 == c0 // 3).all()
    assert (c3 == c2 // 5).all() and (c0 == c2).all()


def test_cpp_shuffle_single_rank_identity(dj, comm):
    # world of 1: shuffle returns the same multiset of rows
    n = 50_000
    k, p = oracle.gen_probe(n, n)
    dk, dp = dj.DeviceArray.from_numpy(k), dj.DeviceArray.from_numpy(p)
    ok, op = dj.cpp_shuffle_on(comm, dk, dp, n, dj.HASH_MURMUR3, 12345678)
    a = oracle.sort_rows(ok, op)
    b = oracle.sort_rows(k, p)
    for x, y in zip(a, b):
        assert (x == y).all()


def test_cpp_shuffle_over_decom_equivalence(dj, comm):
    # over-decomposition must not change the result (distributed_join.cpp:244-329)
    n = 100_000
    bk, bp = oracle.gen_build(n)
    pk, pp = oracle.gen_probe(n, n)
    dlk, dlp = dj.DeviceArray.from_numpy(bk), dj.DeviceArray.from_numpy(bp)
    drk, drp = dj.DeviceArray.from_numpy(pk), dj.DeviceArray.from_numpy(pp)
    a = oracle.sort_rows(*dj.cpp_distributed_inner_join(comm, dlk, dlp, n, drk, drp, n, 1))
    b = oracle.sort_rows(*dj.cpp_distributed_inner_join(comm, dlk, dlp, n, drk, drp, n, 8))
    for x, y in zip(a, b):
        assert (x == y).all()


def test_cpp_join_empty(dj, comm):
    n = 1000
    bk, bp = oracle.gen_build(n)
    dlk, dlp = dj.DeviceArray.from_numpy(bk), dj.DeviceArray.from_numpy(bp)
    empty = dj.DeviceArray(1)
    got = dj.cpp_distributed_inner_join(comm, dlk, dlp, n, empty, empty, 0)
    assert all(len(c) == 0 for c in got)


def test_distribute_and_collect_roundtrip_single_rank(dj, comm):
    # distribute_table / collect_tables round trip (reference
    # distribute_table.hpp:36-49); world size 1: collected == original
    n = 100_000
    k, p = oracle.gen_probe(n, n)
    dk, dp = dj.DeviceArray.from_numpy(k), dj.DeviceArray.from_numpy(p)
    t = dj.lib().dj_cpp_distribute_collect_roundtrip_i64(comm.ptr, dk.ptr, dp.ptr, n)
    c0, c1 = dj.table_to_numpy(t)
    assert (c0 == k).all() and (c1 == p).all()


def test_multirank_loopback():
    """Multi-rank C++ orchestration on one GPU: 2 and 4 ranks as threads with
    a user-style loopback Communicator (tests/cpp/multirank_loopback.cpp).
    Concatenated multi-rank result must equal the single-rank result."""
    import subprocess
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    exe = os.path.join(repo, "tests", "cpp", "multirank_loopback")
    if not os.path.exists(exe):
        subprocess.run(
            ["hipcc", "--offload-arch=gfx950", "-O2", "-std=c++17",
             "-I", os.path.join(repo, "include"),
             os.path.join(repo, "tests", "cpp", "multirank_loopback.cpp"),
             "-o", exe, "-L", os.path.join(repo, "distributed_join_amd"), "-ldistjoin",
             "-Wl,-rpath," + os.path.join(repo, "distributed_join_amd")],
            check=True, capture_output=True)
    # (G, over_decom, n_global, nvlink_domain_size, compression)
    for args in (["2", "2"], ["4", "1"],
                 ["4", "1", "400000", "2", "0"],   # 2-level hierarchy
                 ["4", "1", "400000", "1", "0"],   # nvl=1: full shuffle + local join
                 ["2", "2", "400000", "2", "1"],   # compressed wire
                 ["4", "2", "37", "4", "1"],       # tiny: empty slices/buckets
                 ["8", "1", "800000", "8", "0"],   # the driver's config-3 shape
                 ["8", "1", "800000", "1", "0"]):  # config 3, reference default mode
        r = subprocess.run([exe] + args, capture_output=True, text=True, timeout=240)
        assert r.returncode == 0, " ".join(args) + "\n" + r.stdout + r.stderr
        assert "MULTIRANK OK" in r.stdout


def test_rccl_selftest(dj):
    """RCCL proper (not the loopback transport): world-1 ncclCommInitRank +
    grouped self send/recv through RCCLCommunicator start/send/recv/stop —
    the exact calls the N>1 peer-slice exchange makes. De-risks the driver's
    multi-GPU runs on a single-GPU box."""
    assert dj.lib().dj_rccl_selftest(1 << 20) == 0


def test_full_variant_nvl(dj, comm):
    """The full-featured C ABI wrapper (adds nvlink_domain_size) matches the
    oracle at world=1 for any domain size (get_nvl_partition_size collapses
    to the local join, distributed_join.cpp:200-214)."""
    n = 200_000
    lk, lp = oracle.gen_build(n)
    rk, rp = oracle.gen_probe(n, n)
    dlk, dlp = dj.DeviceArray.from_numpy(lk), dj.DeviceArray.from_numpy(lp)
    drk, drp = dj.DeviceArray.from_numpy(rk), dj.DeviceArray.from_numpy(rp)
    want = oracle.sort_rows(*oracle.inner_join(lk, lp, rk, rp))
    for nvl in (1, 8):
        t = dj.lib().dj_cpp_distributed_inner_join_i64_full(
            comm.ptr, dlk.ptr, dlp.ptr, n, drk.ptr, drp.ptr, n, 1, 0, 0, nvl)
        got = oracle.sort_rows(*dj.table_to_numpy(t))  # frees t
        for a, b in zip(got, want):
            assert (a == b).all()


def test_fused_big_buckets():
    """G=8 od=1 shape on one GPU: when the fused wire path's PA*F fan-out cap
    leaves >1300 rows per bucket, the join must take its 4096-slot LDS table
    (not overflow every bucket into whole-batch redos). DJ_FORCE_FUSED_F=64
    shrinks the fan-out so 2 loopback ranks at 80M global rows reproduce the
    big-bucket regime (~2400 rows/bucket)."""
    import subprocess
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    exe = os.path.join(repo, "tests", "cpp", "multirank_loopback")
    assert os.path.exists(exe), "built by test_multirank_loopback"
    for force_f in ("64", "1024"):
        # 64: ~2400 rows/bucket (4096-slot join); 1024: full fan-out, whose
        # sub-bucket bits must stay disjoint from the pass-A group bits
        # (subF_of) or the effective fan-out collapses 4x
        env = dict(os.environ, DJ_FORCE_FUSED_F=force_f)
        r = subprocess.run([exe, "2", "1", "80000000", "2", "0"], capture_output=True,
                           text=True, timeout=420, env=env)
        assert r.returncode == 0, force_f + "\n" + r.stdout + r.stderr
        assert "MULTIRANK OK" in r.stdout


def test_kernel_path_transport_world1(dj):
    """The raw C-ABI transport (dj_comm_init + dj_all_to_all_i64 +
    dj_exchange_sizes) at world 1 over real RCCL — our analogue of the
    reference's transport-level test (test/buffer_communicator.cu: set a
    pattern, exchange, assert). Self-exchange goes through the same grouped
    ncclSend/ncclRecv the N>1 peer slices use."""
    import ctypes
    L = dj.lib()
    nbytes = L.dj_rccl_unique_id_bytes()
    idb = np.zeros(nbytes, dtype=np.uint8)
    L.dj_rccl_get_unique_id(idb.ctypes.data)
    L.dj_comm_init(0, 1, idb.ctypes.data)
    try:
        assert L.dj_comm_rank() == 0 and L.dj_comm_size() == 1
        # sizes exchange: world of 1 -> recv == send
        send_counts = np.array([123456], dtype=np.int64)
        recv_counts = np.zeros(1, dtype=np.int64)
        L.dj_exchange_sizes(send_counts.ctypes.data, recv_counts.ctypes.data)
        assert recv_counts[0] == 123456
        # buffer all-to-all: pattern round-trips through the self slice
        n = 1 << 20
        pattern = (np.arange(n, dtype=np.int64) * 2654435761) ^ 0x5DEECE66D
        d_send = dj.DeviceArray.from_numpy(pattern)
        d_recv = dj.DeviceArray(n)
        send_off = np.array([0, n], dtype=np.int64)
        recv_off = np.array([0, n], dtype=np.int64)
        L.dj_all_to_all_i64(d_send.ptr, send_off.ctypes.data, d_recv.ptr,
                            recv_off.ctypes.data)
        L.dj_sync()
        assert (d_recv.to_numpy() == pattern).all()
    finally:
        L.dj_comm_finalize()
