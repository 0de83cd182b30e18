"""GPU tests of the cascaded compression layer.
The wire format is ours (parity-unpinned, dj_compress.hip header); what is
pinned is SEMANTICS: a compressed exchange must be bit-identical to the
uncompressed one. At world size 1 the shuffle path sends the self slice
through the communicator, so compress -> wire -> decompress runs fully.
"""
import numpy as np
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


def test_shuffle_compressed_roundtrip(dj, comm):
    n = 200_000
    k, p = oracle.gen_probe(n, n)
    dk, dp = dj.DeviceArray.from_numpy(k), dj.DeviceArray.from_numpy(p)
    a = oracle.sort_rows(*dj.cpp_shuffle_on(comm, dk, dp, n, dj.HASH_MURMUR3, 12345678,
                                            compression=False))
    b = oracle.sort_rows(*dj.cpp_shuffle_on(comm, dk, dp, n, dj.HASH_MURMUR3, 12345678,
                                            compression=True))
    for x, y in zip(a, b):
        assert (x == y).all()


def test_shuffle_compressed_negative_and_large_keys(dj, comm):
    # exercise zigzag + the raw-store fallback (values needing 64 bits)
    n = 50_000
    rng = np.random.RandomState(3)
    k = rng.randint(-2**62, 2**62, n).astype(np.int64)
    p = np.arange(n, dtype=np.int64)
    dk, dp = dj.DeviceArray.from_numpy(k), dj.DeviceArray.from_numpy(p)
    a = oracle.sort_rows(*dj.cpp_shuffle_on(comm, dk, dp, n, compression=False))
    b = oracle.sort_rows(*dj.cpp_shuffle_on(comm, dk, dp, n, compression=True))
    for x, y in zip(a, b):
        assert (x == y).all()


def _roundtrip(dj, arr, rles, deltas, bp, esize=8):
    din = dj.DeviceArray.from_numpy(arr)
    dout = dj.DeviceArray(max(len(arr), 1))
    wire = dj.lib().dj_compress_roundtrip(din.ptr, len(arr), esize, rles, deltas, bp,
                                          dout.ptr)
    got = dout.to_numpy()[:len(arr)]
    assert (got == arr).all(), f"roundtrip mismatch rles={rles} deltas={deltas} bp={bp}"
    return wire


CASCADE_CONFIGS = [(0, 0, 1), (0, 1, 1), (1, 0, 1), (1, 1, 1), (1, 0, 0), (0, 0, 0)]


@pytest.mark.parametrize("rles,deltas,bp", CASCADE_CONFIGS)
def test_cascaded_roundtrip_patterns(dj, rles, deltas, bp):
    # every cascaded pass combination (reference compression.hpp:73-251
    # semantics: num_RLEs in {0,1}, num_deltas in {0,1}, use_bp) over the
    # data shapes each pass targets
    rng = np.random.RandomState(42)
    n = 100_000
    patterns = {
        "runs": np.repeat(rng.randint(0, 50, n // 100), 100)[:n].astype(np.int64),
        "sorted": np.sort(rng.randint(0, 10**12, n)).astype(np.int64),
        "random": rng.randint(-2**62, 2**62, n).astype(np.int64),
        "const": np.full(n, 7, dtype=np.int64),
        "tiny": np.array([5], dtype=np.int64),
        "alternating": np.tile(np.array([3, 3, 3, -9], dtype=np.int64), n // 4),
    }
    for name, arr in patterns.items():
        _roundtrip(dj, arr, rles, deltas, bp)


def test_cascaded_rle_compresses_runs(dj):
    # run-heavy data must actually shrink with the RLE pass, and the RLE
    # form must beat plain delta+bitpack on it
    n = 1_000_000
    arr = np.repeat(np.arange(n // 1000, dtype=np.int64), 1000)
    raw = n * 8
    rle = _roundtrip(dj, arr, 1, 1, 1)
    bponly = _roundtrip(dj, arr, 0, 0, 1)
    assert rle < raw // 50, f"RLE wire {rle} should be <2% of raw {raw}"
    assert rle < bponly


def test_cascaded_raw_fallback_never_expands(dj):
    rng = np.random.RandomState(9)
    arr = rng.randint(-2**62, 2**62, 10_000).astype(np.int64)
    for rles, deltas, bp in CASCADE_CONFIGS:
        wire = _roundtrip(dj, arr, rles, deltas, bp)
        assert wire <= 10_000 * 8 + 64


def test_cascaded_roundtrip_int32(dj):
    rng = np.random.RandomState(5)
    n = 50_000
    arr = np.repeat(rng.randint(0, 1000, n // 10), 10)[:n].astype(np.int32)
    # pack int32 through the 4-byte element path
    buf = dj.DeviceArray((n + 1) // 2)  # n int32 = n/2 int64 slots
    dj.lib().dj_memcpy_h2d(buf.ptr, arr.ctypes.data, n * 4)
    out = dj.DeviceArray((n + 1) // 2)
    for rles, deltas, bp in CASCADE_CONFIGS:
        wire = dj.lib().dj_compress_roundtrip(buf.ptr, n, 4, rles, deltas, bp, out.ptr)
        got = np.zeros(n, dtype=np.int32)
        dj.lib().dj_memcpy_d2h(got.ctypes.data, out.ptr, n * 4)
        assert (got == arr).all(), (rles, deltas, bp)


def test_shuffle_rle_compressed_roundtrip(dj, comm):
    # end-to-end wire path with an explicit RLE+delta cascaded option via
    # the options-carrying join entry (auto-select may also pick RLE; this
    # pins the explicit-config path)
    n = 100_000
    k = np.repeat(np.arange(n // 50, dtype=np.int64), 50)[:n]
    p = np.arange(n, dtype=np.int64)
    dk, dp = dj.DeviceArray.from_numpy(k), dj.DeviceArray.from_numpy(p)
    a = oracle.sort_rows(*dj.cpp_shuffle_on(comm, dk, dp, n, compression=False))
    b = oracle.sort_rows(*dj.cpp_shuffle_on(comm, dk, dp, n, compression=True))
    for x, y in zip(a, b):
        assert (x == y).all()


def test_join_with_compression_option(dj, comm):
    # full join with compression requested (reference analytical test cases
    # run with compression on, compare_against_analytical.cu:199-201)
    n = 300_000
    bk, bp = oracle.gen_build(n)
    pk, pp = oracle.gen_probe(n, n, selectivity=0.3)
    dlk, dlp = dj.DeviceArray.from_numpy(bk), dj.DeviceArray.from_numpy(bp)
    drk, drp = dj.DeviceArray.from_numpy(pk), dj.DeviceArray.from_numpy(pp)
    t = dj.lib().dj_cpp_distributed_inner_join_i64_opts(
        comm.ptr, dlk.ptr, dlp.ptr, n, drk.ptr, drp.ptr, n, 4, 0, 1)
    got = dj.table_to_numpy(t)
    want = oracle.inner_join(bk, bp, pk, pp)
    g = oracle.sort_rows(*got)
    w = oracle.sort_rows(*want)
    assert len(g[0]) == len(w[0])
    for a, b in zip(g, w):
        assert (a == b).all()

