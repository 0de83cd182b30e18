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

