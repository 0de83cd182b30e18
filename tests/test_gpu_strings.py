"""GPU tests of the strings-column path (BASELINE config 4), mirroring the
reference's string-payload test semantics (test/string_payload.cu:40-163):
string for key k has length k%7+1 filled with 'a'+k%26; joining multiples-of-3
keys with multiples-of-5 keys yields n/5 rows whose string payloads (both
sides) must match the formula for their key.
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


def expected_string(k):
    return bytes([ord('a') + k % 26]) * (k % 7 + 1)


def strings_of(col, n):
    off, ch = col
    return [ch[off[i]:off[i + 1]].tobytes() for i in range(n)]


def test_gen_test_strings_matches_formula(dj):
    n = 10_000
    keys = np.arange(n, dtype=np.int64) * 3
    dk = dj.DeviceArray.from_numpy(keys)
    off_p, ch_p, nb = dj.gen_test_strings(dk, n)
    L = dj.lib()
    off = np.empty(n + 1, dtype=np.int32)
    L.dj_memcpy_d2h(off.ctypes.data, off_p, (n + 1) * 4)
    ch = np.empty(nb, dtype=np.uint8)
    L.dj_memcpy_d2h(ch.ctypes.data, ch_p, nb)
    assert off[0] == 0 and off[-1] == nb
    for i in (0, 1, 7, 999, n - 1):
        assert ch[off[i]:off[i + 1]].tobytes() == expected_string(int(keys[i]))
    assert nb == sum(int(k) % 7 + 1 for k in keys)
    L.dj_dfree(off_p)
    L.dj_dfree(ch_p)


@pytest.mark.parametrize("n", [1, 2, 4095, 4096, 4097, 8193, 100_000])
def test_offsets_scan_tile_boundaries(dj, n):
    # the tile-coalesced sizes->offsets scan (dj_strings.hip) must be exact
    # at and around its 4096-element tile boundary; offsets are the prefix
    # sums of the analytical sizes len(k) = k % 7 + 1
    keys = (np.arange(n, dtype=np.int64) * 13 + 5)
    dk = dj.DeviceArray.from_numpy(keys)
    off_p, ch_p, nb = dj.gen_test_strings(dk, n)
    L = dj.lib()
    off = np.empty(n + 1, dtype=np.int32)
    L.dj_memcpy_d2h(off.ctypes.data, off_p, (n + 1) * 4)
    sizes = (keys % 7 + 1).astype(np.int64)
    want = np.zeros(n + 1, dtype=np.int64)
    np.cumsum(sizes, out=want[1:])
    assert (off == want).all()
    assert nb == int(want[-1])
    L.dj_dfree(off_p)
    L.dj_dfree(ch_p)


@pytest.mark.parametrize("over_decom", [1, 4])
def test_string_payload_join(dj, comm, over_decom):
    # reference KAT (string_payload.cu run_test): multiples of 3 x multiples
    # of 5, string payloads on both sides
    size = 30_000
    lk = np.arange(size, dtype=np.int64) * 3
    rk = np.arange(size, dtype=np.int64) * 5
    dlk = dj.DeviceArray.from_numpy(lk)
    drk = dj.DeviceArray.from_numpy(rk)
    ls = dj.gen_test_strings(dlk, size)
    rs = dj.gen_test_strings(drk, size)
    c0, c1, c2, c3 = dj.cpp_distributed_inner_join_str(comm, dlk, ls, size, drk, rs, size,
                                                       over_decom=over_decom)
    n = len(c0)
    assert n == size // 5
    assert (c0 % 15 == 0).all() and (c0 == c2).all()
    s1 = strings_of(c1, n)
    s3 = strings_of(c3, n)
    for i in range(n):
        k = int(c0[i])
        assert s1[i] == expected_string(k), (i, k, s1[i])
        assert s3[i] == expected_string(k)
    L = dj.lib()
    for p in (ls[0], ls[1], rs[0], rs[1]):
        L.dj_dfree(p)


def test_string_payload_join_generated(dj, comm):
    # generated keys (duplicat-able probe side), strings derived from keys;
    # parity vs numpy-computed expectation (order-insensitive)
    n = 50_000
    bk, _ = oracle.gen_build(n)
    pk, _ = oracle.gen_probe(n, n, selectivity=0.5)
    dlk = dj.DeviceArray.from_numpy(bk)
    drk = dj.DeviceArray.from_numpy(pk)
    ls = dj.gen_test_strings(dlk, n)
    rs = dj.gen_test_strings(drk, n)
    c0, c1, c2, c3 = dj.cpp_distributed_inner_join_str(comm, dlk, ls, n, drk, rs, n)
    nout = len(c0)
    # expected matches from the oracle (keys only)
    e0, _, _, _ = oracle.inner_join(bk, np.arange(n, dtype=np.int64),
                                    pk, np.arange(n, dtype=np.int64))
    assert nout == len(e0)
    assert sorted(c0.tolist()) == sorted(e0.tolist())
    # every output string must match the formula of its row's key
    s1 = strings_of(c1, nout)
    s3 = strings_of(c3, nout)
    for i in range(0, nout, 97):
        assert s1[i] == expected_string(int(c0[i]))
        assert s3[i] == expected_string(int(c2[i]))
    L = dj.lib()
    for p in (ls[0], ls[1], rs[0], rs[1]):
        L.dj_dfree(p)
