"""GPU tests for the remaining drop-in surface pieces: INT32 key columns
(dtype coverage — the reference's single-GPU comparison tests span
int32/int64, compare_against_single_gpu.cu:237-268) through the generic
column-descriptor join, and table round-trips.
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


def _upload_i32(dj, a):
    a = np.ascontiguousarray(a, dtype=np.int32)
    L = dj.lib()
    p = L.dj_dmalloc(max(len(a), 1) * 4)
    if len(a):
        L.dj_memcpy_h2d(p, a.ctypes.data, len(a) * 4)
    return p


def test_int32_key_join(dj, comm):
    # int32 keys on both sides + int32 payloads: join through the generic
    # descriptor path; parity vs the oracle on widened keys
    n = 200_000
    bk64, bp64 = oracle.gen_build(n, rand_max=min(2 * n, 2**31 - 2))
    pk64, pp64 = oracle.gen_probe(n, n, rand_max=min(2 * n, 2**31 - 2), selectivity=0.3)
    bk = bk64.astype(np.int32)
    pk = pk64.astype(np.int32)
    bp = bp64.astype(np.int32)
    pp = pp64.astype(np.int32)
    d_bk, d_bp = _upload_i32(dj, bk), _upload_i32(dj, bp)
    d_pk, d_pp = _upload_i32(dj, pk), _upload_i32(dj, pp)
    cols = dj.cpp_distributed_inner_join_cols(
        comm, [(dj.TYPE_INT32, d_bk), (dj.TYPE_INT32, d_bp)], n,
        [(dj.TYPE_INT32, d_pk), (dj.TYPE_INT32, d_pp)], n)
    c0, c1, c2, c3 = [c.astype(np.int64) for c in cols]
    want = oracle.inner_join(bk64, bp64, pk64, pp64)
    g = oracle.sort_rows(c0, c1, c2, c3)
    w = oracle.sort_rows(*want)
    assert len(g[0]) == len(w[0])
    for a, b in zip(g, w):
        assert (a == b).all()
    for p in (d_bk, d_bp, d_pk, d_pp):
        dj.lib().dj_dfree(p)


def test_mixed_int32_key_int64_payload(dj, comm):
    n = 100_000
    bk64, bp = oracle.gen_build(n, rand_max=min(2 * n, 2**31 - 2))
    pk64, pp = oracle.gen_probe(n, n, rand_max=min(2 * n, 2**31 - 2), selectivity=0.5)
    d_bk = _upload_i32(dj, bk64.astype(np.int32))
    d_pk = _upload_i32(dj, pk64.astype(np.int32))
    d_bp = dj.DeviceArray.from_numpy(bp)
    d_pp = dj.DeviceArray.from_numpy(pp)
    cols = dj.cpp_distributed_inner_join_cols(
        comm, [(dj.TYPE_INT32, d_bk), (dj.TYPE_INT64, d_bp.ptr)], n,
        [(dj.TYPE_INT32, d_pk), (dj.TYPE_INT64, d_pp.ptr)], n)
    want = oracle.inner_join(bk64, bp, pk64, pp)
    g = oracle.sort_rows(*[c.astype(np.int64) for c in cols])
    w = oracle.sort_rows(*want)
    assert len(g[0]) == len(w[0])
    for a, b in zip(g, w):
        assert (a == b).all()
    dj.lib().dj_dfree(d_bk)
    dj.lib().dj_dfree(d_pk)


def test_key_not_first_column(dj, comm):
    # join keys at column index 1 (left_on/right_on generality)
    n = 50_000
    bk, bp = oracle.gen_build(n)
    pk, pp = oracle.gen_probe(n, n)
    d = {name: dj.DeviceArray.from_numpy(a) for name, a in
         [("bk", bk), ("bp", bp), ("pk", pk), ("pp", pp)]}
    cols = dj.cpp_distributed_inner_join_cols(
        comm, [(dj.TYPE_INT64, d["bp"].ptr), (dj.TYPE_INT64, d["bk"].ptr)], n,
        [(dj.TYPE_INT64, d["pp"].ptr), (dj.TYPE_INT64, d["pk"].ptr)], n,
        key_l=1, key_r=1)
    # output: left cols (pay, key) then right cols (pay, key)
    c_lp, c_lk, c_rp, c_rk = cols
    want = oracle.inner_join(bk, bp, pk, pp)  # (key, lpay, key, rpay)
    g = oracle.sort_rows(c_lk, c_lp, c_rk, c_rp)
    w = oracle.sort_rows(*want)
    assert len(g[0]) == len(w[0])
    for a, b in zip(g, w):
        assert (a == b).all()


def test_timestamp_duration_dtypes(dj, comm):
    # chrono-typed key/payload columns (reference dtype coverage,
    # compare_against_single_gpu.cu:237-268): join on the integer rep
    n = 100_000
    bk, bp = oracle.gen_build(n)
    pk, pp = oracle.gen_probe(n, n)
    d = {name: dj.DeviceArray.from_numpy(a) for name, a in
         [("bk", bk), ("bp", bp), ("pk", pk), ("pp", pp)]}
    cols = dj.cpp_distributed_inner_join_cols(
        comm,
        [(dj.TYPE_TIMESTAMP_NS, d["bk"].ptr), (dj.TYPE_DURATION_MS, d["bp"].ptr)], n,
        [(dj.TYPE_TIMESTAMP_NS, d["pk"].ptr), (dj.TYPE_INT64, d["pp"].ptr)], n)
    want = oracle.inner_join(bk, bp, pk, pp)
    g = oracle.sort_rows(*[c.astype(np.int64) for c in cols])
    w = oracle.sort_rows(*want)
    assert len(g[0]) == len(w[0])
    for a, b in zip(g, w):
        assert (a == b).all()


def test_timestamp_days_int32_rep(dj, comm):
    n = 50_000
    bk64, bp64 = oracle.gen_build(n, rand_max=min(2 * n, 2**31 - 2))
    pk64, pp64 = oracle.gen_probe(n, n, rand_max=min(2 * n, 2**31 - 2))
    d_bk = _upload_i32(dj, bk64.astype(np.int32))
    d_pk = _upload_i32(dj, pk64.astype(np.int32))
    d_bp = _upload_i32(dj, bp64.astype(np.int32))
    d_pp = _upload_i32(dj, pp64.astype(np.int32))
    cols = dj.cpp_distributed_inner_join_cols(
        comm, [(dj.TYPE_TIMESTAMP_DAYS, d_bk), (dj.TYPE_TIMESTAMP_DAYS, d_bp)], n,
        [(dj.TYPE_TIMESTAMP_DAYS, d_pk), (dj.TYPE_TIMESTAMP_DAYS, d_pp)], n)
    want = oracle.inner_join(bk64, bp64, pk64, pp64)
    g = oracle.sort_rows(*[c.astype(np.int64) for c in cols])
    w = oracle.sort_rows(*want)
    assert len(g[0]) == len(w[0])
    for a, b in zip(g, w):
        assert (a == b).all()
    for p in (d_bk, d_pk, d_bp, d_pp):
        dj.lib().dj_dfree(p)
