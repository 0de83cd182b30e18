"""GPU parity tests: the gfx950 HIP path against the CPU oracle on identical
seeded inputs — the parity gate of SURVEY.md §8(c). All tests call through
the C ABI (the product path); nothing here touches torch compute.

Bar: bit-exact (order-insensitive where row order is unspecified by the API).
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


# -------------------------------------------------------------- generator

@pytest.mark.parametrize("n", [1000, 1_000_000])
def test_generator_parity_build(dj, n):
    gk, gp = dj.generate_build(n)
    ok, op = oracle.gen_build(n)
    assert (gk.to_numpy() == ok).all()
    assert (gp.to_numpy() == op).all()


@pytest.mark.parametrize("sel", [0.0, 0.3, 1.0])
def test_generator_parity_probe(dj, sel):
    n = 500_000
    gk, gp = dj.generate_probe(n, n, selectivity=sel)
    ok, op = oracle.gen_probe(n, n, selectivity=sel)
    assert (gk.to_numpy() == ok).all()
    assert (gp.to_numpy() == op).all()


def test_generator_parity_slice(dj):
    # rank-sliced generation equals the slice of the global table
    n = 100_000
    gk, _ = dj.generate_build(n, row0=40_000, nrows=10_000)
    ok, _ = oracle.gen_build(n, row0=40_000, nrows=10_000)
    assert (gk.to_numpy() == ok).all()


# -------------------------------------------------------------- partition

@pytest.mark.parametrize("nparts,hash_fn,seed", [
    (1, 0, 12345678), (2, 0, 12345678), (8, 0, 12345678), (8, 0, 87654321),
    (13, 0, 0), (64, 0, 12345678), (8, 1, 0),
    # beyond one wave's 64 lanes: the reference's own test envelope reaches
    # 8 ranks x over_decom 10 = 80 (compare_against_single_gpu.cu:237-268);
    # 100/256/1024 cover the multi-chunk dispatch tiers incl. the cap
    (80, 0, 12345678), (100, 0, 0), (256, 0, 12345678), (1024, 1, 0),
])
def test_partition_parity(dj, nparts, hash_fn, seed):
    n = 1_000_000
    k, p = oracle.gen_probe(n, n)
    dk = dj.DeviceArray.from_numpy(k)
    dp = dj.DeviceArray.from_numpy(p)
    gk, gp, goff = dj.hash_partition(dk, dp, n, nparts, hash_fn, seed)
    ok, op, ooff = oracle.partition(k, p, nparts, hash_fn, seed)
    assert (goff == ooff).all()
    # stable partition is fully deterministic => bit-exact, not just set-equal
    assert (gk.to_numpy() == ok).all()
    assert (gp.to_numpy() == op).all()


def test_partition_empty(dj):
    dk = dj.DeviceArray(1)
    dp = dj.DeviceArray(1)
    gk, gp, goff = dj.hash_partition(dk, dp, 0, 8, 0, 0)
    assert (goff == 0).all()


# ------------------------------------------------------------------- join

def _join_parity(dj, lk, lp, rk, rp, cap=None):
    dlk, dlp = dj.DeviceArray.from_numpy(lk), dj.DeviceArray.from_numpy(lp)
    drk, drp = dj.DeviceArray.from_numpy(rk), dj.DeviceArray.from_numpy(rp)
    got = dj.local_inner_join(dlk, dlp, len(lk), drk, drp, len(rk), cap=cap)
    want = oracle.inner_join(lk, lp, rk, rp)
    g = oracle.sort_rows(*got)
    w = oracle.sort_rows(*want)
    assert len(g[0]) == len(w[0])
    for a, b in zip(g, w):
        assert (a == b).all()


def test_join_analytical_multiples(dj):
    # the reference's own KAT (compare_against_analytical.cu:44-54) on GPU
    size = 300_000
    lk = np.arange(size, dtype=np.int64) * 3
    lp = np.arange(size, dtype=np.int64)
    rk = np.arange(size, dtype=np.int64) * 5
    rp = np.arange(size, dtype=np.int64)
    dlk, dlp = dj.DeviceArray.from_numpy(lk), dj.DeviceArray.from_numpy(lp)
    drk, drp = dj.DeviceArray.from_numpy(rk), dj.DeviceArray.from_numpy(rp)
    c0, c1, c2, c3 = dj.local_inner_join(dlk, dlp, size, drk, drp, size)
    assert len(c0) == size // 5
    assert (c0 % 15 == 0).all() and (c1 == c0 // 3).all()
    assert (c3 == c2 // 5).all() and (c0 == c2).all()


def test_join_parity_config1(dj):
    # BASELINE config 1: 1M x 1M, selectivity 0.3, unique build keys
    n = 1_000_000
    bk, bp = oracle.gen_build(n)
    pk, pp = oracle.gen_probe(n, n, selectivity=0.3)
    _join_parity(dj, bk, bp, pk, pp)


def test_join_parity_duplicates(dj):
    # non-unique build keys with a tiny key space => heavy duplicate chains
    n = 100_000
    rng = np.random.RandomState(7)
    lk = rng.randint(0, 1000, n).astype(np.int64)
    lp = np.arange(n, dtype=np.int64)
    rk = rng.randint(0, 1000, n).astype(np.int64)
    rp = np.arange(n, dtype=np.int64)
    # ~n*n/1000 = 10M output rows
    _join_parity(dj, lk, lp, rk, rp)


def test_join_sparse_buckets_drain(dj):
    # ~100 distinct keys over B=256 buckets: most buckets empty, so many
    # 4-bucket flush groups end on an empty bucket while earlier buckets
    # staged matches under the watermark — exercises the join kernel's
    # goto-drain path (the exit-clean guarantee; see lds_join_kernel)
    n = 2_000
    rng = np.random.RandomState(11)
    lk = rng.randint(0, 100, n).astype(np.int64)
    lp = np.arange(n, dtype=np.int64)
    rk = rng.randint(0, 100, n).astype(np.int64)
    rp = np.arange(n, dtype=np.int64)
    _join_parity(dj, lk, lp, rk, rp)


def test_join_sentinel_minus_one_keys(dj):
    # key == -1 is the hash tables' reserved empty marker; legal int64 data
    # must still join (reference cudf::inner_join joins -1 normally) — the
    # paths skip them in-table and append the -1 cross product out-of-band
    rng = np.random.RandomState(31)
    n = 100_000
    lk = rng.randint(0, 50_000, n).astype(np.int64)
    lk[rng.choice(n, 40, replace=False)] = -1
    lp = np.arange(n, dtype=np.int64)
    rk = rng.randint(0, 50_000, n).astype(np.int64)
    rk[rng.choice(n, 25, replace=False)] = -1
    rp = np.arange(n, dtype=np.int64)
    # 40 x 25 = 1000 extra cross-product rows on top of the normal matches
    _join_parity(dj, lk, lp, rk, rp)


def test_join_sentinel_only_one_side(dj):
    # -1 on the probe side only: no matches for those rows, no error
    n = 50_000
    lk = np.arange(n, dtype=np.int64)
    lp = np.arange(n, dtype=np.int64)
    rk = np.arange(n, dtype=np.int64)
    rk[:100] = -1
    rp = np.arange(n, dtype=np.int64)
    _join_parity(dj, lk, lp, rk, rp)


def test_join_duplicate_variance_slack_overflow(dj):
    # the TPC-H lineitem shape: probe keys with multiplicity ~16 inflate
    # per-bucket variance beyond the Poisson slack model, so pass B's capB
    # overflows (bit 2) and the join must redo through the exact compact
    # path (dj_capi compact retry), not the slow global-table fallback
    rng = np.random.RandomState(23)
    n_build, n_probe = 250_000, 4_000_000
    lk = np.arange(n_build, dtype=np.int64) * 4 + 1
    lp = np.arange(n_build, dtype=np.int64)
    oi = rng.randint(0, n_build, n_probe).astype(np.int64)
    rk = oi * 4 + 1
    rp = np.arange(n_probe, dtype=np.int64)
    _join_parity(dj, lk, lp, rk, rp)


def test_join_parity_selectivity_1(dj):
    n = 200_000
    bk, bp = oracle.gen_build(n)
    pk, pp = oracle.gen_probe(n, n, selectivity=1.0)
    _join_parity(dj, bk, bp, pk, pp)


def test_join_empty_sides(dj):
    n = 1000
    bk, bp = oracle.gen_build(n)
    d = dj.DeviceArray.from_numpy(bk)
    dp = dj.DeviceArray.from_numpy(bp)
    empty = dj.DeviceArray(1)
    assert dj.lib().dj_local_inner_join(empty.ptr, empty.ptr, 0, d.ptr, dp.ptr, n,
                                        None, None, None, None, 0) == 0
    assert dj.lib().dj_local_inner_join(d.ptr, dp.ptr, n, empty.ptr, empty.ptr, 0,
                                        None, None, None, None, 0) == 0


def test_join_overflow_retry(dj):
    # cap smaller than the result: count is still exact, retry path works
    n = 10_000
    bk, bp = oracle.gen_build(n)
    pk, pp = oracle.gen_probe(n, n, selectivity=1.0)
    _join_parity(dj, bk, bp, pk, pp, cap=16)


def test_bucket_vs_global_engines(dj):
    # the bucketed-LDS engine and the global-table engine must agree
    n = 500_000
    bk, bp = oracle.gen_build(n)
    pk, pp = oracle.gen_probe(n, n, selectivity=0.5)
    dlk, dlp = dj.DeviceArray.from_numpy(bk), dj.DeviceArray.from_numpy(bp)
    drk, drp = dj.DeviceArray.from_numpy(pk), dj.DeviceArray.from_numpy(pp)
    a = oracle.sort_rows(*dj.local_inner_join(dlk, dlp, n, drk, drp, n))
    b = oracle.sort_rows(*dj.local_inner_join_global(dlk, dlp, n, drk, drp, n))
    assert len(a[0]) == len(b[0])
    for x, y in zip(a, b):
        assert (x == y).all()


def test_skew_fallback_single_hot_key(dj):
    # one key repeated 100k times on the build side => its bucket overflows
    # the LDS row cap and must take the global-table fallback path
    n = 100_000
    lk = np.full(n, 42, dtype=np.int64)
    lk[:100] = np.arange(100, dtype=np.int64) + 1000  # a few normal buckets too
    lp = np.arange(n, dtype=np.int64)
    rk = np.array([42, 1000, 7, 42], dtype=np.int64)
    rp = np.arange(4, dtype=np.int64)
    _join_parity(dj, lk, lp, rk, rp)


def test_negative_keys(dj):
    lk = np.array([-5, -3, 0, 7, 2**62, -2**62], dtype=np.int64)
    lp = np.arange(6, dtype=np.int64)
    rk = np.array([-3, 7, -5, 123, -2**62], dtype=np.int64)
    rp = np.arange(5, dtype=np.int64)
    _join_parity(dj, lk, lp, rk, rp)


def test_slack_partition_overflow_redo(dj):
    # at this size the local partition takes the slack pass-A path (PA > 1,
    # capA ~ n/PA + 6%); a 50k-copy hot key pushes its pass-A group past the
    # slack (bit 2 of any_overflow) and the whole join must redo via the
    # global-table path with identical results (dj_capi.hip
    # dj_bucket_local_join). 50k duplicates keep the fallback's collision
    # chain quadratic-cost bounded (~1.2e9 slot probes).
    n = 1_000_000
    rng = np.random.default_rng(7)
    lk = rng.integers(1 << 40, size=n).astype(np.int64)
    lk[:50_000] = 42  # skew one pass-A group past its slack capacity
    lp = np.arange(n, dtype=np.int64)
    rk = np.concatenate(
        [np.array([42], dtype=np.int64), rng.integers(1 << 40, size=5000).astype(np.int64)]
    )
    rp = np.arange(rk.size, dtype=np.int64)
    _join_parity(dj, lk, lp, rk, rp)


def test_large_table_f512_parity(dj):
    # >209M build rows push B past 262144 into the PA=1024 x F=512
    # decomposition (runtime-F pass B, subF_of extended bit field); parity
    # against the oracle with a small probe side keeps the test fast
    n = 250_000_000
    lk, lp = oracle.gen_build(n)
    m = 1_000_000
    rk, rp = oracle.gen_probe(m, n)
    _join_parity(dj, lk, lp, rk, rp)


def test_benchmark_size_parity(dj):
    """Bit-exact (order-insensitive) parity against the CPU oracle AT the
    headline config-2 size: 100M x 100M int64/int64, selectivity 0.3, unique
    build keys — the exact workload bench.py times. The slowest parity test
    (~2 min: oracle join + 30M-row lexsort) but the strongest claim."""
    n = 100_000_000
    lk, lp = oracle.gen_build(n)
    rk, rp = oracle.gen_probe(n, n, selectivity=0.3)
    _join_parity(dj, lk, lp, rk, rp)
