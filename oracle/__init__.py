"""CPU oracle wrapper — TEST INFRASTRUCTURE ONLY.

Only tests/, __graft_entry__.smoke() and bench.py's cpu_baseline leg may
import this package, and only as the checker / reported CPU baseline. The
product path (distributed_join_amd) never imports it.

See oracle/oracle.c for what is restated from the reference and how the
oracle is pinned.
"""
import ctypes
import os
import subprocess

import numpy as np

_DIR = os.path.dirname(os.path.abspath(__file__))
_SO = os.path.join(_DIR, "liboracle.so")


def build():
    """Compile the oracle shared library (gcc, seconds)."""
    subprocess.run(["make", "-C", _DIR], check=True, capture_output=True)


_lib = None


def lib():
    global _lib
    if _lib is None:
        if not os.path.exists(_SO):
            build()
        _lib = ctypes.CDLL(_SO)
        i64p = ctypes.POINTER(ctypes.c_int64)
        _lib.oracle_gen_build.argtypes = [i64p, i64p, ctypes.c_int64, ctypes.c_int64,
                                          ctypes.c_uint64, ctypes.c_int, ctypes.c_int64,
                                          ctypes.c_int64]
        _lib.oracle_gen_probe.argtypes = [i64p, i64p, ctypes.c_int64, ctypes.c_int64,
                                          ctypes.c_double, ctypes.c_uint64, ctypes.c_int64,
                                          ctypes.c_int64]
        _lib.oracle_row_hash.argtypes = [ctypes.c_int64, ctypes.c_int, ctypes.c_uint32]
        _lib.oracle_row_hash.restype = ctypes.c_uint32
        _lib.oracle_partition.argtypes = [i64p, i64p, ctypes.c_int64, ctypes.c_int,
                                          ctypes.c_int, ctypes.c_uint32, i64p, i64p, i64p]
        _lib.oracle_inner_join.argtypes = [i64p, i64p, ctypes.c_int64,
                                           i64p, i64p, ctypes.c_int64,
                                           i64p, i64p, i64p, i64p, ctypes.c_int64]
        _lib.oracle_inner_join.restype = ctypes.c_int64
        _lib.oracle_cpu_radix_join.argtypes = [i64p, i64p, ctypes.c_int64,
                                               i64p, i64p, ctypes.c_int64,
                                               i64p, i64p, i64p, i64p, ctypes.c_int64,
                                               ctypes.c_int]
        _lib.oracle_cpu_radix_join.restype = ctypes.c_int64
        _lib.oracle_num_threads.restype = ctypes.c_int
    return _lib


def _p(a):
    if a is None:
        return None
    return a.ctypes.data_as(ctypes.POINTER(ctypes.c_int64))


HASH_MURMUR3 = 0
HASH_IDENTITY = 1
DEFAULT_SEED = 1234


def gen_build(build_n, rand_max=None, seed=DEFAULT_SEED, uniq=True, row0=0, nrows=None):
    """Rows [row0, row0+nrows) of the global build table (keys, payloads)."""
    if rand_max is None:
        rand_max = 2 * build_n
    if nrows is None:
        nrows = build_n - row0
    keys = np.empty(nrows, dtype=np.int64)
    pay = np.empty(nrows, dtype=np.int64)
    lib().oracle_gen_build(_p(keys), _p(pay), build_n, rand_max, seed, int(uniq), row0, nrows)
    return keys, pay


def gen_probe(probe_n, build_n, rand_max=None, selectivity=0.3, seed=DEFAULT_SEED,
              row0=0, nrows=None):
    if rand_max is None:
        rand_max = 2 * build_n
    if nrows is None:
        nrows = probe_n - row0
    keys = np.empty(nrows, dtype=np.int64)
    pay = np.empty(nrows, dtype=np.int64)
    lib().oracle_gen_probe(_p(keys), _p(pay), build_n, rand_max, selectivity, seed, row0, nrows)
    return keys, pay


def row_hash(key, hash_fn=HASH_MURMUR3, seed=0):
    return lib().oracle_row_hash(int(key), hash_fn, seed)


def partition(keys, payloads, nparts, hash_fn=HASH_MURMUR3, seed=0):
    n = len(keys)
    ok = np.empty(n, dtype=np.int64)
    op = np.empty(n, dtype=np.int64)
    off = np.empty(nparts + 1, dtype=np.int64)
    lib().oracle_partition(_p(keys), _p(payloads), n, nparts, hash_fn, seed, _p(ok), _p(op), _p(off))
    return ok, op, off


def inner_join(lk, lp, rk, rp, cap=None):
    """Returns (c0, c1, c2, c3) = (lkey, lpay, rkey, rpay), order unspecified."""
    if cap is None:
        cap = max(len(rk), 16) * 2
    while True:
        c0 = np.empty(cap, dtype=np.int64)
        c1 = np.empty(cap, dtype=np.int64)
        c2 = np.empty(cap, dtype=np.int64)
        c3 = np.empty(cap, dtype=np.int64)
        n = lib().oracle_inner_join(_p(lk), _p(lp), len(lk), _p(rk), _p(rp), len(rk),
                                    _p(c0), _p(c1), _p(c2), _p(c3), cap)
        if n <= cap:
            return c0[:n], c1[:n], c2[:n], c3[:n]
        cap = n


def cpu_radix_join(lk, lp, rk, rp, cap=None, nthreads=0, count_only=False):
    """OpenMP radix-partition join — the cpu_baseline leg. Returns count or rows."""
    if count_only:
        n = lib().oracle_cpu_radix_join(_p(lk), _p(lp), len(lk), _p(rk), _p(rp), len(rk),
                                        None, None, None, None, 0, nthreads)
        return n
    if cap is None:
        cap = max(len(rk), 16) * 2
    while True:
        c0 = np.empty(cap, dtype=np.int64)
        c1 = np.empty(cap, dtype=np.int64)
        c2 = np.empty(cap, dtype=np.int64)
        c3 = np.empty(cap, dtype=np.int64)
        n = lib().oracle_cpu_radix_join(_p(lk), _p(lp), len(lk), _p(rk), _p(rp), len(rk),
                                        _p(c0), _p(c1), _p(c2), _p(c3), cap, nthreads)
        if n <= cap:
            return c0[:n], c1[:n], c2[:n], c3[:n]
        cap = n


def num_threads():
    return lib().oracle_num_threads()


def sort_rows(*cols):
    """Canonical order-insensitive form: lexicographic sort of the row tuples."""
    idx = np.lexsort(tuple(reversed(cols)))
    return tuple(c[idx] for c in cols)
