"""distributed_join_amd — MI355X-native distributed repartitioned hash join.

Python binding (ctypes) over the C ABI of libdistjoin.so (see
include/distributed_join.h for the contract and the reference interfaces each
entry point replaces). This is the PRODUCT path: hand-written gfx950 HIP
kernels + RCCL over xGMI. It never falls back to CPU — if the HIP extension
is missing or no GPU is visible, compute calls fail loudly.
"""
import ctypes
import os

import numpy as np

_DIR = os.path.dirname(os.path.abspath(__file__))
_SO = os.path.join(_DIR, "libdistjoin.so")

HASH_MURMUR3 = 0
HASH_IDENTITY = 1
SEED_INTRA = 12345678
SEED_INTER = 87654321
DEFAULT_SEED = 1234

PHASES = {
    "generate": 0, "part_count": 1, "part_scan": 2, "part_scatter": 3,
    "table_init": 4, "build": 5, "probe": 6, "comm": 7, "concat": 8,
    "bucket_count": 9, "bucket_scan": 10, "bucket_scatter": 11, "join_fused": 12,
}

_lib = None


class ExtensionMissing(RuntimeError):
    pass


def lib():
    global _lib
    if _lib is None:
        if not os.path.exists(_SO):
            raise ExtensionMissing(
                f"{_SO} not built. Run `make -C {os.path.join(_DIR, 'csrc')}` "
                "(or __graft_entry__.build()). The product path has no CPU fallback.")
        _lib = ctypes.CDLL(_SO)
        c = ctypes
        i64, u64, u32, i32, dbl = c.c_int64, c.c_uint64, c.c_uint32, c.c_int, c.c_double
        vp = c.c_void_p
        sigs = {
            "dj_device_count": ([], i32),
            "dj_set_device": ([i32], None),
            "dj_dmalloc": ([i64], vp),
            "dj_dfree": ([vp], None),
            "dj_memcpy_h2d": ([vp, vp, i64], None),
            "dj_memcpy_d2h": ([vp, vp, i64], None),
            "dj_memcpy_d2d": ([vp, vp, i64], None),
            "dj_sync": ([], None),
            "dj_generate_build": ([vp, vp, i64, i64, u64, i32, i64, i64], None),
            "dj_generate_probe": ([vp, vp, i64, i64, dbl, u64, i64, i64], None),
            "dj_partition_scratch_bytes": ([i64, i32], i64),
            "dj_hash_partition": ([vp, vp, i64, i32, i32, u32, vp, vp, vp, vp], None),
            "dj_join_table_slots": ([i64], i64),
            "dj_join_table_init": ([vp, i64], None),
            "dj_join_build": ([vp, vp, i64, vp, i64, vp], None),
            "dj_join_probe": ([vp, vp, i64, vp, i64, vp, vp, vp, vp, i64, vp], None),
            "dj_read_counter_i64": ([vp], i64),
            "dj_read_error_i32": ([vp], i32),
            "dj_local_inner_join": ([vp, vp, i64, vp, vp, i64, vp, vp, vp, vp, i64], i64),
            "dj_local_inner_join_global": ([vp, vp, i64, vp, vp, i64, vp, vp, vp, vp, i64], i64),
            "dj_bucket_join_scratch_bytes": ([i64, i64], i64),
            "dj_bucket_local_join": ([vp, vp, i64, vp, vp, i64, vp, vp, vp, vp, i64,
                                      vp, vp, vp], None),
            "dj_timing_enable": ([i32], None),
            "dj_timing_reset": ([], None),
            "dj_timing_total_ms": ([i32], dbl),
            "dj_timing_launches": ([i32], i64),
            "dj_rccl_unique_id_bytes": ([], i32),
            "dj_rccl_get_unique_id": ([vp], None),
            "dj_comm_init": ([i32, i32, vp], None),
            "dj_comm_finalize": ([], None),
            "dj_comm_rank": ([], i32),
            "dj_comm_size": ([], i32),
            "dj_all_to_all_i64": ([vp, vp, vp, vp], None),
            "dj_exchange_sizes": ([vp, vp], None),
            "dj_rccl_selftest": ([i64], i32),
            "dj_compress_roundtrip": ([vp, i64, i32, i32, i32, i32, vp], i64),
            "dj_cpp_comm_create": ([i32, i32, vp], vp),
            "dj_cpp_comm_destroy": ([vp], None),
            "dj_cpp_distributed_inner_join_i64": ([vp, vp, vp, i64, vp, vp, i64, i32, i32], vp),
            "dj_cpp_shuffle_on_i64": ([vp, vp, vp, i64, i32, u32], vp),
            "dj_cpp_shuffle_on_i64_comp": ([vp, vp, vp, i64, i32, u32, i32], vp),
            "dj_cpp_distributed_inner_join_i64_opts": ([vp, vp, vp, i64, vp, vp, i64,
                                                        i32, i32, i32], vp),
            "dj_cpp_distributed_inner_join_i64_full": ([vp, vp, vp, i64, vp, vp, i64,
                                                        i32, i32, i32, i32], vp),
            "dj_gen_test_strings": ([vp, i64, ctypes.POINTER(ctypes.c_void_p),
                                     ctypes.POINTER(ctypes.c_void_p),
                                     ctypes.POINTER(ctypes.c_int64)], None),
            "dj_cpp_distributed_inner_join_i64str": ([vp, vp, vp, vp, i64, i64,
                                                      vp, vp, vp, i64, i64, i32, i32], vp),
            "dj_cpp_distributed_inner_join_cols": ([vp, vp, i32, i64, vp, i32, i64,
                                                    i32, i32, i32, i32], vp),
            "dj_cpp_distributed_inner_join_cols_multi": ([vp, vp, i32, i64, vp, i32, i64,
                                                          vp, vp, i32, i32, i32], vp),
            "dj_cpp_distribute_collect_roundtrip_i64": ([vp, vp, vp, i64], vp),
            "dj_table_column_type": ([vp, i32], i32),
            "dj_table_column_chars": ([vp, i32], vp),
            "dj_table_column_chars_size": ([vp, i32], i64),
            "dj_table_num_rows": ([vp], i64),
            "dj_table_num_columns": ([vp], i32),
            "dj_table_column_data": ([vp, i32], vp),
            "dj_table_free": ([vp], None),
        }
        for name, (argtypes, restype) in sigs.items():
            f = getattr(_lib, name)
            f.argtypes = argtypes
            f.restype = restype
    return _lib


def require_gpu():
    n = lib().dj_device_count()
    if n < 1:
        raise ExtensionMissing("no HIP device visible; the product path requires an MI355X")
    return n


# ---------------------------------------------------------------- helpers

class DeviceArray:
    """Owning device buffer of int64 elements."""

    def __init__(self, n):
        self.n = int(n)
        self.ptr = lib().dj_dmalloc(max(self.n, 1) * 8)

    def free(self):
        if self.ptr:
            lib().dj_dfree(self.ptr)
            self.ptr = None

    def __del__(self):
        try:
            self.free()
        except Exception:
            pass

    def to_numpy(self, n=None):
        n = self.n if n is None else int(n)
        out = np.empty(n, dtype=np.int64)
        lib().dj_memcpy_d2h(out.ctypes.data, self.ptr, n * 8)
        return out

    @classmethod
    def from_numpy(cls, a):
        a = np.ascontiguousarray(a, dtype=np.int64)
        d = cls(len(a))
        if len(a):
            lib().dj_memcpy_h2d(d.ptr, a.ctypes.data, len(a) * 8)
        return d


def generate_build(n_global, rand_max=None, seed=DEFAULT_SEED, uniq=True, row0=0, nrows=None):
    if rand_max is None:
        rand_max = 2 * n_global
    if nrows is None:
        nrows = n_global - row0
    keys, pay = DeviceArray(nrows), DeviceArray(nrows)
    lib().dj_generate_build(keys.ptr, pay.ptr, n_global, rand_max, seed, int(uniq), row0, nrows)
    return keys, pay


def generate_probe(probe_n, build_n_global, rand_max=None, selectivity=0.3, seed=DEFAULT_SEED,
                   row0=0, nrows=None):
    if rand_max is None:
        rand_max = 2 * build_n_global
    if nrows is None:
        nrows = probe_n - row0
    keys, pay = DeviceArray(nrows), DeviceArray(nrows)
    lib().dj_generate_probe(keys.ptr, pay.ptr, build_n_global, rand_max, selectivity, seed,
                            row0, nrows)
    return keys, pay


def hash_partition(d_keys, d_pay, n, nparts, hash_fn=HASH_MURMUR3, seed=0,
                   d_out_keys=None, d_out_pay=None, d_scratch=None):
    """Stable partition; returns (d_out_keys, d_out_pay, host offsets)."""
    L = lib()
    own_scratch = d_scratch is None
    if own_scratch:
        d_scratch = L.dj_dmalloc(L.dj_partition_scratch_bytes(n, nparts))
    if d_out_keys is None:
        d_out_keys = DeviceArray(n)
    if d_out_pay is None:
        d_out_pay = DeviceArray(n)
    offsets = np.zeros(nparts + 1, dtype=np.int64)
    L.dj_hash_partition(d_keys.ptr, d_pay.ptr if d_pay else None, n, nparts, hash_fn, seed,
                        d_out_keys.ptr, d_out_pay.ptr, offsets.ctypes.data, d_scratch)
    if own_scratch:
        L.dj_dfree(d_scratch)
    return d_out_keys, d_out_pay, offsets


def local_inner_join(d_lk, d_lp, ln, d_rk, d_rp, rn, cap=None):
    """One-call local join; returns 4 numpy columns (lkey,lpay,rkey,rpay)."""
    L = lib()
    if cap is None:
        cap = max(int(rn) * 2, 16)
    while True:
        outs = [DeviceArray(cap) for _ in range(4)]
        n = L.dj_local_inner_join(d_lk.ptr, d_lp.ptr, ln, d_rk.ptr, d_rp.ptr, rn,
                                  outs[0].ptr, outs[1].ptr, outs[2].ptr, outs[3].ptr, cap)
        if n <= cap:
            res = tuple(o.to_numpy(n) for o in outs)
            for o in outs:
                o.free()
            return res
        for o in outs:
            o.free()
        cap = n


def local_inner_join_global(d_lk, d_lp, ln, d_rk, d_rp, rn, cap=None):
    """Global-table engine (the skew-fallback path), for cross-checking."""
    L = lib()
    if cap is None:
        cap = max(int(rn) * 2, 16)
    while True:
        outs = [DeviceArray(cap) for _ in range(4)]
        n = L.dj_local_inner_join_global(d_lk.ptr, d_lp.ptr, ln, d_rk.ptr, d_rp.ptr, rn,
                                         outs[0].ptr, outs[1].ptr, outs[2].ptr, outs[3].ptr,
                                         cap)
        if n <= cap:
            res = tuple(o.to_numpy(n) for o in outs)
            for o in outs:
                o.free()
            return res
        for o in outs:
            o.free()
        cap = n


TYPE_INT8, TYPE_INT32, TYPE_INT64, TYPE_STRING = 1, 2, 3, 4
# chrono reps (4-byte DAYS, 8-byte the rest) — join on the integer rep
TYPE_TIMESTAMP_DAYS, TYPE_TIMESTAMP_S, TYPE_TIMESTAMP_MS = 5, 6, 7
TYPE_TIMESTAMP_US, TYPE_TIMESTAMP_NS = 8, 9
TYPE_DURATION_DAYS, TYPE_DURATION_S, TYPE_DURATION_MS = 10, 11, 12
TYPE_DURATION_US, TYPE_DURATION_NS = 13, 14


def table_to_numpy(tbl_ptr):
    """Copy an opaque cudf::table* (C ABI) into numpy columns and free it.
    Fixed-width columns -> int64/int32 arrays; STRING columns ->
    (offsets int32 array, chars uint8 array) tuples."""
    L = lib()
    n = L.dj_table_num_rows(tbl_ptr)
    ncols = L.dj_table_num_columns(tbl_ptr)
    cols = []
    for c in range(ncols):
        t = L.dj_table_column_type(tbl_ptr, c)
        data = L.dj_table_column_data(tbl_ptr, c)
        if t == TYPE_STRING:
            off = np.empty(n + 1, dtype=np.int32)
            L.dj_memcpy_d2h(off.ctypes.data, data, (n + 1) * 4)
            nch = L.dj_table_column_chars_size(tbl_ptr, c)
            ch = np.empty(max(nch, 1), dtype=np.uint8)
            if nch:
                L.dj_memcpy_d2h(ch.ctypes.data, L.dj_table_column_chars(tbl_ptr, c), nch)
            cols.append((off, ch[:nch]))
        elif t in (TYPE_INT32, TYPE_TIMESTAMP_DAYS, TYPE_DURATION_DAYS):
            out = np.empty(n, dtype=np.int32)
            if n:
                L.dj_memcpy_d2h(out.ctypes.data, data, n * 4)
            cols.append(out)
        else:
            out = np.empty(n, dtype=np.int64)
            if n:
                L.dj_memcpy_d2h(out.ctypes.data, data, n * 8)
            cols.append(out)
    L.dj_table_free(tbl_ptr)
    return cols


class CppCommunicator:
    """RCCLCommunicator (or single-process LocalCommunicator) handle."""

    def __init__(self, rank=0, size=1, id_bytes=None):
        L = lib()
        ptr = None if id_bytes is None else id_bytes.ctypes.data
        self.ptr = L.dj_cpp_comm_create(rank, size, ptr)

    def destroy(self):
        if self.ptr:
            lib().dj_cpp_comm_destroy(self.ptr)
            self.ptr = None


def cpp_distributed_inner_join(comm, d_lk, d_lp, ln, d_rk, d_rp, rn, over_decom=1,
                               report_timing=False):
    """The C++ drop-in path (distributed_inner_join) over int64 columns.
    Returns numpy columns (lkey, lpay, rkey, rpay) of this rank's result."""
    t = lib().dj_cpp_distributed_inner_join_i64(comm.ptr, d_lk.ptr, d_lp.ptr, ln,
                                                d_rk.ptr, d_rp.ptr, rn, over_decom,
                                                int(report_timing))
    return table_to_numpy(t)


def gen_test_strings(d_keys, n):
    """Deterministic string payload from keys (len=k%7+1, char='a'+k%26).
    Returns (offsets_ptr, chars_ptr, chars_bytes); caller frees via dj_dfree."""
    L = lib()
    off = ctypes.c_void_p()
    ch = ctypes.c_void_p()
    nb = ctypes.c_int64()
    L.dj_gen_test_strings(d_keys.ptr, n, ctypes.byref(off), ctypes.byref(ch),
                          ctypes.byref(nb))
    return off.value, ch.value, nb.value


def cpp_distributed_inner_join_str(comm, d_lk, l_strings, ln, d_rk, r_strings, rn,
                                   over_decom=1):
    """Distributed join with int64 keys and STRING payloads (config 4)."""
    lo, lc, lb = l_strings
    ro, rc, rb = r_strings
    t = lib().dj_cpp_distributed_inner_join_i64str(
        comm.ptr, d_lk.ptr, lo, lc, lb, ln, d_rk.ptr, ro, rc, rb, rn, over_decom, 0)
    return table_to_numpy(t)


class ColDesc(ctypes.Structure):
    _fields_ = [("type_id", ctypes.c_int), ("data", ctypes.c_void_p),
                ("chars", ctypes.c_void_p), ("chars_bytes", ctypes.c_int64)]


def cpp_distributed_inner_join_cols(comm, lcols, ln, rcols, rn, key_l=0, key_r=0,
                                    over_decom=1):
    """Generic join over column descriptors: each col is
    (type_id, data_ptr[, chars_ptr, chars_bytes])."""
    def pack(cols):
        arr = (ColDesc * len(cols))()
        for i, c in enumerate(cols):
            arr[i].type_id = c[0]
            arr[i].data = c[1]
            arr[i].chars = c[2] if len(c) > 2 else None
            arr[i].chars_bytes = c[3] if len(c) > 3 else 0
        return arr
    la, ra = pack(lcols), pack(rcols)
    t = lib().dj_cpp_distributed_inner_join_cols(
        comm.ptr, ctypes.cast(la, ctypes.c_void_p), len(lcols), ln,
        ctypes.cast(ra, ctypes.c_void_p), len(rcols), rn, key_l, key_r, over_decom, 0)
    return table_to_numpy(t)


def cpp_distributed_inner_join_cols_multi(comm, lcols, ln, rcols, rn, left_on, right_on,
                                          over_decom=1):
    """Composite-key join over column descriptors (left_on/right_on are
    lists of key column indices; the single-key restriction lifts here)."""
    def pack(cols):
        arr = (ColDesc * len(cols))()
        for i, c in enumerate(cols):
            arr[i].type_id = c[0]
            arr[i].data = c[1]
            arr[i].chars = c[2] if len(c) > 2 else None
            arr[i].chars_bytes = c[3] if len(c) > 3 else 0
        return arr
    la, ra = pack(lcols), pack(rcols)
    lon = np.asarray(left_on, dtype=np.int32)
    ron = np.asarray(right_on, dtype=np.int32)
    t = lib().dj_cpp_distributed_inner_join_cols_multi(
        comm.ptr, ctypes.cast(la, ctypes.c_void_p), len(lcols), ln,
        ctypes.cast(ra, ctypes.c_void_p), len(rcols), rn,
        lon.ctypes.data, ron.ctypes.data, len(lon), over_decom, 0)
    return table_to_numpy(t)


def cpp_shuffle_on(comm, d_keys, d_pay, n, hash_fn=HASH_MURMUR3, seed=0,
                   compression=False):
    if compression:
        t = lib().dj_cpp_shuffle_on_i64_comp(comm.ptr, d_keys.ptr, d_pay.ptr, n, hash_fn,
                                             seed, 1)
    else:
        t = lib().dj_cpp_shuffle_on_i64(comm.ptr, d_keys.ptr, d_pay.ptr, n, hash_fn, seed)
    return table_to_numpy(t)


def timing(phase):
    return lib().dj_timing_total_ms(PHASES[phase])
