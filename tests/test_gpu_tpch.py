"""GPU test of the TPC-H-shaped join (config 5 harness) at small scale:
orders (int64 key + priority string) JOIN lineitem (int64 key + payload),
checked against the oracle on keys and the priority formula on strings.
"""
import os
import sys

import numpy as np
import pytest

import oracle

sys.path.insert(0, os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
                                "benchmark"))

pytestmark = pytest.mark.gpu


def test_tpch_shape_join():
    import distributed_join_amd as dj
    from tpch_bench import PRIORITIES, synth_lineitem, synth_orders
    dj.require_gpu()
    L = dj.lib()
    comm = dj.CppCommunicator(0, 1)

    n_o, n_l = 50_000, 200_000
    okeys, ooff, ochars = synth_orders(n_o, 1234, 0, n_o)
    lkeys, lpay = synth_lineitem(n_o, 1234, 0, n_l)

    d_ok = dj.DeviceArray.from_numpy(okeys)
    d_ooff = L.dj_dmalloc(len(ooff) * 4)
    L.dj_memcpy_h2d(d_ooff, ooff.ctypes.data, len(ooff) * 4)
    d_och = L.dj_dmalloc(len(ochars))
    L.dj_memcpy_h2d(d_och, ochars.ctypes.data, len(ochars))
    d_lk = dj.DeviceArray.from_numpy(lkeys)
    d_lp = dj.DeviceArray.from_numpy(lpay)

    cols = dj.cpp_distributed_inner_join_cols(
        comm,
        [(dj.TYPE_INT64, d_ok.ptr), (dj.TYPE_STRING, d_ooff, d_och, len(ochars))],
        n_o,
        [(dj.TYPE_INT64, d_lk.ptr), (dj.TYPE_INT64, d_lp.ptr)],
        n_l)
    c0, c1, c2, c3 = cols
    n = len(c0)

    # oracle on keys: every lineitem key exists in orders => n == n_l
    e0, _, _, e3 = oracle.inner_join(okeys, np.arange(n_o, dtype=np.int64), lkeys, lpay)
    assert n == len(e0) == n_l
    assert sorted(c0.tolist()) == sorted(e0.tolist())
    assert sorted(c3.tolist()) == sorted(e3.tolist())

    # priority strings must match the synthetic formula for their key
    off, ch = c1
    for i in range(0, n, 1031):
        k = int(c0[i])
        want = PRIORITIES[int(abs(k * 2654435761) % 5)]
        assert ch[off[i]:off[i + 1]].tobytes() == want
    comm.destroy()
