"""2-rank CPU dry run of the exact RCCL bootstrap the driver's N>1 bench
executes (bench.py rccl_bootstrap: rank 0 fetches the RCCL unique id, gloo
broadcasts it, every rank hands the same bytes to CppCommunicator — the
reference's MPI_Bcast of ncclGetUniqueId, communicator.cpp:799-817).

Only ncclGetUniqueId itself needs a GPU, so it is stubbed with deterministic
bytes; everything else — the gloo process group, the id-size query through
the real libdistjoin.so, the broadcast, the byte layout handed to the
communicator ctor — is the code path the 8-GPU run will take verbatim.
"""
import os
import sys

import numpy as np
import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from tests.test_distributed_plan import launch  # noqa: E402


class _StubLib:
    """bench.rccl_bootstrap touches exactly these two symbols. The id size
    comes from the real library (ncclUniqueId is 128 bytes in RCCL); the id
    bytes are a deterministic stand-in for the GPU-only ncclGetUniqueId."""

    def __init__(self, nbytes):
        self._n = nbytes

    def dj_rccl_unique_id_bytes(self):
        return self._n

    def dj_rccl_get_unique_id(self, ptr):
        buf = (np.arange(self._n, dtype=np.int64) * 37 + 11).astype(np.uint8)
        import ctypes
        ctypes.memmove(ptr, buf.ctypes.data, self._n)
        return 0


def _bootstrap_rank(rank, world):
    import torch.distributed as dist

    import distributed_join_amd as dj
    from bench import rccl_bootstrap

    nbytes = dj.lib().dj_rccl_unique_id_bytes()  # real .so, host-side query
    assert nbytes > 0
    id_bytes = rccl_bootstrap(dist, rank, world, _StubLib(nbytes))
    assert id_bytes.dtype == np.uint8 and id_bytes.flags["C_CONTIGUOUS"]
    return id_bytes.tobytes()


def test_rccl_bootstrap_broadcasts_rank0_id():
    results = launch(_bootstrap_rank)
    nbytes = None
    import distributed_join_amd as dj
    nbytes = dj.lib().dj_rccl_unique_id_bytes()
    want = bytes((np.arange(nbytes, dtype=np.int64) * 37 + 11).astype(np.uint8))
    assert results[0] == want, "rank 0 must use its own generated id"
    assert results[1] == want, "rank 1 must receive rank 0's id over gloo"


def test_bench_main_bootstrap_wiring():
    """bench.py main() reaches rccl_bootstrap only when world > 1 and passes
    the library handle — pin the wiring by source so a refactor cannot
    silently drop the broadcast before the driver's first 8-GPU run."""
    import inspect

    import bench
    src = inspect.getsource(bench.main)
    assert "rccl_bootstrap(dist, rank, world, L)" in src
    assert "world > 1" in src
