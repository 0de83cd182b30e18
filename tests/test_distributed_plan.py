"""Multi-process (gloo, CPU) tests of the distributed-join PLAN: the same
partition -> size-exchange -> all-to-all -> local-join -> collect dataflow
bench.py and the C++ orchestration run on RCCL, executed here with the oracle
kernels and torch.distributed gloo so the logic is covered without a GPU.

Validates against the reference semantics: the distributed join result
(concatenated over ranks) equals the global join of the concatenated inputs
(reference pin: test/compare_against_single_gpu.cu:96-207, with the CPU
oracle standing in for single-GPU cudf).
"""
import os

import numpy as np
import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

import oracle

WORLD = 2


def _run_rank(rank, world, fn, port, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        res = fn(rank, world)
        q.put((rank, "ok", res))
    except Exception as e:  # pragma: no cover
        import traceback
        q.put((rank, "err", traceback.format_exc()))
    finally:
        dist.destroy_process_group()


def launch(fn, world=WORLD):
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = 29000 + (os.getpid() % 1000)
    ps = [ctx.Process(target=_run_rank, args=(r, world, fn, port, q)) for r in range(world)]
    for p in ps:
        p.start()
    results = {}
    for _ in range(world):
        rank, status, res = q.get(timeout=180)
        assert status == "ok", f"rank {rank} failed:\n{res}"
        results[rank] = res
    for p in ps:
        p.join(timeout=60)
    return results


def _exchange_sizes(send_counts):
    """The communicate_sizes step (all_to_all_comm.cpp:54-100) over gloo."""
    world = dist.get_world_size()
    t = torch.from_numpy(send_counts.astype(np.int64))
    gathered = [torch.zeros_like(t) for _ in range(world)]
    dist.all_gather(gathered, t)
    rank = dist.get_rank()
    return np.array([int(g[rank]) for g in gathered], dtype=np.int64)


def _all_to_all(arr, send_offsets, recv_counts):
    """Personalized all-to-all of one int64 column buffer (the
    send/recv_data_by_offset pattern, all_to_all_comm.cpp:126-189)."""
    world = dist.get_world_size()
    rank = dist.get_rank()
    send = [torch.from_numpy(arr[send_offsets[p]:send_offsets[p + 1]].copy())
            for p in range(world)]
    recv = [torch.zeros(int(recv_counts[p]), dtype=torch.int64) for p in range(world)]
    # gloo has no alltoall: grouped isend/irecv per peer, self via direct copy —
    # the same per-peer pattern the RCCL communicator uses
    reqs = []
    for p in range(world):
        if p == rank:
            recv[p] = send[p]
            continue
        if len(send[p]):
            reqs.append(dist.isend(send[p], p))
        if len(recv[p]):
            reqs.append(dist.irecv(recv[p], p))
    for r in reqs:
        r.wait()
    return np.concatenate([r.numpy() for r in recv])


def _distributed_join_plan(rank, world):
    # local slices of the global 200k x 200k tables (ragged for odd worlds:
    # the first n_global % world ranks take one extra row)
    n_global = 200_000
    base, rem = divmod(n_global, world)
    rows = base + (1 if rank < rem else 0)
    row0 = rank * base + min(rank, rem)
    bk, bp = oracle.gen_build(n_global, row0=row0, nrows=rows)
    pk, pp = oracle.gen_probe(n_global, n_global, selectivity=0.3, row0=row0, nrows=rows)

    # rank-level stable partition, seed 12345678 (distributed_join.cpp:211-226)
    pbk, pbp, boff = oracle.partition(bk, bp, world, oracle.HASH_MURMUR3, 12345678)
    ppk, ppp, poff = oracle.partition(pk, pp, world, oracle.HASH_MURMUR3, 12345678)

    rb = _exchange_sizes(np.diff(boff))
    rp_counts = _exchange_sizes(np.diff(poff))

    lbk = _all_to_all(pbk, boff, rb)
    lbp = _all_to_all(pbp, boff, rb)
    lpk = _all_to_all(ppk, poff, rp_counts)
    lpp = _all_to_all(ppp, poff, rp_counts)

    # local join per rank
    c0, c1, c2, c3 = oracle.inner_join(lbk, lbp, lpk, lpp)
    return np.stack([c0, c1, c2, c3])


@pytest.mark.timeout(300)
@pytest.mark.parametrize("world", [2, 3])
def test_distributed_join_plan_matches_global_join(world):
    # world 3 covers the non-power-of-two paths: murmur % 3 placement and
    # ragged rank slices (the reference's tests run arbitrary mpi_size)
    results = launch(_distributed_join_plan, world=world)
    got = np.concatenate([results[r] for r in sorted(results)], axis=1)
    got = np.stack(oracle.sort_rows(*[got[i] for i in range(4)]))

    n_global = 200_000
    bk, bp = oracle.gen_build(n_global)
    pk, pp = oracle.gen_probe(n_global, n_global, selectivity=0.3)
    want = np.stack(oracle.sort_rows(*oracle.inner_join(bk, bp, pk, pp)))
    assert got.shape == want.shape
    assert (got == want).all()


def _shuffle_plan(rank, world):
    # shuffle_on with identity hash: key k lands on rank k % world
    # (reference pin: test/test_shuffle_on.cpp:78-83)
    n = 10_000
    rng = np.random.RandomState(100 + rank)
    keys = rng.randint(0, 2**31, n).astype(np.int64)
    pays = np.arange(n, dtype=np.int64) + rank * n
    ok, op, off = oracle.partition(keys, pays, world, oracle.HASH_IDENTITY, 0)
    rc = _exchange_sizes(np.diff(off))
    rk = _all_to_all(ok, off, rc)
    return rk


@pytest.mark.timeout(300)
def test_shuffle_identity_placement():
    results = launch(_shuffle_plan)
    for rank, keys in results.items():
        assert (keys % WORLD == rank).all()
