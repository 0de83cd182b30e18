#!/usr/bin/env python3
"""all_to_all_bench.py — all-to-all size-sweep microbenchmark.

Replicates the reference's benchmark/all_to_all.cpp:41-141: for total buffer
sizes 1 MB .. 4.096 GB (doubling), each rank sends size/world bytes to every
peer through the Communicator's grouped peer-slice exchange (here: RCCL
ncclSend/ncclRecv over xGMI, dj_all_to_all_i64), and prints per-GPU egress
GB/s = (size/world) x (world-1) x repeat / elapsed — the number to set
against the >=70% xGMI bisection target (7 links x ~153 GB/s per GPU).

Launch (one rank per GPU):
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 benchmark/all_to_all_bench.py
Single-rank (self-slice through real ncclSend/Recv at world 1) also works:
  python benchmark/all_to_all_bench.py
Prints one JSON line per size from rank 0.
"""
import argparse
import json
import os
import sys
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)

import numpy as np  # noqa: E402

import distributed_join_amd as dj  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--min-mb", type=int, default=1)
    ap.add_argument("--max-mb", type=int, default=4096)
    ap.add_argument("--warmup", type=int, default=2)
    ap.add_argument("--repeat", type=int, default=10)
    args = ap.parse_args()

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))

    dist = None
    if world > 1:
        import torch.distributed as tdist
        dist = tdist
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        tdist.init_process_group("gloo", rank=rank, world_size=world)

    dj.require_gpu()
    L = dj.lib()
    L.dj_set_device(local_rank % max(L.dj_device_count(), 1))

    from bench import rccl_bootstrap
    if world > 1:
        id_bytes = rccl_bootstrap(dist, rank, world, L)
    else:
        id_bytes = np.zeros(L.dj_rccl_unique_id_bytes(), dtype=np.uint8)
        L.dj_rccl_get_unique_id(id_bytes.ctypes.data)
    # dj_all_to_all_i64 drives the raw RCCL comm (dj_comm_init), the same
    # grouped ncclSend/Recv the Communicator path issues per peer
    L.dj_comm_init(rank, world, id_bytes.ctypes.data)

    def barrier():
        L.dj_sync()
        if dist is not None:
            dist.barrier()

    size = args.min_mb * (1 << 20)
    results = []
    while size <= args.max_mb * (1 << 20):
        elems = size // 8
        per_peer = elems // world
        if per_peer == 0:
            size *= 2
            continue
        send = dj.DeviceArray(per_peer * world)
        recv = dj.DeviceArray(per_peer * world)
        offs = np.arange(world + 1, dtype=np.int64) * per_peer
        for _ in range(args.warmup):
            L.dj_all_to_all_i64(send.ptr, offs.ctypes.data, recv.ptr, offs.ctypes.data)
        barrier()
        t0 = time.perf_counter()
        for _ in range(args.repeat):
            L.dj_all_to_all_i64(send.ptr, offs.ctypes.data, recv.ptr, offs.ctypes.data)
        barrier()
        t1 = time.perf_counter()
        elapsed = t1 - t0
        if dist is not None:
            import torch
            e = torch.tensor([elapsed], dtype=torch.float64)
            dist.all_reduce(e, op=dist.ReduceOp.MAX)
            elapsed = float(e.item())
        egress = per_peer * 8 * (world - 1) * args.repeat
        gbs = egress / elapsed / 1e9 if world > 1 else None
        selfcopy_gbs = per_peer * 8 * args.repeat / elapsed / 1e9
        rec = {
            "buffer_mb": size >> 20,
            "world": world,
            "repeat": args.repeat,
            "per_gpu_egress_GBs": gbs,
            "self_copy_GBs": selfcopy_gbs if world == 1 else None,
            "elapsed_s": elapsed,
            "xgmi_link_peak_GBs": 153.0,
            "per_gpu_egress_peak_GBs": 153.0 * 7,
            "frac_of_bisection": (gbs / (153.0 * 7)) if gbs else None,
        }
        results.append(rec)
        if rank == 0:
            print(json.dumps(rec), flush=True)
        send.free()
        recv.free()
        size *= 2

    L.dj_comm_finalize()
    if dist is not None:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
