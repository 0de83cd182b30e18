#!/usr/bin/env python3
"""bench.py — measurement harness for the MI355X-native distributed
repartitioned hash join.

Metric (BASELINE.json): joined rows/sec whole-node for the 800M x 800M
int64/int64 join at selectivity 0.3 (100M rows per GPU per table — the
reference benchmark's per-GPU default, benchmark/distributed_join.cu:96-109).
`value` = total input rows of both tables across all ranks / join wall time
(the README-comparable number: 0.392 s on 8xV100 => ~4.08e9 rows/s,
README.md:73-86).

One step = one full `distributed_inner_join` call through the C++ drop-in
path (include/distributed_join.hpp -> stable hash partition + RCCL all-to-all
over xGMI + bucketed-LDS local join + concat), via its C ABI. Generation and
warmup are excluded; timing is barrier-bracketed, max over ranks — mirroring
the reference's timed region (benchmark/distributed_join.cu:264-286).

Single process (N=1) by default; for N>1 the driver launches this under
torch.distributed.run with one rank per GPU; torch.distributed (gloo) is
used only for bootstrap (RCCL unique id exchange) and barriers — compute and
communication run in libdistjoin.so (HIP + RCCL over xGMI).
"""
import argparse
import json
import os
import sys
import time

REPO = os.path.dirname(os.path.abspath(__file__))
sys.path.insert(0, REPO)

import numpy as np  # noqa: E402

import distributed_join_amd as dj  # noqa: E402

SELECTIVITY = 0.3


def log(rank, *a):
    if rank == 0:
        print(*a, file=sys.stderr, flush=True)


def rccl_bootstrap(dist, rank, world, L):
    """Exchange the RCCL unique id (rank 0 -> all ranks) over the already-
    initialized gloo process group — the reference's MPI_Bcast of
    ncclGetUniqueId (communicator.cpp:799-817). Returns the id bytes every
    rank passes to CppCommunicator. Covered CPU-side (2-rank gloo, stubbed
    ncclGetUniqueId) by tests/test_bootstrap_dryrun.py; the driver's N>1
    bench executes exactly this function."""
    import torch
    nbytes = L.dj_rccl_unique_id_bytes()
    if rank == 0:
        buf = np.zeros(nbytes, dtype=np.uint8)
        L.dj_rccl_get_unique_id(buf.ctypes.data)
        t = torch.from_numpy(buf)
    else:
        t = torch.zeros(nbytes, dtype=torch.uint8)
    dist.broadcast(t, src=0)
    return np.ascontiguousarray(t.numpy())


def cpu_baseline_leg(sample_rows):
    """Oracle OpenMP radix-partition join (the 'port' CPU baseline) on a
    bounded sample of the same workload, timed on this box's host cores.
    Test-infrastructure import — allowed here per oracle/oracle.c header."""
    import oracle
    n = sample_rows
    bk, bp = oracle.gen_build(n)
    pk, pp = oracle.gen_probe(n, n, selectivity=SELECTIVITY)
    oracle.cpu_radix_join(bk[:100_000], bp[:100_000], pk[:100_000], pp[:100_000],
                          count_only=True)
    t0 = time.perf_counter()
    nout = oracle.cpu_radix_join(bk, bp, pk, pp, count_only=True)
    t1 = time.perf_counter()
    secs = t1 - t0
    return {
        "value": (2.0 * n) / secs,
        "unit": "input rows/s",
        "cores": oracle.num_threads(),
        "kind": "port",
        "sample": f"{n}x{n} int64/int64 sel {SELECTIVITY} radix-partition join, "
                  f"{secs:.2f}s wall ({nout} output rows)",
    }


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=20)
    ap.add_argument("--warmup", type=int, default=5)
    ap.add_argument("--rows", type=int, default=100_000_000,
                    help="rows per GPU per table (build and probe)")
    ap.add_argument("--over-decom", type=int, default=1)
    ap.add_argument("--nvlink-domain-size", type=int, default=0,
                    help="0 = world size (one xGMI node is ONE domain -> batched "
                         "all-to-all pipeline / fused wire path). The reference's "
                         "README benchmark ran its default of 1 (IB shuffle + local "
                         "join, distributed_join.cpp:152-214); pass 1 to reproduce "
                         "that configuration.")
    ap.add_argument("--compression", action="store_true",
                    help="cascaded (delta+bitpack) compression on the wire — config 4's "
                         "knob; rarely pays on intra-node xGMI")
    ap.add_argument("--cpu-baseline-rows", type=int, default=100_000_000)
    ap.add_argument("--no-cpu-baseline", action="store_true")
    args = ap.parse_args()

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    if world == 1 and args.gpus > 1:
        print("ERROR: for --gpus N>1 launch under torch.distributed.run", file=sys.stderr)
        sys.exit(2)
    N = world

    dist = None
    if world > 1:
        import torch.distributed as tdist
        dist = tdist
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        tdist.init_process_group("gloo", rank=rank, world_size=world)

    dj.require_gpu()
    L = dj.lib()
    L.dj_set_device(local_rank % max(L.dj_device_count(), 1))

    # RCCL bootstrap: rank 0's unique id broadcast over gloo
    id_bytes = rccl_bootstrap(dist, rank, world, L) if world > 1 else None
    comm = dj.CppCommunicator(rank, world, id_bytes)

    rows = args.rows
    build_n_global = rows * N
    probe_n_global = rows * N
    rand_max = 2 * build_n_global

    log(rank, f"[bench] N={N} rows/GPU={rows} generating inputs...")
    row0 = rank * rows
    bk, bp = dj.generate_build(build_n_global, rand_max, uniq=True, row0=row0, nrows=rows)
    pk, pp = dj.generate_probe(probe_n_global, build_n_global, rand_max,
                               selectivity=SELECTIVITY, row0=row0, nrows=rows)

    state = {"matches": 0}

    nvl = args.nvlink_domain_size if args.nvlink_domain_size > 0 else N

    def step(nvl_mode=None):
        t = L.dj_cpp_distributed_inner_join_i64_full(comm.ptr, bk.ptr, bp.ptr, rows,
                                                     pk.ptr, pp.ptr, rows, args.over_decom,
                                                     0, int(args.compression),
                                                     nvl if nvl_mode is None else nvl_mode)
        state["matches"] = L.dj_table_num_rows(t)
        L.dj_table_free(t)

    def barrier_sync():
        L.dj_sync()
        if dist is not None:
            dist.barrier()

    log(rank, f"[bench] warmup {args.warmup} steps...")
    for _ in range(args.warmup):
        step()

    L.dj_timing_enable(1)
    L.dj_timing_reset()
    barrier_sync()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        step()
    barrier_sync()
    t1 = time.perf_counter()
    elapsed = t1 - t0
    if dist is not None:
        import torch
        e = torch.tensor([elapsed], dtype=torch.float64)
        dist.all_reduce(e, op=dist.ReduceOp.MAX)
        elapsed = float(e.item())

    matches = state["matches"]
    ms_per_step = elapsed / args.steps * 1000.0
    input_rows = 2.0 * rows * N  # both tables, whole node
    value = input_rows / (elapsed / args.steps)
    out_rows_s = matches * N / (elapsed / args.steps)

    # per-phase kernel timing (hipEvents inside the C++ orchestration)
    phases = {name: {"ms": L.dj_timing_total_ms(pid) / args.steps,
                     "launches": L.dj_timing_launches(pid) / args.steps}
              for name, pid in dj.PHASES.items()}
    L.dj_timing_enable(0)  # the alt-mode leg below must not pollute phases

    # At N>1 the default mode (nvlink_domain_size = world: batched all-to-all
    # pipeline, the MI355X-native result) differs from the configuration the
    # reference's README numbers ran (its default nvlink_domain_size = 1:
    # cross-domain shuffle + local join, distributed_join.cpp:152-214). So the
    # ratio against BASELINE.md compares like-for-like, time BOTH modes and
    # attach the alternate one to the record.
    alt = None
    if N > 1 and args.nvlink_domain_size == 0:
        alt_nvl = 1
        for _ in range(min(2, args.warmup)):
            step(alt_nvl)
        barrier_sync()
        a0 = time.perf_counter()
        for _ in range(args.steps):
            step(alt_nvl)
        barrier_sync()
        a1 = time.perf_counter()
        alt_elapsed = a1 - a0
        if dist is not None:
            import torch
            e = torch.tensor([alt_elapsed], dtype=torch.float64)
            dist.all_reduce(e, op=dist.ReduceOp.MAX)
            alt_elapsed = float(e.item())
        alt = {
            "nvlink_domain_size": alt_nvl,
            "mode": "reference-default: cross-domain shuffle + local join",
            "ms_per_step": alt_elapsed / args.steps * 1000.0,
            "value": input_rows / (alt_elapsed / args.steps),
        }

    # roofline for the dominant join kernel (algorithmic bytes per step for
    # this rank; derivation in DESIGN.md §Measurement).
    # N==1 (reference single-rank semantics: local join only, no
    # partition/shuffle stage — distributed_join.cpp:200-214):
    #   bucket_scatter = the two-level local bucket partition: slack pass A
    #     (16 B read + 16 B write, no count pass) + count-free slack pass B
    #     (16 B read + 16 B write at analytic per-bucket starts) = 64 B per
    #     row per table
    # N>1 (fused wire path):
    #   part_scatter   = the fused rank+group partition (8 B count-read +
    #                    16 B read + 16 B write per row, both tables)
    #   bucket_scatter = pass B over per-peer segment lists (8 B count +
    #                    16 B read + 16 B pair write per received row)
    #   join_fused     = streaming read of both bucketed tables (16 B/row) +
    #                    32 B per output row; LDS traffic is on-chip
    lrows = rrows = float(rows)  # hash-uniform => received ~= sent rows
    if N == 1:
        alg = {
            "part_scatter": 1.0,  # not run at N==1
            "bucket_scatter": 64.0 * (lrows + rrows),
            "join_fused": 16.0 * (lrows + rrows) + 32.0 * matches,
        }
    else:
        alg = {
            "part_scatter": 40.0 * rows * 2,
            "bucket_scatter": 40.0 * (lrows + rrows),
            "join_fused": 16.0 * (lrows + rrows) + 32.0 * matches,
        }
    dom = max(alg.keys(), key=lambda k: phases[k]["ms"])
    dom_ms = phases[dom]["ms"]
    achieved = (alg[dom] / 1e9) / (dom_ms / 1e3) if dom_ms > 0 else None
    # measured HBM traffic per step: PMC calibration committed under
    # profiles/r02_pmc_traffic.json (rocprofv3 --pmc FETCH_SIZE / WRITE_SIZE
    # passes; FETCH x2 gfx950 correction — see that file)
    traffic = None
    if N == 1:  # coefficients were calibrated on the N=1 kernel set only
        try:
            with open(os.path.join(REPO, "profiles", "r02_pmc_traffic.json")) as fh:
                coeff = json.load(fh)["phase_bytes_per_input_row_pair"]
            traffic = coeff[dom] * (lrows + rrows)
        except Exception:
            pass
    roofline = {
        "bound": "hbm",
        "kernel": dom,
        "achieved": achieved,
        "peak": 8000.0,
        "unit": "GB/s",
        "frac": (achieved / 8000.0) if achieved else None,
        "traffic": traffic,
    }

    all_to_all_GBs = None
    if N > 1 and phases["comm"]["ms"] > 0:
        # hash-uniform egress estimate: both tables, (G-1)/G of rows, 16 B/row
        est_bytes = 2.0 * rows * 16.0 * (N - 1) / N
        all_to_all_GBs = (est_bytes / 1e9) / (phases["comm"]["ms"] / 1e3)

    cpu_baseline = None
    if rank == 0 and N == 1 and not args.no_cpu_baseline:
        log(rank, "[bench] cpu baseline leg...")
        cpu_baseline = cpu_baseline_leg(args.cpu_baseline_rows)

    if rank == 0:
        rec = {
            "metric": "joined rows/sec whole-node",
            "value": value,
            "unit": "input rows/s",
            "n_gpus": N,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": "weak",
            # README.md:73-86: 0.392 s for 1.6e9 input rows on 8xV100 => 4.08e9.
            # The reference ran its default nvlink_domain_size=1, so the ratio
            # uses the like-for-like leg: the alt-mode (nvl=1) measurement when
            # the primary ran the pipeline mode, else the primary itself.
            "vs_baseline": (((alt["value"] if alt else value) / 4.08e9)
                            if N == 8 else None),
            "dtype": "int64",
            "data": "synthetic",
            "config": {
                "workload": f"{rows*N//1_000_000}Mx{rows*N//1_000_000}M int64/int64 join, "
                            f"selectivity {SELECTIVITY}, unique build keys, "
                            f"{rows//1_000_000}M rows/GPU/table, over_decom {args.over_decom}",
                "rows_per_gpu": rows,
                "selectivity": SELECTIVITY,
                "nvlink_domain_size": nvl,
                "compression": bool(args.compression),
                "output_rows_per_gpu": int(matches),
                "output_rows_per_sec": out_rows_s,
                "engine": "C++ distributed_inner_join (drop-in path); at N=1 the "
                          "reference's single-rank semantics apply (local join only, "
                          "no partition/shuffle stage)",
                "phases_ms": {k: round(v["ms"], 4) for k, v in phases.items()},
                "all_to_all_GBs_per_gpu": all_to_all_GBs,
                "alt_mode": alt,
            },
            "roofline": roofline,
            "cpu_baseline": cpu_baseline,
        }
        print(json.dumps(rec), flush=True)

    comm.destroy()
    if world > 1:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
