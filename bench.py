#!/usr/bin/env python3
"""Flagship benchmark: TeraSort over the MI355X one-sided shuffle.

Driver contract: `python bench.py --gpus N --steps K --warmup W` (launched
under torch.distributed.run for N>1, one rank per GPU). Rank 0 prints ONE
JSON line; `value` is the whole-job aggregate sorted throughput in GB/s
(BASELINE.json metric: 320 GB TeraSort wall-clock + shuffle-read GB/s —
ms_per_step is the wall-clock of one complete sort job of the configured
dataset; value = dataset_bytes / wall_clock aggregated over all ranks).
"""

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))


def _fetch_rate(results, ts):
    """Mean shuffle-read GB/s over steps with a timed fetch phase (steps
    with fetch_s == 0 are excluded, not averaged in as zeros)."""
    rb = getattr(ts, "RECORD_BYTES", None)
    if rb is None:
        return None
    rates = [(getattr(r, "records", 0) * rb / r.fetch_s) / 1e9
             for r in results if getattr(r, "fetch_s", 0) > 0]
    return round(sum(rates) / len(rates), 2) if rates else None


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=5)
    ap.add_argument("--warmup", type=int, default=2)
    ap.add_argument("--gb-per-gpu", type=float,
                    default=float(os.environ.get("TERASORT_GB_PER_GPU", 40)))
    ap.add_argument("--mode",
                    choices=["framework", "rccl", "shuffleread"],
                    default=os.environ.get("TERASORT_MODE", "framework"))
    ap.add_argument("--workload",
                    choices=["terasort", "pagerank", "join", "groupby",
                             "reducebykey"],
                    default="terasort",
                    help="terasort is the headline metric; others cover "
                         "the remaining BASELINE configs")
    ap.add_argument("--partitions-per-executor", type=int, default=0,
                    help="0 = auto (pow2, ~128 per GPU)")
    ap.add_argument("--record-bytes", type=int,
                    default=int(os.environ.get("TERASORT_RECORD_BYTES", 100)),
                    help="terasort record width: 100 (canonical 10B key + "
                         "90B value — the flagship config; GPU only) or 16 "
                         "(u64 key + u64 payload; the CPU/rccl-mode shape)")
    ap.add_argument("--validate", action="store_true")
    ap.add_argument("--cpu", action="store_true",
                    help="force CPU path (plumbing debug)")
    args = ap.parse_args()

    import torch
    use_cuda = torch.cuda.is_available() and not args.cpu
    rank = int(os.environ.get("RANK", 0))
    world = int(os.environ.get("WORLD_SIZE", args.gpus))
    dist = None
    if world > 1:
        import torch.distributed as tdist
        dist = tdist
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        backend = os.environ.get("BENCH_DIST_BACKEND",
                                 "nccl" if use_cuda else "gloo")
        dist.init_process_group(backend)
        if use_cuda:
            ngpu = torch.cuda.device_count()
            torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", rank)) % ngpu)

    from sparkrdma_amd.conf import ShuffleConf
    from sparkrdma_amd.engine import Engine
    from sparkrdma_amd.workloads.terasort import TeraSort

    rec_bytes = args.record_bytes if args.workload == "terasort" else 16
    if rec_bytes != 16 and not use_cuda:
        rec_bytes = 16   # wide records are GPU-only
    n_rec = int(args.gb_per_gpu * (1 << 30) / rec_bytes)
    ppe = args.partitions_per_executor
    if ppe == 0:
        # keep R = 256 total: 8-bit partition pass AND a 7-pass reduce sort
        ppe = max(32, 256 // world)
    conf = ShuffleConf(transport="ipc" if use_cuda else "shm")
    if use_cuda:
        conf.gpu_id = int(os.environ.get("LOCAL_RANK", rank)) % \
            torch.cuda.device_count()
        # served blocks are pow2-rounded (buddy) -> up to 2x data, + slack;
        # keep the pool tight enough that input + fetched + sort ping-pong
        # still fit 288 GB at 40 GB/GPU
        conf.hbm_pool_size = int((args.gb_per_gpu * 2.0 + 2) * (1 << 30))
        # few, large contiguous blocks: adjacent partitions of one map pack
        # into one 512 MiB block, so coalesced fetches become ~GB-scale
        # xGMI copies instead of per-partition ones
        conf.shuffle_write_block_size = 512 << 20
        conf.shuffle_read_block_size = 512 << 20
        conf.max_bytes_in_flight = 8 << 30
    eng = Engine(conf, rank=rank, world_size=world)
    if rank == 0:
        print(f"[bench] engine up: world={world} transport={conf.transport} "
              f"workload={args.workload} rec_bytes={rec_bytes}",
              file=sys.stderr, flush=True)

    device = "cuda" if use_cuda else "cpu"
    if args.workload == "terasort":
        mode = args.mode
        if mode == "framework" and use_cuda and (
                conf.use_rccl or conf.transport == "rccl"):
            mode = "rccl"   # spark.shuffle.rdma.useRccl / transport=rccl
        if mode == "rccl" and not use_cuda:
            mode = "framework"
        ts = TeraSort(eng, n_rec, partitions_per_executor=ppe,
                      device=device, mode=mode, validate=args.validate,
                      record_bytes=rec_bytes)
    elif args.workload == "pagerank":
        from sparkrdma_amd.workloads.pagerank import PageRank
        # 19 GB edge list analog: 16 B records; vertex count scales with
        # the dataset (pow2) so tiny CPU contract runs stay tiny
        edges = n_rec
        vbits = min(26, max(16, edges.bit_length() - 2))
        ts = PageRank(eng, num_vertices=1 << vbits, edges_per_executor=edges,
                      partitions_per_executor=ppe, device=device,
                      iterations=3)
    elif args.workload == "join":
        from sparkrdma_amd.workloads.sql_join import SortMergeJoin
        ts = SortMergeJoin(eng, rows_per_executor=n_rec // 2,
                           partitions_per_executor=ppe, device=device,
                           key_space_bits=40, validate=args.validate)
    elif args.workload == "reducebykey":
        from sparkrdma_amd.workloads.reduce_by_key import ReduceByKey
        ts = ReduceByKey(eng, rows_per_executor=n_rec,
                         partitions_per_executor=ppe, device=device,
                         validate=args.validate)
    else:
        from sparkrdma_amd.workloads.groupby import GroupByKey
        ts = GroupByKey(eng, rows_per_executor=min(n_rec, 1_000_000))

    def barrier_sync():
        if dist is not None:
            dist.barrier()
        else:
            eng.barrier()
        if use_cuda:
            torch.cuda.synchronize()

    debug = os.environ.get("BENCH_DEBUG")
    for _ in range(args.warmup):
        r = ts.run_step()
        if debug and rank == 0 and hasattr(r, "write_s"):
            print(f"[warmup] total={r.seconds*1e3:.1f}ms write={r.write_s*1e3:.1f} "
                  f"fetch={r.fetch_s*1e3:.1f} sort={r.sort_s*1e3:.1f}",
                  file=sys.stderr)
    barrier_sync()
    if rank == 0:
        print(f"[bench] warmup done ({args.warmup}); timing {args.steps} "
              "steps", file=sys.stderr, flush=True)
    t0 = time.perf_counter()
    results = [ts.run_step() for _ in range(args.steps)]
    barrier_sync()
    elapsed = time.perf_counter() - t0

    # MAX over ranks
    if dist is not None:
        t = torch.tensor([elapsed])
        if use_cuda:
            t = t.cuda()
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    ms_per_step = elapsed / args.steps * 1e3
    if args.workload == "terasort" and args.mode == "shuffleread":
        per_step_bytes = n_rec * ts.RECORD_BYTES * world
        metric = "shuffle_read_gb_per_s"
    elif args.workload == "terasort":
        per_step_bytes = n_rec * ts.RECORD_BYTES * world
        metric = "terasort_sorted_gb_per_s"
    elif args.workload == "pagerank":
        per_step_bytes = results[0].iterations * n_rec * 16 * world
        metric = "pagerank_edge_gb_per_s"
    elif args.workload == "join":
        per_step_bytes = n_rec * 16 * world  # both tables
        metric = "join_row_gb_per_s"
    elif args.workload == "reducebykey":
        per_step_bytes = n_rec * 16 * world
        metric = "reducebykey_gb_per_s"
    else:
        per_step_bytes = results[0].rows * world * 64
        metric = "groupby_gb_per_s"
    total_bytes = per_step_bytes
    value = total_bytes / (elapsed / args.steps) / 1e9  # GB/s whole job
    remote_gb = sum(getattr(r, "remote_bytes", getattr(r, "shuffle_bytes", 0))
                    for r in results) / 1e9

    if rank == 0:
        print(json.dumps({
            "metric": metric,
            "value": round(value, 3),
            "unit": "GB/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 2),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": ("10Bkey+90Bval" if rec_bytes == 100 else
                      f"{rec_bytes}B-records" if rec_bytes != 16 else
                      "u64key+u64payload"),
            "data": "synthetic",
            "config": {
                "model": args.workload,
                "global_batch": n_rec * world,
                "seq_len": getattr(ts, "RECORD_BYTES", 16),
                "dataset_gb": round(total_bytes / (1 << 30), 1),
                "partitions": world * ppe,
                "mode": getattr(ts, "mode", args.mode),
                "parallelism": f"shuffle{world}",
                "wall_clock_s_per_job": round(elapsed / args.steps, 3),
                "remote_gb_per_step": round(remote_gb / max(1, args.steps), 2),
                # BASELINE metric names shuffle-read GB/s explicitly;
                # mean over steps that actually timed a fetch phase
                "shuffle_read_gb_per_s": _fetch_rate(results, ts),
                # honesty label: at N=1 every one-sided read is a local
                # D2D copy; xGMI rates only appear at N>1
                "read_locality": (
                    ("local_d2d" if use_cuda else "local_shm")
                    if world == 1 or remote_gb == 0
                    else ("xgmi_remote" if use_cuda else "shm_remote")),
            },
        }))
    eng.shutdown()
    if dist is not None:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
