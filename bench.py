#!/usr/bin/env python3
"""Flagship benchmark: NDS (TPC-DS-derived) power run on N MI355X GPUs.

Driver contract: `python bench.py --gpus N --steps K --warmup W` (launched
under torch.distributed.run for N>1, one rank per GPU over RCCL).

What one step is: the full 11-query NDS-shaped power run (bench/nds_queries
.py — string-keyed dimension joins, decimal aggregation, rollup, windows,
top-N sorts) executed end to end, every query SCANNING ITS INPUT FROM
PARQUET ON DISK through the engine's GPU decode reader. Data is a
deterministic TPC-DS-shaped star schema (bench/nds.py) staged to local disk
before the timed region; the host file cache stays at its default (off) so
every step re-reads and re-decodes.

BASELINE.json metric: NDS power-run wall-clock + speedup vs CPU Spark. The
CPU baseline here is this engine's own CPU backend on multiple worker
processes over the same on-disk data (there is no JVM Spark in the image);
vs_baseline divides that measured speedup by the reference's default 3.0x
operator score — the reference repo publishes no absolute
wall-clock numbers in-tree.

Timing: barrier + torch.cuda.synchronize on both sides, MAX over ranks.
"""
from __future__ import annotations

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

import torch

from spark_rapids_amd import Session
from spark_rapids_amd.bench import nds
from spark_rapids_amd.bench.nds_queries import POWER_RUN, run_power


def _dist_env():
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    return rank, world, local_rank


def _default_data_dir(rows: int, world: int) -> str:
    base = os.environ.get("TMPDIR", "/tmp")
    return os.path.join(base, f"nds_data_{rows}x{world}")


def _open_tables(session: Session, paths):
    tables = {"store_sales": session.read_parquet(paths["store_sales"])}
    for dim in ("date_dim", "item", "store", "customer"):
        tables[dim] = session.read_parquet(paths[dim], replicated=True)
    return tables


def _cpu_worker(data_dir: str, my_fact_files, steps: int) -> float:
    session = Session({"spark.rapids.sql.enabled": False})
    paths = {name: os.path.join(data_dir, name)
             for name in ("date_dim", "item", "store", "customer")}
    tables = {"store_sales": session.read_parquet(list(my_fact_files))}
    for dim, p in paths.items():
        tables[dim] = session.read_parquet(p, replicated=True)
    t0 = time.perf_counter()
    for _ in range(steps):
        run_power(tables)
    return time.perf_counter() - t0


def _cpu_baseline(data_dir: str, procs: int, steps: int) -> float:
    """Power run on `procs` CPU worker processes, each scanning a disjoint
    shard of the on-disk fact files (multi-core CPU Spark analogue).
    Returns wall-clock seconds per step (max over workers)."""
    import concurrent.futures as cf
    import glob as g

    fact_files = sorted(
        g.glob(os.path.join(data_dir, "store_sales", "*.parquet")))
    t0 = time.perf_counter()
    with cf.ProcessPoolExecutor(procs) as pool:
        futs = [pool.submit(_cpu_worker, data_dir,
                            fact_files[i::procs] or fact_files[:1], steps)
                for i in range(procs)]
        [f.result() for f in futs]
    return (time.perf_counter() - t0) / steps


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=3)
    ap.add_argument("--warmup", type=int, default=1)
    ap.add_argument("--rows-per-gpu", type=int, default=60_000_000)
    ap.add_argument("--partitions", type=int, default=8,
                    help="fact parquet files per rank")
    ap.add_argument("--data-dir", default=None)
    ap.add_argument("--io-threads", type=int, default=None,
                    help="override spark.rapids.sql.format.parquet."
                         "multiThreadedRead.numThreads")
    ap.add_argument("--queries", default=None,
                    help="comma list to restrict the suite (debug)")
    ap.add_argument("--per-query", action="store_true",
                    help="print per-query wall times to stderr (untimed)")
    ap.add_argument("--cpu-baseline-steps", type=int, default=1,
                    help="0 disables the CPU-backend baseline measurement")
    ap.add_argument("--cpu-baseline-procs", type=int,
                    default=min(16, os.cpu_count() or 1))
    args = ap.parse_args()

    rank, world, local_rank = _dist_env()
    n_gpus = max(world, 1)
    distributed = world > 1

    use_gpu = torch.cuda.is_available()
    if use_gpu:
        torch.cuda.set_device(local_rank)
    if distributed:
        torch.distributed.init_process_group(
            backend="nccl" if use_gpu else "gloo")

    data_dir = args.data_dir or _default_data_dir(args.rows_per_gpu, world)
    os.makedirs(data_dir, exist_ok=True)
    stage_t0 = time.perf_counter()
    paths = nds.stage(data_dir, args.rows_per_gpu, rank, world,
                      partitions=args.partitions)
    if distributed:
        torch.distributed.barrier()
    stage_s = time.perf_counter() - stage_t0

    conf = {"spark.rapids.sql.enabled": use_gpu}
    if args.io_threads:
        conf["spark.rapids.sql.format.parquet.multiThreadedRead"
             ".numThreads"] = args.io_threads
    session = Session(conf)
    tables = _open_tables(session, paths)
    queries = args.queries.split(",") if args.queries else None

    def barrier_sync():
        if distributed:
            torch.distributed.barrier()
        if use_gpu:
            torch.cuda.synchronize()

    # ---- warmup ----
    for _ in range(args.warmup):
        run_power(tables, queries)
    barrier_sync()

    if args.per_query and rank == 0:
        for qname in (queries or [q for q, _ in POWER_RUN]):
            if use_gpu:
                torch.cuda.synchronize()
            tq = time.perf_counter()
            run_power(tables, [qname])
            if use_gpu:
                torch.cuda.synchronize()
            print(f"[per-query] {qname}: "
                  f"{(time.perf_counter() - tq) * 1000:.1f} ms",
                  file=sys.stderr)
        # per-exec wall breakdown of one representative query
        fn = dict(POWER_RUN)["q3"]
        dfq = fn(tables)
        dfq.collect()
        for m in dfq.metrics():
            print(f"[q3-metrics] {m}", file=sys.stderr)
        barrier_sync()

    # ---- timed region ----
    # GC discipline: freeze the (large) startup object graph and disable
    # collection for the K timed steps — the periodic gen-2 collection
    # otherwise lands an ~80ms spike on one random query per step (the
    # JVM analogue is Spark's tuned GC flags)
    import gc

    gc.collect()
    gc.freeze()
    gc.disable()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        run_power(tables, queries)
    barrier_sync()
    elapsed = time.perf_counter() - t0
    gc.enable()

    # MAX over ranks
    if distributed:
        t = torch.tensor([elapsed], dtype=torch.float64,
                         device="cuda" if use_gpu else "cpu")
        torch.distributed.all_reduce(t, op=torch.distributed.ReduceOp.MAX)
        elapsed = float(t.item())

    n_queries = len(queries) if queries else len(POWER_RUN)
    ms_per_step = elapsed / args.steps * 1000.0
    # whole-job aggregate: fact rows scanned per second across all queries
    rows_per_step = args.rows_per_gpu * n_queries * n_gpus
    value = rows_per_step * args.steps / elapsed

    speedup = None
    if rank == 0 and args.cpu_baseline_steps > 0:
        tcpu = _cpu_baseline(data_dir, args.cpu_baseline_procs,
                             args.cpu_baseline_steps)
        speedup = tcpu / (elapsed / args.steps)

    if rank == 0:
        result = {
            "metric": "NDS power-run fact rows/s from on-disk parquet "
                      "(speedup vs CPU backend in config)",
            "value": value,
            "unit": "rows/s",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": (speedup / 3.0) if speedup is not None else None,
            "dtype": "decimal(7,2)+fp64+string-keys",
            "data": "synthetic TPC-DS-shaped star schema staged to parquet "
                    "on local disk; every step re-scans from disk",
            "config": {
                "model": "nds_power_run_11q",
                "global_batch": args.rows_per_gpu * n_gpus,
                "seq_len": None,
                "parallelism": f"dp{n_gpus}",
                "rows_per_gpu": args.rows_per_gpu,
                "sf_equivalent_per_gpu":
                    round(nds.sf_equivalent(args.rows_per_gpu), 1),
                "queries": [q for q, _ in POWER_RUN] if not queries
                           else queries,
                "dims": "item(102k, string brand/category/id) "
                        "store(1k, string state/name) "
                        "customer(1M, distinct string ids) date_dim(1826)",
                "on_disk_input": data_dir,
                "staging_s_untimed": round(stage_s, 1),
                "scan_stats": __import__(
                    "spark_rapids_amd.io.parquet", fromlist=["SCAN_STATS"]
                ).SCAN_STATS,
                "scan_phases": {
                    k: round(v, 3) for k, v in __import__(
                        "spark_rapids_amd.io.parquet",
                        fromlist=["PHASE_STATS"]).PHASE_STATS.items()},
                "speedup_vs_cpu_backend": speedup,
                "cpu_baseline_procs": args.cpu_baseline_procs,
                "baseline_definition": "vs_baseline = speedup / 3.0: "
                "speedup = wall-clock of this engine's CPU backend on "
                f"{args.cpu_baseline_procs} worker processes scanning the "
                "same on-disk parquet / GPU wall-clock; 3.0x is the "
                "reference's default operator speedup score (BASELINE.md)",
            },
        }
        print(json.dumps(result))

    if distributed:
        torch.distributed.destroy_process_group()


if __name__ == "__main__":
    main()
