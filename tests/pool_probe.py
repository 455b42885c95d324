"""Subprocess probe: hipdf device pool installed as the torch allocator,
spill-before-OOM callback fires under pressure, and the engine runs a
query with every tensor living in the pool.

Run in its own process (the allocator must install before the first
device allocation). Prints POOL_OK on success.
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def main():
    import hipdf

    alloc = torch.cuda.memory.CUDAPluggableAllocator(
        hipdf.__file__, "hipdf_torch_malloc", "hipdf_torch_free")
    torch.cuda.memory.change_current_allocator(alloc)
    # small explicit slab so the pressure path triggers quickly
    rc = hipdf.pool_init(0.0, 512 << 20)
    assert rc == 0, rc
    assert hipdf.pool_active()

    from spark_rapids_amd.memory import device_pool
    from spark_rapids_amd.memory.spill import SpillableBatch, spill_store

    device_pool._state["active"] = True
    device_pool._state["tried"] = True
    calls = []

    def on_exhausted(needed, retry):
        freed = spill_store.spill_device(needed)
        calls.append((needed, retry, freed))
        print(f"[probe] exhausted: needed={needed} retry={retry} "
              f"freed={freed} used={hipdf.pool_used()}", flush=True)
        return 1 if freed else 0

    hipdf.pool_set_spill_cb(on_exhausted)

    from spark_rapids_amd import Column, ColumnBatch, INT64

    # ~300 MiB spillable batch in the 512 MiB slab
    import numpy as np

    a = Column.from_numpy(np.arange(40_000_000, dtype=np.int64),
                          INT64).cuda()
    handle = SpillableBatch(ColumnBatch([a], 40_000_000))
    del a
    used0 = hipdf.pool_used()
    assert used0 >= 300 << 20, used0

    # second ~300 MiB allocation cannot fit -> callback must spill handle
    b = torch.empty(40_000_000, dtype=torch.int64, device="cuda")
    b.fill_(7)
    print(f"[probe] after b: used={hipdf.pool_used()} "
          f"reserved={hipdf.pool_reserved()}", flush=True)
    assert calls, "failure callback never fired"
    from spark_rapids_amd.memory.spill import HOST

    assert handle.state == HOST, handle.state
    assert int(b[123].item()) == 7
    # the spill made room: b lives in the slab, not the overflow path
    assert hipdf.pool_overflow() == 0, hipdf.pool_overflow()

    # spilled batch resurrects and round-trips (b released first: the
    # probe slab cannot hold both)
    del b
    back = handle.get()
    assert back.columns[0].cpu().to_numpy()[12345] == 12345

    # the engine end to end under the pool allocator
    from spark_rapids_amd import Session, col, sum_

    s = Session()
    df = s.create_dataframe({"k": [1, 2, 1, 2], "v": [1.0, 2.0, 3.0, 4.0]})
    out = sorted(df.group_by("k").agg(sum_(col("v"))).collect())
    assert out == [(1, 4.0), (2, 6.0)], out
    st = device_pool.stats()
    assert st["active"] and st["high_watermark"] > (250 << 20), st
    print("POOL_OK calls=%d high_watermark=%d reserved=%d overflow=%d"
          % (len(calls), st["high_watermark"], st["reserved"],
             hipdf.pool_overflow()))


if __name__ == "__main__":
    main()
