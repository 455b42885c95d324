import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
import torch
import hipdf as ext
from spark_rapids_amd import Column, ColumnBatch, INT64, INT32, FLOAT64
from spark_rapids_amd.ops import gpu_backend as gb, cpu_backend as cb

s = torch.cuda.current_stream().cuda_stream

# tiny case
arr = np.array([5, 3, 9, 3, 1, 7, 3, 0, 2, 8], dtype=np.int64)
c = Column.from_numpy(arr, INT64).cuda()
n = len(arr)
keys = torch.empty(n, dtype=torch.int64, device="cuda")
ext.make_sort_keys(4, c.data.data_ptr(), 0, 0, False, False, False,
                   keys.data_ptr(), n, s)
torch.cuda.synchronize()
print("keys:", [hex(k & (2**64-1)) for k in keys.cpu().tolist()][:5])

nb = ext.sort_num_blocks(n)
counts = torch.empty(256 * nb, dtype=torch.int64, device="cuda")
ext.radix_count(keys.data_ptr(), 0, counts.data_ptr(), n, s)
torch.cuda.synchronize()
cc = counts.cpu().numpy()
print("count nonzero bins:", {i: int(v) for i, v in enumerate(cc) if v})

off, total = gb._exclusive_scan_i64(counts)
print("total", total)
keys2 = torch.empty(n, dtype=torch.int64, device="cuda")
perm2 = torch.empty(n, dtype=torch.int32, device="cuda")
ext.radix_scatter(keys.data_ptr(), 0, 0, off.data_ptr(), keys2.data_ptr(),
                  perm2.data_ptr(), n, s)
torch.cuda.synchronize()
print("perm after pass0:", perm2.cpu().tolist())
print("gathered:", arr[perm2.cpu().numpy()])

# full sort_order small
order = gb.sort_order(ColumnBatch([c]), [0], [False], [False]).cpu()
print("sort_order small:", arr[order.to_numpy()])

# bigger no-null int64
arr2 = np.random.default_rng(0).integers(0, 100, 1000).astype(np.int64)
c2 = Column.from_numpy(arr2, INT64).cuda()
o2 = gb.sort_order(ColumnBatch([c2]), [0], [False], [False]).cpu()
g2 = arr2[o2.to_numpy()]
print("big sorted ok:", bool((g2 == np.sort(arr2)).all()))

# with nulls
valid = np.random.default_rng(1).random(1000) >= 0.2
c3 = Column.from_numpy(arr2, INT64, valid).cuda()
o3 = gb.sort_order(ColumnBatch([c3]), [0], [False], [False]).cpu()
o3cpu = cb.sort_order(ColumnBatch([Column.from_numpy(arr2, INT64, valid)]),
                      [0], [False], [False])
g3 = arr2[o3.to_numpy()]
v3 = valid[o3.to_numpy()]
g3c = arr2[o3cpu.to_numpy()]
v3c = valid[o3cpu.to_numpy()]
nn = int((~valid).sum())
print("nulls first ok:", bool((~v3[:nn]).all()),
      "values ok:", bool((g3[nn:] == g3c[nn:]).all()))

# int32 path
c4 = Column.from_numpy(arr2.astype(np.int32), INT32).cuda()
o4 = gb.sort_order(ColumnBatch([c4]), [0], [False], [False]).cpu()
print("i32 ok:", bool((arr2.astype(np.int32)[o4.to_numpy()] == np.sort(arr2.astype(np.int32))).all()))

# float desc
f = np.random.default_rng(2).uniform(-5, 5, 1000)
c5 = Column.from_numpy(f, FLOAT64).cuda()
o5 = gb.sort_order(ColumnBatch([c5]), [0], [True], [True]).cpu()
print("f64 desc ok:", bool((f[o5.to_numpy()] == -np.sort(-f)).all()))
