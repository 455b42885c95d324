import time, torch
from spark_rapids_amd.api import Session
from spark_rapids_amd.bench import datagen
from spark_rapids_amd.bench.queries import POWER_RUN
import bench as B

session = Session()
tables = B._make_tables(session, 20_000_000, seed=1, device="cuda", partitions=8)
for _ in range(2):
    for name, fn in POWER_RUN:
        fn(tables).collect()
torch.cuda.synchronize()
for name, fn in POWER_RUN:
    t0 = time.perf_counter()
    for _ in range(5):
        fn(tables).collect()
    torch.cuda.synchronize()
    print(name, round((time.perf_counter() - t0) / 5 * 1000, 2), "ms")
