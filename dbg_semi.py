import time
import spark_rapids_amd as sr
from spark_rapids_amd import col, count_star, sum_, min_
from spark_rapids_amd.bench import datagen

s = sr.Session()
fact = s.from_batches([datagen.gen_fact_partition(5_000_000, 1)],
                      datagen.fact_schema(), "fact").select(
    col("ss_item_id"), col("ss_store_id"), col("ss_promo"),
    col("ss_sales_price"), col("ss_discount"), col("ss_quantity"))
fact = s.from_batches([b.cuda() for b in
                       fact.collect_batch() and []] or
                      [fact.collect_batch().cuda()],
                      fact.schema, "factg")
hot = fact.group_by("ss_item_id").agg(count_star()).filter(col("count(*)") > 100)
q = fact.join(hot, on="ss_item_id", how="semi").agg(count_star())
q.collect()
import torch
torch.cuda.synchronize(); t0=time.perf_counter()
out = q.collect(); torch.cuda.synchronize()
print("semi_anti", round((time.perf_counter()-t0)*1000,1), "ms", out)
df = q
m = df.metrics()
print(m if not hasattr(m, "items") else {k: round(v,4) if isinstance(v,float) else v for k,v in m.items()})

q2 = fact.group_by("ss_store_id", "ss_promo").agg(
    sum_(col("ss_sales_price")), min_(col("ss_discount")), count_star())
q2.collect()
torch.cuda.synchronize(); t0=time.perf_counter()
q2.collect(); torch.cuda.synchronize()
print("multi_key_agg", round((time.perf_counter()-t0)*1000,1), "ms")
# per-phase timing of the semi query
hotdf = hot
hotdf.collect()
torch.cuda.synchronize(); t0=time.perf_counter()
hotdf.collect(); torch.cuda.synchronize()
print("hot subquery alone", round((time.perf_counter()-t0)*1000,1), "ms")
