import numpy as np
import spark_rapids_amd as sr
from spark_rapids_amd import DType, col

D72 = DType.decimal(7, 2); D104 = DType.decimal(10, 4)
sg = sr.Session(); sc = sr.Session({"spark.rapids.sql.enabled": False})
rng = np.random.default_rng(11)
a = [round(float(x), 2) for x in rng.uniform(-99999, 99999, 20000)]
b = [round(float(x), 4) for x in rng.uniform(-50, 50, 20000)]
for i in range(0, 20000, 97):
    b[i] = 0.0

def _dec_df(s):
    df = s.create_dataframe({"a": a, "b": b})
    return df.select(col("a").cast(D72).alias("a"),
                     col("b").cast(D104).alias("b"))

g = _dec_df(sg).select((col("a") * col("b")).alias("r")).to_pydict()["r"]
c = _dec_df(sc).select((col("a") * col("b")).alias("r")).to_pydict()["r"]
bad = [(i, a[i], b[i], g[i], c[i]) for i in range(20000) if g[i] != c[i]]
print("n_bad:", len(bad))
for row in bad[:10]:
    print(row)
# also dump the casted operand unscaled values for the first bad row
if bad:
    i = bad[0][0]
    ga = _dec_df(sg).to_pydict()
    ca = _dec_df(sc).to_pydict()
    print("gpu operands:", ga["a"][i], ga["b"][i])
    print("cpu operands:", ca["a"][i], ca["b"][i])
