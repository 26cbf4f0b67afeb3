import trino_amd
from trino_amd import tpch_queries as q
s = trino_amd.Session(0)
for i in range(2):
    r = q.q18_gpu(s, 100.0)
    print("q18", round(r["elapsed"] * 1000, 1))
s.close()
