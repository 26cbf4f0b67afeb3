#!/usr/bin/env python3
"""Extract the reference's own tiny (sf0.01) TPC-H dataset into a committed
golden fixture, tests/golden/tiny_sf001.json.gz.

Source (PUBLIC reference test data, read HERE only — /root/reference does not
exist on GPU boxes):
  testing/trino-testing-resources/src/main/resources/io/trino/plugin/
  deltalake/testing/resources/databricks73/{part,supplier,customer,orders,
  lineitem,partsupp,nation,region}/ — a Delta Lake capture of dbgen sf0.01
  output (verified: the scale-independent streams — dates, quantities,
  comments, flags — are byte-identical to the 785/200 canonical SF1 rows in
  plugin/trino-example-http example-data).

The Delta logs contain UPDATE commits over small key ranges (the Delta tests
exercise row rewrites); rows in those ranges may no longer be pristine dbgen
output, so their key ranges are recorded as `tainted` per table and the
generator-pinning tests skip them.

Output layout per table: {"columns": {name: [values...]}, "tainted": [[lo,hi]
(exclusive,inclusive]], "key": keycol}  — dates as epoch days, floats kept as
parquet float64 repr (exact), sorted by primary key.
"""
import datetime
import glob
import gzip
import json
import os
import re

REF = ("/root/reference/testing/trino-testing-resources/src/main/resources/"
       "io/trino/plugin/deltalake/testing/resources/databricks73")
OUT = os.path.join(os.path.dirname(__file__), "..", "tests", "golden",
                   "tiny_sf001.json.gz")

TABLES = {
    "part": ["partkey"],
    "supplier": ["suppkey"],
    "customer": ["custkey"],
    "orders": ["orderkey"],
    "lineitem": ["orderkey", "linenumber"],
    "partsupp": ["partkey", "suppkey"],
    "nation": ["nationkey"],
    "region": ["regionkey"],
}

PRED = re.compile(r"\(\((\w+)#\d+L? <= (\d+)\) && \(\w+#\d+L? > (\d+)\)\)")


def live_files_and_taint(table):
    """Replay the Delta log: live parquet paths + tainted key ranges."""
    live = {}
    taint = []
    for logf in sorted(glob.glob(f"{REF}/{table}/_delta_log/*.json")):
        for line in open(logf):
            d = json.loads(line)
            if "add" in d:
                live[d["add"]["path"]] = True
            elif "remove" in d:
                live.pop(d["remove"]["path"], None)
            elif "commitInfo" in d:
                ci = d["commitInfo"]
                if ci.get("operation") == "UPDATE":
                    m = PRED.match(ci["operationParameters"]["predicate"])
                    if not m:
                        raise ValueError(ci["operationParameters"])
                    taint.append([int(m.group(3)), int(m.group(2))])
    return sorted(live), taint


def main():
    import pyarrow as pa
    import pyarrow.parquet as pq

    out = {}
    for table, key in TABLES.items():
        files, taint = live_files_and_taint(table)
        paths = [f"{REF}/{table}/{p}" for p in files]
        missing = [p for p in paths if not os.path.exists(p)]
        if missing:
            raise FileNotFoundError(missing)
        tb = pa.concat_tables([pq.read_table(p) for p in paths])
        rows = tb.to_pylist()
        rows.sort(key=lambda r: tuple(r[k] for k in key))
        cols = {}
        for name in tb.column_names:
            vals = []
            for r in rows:
                v = r[name]
                if isinstance(v, datetime.date):
                    v = (v - datetime.date(1970, 1, 1)).days
                vals.append(v)
            cols[name] = vals
        out[table] = {"columns": cols, "tainted": taint, "key": key,
                      "rows": len(rows)}
        print(f"{table}: {len(rows)} rows from {len(files)} live files, "
              f"{len(taint)} tainted ranges")
    payload = {
        "source": "testing/trino-testing-resources/.../deltalake/testing/"
                  "resources/databricks73 (dbgen sf0.01 capture; see header)",
        "tables": out,
    }
    with gzip.open(OUT, "wt") as f:
        json.dump(payload, f)
    print(f"wrote {OUT} ({os.path.getsize(OUT)//1024} KiB)")


if __name__ == "__main__":
    main()
