#!/usr/bin/env python3
"""Run one TPC-H query pipeline N times at a given SF (for rocprofv3
kernel attribution): python tools/profile_query.py <q> [sf] [reps]"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import trino_amd
from trino_amd import tpch_queries as q


def main():
    name = sys.argv[1]
    sf = float(sys.argv[2]) if len(sys.argv) > 2 else 100.0
    reps = int(sys.argv[3]) if len(sys.argv) > 3 else 2
    s = trino_amd.Session(0)
    fn = getattr(q, f"{name}_gpu")
    for i in range(reps):
        r = fn(s, sf)
        print(f"{name} rep{i}: {r['elapsed']*1000:.2f} ms", flush=True)
    s.close()


if __name__ == "__main__":
    main()
