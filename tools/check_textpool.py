#!/usr/bin/env python3
"""Iteration harness: compare the restated text pool against every comment
sample in tests/golden/tiny_sf001.json.gz (offsets computed from the pinned
comment streams). Prints the first divergence with context."""
import ctypes, gzip, json, sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np

lib = ctypes.CDLL('oracle/liboracle.so')
lib.tpch_text_pool.restype = ctypes.c_void_p
lib.tpch_text_slices.argtypes = [ctypes.c_int64]*3 + [ctypes.c_int32]*3 + [ctypes.c_void_p]*2

d = json.load(gzip.open('tests/golden/tiny_sf001.json.gz','rt'))['tables']

def slices(seed, count, per_value, usage, avg):
    offs = np.zeros(count*per_value, np.int64)
    lens = np.zeros(count*per_value, np.int32)
    lib.tpch_text_slices(seed, 1, count, per_value, usage, avg,
                         offs.ctypes.data, lens.ctypes.data)
    return offs, lens

samples = []  # (offset, text, table, row)
o = d['orders']['columns']
offs, lens = slices(276090261, len(o['comment']), 1, 2, 49)
for i,c in enumerate(o['comment']):
    assert lens[i]==len(c), (i, lens[i], len(c))
    samples.append((offs[i], c, 'orders', i))
cu = d['customer']['columns']
offs, lens = slices(1335826707, 1500, 1, 2, 73)
for i,c in enumerate(cu['comment']): samples.append((offs[i], c, 'customer', i))
p = d['part']['columns']
offs, lens = slices(804159733, 2000, 1, 2, 14)
for i,c in enumerate(p['comment']): samples.append((offs[i], c, 'part', i))
# partsupp in dbgen order
ps = d['partsupp']['columns']
bykey = {}
for i in range(len(ps['partkey'])): bykey[(ps['partkey'][i],ps['suppkey'][i])]=ps['comment'][i]
S=100
offs, lens = slices(1961692154, 2000, 4, 8, 124)
k=0
for p_ in range(1,2001):
    for j in range(4):
        sk=(p_ + j*(S//4 + (p_-1)//S)) % S + 1
        samples.append((offs[k], bykey[(p_,sk)], 'partsupp', k)); k+=1
# supplier comments (no BBB expected in 100 rows; mismatch would show anyway)
su = d['supplier']['columns']
offs, lens = slices(1341315363, 100, 1, 2, 63)
for i,c in enumerate(su['comment']): samples.append((offs[i], c, 'supplier', i))
# lineitem: per order line_count values, usage 14
import oracle, collections
g = oracle.gen_lineitem(0.01)
lc = collections.Counter(g['orderkey'].tolist())
okeys = sorted(lc)
li = d['lineitem']['columns']
offs, lens = slices(1095462486, 15000, 7, 14, 27)
li_i = 0
for oi, key in enumerate(okeys):
    for j in range(lc[key]):
        samples.append((offs[oi*7+j], li['comment'][li_i], 'lineitem', li_i)); li_i += 1

POOL = 300*1024*1024
pool_ptr = lib.tpch_text_pool()

samples.sort()
bad = 0
shown = 0
covered = 0
for off, text, table, row in samples:
    got = ctypes.string_at(pool_ptr + int(off), len(text)).decode('latin1')
    covered += len(text)
    if got != text:
        bad += 1
        if shown < int(sys.argv[1]) if len(sys.argv)>1 else shown < 3:
            k = next(i for i in range(len(text)) if got[i]!=text[i])
            print(f"MISMATCH {table}[{row}] off={off} at +{k} (pool off {off+k})")
            print("  exp:", repr(text[max(0,k-40):k+40]))
            print("  got:", repr(got[max(0,k-40):k+40]))
            shown += 1
print(f"{bad}/{len(samples)} samples mismatch; {covered/1e6:.1f} MB covered")
