import os, sys
REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO); sys.path.insert(0, os.path.join(REPO, "oracle")); sys.path.insert(0, os.path.join(REPO, "tests"))
import numpy as np, torch
import filodb_amd as f
import pyclient as oracle
from test_topk import make
st, groups = make(f, n_series=100, n_groups=5, n=120, seed=99)
start = 100000 + 30 * 10000
q = f.make_query(start, 15000, start + 60 * 15000, 120000, f.FN_AVG_OVER_TIME, f.AGG_TOPK, 5, param=5)
nw = q.num_windows
want_v, want_i = oracle.query_exec(st.view(), q, st.num_series, nw, out_counts=True)
eng = f.Engine(0)
got_v = np.empty(5 * nw * 5); got_i = np.empty(5 * nw * 5)
eng.query(eng.upload(st), q, out=got_v, out_counts=got_i)
bad = np.nonzero(~((got_v == want_v) | (np.isnan(got_v) & np.isnan(want_v))))[0]
print("value mismatches:", len(bad), bad[:10])
for c in bad[:5]:
    cell = c // 5; g, w = cell // nw, cell % nw
    print(f"cell g={g} w={w} j={c%5} got={got_v[c]} want={want_v[c]} gotid={got_i[c]} wantid={want_i[c]}")
badi = np.nonzero(got_i != want_i)[0]
print("id mismatches:", len(badi), badi[:10])
