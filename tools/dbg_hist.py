import os, sys
REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO); sys.path.insert(0, os.path.join(REPO, "oracle"))
sys.path.insert(0, os.path.join(REPO, "tests"))
import numpy as np
import torch  # init order
import filodb_amd as f
import pyclient as oracle
from test_hist import make_hist_store, synth_hist

rng = np.random.default_rng(4)
series = [synth_hist(rng, 40, nb=8)]
start = int(series[0][0][0]) - 100000
q = f.make_query(start, 60000, start + 30 * 60000, 300000, f.FN_HIST_RATE, f.AGG_SUM, 1, param=0.5)
st = make_hist_store(f, series, nb=8, groups=[0])
ws_, wc_, wq_ = oracle.query_exec_hist(st.view(), q, 8)
eng = f.Engine(0)
ds = eng.upload(st)
nw = q.num_windows
gs = np.zeros(nw*8); gc = np.zeros(nw); gq = np.zeros(nw)
eng.query_hist(ds, q, 8, out_bucket_sums=gs, out_counts=gc, out_quantile=gq)
print("counts diff:", np.nonzero(gc != wc_)[0][:20], "got", gc[:31], "want", wc_[:31])
bad = np.nonzero(~np.isclose(gs, ws_, rtol=1e-9, atol=1e-12))[0]
print("sum mismatch cells:", bad[:10])
for c in bad[:5]:
    print("cell", c, "w", c//8, "b", c%8, "got", gs[c], "want", ws_[c])
ts0 = series[0][0]
tsb,_,_,_,_ = st.chunk(0,0)
enc = oracle.decode_longs(tsb)
print("enc ts[:6]", enc[:6], "start", start)
