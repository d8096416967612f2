"""ROUND-1 tool (stale): FDB_HIST_TIME phase split of the round-1 hist ring
kernel (kept behind FDB_HIST_V1=1). hist2 uses FDB_HIST_ABLATE — see
tools/perf_sweep.py."""
import os, sys
REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)
import torch  # HIP init order
import filodb_amd as fdb

n_series, n_samples, nb, ng = 100_000, 240, 64, 10
st = fdb.ChunkStore(); st.set_max_rows(400)
st.synth_generate(fdb.COL_HIST, n_series, n_samples, n_groups=ng)
st.seal()
eng = fdb.Engine(0); ds = eng.upload(st)
T0 = 100000 + 20 * 15000
q = fdb.make_query(T0, 15000, T0 + (n_samples + 1) * 15000, 300_000,
                   fdb.FN_HIST_RATE, fdb.AGG_SUM, ng, param=0.99)
nw = q.num_windows
sums = torch.zeros(ng * nw * nb, dtype=torch.float64, device="cuda")
cnt = torch.zeros(ng * nw, dtype=torch.float64, device="cuda")
quant = torch.zeros(ng * nw, dtype=torch.float64, device="cuda")
eng.query_hist(ds, q, nb, out_bucket_sums=sums, out_counts=cnt,
               out_quantile=quant, on_device=True)
eng.synchronize()
v = cnt[:4].cpu().numpy()
parse, scan, win, ne = v
tot = parse + scan + win
print(f"elements(wave-summed): {ne:.0f}")
print(f"parse: {parse/ne:8.0f} cyc/elem  ({100*parse/tot:.1f}%)")
print(f"scan:  {scan/ne:8.0f} cyc/elem  ({100*scan/tot:.1f}%)")
print(f"win:   {win/ne:8.0f} cyc/elem  ({100*win/tot:.1f}%)")
print(f"total: {tot/ne:8.0f} cyc/elem")
