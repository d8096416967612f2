#!/usr/bin/env python3
"""Perf sweep on one resident dataset: builds the headline (configs[1]) store
once, then times kernel variants toggled by env knobs — avoids paying the
host-side synth build per variant. Run on a GPU box via gpurun."""
import os
import sys
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)

import filodb_amd as fdb  # noqa: E402

T0 = 100000


def build(n_series=1_000_000, kind=None):
    st = fdb.ChunkStore()
    st.synth_generate(kind if kind is not None else fdb.COL_COUNTER,
                      n_series, 240, start_ts=T0, step_ms=15000, jitter_ms=300,
                      lam=10.0, reset_p=0.001, n_groups=1000, seed=42)
    st.seal()
    return st


def bench(eng, ds, q, out, cnt=None, warmup=2, iters=8):
    return eng.bench(ds, q, out, out_counts=cnt, on_device=True,
                     warmup=warmup, iters=iters)


def main():
    import torch
    n = int(os.environ.get("SWEEP_SERIES", "1000000"))
    st = build(n)
    eng = fdb.Engine(0)
    ds = eng.upload(st)
    span = 240 * 15000
    q_rate = fdb.make_query(T0, 15000, T0 + span, 300000, fdb.FN_RATE)
    q_sum = fdb.make_query(T0, 15000, T0 + span, 300000, fdb.FN_RATE,
                           fdb.AGG_SUM, 1000)
    nw = q_rate.num_windows
    out = torch.empty(n * nw, dtype=torch.float64, device="cuda:0")
    gout = torch.empty(1000 * nw, dtype=torch.float64, device="cuda:0")
    gcnt = torch.empty_like(gout)

    for waves in ("6", "5"):
        os.environ["FDB_RATE_WAVES"] = waves
        ms = bench(eng, ds, q_rate, out)
        print(f"rate waves={waves}: {ms:.3f} ms", flush=True)
    os.environ["FDB_RATE_WAVES"] = "6"
    for gridcap in ("1536", "3072", "16384"):
        os.environ["FDB_GRID"] = gridcap
        ms = bench(eng, ds, q_rate, out)
        print(f"rate grid={gridcap}: {ms:.3f} ms", flush=True)
    os.environ.pop("FDB_GRID")
    for fused in ("1", "0"):
        os.environ["FDB_FUSED_GROUP"] = fused
        ms = bench(eng, ds, q_sum, gout, gcnt)
        print(f"sum-by-group fused={fused}: {ms:.3f} ms", flush=True)
    os.environ.pop("FDB_FUSED_GROUP")
    # phase split (s_memtime; pm bit 3 clobbers out[] with per-wave cycles)
    os.environ["FDB_RATE_WAVES"] = "6"
    q_rate._pad = 8
    eng.query(ds, q_rate, out=out, on_device=True)
    eng.synchronize()
    nwaves = 8192 * 4
    ph = out[:nwaves * 4].reshape(nwaves, 4).cpu().numpy()
    tot = ph.sum(axis=0)
    names = ["decode", "meta", "inversion", "windows"]
    print("rate phase split:",
          " ".join(f"{nm}={v / tot.sum() * 100:.0f}%" for nm, v in zip(names, tot)),
          f"(sum {tot.sum() / 1e9:.2f} Gcyc over {nwaves} waves)", flush=True)
    q_rate._pad = 0

    # gauge workload on the same box (separate store: raw f64 values)
    del ds
    st2 = build(n, kind=fdb.COL_GAUGE)
    ds2 = eng.upload(st2)
    q_avg = fdb.make_query(T0, 15000, T0 + span, 600000, fdb.FN_AVG_OVER_TIME)
    ms = bench(eng, ds2, q_avg, out)
    print(f"avg_over_time[10m]: {ms:.3f} ms", flush=True)
    q_min = fdb.make_query(T0, 15000, T0 + span, 600000, fdb.FN_MIN_OVER_TIME)
    ms = bench(eng, ds2, q_min, out)
    print(f"min_over_time[10m]: {ms:.3f} ms", flush=True)

    # hist workload (config #4 shape): v2 occupancy A/B
    del ds2
    st3 = build(100_000, kind=fdb.COL_HIST)
    ds3 = eng.upload(st3)
    qh = fdb.make_query(T0, 15000, T0 + span, 300000, fdb.FN_HIST_RATE,
                        fdb.AGG_SUM, 1000, param=0.99)
    nwh = qh.num_windows
    hs = torch.zeros(1000 * nwh * 64, dtype=torch.float64, device="cuda:0")
    hc = torch.zeros(1000 * nwh, dtype=torch.float64, device="cuda:0")
    hq = torch.zeros(1000 * nwh, dtype=torch.float64, device="cuda:0")
    def hist_ms(tag):
        eng.query_hist(ds3, qh, 64, out_bucket_sums=hs, out_counts=hc,
                       out_quantile=hq, on_device=True)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(3):
            eng.query_hist(ds3, qh, 64, out_bucket_sums=hs, out_counts=hc,
                           out_quantile=hq, on_device=True)
        torch.cuda.synchronize()
        print(f"hist {tag}: {(time.perf_counter() - t0) / 3 * 1000:.3f} ms",
              flush=True)
    for hw in ("4", "5"):
        os.environ["FDB_HIST_WAVES"] = hw
        hist_ms(f"waves={hw}")
    os.environ.pop("FDB_HIST_WAVES")
    for abl in ("1", "3"):   # 1 = no emits; 3 = no emits + no decodes
        os.environ["FDB_HIST_ABLATE"] = abl
        hist_ms(f"ablate={abl}")
    # in-kernel phase split (abl bit 3): advance / decode / emit cycles
    os.environ["FDB_HIST_ABLATE"] = "9"   # timing + no emits: out_cnt holds only cycles
    hc.zero_()
    eng.query_hist(ds3, qh, 64, out_bucket_sums=hs, out_counts=hc,
                   out_quantile=None, on_device=True)
    eng.synchronize()
    ph = hc[:4].cpu().numpy()
    tot = ph.sum()
    print("hist phase split: advance=%.0f%% decode=%.0f%% emit=%.0f%% "
          "(sum %.2f Gcyc)" % (ph[0]/tot*100, ph[1]/tot*100, ph[2]/tot*100,
                               tot/1e9), flush=True)
    os.environ.pop("FDB_HIST_ABLATE")


if __name__ == "__main__":
    main()
