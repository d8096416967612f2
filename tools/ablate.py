#!/usr/bin/env python3
"""ROUND-1 tool (stale): phase ablation of the deleted round-1 scan kernel
via q._pad bits 1/2. The round-2 fast kernel honors only bit 3 (s_memtime
split — see tools/perf_sweep.py, which supersedes this)."""
import os
import sys

import numpy as np
import torch  # must precede HIP init by the engine

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)
import filodb_amd as fdb  # noqa: E402

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

T0 = 100000


def main():
    workload = sys.argv[1] if len(sys.argv) > 1 else "rate"
    n_series = int(sys.argv[2]) if len(sys.argv) > 2 else 1_000_000
    st = fdb.ChunkStore()
    kind = fdb.COL_COUNTER if workload == "rate" else fdb.COL_GAUGE
    st.synth_generate(kind, n_series, 240, start_ts=T0, step_ms=15000,
                      jitter_ms=250, lam=10.0, reset_p=0.001, n_groups=1000, seed=42)
    st.seal()
    eng = fdb.Engine(0)
    ds = eng.upload(st)
    func = fdb.FN_RATE if workload == "rate" else fdb.FN_AVG_OVER_TIME
    window = 300_000 if workload == "rate" else 600_000
    if len(sys.argv) > 3 and sys.argv[3] == "census":
        # concurrent-block high-water during a full launch
        q = fdb.make_query(T0, 15000, T0 + 240 * 15000, window, func)
        q._pad = 3 | 16
        out = torch.zeros(n_series * q.num_windows, dtype=torch.float64, device="cuda")
        cnt = torch.zeros(1024, dtype=torch.float64, device="cuda")
        nw = q.num_windows
        import ctypes
        rc = fdb.lib().fdb_query_exec(eng._h, ds._h, ctypes.byref(q),
                                      fdb._as_f64_ptr(out), fdb._as_f64_ptr(cnt), 1)
        assert rc == 0, fdb.lib().fdb_last_error()
        eng.synchronize()
        vals = cnt[:2].cpu().numpy().view('uint64') if hasattr(cnt[:2].cpu().numpy(), 'view') else None
        import numpy as np
        raw = cnt[:2].cpu().numpy()
        u = raw.view(np.uint64)
        print(f"census: final_active={u[0]} peak_blocks={u[1]}")
        return
    if len(sys.argv) > 3 and sys.argv[3] == "time":
        # phase-timing mode: kernel reports wall cycles per wave per phase
        q = fdb.make_query(T0, 15000, T0 + 240 * 15000, window, func)
        q._pad = 3 | 8
        out = torch.zeros(n_series * q.num_windows, dtype=torch.float64, device="cuda")
        eng.query(ds, q, out=out, on_device=True)
        eng.synchronize()
        waves = 8192 * 4
        t = out[:waves * 4].reshape(waves, 4).cpu().numpy()
        t = t[t.sum(axis=1) > 0]
        tot = t.sum(axis=0)
        per = tot / tot.sum()
        print(f"phase cycles (avg/wave): decode={t[:,0].mean():,.0f} "
              f"meta={t[:,1].mean():,.0f} windows={t[:,2].mean():,.0f}")
        print(f"phase share: decode={per[0]:.2%} meta={per[1]:.2%} windows={per[2]:.2%}")
        return
    for phase, name in [(3, "full"), (1, "decode-only"), (2, "window-only")]:
        q = fdb.make_query(T0, 15000, T0 + 240 * 15000, window, func)
        q._pad = phase
        out = torch.empty(n_series * q.num_windows, dtype=torch.float64, device="cuda")
        ms = eng.bench(ds, q, out, on_device=True, warmup=2, iters=10)
        print(f"{workload} {name:12s} kernel_ms={ms:8.3f}", flush=True)


if __name__ == "__main__":
    main()
