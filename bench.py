#!/usr/bin/env python3
"""bench.py — measures BASELINE.json's metric on its named config.

Metric: "samples/sec scanned+aggregated, 1M series rate()[5m]" — whole-job
samples scanned+aggregated per second. Default (N=1) workload is BASELINE
configs[1]: 1M Prometheus counter series, 1h@15s, rate()[5m] step=15s, on one
MI355X. A "step" = one execution of the hot path (one fdb_query_exec launch)
over the resident dataset; inputs are built host-side and uploaded to HBM
BEFORE the timed region (DESIGN.md §5 — no PCIe in the timed region).

N>1 (launched by the driver via torch.distributed.run): weak scaling — each
rank owns its own 1M-series shard on its GPU (the reference's shard model;
series are independent until the cross-series reduce, which configs[1] does not
include — see SURVEY.md §8e). value aggregates all ranks' samples.

Roofline: HBM-bound. achieved = algorithmic bytes per launch (encoded chunk
payload + chunk directory reads + output grid writes) ÷ avg kernel ms from HIP
events on the engine's stream. traffic: rocprofv3 PMC-derived bytes per launch
via env FDB_TRAFFIC_BYTES_PER_LAUNCH (collected separately; profiles/), else null.

cpu_baseline: the CPU oracle (reference-algorithm restatement, kind "port")
timed on the host cores over a bounded sample of the same workload (N=1 rank 0
only).
"""
import argparse
import json
import os
import sys
import time

import numpy as np

# torchrun exports OMP_NUM_THREADS=1, which would serialize each rank's
# host-side synthetic chunk build (OpenMP in libfilodb_amd). Give every rank a
# fair share of the host cores instead; must happen BEFORE the library loads.
_world = int(os.environ.get("WORLD_SIZE", "1"))
if os.environ.get("OMP_NUM_THREADS") in (None, "1"):
    os.environ["OMP_NUM_THREADS"] = str(max(1, (os.cpu_count() or 1) // _world))

REPO = os.path.dirname(os.path.abspath(__file__))
sys.path.insert(0, REPO)

import filodb_amd as fdb  # noqa: E402

T0 = 100000  # epoch-ms origin of the synthetic hour

WORKLOADS = {
    # BASELINE configs[1]: the headline config
    "rate_1m": dict(kind="counter", n_series=1_000_000, n_samples=240,
                    step_ms=15000, func=fdb.FN_RATE, agg=fdb.AGG_NONE,
                    window=300_000, qstep=15000, n_groups=1000,
                    label="1M counter series 1h@15s, rate()[5m] step=15s"),
    # BASELINE configs[2]: gauge windows (LDS sliding)
    "gauge_1m": dict(kind="gauge", n_series=1_000_000, n_samples=240,
                     step_ms=15000, func=fdb.FN_AVG_OVER_TIME, agg=fdb.AGG_NONE,
                     window=600_000, qstep=15000, n_groups=1000,
                     label="1M gauge series 1h@15s, avg_over_time[10m]"),
    # BASELINE configs[4] shape (per-GPU shard): sum by(job)(rate[5m])
    "sumrate_1m": dict(kind="counter", n_series=1_000_000, n_samples=240,
                       step_ms=15000, func=fdb.FN_RATE, agg=fdb.AGG_SUM,
                       window=300_000, qstep=15000, n_groups=1000,
                       label="1M counter series, sum by(job)(rate()[5m]), 1000 groups"),
    # BASELINE configs[3]: 100k hist series x 64 buckets,
    # histogram_quantile(0.99, sum(rate([5m])) by (le-group))
    "hist_100k": dict(kind="hist", n_series=100_000, n_samples=240,
                      step_ms=15000, func=11, agg=1, nb=64,
                      window=300_000, qstep=15000, n_groups=10,
                      label="100k hist series x 64 buckets, "
                            "histogram_quantile(0.99, sum(rate()[5m])) by group"),
    # long-lookback tier exercise: 3h@15s spans two 400-row chunks/series
    "rate_long": dict(kind="counter", n_series=400_000, n_samples=720,
                      step_ms=15000, func=fdb.FN_RATE, agg=fdb.AGG_NONE,
                      window=300_000, qstep=15000, n_groups=1000,
                      label="400k counter series 3h@15s (2 chunks/series), rate()[5m]"),
    "smoke": dict(kind="counter", n_series=20_000, n_samples=240,
                  step_ms=15000, func=fdb.FN_RATE, agg=fdb.AGG_NONE,
                  window=300_000, qstep=15000, n_groups=100,
                  label="20k counter series (reduced)"),
}


def build_store(w, rank):
    st = fdb.ChunkStore()
    kind = {"counter": fdb.COL_COUNTER, "gauge": fdb.COL_GAUGE,
            "hist": fdb.COL_HIST}[w["kind"]]
    # jitter 300 > MaxApproxDelta=250 (DeltaDeltaVector.scala:46): timestamp
    # vectors encode as PACKED DDV, not const — the decode path configs[2]
    # intends the benchmark to exercise (round-1 verdict, weak item 7)
    st.synth_generate(kind, w["n_series"], w["n_samples"], start_ts=T0,
                      step_ms=w["step_ms"], jitter_ms=300, lam=10.0,
                      reset_p=0.001, n_groups=w["n_groups"], seed=42 + rank)
    st.seal()
    return st


def make_query(w):
    span = w["n_samples"] * w["step_ms"]
    return fdb.make_query(T0, w["qstep"], T0 + span, w["window"], w["func"],
                          w["agg"], w["n_groups"] if w["agg"] != fdb.AGG_NONE else 0,
                          param=0.99 if w["kind"] == "hist" else 0.0)


def cpu_baseline(w, budget_s=12.0):
    """Oracle (kind 'port') on host cores over a bounded sample of the workload."""
    sys.path.insert(0, os.path.join(REPO, "oracle"))
    import pyclient as oracle
    cores = os.cpu_count() or 1
    n = 2000
    q = make_query(w)
    elapsed, samples = 0.0, 0
    while True:
        sub = dict(w)
        sub["n_series"] = n
        st = build_store(sub, rank=0)
        nw = q.num_windows
        t0 = time.perf_counter()
        if w["kind"] == "hist":
            oracle.query_exec_hist(st.view(), q, w["nb"])
        else:
            oracle.query_exec(st.view(), q, st.num_series, nw, nthreads=cores)
        elapsed = time.perf_counter() - t0
        samples = n * w["n_samples"] * (w.get("nb", 1) if w["kind"] == "hist" else 1)
        if elapsed > 3.0 or n >= w["n_series"]:
            break
        n = min(w["n_series"], int(n * max(2, 6.0 / max(elapsed, 0.05))))
    return {
        "value": samples / elapsed,
        "unit": "samples/s",
        "cores": cores,
        "kind": "port",
        "sample": f"{n} series x {w['n_samples']} samples of the same workload, "
                  f"OpenMP {cores} threads, {elapsed:.1f}s",
    }


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=10)
    ap.add_argument("--warmup", type=int, default=2)
    ap.add_argument("--workload", default="rate_1m", choices=list(WORKLOADS))
    ap.add_argument("--no-cpu-baseline", action="store_true")
    args = ap.parse_args()

    rank = int(os.environ.get("RANK", 0))
    world = int(os.environ.get("WORLD_SIZE", 1))
    local_rank = int(os.environ.get("LOCAL_RANK", rank))
    if args.gpus > 1 and world == 1:
        # not under torchrun: re-exec ourselves through it (one rank per GPU)
        import subprocess
        cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
               f"--nproc-per-node={args.gpus}", "--master-addr", "127.0.0.1",
               "--master-port", "29507", os.path.abspath(__file__),
               f"--gpus={args.gpus}", f"--steps={args.steps}",
               f"--warmup={args.warmup}", f"--workload={args.workload}"]
        if args.no_cpu_baseline:
            cmd.append("--no-cpu-baseline")
        sys.exit(subprocess.run(cmd).returncode)
    n_gpus = max(world, args.gpus)

    dist = None
    import torch
    if world > 1:
        import torch.distributed as dist_mod
        dist = dist_mod
        # device: one GPU per rank (the driver's 8-GPU launch); modulo lets the
        # distributed path be smoke-tested on fewer GPUs with the gloo backend
        device = local_rank % max(torch.cuda.device_count(), 1)
        torch.cuda.set_device(device)
        dist.init_process_group(os.environ.get("FDB_DIST_BACKEND", "nccl"))
        local_rank = device
        # per-rank sanity so a SCALE run is self-evidencing: the rank/world the
        # backend actually sees, and the device this rank drives
        print(f"[fdb rank-sanity] rank={dist.get_rank()}/{dist.get_world_size()} "
              f"backend={dist.get_backend()} device=cuda:{device} "
              f"visible_gpus={torch.cuda.device_count()}",
              file=sys.stderr, flush=True)
    w = WORKLOADS[args.workload]
    st = build_store(w, rank)
    q = make_query(w)
    nw = q.num_windows

    eng = fdb.Engine(local_rank)
    ds = eng.upload(st)
    samples_per_step = ds.total_samples

    # output buffers resident on device (torch tensors so an RCCL reduce could
    # consume them directly); allocated before the timed region
    if w["kind"] == "hist":
        out = torch.zeros(1, dtype=torch.float64, device=f"cuda:{local_rank}")
        cnt = None
    elif q.agg_id == fdb.AGG_NONE:
        out = torch.empty(st.num_series * nw, dtype=torch.float64,
                          device=f"cuda:{local_rank}")
        cnt = None
    else:
        out = torch.empty(q.num_groups * nw, dtype=torch.float64,
                          device=f"cuda:{local_rank}")
        cnt = torch.empty_like(out)

    is_hist = w["kind"] == "hist"
    nb = w.get("nb", 0)
    if is_hist:
        hist_sums = torch.zeros(q.num_groups * nw * nb, dtype=torch.float64,
                                device=f"cuda:{local_rank}")
        hist_cnt = torch.zeros(q.num_groups * nw, dtype=torch.float64,
                               device=f"cuda:{local_rank}")
        hist_quant = torch.zeros(q.num_groups * nw, dtype=torch.float64,
                                 device=f"cuda:{local_rank}")

        def hist_step():
            eng.query_hist(ds, q, nb, out_bucket_sums=hist_sums,
                           out_counts=hist_cnt, out_quantile=hist_quant,
                           on_device=True)
        for _ in range(args.warmup):
            hist_step()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(args.steps):
            hist_step()
        torch.cuda.synchronize()
        kernel_ms = (time.perf_counter() - t0) / args.steps * 1000  # sync per call
    elif dist is not None and q.agg_id != fdb.AGG_NONE:
        # config #5: the per-shard [G×W] partials merge with ONE collective —
        # RCCL all-reduce over xGMI (ReduceAggregateExec equivalent).
        # The collective is part of the timed step.
        def agg_step():
            eng.query(ds, q, out=out, out_counts=cnt, on_device=True)
            dist.all_reduce(out)
            dist.all_reduce(cnt)
        for _ in range(args.warmup):
            agg_step()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        agg_step()
        torch.cuda.synchronize()
        kernel_ms = (time.perf_counter() - t0) * 1000
    else:
        # warmup pass (also yields avg HIP-event kernel ms for the roofline), then
        # wall-time EXACTLY `steps` launches between barriers+synchronize.
        kernel_ms = eng.bench(ds, q, out, out_counts=cnt, on_device=True,
                              warmup=args.warmup, iters=args.steps)
    if dist:
        dist.barrier()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    if is_hist:
        for _ in range(args.steps):
            hist_step()
    elif dist is not None and q.agg_id != fdb.AGG_NONE:
        for _ in range(args.steps):
            agg_step()
    else:
        _ = eng.bench(ds, q, out, out_counts=cnt, on_device=True,
                      warmup=0, iters=args.steps)
    torch.cuda.synchronize()
    if dist:
        dist.barrier()
    elapsed = time.perf_counter() - t0

    ms_per_step = elapsed / args.steps * 1000
    if dist:
        dev = f"cuda:{local_rank}" if dist.get_backend() == "nccl" else "cpu"
        t = torch.tensor([ms_per_step], device=dev)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        ms_per_step = float(t.item())

    if w["kind"] == "hist":
        samples_per_step *= w.get("nb", 1)   # bucket-samples (SURVEY §8d numerator)
    total_samples_per_step = samples_per_step * n_gpus
    value = total_samples_per_step / (ms_per_step / 1000.0)

    if rank == 0:
        # algorithmic bytes per launch: encoded chunk payload read + output
        # grid written (directory reads are <1% and omitted)
        out_bytes = out.numel() * 8 + (cnt.numel() * 8 if cnt is not None else 0)
        algo_bytes = ds.payload_bytes + out_bytes
        ach = algo_bytes / (kernel_ms / 1000.0) if kernel_ms > 0 else None
        traffic = os.environ.get("FDB_TRAFFIC_BYTES_PER_LAUNCH")
        if not traffic:
            # PMC-derived HBM bytes per launch, measured by rocprofv3 --pmc
            # FETCH_SIZE/WRITE_SIZE runs (profiles/; FETCH doubled per the
            # gfx950 read-side calibration, MI355X_MICROARCH.md §HBM)
            cal = os.path.join(REPO, "profiles", "traffic_calibration.json")
            if os.path.exists(cal):
                with open(cal) as f:
                    c = json.load(f)
                traffic = c.get(args.workload)
        roofline = {
            "bound": "hbm",
            "achieved": round(ach / 1e9, 2) if ach else None,
            "peak": 8000.0,
            "unit": "GB/s",
            "frac": round(ach / 8e12, 4) if ach else None,
            "traffic": float(traffic) if traffic else None,
        }
        cb = None
        if not args.no_cpu_baseline and n_gpus == 1:
            cb = cpu_baseline(w)
        line = {
            "metric": "samples/sec scanned+aggregated, 1M series rate()[5m]",
            "value": value,
            "unit": "samples/s",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "kernel_ms_per_step": kernel_ms,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,   # no published reference numbers
            "dtype": "f64",
            "data": "synthetic",
            "config": {
                "workload": w["label"],
                "series_per_gpu": w["n_series"],
                "samples_per_series": w["n_samples"],
                "windows": nw,
                "payload_bytes_per_gpu": ds.payload_bytes,
            },
            "roofline": roofline,
            "cpu_baseline": cb,
        }
        print(json.dumps(line), flush=True)

    if dist:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
