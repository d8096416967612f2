"""Multi-shard reduce correctness on CPU (gloo, world_size=2).

Covers the distributed merge the 8-GPU path uses: each rank computes partial
(sum, count) grids for its shard of series, all-reduces them (RCCL on GPU, gloo
here), and presents. This is the reference's shard model — per-shard leaf plans
reduced by ReduceAggregateExec (AggrOverRangeVectors.scala:18-102) — with the
Akka merge replaced by one collective.
"""
import os

import numpy as np
import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from conftest import REPO, build_store, synth_gauge_series

WORLD = 2


def _make_shard(fdb, rank, n_groups):
    rng = np.random.default_rng(1000 + rank)
    series, groups = [], []
    for s in range(20):
        ts, vs = synth_gauge_series(rng, 50, step=10000, jitter=400, nan_p=0.1)
        series.append([[(int(t), float(v)) for t, v in zip(ts, vs)]])
        groups.append((rank * 20 + s) % n_groups)
    return build_store(fdb, series, groups=groups), groups


def _worker(rank, result_queue):
    import sys
    sys.path.insert(0, REPO)
    sys.path.insert(0, os.path.join(REPO, "oracle"))
    import filodb_amd as fdb
    import pyclient as oracle

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = "29517"
    dist.init_process_group("gloo", rank=rank, world_size=WORLD)

    n_groups = 4
    st, _ = _make_shard(fdb, rank, n_groups)
    start = 100000 + 20 * 10000
    q = fdb.make_query(start, 30000, start + 200000, 100000,
                       fdb.FN_AVG_OVER_TIME, fdb.AGG_AVG, n_groups)
    nw = q.num_windows
    # rank-local partials (the oracle stands in for the GPU engine on CPU;
    # the GPU parity suite covers engine==oracle)
    sums, counts = oracle.query_exec(st.view(), q, st.num_series, nw, out_counts=True)
    t_sum = torch.from_numpy(sums)
    t_cnt = torch.from_numpy(counts)
    dist.all_reduce(t_sum)
    dist.all_reduce(t_cnt)
    merged = torch.where(t_cnt > 0, t_sum / t_cnt, torch.full_like(t_sum, float("nan")))

    if rank == 0:
        result_queue.put(merged.numpy())
    dist.barrier()
    dist.destroy_process_group()


def test_two_shard_avg_merge_equals_global(fdb, oracle):
    ctx = mp.get_context("spawn")
    queue = ctx.Queue()
    procs = [ctx.Process(target=_worker, args=(r, queue)) for r in range(WORLD)]
    for p in procs:
        p.start()
    merged = queue.get(timeout=120)
    for p in procs:
        p.join(timeout=120)
        assert p.exitcode == 0

    # global recomputation: both shards in one store
    n_groups = 4
    series, groups = [], []
    for rank in range(WORLD):
        rng = np.random.default_rng(1000 + rank)
        for s in range(20):
            ts, vs = synth_gauge_series(rng, 50, step=10000, jitter=400, nan_p=0.1)
            series.append([[(int(t), float(v)) for t, v in zip(ts, vs)]])
            groups.append((rank * 20 + s) % n_groups)
    st = build_store(fdb, series, groups=groups)
    start = 100000 + 20 * 10000
    q = fdb.make_query(start, 30000, start + 200000, 100000,
                       fdb.FN_AVG_OVER_TIME, fdb.AGG_AVG, n_groups)
    expected = oracle.query_exec(st.view(), q, st.num_series, q.num_windows)
    np.testing.assert_allclose(merged, expected, rtol=1e-9, equal_nan=True)


def _worker_stddev(rank, result_queue):
    import sys
    sys.path.insert(0, REPO)
    sys.path.insert(0, os.path.join(REPO, "oracle"))
    import filodb_amd as fdb
    import pyclient as oracle

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = "29519"
    dist.init_process_group("gloo", rank=rank, world_size=WORLD)

    n_groups = 3
    st, _ = _make_shard(fdb, rank, n_groups)
    start = 100000 + 20 * 10000
    q = fdb.make_query(start, 30000, start + 200000, 100000,
                       fdb.FN_SUM_OVER_TIME, fdb.AGG_STDDEV, n_groups)
    nw = q.num_windows
    cells = n_groups * nw
    # stacked (raw sums, raw sumsq) partials + counts — the header contract
    stacked, counts = oracle.query_exec(st.view(), q, st.num_series, nw,
                                        out_counts=True)
    assert stacked.shape == (2 * cells,)
    t_sc = torch.from_numpy(stacked)
    t_cnt = torch.from_numpy(counts)
    dist.all_reduce(t_sc)          # ONE collective merges sums and sumsq
    dist.all_reduce(t_cnt)
    s, sq, c = t_sc[:cells], t_sc[cells:], t_cnt
    mean = s / c
    var = sq / c - mean * mean
    merged = torch.where(c > 0, torch.sqrt(var),
                         torch.full_like(s, float("nan")))

    if rank == 0:
        result_queue.put(merged.numpy())
    dist.barrier()
    dist.destroy_process_group()


def test_two_shard_stddev_merge_equals_global(fdb, oracle):
    """Cross-shard stddev via the stacked (sums, sumsq) partial contract
    (StddevRowAggregator.scala:36-52 algebra over one all-reduce)."""
    ctx = mp.get_context("spawn")
    queue = ctx.Queue()
    procs = [ctx.Process(target=_worker_stddev, args=(r, queue))
             for r in range(WORLD)]
    for p in procs:
        p.start()
    merged = queue.get(timeout=120)
    for p in procs:
        p.join(timeout=120)
        assert p.exitcode == 0

    n_groups = 3
    series, groups = [], []
    for rank in range(WORLD):
        rng = np.random.default_rng(1000 + rank)
        for s in range(20):
            ts, vs = synth_gauge_series(rng, 50, step=10000, jitter=400, nan_p=0.1)
            series.append([[(int(t), float(v)) for t, v in zip(ts, vs)]])
            groups.append((rank * 20 + s) % n_groups)
    st = build_store(fdb, series, groups=groups)
    start = 100000 + 20 * 10000
    q = fdb.make_query(start, 30000, start + 200000, 100000,
                       fdb.FN_SUM_OVER_TIME, fdb.AGG_STDDEV, n_groups)
    expected = oracle.query_exec(st.view(), q, st.num_series, q.num_windows)
    np.testing.assert_allclose(merged, expected, rtol=1e-9, atol=1e-12,
                               equal_nan=True)


def _worker_hist(rank, result_queue):
    import sys
    sys.path.insert(0, REPO)
    sys.path.insert(0, os.path.join(REPO, "oracle"))
    import filodb_amd as fdb
    import pyclient as oracle
    from test_hist import make_hist_store, synth_hist

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = "29521"
    dist.init_process_group("gloo", rank=rank, world_size=WORLD)

    n_groups, nb = 3, 8
    rng = np.random.default_rng(2000 + rank)
    series = [synth_hist(rng, 60, nb=nb) for _ in range(12)]
    groups = [(rank * 12 + s) % n_groups for s in range(12)]
    st = make_hist_store(fdb, series, nb=nb, groups=groups)
    start = 100000 + 25 * 15000
    q = fdb.make_query(start, 30000, start + 300000, 300000,
                       fdb.FN_HIST_RATE, fdb.AGG_SUM, n_groups)
    q.param = 0.9
    nw = q.num_windows
    # per-shard [G×W×nb] bucket-rate sums + counts; ONE all-reduce each, then
    # the quantile presentation runs on the MERGED sums (the reference's
    # cross-shard ReduceAggregateExec → present split)
    sums, cnts, _ = oracle.query_exec_hist(st.view(), q, nb)
    t_s = torch.from_numpy(sums)
    t_c = torch.from_numpy(cnts)
    dist.all_reduce(t_s)
    dist.all_reduce(t_c)
    if rank == 0:
        merged = t_s.numpy()
        quant = np.full(n_groups * nw, np.nan)
        for cell in range(n_groups * nw):
            if t_c.numpy()[cell] > 0:
                quant[cell] = oracle.hist_quantile(
                    0.9, merged[cell * nb:(cell + 1) * nb], 2.0, 2.0)
        result_queue.put(quant)
    dist.barrier()
    dist.destroy_process_group()


def test_two_shard_hist_quantile_merge_equals_global(fdb, oracle):
    """Cross-shard histogram pipeline: bucket-rate sums all-reduce BEFORE the
    quantile presentation; merged result equals the single-store global."""
    from test_hist import make_hist_store, synth_hist
    ctx = mp.get_context("spawn")
    queue = ctx.Queue()
    procs = [ctx.Process(target=_worker_hist, args=(r, queue)) for r in range(WORLD)]
    for p in procs:
        p.start()
    merged = queue.get(timeout=180)
    for p in procs:
        p.join(timeout=180)
        assert p.exitcode == 0

    n_groups, nb = 3, 8
    series, groups = [], []
    for rank in range(WORLD):
        rng = np.random.default_rng(2000 + rank)
        series += [synth_hist(rng, 60, nb=nb) for _ in range(12)]
        groups += [(rank * 12 + s) % n_groups for s in range(12)]
    st = make_hist_store(fdb, series, nb=nb, groups=groups)
    start = 100000 + 25 * 15000
    q = fdb.make_query(start, 30000, start + 300000, 300000,
                       fdb.FN_HIST_RATE, fdb.AGG_SUM, n_groups)
    q.param = 0.9
    _, _, want = oracle.query_exec_hist(st.view(), q, nb)
    np.testing.assert_allclose(merged, want, rtol=1e-9, equal_nan=True)


def _worker_w4(rank, result_queue):
    import sys
    sys.path.insert(0, REPO)
    sys.path.insert(0, os.path.join(REPO, "oracle"))
    import filodb_amd as fdb
    import pyclient as oracle

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = "29523"
    dist.init_process_group("gloo", rank=rank, world_size=4)

    n_groups = 4
    st, _ = _make_shard(fdb, rank, n_groups)
    start = 100000 + 20 * 10000
    q = fdb.make_query(start, 30000, start + 200000, 100000,
                       fdb.FN_SUM_OVER_TIME, fdb.AGG_SUM, n_groups)
    nw = q.num_windows
    sums, counts = oracle.query_exec(st.view(), q, st.num_series, nw,
                                     out_counts=True)
    t_sum = torch.from_numpy(sums)
    t_cnt = torch.from_numpy(counts)
    dist.all_reduce(t_sum)
    dist.all_reduce(t_cnt)
    merged = torch.where(t_cnt > 0, t_sum,
                         torch.full_like(t_sum, float("nan")))
    if rank == 0:
        result_queue.put(merged.numpy())
    dist.barrier()
    dist.destroy_process_group()


def test_four_shard_sum_merge_equals_global(fdb, oracle):
    """world_size=4: the same one-collective merge at a deeper reduce tree."""
    ctx = mp.get_context("spawn")
    queue = ctx.Queue()
    procs = [ctx.Process(target=_worker_w4, args=(r, queue)) for r in range(4)]
    for p in procs:
        p.start()
    merged = queue.get(timeout=240)
    for p in procs:
        p.join(timeout=240)
        assert p.exitcode == 0

    n_groups = 4
    series, groups = [], []
    for rank in range(4):
        rng = np.random.default_rng(1000 + rank)
        for s in range(20):
            ts, vs = synth_gauge_series(rng, 50, step=10000, jitter=400,
                                        nan_p=0.1)
            series.append([[(int(t), float(v)) for t, v in zip(ts, vs)]])
            groups.append((rank * 20 + s) % n_groups)
    st = build_store(fdb, series, groups=groups)
    start = 100000 + 20 * 10000
    q = fdb.make_query(start, 30000, start + 200000, 100000,
                       fdb.FN_SUM_OVER_TIME, fdb.AGG_SUM, n_groups)
    expected = oracle.query_exec(st.view(), q, st.num_series, q.num_windows)
    np.testing.assert_allclose(merged, expected, rtol=1e-9, equal_nan=True)
