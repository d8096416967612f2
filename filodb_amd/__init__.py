"""filodb_amd — Python wrapper over the C-ABI of the MI355X-native FiloDB
chunk-scan + range-vector engine (include/filodb_amd.h).

The wrapper is plumbing only: chunk building, dataset upload and query launch all
happen in the native library. PyTorch is used solely for device-tensor interop
(RCCL all-reduce of aggregated grids in multi-GPU runs).
"""
import ctypes
import os

import numpy as np

_ROOT = os.path.dirname(os.path.abspath(__file__))
LIB_PATH = os.environ.get("FDB_LIB", os.path.join(_ROOT, "libfilodb_amd.so"))

# function ids (include/filodb_amd.h; dispatch table RangeFunction.scala:294-410)
FN_RATE, FN_INCREASE, FN_DELTA = 0, 1, 2
FN_SUM_OVER_TIME, FN_COUNT_OVER_TIME, FN_AVG_OVER_TIME = 3, 4, 5
FN_MIN_OVER_TIME, FN_MAX_OVER_TIME = 6, 7
FN_STDDEV_OVER_TIME, FN_STDVAR_OVER_TIME, FN_CHANGES = 8, 9, 10
FN_HIST_RATE = 11
FN_LAST = 12
FN_PRESENT, FN_TIMESTAMP, FN_ZSCORE = 13, 14, 15
FN_QUANTILE_OVER_TIME, FN_MAD_OVER_TIME = 16, 17
FN_PREDICT_LINEAR = 18
FN_RATE_OVER_DELTA = 19
FN_HOLT_WINTERS = 20
# aggregation ids (RowAggregator implementations)
AGG_NONE, AGG_SUM, AGG_COUNT, AGG_MIN, AGG_MAX, AGG_AVG = 0, 1, 2, 3, 4, 5
AGG_TOPK, AGG_BOTTOMK = 6, 7
AGG_STDDEV, AGG_STDVAR, AGG_GROUP = 8, 9, 10
AGG_QUANTILE, AGG_COUNT_VALUES = 11, 12
# column kinds
COL_GAUGE, COL_COUNTER, COL_HIST = 0, 1, 2

_c_double_p = ctypes.POINTER(ctypes.c_double)


class Query(ctypes.Structure):
    _fields_ = [
        ("start", ctypes.c_int64), ("step", ctypes.c_int64),
        ("end", ctypes.c_int64), ("window", ctypes.c_int64),
        ("func_id", ctypes.c_int32), ("agg_id", ctypes.c_int32),
        ("num_groups", ctypes.c_int32), ("_pad", ctypes.c_int32),
        ("param", ctypes.c_double), ("param2", ctypes.c_double),
    ]

    @property
    def num_windows(self):
        return (self.end - self.start) // self.step + 1


class ChunkInfo(ctypes.Structure):
    _fields_ = [
        ("ts_vec", ctypes.POINTER(ctypes.c_uint8)),
        ("val_vec", ctypes.POINTER(ctypes.c_uint8)),
        ("num_rows", ctypes.c_int32),
        ("start_time", ctypes.c_int64), ("end_time", ctypes.c_int64),
        ("ts_vec_len", ctypes.c_int32), ("val_vec_len", ctypes.c_int32),
    ]


class View(ctypes.Structure):
    _fields_ = [
        ("blob", ctypes.POINTER(ctypes.c_uint8)), ("blob_len", ctypes.c_int64),
        ("dir", ctypes.c_void_p), ("num_chunks", ctypes.c_int64),
        ("series_first", ctypes.POINTER(ctypes.c_int32)),
        ("series_nchunks", ctypes.POINTER(ctypes.c_int32)),
        ("group_ids", ctypes.POINTER(ctypes.c_int32)),
        ("num_series", ctypes.c_int32), ("_pad", ctypes.c_int32),
    ]


_lib = None


def lib():
    global _lib
    if _lib is None:
        if not os.path.exists(LIB_PATH):
            raise RuntimeError(
                f"{LIB_PATH} not built — run `python __graft_entry__.py build` "
                "or __graft_entry__.build()")
        L = ctypes.CDLL(LIB_PATH)
        L.fdb_last_error.restype = ctypes.c_char_p
        L.fdb_store_create.restype = ctypes.c_void_p
        L.fdb_store_create.argtypes = [ctypes.c_int64]
        L.fdb_store_destroy.argtypes = [ctypes.c_void_p]
        L.fdb_store_add_series.argtypes = [ctypes.c_void_p, ctypes.c_int32, ctypes.c_int32]
        L.fdb_series_append.argtypes = [ctypes.c_void_p, ctypes.c_int32,
                                        ctypes.POINTER(ctypes.c_int64), _c_double_p,
                                        ctypes.c_int32]
        L.fdb_series_cut_chunk.argtypes = [ctypes.c_void_p, ctypes.c_int32]
        L.fdb_store_set_max_rows.argtypes = [ctypes.c_void_p, ctypes.c_int32]
        L.fdb_store_seal.argtypes = [ctypes.c_void_p]
        L.fdb_store_num_series.argtypes = [ctypes.c_void_p]
        L.fdb_series_num_chunks.argtypes = [ctypes.c_void_p, ctypes.c_int32]
        L.fdb_chunk_get.argtypes = [ctypes.c_void_p, ctypes.c_int32, ctypes.c_int32,
                                    ctypes.POINTER(ChunkInfo)]
        L.fdb_store_view.argtypes = [ctypes.c_void_p, ctypes.POINTER(View)]
        L.fdb_series_append_hist.argtypes = [ctypes.c_void_p, ctypes.c_int32,
                                             ctypes.POINTER(ctypes.c_int64),
                                             ctypes.POINTER(ctypes.c_uint64),
                                             ctypes.c_int32, ctypes.c_int32,
                                             ctypes.c_double, ctypes.c_double]
        L.fdb_query_exec_hist.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                          ctypes.POINTER(Query), ctypes.c_int32,
                                          _c_double_p, _c_double_p, _c_double_p,
                                          ctypes.c_int32]
        L.fdb_synth_generate.argtypes = [ctypes.c_void_p, ctypes.c_int32, ctypes.c_int32,
                                         ctypes.c_int32, ctypes.c_int64, ctypes.c_int32,
                                         ctypes.c_int32, ctypes.c_double, ctypes.c_double,
                                         ctypes.c_int32, ctypes.c_uint64]
        L.fdb_nibblepack_pack8.argtypes = [ctypes.POINTER(ctypes.c_int64),
                                           ctypes.POINTER(ctypes.c_uint8), ctypes.c_int32]
        L.fdb_nibblepack_unpack8.argtypes = [ctypes.POINTER(ctypes.c_uint8), ctypes.c_int32,
                                             ctypes.POINTER(ctypes.c_int64),
                                             ctypes.POINTER(ctypes.c_int32)]
        L.fdb_nibblepack_pack_delta.argtypes = [ctypes.POINTER(ctypes.c_int64), ctypes.c_int32,
                                                ctypes.POINTER(ctypes.c_uint8), ctypes.c_int32]
        L.fdb_nibblepack_pack_doubles.argtypes = [_c_double_p, ctypes.c_int32,
                                                  ctypes.POINTER(ctypes.c_uint8), ctypes.c_int32]
        L.fdb_engine_create.restype = ctypes.c_void_p
        L.fdb_engine_create.argtypes = [ctypes.c_int32]
        L.fdb_engine_destroy.argtypes = [ctypes.c_void_p]
        L.fdb_engine_synchronize.argtypes = [ctypes.c_void_p]
        L.fdb_dataset_upload.restype = ctypes.c_void_p
        L.fdb_dataset_upload.argtypes = [ctypes.c_void_p, ctypes.c_void_p]
        L.fdb_dataset_destroy.argtypes = [ctypes.c_void_p]
        L.fdb_dataset_bytes.restype = ctypes.c_int64
        L.fdb_dataset_bytes.argtypes = [ctypes.c_void_p]
        L.fdb_dataset_samples.restype = ctypes.c_int64
        L.fdb_dataset_samples.argtypes = [ctypes.c_void_p]
        L.fdb_query_exec.argtypes = [ctypes.c_void_p, ctypes.c_void_p, ctypes.POINTER(Query),
                                     _c_double_p, _c_double_p, ctypes.c_int32]
        L.fdb_query_bench.argtypes = [ctypes.c_void_p, ctypes.c_void_p, ctypes.POINTER(Query),
                                      _c_double_p, _c_double_p, ctypes.c_int32,
                                      ctypes.c_int32, ctypes.c_int32, _c_double_p]
        L.fdb_series_append_hist_mm.argtypes = [
            ctypes.c_void_p, ctypes.c_int32, ctypes.POINTER(ctypes.c_int64),
            ctypes.POINTER(ctypes.c_uint64), _c_double_p, _c_double_p,
            ctypes.c_int32, ctypes.c_int32, ctypes.c_double, ctypes.c_double]
        L.fdb_query_exec_hist_mm.argtypes = [
            ctypes.c_void_p, ctypes.c_void_p, ctypes.POINTER(Query),
            ctypes.c_int32, _c_double_p, _c_double_p, _c_double_p,
            _c_double_p, _c_double_p, ctypes.c_int32]
        L.fdb_brv2_builder_create.restype = ctypes.c_void_p
        L.fdb_brv2_builder_create.argtypes = [ctypes.c_int64]
        L.fdb_brv2_builder_destroy.argtypes = [ctypes.c_void_p]
        L.fdb_brv2_add_record.argtypes = [
            ctypes.c_void_p, ctypes.c_int64, ctypes.c_double, ctypes.c_char_p,
            ctypes.POINTER(ctypes.c_char_p), ctypes.c_int32, ctypes.c_int32]
        L.fdb_brv2_finish.argtypes = [ctypes.c_void_p,
                                      ctypes.POINTER(ctypes.c_uint8), ctypes.c_int32]
        L.fdb_brv2_read.argtypes = [
            ctypes.POINTER(ctypes.c_uint8), ctypes.c_int32, ctypes.c_int32,
            ctypes.POINTER(ctypes.c_int64), _c_double_p,
            ctypes.POINTER(ctypes.c_uint8), ctypes.c_int32,
            ctypes.POINTER(ctypes.c_int32), ctypes.POINTER(ctypes.c_int32),
            ctypes.POINTER(ctypes.c_int32)]
        L.fdb_brv2_index_create.restype = ctypes.c_void_p
        L.fdb_brv2_index_destroy.argtypes = [ctypes.c_void_p]
        L.fdb_store_ingest_brv2.argtypes = [
            ctypes.c_void_p, ctypes.c_void_p, ctypes.POINTER(ctypes.c_uint8),
            ctypes.c_int32, ctypes.c_int32, ctypes.POINTER(ctypes.c_int32)]
        L.fdb_chunkid.restype = ctypes.c_int64
        L.fdb_chunkid.argtypes = [ctypes.c_int64, ctypes.c_int64]
        L.fdb_chunkid_start_time.restype = ctypes.c_int64
        L.fdb_chunkid_start_time.argtypes = [ctypes.c_int64]
        L.fdb_store_persist.argtypes = [
            ctypes.c_void_p, ctypes.c_int32, ctypes.POINTER(ctypes.c_uint8),
            ctypes.c_int32, ctypes.c_int64, ctypes.POINTER(ctypes.c_uint8),
            ctypes.c_int32, ctypes.POINTER(ctypes.c_int32)]
        L.fdb_store_restore.argtypes = [
            ctypes.c_void_p, ctypes.c_void_p, ctypes.POINTER(ctypes.c_uint8),
            ctypes.c_int32, ctypes.c_int32, ctypes.POINTER(ctypes.c_int32)]
        L.fdb_store_add_encoded_chunk.argtypes = [
            ctypes.c_void_p, ctypes.c_int32, ctypes.POINTER(ctypes.c_uint8),
            ctypes.c_int32, ctypes.POINTER(ctypes.c_uint8), ctypes.c_int32,
            ctypes.c_int32, ctypes.c_int64, ctypes.c_int64]
        L.fdb_series_append_sc.argtypes = [
            ctypes.c_void_p, ctypes.c_int32, ctypes.POINTER(ctypes.c_int64),
            _c_double_p, _c_double_p, ctypes.c_int32]
        L.fdb_query_exec_avg_sc.argtypes = [
            ctypes.c_void_p, ctypes.c_void_p, ctypes.POINTER(Query),
            _c_double_p, ctypes.c_int32]
        L.fdb_gpu_encode_chunks.argtypes = [
            ctypes.c_void_p, ctypes.POINTER(ctypes.c_int64), _c_double_p,
            ctypes.POINTER(ctypes.c_int64), ctypes.c_int32, ctypes.c_int32,
            ctypes.POINTER(ctypes.c_uint8), ctypes.c_int64,
            ctypes.POINTER(ctypes.c_int64), ctypes.POINTER(ctypes.c_int64),
            ctypes.POINTER(ctypes.c_int32), ctypes.POINTER(ctypes.c_int32)]
        L.fdb_gpu_unpack_doubles_xor.argtypes = [
            ctypes.c_void_p, ctypes.POINTER(ctypes.c_uint8), ctypes.c_int64,
            ctypes.POINTER(ctypes.c_int64), ctypes.POINTER(ctypes.c_int32),
            ctypes.POINTER(ctypes.c_int64), ctypes.c_int32, _c_double_p,
            ctypes.c_int64]
        L.fdb_nibblepack_unpack_doubles.argtypes = [
            ctypes.POINTER(ctypes.c_uint8), ctypes.c_int32, _c_double_p,
            ctypes.c_int32]
        L.fdb_query_exec_count_values.argtypes = [
            ctypes.c_void_p, ctypes.c_void_p, ctypes.POINTER(Query),
            ctypes.c_int32, _c_double_p, _c_double_p,
            ctypes.POINTER(ctypes.c_int32)]
        _lib = L
    return _lib


def _err():
    return lib().fdb_last_error().decode()


def _check(rc, what):
    if rc < 0:
        raise RuntimeError(f"{what} failed (rc={rc}): {_err()}")
    return rc


def _as_f64_ptr(x):
    if x is None:
        return None
    if isinstance(x, np.ndarray):
        assert x.dtype == np.float64 and x.flags["C_CONTIGUOUS"]
        return x.ctypes.data_as(_c_double_p)
    # torch tensor (device pointer) — duck-typed to avoid importing torch here
    return ctypes.cast(ctypes.c_void_p(x.data_ptr()), _c_double_p)


class ChunkStore:
    """Host-side chunk store: encodes samples into the reference's frozen chunk
    format (TimeSeriesPartition.switchBuffers(encode=true) equivalent)."""

    def __init__(self, expected_series=0):
        self._h = lib().fdb_store_create(expected_series)

    def __del__(self):
        if getattr(self, "_h", None):
            try:
                lib().fdb_store_destroy(self._h)
            except (TypeError, AttributeError):
                pass  # interpreter shutdown: ctypes globals already torn down
            self._h = None

    def set_max_rows(self, n):
        _check(lib().fdb_store_set_max_rows(self._h, n), "set_max_rows")

    def add_series(self, group_id=0, kind=COL_GAUGE):
        return _check(lib().fdb_store_add_series(self._h, group_id, kind), "add_series")

    def append(self, sid, ts, vals):
        ts = np.ascontiguousarray(ts, dtype=np.int64)
        vals = np.ascontiguousarray(vals, dtype=np.float64)
        assert len(ts) == len(vals)
        _check(lib().fdb_series_append(
            self._h, sid, ts.ctypes.data_as(ctypes.POINTER(ctypes.c_int64)),
            vals.ctypes.data_as(_c_double_p), len(ts)), "append")

    def append_sc(self, sid, ts, sums, counts):
        """Downsampled rows: per-row pre-aggregated sum + sample count (the
        count column rides the chunk's companion slot)."""
        ts = np.ascontiguousarray(ts, dtype=np.int64)
        sums = np.ascontiguousarray(sums, dtype=np.float64)
        counts = np.ascontiguousarray(counts, dtype=np.float64)
        assert len(ts) == len(sums) == len(counts)
        _check(lib().fdb_series_append_sc(
            self._h, sid, ts.ctypes.data_as(ctypes.POINTER(ctypes.c_int64)),
            sums.ctypes.data_as(_c_double_p),
            counts.ctypes.data_as(_c_double_p), len(ts)), "append_sc")

    def synth_generate(self, kind, n_series, n_samples, start_ts=100000,
                       step_ms=15000, jitter_ms=250, lam=10.0, reset_p=0.001,
                       n_groups=1, seed=42):
        _check(lib().fdb_synth_generate(self._h, kind, n_series, n_samples,
                                        start_ts, step_ms, jitter_ms, lam, reset_p,
                                        n_groups, seed), "synth_generate")

    def append_hist(self, sid, ts, bucket_values, first=2.0, mult=2.0):
        ts = np.ascontiguousarray(ts, dtype=np.int64)
        bv = np.ascontiguousarray(bucket_values, dtype=np.uint64)
        n, nb = bv.shape
        assert len(ts) == n
        _check(lib().fdb_series_append_hist(
            self._h, sid, ts.ctypes.data_as(ctypes.POINTER(ctypes.c_int64)),
            bv.ctypes.data_as(ctypes.POINTER(ctypes.c_uint64)), n, nb,
            first, mult), "append_hist")

    def append_hist_mm(self, sid, ts, bucket_values, maxs, mins,
                       first=2.0, mult=2.0):
        ts = np.ascontiguousarray(ts, dtype=np.int64)
        bv = np.ascontiguousarray(bucket_values, dtype=np.uint64)
        mx = np.ascontiguousarray(maxs, dtype=np.float64)
        mn = np.ascontiguousarray(mins, dtype=np.float64)
        n, nb = bv.shape
        _check(lib().fdb_series_append_hist_mm(
            self._h, sid, ts.ctypes.data_as(ctypes.POINTER(ctypes.c_int64)),
            bv.ctypes.data_as(ctypes.POINTER(ctypes.c_uint64)),
            _as_f64_ptr(mx), _as_f64_ptr(mn), n, nb, first, mult),
            "append_hist_mm")

    def cut_chunk(self, sid):
        _check(lib().fdb_series_cut_chunk(self._h, sid), "cut_chunk")

    def seal(self):
        _check(lib().fdb_store_seal(self._h), "seal")

    @property
    def num_series(self):
        return lib().fdb_store_num_series(self._h)

    def num_chunks(self, sid):
        return _check(lib().fdb_series_num_chunks(self._h, sid), "num_chunks")

    def chunk(self, sid, idx):
        """Returns (ts_bytes, val_bytes, num_rows, start_time, end_time)."""
        info = ChunkInfo()
        _check(lib().fdb_chunk_get(self._h, sid, idx, ctypes.byref(info)), "chunk_get")
        tsb = ctypes.string_at(info.ts_vec, info.ts_vec_len)
        vab = ctypes.string_at(info.val_vec, info.val_vec_len)
        return tsb, vab, info.num_rows, info.start_time, info.end_time

    def view(self):
        v = View()
        _check(lib().fdb_store_view(self._h, ctypes.byref(v)), "store_view")
        return v


class Engine:
    """GPU engine (fails without a HIP device — no CPU fallback)."""

    def __init__(self, device=0):
        h = lib().fdb_engine_create(device)
        if not h:
            raise RuntimeError(f"fdb_engine_create failed: {_err()}")
        self._h = h

    def __del__(self):
        if getattr(self, "_h", None):
            try:
                lib().fdb_engine_destroy(self._h)
            except (TypeError, AttributeError):
                pass
            self._h = None

    def upload(self, store: ChunkStore):
        d = lib().fdb_dataset_upload(self._h, store._h)
        if not d:
            raise RuntimeError(f"fdb_dataset_upload failed: {_err()}")
        ds = Dataset(d)
        ds.num_series = store.num_series
        return ds

    def query_avg_sc(self, dataset, q: Query, out=None, on_device=False):
        """avg over downsampled sum+count columns (AvgWithSumAndCountOverTime):
        SumOverTime(sum)/SumOverTime(count) per window."""
        nw = q.num_windows
        if out is None:
            out = np.empty(dataset.num_series * nw, dtype=np.float64)
        ptr = (out.ctypes.data_as(_c_double_p) if isinstance(out, np.ndarray)
               else ctypes.cast(out.data_ptr(), _c_double_p))
        _check(lib().fdb_query_exec_avg_sc(self._h, dataset._h, q, ptr,
                                           1 if on_device else 0), "avg_sc")
        return out

    def query(self, dataset, q: Query, out=None, out_counts=None, on_device=False):
        """Executes one (shard, query). Returns `out` (allocated as numpy when None
        and on_device is False)."""
        if out is None:
            assert not on_device
            n = dataset_out_len(dataset, q, partial=out_counts is not None)
            out = np.empty(n, dtype=np.float64)
        rc = lib().fdb_query_exec(self._h, dataset._h, ctypes.byref(q),
                                  _as_f64_ptr(out), _as_f64_ptr(out_counts),
                                  1 if on_device else 0)
        _check(rc, "query_exec")
        return out

    def bench(self, dataset, q: Query, out, out_counts=None, on_device=False,
              warmup=2, iters=10):
        ms = ctypes.c_double()
        rc = lib().fdb_query_bench(self._h, dataset._h, ctypes.byref(q),
                                   _as_f64_ptr(out), _as_f64_ptr(out_counts),
                                   1 if on_device else 0, warmup, iters,
                                   ctypes.byref(ms))
        _check(rc, "query_bench")
        return ms.value

    def query_hist(self, dataset, q: Query, num_buckets,
                   out_bucket_sums=None, out_counts=None, out_quantile=None,
                   on_device=False):
        rc = lib().fdb_query_exec_hist(
            self._h, dataset._h, ctypes.byref(q), num_buckets,
            _as_f64_ptr(out_bucket_sums), _as_f64_ptr(out_counts),
            _as_f64_ptr(out_quantile), 1 if on_device else 0)
        _check(rc, "query_exec_hist")

    def query_hist_mm(self, dataset, q: Query, num_buckets,
                      out_bucket_sums=None, out_counts=None, out_max=None,
                      out_min=None, out_quantile=None, on_device=False):
        rc = lib().fdb_query_exec_hist_mm(
            self._h, dataset._h, ctypes.byref(q), num_buckets,
            _as_f64_ptr(out_bucket_sums), _as_f64_ptr(out_counts),
            _as_f64_ptr(out_max), _as_f64_ptr(out_min),
            _as_f64_ptr(out_quantile), 1 if on_device else 0)
        _check(rc, "query_exec_hist_mm")

    def count_values(self, dataset, q: Query, k_cap=64):
        """CountValuesRowAggregator: per (group, window) distinct values with
        frequencies, sorted ascending. Returns (values, counts, n) with
        values/counts shaped [G*W, k_cap] and n [G*W]."""
        nw = q.num_windows
        cells = q.num_groups * nw
        vals = np.zeros(cells * k_cap, dtype=np.float64)
        cnts = np.zeros(cells * k_cap, dtype=np.float64)
        n = np.zeros(cells, dtype=np.int32)
        rc = lib().fdb_query_exec_count_values(
            self._h, dataset._h, ctypes.byref(q), k_cap,
            _as_f64_ptr(vals), _as_f64_ptr(cnts),
            n.ctypes.data_as(ctypes.POINTER(ctypes.c_int32)))
        _check(rc, "count_values")
        return vals.reshape(cells, k_cap), cnts.reshape(cells, k_cap), n

    def synchronize(self):
        _check(lib().fdb_engine_synchronize(self._h), "synchronize")


class Dataset:
    def __init__(self, h):
        self._h = h

    def __del__(self):
        if getattr(self, "_h", None):
            try:
                lib().fdb_dataset_destroy(self._h)
            except (TypeError, AttributeError):
                pass
            self._h = None

    @property
    def payload_bytes(self):
        return lib().fdb_dataset_bytes(self._h)

    @property
    def total_samples(self):
        return lib().fdb_dataset_samples(self._h)


def dataset_out_len(dataset, q: Query, partial=False):
    """Doubles the engine copies into `out` for this query (run_query buf_len):
    G*W*k for top/bottom-k (k = q.param), 2*G*W for the stacked stddev/stdvar
    partial grids (raw sums + sumsq) when out_counts is passed, else G*W."""
    if q.agg_id == AGG_NONE:
        # series count is not stored on Dataset; caller usually knows it
        raise ValueError("pass an explicit out buffer for AGG_NONE queries")
    n = q.num_groups * q.num_windows
    if q.agg_id in (AGG_TOPK, AGG_BOTTOMK):
        return n * max(1, int(q.param))
    if partial and q.agg_id in (AGG_STDDEV, AGG_STDVAR):
        return 2 * n
    return n


def make_query(start, step, end, window, func_id, agg_id=AGG_NONE, num_groups=0,
               param=0.0, param2=0.0):
    q = Query()
    q.start, q.step, q.end, q.window = start, step, end, window
    q.func_id, q.agg_id, q.num_groups = func_id, agg_id, num_groups
    q.param = param
    q.param2 = param2
    return q


def nibblepack_unpack_doubles(data, n):
    """Host-side unpackDoubleXOR (NibblePack.scala:360-394)."""
    buf = (ctypes.c_uint8 * len(data)).from_buffer_copy(data)
    out = np.empty(n, dtype=np.float64)
    _check(lib().fdb_nibblepack_unpack_doubles(buf, len(data), _as_f64_ptr(out), n),
           "unpack_doubles")
    return out


def gpu_unpack_doubles_xor(engine, streams):
    """Decodes a list of packed XOR streams [(bytes, n), ...] on the GPU
    (wavefront prefix-XOR kernel). Returns a list of numpy arrays."""
    blob = b"".join(s for s, _ in streams)
    offs, counts, out_offs = [], [], []
    pos = 0
    opos = 0
    for s, n in streams:
        offs.append(pos)
        counts.append(n)
        out_offs.append(opos)
        pos += len(s)
        opos += n
    b = (ctypes.c_uint8 * len(blob)).from_buffer_copy(blob)
    offs_a = np.array(offs, dtype=np.int64)
    cnts_a = np.array(counts, dtype=np.int32)
    oo_a = np.array(out_offs, dtype=np.int64)
    out = np.empty(opos, dtype=np.float64)
    _check(lib().fdb_gpu_unpack_doubles_xor(
        engine._h, b, len(blob),
        offs_a.ctypes.data_as(ctypes.POINTER(ctypes.c_int64)),
        cnts_a.ctypes.data_as(ctypes.POINTER(ctypes.c_int32)),
        oo_a.ctypes.data_as(ctypes.POINTER(ctypes.c_int64)),
        len(streams), _as_f64_ptr(out), opos), "gpu_unpack_doubles_xor")
    return [out[o:o + n] for o, n in zip(out_offs, counts)]


def nibblepack_pack8(vals8):
    a = np.ascontiguousarray(vals8, dtype=np.int64)
    assert len(a) == 8
    out = (ctypes.c_uint8 * 64)()
    n = _check(lib().fdb_nibblepack_pack8(
        a.ctypes.data_as(ctypes.POINTER(ctypes.c_int64)), out, 64), "pack8")
    return bytes(out[:n])


def nibblepack_unpack8(data):
    buf = (ctypes.c_uint8 * len(data)).from_buffer_copy(data)
    out = (ctypes.c_int64 * 8)()
    consumed = ctypes.c_int32()
    _check(lib().fdb_nibblepack_unpack8(buf, len(data), out, ctypes.byref(consumed)),
           "unpack8")
    return list(out), consumed.value


class BRv2Builder:
    """BinaryRecord v2 ingestion-container builder (layout per
    RecordContainer.scala / RecordSchema.scala / RecordBuilder.scala)."""

    def __init__(self, ts_header=0):
        self._h = lib().fdb_brv2_builder_create(ts_header)

    def __del__(self):
        if getattr(self, "_h", None):
            try:
                lib().fdb_brv2_builder_destroy(self._h)
            except (TypeError, AttributeError):
                pass
            self._h = None

    def add(self, ts, value, metric, tags=None, schema_id=1):
        tags = tags or {}
        arr = (ctypes.c_char_p * (2 * len(tags)))()
        for i, (k, v) in enumerate(tags.items()):
            arr[2 * i] = k.encode()
            arr[2 * i + 1] = v.encode()
        _check(lib().fdb_brv2_add_record(self._h, ts, value, metric.encode(),
                                         arr, len(tags), schema_id), "brv2_add")

    def finish(self):
        cap = 1 << 22
        out = (ctypes.c_uint8 * cap)()
        n = _check(lib().fdb_brv2_finish(self._h, out, cap), "brv2_finish")
        return bytes(out[:n])


def brv2_read(container, idx):
    """Returns (ts, value, partkey_bytes, schema_id, part_hash) of record idx."""
    buf = (ctypes.c_uint8 * len(container)).from_buffer_copy(container)
    ts = ctypes.c_int64()
    val = ctypes.c_double()
    pk = (ctypes.c_uint8 * 4096)()
    pk_len = ctypes.c_int32()
    sid = ctypes.c_int32()
    ph = ctypes.c_int32()
    _check(lib().fdb_brv2_read(buf, len(container), idx, ctypes.byref(ts),
                               ctypes.byref(val), pk, 4096,
                               ctypes.byref(pk_len), ctypes.byref(sid),
                               ctypes.byref(ph)), "brv2_read")
    return ts.value, val.value, bytes(pk[:pk_len.value]), sid.value, ph.value


class BRv2Index:
    """Partition-key → series index for container ingestion."""

    def __init__(self):
        self._h = lib().fdb_brv2_index_create()

    def __del__(self):
        if getattr(self, "_h", None):
            try:
                lib().fdb_brv2_index_destroy(self._h)
            except (TypeError, AttributeError):
                pass
            self._h = None


def ingest_brv2(store, index, container, kind=COL_GAUGE):
    """Consumes a BinaryRecord v2 container into the chunk store. Returns
    (num_records, num_new_series)."""
    buf = (ctypes.c_uint8 * len(container)).from_buffer_copy(container)
    new_series = ctypes.c_int32()
    n = _check(lib().fdb_store_ingest_brv2(store._h, index._h, buf,
                                           len(container), kind,
                                           ctypes.byref(new_series)),
               "ingest_brv2")
    return n, new_series.value


def chunkid(start_time, ingestion_time):
    """Chunk-table row key packing (core/.../store/package.scala:112-123)."""
    return lib().fdb_chunkid(start_time, ingestion_time)


def chunkid_start_time(cid):
    return lib().fdb_chunkid_start_time(cid)


def persist_series(store, sid, partkey, ingestion_time=0):
    """Serializes every chunk of a series as Cassandra chunk-table rows
    (TimeSeriesChunksTable.scala row shape in a flat framing). Returns the
    bytes."""
    pk = (ctypes.c_uint8 * max(1, len(partkey))).from_buffer_copy(
        partkey or b"\x00")
    cap = 1 << 24
    out = (ctypes.c_uint8 * cap)()
    out_len = ctypes.c_int32()
    _check(lib().fdb_store_persist(store._h, sid, pk, len(partkey),
                                   ingestion_time, out, cap,
                                   ctypes.byref(out_len)), "persist")
    return bytes(out[:out_len.value])


def restore_rows(store, index, rows, kind=COL_GAUGE):
    """Restores persisted chunk-table rows into an unsealed store (frozen
    vector bytes reused unchanged). Returns the number of rows consumed."""
    buf = (ctypes.c_uint8 * len(rows)).from_buffer_copy(rows)
    nrows = ctypes.c_int32()
    _check(lib().fdb_store_restore(store._h, index._h, buf, len(rows), kind,
                                   ctypes.byref(nrows)), "restore")
    return nrows.value


def add_encoded_chunk(store, sid, ts_bytes, val_bytes, num_rows,
                      start_time, end_time):
    """Appends a pre-encoded (frozen-bytes) chunk to an unsealed store."""
    tb = (ctypes.c_uint8 * len(ts_bytes)).from_buffer_copy(ts_bytes)
    vb = (ctypes.c_uint8 * len(val_bytes)).from_buffer_copy(val_bytes)
    _check(lib().fdb_store_add_encoded_chunk(store._h, sid, tb, len(ts_bytes),
                                             vb, len(val_bytes), num_rows,
                                             start_time, end_time),
           "add_encoded_chunk")


def gpu_encode_chunks(engine, chunks, kind=COL_GAUGE):
    """GPU ingest encoder: chunks = list of (ts_i64_array, vals_f64_array).
    Returns a list of (ts_bytes, val_bytes) — byte-identical to the host
    encoder's frozen vectors."""
    import numpy as _np
    nch = len(chunks)
    row_offs = _np.zeros(nch + 1, dtype=_np.int64)
    for i, (t, v) in enumerate(chunks):
        assert len(t) == len(v)
        row_offs[i + 1] = row_offs[i] + len(t)
    ts = _np.concatenate([_np.asarray(t, dtype=_np.int64) for t, _ in chunks])
    vals = _np.concatenate([_np.asarray(v, dtype=_np.float64)
                            for _, v in chunks])
    cap = int(sum(2 * ((len(t) * 8 + 64 + 63) & ~63) for t, _ in chunks))
    out = _np.zeros(cap, dtype=_np.uint8)
    toff = _np.zeros(nch, dtype=_np.int64)
    voff = _np.zeros(nch, dtype=_np.int64)
    tlen = _np.zeros(nch, dtype=_np.int32)
    vlen = _np.zeros(nch, dtype=_np.int32)
    _check(lib().fdb_gpu_encode_chunks(
        engine._h,
        ts.ctypes.data_as(ctypes.POINTER(ctypes.c_int64)),
        vals.ctypes.data_as(_c_double_p),
        row_offs.ctypes.data_as(ctypes.POINTER(ctypes.c_int64)),
        nch, kind,
        out.ctypes.data_as(ctypes.POINTER(ctypes.c_uint8)), cap,
        toff.ctypes.data_as(ctypes.POINTER(ctypes.c_int64)),
        voff.ctypes.data_as(ctypes.POINTER(ctypes.c_int64)),
        tlen.ctypes.data_as(ctypes.POINTER(ctypes.c_int32)),
        vlen.ctypes.data_as(ctypes.POINTER(ctypes.c_int32))), "gpu_encode")
    res = []
    for c in range(nch):
        res.append((out[toff[c]:toff[c] + tlen[c]].tobytes(),
                    out[voff[c]:voff[c] + vlen[c]].tobytes()))
    return res
