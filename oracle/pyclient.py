"""ctypes client for the TEST-ONLY CPU oracle (oracle/liboracle.so).

May be imported only by tests/, __graft_entry__.smoke() and bench.py's
cpu_baseline leg. The product GPU path never touches this library.
"""
import ctypes
import os

import numpy as np

_DIR = os.path.dirname(os.path.abspath(__file__))
LIB_PATH = os.path.join(_DIR, "liboracle.so")

_lib = None
_c_double_p = ctypes.POINTER(ctypes.c_double)


def lib():
    global _lib
    if _lib is None:
        if not os.path.exists(LIB_PATH):
            raise RuntimeError(f"{LIB_PATH} not built — run `make -C oracle`")
        L = ctypes.CDLL(LIB_PATH)
        L.oracle_query_exec.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                        _c_double_p, _c_double_p, ctypes.c_int32]
        L.oracle_eval_series.argtypes = [ctypes.c_void_p, ctypes.c_int32,
                                         ctypes.c_void_p, _c_double_p]
        L.oracle_decode_longs.argtypes = [ctypes.POINTER(ctypes.c_uint8),
                                          ctypes.POINTER(ctypes.c_int64), ctypes.c_int32]
        L.oracle_decode_doubles.argtypes = [ctypes.POINTER(ctypes.c_uint8),
                                            _c_double_p, ctypes.c_int32]
        L.oracle_vec_info.argtypes = [ctypes.POINTER(ctypes.c_uint8)] + \
            [ctypes.POINTER(ctypes.c_int32)] * 4
        L.oracle_binary_search.argtypes = [ctypes.POINTER(ctypes.c_uint8), ctypes.c_int64]
        L.oracle_nibblepack_unpack8.argtypes = [ctypes.POINTER(ctypes.c_uint8), ctypes.c_int32,
                                                ctypes.POINTER(ctypes.c_int64),
                                                ctypes.POINTER(ctypes.c_int32)]
        L.oracle_extrapolated_rate.restype = ctypes.c_double
        L.oracle_extrapolated_rate.argtypes = [ctypes.c_int64, ctypes.c_int64, ctypes.c_int32,
                                               ctypes.c_int64, ctypes.c_double,
                                               ctypes.c_int64, ctypes.c_double,
                                               ctypes.c_int32, ctypes.c_int32]
        _lib = L
    return _lib


def query_exec_hist(view, q, num_buckets, with_quantile=True):
    import ctypes as ct
    nw = q.num_windows
    ng = q.num_groups
    sums = np.zeros(ng * nw * num_buckets, dtype=np.float64)
    cnts = np.zeros(ng * nw, dtype=np.float64)
    quant = np.zeros(ng * nw, dtype=np.float64) if with_quantile else None
    L = lib()
    L.oracle_query_exec_hist.argtypes = [ct.c_void_p, ct.c_void_p, ct.c_int32,
                                         _c_double_p, _c_double_p, _c_double_p,
                                         ct.c_int32]
    rc = L.oracle_query_exec_hist(
        ct.byref(view), ct.byref(q), num_buckets,
        sums.ctypes.data_as(_c_double_p), cnts.ctypes.data_as(_c_double_p),
        quant.ctypes.data_as(_c_double_p) if quant is not None else None, 1)
    if rc != 0:
        raise RuntimeError("oracle_query_exec_hist failed")
    return sums, cnts, quant


def hist_decode(vec_bytes, cap=200000):
    import ctypes as ct
    out = np.empty(cap, dtype=np.int64)
    n = ct.c_int32(); nb = ct.c_int32()
    L = lib()
    L.oracle_hist_decode.argtypes = [ct.POINTER(ct.c_uint8), ct.POINTER(ct.c_int64),
                                     ct.c_int32, ct.POINTER(ct.c_int32),
                                     ct.POINTER(ct.c_int32)]
    rc = L.oracle_hist_decode(_u8(vec_bytes),
                              out.ctypes.data_as(ct.POINTER(ct.c_int64)), cap,
                              ct.byref(n), ct.byref(nb))
    if rc != 0:
        raise RuntimeError("hist_decode failed")
    return out[:n.value * nb.value].reshape(n.value, nb.value).copy()


def hist_corrections(vec_bytes, cap=200000):
    """Cumulative in-chunk TypeDrop corrections at each element [n × nb]."""
    import ctypes as ct
    dec = hist_decode(vec_bytes, cap)     # for the shape
    out = np.empty(dec.size, dtype=np.int64)
    L = lib()
    L.oracle_hist_corrections.argtypes = [ct.POINTER(ct.c_uint8),
                                          ct.POINTER(ct.c_int64), ct.c_int32]
    rc = L.oracle_hist_corrections(_u8(vec_bytes),
                                   out.ctypes.data_as(ct.POINTER(ct.c_int64)),
                                   dec.size)
    if rc != 0:
        raise RuntimeError("hist_corrections failed")
    return out.reshape(dec.shape).copy()


def hist_quantile(q, values, first, mult):
    import ctypes as ct
    L = lib()
    L.oracle_hist_quantile.restype = ct.c_double
    L.oracle_hist_quantile.argtypes = [ct.c_double, _c_double_p, ct.c_int32,
                                       ct.c_double, ct.c_double]
    v = np.ascontiguousarray(values, dtype=np.float64)
    return L.oracle_hist_quantile(q, v.ctypes.data_as(_c_double_p), len(v), first, mult)


def query_exec_hist_mm(view, q, num_buckets, with_quantile=True):
    import ctypes as ct
    nw = q.num_windows
    ng = q.num_groups
    sums = np.zeros(ng * nw * num_buckets, dtype=np.float64)
    cnts = np.zeros(ng * nw, dtype=np.float64)
    mx = np.zeros(ng * nw, dtype=np.float64)
    mn = np.zeros(ng * nw, dtype=np.float64)
    quant = np.zeros(ng * nw, dtype=np.float64) if with_quantile else None
    L = lib()
    L.oracle_query_exec_hist_mm.argtypes = [ct.c_void_p, ct.c_void_p,
                                            ct.c_int32, _c_double_p,
                                            _c_double_p, _c_double_p,
                                            _c_double_p, _c_double_p]
    rc = L.oracle_query_exec_hist_mm(
        ct.byref(view), ct.byref(q), num_buckets,
        sums.ctypes.data_as(_c_double_p), cnts.ctypes.data_as(_c_double_p),
        mx.ctypes.data_as(_c_double_p), mn.ctypes.data_as(_c_double_p),
        quant.ctypes.data_as(_c_double_p) if quant is not None else None)
    if rc != 0:
        raise RuntimeError(f"oracle_query_exec_hist_mm failed rc={rc}")
    return sums, cnts, mx, mn, quant


def count_values(view, q, k_cap=64):
    import ctypes as ct
    nw = q.num_windows
    cells = q.num_groups * nw
    vals = np.zeros(cells * k_cap, dtype=np.float64)
    cnts = np.zeros(cells * k_cap, dtype=np.float64)
    n = np.zeros(cells, dtype=np.int32)
    L = lib()
    L.oracle_count_values.argtypes = [ct.c_void_p, ct.c_void_p, ct.c_int32,
                                      _c_double_p, _c_double_p,
                                      ct.POINTER(ct.c_int32)]
    rc = L.oracle_count_values(ct.byref(view), ct.byref(q), k_cap,
                               vals.ctypes.data_as(_c_double_p),
                               cnts.ctypes.data_as(_c_double_p),
                               n.ctypes.data_as(ct.POINTER(ct.c_int32)))
    if rc != 0:
        raise RuntimeError(f"oracle_count_values failed rc={rc}")
    return vals.reshape(cells, k_cap), cnts.reshape(cells, k_cap), n


def query_exec(view, q, num_series, num_windows, out_counts=False, nthreads=1):
    """Runs the oracle over a store view (filodb_amd.View) with query q
    (filodb_amd.Query). Returns the result grid as numpy."""
    if q.agg_id == 0:
        out = np.empty(num_series * num_windows, dtype=np.float64)
        cnt = None
    else:
        cells = q.num_groups * num_windows
        if q.agg_id in (6, 7):        # top/bottom-k: [G × W × k] values + ids
            cells *= int(q.param)
        buf = cells
        if out_counts and q.agg_id in (8, 9):
            buf = cells * 2           # stddev partials: (sums, sumsq) stacked
        out = np.empty(buf, dtype=np.float64)
        cnt = np.zeros(cells, dtype=np.float64) if out_counts else None
    rc = lib().oracle_query_exec(
        ctypes.byref(view), ctypes.byref(q),
        out.ctypes.data_as(_c_double_p),
        cnt.ctypes.data_as(_c_double_p) if cnt is not None else None,
        nthreads)
    if rc != 0:
        raise RuntimeError("oracle_query_exec failed")
    return (out, cnt) if out_counts else out


def eval_series(view, sid, q, num_windows):
    out = np.empty(num_windows, dtype=np.float64)
    rc = lib().oracle_eval_series(ctypes.byref(view), sid, ctypes.byref(q),
                                  out.ctypes.data_as(_c_double_p))
    if rc != 0:
        raise RuntimeError("oracle_eval_series failed")
    return out


def _u8(data):
    return (ctypes.c_uint8 * len(data)).from_buffer_copy(data)


def decode_longs(vec_bytes, cap=100000):
    out = np.empty(cap, dtype=np.int64)
    n = lib().oracle_decode_longs(_u8(vec_bytes),
                                  out.ctypes.data_as(ctypes.POINTER(ctypes.c_int64)), cap)
    if n < 0:
        raise RuntimeError("decode_longs failed")
    return out[:n].copy()


def decode_doubles(vec_bytes, cap=100000):
    out = np.empty(cap, dtype=np.float64)
    n = lib().oracle_decode_doubles(_u8(vec_bytes), out.ctypes.data_as(_c_double_p), cap)
    if n < 0:
        raise RuntimeError("decode_doubles failed")
    return out[:n].copy()


def vec_info(vec_bytes):
    wf, n, nbits, dropped = (ctypes.c_int32() for _ in range(4))
    lib().oracle_vec_info(_u8(vec_bytes), ctypes.byref(wf), ctypes.byref(n),
                          ctypes.byref(nbits), ctypes.byref(dropped))
    return {"wf": wf.value, "n": n.value, "nbits": nbits.value, "dropped": bool(dropped.value)}


def binary_search(vec_bytes, item):
    return lib().oracle_binary_search(_u8(vec_bytes), item)


def nibblepack_unpack8(data):
    out = (ctypes.c_int64 * 8)()
    consumed = ctypes.c_int32()
    rc = lib().oracle_nibblepack_unpack8(_u8(data), len(data), out, ctypes.byref(consumed))
    if rc != 0:
        raise RuntimeError("unpack8 failed")
    return list(out), consumed.value


def extrapolated_rate(ws, we, n, t1, v1, t2, v2, is_counter, is_rate):
    return lib().oracle_extrapolated_rate(ws, we, n, t1, v1, t2, v2,
                                          1 if is_counter else 0, 1 if is_rate else 0)


def corrected_doubles(vec_bytes, cap=100000):
    """Corrected counter series of one chunk (CorrectingDoubleVectorReader)."""
    import ctypes as ct
    out = np.empty(cap, dtype=np.float64)
    L = lib()
    L.oracle_corrected_doubles.argtypes = [ct.POINTER(ct.c_uint8), _c_double_p,
                                           ct.c_int32]
    n = L.oracle_corrected_doubles(_u8(vec_bytes),
                                   out.ctypes.data_as(_c_double_p), cap)
    if n < 0:
        raise RuntimeError("corrected_doubles failed")
    return out[:n].copy()


def query_exec_avg_sc(view, q, num_series, num_windows):
    """AvgWithSumAndCountOverTimeFuncD composition (AggrOverTimeFunctions.
    scala:820-860): SumOverTime over the sum column / SumOverTime over the
    count column, per window — the count column rides each dir entry's
    max_off slot, so the second pass runs the SAME oracle with the value
    offsets swapped."""
    import ctypes as ct

    class DirEntry(ct.Structure):       # fdb_dir_entry_t (chunk_format.h)
        _fields_ = [("ts_off", ct.c_uint64), ("val_off", ct.c_uint64),
                    ("start_time", ct.c_int64), ("end_time", ct.c_int64),
                    ("num_rows", ct.c_int32), ("_pad", ct.c_int32),
                    ("max_off", ct.c_uint64), ("min_off", ct.c_uint64)]

    nq = type(q)()
    ct.memmove(ct.byref(nq), ct.byref(q), ct.sizeof(q))
    nq.func_id = 3                       # FDB_FN_SUM_OVER_TIME
    sums = query_exec(view, nq, num_series, num_windows)
    n = int(view.num_chunks)
    src = ct.cast(view.dir, ct.POINTER(DirEntry))
    swapped = (DirEntry * n)()
    for i in range(n):
        swapped[i] = src[i]
        if swapped[i].max_off == 0:
            raise RuntimeError("chunk %d has no count column" % i)
        swapped[i].val_off = swapped[i].max_off
    view2 = type(view)()
    ct.memmove(ct.byref(view2), ct.byref(view), ct.sizeof(view))
    view2.dir = ct.cast(swapped, ct.c_void_p)
    counts = query_exec(view2, nq, num_series, num_windows)
    return sums / counts
