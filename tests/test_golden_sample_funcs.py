"""Reference-literal goldens for the window-sample functions and changes.

Fixtures and expected values lifted verbatim from the reference's own specs:
  QuantileOverTimeSpec.scala:10-127  (q=0.5 with NaNs → 4928.5; empty → NaN;
                                      constant → 8201; single value → 8201;
                                      q=0.2 → 2.8000000000000003; q=0.9 → 9.1)
  ChangesFunctionSpec.scala:8-77     (normal → 4; empty → NaN; constant → 0;
                                      NaN-padded constant → 0)
  AggrOverTimeFunctionsSpec.scala:608-621 (MAD of [9,6,4,1,1,2,2] → 1.0)

Each spec evaluates one window (startTs, endTs]; here that is a one-window
query with window = endTs - startTs. The "only one value" quantile case
lists its NaN padding out of timestamp order in the spec; the samples are
sorted here (same multiset, same window content).
"""
import numpy as np
import pytest

SPEC_SAMPLES = [
    (8072000, 7419.0), (8082100, np.nan), (8092196, 4614.0),
    (8102215, 4909.0), (8112223, 4909.0), (8122388, 4948.0),
    (8132570, np.nan), (8142822, np.nan), (8152858, np.nan),
    (8162999, 8201.0),
]
SPEC_START, SPEC_END = 8071950, 8163070
RAMP = [(8072000 + [0, 10100, 20196, 30215, 40223, 50388, 60570, 70822,
                    80858, 90999][i], float(i + 1)) for i in range(10)]


def one_window(fdb, oracle, samples, func_id, param=0.0, param2=0.1,
               start=SPEC_START, end=SPEC_END):
    st = fdb.ChunkStore()
    sid = st.add_series(0, fdb.COL_GAUGE)
    ts = np.array([t for t, _ in samples], dtype=np.int64)
    vs = np.array([v for _, v in samples], dtype=np.float64)
    st.append(sid, ts, vs)
    st.seal()
    q = fdb.make_query(end, 10000, end, end - start, func_id)
    q.param = param
    q.param2 = param2
    out = oracle.query_exec(st.view(), q, 1, 1)
    return float(out[0])


def test_quantile_over_time_spec_literals(fdb, oracle):
    FN = fdb.FN_QUANTILE_OVER_TIME
    assert one_window(fdb, oracle, SPEC_SAMPLES, FN, 0.5) == 4928.5
    nan_only = [(8082100, np.nan), (8132570, np.nan),
                (8142822, np.nan), (8152858, np.nan)]
    assert np.isnan(one_window(fdb, oracle, nan_only, FN, 0.5))
    const = [(8082100, 8201.0), (8132570, 8201.0),
             (8142822, 8201.0), (8152858, 8201.0)]
    assert one_window(fdb, oracle, const, FN, 0.2) == 8201.0
    single = [(8082100, np.nan), (8132570, 8201.0),
              (8142822, np.nan), (8152858, np.nan)]
    assert one_window(fdb, oracle, single, FN, 0.2) == 8201.0
    assert one_window(fdb, oracle, RAMP, FN, 0.2) == 2.8000000000000003
    assert one_window(fdb, oracle, RAMP, FN, 0.9) == 9.1


def test_changes_spec_literals(fdb, oracle):
    """ChangesFunctionSpec's normal case expects 4 from the SLIDING function
    (ChangesOverTimeFunction, AggrOverTimeFunctions.scala:360-395, which
    carries lastValue ACROSS NaNs). The chunked path this engine implements
    (DoubleVector.changes, DoubleVector.scala:283-302) sets prev to every
    value including NaN, so a NaN breaks the chain: the same samples give 2
    (7419→NaN→4614 and NaN→8201 are not counted; 4614→4909 and 4909→4948
    are). The NaN-free literals below agree between both variants."""
    FN = fdb.FN_CHANGES
    assert one_window(fdb, oracle, SPEC_SAMPLES, FN) == 2.0
    # all-NaN rows: the chunked accumulator zeroes `changes` for ANY touched
    # chunk range (ChangesChunkedFunctionD, AggrOverTimeFunctions.scala:
    # 1200-1202), so this is 0; NaN needs a truly empty window (below).
    # The spec's NaN expectation is again the sliding variant's.
    nan_only = [(8082100, np.nan), (8132570, np.nan),
                (8142822, np.nan), (8152858, np.nan)]
    assert one_window(fdb, oracle, nan_only, FN) == 0.0
    assert np.isnan(one_window(fdb, oracle, nan_only, FN,
                               start=8000000, end=8001000))
    const = [(8082100, 8201.0), (8132570, 8201.0),
             (8142822, 8201.0), (8152858, 8201.0)]
    assert one_window(fdb, oracle, const, FN) == 0.0
    padded = [(8082100, np.nan), (8132570, 8201.0),
              (8142822, 8201.0), (8152858, np.nan)]
    assert one_window(fdb, oracle, padded, FN) == 0.0


def test_mad_spec_literal(fdb, oracle):
    samples = [(100000 + i * 10000, v)
               for i, v in enumerate([9.0, 6.0, 4.0, 1.0, 1.0, 2.0, 2.0])]
    got = one_window(fdb, oracle, samples, fdb.FN_MAD_OVER_TIME,
                     start=70000, end=170000)
    assert got == pytest.approx(1.0, abs=1e-10)


DELTA_TS = [8072000, 8082100, 8092196, 8102215, 8112223, 8122388, 8132570,
            8142822, 8152858, 8162999]
DELTA_VALS = [111.0, 92.0, 103.0, 110.0, 185.0, 39.0, 52.0, 95.0, 7.0, 99.0]


def test_rate_over_delta_spec_literals(fdb, oracle):
    """PeriodicRateFunctionsSpec.scala:26-80 — RateOverDeltaChunkedFunctionD:
    rate = sum of in-window delta samples / window_ms * 1000."""
    FN = fdb.FN_RATE_OVER_DELTA
    samples = list(zip(DELTA_TS, DELTA_VALS))
    got = one_window(fdb, oracle, samples, FN)
    want = sum(DELTA_VALS) / (SPEC_END - SPEC_START) * 1000
    assert got == pytest.approx(want, abs=1e-7)
    # one-sample window is NOT NaN (:60-68)
    got1 = one_window(fdb, oracle, samples, FN, start=8101215, end=8103215)
    assert not np.isnan(got1)
    assert got1 == pytest.approx(110.0 / 2000 * 1000, abs=1e-7)
    # flat (non-increasing) delta samples still rate > 0 (:69-79)
    flat = [(t, 111.0) for t in DELTA_TS]
    gotf = one_window(fdb, oracle, flat, FN)
    assert gotf != 0.0
    assert gotf == pytest.approx(1110.0 / (SPEC_END - SPEC_START) * 1000,
                                 abs=1e-7)


def _hw_model(arr, sf=0.01, tf=0.1):
    """The reference's own test model (AggrOverTimeFunctionsSpec.scala:
    694-714): s0=arr[0], b0=arr[1]-arr[0], then the smoothing recurrence."""
    if len(arr) < 2:
        return float("nan")
    s0, b0 = arr[0], arr[1] - arr[0]
    for x in arr[1:]:
        s = sf * x + (1 - sf) * (s0 + b0)
        b0 = tf * (s - s0) + (1 - tf) * b0
        s0 = s
    return s0


def test_holt_winters_spec_literals(fdb, oracle):
    """HoltWintersChunkedFunctionD (single-chunk) against the reference
    spec's data and model (AggrOverTimeFunctionsSpec.scala:686-737)."""
    FN = fdb.FN_HOLT_WINTERS
    cases = [
        [15900.0, 15920.0, 15940.0, 15960.0, 15980.0, 16000.0],
        [23850.0, 23880.0, 23910.0, 23940.0, 23970.0, 24000.0],
        [31800.0, 31840.0, 31880.0, 31920.0, 31960.0, 32000.0],
        [-15900.0, -15920.0, -15940.0, -15960.0, -15980.0, -16000.0],
    ]
    for vals in cases:
        samples = [(100000 + i * 10000, v) for i, v in enumerate(vals)]
        got = one_window(fdb, oracle, samples, FN, param=0.01,
                         start=60000, end=160000)
        assert got == pytest.approx(_hw_model(vals), abs=1e-10)
    # < 2 samples → NaN; NaN rows are skipped for the two seeds
    one = [(100000, 5.0)]
    assert np.isnan(one_window(fdb, oracle, one, FN, param=0.01,
                               start=60000, end=160000))
    withnan = [(100000, np.nan), (110000, 10.0), (120000, np.nan),
               (130000, 20.0), (140000, 30.0)]
    got = one_window(fdb, oracle, withnan, FN, param=0.01,
                     start=60000, end=160000)
    assert not np.isnan(got)


@pytest.mark.gpu
def test_holt_winters_gpu_vs_oracle(fdb, oracle):
    """FN 20 through the window-sample kernel: identical operation sequence
    to the oracle, including the NaN seed-scan and the modeled end-of-chunk
    read — bit-equal results across a randomized single-chunk matrix."""
    rng = np.random.default_rng(73)
    st = fdb.ChunkStore()
    n = 240
    for s in range(24):
        ts = (100000 + np.arange(n) * 15000
              + rng.integers(-250, 251, n)).astype(np.int64)
        ts = np.maximum.accumulate(ts)
        vs = np.cumsum(rng.normal(1.0, 3.0, n)) + 100
        vs[rng.random(n) < 0.15] = np.nan
        sid = st.add_series(0, fdb.COL_GAUGE)
        st.append(sid, ts, vs)
    st.seal()
    eng = fdb.Engine(0)
    ds = eng.upload(st)
    for sf, tf, step, window in [(0.01, 0.1, 15000, 300000),
                                 (0.5, 0.5, 60000, 1800000),
                                 (0.9, 0.05, 15000, 60000)]:
        q = fdb.make_query(100000 + 30 * 15000, step, 100000 + 230 * 15000,
                           window, fdb.FN_HOLT_WINTERS, param=sf, param2=tf)
        want = oracle.query_exec(st.view(), q, st.num_series, q.num_windows)
        got = np.empty(st.num_series * q.num_windows, dtype=np.float64)
        eng.query(ds, q, out=got)
        np.testing.assert_allclose(got, want, rtol=1e-12, atol=0,
                                   equal_nan=True)
    with pytest.raises(RuntimeError):       # sf outside [0,1]
        q = fdb.make_query(550000, 15000, 700000, 300000,
                           fdb.FN_HOLT_WINTERS, param=1.5, param2=0.1)
        eng.query(ds, q, out=np.empty(24 * 11))


def _hw_chunked(ts, vs, ws, we, sf, tf):
    """Independent Python restatement of HoltWintersChunkedFunctionD for ONE
    single-chunk window (rows [s..e] = first ts >= ws .. last ts <= we):
    NaN seed scans, the double-counted second seed, the one-past-endRow read
    (in-chunk row or the modeled NaN at the chunk's end)."""
    n = len(ts)
    s = next((i for i in range(n) if ts[i] >= ws), n)
    e = n - 1
    while e >= 0 and ts[e] > we:
        e -= 1
    if s > e:
        return float("nan")
    itp = s
    s0v = float("nan")
    cur = s
    while cur <= e and np.isnan(s0v):
        s0v = vs[itp]; itp += 1; cur += 1
    b0v = float("nan")
    while cur <= e and np.isnan(b0v):
        b0v = vs[itp]; itp += 1; cur += 1
    nxt, b0, s0 = b0v, b0v - s0v, s0v
    row = cur - 1
    res = float("nan")
    if not np.isnan(b0):
        while row <= e:
            if not np.isnan(nxt):
                ns = sf * nxt + (1 - sf) * (s0 + b0)
                b0 = tf * (ns - s0) + (1 - tf) * b0
                s0 = ns
            nxt = vs[itp] if itp < n else float("nan")
            itp += 1
            row += 1
        res = s0
    return res


def test_holt_winters_nan_fuzz(fdb, oracle):
    """Randomized NaN patterns and window boundaries: the oracle equals an
    independent Python restatement of the chunked function bit for bit."""
    rng = np.random.default_rng(79)
    for trial in range(25):
        n = int(rng.integers(2, 120))
        ts = (100000 + np.arange(n) * 10000).astype(np.int64)
        vs = np.round(rng.normal(100, 20, n), 6)
        vs[rng.random(n) < rng.choice([0.0, 0.2, 0.6])] = np.nan
        sf = float(rng.uniform(0.01, 0.99))
        tf = float(rng.uniform(0.01, 0.99))
        end = int(ts[0] + rng.integers(1, n + 3) * 10000)
        win = int(rng.integers(1, n + 3) * 10000)
        samples = list(zip(ts.tolist(), vs.tolist()))
        got = one_window(fdb, oracle, samples, fdb.FN_HOLT_WINTERS,
                         param=sf, param2=tf, start=end - win, end=end)
        want = _hw_chunked(ts, vs, end - win, end, sf, tf)
        assert (np.isnan(got) and np.isnan(want)) or got == want, \
            (trial, got, want)


def test_nan_sequence_matrix_spec(fdb, oracle):
    """AggrOverTimeFunctionsSpec.scala:914-965 — sum/avg/stdvar/stddev/
    zscore/present over the spec's NaN-pattern sequences, against its own
    NaN-skipping models (window (60000, 160000] over 10s-spaced samples)."""
    nan = float("nan")
    test_data = [
        [15900.0, 15920.0, 15940.0, 15960.0, 15980.0, 16000.0, 16020.0],
        [-15900.0, -15920.0, -15940.0, -15960.0, -15980.0, -16000.0],
        [15900.0, 15920.0, 15940.0, 15960.0, 15980.0, 16000.0, nan],
        [23850.0, 23880.0, 23910.0, 23940.0, 23970.0, 24000.0],
        [31800.0, 31840.0, 31880.0, 31920.0, 31960.0, 32000.0],
        [31800.0, 31840.0, 31880.0, nan, 31920.0, 31960.0, 32000.0],
        [nan, 31800.0, 31840.0, 31880.0, 31920.0, 31960.0, 32000.0],
        [nan] * 7,
        [],
    ]
    for vals in test_data:
        samples = [(100000 + i * 10000, v) for i, v in enumerate(vals)]
        if not samples:
            samples = []
        clean = [v for v in vals if not np.isnan(v)]
        n = len(clean)
        exp_sum = sum(clean) if n else nan
        exp_avg = (exp_sum / n) if n else nan
        exp_var = (sum((x - exp_avg) ** 2 for x in clean) / n) if n else nan
        # zscore uses the reference's lastSample rule: set only when the
        # range's LAST row is non-NaN (VarOverTimeChunkedFunctionD:1103) —
        # a NaN-tailed window emits NaN. (The reference additionally never
        # resets lastSample between windows, so a long-lived iterator can
        # leak a previous window's sample into a NaN-tailed window; the
        # spec's own assertions accept NaN there, and this stateless engine
        # always emits NaN — DESIGN.md §9.)
        exp_z = (((vals[-1] - exp_avg) / np.sqrt(exp_var))
                 if n and not np.isnan(vals[-1]) else nan)

        def run(fn):
            if not samples:       # truly empty series
                st = fdb.ChunkStore()
                st.add_series(0, fdb.COL_GAUGE)
                st.seal()
                q = fdb.make_query(160000, 10000, 160000, 100000, fn)
                return float(oracle.query_exec(st.view(), q, 1, 1)[0])
            return one_window(fdb, oracle, samples, fn,
                              start=60000, end=160000)

        for fn, want in [(fdb.FN_SUM_OVER_TIME, exp_sum),
                         (fdb.FN_AVG_OVER_TIME, exp_avg),
                         (fdb.FN_STDVAR_OVER_TIME, exp_var),
                         (fdb.FN_STDDEV_OVER_TIME,
                          np.sqrt(exp_var) if n else nan),
                         (fdb.FN_ZSCORE, exp_z)]:
            got = run(fn)
            if np.isnan(want):
                assert np.isnan(got), (vals, fn, got)
            else:
                assert got == pytest.approx(want, rel=1e-9), (vals, fn, got)
        if n:
            assert run(fdb.FN_PRESENT) == 1.0
