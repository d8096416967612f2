"""Oracle rate/increase parity against the reference's own golden fixtures.

Fixture series and expected-value formulas copied from
query/src/test/scala/filodb/query/exec/rangefn/RateFunctionsSpec.scala:16-160
(the expected values there are closed-form expressions over the fixtures;
we evaluate the same expressions). Tolerance matches the spec's errorOk=1e-7.
"""
import numpy as np
import pytest

from conftest import build_store

# RateFunctionsSpec.scala:17-26
COUNTER_SAMPLES = [
    (8072000, 4419.00), (8082100, 4511.00), (8092196, 4614.00),
    (8102215, 4724.00), (8112223, 4909.00), (8122388, 4948.00),
    (8132570, 5000.00), (8142822, 5095.00), (8152858, 5102.00),
    (8162999, 5201.00),
]

# :72-76 (reset-at-chunk-boundary second chunk)
CHUNK2 = [(8173000, 325.00), (8183000, 511.00), (8193000, 614.00),
          (8203000, 724.00), (8213000, 909.00)]

# :117-135 (drops in middle of chunks)
RESET_CHUNK1 = [
    (8072000, 4419.00), (8082100, 4511.00), (8092196, 4614.00),
    (8102215, 4724.00), (8112223, 4909.00), (8122388, 948.00),
    (8132570, 1000.00), (8142822, 1095.00), (8152858, 1102.00),
    (8162999, 1201.00),
]
RESET_CHUNK2 = [(8173000, 1325.00), (8183000, 1511.00), (8193000, 214.00),
                (8203000, 324.00), (8213000, 409.00)]

ERR = 1e-7


def run_rate(fdb, oracle, chunks, start, step, end, window, func=None):
    st = build_store(fdb, [chunks])
    q = fdb.make_query(start, step, end, window, fdb.FN_RATE if func is None else func)
    return oracle.eval_series(st.view(), 0, q, q.num_windows)


def test_rate_start_end_outside_window(fdb, oracle):
    # RateFunctionsSpec :59-70: one window, start=end=endTs, window=endTs-startTs
    start_ts, end_ts = 8071950, 8163070
    expected = (5201.0 - 4419.0) / (8162999 - 8072000) * 1000
    out = run_rate(fdb, oracle, [COUNTER_SAMPLES], end_ts, 10000, end_ts,
                   end_ts - start_ts)
    assert out[0] == pytest.approx(expected, abs=ERR)


def test_rate_reset_at_chunk_boundary(fdb, oracle):
    # :72-92: correction = last value of chunk1
    start_ts, end_ts = 8071950, 8213070
    correction = 5201.0
    expected = (909.0 + correction - 4419.0) / (8213000 - 8072000) * 1000
    out = run_rate(fdb, oracle, [COUNTER_SAMPLES, CHUNK2], end_ts, 10000, end_ts,
                   end_ts - start_ts)
    assert out[0] == pytest.approx(expected, abs=ERR)


def test_rate_nan_at_chunk_start(fdb, oracle):
    # :94-114: chunk2 leads with NaN; same expected as clean reset
    chunk2 = [(8173000, float("nan"))] + CHUNK2[1:]
    start_ts, end_ts = 8071950, 8213070
    correction = 5201.0
    expected = (909.0 + correction - 4419.0) / (8213000 - 8072000) * 1000
    out = run_rate(fdb, oracle, [COUNTER_SAMPLES, chunk2], end_ts, 10000, end_ts,
                   end_ts - start_ts)
    assert out[0] == pytest.approx(expected, abs=ERR)


def test_rate_drops_in_middle_of_chunks(fdb, oracle):
    # :137-160: one drop in each chunk
    start_ts, end_ts = 8071950, 8213070
    correction1 = RESET_CHUNK1[4][1]        # 4909
    corr2 = RESET_CHUNK2[1][1]              # 1511
    corrections = correction1 + corr2
    expected = (RESET_CHUNK2[-1][1] + corrections - RESET_CHUNK1[0][1]) / \
               (RESET_CHUNK2[-1][0] - RESET_CHUNK1[0][0]) * 1000
    out = run_rate(fdb, oracle, [RESET_CHUNK1, RESET_CHUNK2], end_ts, 10000, end_ts,
                   end_ts - start_ts)
    assert out[0] == pytest.approx(expected, abs=ERR)
    # two drops in one chunk (:152-157)
    out2 = run_rate(fdb, oracle, [RESET_CHUNK1 + RESET_CHUNK2], end_ts, 10000, end_ts,
                    end_ts - start_ts)
    assert out2[0] == pytest.approx(expected, abs=ERR)


def test_rate_single_sample_window_is_nan(fdb, oracle):
    # :162+: window containing one sample → NaN (needs 2 samples)
    out = run_rate(fdb, oracle, [COUNTER_SAMPLES], 8102215, 10000, 8102215, 5000)
    assert np.isnan(out[0])


def test_rate_empty_window_is_nan(fdb, oracle):
    out = run_rate(fdb, oracle, [COUNTER_SAMPLES], 8071000, 10000, 8071000, 5000)
    assert np.isnan(out[0])


def test_increase(fdb, oracle):
    # increase = extrapolated delta without per-second scaling
    start_ts, end_ts = 8071950, 8163070
    out_rate = run_rate(fdb, oracle, [COUNTER_SAMPLES], end_ts, 10000, end_ts,
                        end_ts - start_ts, func=0)
    out_inc = run_rate(fdb, oracle, [COUNTER_SAMPLES], end_ts, 10000, end_ts,
                       end_ts - start_ts, func=1)
    assert out_inc[0] == pytest.approx(out_rate[0] * (end_ts - (end_ts - (8163070 - 8071950))) / 1000, rel=1e-12)


def test_extrapolated_rate_formula(oracle):
    # spot-check the epilogue itself against a hand-computed Prometheus value
    # (RateFunctions.scala:72-111): samples exactly at window edges
    ws, we = 0, 100000
    r = oracle.extrapolated_rate(ws, we, 11, 0, 0.0, 100000, 100.0, True, True)
    assert r == pytest.approx(1.0, abs=1e-12)  # 100 over 100s → 1/s, no extrapolation


def test_rate_many_windows_vs_naive(fdb, oracle):
    """Sliding windows over the fixture: spec-independent consistency — each
    window's rate recomputed naively from the in-window samples (no resets)."""
    st = build_store(fdb, [[COUNTER_SAMPLES]])
    # the naive model must see the ENCODED timestamps: this fixture's jitter is
    # within ±250 of the slope line, so the reference's approx-const timestamp
    # encoding (faithfully reproduced by the builder) replaces them with the line
    tsb, _, _, _, _ = st.chunk(0, 0)
    ts = oracle.decode_longs(tsb)
    vs = np.array([v for _, v in COUNTER_SAMPLES])
    start, step, end, window = 8092000, 10000, 8162000, 30000
    q = st_q = None
    import filodb_amd as f
    q = f.make_query(start, step, end, window, f.FN_RATE)
    out = oracle.eval_series(st.view(), 0, q, q.num_windows)
    for w in range(q.num_windows):
        w_end = start + w * step
        w_start = w_end - window
        m = (ts >= w_start) & (ts <= w_end)
        if m.sum() < 2:
            assert np.isnan(out[w])
            continue
        t1, t2 = ts[m][0], ts[m][-1]
        expected = oracle.extrapolated_rate(int(w_start), int(w_end), int(m.sum()),
                                            int(t1), float(vs[m][0]),
                                            int(t2), float(vs[m][-1]), True, True)
        assert out[w] == pytest.approx(expected, abs=ERR), w


def test_counter_correction_goldens(fdb, oracle):
    """BufferableCounterCorrectionIteratorSpec.scala:9-26 literal fixtures:
    the corrected counter series for dips / multiple dips / no dips."""
    cases = [
        ([3, 5, 7, 13, 2, 34], [3, 5, 7, 13, 15, 47]),
        ([3, 5, 7, 13, 2, 34, 4, 6], [3, 5, 7, 13, 15, 47, 51, 53]),
        ([3, 5, 7, 13, 22, 34], [3, 5, 7, 13, 22, 34]),
    ]
    for raw, expected in cases:
        ts = (100000 + np.arange(len(raw)) * 10000).astype(np.int64)
        st = build_store(fdb, [[[(int(t), float(v)) for t, v in zip(ts, raw)]]],
                         kind=fdb.COL_COUNTER)
        _, vab, _, _, _ = st.chunk(0, 0)
        got = oracle.corrected_doubles(vab)
        np.testing.assert_array_equal(got, np.array(expected, dtype=np.float64))
