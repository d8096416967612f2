"""The committed golden fixtures must agree with the executable test literals
and with the builder/oracle (guards against fixture drift)."""
import json
import os

import numpy as np

HERE = os.path.join(os.path.dirname(os.path.abspath(__file__)), "golden")


def _load(name):
    with open(os.path.join(HERE, name)) as f:
        return json.load(f)


def test_nibblepack_golden_file(fdb):
    g = _load("nibblepack.json")
    for case in ("even", "odd"):
        c = g[case]
        inp = [int(v, 16) for v in c["input_hex"]]
        assert fdb.nibblepack_pack8(inp) == bytes.fromhex(c["packed_hex"])


def test_hist_quantile_golden_file(oracle):
    g = _load("hist_quantile.json")
    s = g["bucket_scheme"]
    for vals, exp in zip(g["raw_hist_buckets"], g["expected"]):
        got = oracle.hist_quantile(g["quantile_q"],
                                   np.array(vals, dtype=np.float64),
                                   s["first"], s["mult"])
        assert abs(got - exp) < 1e-12


def test_rate_golden_file(fdb, oracle):
    from conftest import build_store
    g = _load("rate_fixtures.json")
    tol = g["tolerance_abs"]
    cs = [tuple(x) for x in g["counter_samples"]]
    c2 = [tuple(x) for x in g["chunk2"]]
    for name, chunks in [("basic_rate", [cs]),
                         ("reset_at_chunk_boundary", [cs, c2]),
                         ("drops_in_middle",
                          [[tuple(x) for x in g["reset_chunk1"]],
                           [tuple(x) for x in g["reset_chunk2"]]])]:
        case = g["cases"][name]
        st = build_store(fdb, [chunks])
        end = case["end"]
        q = fdb.make_query(end, 10000, end, end - case["start"], fdb.FN_RATE)
        out = oracle.eval_series(st.view(), 0, q, 1)
        assert abs(out[0] - case["expected"]) < tol, name
