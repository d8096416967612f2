"""NibblePack golden-vector parity.

Byte-exact expected outputs copied from the reference's own test suite:
core/src/test/scala/filodb.memory/format/NibblePackTest.scala:13-78 (the
"partial non-zero even/odd nibbles" and unpack cases). Both the product builder
(fdb_nibblepack_*) and the test oracle (oracle_nibblepack_unpack8) are checked.
"""
import numpy as np
import pytest


# NibblePackTest.scala:16-30
EVEN_INPUT = [0, 0x0000003322110000, 0x0000004433220000,
              0x0000005544330000, 0x0000006655440000, 0, 0, 0]
EVEN_PACKED = bytes([0x1e, 0x54,
                     0x11, 0x22, 0x33, 0x22, 0x33, 0x44,
                     0x33, 0x44, 0x55, 0x44, 0x55, 0x66])

# NibblePackTest.scala:33-52
ODD_INPUT = [0, 0x0000003322100000, 0x0000004433200000,
             0x0000005544300000, 0x0000006655400000, 0x0000007654300000, 0, 0]
ODD_PACKED = bytes([0x3e, 0x45,
                    0x21, 0x32, 0x23, 0x33, 0x44,
                    0x43, 0x54, 0x45, 0x55, 0x66,
                    0x43, 0x65, 0x07])


def test_pack8_even_nibbles(fdb):
    assert fdb.nibblepack_pack8(EVEN_INPUT) == EVEN_PACKED


def test_pack8_odd_nibbles(fdb):
    assert fdb.nibblepack_pack8(ODD_INPUT) == ODD_PACKED


def test_unpack8_odd_nibbles(fdb):
    out, consumed = fdb.nibblepack_unpack8(ODD_PACKED)
    assert out == ODD_INPUT
    assert consumed == len(ODD_PACKED)


def test_oracle_unpack8_golden(oracle):
    out, consumed = oracle.nibblepack_unpack8(ODD_PACKED)
    assert out == ODD_INPUT
    assert consumed == len(ODD_PACKED)
    out, consumed = oracle.nibblepack_unpack8(EVEN_PACKED)
    assert out == EVEN_INPUT


def test_pack8_all_zero(fdb):
    assert fdb.nibblepack_pack8([0] * 8) == bytes([0])
    out, consumed = fdb.nibblepack_unpack8(bytes([0]))
    assert out == [0] * 8 and consumed == 1


def test_pack8_roundtrip_random(fdb, oracle):
    rng = np.random.default_rng(42)
    for _ in range(200):
        shift = int(rng.integers(0, 50))
        vals = [int(v) << shift for v in rng.integers(0, 1 << 12, 8)]
        packed = fdb.nibblepack_pack8(vals)
        got, consumed = fdb.nibblepack_unpack8(packed)
        assert got == vals and consumed == len(packed)
        got2, _ = oracle.nibblepack_unpack8(packed)
        assert got2 == vals


def test_pack_delta_roundtrip(fdb, oracle):
    """NibblePackTest.scala:79+ 'pack and unpack delta values' input."""
    inputs = [0, 1000, 1001, 1002, 1003, 2005, 2010, 3034, 4045, 5056, 6067, 7078]
    import ctypes
    import filodb_amd as f
    arr = np.array(inputs, dtype=np.int64)
    out = (ctypes.c_uint8 * 256)()
    n = f.lib().fdb_nibblepack_pack_delta(
        arr.ctypes.data_as(ctypes.POINTER(ctypes.c_int64)), len(inputs), out, 256)
    assert n > 0
    # delta-unpack via oracle unpack8 + running sum (NibblePack DeltaSink semantics)
    data = bytes(out[:n])
    vals, pos, current = [], 0, 0
    while len(vals) < len(inputs):
        eight, consumed = oracle.nibblepack_unpack8(data[pos:])
        for d in eight:
            current += d
            vals.append(current)
        pos += consumed
    assert vals[:len(inputs)] == inputs


def test_unpack_doubles_xor_roundtrip(fdb, oracle):
    """§8a9 close-out: packDoubles → unpackDoubleXOR bit-exact roundtrip,
    host/product decoder vs the oracle restatement (NibblePack.scala:360-394)."""
    import ctypes as ct
    rng = np.random.default_rng(11)
    for n in (1, 2, 7, 8, 9, 63, 64, 65, 500):
        vals = rng.normal(0, 100, n)
        vals[rng.random(n) < 0.1] = 0.0
        packed = _pack_doubles(fdb, vals)
        got = fdb.nibblepack_unpack_doubles(packed, n)
        np.testing.assert_array_equal(got, vals)
        # oracle restatement decodes identically
        L = oracle.lib()
        L.oracle_nibblepack_unpack_doubles.argtypes = [
            ct.POINTER(ct.c_uint8), ct.c_int32,
            ct.POINTER(ct.c_double), ct.c_int32]
        buf = (ct.c_uint8 * len(packed)).from_buffer_copy(packed)
        out = np.empty(n, dtype=np.float64)
        rc = L.oracle_nibblepack_unpack_doubles(
            buf, len(packed), out.ctypes.data_as(ct.POINTER(ct.c_double)), n)
        assert rc == 0
        np.testing.assert_array_equal(out, vals)


def _pack_doubles(fdb, vals):
    import ctypes as ct
    a = np.ascontiguousarray(vals, dtype=np.float64)
    cap = len(a) * 10 + 64
    out = (ct.c_uint8 * cap)()
    n = fdb.lib().fdb_nibblepack_pack_doubles(
        a.ctypes.data_as(ct.POINTER(ct.c_double)), len(a), out, cap)
    assert n > 0
    return bytes(out[:n])


@pytest.mark.gpu
def test_gpu_xor_decode_matches_host(fdb, oracle):
    """The wavefront prefix-XOR kernel decodes packed streams bit-exactly
    (vs the host decoder) across sizes and batches."""
    rng = np.random.default_rng(12)
    eng = fdb.Engine(0)
    streams = []
    wants = []
    for n in (1, 2, 9, 64, 65, 129, 400, 1000):
        vals = np.round(rng.normal(50, 20, n), 3)
        vals[rng.random(n) < 0.05] = 0.0
        streams.append((_pack_doubles(fdb, vals), n))
        wants.append(vals)
    outs = fdb.gpu_unpack_doubles_xor(eng, streams)
    for got, want in zip(outs, wants):
        np.testing.assert_array_equal(got, want)
