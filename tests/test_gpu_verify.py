"""GPU parity tests: HIP batch-verify kernels vs the oracle, on a real MI355X.

These are the parity tests proper for the EC-verify stage of the hot path —
bit-exact verdict bitmaps on seeded inputs, including adversarial cases
(tampered signatures, invalid pubkeys, overflowing r/s, high-S ECDSA).
"""
import ctypes

import pytest

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def engine():
    from rusty_kaspa_amd.engine import Engine
    eng = Engine()
    yield eng
    eng.close()


def popcount(bitmap):
    return sum(bin(w).count("1") for w in bitmap)


def test_schnorr_batch_parity(oracle, engine):
    n = 4096
    tuples = ctypes.create_string_buffer(n * 128)
    oracle.ok_gen_schnorr_tuples(ctypes.c_uint64(42), ctypes.c_size_t(n), 100, tuples, 8)
    expected = (ctypes.c_uint64 * ((n + 63) // 64))()
    oracle.ok_verify_schnorr_batch(tuples, ctypes.c_size_t(n), 8, expected)
    bitmap, status = engine.verify_schnorr_batch(tuples.raw[: n * 128], n,
                                                 with_status=True)
    assert bitmap == list(expected)
    # every status must agree with the oracle's verdict for valid/invalid
    for i in range(n):
        bit = (expected[i // 64] >> (i % 64)) & 1
        assert (status[i] == 0) == bool(bit), i


def test_schnorr_adversarial_edges(oracle, engine):
    """Hand-built edge tuples: r >= p, s >= n, pk not on curve, zero sig."""
    P = 0xFFFFFFFFFFFFFFFFFFFFFFFFFFFFFFFFFFFFFFFFFFFFFFFFFFFFFFFEFFFFFC2F
    N = 0xFFFFFFFFFFFFFFFFFFFFFFFFFFFFFFFEBAAEDCE6AF48A03BBFD25E8CD0364141
    base = ctypes.create_string_buffer(1 * 128)
    oracle.ok_gen_schnorr_tuples(ctypes.c_uint64(5), ctypes.c_size_t(1), 0, base, 1)
    r, s, pk, msg = (base.raw[0:32], base.raw[32:64], base.raw[64:96], base.raw[96:128])
    cases = [
        r + s + pk + msg,                                   # valid
        (P + 1).to_bytes(32, "big") + s + pk + msg,         # r >= p → invalid
        r + (N + 5).to_bytes(32, "big") + pk + msg,         # s >= n → invalid
        r + s + (5).to_bytes(32, "big") + msg,              # x=5 not on curve → bad pk
        bytes(64) + pk + msg,                               # zero sig → invalid
        r + s + pk + bytes(32),                             # wrong msg → invalid
    ]
    tuples = b"".join(cases)
    n = len(cases)
    bitmap, status = engine.verify_schnorr_batch(tuples, n, with_status=True)
    assert (bitmap[0] & 1) == 1
    assert popcount(bitmap) == 1
    assert status[0] == 0
    assert status[3] == 2  # bad pubkey — maps to TxScriptError::InvalidPubkey
    # oracle agreement on each
    for i, t in enumerate(cases):
        expect = oracle.ok_schnorr_verify(t[64:96], t[96:128], t[0:64])
        got_bit = (bitmap[i // 64] >> (i % 64)) & 1
        assert got_bit == (1 if expect == 1 else 0), i
        if expect == -1:
            assert status[i] == 2


def test_ecdsa_batch_parity(oracle, engine):
    n = 2048
    tuples = ctypes.create_string_buffer(n * 132)
    oracle.ok_gen_ecdsa_tuples(ctypes.c_uint64(77), ctypes.c_size_t(n), 150, tuples, 8)
    bitmap, status = engine.verify_ecdsa_batch(tuples.raw[: n * 132], n,
                                               with_status=True)
    for i in range(n):
        t = tuples.raw[i * 132:(i + 1) * 132]
        expect = oracle.ok_ecdsa_verify(t[64:97], t[97:129], t[0:64])
        got = (bitmap[i // 64] >> (i % 64)) & 1
        assert got == (1 if expect == 1 else 0), i


def test_ecdsa_high_s_rejected(oracle, engine):
    N = 0xFFFFFFFFFFFFFFFFFFFFFFFFFFFFFFFEBAAEDCE6AF48A03BBFD25E8CD0364141
    base = ctypes.create_string_buffer(132)
    oracle.ok_gen_ecdsa_tuples(ctypes.c_uint64(3), ctypes.c_size_t(1), 0, base, 1)
    r, s = base.raw[0:32], base.raw[32:64]
    rest = base.raw[64:132]
    high_s = (N - int.from_bytes(s, "big")).to_bytes(32, "big")
    tuples = bytes(base.raw) + r + high_s + rest
    bitmap, status = engine.verify_ecdsa_batch(tuples, 2, with_status=True)
    assert (bitmap[0] >> 0) & 1 == 1
    assert (bitmap[0] >> 1) & 1 == 0


def test_muhash_finalize_parity(oracle, engine):
    """Engine host-side U3072 (independent impl) vs oracle on combine+finalize."""
    import random
    rng = random.Random(11)
    U = ctypes.c_uint64 * 48
    num, den = U(), U()
    oracle.ok_u3072_one(num)
    oracle.ok_u3072_one(den)
    elem = U()
    acc = bytearray(768)
    acc[0:1] = b"\x01"
    acc[384:385] = b"\x01"
    for i in range(8):
        data = bytes(rng.randrange(256) for _ in range(80))
        oracle.ok_muhash_element(data, len(data), elem)
        target = num if i % 3 else den
        oracle.ok_u3072_mul(target, elem)
        # engine-side: combine with a partial that has this element in num or den
        other = bytearray(768)
        other[0:1] = b"\x01"
        other[384:385] = b"\x01"
        ser = b"".join(int(elem[k]).to_bytes(8, "little") for k in range(48))
        if i % 3:
            other[0:384] = ser
        else:
            other[384:768] = ser
        engine.muhash_combine(acc, bytes(other))
    expected = (ctypes.c_uint8 * 32)()
    oracle.ok_muhash_finalize(num, den, expected)
    got = engine.muhash_finalize(bytes(acc))
    assert got == bytes(expected)
