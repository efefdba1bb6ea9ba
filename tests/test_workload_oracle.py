"""CPU tests: the workload generator's blocks validate through the oracle with
the expected verdict pattern (valid txs OK, corrupted first-sigs rejected)."""
import ctypes
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", "oracle"))

from workload import gen_block  # noqa: E402

KV_OK = 0
SKIP_MASS = 2


def validate(oracle, blob, n, parallel=True):
    codes = (ctypes.c_int32 * n)()
    fees = (ctypes.c_uint64 * n)()
    mh = (ctypes.c_uint8 * 32)()
    if parallel:
        rc = oracle.ok_validate_block_parallel(blob, len(blob), 10**9, 10**9,
                                               SKIP_MASS, 8, codes, fees, mh)
    else:
        rc = oracle.ok_validate_block(blob, len(blob), 10**9, 10**9, SKIP_MASS,
                                      codes, fees, mh)
    assert rc == 0
    return list(codes), list(fees), bytes(mh)


def test_p2pk_block_validates(oracle):
    blob, meta = gen_block(oracle, seed=1, n_txs=24)
    codes, fees, mh = validate(oracle, blob, 24)
    for t, spec in enumerate(meta["specs"]):
        assert codes[t] == KV_OK, (t, codes[t])
        assert fees[t] >= 1000


def test_mixed_block_with_invalid(oracle):
    blob, meta = gen_block(oracle, seed=2, n_txs=40, pct_multi_input=20,
                           pct_ecdsa=10, pct_multisig=10, pct_invalid=25)
    codes, fees, mh = validate(oracle, blob, 40)
    n_bad = 0
    for t, spec in enumerate(meta["specs"]):
        if spec["invalid"]:
            assert codes[t] != KV_OK, t
            n_bad += 1
        else:
            assert codes[t] == KV_OK, (t, codes[t])
    assert n_bad > 0
    # sequential and parallel paths agree bit-exactly
    codes2, fees2, mh2 = validate(oracle, blob, 40, parallel=False)
    assert codes2 == codes and fees2 == fees and mh2 == mh


def test_muhash_changes_with_verdicts(oracle):
    blob, _ = gen_block(oracle, seed=3, n_txs=8)
    _, _, mh_all = validate(oracle, blob, 8)
    blob2, _ = gen_block(oracle, seed=3, n_txs=8, pct_invalid=100)
    _, _, mh_none = validate(oracle, blob2, 8)
    assert mh_all != mh_none
    # all txs invalid → muhash over empty set = EMPTY_MUHASH
    assert mh_none.hex() == (
        "544eb3142c000f0ad2c76ac41f4222abbababed830eeafee4b6dc56b52d5cac0")


def test_full_flags_storage_mass(oracle):
    """KV_FLAGS_FULL: the generator's storage-mass commitments must satisfy the
    KIP-9 check; a corrupted commitment must yield WrongMass (5)."""
    import ctypes
    blob, meta = gen_block(oracle, seed=6, n_txs=16, pct_multi_input=30)
    n = 16
    codes = (ctypes.c_int32 * n)()
    fees = (ctypes.c_uint64 * n)()
    mh = (ctypes.c_uint8 * 32)()
    rc = oracle.ok_validate_block_parallel(blob, len(blob), 10**9, 10**9, 0, 8,
                                           codes, fees, mh)
    assert rc == 0
    assert all(c == 0 for c in codes), list(codes)
    # corrupt one tx's committed mass → WrongMass
    import struct
    off = struct.unpack("<I", blob[4:8])[0]
    bad = bytearray(blob)
    bad[off + 48:off + 56] = struct.pack("<Q", 12345678)
    rc = oracle.ok_validate_block_parallel(bytes(bad), len(bad), 10**9, 10**9, 0, 8,
                                           codes, fees, mh)
    assert rc == 0
    assert codes[0] == 5  # KV_ERR_WRONG_MASS
    assert all(c == 0 for c in list(codes)[1:])
