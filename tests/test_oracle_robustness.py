"""Parser robustness: truncated/corrupted blobs must fail cleanly (nonzero rc
or sensible codes), never crash — the blob is the FFI trust boundary."""
import ctypes
import os
import random
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", "oracle"))
from workload import gen_block  # noqa: E402


def validate(oracle, blob):
    n = 64  # upper bound; parser reads the real count
    codes = (ctypes.c_int32 * n)()
    fees = (ctypes.c_uint64 * n)()
    mh = (ctypes.c_uint8 * 32)()
    return oracle.ok_validate_block_parallel(blob, ctypes.c_size_t(len(blob)),
                                             10**9, 10**9, 2, 4, codes, fees, mh)


def test_truncations(oracle):
    blob, _ = gen_block(oracle, seed=91, n_txs=12, pct_multi_input=25)
    assert validate(oracle, blob) == 0
    rng = random.Random(3)
    for _ in range(200):
        cut = rng.randrange(len(blob))
        rc = validate(oracle, blob[:cut])
        assert rc != 0 or cut >= len(blob)  # truncated must not claim success
    for n in range(0, 12):
        rc = validate(oracle, blob[:n])
        assert rc != 0


def test_bitflips_never_crash(oracle):
    blob, _ = gen_block(oracle, seed=92, n_txs=10, pct_multi_input=20)
    rng = random.Random(4)
    for _ in range(300):
        b = bytearray(blob)
        for _ in range(rng.randrange(1, 4)):
            b[rng.randrange(len(b))] ^= 1 << rng.randrange(8)
        validate(oracle, bytes(b))  # any rc is fine; must not crash
        oracle.ok_body_check(bytes(b), ctypes.c_size_t(len(b)))
