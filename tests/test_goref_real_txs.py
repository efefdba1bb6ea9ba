"""End-to-end parity on REAL transactions from the reference's
goref-1060-tx-265-blocks integration fixture (an independently produced
Go-implementation DAG): 223 transactions with genuine schnorr signatures and
UtxoEntries rebuilt from their creating blocks (extract_goref.py). Every
signature must verify; any tampering must reject."""
import ctypes
import gzip
import json
import os
import struct
import sys

import pytest

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", "oracle"))

GOLD_DIR = os.path.join(os.path.dirname(__file__), "golden")
FIXTURES = ["goref_txs.json.gz", "goref_pruning_txs.json.gz"]
EXPECT_TOTAL = {"goref_txs.json.gz": 223, "goref_pruning_txs.json.gz": 4789}
SKIP_MASS = 2


def load_batches(name="goref_txs.json.gz"):
    with gzip.open(os.path.join(GOLD_DIR, name), "rt") as f:
        return [bytes.fromhex(h) for h in json.load(f)["batches"]]


def oracle_validate(oracle, blob, n):
    codes = (ctypes.c_int32 * n)()
    fees = (ctypes.c_uint64 * n)()
    mh = (ctypes.c_uint8 * 32)()
    rc = oracle.ok_validate_block_parallel(
        blob, ctypes.c_size_t(len(blob)), ctypes.c_uint64(10**9),
        ctypes.c_uint64(10**9), SKIP_MASS, 8, codes, fees, mh)
    assert rc == 0
    return list(codes), list(fees), bytes(mh)


@pytest.mark.parametrize("name", FIXTURES)
def test_oracle_accepts_all_real_sigs(oracle, name):
    total = 0
    for blob in load_batches(name):
        n, = struct.unpack_from("<I", blob, 0)
        codes, fees, _ = oracle_validate(oracle, blob, n)
        assert all(c == 0 for c in codes), codes[:8]
        total += n
    assert total == EXPECT_TOTAL[name]


def test_oracle_rejects_tampered_sig(oracle):
    blob = bytearray(load_batches()[0])
    n, = struct.unpack_from("<I", blob, 0)
    off0, = struct.unpack_from("<I", blob, 4)
    # first tx's first input's signature starts at off0+88 (no payload) + 52 + 1
    payload_len, = struct.unpack_from("<I", blob, off0 + 36)
    sig_off = off0 + 88 + payload_len + 52 + 1
    blob[sig_off + 10] ^= 0x40
    codes, _, _ = oracle_validate(oracle, bytes(blob), n)
    assert codes[0] != 0
    assert all(c == 0 for c in codes[1:])


@pytest.mark.gpu
def test_engine_matches_oracle_on_real_txs(oracle):
    from rusty_kaspa_amd.engine import Engine
    eng = Engine()
    try:
        for name in FIXTURES:
          for blob in load_batches(name):
            n, = struct.unpack_from("<I", blob, 0)
            oc, of, omh = oracle_validate(oracle, blob, n)
            ec, ef, ep = eng.validate_block(blob, n, 10**9, 10**9, SKIP_MASS)
            assert ec == oc and ef == of
            assert eng.muhash_finalize(ep) == omh
            assert all(c == 0 for c in ec)
    finally:
        eng.close()
