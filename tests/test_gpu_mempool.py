"""Mempool batch validation (kv_validate_mempool) vs the oracle: SkipMassCheck
semantics, computed contextual mass, feerate-threshold rejection, and the
table-resolved populate variant — ⇔ utxo_validation.rs:418-457."""
import ctypes
import os
import sys

import pytest

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", "oracle"))
from workload import gen_block  # noqa: E402

from rusty_kaspa_amd.blob import strip_utxo_entries  # noqa: E402

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def engine():
    from rusty_kaspa_amd.engine import Engine
    eng = Engine()
    yield eng
    eng.close()


def oracle_mempool(oracle, blob, n, threshold):
    codes = (ctypes.c_int32 * n)()
    fees = (ctypes.c_uint64 * n)()
    rc = oracle.ok_validate_mempool(blob, ctypes.c_size_t(len(blob)),
                                    ctypes.c_uint64(10**9),
                                    ctypes.c_double(threshold), 8, codes, fees)
    assert rc == 0
    return list(codes), list(fees)


@pytest.mark.parametrize("threshold", [0.0, 0.05, 0.5, 5.0, 1e9])
def test_mempool_vs_oracle(oracle, engine, threshold):
    n = 80
    blob, _ = gen_block(oracle, seed=71, n_txs=n, pct_multi_input=20,
                        pct_ecdsa=10, pct_invalid=10)
    oc, of = oracle_mempool(oracle, blob, n, threshold)
    ec, ef = engine.validate_mempool(blob, n, 10**9, threshold)
    assert ec == oc, [(i, a, b) for i, (a, b) in enumerate(zip(ec, oc)) if a != b][:5]
    assert ef == of
    if threshold == 1e9:
        assert all(c != 0 for c in ec)  # everything fails an absurd threshold
    if threshold == 0.0:
        assert any(c == 0 for c in ec)


def test_mempool_from_table(oracle, engine):
    n = 60
    blob, _ = gen_block(oracle, seed=72, n_txs=n, pct_multi_input=25,
                        pct_ecdsa=10)
    oc, of = oracle_mempool(oracle, blob, n, 0.1)
    stripped, seeds = strip_utxo_entries(blob)
    lib = engine.lib
    ctx = ctypes.c_void_p(engine.ctx)
    assert lib.kv_utxo_reset(ctx, ctypes.c_uint64(2 * len(seeds))) == 0
    # withhold one input: its tx must come back MISSING_OUTPOINT
    kept = seeds[1:]
    assert lib.kv_utxo_upsert(ctx, b"".join(op for op, _ in kept),
                              b"".join(e for _, e in kept),
                              ctypes.c_size_t(len(kept))) == 0
    ec, ef = engine.validate_mempool(stripped, n, 10**9, 0.1,
                                     from_utxo_table=True)
    assert ec[0] == 9 and ef[0] == 0
    assert ec[1:] == oc[1:] and ef[1:] == of[1:]


def test_mempool_from_table_arena_and_interp(oracle, engine):
    """Mempool validation resolving entries from the GPU table where the spent
    UTXOs carry LONG scripts (arena path) executed by the general interpreter
    with real signatures (collect/replay path) — populate + arena gather +
    interpreter + mempool post-pass in one flow, vs the oracle on the
    inline-populated blob."""
    import struct
    import sys as _sys
    import os as _os
    _sys.path.insert(0, _os.path.join(_os.path.dirname(__file__), ".."))
    import rusty_kaspa_amd.blob as B
    from rusty_kaspa_amd.blob import strip_utxo_entries
    lib = engine.lib
    assert lib.kv_utxo_reset(ctypes.c_void_p(engine.ctx),
                             ctypes.c_uint64(256)) == 0

    def push(data):
        return (bytes([len(data)]) + data) if data else b"\x00"

    # keypair via the oracle
    key = bytes([21]) * 31 + b"\x01"
    pk = (ctypes.c_uint8 * 32)()
    assert oracle.ok_pubkey_xonly(key, pk) == 1
    pk = bytes(pk)
    # non-template, >36B spk: NOP padding + <pk> CHECKSIG
    spk_long = b"\x61" * 60 + push(pk) + b"\xac"
    prev = bytes([7]) * 32
    outpoint = prev + struct.pack("<I", 0)
    entry = struct.pack("<QQHHI", 70_000, 5, 0, 0, len(spk_long)) + bytes(36 + 4)
    rc = lib.kv_utxo_upsert_spk(ctypes.c_void_p(engine.ctx), outpoint, entry,
                                spk_long, ctypes.c_size_t(len(spk_long)),
                                ctypes.c_size_t(1))
    assert rc == 0, lib.kv_last_error().decode()

    tx = B.tx_dict(
        1,
        [B.tx_input(prev, 0, sequence=2**64 - 1,
                    sig_script=push(bytes(64) + b"\x01"),
                    commit_kind=0, commit_value=20,
                    utxo=B.utxo_entry(70_000, spk_long, daa_score=5))],
        [B.tx_output(60_000, b"\x51")])
    blob = B.build_blob([tx])
    msg = (ctypes.c_uint8 * 32)()
    assert oracle.ok_sighash(blob, len(blob), 0, 0, 1, 0, msg) == 0
    sig = (ctypes.c_uint8 * 64)()
    assert oracle.ok_schnorr_sign(key, msg, None, sig) == 1
    ins = list(tx["inputs"])
    ins[0] = dict(ins[0], sig_script=push(bytes(sig) + b"\x01"))
    tx = dict(tx, inputs=ins)
    blob = B.build_blob([tx])

    # oracle (inline entries) vs engine (from the table, arena-resolved)
    oc = (ctypes.c_int32 * 1)()
    of = (ctypes.c_uint64 * 1)()
    assert oracle.ok_validate_mempool(blob, ctypes.c_size_t(len(blob)),
                                      ctypes.c_uint64(10**9),
                                      ctypes.c_double(0.0), 2, oc, of) == 0
    stripped, _ = strip_utxo_entries(blob)
    codes, fees = engine.validate_mempool(stripped, 1, 10**9,
                                          feerate_threshold=0.0,
                                          from_utxo_table=True)
    assert codes == list(oc) == [0], (codes, list(oc))
    assert fees == list(of) == [10_000]
    # corrupted signature rejects identically through the same path
    bad = bytearray(bytes(sig))
    bad[5] ^= 1
    ins[0] = dict(ins[0], sig_script=push(bytes(bad) + b"\x01"))
    blob2 = B.build_blob([dict(tx, inputs=ins)])
    assert oracle.ok_validate_mempool(blob2, ctypes.c_size_t(len(blob2)),
                                      ctypes.c_uint64(10**9),
                                      ctypes.c_double(0.0), 2, oc, of) == 0
    stripped2, _ = strip_utxo_entries(blob2)
    codes2, _ = engine.validate_mempool(stripped2, 1, 10**9,
                                        feerate_threshold=0.0,
                                        from_utxo_table=True)
    assert codes2 == list(oc) and codes2[0] != 0
