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
