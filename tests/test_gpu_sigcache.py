"""Sig-cache behavior (⇔ TransactionValidator sig_cache, caches.rs:57-82):
revalidating a block hits the cache for every signature check and returns
identical results; distinct blocks never false-hit (results stay bit-exact vs
the oracle)."""
import ctypes
import os
import sys

import pytest

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", "oracle"))
from workload import gen_block  # noqa: E402

pytestmark = pytest.mark.gpu

SKIP_MASS = 2


def oracle_validate(oracle, blob, n, flags=SKIP_MASS):
    codes = (ctypes.c_int32 * n)()
    fees = (ctypes.c_uint64 * n)()
    mh = (ctypes.c_uint8 * 32)()
    rc = oracle.ok_validate_block_parallel(blob, len(blob), 10**9, 10**9, flags,
                                           8, codes, fees, mh)
    assert rc == 0
    return list(codes), list(fees), bytes(mh)


def test_sig_cache_revalidation(oracle):
    from rusty_kaspa_amd.engine import Engine
    eng = Engine()
    try:
        n = 80
        blob, _ = gen_block(oracle, seed=31, n_txs=n, pct_multi_input=20,
                            pct_ecdsa=10, pct_invalid=10)
        oc, of, omh = oracle_validate(oracle, blob, n)

        c1, f1, p1 = eng.validate_block(blob, n, 10**9, 10**9, SKIP_MASS)
        ins1, hits1, miss1 = eng.sig_cache_stats()
        assert hits1 == 0 and ins1 > 0 and miss1 == ins1
        assert c1 == oc and f1 == of and eng.muhash_finalize(p1) == omh

        # revalidate: all checks must come from the cache, results identical
        c2, f2, p2 = eng.validate_block(blob, n, 10**9, 10**9, SKIP_MASS)
        ins2, hits2, miss2 = eng.sig_cache_stats()
        assert hits2 == ins1, (hits2, ins1)
        assert ins2 == ins1 and miss2 == miss1
        assert c2 == c1 and f2 == f1 and p2 == p1

        # a different block shares no checks and must stay oracle-exact
        blob3, _ = gen_block(oracle, seed=32, n_txs=n, pct_multi_input=20,
                             pct_ecdsa=10, pct_invalid=10)
        oc3, of3, omh3 = oracle_validate(oracle, blob3, n)
        c3, f3, p3 = eng.validate_block(blob3, n, 10**9, 10**9, SKIP_MASS)
        _, hits3, _ = eng.sig_cache_stats()
        assert hits3 == hits2  # no false hits
        assert c3 == oc3 and f3 == of3 and eng.muhash_finalize(p3) == omh3
    finally:
        eng.close()


def test_sig_cache_disabled(oracle):
    from rusty_kaspa_amd.engine import Engine
    eng = Engine(sig_cache_size=0)
    try:
        n = 20
        blob, _ = gen_block(oracle, seed=33, n_txs=n)
        oc, of, _ = oracle_validate(oracle, blob, n)
        for _ in range(2):
            c, f, _p = eng.validate_block(blob, n, 10**9, 10**9, SKIP_MASS)
            assert c == oc and f == of
        ins, hits, miss = eng.sig_cache_stats()
        assert ins == 0 and hits == 0 and miss == 0
    finally:
        eng.close()
