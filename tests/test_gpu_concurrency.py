"""Two engine contexts in ONE process validating concurrently — the
'one process driving N contexts' integration mode. Validate scratch is
per-context; results must stay bit-exact under interleaving."""
import ctypes
import os
import sys
import threading

import pytest

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", "oracle"))
from workload import gen_block  # noqa: E402

pytestmark = pytest.mark.gpu


def test_two_contexts_concurrent(oracle):
    from rusty_kaspa_amd.engine import Engine
    blob1, _ = gen_block(oracle, seed=61, n_txs=300, pct_multi_input=20,
                         pct_ecdsa=10)
    blob2, _ = gen_block(oracle, seed=62, n_txs=300, pct_multi_input=20,
                         pct_ecdsa=10)

    def oracle_codes(blob, n):
        c = (ctypes.c_int32 * n)()
        f = (ctypes.c_uint64 * n)()
        m = (ctypes.c_uint8 * 32)()
        assert oracle.ok_validate_block_parallel(blob, len(blob), 10**9, 10**9,
                                                 2, 8, c, f, m) == 0
        return list(c), bytes(m)

    exp1, mh1 = oracle_codes(blob1, 300)
    exp2, mh2 = oracle_codes(blob2, 300)
    e1, e2 = Engine(), Engine()
    errs = []

    def worker(eng, blob, exp, mh, tag):
        try:
            for _ in range(12):
                c, f, p = eng.validate_block(blob, 300, 10**9, 10**9, 2)
                assert c == exp, f"{tag} codes diverged"
                assert eng.muhash_finalize(p) == mh, f"{tag} muhash diverged"
        except Exception as ex:
            errs.append(f"{tag}: {ex}")

    ts = [threading.Thread(target=worker, args=(e1, blob1, exp1, mh1, "ctx1")),
          threading.Thread(target=worker, args=(e2, blob2, exp2, mh2, "ctx2"))]
    for t in ts:
        t.start()
    for t in ts:
        t.join()
    e1.close()
    e2.close()
    assert not errs, errs
