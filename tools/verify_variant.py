#!/usr/bin/env python3
"""Load a given engine .so variant and time the staged schnorr verify kernel
(262144 tuples x 4 passes), parity-checked against the oracle. Perf-experiment
driver: python tools/verify_variant.py <path-to-.so> [n_tuples]."""
import ctypes
import os
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


class KvParams(ctypes.Structure):
    _fields_ = [("coinbase_maturity", ctypes.c_uint64),
                ("mass_per_sig_op", ctypes.c_uint64),
                ("sig_cache_size", ctypes.c_uint64),
                ("device", ctypes.c_int)]


def main():
    so = sys.argv[1]
    n = int(sys.argv[2]) if len(sys.argv) > 2 else 262144
    O = ctypes.CDLL(os.path.join(REPO, "oracle", "liboracle.so"))
    buf = ctypes.create_string_buffer(n * 128)
    O.ok_gen_schnorr_tuples(ctypes.c_uint64(1), ctypes.c_size_t(n), 0, buf,
                            os.cpu_count() or 16)
    lib = ctypes.CDLL(so)
    lib.kv_create.restype = ctypes.c_void_p
    p = KvParams(1000, 1000, 0, -1)
    ctx = ctypes.c_void_p(lib.kv_create(ctypes.byref(p)))
    assert ctx.value, "kv_create failed"
    assert lib.kv_stage_tuples(ctx, buf, ctypes.c_size_t(n), 0) == 0
    ms = ctypes.c_double()
    times = []
    for _ in range(4):
        assert lib.kv_verify_staged(ctx, ctypes.c_size_t(n), 0,
                                    ctypes.byref(ms)) == 0
        times.append(ms.value)
    words = (n + 63) // 64
    got = (ctypes.c_uint64 * words)()
    assert lib.kv_fetch_bitmap(ctx, ctypes.c_size_t(n), got) == 0
    sample = min(n, 1 << 14)
    exp = (ctypes.c_uint64 * words)()
    O.ok_verify_schnorr_batch(buf, ctypes.c_size_t(sample),
                              os.cpu_count() or 16, exp)
    for i in range(sample // 64):
        assert got[i] == exp[i], f"bitmap mismatch word {i}"
    best = min(times)
    print(f"{os.path.basename(so)}: n={n} best={best:.3f}ms "
          f"= {n/best*1000/1e6:.2f}M verifies/s (parity OK)", flush=True)
    lib.kv_destroy(ctx)


if __name__ == "__main__":
    main()
