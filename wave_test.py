import ctypes, os, sys
os.chdir(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, '.'); sys.path.insert(0, 'oracle')
O = ctypes.CDLL("oracle/liboracle.so")
from rusty_kaspa_amd.engine import Engine
from workload import gen_block
import random, time
eng = Engine()
# full block parity (muhash now computed via the wave reduce)
for seed, kw in [(12, dict(n_txs=80, pct_multi_input=20, pct_ecdsa=10, pct_multisig=10, pct_invalid=15)),
                 (31, dict(n_txs=300, pct_multi_input=20, pct_ecdsa=10))]:
    blob, _ = gen_block(O, seed=seed, **kw)
    n = kw["n_txs"]
    codes, fees, partial = eng.validate_block(blob, n, 10**9, 10**9, 2)
    mh = eng.muhash_finalize(partial)
    ocodes = (ctypes.c_int32*n)(); ofees = (ctypes.c_uint64*n)(); omh = (ctypes.c_uint8*32)()
    O.ok_validate_block_parallel(blob, len(blob), 10**9, 10**9, 2, 16, ocodes, ofees, omh)
    assert list(codes) == list(ocodes), "codes mismatch"
    assert mh == bytes(omh), f"muhash mismatch seed {seed}"
    print(f"seed {seed}: wave-reduce muhash parity OK")
# block bench quick
blob, _ = gen_block(O, seed=42, n_txs=4800, pct_multi_input=20, pct_ecdsa=10)
for _ in range(2):
    t0 = time.perf_counter()
    codes, fees, partial = eng.validate_block(blob, 4800, 10**9, 10**9, 0)
    mh = eng.muhash_finalize(partial)
    dt = time.perf_counter() - t0
print(f"block step: {dt*1000:.1f} ms -> {4800/dt/1000:.1f}k txs/s")
eng.close()
