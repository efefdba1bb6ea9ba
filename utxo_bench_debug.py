import ctypes, os, sys, time
os.chdir(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, '.'); sys.path.insert(0, 'oracle')
O = ctypes.CDLL("oracle/liboracle.so")
from workload import gen_block
from rusty_kaspa_amd.engine import Engine
from rusty_kaspa_amd.blob import strip_utxo_entries
n_txs = 32 * 300
blob, _ = gen_block(O, seed=1234, n_txs=n_txs, pct_multi_input=20, pct_ecdsa=10)
eng = Engine(sig_cache_size=0)
stripped, seeds = strip_utxo_entries(blob)
lib = eng.lib; ctx = ctypes.c_void_p(eng.ctx)
assert lib.kv_utxo_reset(ctx, ctypes.c_uint64(2*len(seeds))) == 0
assert lib.kv_utxo_upsert(ctx, b"".join(op for op,_ in seeds),
                          b"".join(e for _,e in seeds), ctypes.c_size_t(len(seeds))) == 0
for name, fn in [
    ("validate_utxo", lambda: eng.validate_block_utxo(stripped, n_txs, 10**9, 10**9, 0, apply_diff=False)),
    ("validate_inline", lambda: eng.validate_block(blob, n_txs, 10**9, 10**9, 0)),
]:
    fn(); fn()
    ts = []
    for _ in range(8):
        t0 = time.perf_counter(); c, f, p = fn(); ts.append(time.perf_counter()-t0)
    t0 = time.perf_counter(); mh = eng.muhash_finalize(p); tfin = time.perf_counter()-t0
    print(f"{name}: validate {min(ts)*1000:.2f}ms (min of 8), finalize {tfin*1000:.2f}ms")
eng.close()
