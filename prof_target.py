import ctypes, os, sys
os.chdir(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, '.')
sys.path.insert(0, 'oracle')
O = ctypes.CDLL("oracle/liboracle.so")
from rusty_kaspa_amd.engine import Engine
n = 262144
buf = ctypes.create_string_buffer(n*128)
O.ok_gen_schnorr_tuples(ctypes.c_uint64(1), ctypes.c_size_t(n), 0, buf, os.cpu_count() or 16)
eng = Engine(); lib = eng.lib; ctx = ctypes.c_void_p(eng.ctx)
lib.kv_stage_tuples(ctx, buf, ctypes.c_size_t(n), 0)
ms = ctypes.c_double()
for _ in range(4):
    lib.kv_verify_staged(ctx, ctypes.c_size_t(n), 0, ctypes.byref(ms))
print("verify kernel ms:", ms.value, "->", n/ms.value*1000, "verifies/s")
# one block-validate pass so its kernels appear in the trace
from workload import gen_block
blob, _ = gen_block(O, seed=30, n_txs=256, pct_multi_input=20, pct_ecdsa=10)
codes, fees, partial = eng.validate_block(blob, 256, 10**9, 10**9, 2)
print("validate ok:", sum(1 for c in codes if c == 0), "/", 256)
eng.close()
