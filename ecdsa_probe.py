"""Standalone ECDSA staged-verify rate (GLV + chain sc_inv), for DESIGN.md."""
import ctypes, os, sys
os.chdir(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, '.')
O = ctypes.CDLL("oracle/liboracle.so")
from rusty_kaspa_amd.engine import Engine
n = 1 << 20
buf = ctypes.create_string_buffer(n * 132)
O.ok_gen_ecdsa_tuples(ctypes.c_uint64(9), ctypes.c_size_t(n), 0, buf, 32)
eng = Engine(); lib = eng.lib; ctx = ctypes.c_void_p(eng.ctx)
assert lib.kv_stage_tuples(ctx, buf, ctypes.c_size_t(n), 1) == 0
ms = ctypes.c_double()
for _ in range(2):
    lib.kv_verify_staged(ctx, ctypes.c_size_t(n), 1, ctypes.byref(ms))
best = 1e9
for _ in range(5):
    lib.kv_verify_staged(ctx, ctypes.c_size_t(n), 1, ctypes.byref(ms))
    best = min(best, ms.value)
bm = (ctypes.c_uint64 * ((n + 63) // 64))()
lib.kv_fetch_bitmap(ctx, ctypes.c_size_t(n), bm)
ok = sum(bin(w).count('1') for w in bm)
print(f"ecdsa: {best:.2f} ms -> {n/best*1000/1e6:.2f} M/s, valid {ok}/{n}")
eng.close()
