import ctypes, sys
opt = sys.argv[1] if len(sys.argv)>1 else "O3"
mod = ctypes.CDLL(f"./rusty_kaspa_amd/libkvdebug_{opt}.so")
O = ctypes.CDLL("oracle/liboracle.so")
print(f"=== {opt} pure-arith ecmult ===", flush=True)
mod.kv_debug_run_pure(45)
print(f"=== {opt} full verify ===", flush=True)
tup = ctypes.create_string_buffer(128)
O.ok_gen_schnorr_tuples(ctypes.c_uint64(5), ctypes.c_size_t(1), 0, tup, 1)
mod.kv_debug_run(tup, 45)
