#!/bin/bash
export PYTHONUNBUFFERED=1
timeout 400 python -m pytest tests/ -m gpu -q 2>&1 | tail -2
echo "=== block bench after reduce fix ==="
timeout 250 python bench.py --mode block --block-batch 16 --steps 8 --warmup 2 --skip-cpu-baseline 2>&1 | tail -1
echo "=== adversarial verify 10pct invalid ==="
timeout 200 python bench.py --tuples 262144 --invalid-permille 100 --steps 6 --warmup 2 --skip-cpu-baseline 2>&1 | tail -1
echo "=== ecdsa staged ==="
timeout 200 python - <<'PYEOF'
import ctypes
O = ctypes.CDLL("oracle/liboracle.so")
from rusty_kaspa_amd.engine import Engine
n = 262144
buf = ctypes.create_string_buffer(n*132)
O.ok_gen_ecdsa_tuples(ctypes.c_uint64(2), ctypes.c_size_t(n), 100, buf, 32)
eng = Engine(); lib = eng.lib; ctx = ctypes.c_void_p(eng.ctx)
lib.kv_stage_tuples(ctx, buf, ctypes.c_size_t(n), 1)
ms = ctypes.c_double()
for _ in range(3): lib.kv_verify_staged(ctx, ctypes.c_size_t(n), 1, ctypes.byref(ms))
print(f"ecdsa: {ms.value:.1f} ms -> {n/ms.value*1000/1e6:.2f} M/s")
eng.close()
PYEOF
