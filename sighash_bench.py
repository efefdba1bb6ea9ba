"""Sighash-kernel roofline measurement at SCALE: a 1M-input batch through the
subhash + assemble kernels (the bandwidth-facing kernels of SURVEY §8d; the
north star demands achieved HBM GB/s here). Run standalone for hipEvent
timings, or under rocprofv3 (--stats / --pmc FETCH_SIZE / --pmc WRITE_SIZE,
separate passes) for the PMC traffic.

The blob is built with numpy (identical fixed-size 315B tx records, per-tx
prevout patched) so construction takes seconds, not minutes; sig scripts are
dummies — the sighash message does not depend on them.

Algorithmic bytes per input (1-in/1-out tx, derivation in DESIGN.md):
  subhash : reads the 315B tx record once, writes 160B of subhashes
  assemble: reads ~250B (record fields + 160B subhashes) + writes 128B tuple
"""
import ctypes
import os
import struct
import sys
import time

import numpy as np

os.chdir(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, '.')
from rusty_kaspa_amd import blob as B
from rusty_kaspa_amd.engine import Engine

N = int(os.environ.get("SIGHASH_N", 1_000_000))

# one template tx record via the reference builder, then numpy-tile
spk = bytes([0x20]) + bytes(32) + bytes([0xAC])
tmpl_tx = B.tx_dict(0, [B.tx_input(bytes(32), 0,
                                   sig_script=bytes([0x41]) + bytes(65),
                                   commit_kind=0, commit_value=1,
                                   utxo=B.utxo_entry(10_000, spk, 42))],
                    [B.tx_output(9_000, spk)], tx_id=bytes(32))
one = B.build_blob([tmpl_tx])
rec = one[8:]  # skip n_txs u32 + 1 offset u32
REC = len(rec)
assert REC == 315, REC

t0 = time.time()
header = struct.pack("<I", N) + np.arange(
    4 + 4 * N, 4 + 4 * N + N * REC, REC, dtype=np.uint32).tobytes()
body = np.tile(np.frombuffer(rec, dtype=np.uint8), N)
# patch per-tx prevout tx-id (offset 88 in the record) and carried tx_id
# (offset 56) with the counter so every tx/input is distinct
ctr = np.arange(N, dtype=np.uint64)
for off in (56, 88):
    view = np.lib.stride_tricks.as_strided(
        body[off:], shape=(N, 8), strides=(REC, 1))
    view[:] = ctr.view(np.uint8).reshape(N, 8)
blob = header + body.tobytes()
print(f"blob: {len(blob)/1e6:.1f} MB, {N} txs/inputs, built in "
      f"{time.time()-t0:.1f}s", flush=True)


class Job(ctypes.Structure):
    _fields_ = [("tx_index", ctypes.c_uint32), ("input_index", ctypes.c_uint32),
                ("hash_type", ctypes.c_uint8), ("ecdsa", ctypes.c_uint8),
                ("_pad", ctypes.c_uint16)]


jobs = (Job * N)(*[Job(t, 0, 0x01, 0, 0) for t in range(N)])
out = (ctypes.c_uint8 * (32 * N))()
eng = Engine()
lib = eng.lib
for rep in range(3):
    t0 = time.perf_counter()
    rc = lib.kv_sighash_batch(ctypes.c_void_p(eng.ctx), blob, len(blob), jobs,
                              ctypes.c_size_t(N), out)
    dt = time.perf_counter() - t0
    assert rc == 0, lib.kv_last_error().decode()
print(f"sighash batch e2e (incl. H2D blob + D2H tuples): {dt*1000:.1f} ms "
      f"= {N/dt/1e6:.2f}M sighashes/s", flush=True)
# spot-check one hash against the oracle
O = ctypes.CDLL("oracle/liboracle.so")
exp = (ctypes.c_uint8 * 32)()
probe = N // 2
assert O.ok_sighash(blob, len(blob), probe, 0, 0x01, 0, exp) == 0
assert bytes(out[32 * probe:32 * probe + 32]) == bytes(exp), "sighash mismatch"
print("oracle spot-check OK", flush=True)
eng.close()
