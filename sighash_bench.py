"""Sighash-kernel roofline measurement: a 100k-input batch through the
subhash + assemble kernels (the bandwidth-facing kernels of SURVEY §8d).
Run under rocprofv3 --kernel-trace; GB/s derived from the dispatch times.
Sig scripts are dummies — the sighash message does not depend on them."""
import ctypes, os, sys, struct, time
os.chdir(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, '.')
from rusty_kaspa_amd import blob as B
from rusty_kaspa_amd.engine import Engine

N = 100_000
txs = []
prev = bytes(range(32))
spk = bytes([0x20]) + bytes(32) + bytes([0xAC])
for t in range(N):
    pid = struct.pack("<Q", t) + prev[8:]
    inp = B.tx_input(pid, t & 3, sig_script=bytes([0x41]) + bytes(65),
                     commit_kind=0, commit_value=1,
                     utxo=B.utxo_entry(10_000 + t, spk, 42))
    txs.append(B.tx_dict(0, [inp], [B.tx_output(9_000, spk)], tx_id=pid))
blob = B.build_blob(txs)
print(f"blob: {len(blob)/1e6:.1f} MB, {N} txs/inputs", flush=True)

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
# spot-check two hashes vs oracle
O = ctypes.CDLL("oracle/liboracle.so")
exp = (ctypes.c_uint8 * 32)()
for t in (0, N - 1):
    assert O.ok_sighash(blob, len(blob), t, 0, 0x01, 0, exp) == 0
    assert bytes(out[32*t:32*t+32]) == bytes(exp), t
print("sighash parity spot-check OK", flush=True)
eng.close()
