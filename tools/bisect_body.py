import ctypes, os, sys, struct, json
sys.path.insert(0, "/root/repo"); sys.path.insert(0, "/root/repo/tests")
os.chdir("/root/repo")
from rusty_kaspa_amd.engine import Engine
import rusty_kaspa_amd.blob as B
from rusty_kaspa_amd.blob import strip_utxo_entries
oracle = ctypes.CDLL("oracle/liboracle.so")
g = json.load(open("tests/golden/merkle.json"))
gold = bytes.fromhex(g["blob_mass0"])

def body_check(tag):
    eng = Engine()
    try:
        root, code = eng.block_body_check(gold)
        print(f"{tag}: body_check OK code={code}")
    except Exception as e:
        print(f"{tag}: body_check FAILED: {e}")
    eng.close()

body_check("baseline")

def push(d): return (bytes([len(d)]) + d) if d else b"\x00"
key = bytes([21]) * 31 + b"\x01"
pk = (ctypes.c_uint8 * 32)(); oracle.ok_pubkey_xonly(key, pk); pk = bytes(pk)
spk_long = b"\x61" * 60 + push(pk) + b"\xac"
prev = bytes([7]) * 32
outpoint = prev + struct.pack("<I", 0)
entry = struct.pack("<QQHHI", 70_000, 5, 0, 0, len(spk_long)) + bytes(40)

# step 1: upsert_spk only
eng = Engine()
lib = eng.lib
lib.kv_utxo_reset(ctypes.c_void_p(eng.ctx), ctypes.c_uint64(256))
assert lib.kv_utxo_upsert_spk(ctypes.c_void_p(eng.ctx), outpoint, entry, spk_long,
                              ctypes.c_size_t(len(spk_long)), ctypes.c_size_t(1)) == 0
eng.close()
body_check("after upsert_spk")

# step 2: + mempool from table (interp + arena)
eng = Engine()
lib = eng.lib
lib.kv_utxo_reset(ctypes.c_void_p(eng.ctx), ctypes.c_uint64(256))
lib.kv_utxo_upsert_spk(ctypes.c_void_p(eng.ctx), outpoint, entry, spk_long,
                       ctypes.c_size_t(len(spk_long)), ctypes.c_size_t(1))
tx = B.tx_dict(1, [B.tx_input(prev, 0, sequence=2**64-1,
               sig_script=push(bytes(64) + b"\x01"), commit_kind=0, commit_value=20,
               utxo=B.utxo_entry(70_000, spk_long, daa_score=5))],
               [B.tx_output(60_000, b"\x51")])
blob = B.build_blob([tx])
msg = (ctypes.c_uint8 * 32)(); oracle.ok_sighash(blob, len(blob), 0, 0, 1, 0, msg)
sig = (ctypes.c_uint8 * 64)(); oracle.ok_schnorr_sign(key, msg, None, sig)
ins = list(tx["inputs"]); ins[0] = dict(ins[0], sig_script=push(bytes(sig) + b"\x01"))
blob = B.build_blob([dict(tx, inputs=ins)])
stripped, _ = strip_utxo_entries(blob)
codes, fees = eng.validate_mempool(stripped, 1, 10**9, feerate_threshold=0.0,
                                   from_utxo_table=True)
print("mempool codes:", codes)
eng.close()
body_check("after mempool+interp+arena")

# step 3: same but body_check on the SAME engine before close
eng = Engine()
lib = eng.lib
lib.kv_utxo_reset(ctypes.c_void_p(eng.ctx), ctypes.c_uint64(256))
lib.kv_utxo_upsert_spk(ctypes.c_void_p(eng.ctx), outpoint, entry, spk_long,
                       ctypes.c_size_t(len(spk_long)), ctypes.c_size_t(1))
eng.validate_mempool(stripped, 1, 10**9, feerate_threshold=0.0, from_utxo_table=True)
try:
    root, code = eng.block_body_check(gold)
    print("same-engine body_check OK")
except Exception as e:
    print("same-engine body_check FAILED:", e)
eng.close()
