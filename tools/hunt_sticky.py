import ctypes, os, sys, struct
sys.path.insert(0, "/root/repo"); sys.path.insert(0, "/root/repo/tests")
os.chdir("/root/repo")
hip = ctypes.CDLL("libamdhip64.so")
hip.hipGetErrorString.restype = ctypes.c_char_p

def probe(tag):
    e = hip.hipGetLastError()
    if e != 0:
        print(f"STICKY after {tag}: {e} {hip.hipGetErrorString(e).decode()}", flush=True)
    else:
        print(f"clean after {tag}", flush=True)

sys.path.insert(0, os.path.join("tests"))
from conftest import load_oracle
oracle = load_oracle()
import importlib
tm = importlib.import_module("test_gpu_mempool")
from rusty_kaspa_amd.engine import Engine
eng = Engine()
probe("engine create")
import pytest
# replicate module order
for th in getattr(tm, "THRESHOLDS", None) or [0.0]:
    pass
# call the actual test functions with fixtures by hand
import inspect
thresholds = []
for mark in getattr(tm.test_mempool_vs_oracle, "pytestmark", []):
    if mark.name == "parametrize":
        thresholds = list(mark.args[1])
for th in thresholds or [0.0]:
    tm.test_mempool_vs_oracle(oracle, eng, th)
    probe(f"vs_oracle({th})")
tm.test_mempool_from_table(oracle, eng)
probe("from_table")
tm.test_mempool_from_table_arena_and_interp(oracle, eng)
probe("combo")
eng.close()
probe("engine close")
eng2 = Engine()
probe("engine2 create")
import json
g = json.load(open("tests/golden/merkle.json"))
gold = bytes.fromhex(g["blob_mass0"])
try:
    root, code = eng2.block_body_check(gold)
    print("body_check OK", flush=True)
except Exception as ex:
    print("body_check FAILED:", ex, flush=True)
probe("body_check")
eng2.close()
