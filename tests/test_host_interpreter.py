"""The PRODUCT host interpreter (kv_script_host.inc — the non-template
fallback inside classify_input) differentially tested against the oracle:
every canonical script vector row, plus random-script fuzz. Rows reaching a
signature opcode DEFER (KVH_SCRIPT_DEFER) by design — the GPU owns EC."""
import ctypes
import json
import os
import random
import subprocess
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
from test_script_vectors import (BuildError, CODE_NAMES, GOLD, assemble,
                                 build_case_blob)  # noqa: E402

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
SHIM_DIR = os.path.join(REPO, "tests", "host_shim")
LIB = os.path.join(SHIM_DIR, "libscriptshim.so")
DEFER = -100
UNSUPPORTED = 63
# opcodes whose oracle path is supported but the product defers (sig family,
# blake3, seq-commit, zk)
DEFER_OPS = {0xa6, 0xa9, 0xab, 0xac, 0xad, 0xae, 0xaf, 0xd4, 0xd7, 0xd8,
             0xd9, 0xda}


@pytest.fixture(scope="module")
def shim():
    src = os.path.join(SHIM_DIR, "script_main.cpp")
    deps = [src,
            os.path.join(REPO, "rusty_kaspa_amd", "csrc", "kv_script_host.inc"),
            os.path.join(REPO, "rusty_kaspa_amd", "csrc", "kv_validate_host.inc")]
    if (not os.path.exists(LIB)
            or os.path.getmtime(LIB) < max(os.path.getmtime(p) for p in deps)):
        subprocess.run(["g++", "-O1", "-fPIC", "-shared",
                        "-I", os.path.join(REPO, "include"),
                        "-I", os.path.join(REPO, "rusty_kaspa_amd", "csrc"),
                        src, "-o", LIB], check=True)
    lib = ctypes.CDLL(LIB)
    lib.host_run_input_script.restype = ctypes.c_int
    return lib


def run_product(shim, blob):
    return shim.host_run_input_script(blob, ctypes.c_size_t(len(blob)), 0, 0,
                                      ctypes.c_uint64(1000))


def test_vector_suite_differential(oracle, shim):
    blk = b"input_block".ljust(32, b"f")
    com = b"output_root_hash".ljust(32, b"f")
    oracle.ok_script_set_seq_commit_mock(blk, com)
    rows = json.load(open(os.path.join(GOLD, "script_tests.json")))
    ran = deferred = 0
    failures = []
    for row in rows:
        if len(row) < 4 or not isinstance(row[0], str):
            continue
        try:
            sig = assemble(row[0])
            spk = assemble(row[1])
        except (BuildError, AssertionError):
            continue
        blob = build_case_blob(oracle, sig, spk)
        ocode = oracle.ok_check_input_script(blob, ctypes.c_size_t(len(blob)),
                                             0, 0)
        pcode = run_product(shim, blob)
        if pcode == DEFER:
            deferred += 1
            continue
        if pcode != ocode:
            failures.append((row[0][:40], row[1][:60], ocode, pcode))
        else:
            ran += 1
    print(f"\n[product interpreter] {ran} rows identical to the oracle, "
          f"{deferred} deferred (signature family), {len(failures)} diverged")
    assert not failures, failures[:10]
    assert ran > 700


def test_random_script_fuzz(oracle, shim):
    """Random byte scripts: the product interpreter and the oracle must agree
    on every outcome (or the product defers on a sig opcode)."""
    rng = random.Random(99)
    agreed = deferred = 0
    mismatches = []
    for trial in range(12000):
        slen = rng.randrange(0, 40)
        spk = bytes(rng.randrange(256) for _ in range(slen))
        sig_len = rng.randrange(0, 30)
        # push-only-ish sig script: random small pushes
        sig = bytearray()
        while len(sig) < sig_len:
            dl = rng.randrange(0, 8)
            sig.append(dl)
            sig.extend(rng.randrange(256) for _ in range(dl))
        blob = build_case_blob(oracle, bytes(sig), spk)
        ocode = oracle.ok_check_input_script(blob, ctypes.c_size_t(len(blob)),
                                             0, 0)
        pcode = run_product(shim, blob)
        if pcode == DEFER:
            deferred += 1
            continue
        if pcode != ocode:
            mismatches.append((sig.hex(), spk.hex(), ocode, pcode))
        else:
            agreed += 1
    print(f"\n[fuzz] {agreed} agreed, {deferred} deferred, "
          f"{len(mismatches)} mismatched")
    assert not mismatches, mismatches[:5]
