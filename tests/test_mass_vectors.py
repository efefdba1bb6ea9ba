"""KIP-9 storage mass / plurality / cofactor golden vectors.

Pins BOTH the oracle's and the engine's mass restatements against the
reference's own test vectors (consensus/core/src/mass/mod.rs:531-953),
extracted into tests/golden/mass.json by tests/golden/extract_mass.py.
CPU-only: the engine's mass helpers are host code exposed via kv_test_*.
"""
import ctypes
import json
import os

import pytest

HERE = os.path.dirname(os.path.abspath(__file__))
REPO = os.path.dirname(HERE)

with open(os.path.join(HERE, "golden", "mass.json")) as f:
    VECTORS = json.load(f)


def _sig_setup(lib):
    lib.ok_test_plurality.restype = ctypes.c_uint64
    lib.ok_test_plurality.argtypes = [ctypes.c_uint32, ctypes.c_int]
    lib.ok_normalized_max_limits.restype = ctypes.c_uint64
    lib.ok_normalized_max_limits.argtypes = [ctypes.c_uint64] * 6


@pytest.fixture(scope="module", params=["oracle", "engine"])
def massfns(request):
    """(plurality, storage_mass, normalized_max) callables for each impl."""
    if request.param == "oracle":
        lib = ctypes.CDLL(os.path.join(REPO, "oracle", "liboracle.so"))
        plu, sm, nm = (lib.ok_test_plurality, lib.ok_test_storage_mass,
                       lib.ok_normalized_max_limits)
    else:
        lib = ctypes.CDLL(os.path.join(REPO, "rusty_kaspa_amd", "libkaspa_gpu.so"))
        plu, sm, nm = (lib.kv_test_plurality, lib.kv_test_storage_mass,
                       lib.kv_test_normalized_max)
    plu.restype = ctypes.c_uint64
    plu.argtypes = [ctypes.c_uint32, ctypes.c_int]
    nm.restype = ctypes.c_uint64
    nm.argtypes = [ctypes.c_uint64] * 6
    sm.restype = ctypes.c_int

    def storage_mass(ins, outs):
        """ins/outs: lists of (amount, spk_len, has_cov). -> mass or None."""
        n_i, n_o = len(ins), len(outs)
        ia = (ctypes.c_uint64 * n_i)(*[a for a, _, _ in ins])
        il = (ctypes.c_uint32 * n_i)(*[l for _, l, _ in ins])
        ic = (ctypes.c_uint8 * n_i)(*[1 if c else 0 for _, _, c in ins])
        oa = (ctypes.c_uint64 * n_o)(*[a for a, _, _ in outs])
        ol = (ctypes.c_uint32 * n_o)(*[l for _, l, _ in outs])
        oc = (ctypes.c_uint8 * n_o)(*[1 if c else 0 for _, _, c in outs])
        out = ctypes.c_uint64()
        rc = sm(ctypes.c_uint32(n_i), ia, il, ic, ctypes.c_uint32(n_o), oa, ol, oc,
                ctypes.byref(out))
        assert rc in (0, -1), rc
        return None if rc == -1 else out.value

    return plu, storage_mass, nm


def test_plurality_vectors(massfns):
    plu, _, _ = massfns
    for case in VECTORS["plurality"]:
        got = plu(case["spk_len"], 1 if case["has_covenant"] else 0)
        assert got == case["expected"], case


def test_storage_mass_vectors(massfns):
    _, storage_mass, _ = massfns
    assert VECTORS["storm_param"] == 10**12  # both impls hardwire mainnet C
    for case in VECTORS["storage_mass"]:
        assert case["storm"] == 10**12
        got = storage_mass([(a, 0, False) for a in case["ins"]],
                           [(a, 0, False) for a in case["outs"]])
        assert got == case["expected"], case


def test_storage_mass_plurality_equalities(massfns):
    """Super-entry equivalence: a plurality-p entry behaves like p sub-entries
    (test_storage_mass_pluralities, mass/mod.rs:631-748)."""
    _, storage_mass, _ = massfns
    for case in VECTORS["plurality_equalities"]:
        ins1 = [(a, 0, False) for a in case["ins1"]]
        outs1 = [(a, 0, False) for a in case["outs1"]]
        ins2 = [(a, 0, False) for a in case["ins2"]]
        outs2 = [(a, 0, False) for a in case["outs2"]]
        idx, spk = case["override_index"], case["override_spk_len"]
        if case["override_output"]:
            outs2[idx] = (outs2[idx][0], spk, False)
        else:
            ins2[idx] = (ins2[idx][0], spk, False)
        m1 = storage_mass(ins1, outs1)
        m2 = storage_mass(ins2, outs2)
        assert m1 == m2 == case["expected"] and m1 not in (0, None), case["name"]


def test_cofactor_vectors(massfns):
    """cofactors + the normalized_max invariant: filling one dimension to its
    raw limit normalizes to the reference (test_mass_cofactors)."""
    _, _, nm = massfns
    for case in VECTORS["cofactors"]:
        ls, lc, lt = case["storage"], case["compute"], case["transient"]
        ref = case["reference"]
        assert ref == lc
        assert nm(ls, 0, 0, ls, lc, lt) == ref, case
        assert nm(0, lc, 0, ls, lc, lt) == ref, case
        assert nm(0, 0, lt, ls, lc, lt) == ref, case


def test_normalized_max_ranking(massfns):
    """Exact normalized values + bottleneck ranking (test_normalized_max_ranking)."""
    _, _, nm = massfns
    got = []
    for case in VECTORS["normalized_max"]:
        ls, lc, lt = case["limits"]
        v = nm(case["storage_mass"], case["compute_mass"], case["transient_mass"],
               ls, lc, lt)
        assert v == case["expected"], case
        got.append(v)
    # tx_c (80% transient) > tx_b (60% compute) > tx_a (50% storage)
    assert got[2] > got[1] > got[0]
