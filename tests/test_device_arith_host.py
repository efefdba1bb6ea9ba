"""Device EC arithmetic fuzzed ON CPU: the HIP headers compile under
tests/host_shim (g++, KV_HOST_TEST=magnitude asserts live) and the exact same
code paths the GPU runs are checked against the oracle and exact bigint
arithmetic. This is the validation harness every kernel change went through
before touching the GPU."""
import ctypes
import os
import random
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
SHIM_DIR = os.path.join(REPO, "tests", "host_shim")
LIB = os.path.join(SHIM_DIR, "libhostshim.so")

P = 2**256 - 0x1000003D1


@pytest.fixture(scope="module")
def shim(oracle):
    srcs = [os.path.join(SHIM_DIR, "main.cpp")]
    hdrs = [os.path.join(REPO, "rusty_kaspa_amd", "csrc", "kv_secp_device.h"),
            os.path.join(SHIM_DIR, "shim.h")]
    if (not os.path.exists(LIB)
            or os.path.getmtime(LIB) < max(os.path.getmtime(p)
                                           for p in srcs + hdrs)):
        subprocess.run(["g++", "-O1", "-fPIC", "-shared",
                        "-I", os.path.join(REPO, "rusty_kaspa_amd", "csrc"),
                        "-I", SHIM_DIR, srcs[0], "-o", LIB], check=True)
    lib = ctypes.CDLL(LIB)
    lib.host_init_gtable()
    return lib


def rnd_limbs(rng, m):
    return [rng.randrange(m * (1 << 26)) for _ in range(9)] + \
           [rng.randrange(m * (1 << 22))]


def val(a):
    return sum(a[i] << (26 * i) for i in range(10))


def test_fe26_mul_sqr_vs_bigint(shim):
    rng = random.Random(11)
    arr = lambda l: (ctypes.c_uint32 * 10)(*l)
    for _ in range(8000):
        m = rng.choice([1, 2, 4, 8])
        a, b = rnd_limbs(rng, m), rnd_limbs(rng, m)
        r = (ctypes.c_uint32 * 10)()
        shim.host_fe26_mul(arr(a), arr(b), r)
        assert val(r) % P == (val(a) * val(b)) % P
        shim.host_fe26_sqr(arr(a), r)
        assert val(r) % P == (val(a) ** 2) % P


def test_fe26_normalize_neg_roundtrip(shim):
    rng = random.Random(12)
    arr = lambda l: (ctypes.c_uint32 * 10)(*l)
    for _ in range(3000):
        m = rng.choice([1, 8, 16, 31])
        l = rnd_limbs(rng, m)
        a = arr(l)
        shim.host_fe26_normalize(a)
        assert val(a) == val(l) % P and val(a) < P
    for _ in range(2000):
        m = rng.choice([1, 2, 8, 15])
        l = rnd_limbs(rng, m)
        r = (ctypes.c_uint32 * 10)()
        shim.host_fe26_neg_norm(arr(l), ctypes.c_uint32(m), r)
        assert val(r) == (-val(l)) % P
    for _ in range(2000):
        v = random.Random(13).randrange(P)
        a = (ctypes.c_uint64 * 4)(*[(v >> (64 * i)) & (2**64 - 1)
                                    for i in range(4)])
        r = (ctypes.c_uint64 * 4)()
        shim.host_fe26_roundtrip(a, r)
        assert sum(r[i] << (64 * i) for i in range(4)) == v


def test_schnorr_ecdsa_vs_oracle(shim, oracle):
    n = 512
    tu = ctypes.create_string_buffer(n * 128)
    oracle.ok_gen_schnorr_tuples(ctypes.c_uint64(21), ctypes.c_size_t(n), 100,
                                 tu, 8)
    exp = (ctypes.c_uint64 * ((n + 63) // 64))()
    oracle.ok_verify_schnorr_batch(tu, ctypes.c_size_t(n), 8, exp)
    raw = tu.raw
    for i in range(n):
        got = shim.host_schnorr_verify(raw[128 * i + 64:128 * i + 96],
                                       raw[128 * i + 96:128 * i + 128],
                                       raw[128 * i:128 * i + 64])
        assert got == ((exp[i // 64] >> (i % 64)) & 1), i
    ne = 256
    etu = ctypes.create_string_buffer(ne * 132)
    oracle.ok_gen_ecdsa_tuples(ctypes.c_uint64(22), ctypes.c_size_t(ne), 150,
                               etu, 8)
    eraw = etu.raw
    for i in range(ne):
        t = eraw[132 * i:132 * (i + 1)]
        got = shim.host_ecdsa_verify(t[64:97], t[97:129], t[:64])
        expd = oracle.ok_ecdsa_verify(t[64:97], t[97:129], t[:64])
        assert got == expd, i


def test_fe26_inv_sqrt_vs_pow(shim):
    """Addition-chain Fermat powers (fe26_inv, fe26_sqrt) against python pow."""
    rng = random.Random(55)
    arr = lambda l: (ctypes.c_uint32 * 10)(*l)
    M26 = (1 << 26) - 1
    to26 = lambda v: [(v >> (26 * i)) & M26 for i in range(10)]
    for _ in range(300):
        v = rng.randrange(1, P)
        r = (ctypes.c_uint32 * 10)()
        shim.host_fe26_inv(arr(to26(v)), r)
        assert val(r) == pow(v, P - 2, P)
        ok = shim.host_fe26_sqrt(arr(to26(v)), r)
        is_qr = pow(v, (P - 1) // 2, P) == 1
        assert bool(ok) == is_qr
        if ok:
            assert (val(r) * val(r)) % P == v
