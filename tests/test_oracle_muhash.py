"""Oracle MuHash/U3072 vs reference KATs (crypto/muhash/src/lib.rs:19-346)."""
import ctypes
import random

U = ctypes.c_uint64 * 48
OUT32 = ctypes.c_uint8 * 32


def element_from_byte(b):
    return bytes([b]) + bytes(31)


def finalize(oracle, num, den):
    n2, d2 = U(*num), U(*den)
    out = OUT32()
    oracle.ok_muhash_finalize(n2, d2, out)
    return bytes(out)


def test_empty_muhash(oracle, golden):
    g = golden("muhash.json")
    num, den = U(), U()
    oracle.ok_u3072_one(num)
    oracle.ok_u3072_one(den)
    assert finalize(oracle, num, den) == bytes(g["empty_muhash"])


def test_vectors_cumulative_and_multiset(oracle, golden):
    g = golden("muhash.json")
    elem = U()
    num, den = U(), U()
    oracle.ok_u3072_one(num)
    oracle.ok_u3072_one(den)
    for v in g["vectors"]:
        data = bytes(v["data"])
        oracle.ok_muhash_element(data, len(data), elem)
        # multiset: single-element set
        n1, d1 = U(), U()
        oracle.ok_u3072_one(n1)
        oracle.ok_u3072_one(d1)
        oracle.ok_u3072_mul(n1, elem)
        assert finalize(oracle, n1, d1) == bytes(v["multiset_hash"])
        # cumulative
        oracle.ok_u3072_mul(num, elem)
        assert finalize(oracle, num, den) == bytes(v["cumulative_hash"])


def test_add_remove_roundtrip(oracle, golden):
    g = golden("muhash.json")
    elem = U()
    num, den = U(), U()
    oracle.ok_u3072_one(num)
    oracle.ok_u3072_one(den)
    rng = random.Random(7)
    datas = [bytes(rng.randrange(256) for _ in range(100)) for _ in range(50)]
    for d in datas:
        oracle.ok_muhash_element(d, len(d), elem)
        oracle.ok_u3072_mul(num, elem)
    assert finalize(oracle, num, den) != bytes(g["empty_muhash"])
    for d in datas:
        oracle.ok_muhash_element(d, len(d), elem)
        oracle.ok_u3072_mul(den, elem)
    assert finalize(oracle, num, den) == bytes(g["empty_muhash"])


def test_precomputed_with_inverse(oracle, golden):
    g = golden("muhash.json")
    elem = U()
    num, den = U(), U()
    oracle.ok_u3072_one(num)
    oracle.ok_u3072_one(den)
    for b, target in ((0, num), (1, num), (2, den)):
        e = element_from_byte(b)
        oracle.ok_muhash_element(e, 32, elem)
        oracle.ok_u3072_mul(target, elem)
    assert finalize(oracle, num, den).hex() == g["precomputed_add0_add1_remove2"]


def test_serialize_vector(oracle, golden):
    g = golden("muhash.json")
    elem = U()
    num, den = U(), U()
    oracle.ok_u3072_one(num)
    oracle.ok_u3072_one(den)
    for b in (1, 2):
        e = element_from_byte(b)
        oracle.ok_muhash_element(e, 32, elem)
        oracle.ok_u3072_mul(num, elem)
    oracle.ok_u3072_div(num, den)
    ser = b"".join(int(num[i]).to_bytes(8, "little") for i in range(48))
    assert list(ser) == g["serialize_add1_add2"]


def test_u3072_mul_vs_python_bigint(oracle):
    """Randomized cross-check of the reduce-as-you-go multiply against python
    bigint arithmetic (independent implementation)."""
    P = 2**3072 - 1103717
    rng = random.Random(3)
    for _ in range(10):
        a = rng.getrandbits(3072)
        b = rng.getrandbits(3072)
        ua = U(*[(a >> (64 * i)) & (2**64 - 1) for i in range(48)])
        ub = U(*[(b >> (64 * i)) & (2**64 - 1) for i in range(48)])
        oracle.ok_u3072_mul(ua, ub)
        got = sum(int(ua[i]) << (64 * i) for i in range(48))
        # engine state may be in [0, 2^3072); reduce both for comparison
        assert got % P == (a * b) % P


def test_u3072_inverse_edge_case(oracle, golden):
    g = golden("u3072.json")
    limbs = g["inverse_edge_case_limbs"]
    P = 2**3072 - 1103717
    a = sum(v << (64 * i) for i, v in enumerate(limbs))
    ua = U(*limbs)
    one = U()
    oracle.ok_u3072_one(one)
    # div: one/a = a^-1; then multiply back
    oracle.ok_u3072_div(one, ua)
    inv = sum(int(one[i]) << (64 * i) for i in range(48))
    assert (inv * a) % P == 1
