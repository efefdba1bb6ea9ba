"""GPU parity: the full kv_validate_block pipeline vs the oracle on
generator blocks — codes, fees, and the muhash commitment, bit-exact."""
import ctypes
import os
import sys

import pytest

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", "oracle"))
from workload import gen_block  # noqa: E402

pytestmark = pytest.mark.gpu

SKIP_MASS = 2


@pytest.fixture(scope="module")
def engine():
    from rusty_kaspa_amd.engine import Engine
    eng = Engine()
    yield eng
    eng.close()


def oracle_validate(oracle, blob, n, flags=SKIP_MASS):
    codes = (ctypes.c_int32 * n)()
    fees = (ctypes.c_uint64 * n)()
    mh = (ctypes.c_uint8 * 32)()
    rc = oracle.ok_validate_block_parallel(blob, len(blob), 10**9, 10**9, flags,
                                           8, codes, fees, mh)
    assert rc == 0
    return list(codes), list(fees), bytes(mh)


def engine_validate(engine, blob, n, flags=SKIP_MASS):
    codes, fees, partial = engine.validate_block(blob, n, 10**9, 10**9, flags)
    mh = engine.muhash_finalize(partial)
    return codes, fees, mh


@pytest.mark.parametrize("kwargs,flags", [
    (dict(seed=10, n_txs=64), SKIP_MASS),                               # config-1 shape
    (dict(seed=11, n_txs=120, pct_multi_input=20, pct_ecdsa=10), SKIP_MASS),  # config-3
    (dict(seed=12, n_txs=80, pct_multi_input=20, pct_ecdsa=10,
          pct_multisig=10, pct_invalid=15), SKIP_MASS),                 # adversarial
    (dict(seed=13, n_txs=30, payload_len=500), SKIP_MASS),              # payloads
    (dict(seed=14, n_txs=60, pct_multi_input=30, pct_ecdsa=10), 0),     # FULL (KIP-9 mass)
    (dict(seed=16, n_txs=80, pct_multi_input=20, pct_ecdsa=15,
          pct_alt_hashtype=50), SKIP_MASS),                             # all 6 SigHashTypes
])
def test_validate_block_parity(oracle, engine, kwargs, flags):
    n = kwargs["n_txs"]
    blob, meta = gen_block(oracle, **kwargs)
    oc, of, omh = oracle_validate(oracle, blob, n, flags)
    ec, ef, emh = engine_validate(engine, blob, n, flags)
    assert ec == oc, [(i, a, b) for i, (a, b) in enumerate(zip(ec, oc)) if a != b][:5]
    assert ef == of
    assert emh == omh


def test_sighash_batch_parity(oracle, engine):
    import random
    rng = random.Random(5)
    n_txs = 20
    blob, _ = gen_block(oracle, seed=20, n_txs=n_txs, pct_multi_input=40)
    # pick random (tx, input, type, ecdsa) jobs
    from rusty_kaspa_amd.engine import KvParams  # noqa
    lib = engine.lib

    class Job(ctypes.Structure):
        _fields_ = [("tx_index", ctypes.c_uint32), ("input_index", ctypes.c_uint32),
                    ("hash_type", ctypes.c_uint8), ("ecdsa", ctypes.c_uint8),
                    ("_pad", ctypes.c_uint16)]

    types = [0x01, 0x02, 0x04, 0x81, 0x82, 0x84]
    jobs = []
    import struct
    for _ in range(64):
        t = rng.randrange(n_txs)
        # count inputs of tx t from the blob
        off = struct.unpack("<I", blob[4 + 4 * t:8 + 4 * t])[0]
        n_in = struct.unpack("<H", blob[off + 2:off + 4])[0]
        jobs.append((t, rng.randrange(n_in), rng.choice(types), rng.randrange(2)))
    arr = (Job * len(jobs))(*[Job(a, b, c, d, 0) for a, b, c, d in jobs])
    out = (ctypes.c_uint8 * (32 * len(jobs)))()
    rc = lib.kv_sighash_batch(ctypes.c_void_p(engine.ctx), blob, len(blob), arr,
                              len(jobs), out)
    assert rc == 0, lib.kv_last_error().decode()
    exp = (ctypes.c_uint8 * 32)()
    for i, (t, inp, ht, ec) in enumerate(jobs):
        assert oracle.ok_sighash(blob, len(blob), t, inp, ht, ec, exp) == 0
        assert bytes(out[32 * i:32 * i + 32]) == bytes(exp), (i, jobs[i])


def test_empty_and_edge_blobs(oracle, engine):
    # block with zero txs
    import struct
    blob = struct.pack("<I", 0)
    codes, fees, partial = engine.validate_block(blob, 0, 0, 0, SKIP_MASS)
    mh = engine.muhash_finalize(partial)
    assert mh.hex() == (
        "544eb3142c000f0ad2c76ac41f4222abbababed830eeafee4b6dc56b52d5cac0")


def test_malformed_blob_fails_cleanly(oracle, engine):
    """Truncated/corrupt blobs error out (<0 rc → RuntimeError) without
    wedging the context; the next valid call succeeds."""
    import pytest as _pytest
    blob, _ = gen_block(oracle, seed=15, n_txs=8)
    for cut in (0, 3, 10, len(blob) // 2, len(blob) - 1):
        with _pytest.raises(RuntimeError):
            engine.validate_block(blob[:cut], 8, 10**9, 10**9, SKIP_MASS)
    codes, _, _ = engine.validate_block(blob, 8, 10**9, 10**9, SKIP_MASS)
    assert all(c == 0 for c in codes)


def test_non_template_script_fallback(oracle, engine):
    """A signature-free non-template script (sha256 hash puzzle) resolves
    through the host general interpreter inside classify (kv_script_host.inc)
    — engine codes/fees stay oracle-bit-exact for solve and fail cases."""
    import hashlib
    import struct
    n = 12
    blob, _ = gen_block(oracle, seed=17, n_txs=n)  # all 1-input P2PK txs

    def with_puzzle(blob, preimage):
        # rewrite tx 0's only input: spk = OpSHA256 <h32> OpEqual,
        # sig_script = push(preimage). 1-input tx → no signatures anywhere.
        b = bytearray(blob)
        n_txs, = struct.unpack_from("<I", b, 0)
        off, = struct.unpack_from("<I", b, 4)
        payload_len, = struct.unpack_from("<I", b, off + 36)
        p = off + 88 + payload_len  # input record
        sig_len, = struct.unpack_from("<I", b, p + 48)
        h = hashlib.sha256(b"secret").digest()
        new_sig = bytes([len(preimage)]) + preimage
        new_spk = bytes([0xA8, 0x20]) + h + bytes([0x87])
        entry_off = p + 52 + sig_len
        spk_len, = struct.unpack_from("<I", b, entry_off + 20)
        rest = bytes(b[entry_off + 24 + spk_len:])
        head = bytes(b[:p + 48])
        mid = bytes(b[entry_off:entry_off + 20])
        out = bytearray()
        out += head
        out += struct.pack("<I", len(new_sig)) + new_sig
        out += mid + struct.pack("<I", len(new_spk)) + new_spk
        out += rest
        # fix the offset table for the shifted following txs
        delta = len(out) - len(b)
        for t in range(1, n_txs):
            o, = struct.unpack_from("<I", bytes(out), 4 + 4 * t)
            struct.pack_into("<I", out, 4 + 4 * t, o + delta)
        return bytes(out)

    for preimage, expect0 in [(b"secret", 0), (b"wrong!", 101)]:
        pb = with_puzzle(blob, preimage)
        oc, of, omh = oracle_validate(oracle, pb, n)
        assert oc[0] == expect0, (oc[0], expect0)
        ec, ef, emh = engine_validate(engine, pb, n)
        assert ec == oc and ef == of and emh == omh


def test_validate_timings_exposed(oracle, engine):
    """kv_get_validate_timings reports per-kernel hipEvent times and job
    counts of the last validate call (the bench derives the block-path
    roofline from these)."""
    class KvTimings(ctypes.Structure):
        _fields_ = [("subhash_ms", ctypes.c_double),
                    ("s_assemble_ms", ctypes.c_double),
                    ("e_assemble_ms", ctypes.c_double),
                    ("schnorr_ms", ctypes.c_double),
                    ("ecdsa_ms", ctypes.c_double),
                    ("muhash_ms", ctypes.c_double),
                    ("n_schnorr", ctypes.c_uint64),
                    ("n_ecdsa", ctypes.c_uint64)]
    blob, _ = gen_block(oracle, seed=99, n_txs=64, pct_ecdsa=20)
    engine.validate_block(blob, 64, 10**9, 10**9, SKIP_MASS)
    tm = KvTimings()
    assert engine.lib.kv_get_validate_timings(ctypes.c_void_p(engine.ctx),
                                              ctypes.byref(tm)) == 0
    assert tm.n_schnorr > 0 and tm.n_ecdsa > 0
    assert tm.subhash_ms > 0 and tm.schnorr_ms > 0 and tm.ecdsa_ms > 0
    assert tm.muhash_ms > 0  # muhash partial requested by default wrapper
