"""GPU UTXO arena: out-of-line scripts beyond the 36B inline slot
(⇔ UtxoEntry spk up to max_script_public_key_len = 10,000B,
consensus/core/src/utxo/utxo_entry.rs:20), plus beyond-Infinity-Cache table
scaling (config-5's >L3 rung)."""
import ctypes
import random
import struct
import sys
import os

import pytest

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import rusty_kaspa_amd.blob as B  # noqa: E402

pytestmark = pytest.mark.gpu

SKIP_MASS = 2
ARENA_FLAG = 2


@pytest.fixture(scope="module")
def engine():
    from rusty_kaspa_amd.engine import Engine
    eng = Engine()
    yield eng
    eng.close()


def pack_entry(amount, daa, coinbase, spk, spk_ver=0):
    e = struct.pack("<QQHHI", amount, daa, 1 if coinbase else 0, spk_ver,
                    len(spk))
    return e + (spk.ljust(36, b"\0")[:36] if len(spk) <= 36 else bytes(36)) \
        + bytes(4)


def upsert_spk(engine, items):
    """items: [(outpoint36, amount, daa, coinbase, spk)]"""
    lib = engine.lib
    ops = b"".join(it[0] for it in items)
    ents = b"".join(pack_entry(a, d, c, s) for _, a, d, c, s in items)
    blob = b"".join(s for _, _, _, _, s in items if len(s) > 36)
    rc = lib.kv_utxo_upsert_spk(ctypes.c_void_p(engine.ctx), ops, ents, blob,
                                ctypes.c_size_t(len(blob)),
                                ctypes.c_size_t(len(items)))
    assert rc == 0, lib.kv_last_error().decode()


def lookup_spk(engine, ops, spk_cap=1 << 20):
    lib = engine.lib
    n = len(ops)
    out = (ctypes.c_uint8 * (64 * n))()
    bm = (ctypes.c_uint64 * ((n + 63) // 64))()
    spk = (ctypes.c_uint8 * max(1, spk_cap))()
    used = ctypes.c_size_t()
    rc = lib.kv_utxo_lookup_spk(ctypes.c_void_p(engine.ctx), b"".join(ops),
                                ctypes.c_size_t(n), out, bm, spk,
                                ctypes.c_size_t(spk_cap), ctypes.byref(used))
    found = [(bm[i // 64] >> (i % 64)) & 1 for i in range(n)]
    return rc, found, bytes(out), bytes(spk[:used.value]), used.value


def test_arena_roundtrip_and_growth(engine):
    """Inline + 100B/1KB/10KB arena entries round-trip through upsert/lookup;
    the arena grows past its initial 1MB allocation; a too-small output
    buffer reports the needed size."""
    lib = engine.lib
    assert lib.kv_utxo_reset(ctypes.c_void_p(engine.ctx),
                             ctypes.c_uint64(4096)) == 0
    rng = random.Random(31)

    def op():
        return bytes(rng.randrange(256) for _ in range(32)) + struct.pack(
            "<I", rng.randrange(4))

    items = []
    for ln in [0, 20, 36, 37, 100, 1000, 10000]:
        spk = bytes(rng.randrange(256) for _ in range(ln))
        items.append((op(), 1000 + ln, 7, ln % 2 == 0, spk))
    # growth: 150 x 10KB = 1.5MB of arena (initial allocation is 1MB)
    for k in range(150):
        items.append((op(), k, 1, False, bytes([k & 0xFF]) * 10000))
    upsert_spk(engine, items)

    ops = [it[0] for it in items] + [op() for _ in range(5)]
    rc, found, ents, spks, used = lookup_spk(engine, ops, spk_cap=2 << 20)
    assert rc == 0
    assert found == [1] * len(items) + [0] * 5
    for i, (_, amount, daa, cb, spk) in enumerate(items):
        e = ents[64 * i:64 * (i + 1)]
        am, da, fl, _, ln = struct.unpack_from("<QQHHI", e)
        assert (am, da, ln) == (amount, daa, len(spk)), i
        assert fl & 1 == (1 if cb else 0)
        if len(spk) <= 36:
            assert not fl & ARENA_FLAG
            assert e[24:24 + len(spk)] == spk
        else:
            assert fl & ARENA_FLAG
            off, = struct.unpack_from("<I", e, 24)
            assert spks[off:off + len(spk)] == spk, i
    # overwrite one long entry with new bytes (fresh arena span)
    tgt = items[5]
    new_spk = bytes(rng.randrange(256) for _ in range(1000))
    upsert_spk(engine, [(tgt[0], 42, 42, False, new_spk)])
    rc, found, ents, spks, _ = lookup_spk(engine, [tgt[0]])
    off, = struct.unpack_from("<I", ents, 24)
    assert found == [1] and spks[off:off + 1000] == new_spk
    # undersized output buffer: -3 with the needed size reported
    rc, found, _, _, used = lookup_spk(engine, [it[0] for it in items],
                                       spk_cap=16)
    assert rc == -3 and used > 16 and all(found[:len(items)])


def _spend_tx(prev_id, prev_index, entry_spk, amount, daa, out_spk=b"",
              sig_script=b""):
    return B.tx_dict(
        1,
        [B.tx_input(prev_id, prev_index, sequence=2**64 - 1,
                    sig_script=sig_script, commit_kind=0, commit_value=20,
                    utxo=B.utxo_entry(amount, entry_spk, daa_score=daa))],
        [B.tx_output(amount, out_spk)])


def test_validate_block_with_arena_entries(engine, oracle):
    """End-to-end: txs spending long-spk UTXOs resolve from the arena through
    kv_validate_block_utxo, bit-exact vs the oracle on the inline-populated
    blob; a created long-spk output round-trips through the applied diff and
    is spent by the next block."""
    from rusty_kaspa_amd.blob import strip_utxo_entries
    lib = engine.lib
    assert lib.kv_utxo_reset(ctypes.c_void_p(engine.ctx),
                             ctypes.c_uint64(1024)) == 0
    rng = random.Random(77)
    # anyone-can-spend scripts padded with NOPs beyond the inline slot
    spk_100 = b"\x61" * 99 + b"\x51"
    spk_1k = b"\x61" * 999 + b"\x51"
    long_out = b"\x61" * 199 + b"\x51"  # created by block 1, spent by block 2

    prev1 = bytes(rng.randrange(256) for _ in range(32))
    prev2 = bytes(rng.randrange(256) for _ in range(32))
    upsert_spk(engine, [
        (prev1 + struct.pack("<I", 0), 50_000, 5, False, spk_100),
        (prev2 + struct.pack("<I", 1), 70_000, 5, False, spk_1k),
    ])

    txs1 = [
        _spend_tx(prev1, 0, spk_100, 50_000, 5, out_spk=long_out),
        _spend_tx(prev2, 1, spk_1k, 70_000, 5, out_spk=b"\x51"),
    ]
    blob1 = B.build_blob(txs1)
    # carry computed tx ids so the diff keys match
    n, = struct.unpack_from("<I", blob1, 0)
    offs = struct.unpack_from(f"<{n}I", blob1, 4)
    out = bytearray(blob1)
    ids = []
    for ti in range(n):
        idb = (ctypes.c_uint8 * 32)()
        assert oracle.ok_tx_id(bytes(blob1), len(blob1), ti, idb) == 0
        out[offs[ti] + 56:offs[ti] + 88] = bytes(idb)
        ids.append(bytes(idb))
    blob1 = bytes(out)

    # oracle on the populated blob
    oc = (ctypes.c_int32 * n)()
    of = (ctypes.c_uint64 * n)()
    omh = (ctypes.c_uint8 * 32)()
    assert oracle.ok_validate_block_parallel(blob1, len(blob1), 10**9, 10**9,
                                             SKIP_MASS, 2, oc, of, omh) == 0
    # engine from the table (entries stripped)
    stripped, _ = strip_utxo_entries(blob1)
    ec, ef, ep = engine.validate_block_utxo(stripped, n, 10**9, 10**9,
                                            SKIP_MASS, apply_diff=True)
    assert ec == list(oc) == [0, 0], (ec, list(oc))
    assert ef == list(of)
    assert engine.muhash_finalize(ep) == bytes(omh)

    # block 2 spends the long-spk output created by block 1's diff
    txs2 = [_spend_tx(ids[0], 0, long_out, 50_000, 10**9, out_spk=b"\x51")]
    blob2 = B.build_blob(txs2)
    oc2 = (ctypes.c_int32 * 1)()
    of2 = (ctypes.c_uint64 * 1)()
    omh2 = (ctypes.c_uint8 * 32)()
    assert oracle.ok_validate_block_parallel(blob2, len(blob2), 10**9 + 1,
                                             10**9 + 1, SKIP_MASS, 2, oc2, of2,
                                             omh2) == 0
    stripped2, _ = strip_utxo_entries(blob2)
    ec2, ef2, ep2 = engine.validate_block_utxo(stripped2, 1, 10**9 + 1,
                                               10**9 + 1, SKIP_MASS,
                                               apply_diff=True)
    assert ec2 == list(oc2) == [0]
    assert engine.muhash_finalize(ep2) == bytes(omh2)
    # the spent long-spk outpoint is gone
    rc, found, _, _, _ = lookup_spk(engine, [ids[0] + struct.pack("<I", 0)])
    assert rc == 0 and found == [0]


def test_utxo_4m_beyond_infinity_cache(engine):
    """Config-5 scaling rung past the 256MB Infinity Cache: a 4M-entry table
    (~470MB of slots) with random lookups — throughput recorded."""
    import numpy as np
    lib = engine.lib
    n = 4 << 20
    assert lib.kv_utxo_reset(ctypes.c_void_p(engine.ctx),
                             ctypes.c_uint64(n)) == 0
    batch = 1 << 19
    rs = np.random.RandomState(9)
    first_ops = None
    for b in range(n // batch):
        ops = rs.randint(0, 256, size=(batch, 36), dtype=np.uint8)
        ops[:, 32:] = 0  # index 0
        ents = np.zeros((batch, 64), dtype=np.uint8)
        ents[:, 0] = 1  # amount = 1
        ents[:, 20] = 34  # spk_len
        ents[:, 24:58] = ops[:, :34]  # spk bytes (deterministic)
        raw_ops = ops.tobytes()
        rc = lib.kv_utxo_upsert(ctypes.c_void_p(engine.ctx), raw_ops,
                                ents.tobytes(), ctypes.c_size_t(batch))
        assert rc == 0, lib.kv_last_error().decode()
        if b == 0:
            first_ops = raw_ops
    m = batch
    out = (ctypes.c_uint8 * (64 * m))()
    bm = (ctypes.c_uint64 * ((m + 63) // 64))()
    ms = ctypes.c_double()
    rc = lib.kv_utxo_lookup(ctypes.c_void_p(engine.ctx), first_ops,
                            ctypes.c_size_t(m), out, bm, ctypes.byref(ms))
    assert rc == 0
    missing = sum(1 for i in range(m) if not (bm[i // 64] >> (i % 64)) & 1)
    assert missing == 0, missing
    print(f"\n[utxo] {m} random lookups over a 4M-entry (~470MB) table: "
          f"{ms.value:.2f} ms = {m/ms.value*1000/1e6:.1f}M lookups/s "
          f"(beyond-L3 rung)")
    # restore a small table so later tests in the session start clean
    assert lib.kv_utxo_reset(ctypes.c_void_p(engine.ctx),
                             ctypes.c_uint64(1024)) == 0
