"""GPU UTXO table parity vs a python dict model: upsert/overwrite/remove/lookup
sequences with forced collisions, at config-5 scale (1M entries)."""
import ctypes
import random
import struct

import pytest

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def engine():
    from rusty_kaspa_amd.engine import Engine
    eng = Engine()
    yield eng
    eng.close()


def pack_entry(amount, daa, coinbase, spk):
    assert len(spk) <= 36
    return struct.pack("<QQHHI", amount, daa, 1 if coinbase else 0, 0,
                       len(spk)) + spk.ljust(36, b"\0") + bytes(4)


def outpoint(rng):
    return bytes(rng.randrange(256) for _ in range(32)) + struct.pack(
        "<I", rng.randrange(4))


def lookup(engine, ops):
    lib = engine.lib
    n = len(ops)
    flat = b"".join(ops)
    out = (ctypes.c_uint8 * (64 * n))()
    words = (n + 63) // 64
    bm = (ctypes.c_uint64 * words)()
    ms = ctypes.c_double()
    rc = lib.kv_utxo_lookup(ctypes.c_void_p(engine.ctx), flat, ctypes.c_size_t(n),
                            out, bm, ctypes.byref(ms))
    assert rc == 0, lib.kv_last_error().decode()
    found = [(bm[i // 64] >> (i % 64)) & 1 for i in range(n)]
    return found, bytes(out), ms.value


def test_utxo_model_parity(engine):
    lib = engine.lib
    rng = random.Random(77)
    assert lib.kv_utxo_reset(ctypes.c_void_p(engine.ctx), ctypes.c_uint64(4000)) == 0

    model = {}
    ops, vals = [], []
    for i in range(3000):
        op = outpoint(rng)
        e = pack_entry(rng.randrange(1, 10**12), rng.randrange(10**7),
                       rng.randrange(2) == 0, bytes(rng.randrange(256)
                                                    for _ in range(34)))
        model[op] = e
        ops.append(op)
        vals.append(e)
    rc = lib.kv_utxo_upsert(ctypes.c_void_p(engine.ctx), b"".join(ops),
                            b"".join(vals), ctypes.c_size_t(len(ops)))
    assert rc == 0, lib.kv_last_error().decode()

    # overwrite a subset (upsert semantics) and remove another subset
    over = rng.sample(ops, 300)
    new_vals = []
    for op in over:
        e = pack_entry(1234, 5678, False, b"\x20" + bytes(33))
        model[op] = e
        new_vals.append(e)
    assert lib.kv_utxo_upsert(ctypes.c_void_p(engine.ctx), b"".join(over),
                              b"".join(new_vals), ctypes.c_size_t(len(over))) == 0
    gone = rng.sample([o for o in ops if o not in over], 400)
    for op in gone:
        del model[op]
    assert lib.kv_utxo_remove(ctypes.c_void_p(engine.ctx), b"".join(gone),
                              ctypes.c_size_t(len(gone))) == 0

    # lookups: all inserted + some misses; order shuffled
    probe = ops + [outpoint(rng) for _ in range(500)]
    rng.shuffle(probe)
    found, entries, _ = lookup(engine, probe)
    for i, op in enumerate(probe):
        if op in model:
            assert found[i] == 1, i
            assert entries[64 * i:64 * i + 64] == model[op], i
        else:
            assert found[i] == 0, i


def test_utxo_1m_scale_and_throughput(engine):
    """Config-5 scale: 1M entries, measure random-lookup throughput."""
    lib = engine.lib
    n = 1 << 20
    assert lib.kv_utxo_reset(ctypes.c_void_p(engine.ctx), ctypes.c_uint64(n)) == 0
    # deterministic outpoints without python-loop overhead: counter-based
    import hashlib
    rng = random.Random(5)
    batch = 1 << 18
    all_ops = []
    for b in range(n // batch):
        seed = struct.pack("<QQ", 12345, b)
        raw = bytearray()
        ent = bytearray()
        for i in range(batch):
            h = hashlib.blake2b(seed + struct.pack("<I", i), digest_size=32).digest()
            raw += h + struct.pack("<I", i & 3)
            ent += pack_entry(1000 + i, 42, False, b"\x20" + h[:33])
        all_ops.append(bytes(raw))
        rc = lib.kv_utxo_upsert(ctypes.c_void_p(engine.ctx), bytes(raw), bytes(ent),
                                ctypes.c_size_t(batch))
        assert rc == 0, lib.kv_last_error().decode()
    # random lookups over the whole set
    probe_ops = all_ops[0]
    found, entries, ms = lookup(engine, [probe_ops[i * 36:(i + 1) * 36]
                                         for i in range(batch)])
    assert all(found), found.count(0)
    print(f"\n[utxo] {batch} random lookups over 1M-entry table: {ms:.2f} ms "
          f"= {batch / ms * 1000 / 1e6:.1f}M lookups/s")
