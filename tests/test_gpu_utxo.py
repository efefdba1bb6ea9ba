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


def _op_home(op36, mask):
    """Python model of kv_utxo_kernels.hip op_hash()."""
    h = int.from_bytes(op36[:8], "little")
    idx = int.from_bytes(op36[32:36], "little")
    return (h ^ (0x9E3779B97F4A7C15 * (idx + 1) & (1 << 64) - 1)) & mask


def test_utxo_tombstone_chain_upsert(engine):
    """Regression (advisor r1): a tombstone left early in a live key's probe
    chain must not be claimed by an upsert of that key — the upsert must keep
    probing, find the existing READY slot and overwrite it, or the table holds
    two slots for one outpoint and a later remove resurrects the stale one."""
    lib = engine.lib
    assert lib.kv_utxo_reset(ctypes.c_void_p(engine.ctx), ctypes.c_uint64(16)) == 0
    mask = 63  # reset(16) allocates the 64-slot minimum

    # brute-force three distinct outpoints sharing one home slot
    rng = random.Random(99)
    home, cluster = None, []
    while len(cluster) < 3:
        op = outpoint(rng)
        h = _op_home(op, mask)
        if home is None:
            home, cluster = h, [op]
        elif h == home and op not in cluster:
            cluster.append(op)
    k1, k2, k3 = cluster

    def upsert(op, ent):
        rc = lib.kv_utxo_upsert(ctypes.c_void_p(engine.ctx), op, ent,
                                ctypes.c_size_t(1))
        assert rc == 0, lib.kv_last_error().decode()

    def remove(op):
        assert lib.kv_utxo_remove(ctypes.c_void_p(engine.ctx), op,
                                  ctypes.c_size_t(1)) == 0

    e_old = pack_entry(111, 1, False, b"\x01" * 34)
    e_new = pack_entry(222, 2, False, b"\x02" * 34)
    e_k1 = pack_entry(333, 3, False, b"\x03" * 34)

    upsert(k1, e_k1)   # lands at home
    upsert(k2, e_old)  # chains to home+1
    remove(k1)         # tombstone at home, k2's chain passes through it
    upsert(k2, e_new)  # MUST overwrite home+1, not claim the tombstone

    found, entries, _ = lookup(engine, [k2])
    assert found == [1]
    assert entries[:64] == e_new

    remove(k2)         # must delete the ONE slot holding k2
    found, _, _ = lookup(engine, [k1, k2])
    assert found == [0, 0], "stale k2 entry resurrected after remove"

    # a fresh key may still reuse the tombstones
    upsert(k3, e_old)
    found, entries, _ = lookup(engine, [k3])
    assert found == [1] and entries[:64] == e_old
