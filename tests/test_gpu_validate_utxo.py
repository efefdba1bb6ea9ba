"""GPU parity for the populate + validate + diff-apply path
(kv_validate_block_utxo): entries resolve from the GPU-resident UTXO table,
results must match the inline-populated kv_validate_block bit-for-bit, and the
table afterwards reflects the applied diff (spent gone, created present) —
⇔ utxo_validation.rs:351-390 + utxo_diff.rs:224.
"""
import ctypes
import os
import struct
import sys

import pytest

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", "oracle"))
from workload import gen_block  # noqa: E402

from rusty_kaspa_amd.blob import strip_utxo_entries  # noqa: E402

pytestmark = pytest.mark.gpu

SKIP_MASS = 2
MISSING_OUTPOINT = 9


@pytest.fixture(scope="module")
def engine():
    from rusty_kaspa_amd.engine import Engine
    eng = Engine()
    yield eng
    eng.close()


def seed_table(engine, seeds, capacity=None):
    lib = engine.lib
    ctx = ctypes.c_void_p(engine.ctx)
    assert lib.kv_utxo_reset(ctx, ctypes.c_uint64(capacity or
                                                  max(64, 2 * len(seeds)))) == 0
    if seeds:
        ops = b"".join(op for op, _ in seeds)
        ents = b"".join(e for _, e in seeds)
        rc = lib.kv_utxo_upsert(ctx, ops, ents, ctypes.c_size_t(len(seeds)))
        assert rc == 0, lib.kv_last_error().decode()


def lookup(engine, ops):
    lib = engine.lib
    n = len(ops)
    out = (ctypes.c_uint8 * (64 * n))()
    bm = (ctypes.c_uint64 * ((n + 63) // 64))()
    rc = lib.kv_utxo_lookup(ctypes.c_void_p(engine.ctx), b"".join(ops),
                            ctypes.c_size_t(n), out, bm, None)
    assert rc == 0, lib.kv_last_error().decode()
    return [(bm[i // 64] >> (i % 64)) & 1 for i in range(n)], bytes(out)


def parse_created(blob):
    """(outpoint36, value, spk) of every output, per tx index."""
    n_txs, = struct.unpack_from("<I", blob, 0)
    offs = list(struct.unpack_from(f"<{n_txs}I", blob, 4))
    created = []
    for t in range(n_txs):
        p = offs[t]
        n_in, n_out = struct.unpack_from("<HH", blob, p + 2)
        payload_len, = struct.unpack_from("<I", blob, p + 36)
        tx_id = blob[p + 56:p + 88]
        p += 88 + payload_len
        for _ in range(n_in):
            sig_len, = struct.unpack_from("<I", blob, p + 48)
            p += 52 + sig_len
            spk_len, = struct.unpack_from("<I", blob, p + 20)
            p += 24 + spk_len + (32 if blob[p + 17] else 0)
        outs = []
        for i in range(n_out):
            value, spkv = struct.unpack_from("<QH", blob, p)
            spk_len, = struct.unpack_from("<I", blob, p + 12)
            spk = blob[p + 16:p + 16 + spk_len]
            p += 16 + spk_len + 1
            if blob[p - 1]:
                p += 34
            outs.append((tx_id + struct.pack("<I", i), value, spk))
        created.append(outs)
    return created


@pytest.mark.parametrize("kwargs,flags", [
    (dict(seed=21, n_txs=96, pct_multi_input=20, pct_ecdsa=10), SKIP_MASS),
    (dict(seed=22, n_txs=60, pct_multi_input=25, pct_ecdsa=10,
          pct_multisig=10, pct_invalid=15), SKIP_MASS),
    (dict(seed=23, n_txs=50, pct_multi_input=30, pct_ecdsa=10), 0),  # FULL
])
def test_populate_validate_matches_inline(oracle, engine, kwargs, flags):
    n = kwargs["n_txs"]
    blob, _ = gen_block(oracle, **kwargs)
    # reference result: the inline-populated path (itself oracle-verified)
    ic, if_, ip = engine.validate_block(blob, n, 10**9, 10**9, flags)
    stripped, seeds = strip_utxo_entries(blob)
    seed_table(engine, seeds)
    uc, uf, up = engine.validate_block_utxo(stripped, n, 10**9, 10**9, flags,
                                            apply_diff=True)
    assert uc == ic, [(i, a, b) for i, (a, b) in enumerate(zip(uc, ic)) if a != b][:5]
    assert uf == if_
    assert up == ip

    # diff applied: accepted txs' inputs gone, outputs present with the right entry
    created = parse_created(blob)
    spent_by_tx = []
    # seeds are in tx-then-input order; regroup
    counts = []
    st = struct.unpack_from(f"<{n}I", blob, 4)
    for t in range(n):
        n_in, = struct.unpack_from("<H", blob, st[t] + 2)
        counts.append(n_in)
    idx = 0
    for t in range(n):
        spent_by_tx.append([seeds[idx + i][0] for i in range(counts[t])])
        idx += counts[t]

    probe, expect_found, expect_entry = [], [], []
    for t in range(n):
        if uc[t] == 0:
            for op in spent_by_tx[t]:
                probe.append(op)
                expect_found.append(0)
                expect_entry.append(None)
            for op, value, spk in created[t]:
                probe.append(op)
                expect_found.append(1)
                expect_entry.append((value, spk))
        else:
            for op in spent_by_tx[t]:
                probe.append(op)
                expect_found.append(1)  # rejected tx spends nothing
                expect_entry.append(None)
    found, entries = lookup(engine, probe)
    for i in range(len(probe)):
        assert found[i] == expect_found[i], i
        if expect_entry[i] and found[i]:
            e = entries[64 * i:64 * (i + 1)]
            value, daa = struct.unpack_from("<QQ", e)
            spk_len, = struct.unpack_from("<I", e, 20)
            assert value == expect_entry[i][0]
            assert daa == 10**9  # block_daa_score
            assert e[24:24 + spk_len] == expect_entry[i][1]


def test_missing_outpoint(oracle, engine):
    n = 40
    blob, _ = gen_block(oracle, seed=24, n_txs=n, pct_multi_input=20)
    ic, if_, _ = engine.validate_block(blob, n, 10**9, 10**9, SKIP_MASS)
    stripped, seeds = strip_utxo_entries(blob)
    # withhold the first input of tx 0 and every input of tx 3
    counts = []
    st = struct.unpack_from(f"<{n}I", blob, 4)
    for t in range(n):
        n_in, = struct.unpack_from("<H", blob, st[t] + 2)
        counts.append(n_in)
    withheld = {0}
    base3 = sum(counts[:3])
    withheld |= set(range(base3, base3 + counts[3]))
    seed_table(engine, [s for i, s in enumerate(seeds) if i not in withheld],
               capacity=2 * len(seeds))
    uc, uf, _ = engine.validate_block_utxo(stripped, n, 10**9, 10**9, SKIP_MASS,
                                           apply_diff=True, want_muhash=False)
    for t in range(n):
        if t in (0, 3):
            assert uc[t] == MISSING_OUTPOINT
            assert uf[t] == 0
        else:
            assert uc[t] == ic[t]
            assert uf[t] == if_[t]
    # tx0's other inputs (still in the table) were NOT spent — tx0 was rejected
    if counts[0] > 1:
        found, _ = lookup(engine, [seeds[1][0]])
        assert found[0] == 1


def test_chained_blocks_through_table(oracle, engine):
    """Block B spends block A's created outputs straight from the table
    (the cross-block UTXO-diff lifecycle, utxo_diff.rs:224), then a replay of
    B double-spends and every tx fails with MISSING_OUTPOINT."""
    from workload import gen_spend_block
    D1, D2 = 10**9, 10**9 + 10
    nA = 40
    blobA, metaA = gen_block(oracle, seed=41, n_txs=nA, pct_multi_input=15,
                             pct_invalid=10)
    strippedA, seedsA = strip_utxo_entries(blobA)
    seed_table(engine, seedsA, capacity=4 * len(seedsA))
    cA, _, _ = engine.validate_block_utxo(strippedA, nA, D1, D1,
                                          SKIP_MASS, apply_diff=True)
    icA, _, _ = engine.validate_block(blobA, nA, D1, D1, SKIP_MASS)
    assert cA == icA

    accepted = [c == 0 for c in cA]
    blobB, metaB = gen_spend_block(oracle, seed=42, prev_meta=metaA,
                                   prev_block_daa=D1, accepted=accepted)
    nB = metaB["n_txs"]
    assert nB > 10
    # ground truth: the inline-populated form of B (entries as the diff wrote
    # them: value/spk from A's outputs, daa = D1, non-coinbase)
    iB, fB, pB = engine.validate_block(blobB, nB, D2, D2, SKIP_MASS)
    assert all(c == 0 for c in iB), iB[:5]
    strippedB, _ = strip_utxo_entries(blobB)
    uB, ufB, upB = engine.validate_block_utxo(strippedB, nB, D2, D2,
                                              SKIP_MASS, apply_diff=True)
    assert uB == iB and ufB == fB and upB == pB

    # replay: every input was just spent
    rB, rfB, _ = engine.validate_block_utxo(strippedB, nB, D2, D2, SKIP_MASS,
                                            apply_diff=True, want_muhash=False)
    assert all(c == MISSING_OUTPOINT for c in rB), rB[:5]
    assert all(f == 0 for f in rfB)
