"""Full goref DAG replay (goref-1060-tx-265-blocks, the reference's own
integration fixture produced by the independent Go implementation): blocks
applied in consensus file order, each block's UTXO diff applied to the
GPU-resident table, and the muhash commitment rolled block by block — the
engine chain bit-exact against the oracle chain, mirroring the reference's
json_test end-to-end replay (consensus_integration_tests.rs:712-830).

Fixture: tests/golden/goref_replay.json.gz (extract_goref_replay.py)."""
import ctypes
import gzip
import json
import os
import struct
import sys

import pytest

HERE = os.path.dirname(os.path.abspath(__file__))
REPO = os.path.dirname(HERE)
sys.path.insert(0, REPO)

SKIP_MASS = 2
POV = 10**9
LIMBS = 48


def load_fixture():
    with gzip.open(os.path.join(HERE, "golden", "goref_replay.json.gz"),
                   "rt") as f:
        return json.load(f)


def parse_blob(blob):
    """Light parser: per tx -> (spent outpoints, [(index, value, spk_ver, spk)])."""
    n, = struct.unpack_from("<I", blob, 0)
    offs = struct.unpack_from(f"<{n}I", blob, 4)
    txs = []
    for t in range(n):
        p = offs[t]
        n_in, n_out = struct.unpack_from("<HH", blob, p + 2)
        payload_len, = struct.unpack_from("<I", blob, p + 36)
        tx_id = bytes(blob[p + 56:p + 88])
        p += 88 + payload_len
        spends = []
        for _ in range(n_in):
            spends.append(bytes(blob[p:p + 32]) +
                          blob[p + 32:p + 36])
            sslen, = struct.unpack_from("<I", blob, p + 48)
            p += 52 + sslen
            has_cov = blob[p + 17]
            spk_len, = struct.unpack_from("<I", blob, p + 20)
            p += 24 + spk_len + (32 if has_cov else 0)
        outs = []
        for oi in range(n_out):
            value, spk_ver = struct.unpack_from("<QH", blob, p)
            spk_len, = struct.unpack_from("<I", blob, p + 12)
            spk = bytes(blob[p + 16:p + 16 + spk_len])
            p += 16 + spk_len
            has_cov = blob[p]
            p += 1 + (34 if has_cov else 0)
            outs.append((oi, value, spk_ver, spk))
        txs.append((tx_id, spends, outs))
    return txs


def pack_entry(amount, daa, coinbase, spk_ver, spk):
    assert len(spk) <= 36
    return struct.pack("<QQHHI", amount, daa, 1 if coinbase else 0, spk_ver,
                       len(spk)) + spk.ljust(36, b"\0") + bytes(4)


def u3072_one():
    a = (ctypes.c_uint64 * LIMBS)()
    a[0] = 1
    return a


def test_fixture_consistency_oracle_chain(oracle):
    """CPU: the oracle replay over the fixture — every block validates clean
    and the cumulative muhash chain is reproducible (the expected values the
    GPU test compares against)."""
    fx = load_fixture()
    assert fx["applied"] == 223 and fx["skipped_unresolvable"] == 0
    num, den = u3072_one(), u3072_one()
    n_blocks = n_txs_total = 0
    for ob in fx["blocks"]:
        if not ob["n_txs"]:
            continue
        blob = bytes.fromhex(ob["blob"])
        n = ob["n_txs"]
        codes = (ctypes.c_int32 * n)()
        fees = (ctypes.c_uint64 * n)()
        mh = (ctypes.c_uint8 * 32)()
        rc = oracle.ok_validate_block_parallel(
            blob, ctypes.c_size_t(len(blob)), ctypes.c_uint64(POV),
            ctypes.c_uint64(ob["daa"]), SKIP_MASS, 4, codes, fees, mh)
        assert rc == 0 and all(c == 0 for c in codes)
        for t in range(n):
            assert oracle.ok_muhash_add_tx(blob, ctypes.c_size_t(len(blob)),
                                           ctypes.c_uint32(t),
                                           ctypes.c_uint64(ob["daa"]), num,
                                           den) == 0
        n_blocks += 1
        n_txs_total += n
    out = (ctypes.c_uint8 * 32)()
    oracle.ok_muhash_finalize(num, den, out)
    assert n_txs_total == 223 and n_blocks > 50
    assert bytes(out) != bytes(32)


@pytest.mark.gpu
def test_engine_replays_whole_dag(oracle):
    """GPU: walk the DAG block by block — coinbase outputs upserted, each
    block validated from the GPU-resident table with its diff applied, the
    muhash partial folded into the running commitment. Per-block AND
    cumulative commitments bit-exact vs the oracle chain; final UTXO set
    content verified against a host replay."""
    from rusty_kaspa_amd.engine import Engine
    from rusty_kaspa_amd.blob import strip_utxo_entries
    fx = load_fixture()
    eng = Engine()
    lib = eng.lib
    ctx = ctypes.c_void_p(eng.ctx)
    try:
        assert lib.kv_utxo_reset(ctx, ctypes.c_uint64(4096)) == 0
        acc = bytearray(b"\x01" + bytes(383) + b"\x01" + bytes(383))
        num, den = u3072_one(), u3072_one()
        model = {}  # outpoint -> entry64 (host replay of the final set)
        spent = []
        for ob in fx["blocks"]:
            daa = ob["daa"]
            if ob["coinbase"]:
                ops = b"".join(bytes.fromhex(t) + struct.pack("<I", i)
                               for t, i, _, _, _ in ob["coinbase"])
                ents = b"".join(
                    pack_entry(v, daa, True, ver, bytes.fromhex(spk))
                    for _, _, v, spk, ver in ob["coinbase"])
                n_cb = len(ob["coinbase"])
                assert lib.kv_utxo_upsert(ctx, ops, ents,
                                          ctypes.c_size_t(n_cb)) == 0
                for k in range(n_cb):
                    model[ops[k * 36:(k + 1) * 36]] = ents[k * 64:(k + 1) * 64]
            if not ob["n_txs"]:
                continue
            blob = bytes.fromhex(ob["blob"])
            n = ob["n_txs"]
            stripped, _ = strip_utxo_entries(blob)
            ec, ef, ep = eng.validate_block_utxo(stripped, n, POV, daa,
                                                 SKIP_MASS, apply_diff=True)
            assert all(c == 0 for c in ec), (daa, ec)
            # oracle expected: per-block commitment + cumulative chain
            oc = (ctypes.c_int32 * n)()
            of = (ctypes.c_uint64 * n)()
            omh = (ctypes.c_uint8 * 32)()
            rc = oracle.ok_validate_block_parallel(
                blob, ctypes.c_size_t(len(blob)), ctypes.c_uint64(POV),
                ctypes.c_uint64(daa), SKIP_MASS, 4, oc, of, omh)
            assert rc == 0 and list(oc) == ec and list(of) == ef
            assert eng.muhash_finalize(ep) == bytes(omh), daa
            eng.muhash_combine(acc, bytes(ep))
            for t in range(n):
                assert oracle.ok_muhash_add_tx(blob, ctypes.c_size_t(len(blob)),
                                               ctypes.c_uint32(t),
                                               ctypes.c_uint64(daa), num,
                                               den) == 0
            # host replay of the diff
            for tx_id, spends, outs in parse_blob(blob):
                for op in spends:
                    del model[op]
                    spent.append(op)
                for oi, value, ver, spk in outs:
                    model[tx_id + struct.pack("<I", oi)] = pack_entry(
                        value, daa, False, ver, spk)
        # cumulative commitment: engine fold == oracle chain
        exp = (ctypes.c_uint8 * 32)()
        oracle.ok_muhash_finalize(num, den, exp)
        assert eng.muhash_finalize(bytes(acc)) == bytes(exp)
        # final UTXO set: every live outpoint present with the exact entry,
        # a sample of spent outpoints absent
        assert len(model) == fx["final_utxos"]
        ops = list(model.keys())
        flat = b"".join(ops)
        out = (ctypes.c_uint8 * (64 * len(ops)))()
        words = (len(ops) + 63) // 64
        bm = (ctypes.c_uint64 * words)()
        ms = ctypes.c_double()
        assert lib.kv_utxo_lookup(ctx, flat, ctypes.c_size_t(len(ops)), out,
                                  bm, ctypes.byref(ms)) == 0
        for i, op in enumerate(ops):
            assert (bm[i // 64] >> (i % 64)) & 1, i
            assert bytes(out[64 * i:64 * i + 64]) == model[op], i
        sample = spent[:: max(1, len(spent) // 64)]
        flat = b"".join(sample)
        bm2 = (ctypes.c_uint64 * ((len(sample) + 63) // 64))()
        assert lib.kv_utxo_lookup(ctx, flat, ctypes.c_size_t(len(sample)),
                                  None, bm2, None) == 0
        assert all((bm2[i // 64] >> (i % 64)) & 1 == 0
                   for i in range(len(sample))), "spent outpoint still live"
        print(f"\n[goref replay] {fx['applied']} txs over "
              f"{len(fx['blocks'])} blocks: per-block + cumulative muhash "
              f"chain and final {len(model)}-entry UTXO set bit-exact")
    finally:
        eng.close()
