"""Oracle sighash vs the reference's 33 golden vectors
(consensus/core/src/hashing/sighash.rs:308-818 → tests/golden/sighash.json)."""
import copy

import pytest

from conftest import oracle_sighash
from rusty_kaspa_amd import blob as B


def base_txs(g):
    prev = bytes.fromhex(g["prev_tx_id"])
    spk1 = bytes.fromhex(g["spk1"])
    spk2 = bytes.fromhex(g["spk2"])
    entries = [B.utxo_entry(100, spk1), B.utxo_entry(200, spk2), B.utxo_entry(300, spk2)]

    def mk(version, commits):
        inputs = []
        for i in range(3):
            kind, val = commits[i]
            inputs.append(B.tx_input(prev, i, sequence=i, commit_kind=kind,
                                     commit_value=val, utxo=copy.deepcopy(entries[i])))
        outputs = [B.tx_output(300, spk2), B.tx_output(300, spk1)]
        return B.tx_dict(version, inputs, outputs, g["lock_time"])

    native = mk(0, [(0, 0)] * 3)
    native_v1 = mk(1, [(1, 11), (1, 22), (1, 33)])
    sub = mk(0, [(0, 0)] * 3)
    s = g["subnetwork_tx"]
    sub["subnetwork_id"] = bytes(s["subnetwork_id"])
    sub["gas"] = s["gas"]
    sub["payload"] = bytes(s["payload"])
    return {"native": native, "native_v1": native_v1, "subnetwork": sub}


def apply_action(g, tx, action, idx):
    m = g["mutations"]
    if action == "none":
        return
    if action == "output":
        tx["outputs"][idx]["value"] = m["output_value_new"]
    elif action == "input":
        tx["inputs"][idx]["prev_index"] = m["input_prev_index_new"]
    elif action == "compute_budget":
        tx["inputs"][idx]["commit_kind"] = 1
        tx["inputs"][idx]["commit_value"] = m["compute_budget_new"]
    elif action == "sigop_count":
        tx["inputs"][idx]["commit_kind"] = 0
        tx["inputs"][idx]["commit_value"] = m["sigop_count_new"]
    elif action == "amount":
        tx["inputs"][idx]["utxo"]["amount"] = m["amount_new"]
    elif action == "spk":
        u = tx["inputs"][idx]["utxo"]
        u["spk"] = u["spk"] + bytes(m["spk_append"])
    elif action == "sequence":
        tx["inputs"][idx]["sequence"] = m["sequence_new"]
    elif action == "payload":
        tx["payload"] = bytes(m["payload_new"])
    elif action == "gas":
        tx["gas"] = m["gas_new"]
    elif action == "subnetwork":
        tx["subnetwork_id"] = bytes(m["subnetwork_new"])
    else:
        raise ValueError(action)


def test_sighash_vectors(oracle, golden):
    g = golden("sighash.json")
    for name, txkey, hash_type, input_index, action, aidx, expected in g["tests"]:
        txs = base_txs(g)
        tx = txs[txkey]
        apply_action(g, tx, action, aidx)
        blob = B.build_blob([tx])
        got = oracle_sighash(oracle, blob, 0, input_index, hash_type)
        assert got.hex() == expected, name


def test_sighash_ecdsa_wrap(oracle, golden):
    """calc_ecdsa_signature_hash = domain-sha256(schnorr hash) (sighash.rs:282-292)"""
    import ctypes
    g = golden("sighash.json")
    txs = base_txs(g)
    blob = B.build_blob([txs["native"]])
    schnorr = oracle_sighash(oracle, blob, 0, 0, 0x01)
    ecdsa = oracle_sighash(oracle, blob, 0, 0, 0x01, ecdsa=True)
    out = (ctypes.c_uint8 * 32)()
    dom = b"TransactionSigningHashECDSA"
    oracle.ok_sha256_domain(dom, len(dom), schnorr, 32, out)
    assert bytes(out) == ecdsa
