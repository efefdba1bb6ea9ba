"""Oracle transaction id/hash vs reference KATs
(consensus/core/src/hashing/tx.rs:247-434 → tests/golden/txid.json)."""
import ctypes

from conftest import oracle_tx_id
from rusty_kaspa_amd import blob as B

SUBNETS = {"00": bytes(20), "01": bytes([1] + [0] * 19), "02": bytes([2] + [0] * 19),
           "native": bytes(20)}


def tx_from_case(case):
    t = case["tx"]
    inputs = []
    for i in t["inputs"]:
        if i.get("prev_id") is not None:
            pid = bytes.fromhex(i["prev_id"])
        elif i.get("prev_id_u64") is not None:
            pid = i["prev_id_u64"].to_bytes(8, "little") + bytes(24)
        else:
            pid = bytes(32)
        commit_kind, commit_value = 0, i.get("sigop_count", 0)
        if "compute_budget" in i:
            commit_kind, commit_value = 1, i["compute_budget"]
        inputs.append(B.tx_input(pid, i["prev_index"], i["sequence"],
                                 bytes.fromhex(i["sig_script"]),
                                 commit_kind, commit_value,
                                 B.utxo_entry(0, b"")))
    outputs = [B.tx_output(o["value"], bytes.fromhex(o["spk"]), o["spk_version"])
               for o in t["outputs"]]
    return B.tx_dict(t["version"], inputs, outputs, t["lock_time"],
                     SUBNETS[t["subnetwork"]], t["gas"],
                     bytes.fromhex(t["payload"]), t["mass"])


def test_tx_ids(oracle, golden):
    g = golden("txid.json")
    for case in g["cases"]:
        tx = tx_from_case(case)
        blob = B.build_blob([tx])
        got = oracle_tx_id(oracle, blob, 0)
        assert got.hex() == case["id"], case["name"]


def test_tx_hash(oracle, golden):
    g = golden("txid.json")
    out = (ctypes.c_uint8 * 32)()
    for case in g["cases"]:
        tx = tx_from_case(case)
        blob = B.build_blob([tx])
        # ok_tx_compute_hash is not exported at blob level; reuse via ok_tx_id path:
        # hash vectors are covered through a dedicated export
        rc = oracle.ok_tx_hash_blob(bytes(blob), len(blob), 0, out)
        assert rc == 0
        assert bytes(out).hex() == case["hash"], case["name"]


def test_zero_payload_digest(oracle, golden):
    g = golden("txid.json")
    out = (ctypes.c_uint8 * 32)()
    key = b"PayloadDigest".ljust(32, b"\0")
    oracle.ok_blake3_keyed(key, b"", 0, out)
    assert list(out) == g["zero_payload_digest"]
