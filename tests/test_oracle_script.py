"""Oracle script engine + validator vs the reference's embedded mainnet-signature
transactions (tx_validation_in_utxo_context.rs:489-1041 → golden/mainnet_txs.json).

These carry REAL mainnet Schnorr/ECDSA-era signatures, so they pin the whole
sighash → secp256k1 → script pipeline end to end.
"""
import ctypes

from rusty_kaspa_amd import blob as B

# KV codes (include/kaspa_engine_abi.h)
KV_OK = 0
SIG_INVALID = 100
EVAL_FALSE = 1
NULL_FAIL = 4
NOT_PUSH_ONLY = 5

EXPECT = {
    "ok": lambda c: c == KV_OK,
    "err": lambda c: c != KV_OK,
    "SignatureInvalid:EvalFalse": lambda c: c == SIG_INVALID + EVAL_FALSE,
    "SignatureInvalid:NullFail": lambda c: c == SIG_INVALID + NULL_FAIL,
    "SignatureInvalid:SignatureScriptNotPushOnly": lambda c: c == SIG_INVALID + NOT_PUSH_ONLY,
}


def case_to_tx(case, dup_input=False):
    inp = B.tx_input(
        bytes.fromhex(case["prev_tx_id"]), case["prev_index"],
        sig_script=bytes.fromhex(case["sig_script"]),
        commit_kind=0, commit_value=case["sigop_count"],
        utxo=B.utxo_entry(case["utxo_amount"], bytes.fromhex(case["utxo_spk"]),
                          case["utxo_daa"]))
    inputs = [inp]
    if dup_input:
        import copy
        inputs.append(copy.deepcopy(inp))
    outputs = [B.tx_output(v, bytes.fromhex(spk)) for v, spk in case["outputs"]]
    return B.tx_dict(0, inputs, outputs)


def run_check_scripts(oracle, tx):
    """check_scripts equivalent: first failing input's mapped error, else 0."""
    blob = B.build_blob([tx])
    for i in range(len(tx["inputs"])):
        rc = oracle.ok_check_input_script(bytes(blob), len(blob), 0, i)
        assert rc >= 0
        if rc:
            empty = len(tx["inputs"][i]["sig_script"]) == 0
            return (200 if empty else 100) + rc
    return 0


def test_mainnet_script_cases(oracle, golden):
    g = golden("mainnet_txs.json")
    for case in g["cases"]:
        code = run_check_scripts(oracle, case_to_tx(case))
        assert EXPECT[case["expect"]](code), (case["name"], code)


def test_mainnet_script_cases_dup_input(oracle, golden):
    """Two-input variants (duplicate_input helper in the reference tests) — the
    duplicated signature signs a different sighash and must fail exactly as the
    reference asserts."""
    g = golden("mainnet_txs.json")
    for case in g["cases"]:
        code = run_check_scripts(oracle, case_to_tx(case, dup_input=True))
        assert EXPECT[case["expect_dup_input"]](code), (case["name"], code)


def test_validate_block_end_to_end(oracle, golden):
    """Full validate over a blob of the accepting mainnet txs: fees + codes."""
    g = golden("mainnet_txs.json")
    ok_cases = [c for c in g["cases"] if c["expect"] == "ok"]
    txs = [case_to_tx(c) for c in ok_cases]
    blob = B.build_blob(txs)
    n = len(txs)
    codes = (ctypes.c_int32 * n)()
    fees = (ctypes.c_uint64 * n)()
    muhash = (ctypes.c_uint8 * 32)()
    rc = oracle.ok_validate_block(bytes(blob), len(blob), 10**9, 10**9, 2,  # SkipMassCheck
                                  codes, fees, muhash)
    assert rc == 0
    for i, c in enumerate(ok_cases):
        assert codes[i] == KV_OK, c["name"]
        total_in = c["utxo_amount"]
        total_out = sum(v for v, _ in c["outputs"])
        assert fees[i] == total_in - total_out

    # parallel variant must agree bit-exactly (incl. the muhash commitment)
    codes2 = (ctypes.c_int32 * n)()
    fees2 = (ctypes.c_uint64 * n)()
    muhash2 = (ctypes.c_uint8 * 32)()
    rc = oracle.ok_validate_block_parallel(bytes(blob), len(blob), 10**9, 10**9, 2, 8,
                                           codes2, fees2, muhash2)
    assert rc == 0
    assert list(codes2) == list(codes)
    assert list(fees2) == list(fees)
    assert bytes(muhash2) == bytes(muhash)
