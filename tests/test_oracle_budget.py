"""Compute-budget meter boundaries ⇔ check_scripts_parallel_budget_behavior
(tx_validation_in_utxo_context.rs:263-345): a P2SH input whose redeem
OpDup-licates a 10,000-byte (= SCRIPT_UNITS_PER_COMPUTE_BUDGET_UNIT) push.
With ComputeBudget(0) the single metered duplicate exceeds the 9,999 free
units; ComputeBudget(1) and (10) pass."""
import ctypes
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", "oracle"))

import rusty_kaspa_amd.blob as B  # noqa: E402

EXCEEDED = 28  # KV_SCRIPT_EXCEEDED_SCRIPT_UNITS
OP_DUP, OP_DROP = 0x76, 0x75


def p2sh_spk(oracle, redeem):
    h = (ctypes.c_uint8 * 32)()
    oracle.ok_blake2b_keyed(None, 0, redeem, ctypes.c_size_t(len(redeem)), h)
    return bytes([0xAA, 0x20]) + bytes(h) + bytes([0x87])


def build_blob(oracle, n_inputs, budget):
    redeem = bytes([OP_DUP, OP_DROP])
    prefix = b"\x01" * 10_000
    sig_script = (bytes([0x4D]) + len(prefix).to_bytes(2, "little") + prefix +
                  bytes([len(redeem)]) + redeem)
    spk = p2sh_spk(oracle, redeem)
    ins = [B.tx_input(bytes([i + 1]) * 32, 0, sequence=0, sig_script=sig_script,
                      commit_kind=1, commit_value=budget,
                      utxo=B.utxo_entry(1, spk))
           for i in range(n_inputs)]
    outs = [B.tx_output(1, bytes([0x51]))]
    return B.build_blob([B.tx_dict(1, ins, outs)])


def test_budget_boundaries(oracle):
    # (a) budget 0: the OpDup'd 10,000B element exceeds the 9,999 free units
    blob = build_blob(oracle, 2, 0)
    code = oracle.ok_check_input_script(blob, ctypes.c_size_t(len(blob)), 0, 0)
    assert code == EXCEEDED, code
    # (b) budget 1 (= +10,000 units): passes
    blob = build_blob(oracle, 3, 1)
    for i in range(3):
        assert oracle.ok_check_input_script(blob, ctypes.c_size_t(len(blob)),
                                            0, i) == 0
    # (c) larger budget also passes
    blob = build_blob(oracle, 3, 10)
    assert oracle.ok_check_input_script(blob, ctypes.c_size_t(len(blob)), 0, 0) == 0
