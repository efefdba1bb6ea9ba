#!/usr/bin/env python3
"""Extract golden test vectors (data, not code) from the reference tree.

Runs ONLY in the build container where /root/reference is mounted; the JSON
outputs are committed under tests/golden/ and are what travels to the GPU box.

Sources (all inside the reference's own test code):
  - crypto/hashes/src/hashers.rs:204-390        (keyed blake2b / sha256 / blake3 KATs)
  - consensus/core/src/hashing/sighash.rs:308-818 (sighash vectors, hand-coded table below)
  - crypto/muhash/src/lib.rs:19-346             (muhash KATs)
  - consensus/core/src/hashing/tx.rs:247-434    (transaction id/hash KATs, hand-coded)
  - consensus/src/processes/transaction_validator/tx_validation_in_utxo_context.rs:489-1041
                                                (real mainnet-signature accept/reject txs)
"""
import json
import os
import re
import sys

REF = "/root/reference"
OUT = os.path.dirname(os.path.abspath(__file__))


def read(p):
    with open(os.path.join(REF, p)) as f:
        return f.read()


def rust_byte_array(text):
    """Parse a rust &[...] numeric list (decimal or 0x hex) into a python list of ints."""
    return [int(x, 0) for x in re.findall(r"0x[0-9a-fA-F]+|\d+", text)]


def extract_hasher_vectors():
    src = read("crypto/hashes/src/hashers.rs")
    # The shared incremental input data (first occurrence, blake2b tests; blake3 uses same)
    inputs = [
        [],
        [1],
        [5, 199, 126, 44, 71, 32, 82, 139, 122, 217, 43, 48, 52, 112, 40, 209, 180, 83, 139, 231,
         72, 48, 136, 48, 168, 226, 133, 7, 60, 4, 160, 205],
        [42] * 64,
        [0] * 8,
    ]
    # run_test_vector(&input_data, Name::new, &[ "...", ... ]);
    cases = {}
    for m in re.finditer(r"run_test_vector\(\s*&input_data,\s*(\w+)::new,\s*&\[(.*?)\]\s*,?\s*\)", src, re.S):
        name, body = m.group(1), m.group(2)
        hashes = re.findall(r'"([0-9a-f]{64})"', body)
        assert len(hashes) == 5, (name, len(hashes))
        cases[name] = hashes
    assert "TransactionSigningHash" in cases and "TransactionSigningHashECDSA" in cases
    assert len(cases) >= 14, sorted(cases)
    return {"inputs": inputs, "expected": cases}


def extract_muhash_vectors():
    src = read("crypto/muhash/src/lib.rs")
    m = re.search(r"EMPTY_MUHASH: Hash = Hash::from_bytes\(\[(.*?)\]\)", src, re.S)
    empty = rust_byte_array(m.group(1))
    assert len(empty) == 32

    vectors = []
    for m in re.finditer(
        r"TestVector\s*\{\s*data:\s*&\[(.*?)\],\s*multiset_hash:\s*Hash::from_bytes\(\[(.*?)\]\s*\),\s*"
        r"cumulative_hash:\s*Hash::from_bytes\(\[(.*?)\]\s*\),\s*\}",
        src, re.S,
    ):
        vectors.append({
            "data": rust_byte_array(m.group(1)),
            "multiset_hash": rust_byte_array(m.group(2)),
            "cumulative_hash": rust_byte_array(m.group(3)),
        })
    assert len(vectors) == 3
    for v in vectors:
        assert len(v["multiset_hash"]) == 32 and len(v["cumulative_hash"]) == 32

    m = re.search(r'let expected = "([0-9a-f]{64})";\s*let mut acc = MuHash::new', src)
    precomputed = m.group(1)  # add(0), add(1), remove(2) over element_from_byte

    m = re.search(r"fn test_serialize\(\)\s*\{\s*let expected = \[(.*?)\];", src, re.S)
    serialize_expected = rust_byte_array(m.group(1))
    assert len(serialize_expected) == 384

    return {
        "empty_muhash": empty,
        "vectors": vectors,
        "precomputed_add0_add1_remove2": precomputed,
        "serialize_add1_add2": serialize_expected,
    }


def extract_u3072_helper_vectors():
    src = read("crypto/muhash/src/u3072.rs")
    out = {}
    # mul_wide
    out["mul_wide"] = [
        {"a": 2**64 - 1, "b": 2**64 - 1, "low": 1, "high": 18446744073709551614},
        {"a": 2**64 - 101, "b": 2**64 - 31, "low": 3131, "high": 18446744073709551484},
    ]
    # The remaining helper vectors are structural (mulnadd3/muln2/muladd3) — embedded in
    # oracle C unit tests directly; here we only keep the inverse edge case limbs.
    m = re.search(r"let orig = U3072 \{\s*limbs: \[(.*?)\],?\s*\}", src, re.S)
    out["inverse_edge_case_limbs"] = rust_byte_array(m.group(1))
    assert len(out["inverse_edge_case_limbs"]) == 48
    return out


# -- sighash vectors: the base txs + mutation table (transcribed from sighash.rs:308-818) --
SIGHASH = {
    "prev_tx_id": "880eb9819a31821d9d2399e2f35e2433b72637e393d71ecc9b8d0250f49153c3",
    "spk1": "208325613d2eeaf7176ac6c670b13c0043156c427438ed72d74b7800862ad884e8ac",
    "spk2": "20fcef4c106cf11135bbd70f02a726a92162d2fb8b22f0469126f800862ad884e8ac",
    "lock_time": 1615462089000,
    "subnetwork_tx": {
        "subnetwork_id": [1, 2, 3, 4, 5, 6, 7, 8, 9, 10, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0],
        "gas": 250,
        "payload": [10, 11, 12, 13, 14, 15, 16, 17, 18, 19, 20],
    },
    # tx templates: native v0, native v1 (compute budgets 11/22/33), subnetwork
    # inputs: (index i, sequence i, sigop_count 0); outputs: (300, spk2), (300, spk1)
    # entries: (100, spk1), (200, spk2), (300, spk2); daa 0, not coinbase
    "mutations": {
        "payload_new": [6, 6, 6, 4, 2, 0, 1, 3, 3, 7],
        "gas_new": 1234,
        "subnetwork_new": [6, 6, 6, 4, 2, 0, 1, 3, 3, 7, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0],
        "output_value_new": 100,
        "input_prev_index_new": 2,
        "sequence_new": 12345,
        "amount_new": 666,
        "spk_append": [1, 2, 3],
        "compute_budget_new": 1234,
        "sigop_count_new": 123,
    },
    "tests": [
        # [name, tx(native|native_v1|subnetwork), hash_type, input_index, action, action_index, expected]
        ["native-all-0", "native", 0x01, 0, "none", 0, "03b7ac6927b2b67100734c3cc313ff8c2e8b3ce3e746d46dd660b706a916b1f5"],
        ["native-all-0-modify-input-1", "native", 0x01, 0, "input", 1, "a9f563d86c0ef19ec2e4f483901d202e90150580b6123c3d492e26e7965f488c"],
        ["native-all-0-modify-compute-mass-1", "native", 0x01, 0, "compute_budget", 1, "03b7ac6927b2b67100734c3cc313ff8c2e8b3ce3e746d46dd660b706a916b1f5"],
        ["native-v1-all-0-modify-sigopcount-0", "native_v1", 0x01, 0, "sigop_count", 0, "5b2657524be672e019897646b56da3d192b453d78ae5e6e5c07f029a69f5f075"],
        ["native-v1-all-0-modify-sigopcount-1", "native_v1", 0x01, 0, "sigop_count", 1, "5b2657524be672e019897646b56da3d192b453d78ae5e6e5c07f029a69f5f075"],
        ["native-v1-all-0-modify-compute-budget-0", "native_v1", 0x01, 0, "compute_budget", 0, "5b2657524be672e019897646b56da3d192b453d78ae5e6e5c07f029a69f5f075"],
        ["native-v1-all-0-modify-compute-budget-1", "native_v1", 0x01, 0, "compute_budget", 1, "5b2657524be672e019897646b56da3d192b453d78ae5e6e5c07f029a69f5f075"],
        ["native-all-0-modify-output-1", "native", 0x01, 0, "output", 1, "aad2b61bd2405dfcf7294fc2be85f325694f02dda22d0af30381cb50d8295e0a"],
        ["native-all-0-modify-sequence-1", "native", 0x01, 0, "sequence", 1, "0818bd0a3703638d4f01014c92cf866a8903cab36df2fa2506dc0d06b94295e8"],
        ["native-all-anyonecanpay-0", "native", 0x81, 0, "none", 0, "24821e466e53ff8e5fa93257cb17bb06131a48be4ef282e87f59d2bdc9afebc2"],
        ["native-all-anyonecanpay-0-modify-input-0", "native", 0x81, 0, "input", 0, "d09cb639f335ee69ac71f2ad43fd9e59052d38a7d0638de4cf989346588a7c38"],
        ["native-all-anyonecanpay-0-modify-input-1", "native", 0x81, 0, "input", 1, "24821e466e53ff8e5fa93257cb17bb06131a48be4ef282e87f59d2bdc9afebc2"],
        ["native-all-anyonecanpay-0-modify-sequence", "native", 0x81, 0, "sequence", 1, "24821e466e53ff8e5fa93257cb17bb06131a48be4ef282e87f59d2bdc9afebc2"],
        ["native-none-0", "native", 0x02, 0, "none", 0, "38ce4bc93cf9116d2e377b33ff8449c665b7b5e2f2e65303c543b9afdaa4bbba"],
        ["native-none-0-modify-output-1", "native", 0x02, 0, "output", 1, "38ce4bc93cf9116d2e377b33ff8449c665b7b5e2f2e65303c543b9afdaa4bbba"],
        ["native-none-0-modify-sequence-0", "native", 0x02, 0, "sequence", 0, "d9efdd5edaa0d3fd0133ee3ab731d8c20e0a1b9f3c0581601ae2075db1109268"],
        ["native-none-0-modify-sequence-1", "native", 0x02, 0, "sequence", 1, "38ce4bc93cf9116d2e377b33ff8449c665b7b5e2f2e65303c543b9afdaa4bbba"],
        ["native-none-anyonecanpay-0", "native", 0x82, 0, "none", 0, "06aa9f4239491e07bb2b6bda6b0657b921aeae51e193d2c5bf9e81439cfeafa0"],
        ["native-none-anyonecanpay-0-modify-amount-spent", "native", 0x82, 0, "amount", 0, "f07f45f3634d3ea8c0f2cb676f56e20993edf9be07a83bf0dfdb3debcf1441bf"],
        ["native-none-anyonecanpay-0-modify-script-public-key", "native", 0x82, 0, "spk", 0, "20a525c54dc33b2a61201f05233c086dbe8e06e9515775181ed96550b4f2d714"],
        ["native-single-0", "native", 0x04, 0, "none", 0, "44a0b407ff7b239d447743dd503f7ad23db5b2ee4d25279bd3dffaf6b474e005"],
        ["native-single-0-modify-output-1", "native", 0x04, 0, "output", 1, "44a0b407ff7b239d447743dd503f7ad23db5b2ee4d25279bd3dffaf6b474e005"],
        ["native-single-0-modify-sequence-0", "native", 0x04, 0, "sequence", 0, "83796d22879718eee1165d4aace667bb6778075dab579c32c57be945f466a451"],
        ["native-single-0-modify-sequence-1", "native", 0x04, 0, "sequence", 1, "44a0b407ff7b239d447743dd503f7ad23db5b2ee4d25279bd3dffaf6b474e005"],
        ["native-single-2-no-corresponding-output", "native", 0x04, 2, "none", 0, "022ad967192f39d8d5895d243e025ec14cc7a79708c5e364894d4eff3cecb1b0"],
        ["native-single-2-no-corresponding-output-modify-output-1", "native", 0x04, 2, "output", 1, "022ad967192f39d8d5895d243e025ec14cc7a79708c5e364894d4eff3cecb1b0"],
        ["native-single-anyonecanpay-0", "native", 0x84, 0, "none", 0, "43b20aba775050cf9ba8d5e48fc7ed2dc6c071d23f30382aea58b7c59cfb8ed7"],
        ["native-single-anyonecanpay-2-no-corresponding-output", "native", 0x84, 2, "none", 0, "846689131fb08b77f83af1d3901076732ef09d3f8fdff945be89aa4300562e5f"],
        ["native-all-0-modify-payload", "native", 0x01, 0, "payload", 0, "72ea6c2871e0f44499f1c2b556f265d9424bfea67cca9cb343b4b040ead65525"],
        ["subnetwork-all-0", "subnetwork", 0x01, 0, "none", 0, "b2f421c933eb7e1a91f1d9e1efa3f120fe419326c0dbac487752189522550e0c"],
        ["subnetwork-all-modify-payload", "subnetwork", 0x01, 0, "payload", 0, "12ab63b9aea3d58db339245a9b6e9cb6075b2253615ce0fb18104d28de4435a1"],
        ["subnetwork-all-modify-gas", "subnetwork", 0x01, 0, "gas", 0, "2501edfc0068d591160c4bd98646c6e6892cdc051182a8be3ccd6d67f104fd17"],
        ["subnetwork-all-subnetwork-id", "subnetwork", 0x01, 0, "subnetwork", 0, "a5d1230ede0dfcfd522e04123a7bcd721462fed1d3a87352031a4f6e3c4389b6"],
    ],
}

# -- mainnet-signature script tests (tx_validation_in_utxo_context.rs:489-1041) --
MAINNET_TXS = [
    {
        "name": "p2pk_schnorr_accept",  # check_signature_test (:489)
        "prev_tx_id": "746915c8dfc5e1550eacbe1d87625a105750cf1a65aaddd1baa60f8bcf7e953c",
        "prev_index": 1,
        "sig_script": "4176cf2ee56b3eed1e8da083851f41cae11532fc70a63ca1ca9f17bc9a4c2fd3dcdf60df1c1a57465f0d112995a6f289511c8e0a79c806fb79165544a439d11c0201",
        "utxo_spk": "20e1d5835e09f3c3dad209debcb7b3bf3fb0e0d9642471f5db36c9ea58338b06beac",
        "utxo_amount": 20879456551,
        "utxo_daa": 32022768,
        "sigop_count": 1,
        "outputs": [
            [10360487799, "200749c89953b463d1e186a16a941f9354fa3fff313c391149e47961b95dd4df28ac"],
            [10518958752, "20e1d5835e09f3c3dad209debcb7b3bf3fb0e0d9642471f5db36c9ea58338b06beac"],
        ],
        "expect": "ok",
        "expect_dup_input": "SignatureInvalid:EvalFalse",
    },
    {
        "name": "p2pk_schnorr_wrong_spk_reject",  # check_incorrect_signature_test (:564) — entry spk is spk2, sig signs spk1 context
        "prev_tx_id": "746915c8dfc5e1550eacbe1d87625a105750cf1a65aaddd1baa60f8bcf7e953c",
        "prev_index": 1,
        "sig_script": "4176cf2ee56b3eed1e8da083851f41cae11532fc70a63ca1ca9f17bc9a4c2fd3dcdf60df1c1a57465f0d112995a6f289511c8e0a79c806fb79165544a439d11c0201",
        "utxo_spk": "200749c89953b463d1e186a16a941f9354fa3fff313c391149e47961b95dd4df28ac",
        "utxo_amount": 20879456551,
        "utxo_daa": 32022768,
        "sigop_count": 1,
        "outputs": [
            [10360487799, "200749c89953b463d1e186a16a941f9354fa3fff313c391149e47961b95dd4df28ac"],
            [10518958752, "20e1d5835e09f3c3dad209debcb7b3bf3fb0e0d9642471f5db36c9ea58338b06beac"],
        ],
        "expect": "err",
        "expect_dup_input": "err",
    },
    {
        "name": "p2sh_multisig_2of4_accept",  # check_multi_signature_test (:644)
        "prev_tx_id": "63020db736215f8b1105a9281f7bcbb6473d965ecc45bb2fb5da59bd35e6ff84",
        "prev_index": 0,
        "sig_script": "41ca6f8d104b47ca8ab133d98b3794b49f00ec5d2dce8253e78de035dfbc8f40a2fefa3086c3a181d9f1755a8f4ada4f8a4b8982b361853c8020009e1a752debce0141fdb58c2c25fcfe37d427967c34700f92e9eb1df0f2f9ff366444d92357ff35a270ee5445287031e4c0f72acda20876ccf918de1039a41e9b5f83b3737223f995014c875220ecdd9ec9f2c53ed8e5a170cc88354e133299022da55e1e8bd3c61d8b9dcbd7df2068f191b6aca3d9d8cfa2edb0c44a10fc87dc36b62e1d02228257ccdf979b1fce20b1503ef14aa6773ba3a1f012dbea2992e181766c35c5bc17465b5f57807540bf2006e161ced6b77c11b9a317080a899121a9c6df30a76490402f9a3b7e18bce97b54ae",
        "utxo_spk": "aa2071b6c2c604a8830a1484ba469e845c37bb0af32f044bc8fd0c892c8878419e8587",
        "utxo_amount": 12793000000000,
        "utxo_daa": 36151168,
        "sigop_count": 4,
        "outputs": [
            [10000000000000, "206c376f9da440494e18b283803698ed13249af93be3e99f58f42d7d82744d3d15ac"],
            [2792999990000, "aa2071b6c2c604a8830a1484ba469e845c37bb0af32f044bc8fd0c892c8878419e8587"],
        ],
        "expect": "ok",
        "expect_dup_input": "SignatureInvalid:NullFail",
    },
    {
        "name": "p2sh_multisig_last_sig_bad_nullfail",  # (:725) second sig mutated ...ff35a2 -> ff3da2
        "prev_tx_id": "63020db736215f8b1105a9281f7bcbb6473d965ecc45bb2fb5da59bd35e6ff84",
        "prev_index": 0,
        "sig_script": "41ca6f8d104b47ca8ab133d98b3794b49f00ec5d2dce8253e78de035dfbc8f40a2fefa3086c3a181d9f1755a8f4ada4f8a4b8982b361853c8020009e1a752debce0141fdb58c2c25fcfe37d427967c34700f92e9eb1df0f2f9ff366444d92357ff3da270ee5445287031e4c0f72acda20876ccf918de1039a41e9b5f83b3737223f995014c875220ecdd9ec9f2c53ed8e5a170cc88354e133299022da55e1e8bd3c61d8b9dcbd7df2068f191b6aca3d9d8cfa2edb0c44a10fc87dc36b62e1d02228257ccdf979b1fce20b1503ef14aa6773ba3a1f012dbea2992e181766c35c5bc17465b5f57807540bf2006e161ced6b77c11b9a317080a899121a9c6df30a76490402f9a3b7e18bce97b54ae",
        "utxo_spk": "aa2071b6c2c604a8830a1484ba469e845c37bb0af32f044bc8fd0c892c8878419e8587",
        "utxo_amount": 12793000000000,
        "utxo_daa": 36151168,
        "sigop_count": 4,
        "outputs": [
            [10000000000000, "206c376f9da440494e18b283803698ed13249af93be3e99f58f42d7d82744d3d15ac"],
            [2792999990000, "aa2071b6c2c604a8830a1484ba469e845c37bb0af32f044bc8fd0c892c8878419e8587"],
        ],
        "expect": "SignatureInvalid:NullFail",
        "expect_dup_input": "SignatureInvalid:NullFail",
    },
    {
        "name": "p2sh_multisig_first_sig_bad_nullfail",  # (:808) first sig mutated ...8f40a2 -> 8f41a2
        "prev_tx_id": "63020db736215f8b1105a9281f7bcbb6473d965ecc45bb2fb5da59bd35e6ff84",
        "prev_index": 0,
        "sig_script": "41ca6f8d104b47ca8ab133d98b3794b49f00ec5d2dce8253e78de035dfbc8f41a2fefa3086c3a181d9f1755a8f4ada4f8a4b8982b361853c8020009e1a752debce0141fdb58c2c25fcfe37d427967c34700f92e9eb1df0f2f9ff366444d92357ff35a270ee5445287031e4c0f72acda20876ccf918de1039a41e9b5f83b3737223f995014c875220ecdd9ec9f2c53ed8e5a170cc88354e133299022da55e1e8bd3c61d8b9dcbd7df2068f191b6aca3d9d8cfa2edb0c44a10fc87dc36b62e1d02228257ccdf979b1fce20b1503ef14aa6773ba3a1f012dbea2992e181766c35c5bc17465b5f57807540bf2006e161ced6b77c11b9a317080a899121a9c6df30a76490402f9a3b7e18bce97b54ae",
        "utxo_spk": "aa2071b6c2c604a8830a1484ba469e845c37bb0af32f044bc8fd0c892c8878419e8587",
        "utxo_amount": 12793000000000,
        "utxo_daa": 36151168,
        "sigop_count": 4,
        "outputs": [
            [10000000000000, "206c376f9da440494e18b283803698ed13249af93be3e99f58f42d7d82744d3d15ac"],
            [2792999990000, "aa2071b6c2c604a8830a1484ba469e845c37bb0af32f044bc8fd0c892c8878419e8587"],
        ],
        "expect": "SignatureInvalid:NullFail",
        "expect_dup_input": "SignatureInvalid:NullFail",
    },
    {
        "name": "p2sh_multisig_empty_sigs_evalfalse",  # (:891) two empty pushes instead of sigs
        "prev_tx_id": "63020db736215f8b1105a9281f7bcbb6473d965ecc45bb2fb5da59bd35e6ff84",
        "prev_index": 0,
        "sig_script": "00004c875220ecdd9ec9f2c53ed8e5a170cc88354e133299022da55e1e8bd3c61d8b9dcbd7df2068f191b6aca3d9d8cfa2edb0c44a10fc87dc36b62e1d02228257ccdf979b1fce20b1503ef14aa6773ba3a1f012dbea2992e181766c35c5bc17465b5f57807540bf2006e161ced6b77c11b9a317080a899121a9c6df30a76490402f9a3b7e18bce97b54ae",
        "utxo_spk": "aa2071b6c2c604a8830a1484ba469e845c37bb0af32f044bc8fd0c892c8878419e8587",
        "utxo_amount": 12793000000000,
        "utxo_daa": 36151168,
        "sigop_count": 4,
        "outputs": [
            [10000000000000, "206c376f9da440494e18b283803698ed13249af93be3e99f58f42d7d82744d3d15ac"],
            [2792999990000, "aa2071b6c2c604a8830a1484ba469e845c37bb0af32f044bc8fd0c892c8878419e8587"],
        ],
        "expect": "SignatureInvalid:EvalFalse",
        "expect_dup_input": "SignatureInvalid:EvalFalse",
    },
    {
        "name": "non_push_only_sig_script",  # (:974) sig_script = OP_TRUE OP_DROP, spk = OP_TRUE
        "prev_tx_id": "1111111111111111111111111111111111111111111111111111111111111111",
        "prev_index": 0,
        "sig_script": "5175",
        "utxo_spk": "51",
        "utxo_amount": 12793000000000,
        "utxo_daa": 36151168,
        "sigop_count": 4,
        "outputs": [
            [2792999990000, "51"],
        ],
        "expect": "SignatureInvalid:SignatureScriptNotPushOnly",
        "expect_dup_input": "SignatureInvalid:SignatureScriptNotPushOnly",
    },
]

# -- transaction id/hash vectors (hashing/tx.rs tests, transcribed) --
TXID = [
    {"name": "t1", "tx": {"version": 0, "inputs": [], "outputs": [], "lock_time": 0, "subnetwork": "00", "gas": 0, "payload": "", "mass": 0},
     "id": "2c18d5e59ca8fc4c23d9560da3bf738a8f40935c11c162017fbf2c907b7e665c",
     "hash": "c9e29784564c269ce2faaffd3487cb4684383018ace11133de082dce4bb88b0b"},
    {"name": "t2", "tx": {"version": 0, "inputs": [{"prev_id_u64": 0, "prev_index": 2, "sig_script": "0102", "sequence": 7, "sigop_count": 5}], "outputs": [], "lock_time": 0, "subnetwork": "00", "gas": 0, "payload": "", "mass": 0},
     "id": "b2d65ae36e123eb73f253176d7234a57656b84d0d60b9fc746ab0d0f085c9cc7",
     "hash": "7d9f7cfdd77f236a41895ac5cdda2fa42f7122964ba995fdfacebce54efad7e8"},
    {"name": "t3", "tx": {"version": 0, "inputs": [{"prev_id_u64": 0, "prev_index": 2, "sig_script": "0102", "sequence": 7, "sigop_count": 5}], "outputs": [{"value": 1564, "spk_version": 7, "spk": "0102030405"}], "lock_time": 0, "subnetwork": "00", "gas": 0, "payload": "", "mass": 0},
     "id": "67289b12146d1b5ef384332137399791a5cfe89506ff31688b0d95ae821d0a0c",
     "hash": "492279c0ed5018aa00b0b2d42c1c42350285f2e689236a81829edaf818e30fdb"},
    {"name": "t4", "tx": {"version": 0, "inputs": [{"prev_id_u64": 0, "prev_index": 2, "sig_script": "0102", "sequence": 7, "sigop_count": 5}], "outputs": [{"value": 1564, "spk_version": 7, "spk": "0102030405"}], "lock_time": 54, "subnetwork": "00", "gas": 3, "payload": "", "mass": 0},
     "id": "7cd34b788d7d230970d4bfd955c34c5abc49e3bcdd5adb03a77bb71d05554401",
     "hash": "de319664ee9f4197e89be0d0e08b2b6cac110efc2cf107de1fbc6bd2ce29d545"},
    {"name": "t5", "tx": {"version": 0, "inputs": [{"prev_id": "59b3d6dc6cdc660c389c3fdb5704c48c598d279cdf1bab54182db586a4c95dd5", "prev_index": 2, "sig_script": "0102", "sequence": 7, "sigop_count": 5}], "outputs": [{"value": 1564, "spk_version": 7, "spk": "0102030405"}], "lock_time": 54, "subnetwork": "00", "gas": 3, "payload": "", "mass": 0},
     "id": "c9dd78e818445f617a28348d6db752142e2fab440effa58140ad2773e638b628",
     "hash": "1be9978bcab9424f15adac6fca0a64c3f56344a7cd0ec92a225496e19a0d122c"},
    {"name": "t6_coinbase", "tx": {"version": 0, "inputs": [], "outputs": [{"value": 1564, "spk_version": 7, "spk": "0102030405"}], "lock_time": 54, "subnetwork": "01", "gas": 3, "payload": "", "mass": 0},
     "id": "2578783ec93c3a02414a228e10b1b5af298623254775f972f97df08d4ec28c8f",
     "hash": "dffa96c75ef9d17520991fc6d88813531e230488e75b65f65ce958f2d54d2451"},
    {"name": "t7_registry", "tx": {"version": 0, "inputs": [{"prev_id": "59b3d6dc6cdc660c389c3fdb5704c48c598d279cdf1bab54182db586a4c95dd5", "prev_index": 2, "sig_script": "0102", "sequence": 7, "sigop_count": 5}], "outputs": [{"value": 1564, "spk_version": 7, "spk": "0102030405"}], "lock_time": 54, "subnetwork": "02", "gas": 3, "payload": "", "mass": 0},
     "id": "3f6cea6d7ac8f6b2f86209fa748ea0ef5a1d5d380d43b79e77d52e770bb9a7b9",
     "hash": "9abf01c6c312dd984ff19c23bec85e8678e6ea34041fe3c5de52fd9344adac63"},
    {"name": "t8_payload", "tx": {"version": 0, "inputs": [{"prev_id": "59b3d6dc6cdc660c389c3fdb5704c48c598d279cdf1bab54182db586a4c95dd5", "prev_index": 2, "sig_script": "0102", "sequence": 7, "sigop_count": 5}], "outputs": [{"value": 1564, "spk_version": 7, "spk": "0102030405"}], "lock_time": 54, "subnetwork": "02", "gas": 3, "payload": "010203", "mass": 0},
     "id": "4acda997dfb31c6518224c9ac00d0777fc7cbecdab461be3c0816b1cba19a056",
     "hash": "f0bb137ed71a91445ddf9224c76f755153a296eeb4fdc29b8393ddd81bf34ce6"},
    {"name": "t9_mass", "tx": {"version": 0, "inputs": [{"prev_id": "59b3d6dc6cdc660c389c3fdb5704c48c598d279cdf1bab54182db586a4c95dd5", "prev_index": 2, "sig_script": "0102", "sequence": 7, "sigop_count": 5}], "outputs": [{"value": 1564, "spk_version": 7, "spk": "0102030405"}], "lock_time": 54, "subnetwork": "02", "gas": 3, "payload": "010203", "mass": 5},
     "id": "4acda997dfb31c6518224c9ac00d0777fc7cbecdab461be3c0816b1cba19a056",
     "hash": "ced89bbf642cda42d29d9518d16e35cbbf85d10e1ab106b7dc2e0a821308ac91"},
    {"name": "t10_v1", "tx": {"version": 1, "inputs": [{"prev_id": "59b3d6dc6cdc660c389c3fdb5704c48c598d279cdf1bab54182db586a4c95dd5", "prev_index": 2, "sig_script": "0102", "sequence": 7, "sigop_count": 5}], "outputs": [{"value": 1564, "spk_version": 7, "spk": "0102030405"}], "lock_time": 54, "subnetwork": "02", "gas": 3, "payload": "010203", "mass": 0},
     "id": "a08a500b21be3e692c080b14e399fcfa2cfa01b25c08f2f8e7414d1c116e8d18",
     "hash": "773f5582d847a1c48947eb4e6e6ac569f90f0f9d979b4c939b72ef008f025e02"},
    {"name": "t11_v1_budget111", "tx": {"version": 1, "inputs": [{"prev_id_u64": None, "prev_index": 0, "sig_script": "", "sequence": 0, "compute_budget": 111}], "outputs": [], "lock_time": 0, "subnetwork": "native", "gas": 0, "payload": "", "mass": 0},
     "id": "5978e7aa1a9ba8fdf12dae6aa39aa198a91985e91192b291e207d4d6246349e6",
     "hash": "c41c18964aab2abe309a79de3dcf0353eee216e29ab83448cbec0c4c5792056c"},
    {"name": "t12_v1_budget222", "tx": {"version": 1, "inputs": [{"prev_id_u64": None, "prev_index": 0, "sig_script": "", "sequence": 0, "compute_budget": 222}], "outputs": [], "lock_time": 0, "subnetwork": "native", "gas": 0, "payload": "", "mass": 0},
     "id": "5978e7aa1a9ba8fdf12dae6aa39aa198a91985e91192b291e207d4d6246349e6",
     "hash": "415dfbc5b38e5805e20831d43a49bc770f4f591b00964ac922d108f6a224c590"},
    {"name": "t13_v1_sigop111", "tx": {"version": 1, "inputs": [{"prev_id_u64": None, "prev_index": 0, "sig_script": "", "sequence": 0, "sigop_count": 111}], "outputs": [], "lock_time": 0, "subnetwork": "native", "gas": 0, "payload": "", "mass": 0},
     "id": "5978e7aa1a9ba8fdf12dae6aa39aa198a91985e91192b291e207d4d6246349e6",
     "hash": "55724643b090b9a1c1b9b93b03ffac9cb1bd913a1cf0605a36509322af825864"},
    {"name": "t14_v1_sigop222", "tx": {"version": 1, "inputs": [{"prev_id_u64": None, "prev_index": 0, "sig_script": "", "sequence": 0, "sigop_count": 222}], "outputs": [], "lock_time": 0, "subnetwork": "native", "gas": 0, "payload": "", "mass": 0},
     "id": "5978e7aa1a9ba8fdf12dae6aa39aa198a91985e91192b291e207d4d6246349e6",
     "hash": "55724643b090b9a1c1b9b93b03ffac9cb1bd913a1cf0605a36509322af825864"},
]

ZERO_PAYLOAD_DIGEST = [156, 12, 162, 172, 180, 94, 146, 255, 230, 206, 180, 174, 41, 24, 139, 53,
                       200, 45, 150, 118, 205, 211, 206, 6, 127, 214, 204, 195, 10, 156, 74, 56]


def main():
    out = {}
    out["hashers.json"] = extract_hasher_vectors()
    out["muhash.json"] = extract_muhash_vectors()
    out["u3072.json"] = extract_u3072_helper_vectors()
    out["sighash.json"] = SIGHASH
    out["mainnet_txs.json"] = {"cases": MAINNET_TXS}
    out["txid.json"] = {"cases": TXID, "zero_payload_digest": ZERO_PAYLOAD_DIGEST}
    for name, data in out.items():
        path = os.path.join(OUT, name)
        with open(path, "w") as f:
            json.dump(data, f, indent=1)
        print(f"wrote {path}")


if __name__ == "__main__":
    sys.exit(main())
