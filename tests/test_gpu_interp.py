"""GPU end-to-end parity for sig-bearing NON-TEMPLATE scripts: the general
interpreter's collect/replay rounds resolved by the real GPU verify kernels
inside kv_validate_block (no template fast path involved), bit-exact vs the
oracle — including OpCheckSigFromStack, verdict-dependent conditionals,
multi-round (two-site) scripts, wrong-length multisig signatures (the
error-order case) and the KIP-21 seq-commit accessor over the C-ABI."""
import ctypes
import os
import sys

import pytest

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
import rusty_kaspa_amd.blob as B  # noqa: E402

pytestmark = pytest.mark.gpu

SKIP_MASS = 2
DEFER = -63  # KV_TX_DEFER_TO_CPU
ORACLE_UNSUPPORTED = 63


@pytest.fixture(scope="module")
def engine():
    from rusty_kaspa_amd.engine import Engine
    eng = Engine()
    yield eng
    eng.close()


def push(data):
    assert len(data) <= 0x4B
    return (bytes([len(data)]) + data) if data else b"\x00"


def spend_tx(oracle, sig_script, spk, prev_seed=0):
    """create_spending_transaction shape (lib.rs:2413-2443) as a dict."""
    funding = B.tx_dict(
        1,
        [B.tx_input(bytes([prev_seed % 256]) * 32, 0xFFFFFFFF,
                    sequence=2**64 - 1, sig_script=bytes([0, 0]),
                    commit_kind=0, commit_value=20,
                    utxo=B.utxo_entry(0, b""))],
        [B.tx_output(0, spk)])
    fblob = B.build_blob([funding])
    fid = (ctypes.c_uint8 * 32)()
    assert oracle.ok_tx_id(fblob, len(fblob), 0, fid) == 0
    return B.tx_dict(
        1,
        [B.tx_input(bytes(fid), 0, sequence=2**64 - 1, sig_script=sig_script,
                    commit_kind=0, commit_value=20,
                    utxo=B.utxo_entry(0, spk, daa_score=0, is_coinbase=True))],
        [B.tx_output(0, b"")])


def validate_both(oracle, engine, txs):
    blob = B.build_blob(txs)
    n = len(txs)
    ocodes = (ctypes.c_int32 * n)()
    ofees = (ctypes.c_uint64 * n)()
    omh = (ctypes.c_uint8 * 32)()
    rc = oracle.ok_validate_block_parallel(blob, len(blob), 10**9, 10**9,
                                           SKIP_MASS, 4, ocodes, ofees, omh)
    assert rc == 0
    ecodes, efees, _ = engine.validate_block(blob, n, 10**9, 10**9, SKIP_MASS)
    return list(ocodes), ecodes


def keypair(oracle, tag):
    key = bytes([tag]) * 31 + b"\x01"
    pk = (ctypes.c_uint8 * 32)()
    assert oracle.ok_pubkey_xonly(key, pk) == 1
    return key, bytes(pk)


def sign_input(oracle, txs, tx_index, hash_type, key, ecdsa=False):
    blob = B.build_blob(txs)
    msg = (ctypes.c_uint8 * 32)()
    assert oracle.ok_sighash(blob, len(blob), tx_index, 0, hash_type,
                             1 if ecdsa else 0, msg) == 0
    sig = (ctypes.c_uint8 * 64)()
    if ecdsa:
        assert oracle.ok_ecdsa_sign(key, msg, sig) == 1
    else:
        assert oracle.ok_schnorr_sign(key, msg, None, sig) == 1
    return bytes(sig)


def test_nontemplate_checksig_shapes(oracle, engine):
    """Real signatures through non-template script shapes, valid + corrupted,
    in ONE block batch (template and interpreter paths mixed)."""
    key, pk = keypair(oracle, 7)
    ekey = bytes([9]) * 31 + b"\x02"
    epk = (ctypes.c_uint8 * 33)()
    assert oracle.ok_pubkey_compressed(ekey, epk) == 1
    epk = bytes(epk)

    shapes = [
        # (spk, hash_type, ecdsa) — every spk is NON-template
        (push(pk) + b"\xac\x61", 0x01, False),            # checksig + NOP
        (push(pk) + b"\xad\x51", 0x01, False),            # checksig-verify, true
        (push(pk) + b"\xac\x63\x51\x67\x00\x68", 0x81, False),  # in conditional
        (push(epk) + b"\xab\x61", 0x01, True),            # ecdsa + NOP
        (b"\x51" + push(pk) + b"\x51\xae\x61", 0x01, False),  # bare 1-of-1 multisig
    ]
    txs = []
    metas = []
    for i, (spk, ht, ecdsa) in enumerate(shapes):
        placeholder = push(bytes(64) + bytes([ht]))
        txs.append(spend_tx(oracle, placeholder, spk, prev_seed=i))
        metas.append((ht, ecdsa))
    # sign each (sighash independent of sig_script contents)
    for i, (ht, ecdsa) in enumerate(metas):
        k = ekey if ecdsa else key
        sig = sign_input(oracle, txs, i, ht, k, ecdsa)
        txs[i] = dict(txs[i])
        ins = list(txs[i]["inputs"])
        ins[0] = dict(ins[0], sig_script=push(sig + bytes([ht])))
        txs[i]["inputs"] = ins
    # add corrupted twins
    n_valid = len(txs)
    for i in range(n_valid):
        t = dict(txs[i])
        ins = list(t["inputs"])
        ss = bytearray(ins[0]["sig_script"])
        ss[10] ^= 1
        ins[0] = dict(ins[0], sig_script=bytes(ss))
        t["inputs"] = ins
        txs.append(t)

    ocodes, ecodes = validate_both(oracle, engine, txs)
    assert ecodes == ocodes, list(zip(ocodes, ecodes))
    assert ocodes[:n_valid] == [0] * n_valid  # sanity: all valid shapes accept
    assert all(c != 0 for c in ocodes[n_valid:])  # corrupted all reject


def test_checksig_from_stack(oracle, engine):
    """OpCheckSigFromStack over a literal digest (no sighash), valid + invalid,
    plus a wrong-size digest (InvalidState parity)."""
    key, pk = keypair(oracle, 11)
    digest = bytes(range(32))
    sig = (ctypes.c_uint8 * 64)()
    assert oracle.ok_schnorr_sign(key, digest, None, sig) == 1
    sig = bytes(sig)
    good = push(sig) + push(digest) + push(pk) + b"\xd7"
    bad_sig = push(sig[:32] + bytes(32)) + push(digest) + push(pk) + b"\xd7"
    bad_digest = push(sig) + push(digest[:31]) + push(pk) + b"\xd7"
    txs = [spend_tx(oracle, b"", spk, prev_seed=i)
           for i, spk in enumerate([good, bad_sig, bad_digest])]
    ocodes, ecodes = validate_both(oracle, engine, txs)
    assert ecodes == ocodes, list(zip(ocodes, ecodes))
    assert ocodes[0] == 0 and ocodes[1] != 0 and ocodes[2] != 0


def test_two_sites_two_rounds(oracle, engine):
    """A script with two sequential checksig sites (two GPU rounds): the
    first verdict parks on the alt stack while the second verifies."""
    key, pk = keypair(oracle, 13)
    # [s1 s2] pk CHECKSIG TOALT pk CHECKSIG FROMALT BOOLAND
    spk = (push(pk) + b"\xac\x6b" + push(pk) + b"\xac\x6c\x9a")
    placeholder = push(bytes(65)) + push(bytes(65))
    txs = [spend_tx(oracle, placeholder, spk)]
    blob = B.build_blob(txs)
    msg = (ctypes.c_uint8 * 32)()
    assert oracle.ok_sighash(blob, len(blob), 0, 0, 1, 0, msg) == 0
    sig = (ctypes.c_uint8 * 64)()
    assert oracle.ok_schnorr_sign(key, bytes(msg), None, sig) == 1
    sp = push(bytes(sig) + b"\x01")
    ins = list(txs[0]["inputs"])
    ins[0] = dict(ins[0], sig_script=sp + sp)
    txs[0] = dict(txs[0], inputs=ins)
    ocodes, ecodes = validate_both(oracle, engine, txs)
    assert ecodes == ocodes == [0]


def test_multisig_wrong_length_sig_error_order(oracle, engine):
    """A multisig sig of invalid length must surface the PUBKEY status error
    first when the key is bad (check order: cost, pubkey, signature) —
    the advisor's round-1 error-order finding, now via the interpreter."""
    key, pk = keypair(oracle, 17)
    off_curve = bytes([0xFF]) * 32  # x not on curve → InvalidPubkey
    redeem_good_key = b"\x51" + push(pk) + b"\x51\xae"
    redeem_bad_key = b"\x51" + push(off_curve) + b"\x51\xae"
    short_sig = push(bytes(29) + b"\x01")  # 30B: valid type byte, bad length
    txs = [
        spend_tx(oracle, short_sig, redeem_bad_key, prev_seed=0),
        spend_tx(oracle, short_sig, redeem_good_key, prev_seed=1),
    ]
    ocodes, ecodes = validate_both(oracle, engine, txs)
    assert ecodes == ocodes, list(zip(ocodes, ecodes))
    # bad key → InvalidPubkey (107), good key → InvalidSignature (108)
    assert ocodes[0] % 100 == 7 and ocodes[1] % 100 == 8, ocodes


def test_seq_commit_accessor_roundtrip(oracle, engine):
    """OpChainblockSeqCommit through the kv_set_seq_commit_accessor callback:
    known block → commitment pushed; unknown → InvalidSource; unset →
    InvalidOpcode. Oracle mock configured identically."""
    lib = engine.lib
    blk = b"chain_block".ljust(32, b"b")
    com = b"commitment".ljust(32, b"c")
    oracle.ok_script_set_seq_commit_mock(blk, com)

    CB = ctypes.CFUNCTYPE(ctypes.c_int, ctypes.c_void_p,
                          ctypes.POINTER(ctypes.c_uint8),
                          ctypes.POINTER(ctypes.c_uint8))

    def accessor(_user, block_p, commit_p):
        block = bytes(block_p[:32])
        if block != blk:
            return 1
        for i, b in enumerate(com):
            commit_p[i] = b
        return 0

    cb = CB(accessor)
    assert lib.kv_set_seq_commit_accessor(ctypes.c_void_p(engine.ctx), cb,
                                          None) == 0
    try:
        good = push(blk) + b"\xd4" + push(com) + b"\x87"     # commit == expected
        wrong = push(blk) + b"\xd4" + push(b"x" * 32) + b"\x87"
        unknown = push(b"u" * 32) + b"\xd4" + push(com) + b"\x87"
        txs = [spend_tx(oracle, b"", spk, prev_seed=i)
               for i, spk in enumerate([good, wrong, unknown])]
        ocodes, ecodes = validate_both(oracle, engine, txs)
        assert ecodes == ocodes, list(zip(ocodes, ecodes))
        assert ocodes[0] == 0 and ocodes[1] != 0 and ocodes[2] != 0
    finally:
        assert lib.kv_set_seq_commit_accessor(ctypes.c_void_p(engine.ctx),
                                              ctypes.cast(None, CB), None) == 0
        oracle.ok_script_set_seq_commit_mock(None, None)

    # with the accessor unset both sides treat the opcode as InvalidOpcode
    txs = [spend_tx(oracle, b"", push(blk) + b"\xd4")]
    ocodes, ecodes = validate_both(oracle, engine, txs)
    assert ecodes == ocodes and ocodes[0] % 100 == 13, (ocodes, ecodes)


def test_blake3_ops_on_gpu_path(oracle, engine):
    """OpBlake3 / OpBlake3WithKey through kv_validate_block (host interpreter
    inside the engine) — parity with the oracle."""
    data = b"kaspa-blake3-test"
    spk = push(data) + b"\xd9\x75\x51"  # blake3, drop, true
    spk_keyed = push(data) + push(b"k" * 32) + b"\xda\x75\x51"  # keyed variant
    spk_badkey = push(data) + push(b"k" * 31) + b"\xda\x75\x51"  # MalformedPush
    txs = [spend_tx(oracle, b"", s, prev_seed=i)
           for i, s in enumerate([spk, spk_keyed, spk_badkey])]
    ocodes, ecodes = validate_both(oracle, engine, txs)
    assert ecodes == ocodes, list(zip(ocodes, ecodes))
    assert ocodes[0] == 0 and ocodes[1] == 0 and ocodes[2] % 100 == 14


def test_zk_defers_with_distinct_code(oracle, engine):
    """OpZkPrecompile: the oracle rejects with UNSUPPORTED (63-class); the
    engine signals the structurally distinct KV_TX_DEFER_TO_CPU (-63)."""
    spk = b"\x51\xa6"  # push 1, zk precompile
    txs = [spend_tx(oracle, b"", spk)]
    blob = B.build_blob(txs)
    ocodes = (ctypes.c_int32 * 1)()
    ofees = (ctypes.c_uint64 * 1)()
    omh = (ctypes.c_uint8 * 32)()
    assert oracle.ok_validate_block_parallel(blob, len(blob), 10**9, 10**9,
                                             SKIP_MASS, 1, ocodes, ofees,
                                             omh) == 0
    ecodes, _, _ = engine.validate_block(blob, 1, 10**9, 10**9, SKIP_MASS)
    assert ocodes[0] % 100 == ORACLE_UNSUPPORTED
    assert ecodes[0] == DEFER


def test_forty_sites_forty_rounds(oracle, engine):
    """Depth stress: 40 sequential checksig sites (one GPU round each),
    verdicts parked on the alt stack and AND-folded — all real signatures."""
    key, pk = keypair(oracle, 29)
    n_sites = 40
    spk = (push(pk) + b"\xac\x6b") * n_sites + b"\x6c" * n_sites + \
          b"\x9a" * (n_sites - 1)
    placeholder = push(bytes(65)) * n_sites
    funding_seq = spend_tx(oracle, placeholder, spk)
    # raise the compute-budget commitment: 40 sigops > the default 20
    ins = list(funding_seq["inputs"])
    ins[0] = dict(ins[0], commit_value=50)
    funding_seq = dict(funding_seq, inputs=ins)
    txs = [funding_seq]
    blob = B.build_blob(txs)
    msg = (ctypes.c_uint8 * 32)()
    assert oracle.ok_sighash(blob, len(blob), 0, 0, 1, 0, msg) == 0
    sig = (ctypes.c_uint8 * 64)()
    assert oracle.ok_schnorr_sign(key, bytes(msg), None, sig) == 1
    sp = push(bytes(sig) + b"\x01")
    ins = list(txs[0]["inputs"])
    ins[0] = dict(ins[0], sig_script=sp * n_sites)
    txs[0] = dict(txs[0], inputs=ins)
    ocodes, ecodes = validate_both(oracle, engine, txs)
    assert ecodes == ocodes == [0], (ocodes, ecodes)
    # flip one middle signature: both sides reject identically
    bad = bytearray(bytes(sig))
    bad[7] ^= 2
    scripts = [sp] * n_sites
    scripts[20] = push(bytes(bad) + b"\x01")
    ins[0] = dict(ins[0], sig_script=b"".join(scripts))
    txs[0] = dict(txs[0], inputs=ins)
    ocodes, ecodes = validate_both(oracle, engine, txs)
    assert ecodes == ocodes and ocodes[0] != 0


def test_p2sh_noncanonical_redeem_with_sig(oracle, engine):
    """A P2SH spend whose redeem script is NOT the canonical multisig template
    (checksig + NOP) — the classify fast path rejects it and the general
    interpreter runs the full [sig, spk, redeem] sequence with the signature
    resolved on GPU."""
    key, pk = keypair(oracle, 33)
    redeem = push(pk) + b"\xac\x61"  # <pk> CHECKSIG NOP — non-template
    # spk = OpBlake2b <32B hash> OpEqual (0xaa 0x20 h 0x87)
    import hashlib
    h = hashlib.blake2b(redeem, digest_size=32).digest()
    spk = b"\xaa\x20" + h + b"\x87"
    placeholder = push(bytes(65)) + push(redeem)
    txs = [spend_tx(oracle, placeholder, spk)]
    blob = B.build_blob(txs)
    msg = (ctypes.c_uint8 * 32)()
    assert oracle.ok_sighash(blob, len(blob), 0, 0, 1, 0, msg) == 0
    sig = (ctypes.c_uint8 * 64)()
    assert oracle.ok_schnorr_sign(key, bytes(msg), None, sig) == 1
    ins = list(txs[0]["inputs"])
    ins[0] = dict(ins[0], sig_script=push(bytes(sig) + b"\x01") + push(redeem))
    txs[0] = dict(txs[0], inputs=ins)
    ocodes, ecodes = validate_both(oracle, engine, txs)
    assert ecodes == ocodes == [0], (ocodes, ecodes)
    # wrong redeem hash rejects identically (EvalFalse from the spk phase)
    bad_spk = b"\xaa\x20" + bytes(32) + b"\x87"
    txs2 = [spend_tx(oracle, ins[0]["sig_script"], bad_spk, prev_seed=1)]
    ocodes, ecodes = validate_both(oracle, engine, txs2)
    assert ecodes == ocodes and ocodes[0] != 0
