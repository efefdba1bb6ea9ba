"""Synthetic block workload generator (harness/bench input only — uses the
oracle's signer; mirrors simpa's Miner::build_txs workload shape,
/root/reference/simpa/src/simulator/miner.rs:165-251 and sign.rs:83-103).

Deterministic by seed. Produces tx-batch blobs (include/kaspa_engine_abi.h)
of the BASELINE config shapes:
  - config 1: 1-input P2PK Schnorr txs (spk = 0x20‖pk‖0xac)
  - config 3: 10-BPS mix — 70% 1-in P2PK schnorr, 20% multi-input (2-8),
    10% ECDSA P2PK
  - config 5 additions: invalid sigs, P2SH multisig (m-of-n), large payloads
"""
from __future__ import annotations

import ctypes
import os
import random
import sys

_REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, _REPO)

from rusty_kaspa_amd import blob as B  # noqa: E402

SIGHASH_ALL = 0x01


class Signer:
    def __init__(self, oracle):
        self.o = oracle
        self._xonly = {}
        self._pk33 = {}

    def seckey(self, i: int) -> bytes:
        out = (ctypes.c_uint8 * 32)()
        data = b"wl-key" + i.to_bytes(8, "little")
        self.o.ok_blake2b_keyed(b"workload", 8, data, len(data), out)
        return bytes(out)

    def xonly(self, i: int) -> bytes:
        if i not in self._xonly:
            pk = (ctypes.c_uint8 * 32)()
            assert self.o.ok_pubkey_xonly(self.seckey(i), pk) == 1
            self._xonly[i] = bytes(pk)
        return self._xonly[i]

    def pk33(self, i: int) -> bytes:
        if i not in self._pk33:
            pk = (ctypes.c_uint8 * 33)()
            assert self.o.ok_pubkey_compressed(self.seckey(i), pk) == 1
            self._pk33[i] = bytes(pk)
        return self._pk33[i]

    def p2pk_spk(self, i: int) -> bytes:
        return bytes([0x20]) + self.xonly(i) + bytes([0xAC])

    def p2pk_ecdsa_spk(self, i: int) -> bytes:
        return bytes([0x21]) + self.pk33(i) + bytes([0xAB])

    def p2sh_spk(self, redeem: bytes) -> bytes:
        h = (ctypes.c_uint8 * 32)()
        self.o.ok_blake2b_keyed(None, 0, redeem, len(redeem), h)
        return bytes([0xAA, 0x20]) + bytes(h) + bytes([0x87])

    def multisig_redeem(self, m: int, key_ids) -> bytes:
        r = bytes([0x50 + m])
        for k in key_ids:
            r += bytes([0x20]) + self.xonly(k)
        r += bytes([0x50 + len(key_ids), 0xAE])
        return r

    def schnorr_sign(self, key_id: int, msg: bytes) -> bytes:
        sig = (ctypes.c_uint8 * 64)()
        assert self.o.ok_schnorr_sign(self.seckey(key_id), msg, None, sig) == 1
        return bytes(sig)

    def ecdsa_sign(self, key_id: int, msg: bytes) -> bytes:
        sig = (ctypes.c_uint8 * 64)()
        assert self.o.ok_ecdsa_sign(self.seckey(key_id), msg, sig) == 1
        return bytes(sig)

    def sighash(self, blob: bytes, tx_i: int, in_i: int, hash_type=SIGHASH_ALL,
                ecdsa=False) -> bytes:
        out = (ctypes.c_uint8 * 32)()
        rc = self.o.ok_sighash(blob, len(blob), tx_i, in_i, hash_type,
                               1 if ecdsa else 0, out)
        assert rc == 0
        return bytes(out)

    def tx_id(self, blob: bytes, tx_i: int) -> bytes:
        out = (ctypes.c_uint8 * 32)()
        assert self.o.ok_tx_id(blob, len(blob), tx_i, out) == 0
        return bytes(out)


def gen_block(oracle, seed: int, n_txs: int, *, pct_multi_input=0, pct_ecdsa=0,
              pct_invalid=0, pct_multisig=0, payload_len=0, utxo_daa=1000,
              pct_alt_hashtype=0):
    """Build a block's worth of independent txs + populated entries.

    Returns (blob_bytes, meta) where meta notes which txs carry invalid sigs.
    The consensus no-chained-txs rule (body_validation_in_isolation.rs:126-152)
    guarantees in-block independence — every input spends a pre-block utxo.
    """
    rng = random.Random(seed)
    sg = Signer(oracle)
    txs = []
    specs = []
    key_seq = 0

    for t in range(n_txs):
        roll = rng.randrange(100)
        kind = "p2pk"
        n_in = 1
        if roll < pct_multisig:
            kind = "multisig"
        elif roll < pct_multisig + pct_ecdsa:
            kind = "ecdsa"
        elif roll < pct_multisig + pct_ecdsa + pct_multi_input:
            kind = "p2pk"
            n_in = rng.randrange(2, 9)
        invalid = rng.randrange(100) < pct_invalid

        inputs = []
        in_specs = []
        total = 0
        for i in range(n_in):
            key = key_seq
            key_seq += 1
            amount = rng.randrange(10_000, 1_000_000)
            total += amount
            prev = bytes(rng.randrange(256) for _ in range(32))
            if kind == "p2pk":
                spk = sg.p2pk_spk(key)
            elif kind == "ecdsa":
                spk = sg.p2pk_ecdsa_spk(key)
            else:  # multisig m-of-n
                m = rng.randrange(2, 4)
                n = rng.randrange(m, min(m + 3, 6))
                keys = [key_seq + j for j in range(n)]
                key_seq += n
                redeem = sg.multisig_redeem(m, keys)
                spk = sg.p2sh_spk(redeem)
                in_specs.append({"kind": kind, "key": keys[:m], "m": m, "n": n,
                                 "redeem": redeem, "keys": keys})
                inputs.append(B.tx_input(prev, rng.randrange(4),
                                         sequence=0, commit_kind=0,
                                         commit_value=n,
                                         utxo=B.utxo_entry(amount, spk, utxo_daa)))
                continue
            htype = SIGHASH_ALL
            if rng.randrange(100) < pct_alt_hashtype:
                # the other five valid SigHashTypes (sighash_type.rs:15-22 —
                # bit flags, so Single is 0x04): None, Single, and the
                # AnyOneCanPay variants
                htype = rng.choice([0x02, 0x04, 0x81, 0x82, 0x84])
            in_specs.append({"kind": kind, "key": key, "htype": htype})
            inputs.append(B.tx_input(prev, rng.randrange(4), sequence=0,
                                     commit_kind=0, commit_value=1,
                                     utxo=B.utxo_entry(amount, spk, utxo_daa)))
        fee = rng.randrange(1000, 5000)
        n_out = rng.randrange(1, 3)
        out_vals = []
        remain = total - fee
        for i in range(n_out - 1):
            v = remain // 2
            out_vals.append(v)
            remain -= v
        out_vals.append(remain)
        out_keys = [key_seq + 1000 + i for i in range(len(out_vals))]
        outputs = [B.tx_output(v, sg.p2pk_spk(k))
                   for k, v in zip(out_keys, out_vals)]
        payload = bytes(rng.randrange(256) for _ in range(payload_len))
        txs.append(B.tx_dict(0, inputs, outputs, payload=payload))
        specs.append({"inputs": in_specs, "invalid": invalid,
                      "out_keys": out_keys, "out_vals": out_vals})

    # pass 1: tx ids (exclude sig scripts, so ids are final before signing)
    blob = B.build_blob(txs)
    for t in range(n_txs):
        txs[t]["tx_id"] = sg.tx_id(blob, t)

    # pass 2: sighash + sign
    blob = B.build_blob(txs)
    for t, spec in enumerate(specs):
        for i, ins in enumerate(spec["inputs"]):
            if ins["kind"] == "multisig":
                sig_script = b""
                for key in ins["key"]:
                    msg = sg.sighash(blob, t, i)
                    sig = sg.schnorr_sign(key, msg)
                    sig_script += bytes([0x41]) + sig + bytes([SIGHASH_ALL])
                sig_script += bytes([0x4C, len(ins["redeem"])]) + ins["redeem"]
            else:
                ecdsa = ins["kind"] == "ecdsa"
                ht = ins.get("htype", SIGHASH_ALL)
                msg = sg.sighash(blob, t, i, hash_type=ht, ecdsa=ecdsa)
                sig = (sg.ecdsa_sign if ecdsa else sg.schnorr_sign)(ins["key"], msg)
                sig_script = bytes([0x41]) + sig + bytes([ht])
            if spec["invalid"] and i == 0:
                sig_script = bytearray(sig_script)
                sig_script[10] ^= 0x40  # corrupt the first signature
                sig_script = bytes(sig_script)
            txs[t]["inputs"][i]["sig_script"] = sig_script

    # storage-mass commitment (KIP-0009; mirrors mass/mod.rs:385-514) so the
    # blocks validate under KV_FLAGS_FULL as well
    for t in txs:
        t["storage_mass"] = storage_mass(t)

    blob = B.build_blob(txs)
    for t in range(n_txs):
        specs[t]["tx_id"] = txs[t]["tx_id"]
    return blob, {"n_txs": n_txs, "specs": specs}


def gen_spend_block(oracle, seed: int, prev_meta, prev_block_daa: int,
                    accepted=None):
    """Build a follow-up block whose txs spend the PREVIOUS block's outputs
    (P2PK schnorr, 1-in/1-out) — the cross-block chain the UTXO-diff apply
    must sustain (utxo_diff.rs:224). `accepted` filters which prev txs'
    outputs exist (default: the non-invalid ones). Returns (blob, meta) in the
    same populated form as gen_block; strip_utxo_entries turns it into the
    table-resolved shape."""
    rng = random.Random(seed)
    sg = Signer(oracle)
    txs = []
    chosen = []
    for t, spec in enumerate(prev_meta["specs"]):
        ok = spec.get("invalid") is False if accepted is None else accepted[t]
        if not ok:
            continue
        for i, (k, v) in enumerate(zip(spec["out_keys"], spec["out_vals"])):
            chosen.append((spec["tx_id"], i, k, v))
    for tx_id, idx, key, value in chosen:
        spk = sg.p2pk_spk(key)
        fee = min(1000, value - 1) if value > 1 else 0
        inp = B.tx_input(tx_id, idx, sequence=0, commit_kind=0, commit_value=1,
                         utxo=B.utxo_entry(value, spk, prev_block_daa))
        out = B.tx_output(value - fee, sg.p2pk_spk(key + 500000))
        txs.append(B.tx_dict(0, [inp], [out]))
    blob = B.build_blob(txs)
    for t in range(len(txs)):
        txs[t]["tx_id"] = sg.tx_id(blob, t)
    blob = B.build_blob(txs)
    for t, (tx_id, idx, key, value) in enumerate(chosen):
        msg = sg.sighash(blob, t, 0)
        sig = sg.schnorr_sign(key, msg)
        txs[t]["inputs"][0]["sig_script"] = bytes([0x41]) + sig + bytes([SIGHASH_ALL])
    for t in txs:
        t["storage_mass"] = storage_mass(t)
    return B.build_blob(txs), {"n_txs": len(txs)}


STORM = 10**8 * 10**4


def _plurality(spk_len, has_cov):
    return -(-(63 + spk_len + (32 if has_cov else 0)) // 100)


def storage_mass(tx):
    outs_p = 0
    harm_outs = 0
    for o in tx["outputs"]:
        p = _plurality(len(o["spk"]), bool(o["covenant"]))
        harm_outs += STORM * p * p // o["value"]
        outs_p += p
    ins = tx["inputs"]
    if outs_p == 1:
        relaxed = True
    elif len(ins) > 2:
        relaxed = False
    else:
        ins_p = sum(_plurality(len(i["utxo"]["spk"]), bool(i["utxo"]["covenant_id"]))
                    for i in ins)
        relaxed = ins_p == 1 or (outs_p == 2 and ins_p == 2)
    if relaxed:
        harm_ins = sum(STORM * _plurality(len(i["utxo"]["spk"]),
                                          bool(i["utxo"]["covenant_id"])) ** 2
                       // i["utxo"]["amount"] for i in ins)
        return max(0, harm_outs - harm_ins)
    ins_p = sum(_plurality(len(i["utxo"]["spk"]), bool(i["utxo"]["covenant_id"]))
                for i in ins)
    mean = max(1, sum(i["utxo"]["amount"] for i in ins) // ins_p)
    return max(0, harm_outs - ins_p * (STORM // mean))
