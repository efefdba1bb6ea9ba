"""Transaction-batch blob builder — the host-side flattening of a block's
transactions + populated UTXO entries into the C-ABI format defined in
include/kaspa_engine_abi.h.

This mirrors what a Rust host would do at the FFI seam before calling
kv_validate_block (the borrow of PopulatedTransaction: tx + entries,
consensus/core/src/tx.rs:254 + utxo/utxo_entry.rs:20).
"""
from __future__ import annotations

import struct

SUBNETWORK_NATIVE = bytes(20)
SUBNETWORK_COINBASE = bytes([1] + [0] * 19)


def tx_dict(version=0, inputs=(), outputs=(), lock_time=0,
            subnetwork_id=SUBNETWORK_NATIVE, gas=0, payload=b"", storage_mass=0,
            tx_id=bytes(32)):
    return {
        "version": version, "inputs": list(inputs), "outputs": list(outputs),
        "lock_time": lock_time, "subnetwork_id": subnetwork_id, "gas": gas,
        "payload": payload, "storage_mass": storage_mass, "tx_id": tx_id,
    }


def tx_input(prev_tx_id, prev_index, sequence=0, sig_script=b"", commit_kind=0,
             commit_value=0, utxo=None):
    return {
        "prev_tx_id": prev_tx_id, "prev_index": prev_index, "sequence": sequence,
        "sig_script": sig_script, "commit_kind": commit_kind,
        "commit_value": commit_value, "utxo": utxo,
    }


def utxo_entry(amount, spk, daa_score=0, is_coinbase=False, spk_version=0,
               covenant_id=None):
    return {
        "amount": amount, "spk": spk, "daa_score": daa_score,
        "is_coinbase": is_coinbase, "spk_version": spk_version,
        "covenant_id": covenant_id,
    }


def tx_output(value, spk, spk_version=0, covenant=None):
    return {"value": value, "spk": spk, "spk_version": spk_version,
            "covenant": covenant}


def _encode_tx(tx) -> bytes:
    parts = [struct.pack("<HHHH", tx["version"], len(tx["inputs"]),
                         len(tx["outputs"]), 0)]
    parts.append(struct.pack("<Q", tx["lock_time"]))
    assert len(tx["subnetwork_id"]) == 20
    parts.append(tx["subnetwork_id"])
    parts.append(struct.pack("<IQQ", len(tx["payload"]), tx["gas"],
                             tx["storage_mass"]))
    assert len(tx["tx_id"]) == 32
    parts.append(tx["tx_id"])
    parts.append(tx["payload"])
    for i in tx["inputs"]:
        assert len(i["prev_tx_id"]) == 32
        parts.append(i["prev_tx_id"])
        parts.append(struct.pack("<IQ", i["prev_index"], i["sequence"]))
        parts.append(struct.pack("<BBH", i["commit_kind"], 0, i["commit_value"]))
        parts.append(struct.pack("<I", len(i["sig_script"])))
        parts.append(i["sig_script"])
        u = i["utxo"]
        parts.append(struct.pack("<QQBBH", u["amount"], u["daa_score"],
                                 1 if u["is_coinbase"] else 0,
                                 1 if u["covenant_id"] else 0, u["spk_version"]))
        parts.append(struct.pack("<I", len(u["spk"])))
        parts.append(u["spk"])
        if u["covenant_id"]:
            assert len(u["covenant_id"]) == 32
            parts.append(u["covenant_id"])
    for o in tx["outputs"]:
        parts.append(struct.pack("<QHHI", o["value"], o["spk_version"], 0,
                                 len(o["spk"])))
        parts.append(o["spk"])
        if o["covenant"]:
            auth, cid = o["covenant"]
            assert len(cid) == 32
            parts.append(struct.pack("<BH", 1, auth))
            parts.append(cid)
        else:
            parts.append(b"\x00")
    return b"".join(parts)


def build_blob(txs) -> bytes:
    """txs: list of tx dicts (see tx_dict)."""
    encoded = [_encode_tx(t) for t in txs]
    header_len = 4 + 4 * len(txs)
    offsets = []
    off = header_len
    for e in encoded:
        offsets.append(off)
        off += len(e)
    out = [struct.pack("<I", len(txs))]
    out += [struct.pack("<I", o) for o in offsets]
    out += encoded
    return b"".join(out)


def finalize_tx_ids(txs, compute_id) -> None:
    """Fill each tx's tx_id using compute_id(blob_bytes, tx_index) -> bytes32.

    Mirrors Transaction::finalize (consensus/core/src/tx.rs:399) which caches the
    id at construction time — the id is an input to validation, not part of it.
    """
    blob = build_blob(txs)
    for i, t in enumerate(txs):
        t["tx_id"] = compute_id(blob, i)


def strip_utxo_entries(blob: bytes):
    """Split a populated blob into (unpopulated_blob, seeds).

    The unpopulated blob has every input's UtxoEntry fields zeroed with
    spk_len = 0 — the shape a node hands to kv_validate_block_utxo, where
    entries resolve from the GPU-resident UTXO table instead. `seeds` is the
    list of (outpoint36, entry64) pairs to kv_utxo_upsert beforehand
    (packed-64B table layout). Mirrors the populate contract of
    utxo_validation.rs:351-390.
    """
    n_txs, = struct.unpack_from("<I", blob, 0)
    offs = list(struct.unpack_from(f"<{n_txs}I", blob, 4))
    seeds = []
    out_txs = []
    for t in range(n_txs):
        off = offs[t]
        end = offs[t + 1] if t + 1 < n_txs else len(blob)
        p = off
        n_in, n_out = struct.unpack_from("<HH", blob, p + 2)
        payload_len, = struct.unpack_from("<I", blob, p + 36)
        p += 88 + payload_len
        chunks = [blob[off:p]]
        for _ in range(n_in):
            outpoint = blob[p:p + 36]
            sig_len, = struct.unpack_from("<I", blob, p + 48)
            chunks.append(blob[p:p + 52 + sig_len])
            p += 52 + sig_len
            amount, daa = struct.unpack_from("<QQ", blob, p)
            is_cb, has_cov = blob[p + 16], blob[p + 17]
            spkv, = struct.unpack_from("<H", blob, p + 18)
            spk_len, = struct.unpack_from("<I", blob, p + 20)
            spk = blob[p + 24:p + 24 + spk_len]
            # seeds carry inline spks only; long scripts (spk_len > 36) go
            # through kv_utxo_upsert_spk — their seed entry keeps the real
            # spk_len with a zeroed inline area and the caller supplies the
            # bytes via the spk blob (tests do this explicitly)
            entry64 = struct.pack("<QQHHI", amount, daa, is_cb & 1, spkv,
                                  spk_len) + spk[:36].ljust(36, b"\0") + bytes(4)
            seeds.append((outpoint, entry64))
            chunks.append(bytes(24))  # zero entry, spk_len 0
            p += 24 + spk_len + (32 if has_cov else 0)
        chunks.append(blob[p:end])  # outputs verbatim
        out_txs.append(b"".join(chunks))
    header = [struct.pack("<I", n_txs)]
    off = 4 + 4 * n_txs
    for e in out_txs:
        header.append(struct.pack("<I", off))
        off += len(e)
    return b"".join(header + out_txs), seeds
