"""CPU round-trip of the populate-mode blob transform.

strip_utxo_entries removes every input's UtxoEntry (the shape handed to
kv_validate_block_utxo); repopulating from the extracted (outpoint, entry)
seeds must reproduce the original populated blob byte-for-byte — the same
rebuild the engine performs internally after the GPU table lookup
(⇔ utxo_validation.rs:351-390 populate contract).
"""
import os
import struct
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", "oracle"))

from workload import gen_block  # noqa: E402

from rusty_kaspa_amd.blob import strip_utxo_entries  # noqa: E402


def repopulate(stripped: bytes, seeds):
    """Pure-python inverse of strip_utxo_entries (mirrors the engine rebuild)."""
    n_txs, = struct.unpack_from("<I", stripped, 0)
    offs = list(struct.unpack_from(f"<{n_txs}I", stripped, 4))
    out_txs = []
    it = iter(seeds)
    for t in range(n_txs):
        off = offs[t]
        end = offs[t + 1] if t + 1 < n_txs else len(stripped)
        p = off
        n_in, _ = struct.unpack_from("<HH", stripped, p + 2)
        payload_len, = struct.unpack_from("<I", stripped, p + 36)
        p += 88 + payload_len
        chunks = [stripped[off:p]]
        for _ in range(n_in):
            sig_len, = struct.unpack_from("<I", stripped, p + 48)
            chunks.append(stripped[p:p + 52 + sig_len])
            p += 52 + sig_len
            _outpoint, e = next(it)
            amount, daa, flags, spkv, spk_len = struct.unpack_from("<QQHHI", e)
            chunks.append(struct.pack("<QQBBH I".replace(" ", ""), amount, daa,
                                      flags & 1, 0, spkv, spk_len))
            chunks.append(e[24:24 + spk_len])
            p += 24  # stripped entry is the 24B zero header
        chunks.append(stripped[p:end])
        out_txs.append(b"".join(chunks))
    header = [struct.pack("<I", n_txs)]
    off = 4 + 4 * n_txs
    for e in out_txs:
        header.append(struct.pack("<I", off))
        off += len(e)
    return b"".join(header + out_txs)


def test_strip_repopulate_roundtrip(oracle):
    blob, meta = gen_block(oracle, seed=11, n_txs=40, pct_multi_input=25,
                           pct_ecdsa=10, pct_multisig=10, payload_len=16)
    stripped, seeds = strip_utxo_entries(blob)
    n_inputs = sum(1 for _ in seeds)
    assert n_inputs > 40  # multi-input txs present
    assert len(stripped) < len(blob)
    # outpoints are unique in a block (no chained txs) and 36B
    assert len({op for op, _ in seeds}) == n_inputs
    assert all(len(op) == 36 and len(e) == 64 for op, e in seeds)
    assert repopulate(stripped, seeds) == blob


def test_strip_is_stable(oracle):
    blob, _ = gen_block(oracle, seed=12, n_txs=8)
    stripped, seeds = strip_utxo_entries(blob)
    # stripping an already-stripped blob is the identity with zero-entry seeds
    stripped2, seeds2 = strip_utxo_entries(stripped)
    assert stripped2 == stripped
    assert all(e == b"\0" * 64 for _, e in seeds2)
