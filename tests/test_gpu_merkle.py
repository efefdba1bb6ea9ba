"""GPU parity for the body-in-isolation batch (kv_block_body_check):
merkle root (GPU leaf hashes + host fold) bit-exact vs the oracle and the
reference's merkle_root_test vector; duplicate/double-spend/chained rule
codes match the oracle's."""
import ctypes
import json
import os
import struct
import sys

import pytest

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", "oracle"))
from workload import gen_block  # noqa: E402

pytestmark = pytest.mark.gpu

GOLDEN = os.path.join(os.path.dirname(__file__), "golden", "merkle.json")


@pytest.fixture(scope="module")
def engine():
    from rusty_kaspa_amd.engine import Engine
    eng = Engine()
    yield eng
    eng.close()


def test_reference_vector(engine):
    g = json.load(open(GOLDEN))
    for blob_hex, root_hex in [(g["blob_mass0"], g["root_mass0"]),
                               (g["blob_mass7"], g["root_mass7"])]:
        root, code = engine.block_body_check(bytes.fromhex(blob_hex))
        assert root.hex() == root_hex
        assert code == 0


@pytest.mark.parametrize("kwargs", [
    dict(seed=51, n_txs=63, pct_multi_input=20, pct_ecdsa=10),
    dict(seed=52, n_txs=256, pct_multisig=10, payload_len=40),
    dict(seed=53, n_txs=1),
])
def test_merkle_vs_oracle(oracle, engine, kwargs):
    blob, _ = gen_block(oracle, **kwargs)
    expect = (ctypes.c_uint8 * 32)()
    assert oracle.ok_blob_merkle_root(blob, ctypes.c_size_t(len(blob)), expect) == 0
    root, code = engine.block_body_check(blob)
    assert root == bytes(expect)
    assert code == oracle.ok_body_check(blob, ctypes.c_size_t(len(blob))) == 0


def test_rule_codes_vs_oracle(oracle, engine):
    blob, _ = gen_block(oracle, seed=54, n_txs=20, pct_multi_input=25)
    n, = struct.unpack_from("<I", blob, 0)
    offs = list(struct.unpack_from(f"<{n}I", blob, 4))
    raws = []
    for t in range(n):
        end = offs[t + 1] if t + 1 < n else len(blob)
        raws.append(blob[offs[t]:end])

    def rebuild(raws2):
        out = [struct.pack("<I", len(raws2))]
        off = 4 + 4 * len(raws2)
        for rw in raws2:
            out.append(struct.pack("<I", off))
            off += len(rw)
        return b"".join(out) + b"".join(raws2)

    cases = []
    cases.append(rebuild(raws + [raws[0]]))            # duplicate tx
    r = bytearray(raws[1])
    r[88:88 + 36] = raws[0][88:88 + 36]                # double spend
    cases.append(rebuild([raws[0], bytes(r)] + raws[2:]))
    r = bytearray(raws[2])
    r[88:88 + 32] = raws[0][56:88]                     # chained
    r[88 + 32:88 + 36] = struct.pack("<I", 0)
    cases.append(rebuild(raws[:2] + [bytes(r)] + raws[3:]))
    for b in cases:
        root, code = engine.block_body_check(b)
        assert code == oracle.ok_body_check(b, ctypes.c_size_t(len(b)))
        assert code in (10, 11, 12)
        expect = (ctypes.c_uint8 * 32)()
        assert oracle.ok_blob_merkle_root(b, ctypes.c_size_t(len(b)), expect) == 0
        assert root == bytes(expect)
