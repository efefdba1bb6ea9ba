"""Extract the goref-1060-tx-265-blocks DAG fixture (the reference's own
integration-test data, an independently produced Go-implementation DAG) into
batched populated tx blobs: every non-coinbase transaction whose inputs all
resolve to non-coinbase outputs inside the DAG, with UtxoEntries rebuilt from
the creating blocks. These carry REAL schnorr signatures made by the Go
implementation — end-to-end parity anchors for the sighash+verify pipeline.
Run in the build container (reads /root/reference); writes
tests/golden/goref_txs.json.gz."""
import ctypes
import gzip
import json
import sys

sys.path.insert(0, '/root/repo')
import rusty_kaspa_amd.blob as B

FIXTURES = {
    'goref_txs.json.gz': 'goref-1060-tx-265-blocks',
    'goref_pruning_txs.json.gz': 'goref_custom_pruning_depth',
}
import os
NAME = os.environ.get('GOREF_OUT', 'goref_txs.json.gz')
FIXTURE = ('/root/reference/testing/integration/testdata/dags_for_json_tests/'
           f'{FIXTURES[NAME]}/blocks.json.gz')
O = ctypes.CDLL('/root/repo/oracle/liboracle.so')

with gzip.open(FIXTURE) as f:
    lines = f.read().decode().splitlines()
blocks = [json.loads(ln) for ln in lines[1:]]


def tx_to_dict(tx, entries=None):
    ins = []
    for i, inp in enumerate(tx['inputs']):
        utxo = entries[i] if entries else B.utxo_entry(0, b'')
        ins.append(B.tx_input(
            bytes.fromhex(inp['previousOutpoint']['transactionId']),
            inp['previousOutpoint']['index'], sequence=inp['sequence'],
            sig_script=bytes.fromhex(inp['signatureScript']),
            commit_kind=0, commit_value=inp.get('sigOpCount', 0), utxo=utxo))
    outs = []
    for o in tx['outputs']:
        spk_raw = bytes.fromhex(o['scriptPublicKey'])
        outs.append(B.tx_output(o['value'], spk_raw[2:],
                                spk_version=int.from_bytes(spk_raw[:2], 'little')))
    d = B.tx_dict(tx['version'], ins, outs, lock_time=tx['lockTime'],
                  subnetwork_id=bytes.fromhex(tx['subnetworkId']),
                  gas=tx['gas'], payload=bytes.fromhex(tx['payload']))
    d['storage_mass'] = tx.get('storageMass', 0) or 0
    return d


out_map = {}
for blk in blocks:
    txd = [tx_to_dict(t) for t in blk['transactions']]
    blob = B.build_blob(txd)
    for ti, t in enumerate(blk['transactions']):
        idb = (ctypes.c_uint8 * 32)()
        assert O.ok_tx_id(blob, len(blob), ti, idb) == 0
        tid = bytes(idb).hex()
        is_cb = t['subnetworkId'].startswith('01')
        for oi, o in enumerate(t['outputs']):
            spk_raw = bytes.fromhex(o['scriptPublicKey'])
            out_map[(tid, oi)] = (o['value'], spk_raw[2:],
                                  int.from_bytes(spk_raw[:2], 'little'),
                                  blk['header']['daaScore'], is_cb)

picked, seen = [], set()
for blk in blocks:
    for t in blk['transactions']:
        if not t['inputs']:
            continue
        key = json.dumps(t, sort_keys=True)
        if key in seen:
            continue
        seen.add(key)
        entries = []
        ok = True
        for inp in t['inputs']:
            k = (inp['previousOutpoint']['transactionId'],
                 inp['previousOutpoint']['index'])
            if k not in out_map:
                ok = False
                break
            value, spk, spkv, daa, is_cb = out_map[k]
            entries.append(B.utxo_entry(value, spk, daa_score=daa,
                                        is_coinbase=is_cb, spk_version=spkv))
        if ok:
            picked.append(tx_to_dict(t, entries))

print(f"{len(picked)} resolvable real txs")
BATCH = 64
batches = []
for i in range(0, len(picked), BATCH):
    blob = B.build_blob(picked[i:i + BATCH])
    # fill carried ids
    import struct
    n, = struct.unpack_from('<I', blob, 0)
    offs = list(struct.unpack_from(f'<{n}I', blob, 4))
    out = bytearray(blob)
    for ti in range(n):
        idb = (ctypes.c_uint8 * 32)()
        assert O.ok_tx_id(bytes(blob), len(blob), ti, idb) == 0
        out[offs[ti] + 56:offs[ti] + 88] = bytes(idb)
    batches.append(bytes(out))

# validate NOW: every real signature must verify
total = bad = 0
for blob in batches:
    import struct
    n, = struct.unpack_from('<I', blob, 0)
    codes = (ctypes.c_int32 * n)()
    fees = (ctypes.c_uint64 * n)()
    mh = (ctypes.c_uint8 * 32)()
    rc = O.ok_validate_block_parallel(blob, ctypes.c_size_t(len(blob)),
                                      ctypes.c_uint64(10**9), ctypes.c_uint64(10**9),
                                      2, 8, codes, fees, mh)
    assert rc == 0
    total += n
    bad += sum(1 for c in codes if c != 0)
    if bad:
        print("codes:", [c for c in codes][:10])
print(f"oracle: {total} txs, {bad} rejected")
if bad == 0:
    data = {"note": f"{FIXTURES[NAME]} real Go-implementation txs with "
                    "rebuilt UtxoEntries; every signature must verify",
            "batches": [b.hex() for b in batches]}
    with gzip.open(f'/root/repo/tests/golden/{NAME}', 'wt') as f:
        json.dump(data, f)
    print("golden written:", NAME)
