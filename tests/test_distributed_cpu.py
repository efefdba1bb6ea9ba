"""Multi-process (gloo, world_size=2) tests of the cross-GPU exchange logic:
verdict-bitmap all-reduce + muhash-partial all-gather/fold — the only
collectives the sharded path needs. Runs on CPU here; the same
tensor logic runs over RCCL/xGMI in bench.py on the GPU box."""
import ctypes
import os

import torch
import torch.multiprocessing as mp


def _worker(rank, world, port, results):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    import torch.distributed as dist
    dist.init_process_group("gloo", rank=rank, world_size=world)

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    oracle = ctypes.CDLL(os.path.join(repo, "oracle", "liboracle.so"))
    engine_lib = ctypes.CDLL(os.path.join(repo, "rusty_kaspa_amd", "libkaspa_gpu.so"))

    # each rank builds a muhash partial from its own shard of elements
    U = ctypes.c_uint64 * 48
    num, den = U(), U()
    oracle.ok_u3072_one(num)
    oracle.ok_u3072_one(den)
    elem = U()
    for i in range(6):
        data = bytes([rank * 100 + i]) * 50
        oracle.ok_muhash_element(data, 50, elem)
        oracle.ok_u3072_mul(num if i % 2 else den, elem)
    partial = bytearray(768)
    partial[0:384] = b"".join(int(num[k]).to_bytes(8, "little") for k in range(48))
    partial[384:768] = b"".join(int(den[k]).to_bytes(8, "little") for k in range(48))

    # bitmap exchange: rank r marks bits [r*64, r*64+64) valid
    bitmap = torch.zeros(world, dtype=torch.int64)
    bitmap[rank] = -1  # all 64 bits set
    dist.all_reduce(bitmap, op=dist.ReduceOp.SUM)  # disjoint shards: SUM == OR

    # the bench's verify leg runs IDENTICAL cached batches per rank and
    # exchanges with MAX (bit-exact when every rank's words agree)
    bitmap_max = torch.full((world,), 0x0F0F, dtype=torch.int64)
    dist.all_reduce(bitmap_max, op=dist.ReduceOp.MAX)
    assert (bitmap_max == 0x0F0F).all()

    # muhash partial exchange: all-gather + local multiplicative fold
    t = torch.frombuffer(bytes(partial), dtype=torch.uint8).clone()
    gathered = [torch.zeros_like(t) for _ in range(world)]
    dist.all_gather(gathered, t)
    acc = bytearray(768)
    acc[0] = 1
    acc[384] = 1
    accbuf = (ctypes.c_uint8 * 768).from_buffer(acc)
    for g in gathered:
        other = bytes(g.numpy().tobytes())
        assert engine_lib.kv_muhash_combine(None, accbuf, other) == 0
    out = (ctypes.c_uint8 * 32)()
    assert engine_lib.kv_muhash_finalize(None, bytes(acc), out) == 0

    results[rank] = (list(bitmap), bytes(out))
    dist.destroy_process_group()


def test_bitmap_and_muhash_exchange(oracle):
    import random
    port = random.Random(os.getpid()).randrange(20000, 40000)
    mgr = mp.Manager()
    results = mgr.dict()
    mp.spawn(_worker, args=(2, port, results), nprocs=2, join=True)
    b0, m0 = results[0]
    b1, m1 = results[1]
    assert b0 == b1 == [-1, -1]
    assert m0 == m1

    # the folded commitment must equal the oracle computing ALL elements locally
    U = ctypes.c_uint64 * 48
    num, den = U(), U()
    oracle.ok_u3072_one(num)
    oracle.ok_u3072_one(den)
    elem = U()
    for rank in range(2):
        for i in range(6):
            data = bytes([rank * 100 + i]) * 50
            oracle.ok_muhash_element(data, 50, elem)
            oracle.ok_u3072_mul(num if i % 2 else den, elem)
    exp = (ctypes.c_uint8 * 32)()
    oracle.ok_muhash_finalize(num, den, exp)
    assert m0 == bytes(exp)

