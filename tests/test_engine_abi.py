"""CPU-side checks of the product engine library: it must build for gfx950,
load without a GPU, and export every symbol include/kaspa_engine_abi.h declares.
No compute calls here (no GPU in the build container)."""
import ctypes
import os
import re
import subprocess

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
SO = os.path.join(REPO, "rusty_kaspa_amd", "libkaspa_gpu.so")


@pytest.fixture(scope="module")
def engine_lib():
    import __graft_entry__
    __graft_entry__.build()
    return ctypes.CDLL(SO)


def declared_symbols():
    hdr = open(os.path.join(REPO, "include", "kaspa_engine_abi.h")).read()
    # function declarations: return-type kv_xxx(
    return sorted(set(re.findall(r"\b(kv_[a-z0-9_]+)\s*\(", hdr)) - {"kv_create"}
                  | {"kv_create"})


def test_engine_builds_and_loads(engine_lib):
    assert engine_lib is not None


def test_all_abi_symbols_exported(engine_lib):
    missing = []
    for sym in declared_symbols():
        try:
            getattr(engine_lib, sym)
        except AttributeError:
            missing.append(sym)
    assert not missing, f"ABI symbols missing from libkaspa_gpu.so: {missing}"


def test_engine_fails_loudly_without_gpu(engine_lib):
    """In this CPU-only container kv_create must return NULL with a clear error
    (the product has no CPU fallback); on a GPU box it must succeed instead."""
    engine_lib.kv_create.restype = ctypes.c_void_p
    engine_lib.kv_last_error.restype = ctypes.c_char_p
    ctx = engine_lib.kv_create(None)
    import torch
    if torch.cuda.is_available():
        assert ctx, engine_lib.kv_last_error().decode()
        engine_lib.kv_destroy(ctypes.c_void_p(ctx))
    else:
        assert not ctx
        assert b"no HIP device" in engine_lib.kv_last_error()


def test_muhash_host_ops_work_without_gpu(engine_lib):
    """kv_muhash_combine/finalize are host-side (by design) — they must work
    even without a device context, pinned against the reference EMPTY_MUHASH."""
    acc = (ctypes.c_uint8 * 768)()
    acc[0] = 1
    acc[384] = 1
    out = (ctypes.c_uint8 * 32)()
    rc = engine_lib.kv_muhash_finalize(None, acc, out)
    assert rc == 0
    assert bytes(out).hex() == (
        "544eb3142c000f0ad2c76ac41f4222abbababed830eeafee4b6dc56b52d5cac0")
