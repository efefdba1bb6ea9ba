"""ctypes host wrapper over the MI355X-native engine C-ABI (libkaspa_gpu.so).

This is plumbing over include/kaspa_engine_abi.h — the same surface a Rust host
would bind over FFI. The engine has NO CPU fallback:
constructing an Engine raises if the shared library is missing or no HIP
device is present.
"""
from __future__ import annotations

import ctypes
import os

_REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
_LIB_PATH = os.path.join(_REPO, "rusty_kaspa_amd", "libkaspa_gpu.so")


class KvParams(ctypes.Structure):
    _fields_ = [("coinbase_maturity", ctypes.c_uint64),
                ("mass_per_sig_op", ctypes.c_uint64),
                ("sig_cache_size", ctypes.c_uint64),
                ("device", ctypes.c_int)]


class KvCacheStats(ctypes.Structure):
    _fields_ = [("insertions", ctypes.c_uint64), ("hits", ctypes.c_uint64),
                ("misses", ctypes.c_uint64)]


def load_library() -> ctypes.CDLL:
    if not os.path.exists(_LIB_PATH):
        raise RuntimeError(
            f"libkaspa_gpu.so not built at {_LIB_PATH} — run __graft_entry__.build()")
    lib = ctypes.CDLL(_LIB_PATH)
    lib.kv_create.restype = ctypes.c_void_p
    lib.kv_create.argtypes = [ctypes.POINTER(KvParams)]
    lib.kv_last_error.restype = ctypes.c_char_p
    return lib


class Engine:
    """One engine context == one GPU (⇔ TransactionValidator instance,
    consensus/src/processes/transaction_validator/mod.rs:15)."""

    def __init__(self, device: int = -1, coinbase_maturity: int = 1000,
                 mass_per_sig_op: int = 1000, sig_cache_size: int = 10_000):
        self.lib = load_library()
        params = KvParams(coinbase_maturity, mass_per_sig_op, sig_cache_size,
                          device)
        self.ctx = self.lib.kv_create(ctypes.byref(params))
        if not self.ctx:
            err = self.lib.kv_last_error().decode()
            raise RuntimeError(f"kv_create failed: {err}")

    def close(self):
        if getattr(self, "ctx", None):
            self.lib.kv_destroy(ctypes.c_void_p(self.ctx))
            self.ctx = None

    def __del__(self):
        try:
            self.close()
        except Exception:
            pass

    def _check(self, rc: int):
        if rc != 0:
            raise RuntimeError(
                f"engine call failed rc={rc}: {self.lib.kv_last_error().decode()}")

    def verify_schnorr_batch(self, tuples: bytes, n: int, with_status=False):
        words = (n + 63) // 64
        bitmap = (ctypes.c_uint64 * words)()
        status = (ctypes.c_uint8 * n)() if with_status else None
        if with_status:
            rc = self.lib.kv_verify_schnorr_batch_status(
                ctypes.c_void_p(self.ctx), tuples, ctypes.c_size_t(n), bitmap, status)
        else:
            rc = self.lib.kv_verify_schnorr_batch(
                ctypes.c_void_p(self.ctx), tuples, ctypes.c_size_t(n), bitmap)
        self._check(rc)
        return (list(bitmap), bytes(status) if with_status else None)

    def verify_ecdsa_batch(self, tuples: bytes, n: int, with_status=False):
        words = (n + 63) // 64
        bitmap = (ctypes.c_uint64 * words)()
        status = (ctypes.c_uint8 * n)() if with_status else None
        if with_status:
            rc = self.lib.kv_verify_ecdsa_batch_status(
                ctypes.c_void_p(self.ctx), tuples, ctypes.c_size_t(n), bitmap, status)
        else:
            rc = self.lib.kv_verify_ecdsa_batch(
                ctypes.c_void_p(self.ctx), tuples, ctypes.c_size_t(n), bitmap)
        self._check(rc)
        return (list(bitmap), bytes(status) if with_status else None)

    def muhash_finalize(self, partial768: bytes) -> bytes:
        out = (ctypes.c_uint8 * 32)()
        rc = self.lib.kv_muhash_finalize(ctypes.c_void_p(self.ctx), partial768, out)
        self._check(rc)
        return bytes(out)

    def muhash_combine(self, acc768: bytearray, other768: bytes) -> None:
        buf = (ctypes.c_uint8 * 768).from_buffer(acc768)
        rc = self.lib.kv_muhash_combine(ctypes.c_void_p(self.ctx), buf, other768)
        self._check(rc)

    def validate_block(self, blob: bytes, n_txs: int, pov_daa: int, block_daa: int,
                       flags: int = 2, want_muhash: bool = True, raw: bool = False):
        codes = (ctypes.c_int32 * n_txs)()
        fees = (ctypes.c_uint64 * n_txs)()
        partial = (ctypes.c_uint8 * 768)() if want_muhash else None
        rc = self.lib.kv_validate_block(
            ctypes.c_void_p(self.ctx), blob, ctypes.c_size_t(len(blob)),
            ctypes.c_uint64(pov_daa), ctypes.c_uint64(block_daa),
            ctypes.c_uint32(flags), codes, fees, partial)
        self._check(rc)
        if raw:  # bench hot loop: skip the O(n) ctypes->list conversions
            return codes, fees, partial
        return list(codes), list(fees), bytes(partial) if want_muhash else None

    def validate_block_utxo(self, blob: bytes, n_txs: int, pov_daa: int,
                            block_daa: int, flags: int = 2, apply_diff: bool = True,
                            want_muhash: bool = True):
        """Populate from the GPU-resident UTXO table, validate, apply diff."""
        codes = (ctypes.c_int32 * n_txs)()
        fees = (ctypes.c_uint64 * n_txs)()
        partial = (ctypes.c_uint8 * 768)() if want_muhash else None
        rc = self.lib.kv_validate_block_utxo(
            ctypes.c_void_p(self.ctx), blob, ctypes.c_size_t(len(blob)),
            ctypes.c_uint64(pov_daa), ctypes.c_uint64(block_daa),
            ctypes.c_uint32(flags), ctypes.c_int(1 if apply_diff else 0),
            codes, fees, partial)
        self._check(rc)
        return list(codes), list(fees), bytes(partial) if want_muhash else None

    def sig_cache_stats(self):
        out = KvCacheStats()
        self._check(self.lib.kv_sig_cache_stats(ctypes.c_void_p(self.ctx),
                                                ctypes.byref(out)))
        return out.insertions, out.hits, out.misses

    def block_body_check(self, blob: bytes):
        """Merkle root over tx hashes + duplicate/double-spend/chained checks."""
        root = (ctypes.c_uint8 * 32)()
        code = ctypes.c_int32()
        rc = self.lib.kv_block_body_check(
            ctypes.c_void_p(self.ctx), blob, ctypes.c_size_t(len(blob)),
            root, ctypes.byref(code))
        self._check(rc)
        return bytes(root), code.value

    def validate_mempool(self, blob: bytes, n_txs: int, pov_daa: int,
                         feerate_threshold: float = 0.0,
                         from_utxo_table: bool = False):
        codes = (ctypes.c_int32 * n_txs)()
        fees = (ctypes.c_uint64 * n_txs)()
        rc = self.lib.kv_validate_mempool(
            ctypes.c_void_p(self.ctx), blob, ctypes.c_size_t(len(blob)),
            ctypes.c_uint64(pov_daa), ctypes.c_double(feerate_threshold),
            ctypes.c_int(1 if from_utxo_table else 0), codes, fees)
        self._check(rc)
        return list(codes), list(fees)
