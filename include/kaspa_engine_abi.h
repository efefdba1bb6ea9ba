/* kaspa_engine_abi.h — the C-ABI drop-in boundary of the MI355X-native
 * transaction-validation engine.
 *
 * This header declares exactly what a Rust host (rusty-kaspa) would bind over FFI
 * to replace its validation hot path. Each entry point cites the reference
 * interface it replaces (kaspanet/rusty-kaspa v2.0.1, paths relative to the
 * reference root):
 *
 *   kv_validate_block       ⇔ TransactionValidator::validate_populated_transaction_and_get_fee
 *                             fanned out per tx, plus the muhash monoid reduce
 *                             (consensus/src/processes/transaction_validator/
 *                              tx_validation_in_utxo_context.rs:37-67;
 *                              consensus/src/pipeline/virtual_processor/
 *                              utxo_validation.rs:319-348)
 *   kv_check_scripts        ⇔ check_scripts (tx_validation_in_utxo_context.rs:181-218)
 *                             for a single tx of the batch
 *   kv_verify_schnorr_batch ⇔ batched secp256k1::schnorr::Signature::verify
 *                             (crypto/txscript/src/lib.rs:869)
 *   kv_verify_ecdsa_batch   ⇔ batched secp256k1::ecdsa::Signature::verify
 *                             (crypto/txscript/src/lib.rs:899)
 *   kv_sighash_batch        ⇔ calc_schnorr_signature_hash / calc_ecdsa_signature_hash
 *                             (consensus/core/src/hashing/sighash.rs:245-292)
 *   kv_muhash_finalize      ⇔ MuHash::finalize (crypto/muhash/src/lib.rs:93)
 *
 * No torch types cross this boundary: plain pointers and sizes only.
 * See INTEGRATION.md for the Rust-side binding a maintainer would add.
 */
#ifndef KASPA_ENGINE_ABI_H
#define KASPA_ENGINE_ABI_H

#include <stddef.h>
#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

/* ======================================================================
 * Transaction batch blob format (little-endian throughout)
 *
 * The host (Rust over FFI, or the C++/Python harness here) flattens a block's
 * transactions WITH their populated UTXO entries into one contiguous buffer —
 * the borrow-only contract of validate_populated_transaction_and_get_fee
 * (caller owns everything, engine retains nothing).
 *
 *   Blob     := u32 n_txs, u32 tx_offset[n_txs], Tx...
 *   Tx       := u16 version, u16 n_inputs, u16 n_outputs, u16 _pad,
 *               u64 lock_time, u8 subnetwork_id[20], u32 payload_len, u64 gas,
 *               u64 storage_mass, u8 tx_id[32],   // id is carried AND re-derivable
 *               u8 payload[payload_len],
 *               Input[n_inputs], Output[n_outputs]
 *   Input    := u8 prev_tx_id[32], u32 prev_index, u64 sequence,
 *               u8 commit_kind (0 = sigop_count, 1 = compute_budget),
 *               u8 _pad, u16 commit_value,
 *               u32 sig_script_len, u8 sig_script[...],
 *               // populated UtxoEntry (consensus/core/src/utxo/utxo_entry.rs:20)
 *               u64 utxo_amount, u64 utxo_daa_score, u8 utxo_is_coinbase,
 *               u8 utxo_has_covenant_id, u16 utxo_spk_version,
 *               u32 utxo_spk_len, u8 utxo_spk[...],
 *               u8 utxo_covenant_id[32 if utxo_has_covenant_id else 0]
 *   Output   := u64 value, u16 spk_version, u16 _pad, u32 spk_len, u8 spk[...],
 *               u8 has_covenant, (u16 cov_authorizing_input, u8 cov_id[32] if has)
 * ====================================================================== */

/* ---- validation flags (TxValidationFlags, tx_validation_in_utxo_context.rs:24-34) */
#define KV_FLAGS_FULL 0u
#define KV_FLAGS_SKIP_SCRIPT_CHECKS 1u
#define KV_FLAGS_SKIP_MASS_CHECK 2u

/* ---- per-tx result codes ----
 * 0 = valid. 1..99 mirror TxRuleError variants on the non-script path
 * (consensus/src/processes/transaction_validator/errors.rs).
 * 100+s = SignatureInvalid(script error s); 200+s = SignatureEmpty(script error s)
 * (the map_script_err convention, tx_validation_in_utxo_context.rs:220-222). */
#define KV_OK 0
#define KV_ERR_IMMATURE_COINBASE 1
#define KV_ERR_INPUT_AMOUNT_OVERFLOW 2
#define KV_ERR_INPUT_AMOUNT_TOO_HIGH 3
#define KV_ERR_SPEND_TOO_HIGH 4
#define KV_ERR_WRONG_MASS 5
#define KV_ERR_SEQUENCE_LOCK 6
#define KV_ERR_FEERATE_TOO_LOW 7
#define KV_ERR_MASS_INCOMPUTABLE 8
#define KV_ERR_MISSING_OUTPOINT 9   /* populate failure: outpoint not in the UTXO set
                                       (ruleerrors::RuleError::MissingTxOutpoints) */
/* body-in-isolation rule codes (RuleError::{DuplicateTransactions,
 * DoubleSpendInSameBlock, ChainedTransaction}) */
#define KV_ERR_BODY_DUP_TX 10
#define KV_ERR_BODY_DOUBLE_SPEND 11
#define KV_ERR_BODY_CHAINED 12
#define KV_ERR_BAD_BLOB 90
#define KV_ERR_SIGNATURE_INVALID_BASE 100
#define KV_ERR_SIGNATURE_EMPTY_BASE 200

/* TxScriptError variants (kaspa-txscript-errors crate) used as s above */
#define KV_SCRIPT_EVAL_FALSE 1
#define KV_SCRIPT_EMPTY_STACK 2
#define KV_SCRIPT_CLEAN_STACK 3
#define KV_SCRIPT_NULL_FAIL 4
#define KV_SCRIPT_NOT_PUSH_ONLY 5
#define KV_SCRIPT_INVALID_SIG_HASH_TYPE 6
#define KV_SCRIPT_INVALID_PUBKEY 7
#define KV_SCRIPT_INVALID_SIGNATURE 8
#define KV_SCRIPT_VERIFY_ERROR 9
#define KV_SCRIPT_EARLY_RETURN 10
#define KV_SCRIPT_OPCODE_RESERVED 11
#define KV_SCRIPT_OPCODE_DISABLED 12
#define KV_SCRIPT_INVALID_OPCODE 13
#define KV_SCRIPT_MALFORMED_PUSH 14
#define KV_SCRIPT_STACK_SIZE_EXCEEDED 15
#define KV_SCRIPT_TOO_MANY_OPERATIONS 16
#define KV_SCRIPT_ELEMENT_TOO_BIG 17
#define KV_SCRIPT_INVALID_STACK_OPERATION 18
#define KV_SCRIPT_UNBALANCED_CONDITIONAL 19
#define KV_SCRIPT_INVALID_STATE 20
#define KV_SCRIPT_NUMBER_TOO_BIG 21
#define KV_SCRIPT_NOT_MINIMAL_DATA 22
#define KV_SCRIPT_INVALID_PUBKEY_COUNT 23
#define KV_SCRIPT_INVALID_SIGNATURE_COUNT 24
#define KV_SCRIPT_PUBKEY_FORMAT 25
#define KV_SCRIPT_SCRIPT_SIZE 26
#define KV_SCRIPT_NO_SCRIPTS 27
#define KV_SCRIPT_EXCEEDED_SCRIPT_UNITS 28
#define KV_SCRIPT_UNSATISFIED_LOCKTIME 29
#define KV_SCRIPT_INVALID_INPUT_INDEX 30
#define KV_SCRIPT_INVALID_SOURCE 31
#define KV_SCRIPT_INVALID_INDEX 32
#define KV_SCRIPT_INVALID_RANGE 33
#define KV_SCRIPT_UNSUPPORTED_OPCODE 63 /* engine-only: routed to CPU fallback */

/* Structurally-distinct "route this tx back to the reference CPU interpreter"
 * signal: an engine per-tx code that is NEGATIVE, so no binder can fold it
 * into the reject path by accident (every reject code is >= 1; 0 is accept).
 * Emitted when a script's execution reaches an opcode the engine does not
 * decide (currently only OpZkPrecompile, 0xa6). The legacy 1xx/2xx+63 codes
 * are no longer produced by the engine. */
#define KV_TX_DEFER_TO_CPU (-63)

/* ---- engine lifecycle ---- */

typedef struct kv_ctx kv_ctx; /* opaque; thread-safe (Sync), internally synchronized */

typedef struct {
  uint64_t coinbase_maturity;   /* params.coinbase_maturity(): mainnet 1000 (Crescendo) */
  uint64_t mass_per_sig_op;     /* params.mass_per_sig_op: mainnet 1000 grams */
  uint64_t sig_cache_size;      /* TransactionValidator sig_cache entries (10_000) */
  int device;                   /* HIP device ordinal, -1 = default */
} kv_params;

/* Create / destroy an engine context. Fails (returns NULL) if no HIP device is
 * available — there is NO CPU fallback in the product path. */
kv_ctx *kv_create(const kv_params *params);
void kv_destroy(kv_ctx *ctx);
/* Human-readable description of the last error on this thread. */
const char *kv_last_error(void);

/* ---- the drop-in entry points ---- */

/* Validate a block's transactions (blob format above) against their populated
 * entries. Writes per-tx result codes and fees, and the muhash partial
 * (numerator 384B ‖ denominator 384B LE) over the txs that validated OK —
 * mirroring validate_transactions_with_muhash_in_parallel
 * (utxo_validation.rs:319-348; coinbase txs are skipped by the caller there,
 * so the blob should not contain them — if it does, pass their index in
 * skip_mask). Returns 0 on success (even when some txs are invalid), <0 on
 * malformed blob / device error. Blocking; callable from multiple threads. */
int kv_validate_block(kv_ctx *ctx, const uint8_t *blob, size_t blob_len,
                      uint64_t pov_daa_score, uint64_t block_daa_score, uint32_t flags,
                      int32_t *tx_codes_out /*[n_txs]*/, uint64_t *fees_out /*[n_txs]*/,
                      uint8_t *muhash_partial_out /*[768] or NULL*/);

/* Batched BIP-340 Schnorr verify — the "verify_schnorr batch API" of the north
 * star. tuples = n × 128 bytes: r(32) ‖ s(32) ‖ pk_xonly(32) ‖ msg(32), SoA-major
 * not required (the engine re-lays-out on device). bitmap_out: bit i set ⇔ tuple
 * i valid. Returns 0, or <0 on device error. */
int kv_verify_schnorr_batch(kv_ctx *ctx, const uint8_t *tuples, size_t n,
                            uint64_t *bitmap_out /*[ceil(n/64)]*/);

/* Batched ECDSA verify. tuples = n × 129 bytes: r(32) ‖ s(32) ‖ pk33(33) ‖ msg(32). */
int kv_verify_ecdsa_batch(kv_ctx *ctx, const uint8_t *tuples, size_t n,
                          uint64_t *bitmap_out);

/* Batched sighash: for each (tx_index, input_index, hash_type, ecdsa) job, compute
 * the 32-byte signing hash of the blob's tx input. */
typedef struct {
  uint32_t tx_index;
  uint32_t input_index;
  uint8_t hash_type;
  uint8_t ecdsa;
  uint16_t _pad;
} kv_sighash_job;
int kv_sighash_batch(kv_ctx *ctx, const uint8_t *blob, size_t blob_len,
                     const kv_sighash_job *jobs, size_t n, uint8_t *hashes_out /*[32n]*/);

/* Finalize a muhash partial (num‖den, 768B LE) into the 32-byte commitment. */
int kv_muhash_finalize(kv_ctx *ctx, const uint8_t *partial768, uint8_t *hash32_out);

/* Fold another partial into an accumulator partial: acc *= other (num and den
 * multiplied mod 2^3072-1103717) — MuHash::combine (crypto/muhash/src/lib.rs:87). */
int kv_muhash_combine(kv_ctx *ctx, uint8_t *acc_partial768, const uint8_t *other_partial768);

/* Batched verifies with per-signature status bytes (0 valid / 1 invalid /
 * 2 pubkey-parse-error / 3 sig-parse-error) — the validate path uses these to
 * distinguish TxScriptError::InvalidPubkey/InvalidSignature from boolean false. */
int kv_verify_schnorr_batch_status(kv_ctx *ctx, const uint8_t *tuples, size_t n,
                                   uint64_t *bitmap_out, uint8_t *status_out);
int kv_verify_ecdsa_batch_status(kv_ctx *ctx, const uint8_t *tuples, size_t n,
                                 uint64_t *bitmap_out, uint8_t *status_out);

/* Staged mode: stage a batch into HBM once, then time repeated launches with
 * inputs resident (bench harness; kernel_ms from hipEvents on the engine stream). */
int kv_stage_tuples(kv_ctx *ctx, const uint8_t *tuples, size_t n, int ecdsa);
int kv_verify_staged(kv_ctx *ctx, size_t n, int ecdsa, double *kernel_ms);
int kv_fetch_bitmap(kv_ctx *ctx, size_t n, uint64_t *bitmap_out);

/* ---- GPU-resident UTXO set ----
 * ⇔ UtxoCollection (consensus/core/src/utxo/utxo_collection.rs:5) + the
 * populate/diff steps (utxo_validation.rs:351-390, utxo_diff.rs:224).
 * Outpoints are 36B (tx_id‖index LE); entries are packed 64B records:
 * amount u64 ‖ daa_score u64 ‖ flags u16 ‖ spk_version u16 ‖
 * spk_len u32 ‖ spk[36] inline when spk_len ≤ 36.
 * Scripts larger than 36B (the reference allows spk up to
 * max_script_public_key_len = 10,000B, utxo_entry.rs:20 / params.rs) live in
 * a device-side ARENA: the entry carries flags bit1
 * (KV_UTXO_F_SPK_ARENA) and its spk field's first 4 bytes are the u32 arena
 * byte offset. Arena space is bump-allocated; removing an entry leaks its
 * arena span until the next kv_utxo_reset (documented trade: non-standard
 * outputs are rare and the arena compacts on reset). */
#define KV_UTXO_F_COINBASE 1u
#define KV_UTXO_F_SPK_ARENA 2u
#define KV_UTXO_MAX_SPK 10000u
int kv_utxo_reset(kv_ctx *ctx, uint64_t capacity);
/* inline-only upsert: every entry's spk_len must be ≤ 36 */
int kv_utxo_upsert(kv_ctx *ctx, const uint8_t *outpoints, const uint8_t *entries64,
                   size_t n);
/* general upsert: entries with spk_len > 36 take their script bytes from
 * spk_blob, concatenated in entry order (total = sum of those spk_lens);
 * the engine stores them in the arena and rewrites the entry in the table
 * with KV_UTXO_F_SPK_ARENA + the arena offset. */
int kv_utxo_upsert_spk(kv_ctx *ctx, const uint8_t *outpoints,
                       const uint8_t *entries64, const uint8_t *spk_blob,
                       size_t spk_blob_len, size_t n);
int kv_utxo_remove(kv_ctx *ctx, const uint8_t *outpoints, size_t n);
int kv_utxo_lookup(kv_ctx *ctx, const uint8_t *outpoints, size_t n,
                   uint8_t *entries_out /*[64n] or NULL*/, uint64_t *found_bitmap,
                   double *kernel_ms /*optional*/);
/* lookup that also returns out-of-line scripts: arena entries' bytes are
 * gathered (device-side) into spk_out in entry order and each such returned
 * entry's spk field is rewritten to the u32 offset INTO spk_out. *spk_used
 * receives the total bytes (call with spk_cap 0 to size). Returns -3 when
 * spk_cap is too small (entries_out/bitmap still valid, *spk_used = needed). */
int kv_utxo_lookup_spk(kv_ctx *ctx, const uint8_t *outpoints, size_t n,
                       uint8_t *entries_out, uint64_t *found_bitmap,
                       uint8_t *spk_out, size_t spk_cap, size_t *spk_used);

/* Populate + validate + (optionally) apply the UTXO diff — the full virtual
 * processor step for one block (utxo_validation.rs:351-390 populate, then
 * validation, then utxo_diff.rs:224 add_transaction per accepted tx).
 * Blob format as kv_validate_block, but each input's UtxoEntry fields are
 * IGNORED (builders write zeros with utxo_spk_len = 0); entries resolve from
 * the GPU-resident table. A missing outpoint pre-fails that tx with
 * KV_ERR_MISSING_OUTPOINT (it is skipped by validation and muhash). When
 * apply_diff != 0, accepted txs' spent outpoints are removed and their created
 * outputs upserted with block_daa_score (coinbase txs never enter this path). */
int kv_validate_block_utxo(kv_ctx *ctx, const uint8_t *blob, size_t blob_len,
                           uint64_t pov_daa_score, uint64_t block_daa_score,
                           uint32_t flags, int apply_diff, int32_t *tx_codes_out,
                           uint64_t *fees_out, uint8_t *muhash_partial_out);

/* Mempool batch validation ⇔ validate_mempool_transaction_in_utxo_context
 * (utxo_validation.rs:418-457) fanned over independent txs: SkipMassCheck +
 * per-tx COMPUTED contextual storage mass + optional feerate threshold
 * (<= 0 disables; fee / normalized_max(mass) <= threshold →
 * KV_ERR_FEERATE_TOO_LOW). from_utxo_table = 1 resolves entries from the
 * GPU-resident table (missing outpoint → KV_ERR_MISSING_OUTPOINT), 0 takes
 * them inline from the blob. */
int kv_validate_mempool(kv_ctx *ctx, const uint8_t *blob, size_t blob_len,
                        uint64_t pov_daa_score, double feerate_threshold,
                        int from_utxo_table, int32_t *tx_codes_out,
                        uint64_t *fees_out);

/* Block-body-in-isolation batch (body_validation_in_isolation.rs):
 * merkle_root_out ← calc_hash_merkle_root over the txs (GPU leaf hashes,
 * caller compares against the header commitment); rule_code_out ← 0 or the
 * first KV_ERR_BODY_* violation (duplicate tx / in-block double spend /
 * chained tx). Either out-pointer may be NULL. */
int kv_block_body_check(kv_ctx *ctx, const uint8_t *blob, size_t blob_len,
                        uint8_t merkle_root_out[32], int32_t *rule_code_out);

/* ---- KIP-21 sequencing-commitment accessor ----
 * ⇔ SeqCommitAccessor consulted by OpChainblockSeqCommit
 * (crypto/txscript/src/opcodes/mod.rs:1389-1405; threaded through
 * tx_validation_in_utxo_context.rs:44,170). The host (Rust over FFI) supplies
 * a callback so DAG reachability stays on the node side. Contract:
 *   return 0  → block is a chain ancestor within depth; write the 32-byte
 *               commitment to commitment_out32
 *   return nonzero → not an ancestor / pruned / too deep (the script fails
 *               with the InvalidSource class, matching the reference's
 *               BlockNotSelected/BlockAlreadyPruned/BlockIsTooDeep rejects)
 * A NULL accessor (the default) disables the opcode: executing it fails with
 * InvalidOpcode, matching a None accessor in the reference.
 * The callback may be invoked concurrently from engine worker threads. */
typedef int (*kv_seq_commit_accessor_fn)(void *user, const uint8_t block_hash32[32],
                                         uint8_t commitment_out32[32]);
int kv_set_seq_commit_accessor(kv_ctx *ctx, kv_seq_commit_accessor_fn fn, void *user);

/* Per-kernel GPU timings (hipEvents on the engine stream) and job counts of
 * the LAST kv_validate_block/kv_validate_block_utxo call on this context —
 * the bench harness derives the block-path kernel roofline from these.
 * A kernel that did not launch reports 0. */
typedef struct {
  double subhash_ms;   /* kv_tx_subhash_kernel */
  double s_assemble_ms; /* sighash+tuple assembly, schnorr jobs */
  double e_assemble_ms; /* sighash+tuple assembly, ecdsa jobs */
  double schnorr_ms;   /* kv_schnorr_verify_kernel */
  double ecdsa_ms;     /* kv_ecdsa_verify_kernel */
  double muhash_ms;    /* element + reduce kernels */
  uint64_t n_schnorr;  /* schnorr verify lane-jobs (cache misses only) */
  uint64_t n_ecdsa;    /* ecdsa verify lane-jobs */
} kv_validate_timings;
int kv_get_validate_timings(kv_ctx *ctx, kv_validate_timings *out);

/* Sig-cache statistics (crypto/txscript/src/caches.rs:57-82 counters). */
typedef struct {
  uint64_t insertions;
  uint64_t hits;
  uint64_t misses;
} kv_cache_stats;
int kv_sig_cache_stats(kv_ctx *ctx, kv_cache_stats *out);

#ifdef __cplusplus
}
#endif

#endif /* KASPA_ENGINE_ABI_H */
