/* ORACLE — TEST INFRASTRUCTURE ONLY.
 *
 * CPU restatement of rusty-kaspa's per-block transaction-validation hot path
 * (kaspanet/rusty-kaspa v2.0.1, /root/reference), pinned by the reference's own
 * golden vectors (see tests/golden/). Each function cites the reference file:line
 * it follows. This library is the parity checker and the bench.py `cpu_baseline`
 * ("port" kind) — it is NEVER the product path; the product path is the HIP
 * engine in rusty_kaspa_amd/ and must fail loudly when its extension is missing.
 *
 * Only tests/, __graft_entry__.smoke() and bench.py's cpu_baseline leg may
 * import, call, link or execute anything in oracle/.
 */
#ifndef OK_ORACLE_H
#define OK_ORACLE_H

#include <stddef.h>
#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

/* ---------------- hashing (crypto/hashes/src/hashers.rs) ---------------- */

typedef struct {
  uint64_t h[8];
  uint64_t t[2];
  uint8_t buf[128];
  size_t buflen;
  size_t outlen;
} ok_blake2b_state;

void ok_blake2b_init(ok_blake2b_state *S, const uint8_t *key, size_t keylen, size_t outlen);
void ok_blake2b_update(ok_blake2b_state *S, const uint8_t *data, size_t len);
void ok_blake2b_final(ok_blake2b_state *S, uint8_t *out);
void ok_blake2b_keyed(const uint8_t *key, size_t keylen, const uint8_t *data, size_t len,
                      uint8_t out32[32]);

typedef struct {
  uint32_t h[8];
  uint64_t nbytes;
  uint8_t buf[64];
  size_t buflen;
} ok_sha256_state;

void ok_sha256_init(ok_sha256_state *S);
void ok_sha256_update(ok_sha256_state *S, const uint8_t *data, size_t len);
void ok_sha256_final(ok_sha256_state *S, uint8_t out32[32]);
void ok_sha256(const uint8_t *data, size_t len, uint8_t out32[32]);
void ok_sha256_domain(const uint8_t *domain, size_t domain_len, const uint8_t *data,
                      size_t len, uint8_t out32[32]);

void ok_blake3(const uint8_t *data, size_t len, uint8_t out32[32]);
void ok_blake3_keyed(const uint8_t key[32], const uint8_t *data, size_t len, uint8_t out32[32]);

void ok_chacha20_block384(const uint8_t seed[32], uint8_t out[384]);

/* ---------------- U3072 / MuHash (crypto/muhash) ---------------- */

#define OK_U3072_LIMBS 48

/* a *= b mod (2^3072 - 1103717); restates crypto/muhash/src/u3072.rs:90-155 */
void ok_u3072_mul(uint64_t a[OK_U3072_LIMBS], const uint64_t b[OK_U3072_LIMBS]);
/* a /= b (multiply by modular inverse); restates u3072.rs:156-191 */
void ok_u3072_div(uint64_t a[OK_U3072_LIMBS], const uint64_t b[OK_U3072_LIMBS]);
void ok_u3072_one(uint64_t a[OK_U3072_LIMBS]);
int ok_u3072_is_overflow(const uint64_t a[OK_U3072_LIMBS]);
void ok_u3072_full_reduce(uint64_t a[OK_U3072_LIMBS]);

/* data → U3072 element: blake2b-keyed("MuHashElement") → ChaCha20 384B → LE limbs
 * (crypto/muhash/src/lib.rs:162-169) */
void ok_muhash_element(const uint8_t *data, size_t len, uint64_t out[OK_U3072_LIMBS]);
/* finalize: num/den, serialize 384B LE, blake2b-keyed("MuHashFinalize")
 * (crypto/muhash/src/lib.rs:93-107); num is normalized in place */
void ok_muhash_finalize(uint64_t num[OK_U3072_LIMBS], uint64_t den[OK_U3072_LIMBS],
                        uint8_t out32[32]);

/* ---------------- secp256k1 (from scratch; parity pinned by mainnet-signature
 * vectors in tests/golden/mainnet_txs.json — the reference's secp256k1-sys crate
 * (vendored libsecp256k1 0.10.1) is NOT under /root/reference; call sites:
 * crypto/txscript/src/lib.rs:869,899) ---------------- */

/* BIP-340 verify. Returns 1 valid, 0 invalid, -1 pubkey parse error (x not on curve
 * or >= p — reference maps this to TxScriptError::InvalidPubkey). */
int ok_schnorr_verify(const uint8_t pk32[32], const uint8_t msg32[32], const uint8_t sig64[64]);

/* ECDSA verify, compact 64B sig, 33B compressed pubkey. Returns 1 valid, 0 invalid
 * (incl. high-S per libsecp256k1 secp256k1_ecdsa_verify), -1 pubkey parse error,
 * -2 signature parse error (r or s >= n, per parse_compact overflow). */
int ok_ecdsa_verify(const uint8_t pk33[33], const uint8_t msg32[32], const uint8_t sig64[64]);

/* BIP-340 sign (for synthetic workload generation only). aux32 may be NULL (zeros). */
int ok_schnorr_sign(const uint8_t seckey32[32], const uint8_t msg32[32],
                    const uint8_t *aux32, uint8_t sig_out[64]);
/* ECDSA sign (RFC6979-free deterministic nonce via tagged hash — NOT a consensus
 * surface; only used to build synthetic valid signatures). */
int ok_ecdsa_sign(const uint8_t seckey32[32], const uint8_t msg32[32], uint8_t sig_out[64]);

/* derive pubkeys; returns 0 on bad seckey */
int ok_pubkey_xonly(const uint8_t seckey32[32], uint8_t xonly_out[32]);
int ok_pubkey_compressed(const uint8_t seckey32[32], uint8_t pk33_out[33]);

/* ---------------- tx validation over the shared blob format
 * (format spec: include/kaspa_engine_abi.h; semantics:
 * consensus/src/processes/transaction_validator/tx_validation_in_utxo_context.rs:37-218)
 * ---------------- */

/* per-tx result codes: see include/kaspa_engine_abi.h (KV_OK / KV_ERR_*) */
int ok_validate_block(const uint8_t *blob, size_t blob_len, uint64_t pov_daa_score,
                      uint64_t block_daa_score, uint32_t flags, int32_t *tx_codes_out,
                      uint64_t *fees_out, uint8_t muhash_out[32]);

/* parallel (OpenMP) variant over `threads` threads — the CPU baseline driver,
 * structured like check_scripts_par_iter + validate_transactions_with_muhash_in_parallel */
int ok_validate_block_parallel(const uint8_t *blob, size_t blob_len, uint64_t pov_daa_score,
                               uint64_t block_daa_score, uint32_t flags, int threads,
                               int32_t *tx_codes_out, uint64_t *fees_out, uint8_t muhash_out[32]);

/* sighash of one input (consensus/core/src/hashing/sighash.rs:245-292) */
int ok_sighash(const uint8_t *blob, size_t blob_len, uint32_t tx_index, uint32_t input_index,
               uint8_t hash_type, int ecdsa, uint8_t out32[32]);

/* transaction hash (hashing/tx.rs:20-24) */
int ok_tx_hash_blob(const uint8_t *blob, size_t blob_len, uint32_t tx_index, uint8_t out32[32]);

/* transaction id (consensus/core/src/hashing/tx.rs:34-48,207-218) */
int ok_tx_id(const uint8_t *blob, size_t blob_len, uint32_t tx_index, uint8_t out32[32]);

/* synthetic workload generation (harness only; deterministic by seed) */
int ok_gen_schnorr_tuples(uint64_t seed, size_t n, uint32_t invalid_permille, uint8_t *out, int threads);
int ok_gen_ecdsa_tuples(uint64_t seed, size_t n, uint32_t invalid_permille, uint8_t *out, int threads);

/* CPU-baseline batched schnorr verify over n × 128B (r‖s‖pk‖msg) tuples */
int ok_verify_schnorr_batch(const uint8_t *tuples, size_t n, int threads,
                            uint64_t *bitmap_out);

/* run the script engine for one input; returns 0 ok, else KV script error code */
int ok_check_input_script(const uint8_t *blob, size_t blob_len, uint32_t tx_index,
                          uint32_t input_index);

/* muhash contribution of one valid tx (consensus/core/src/muhash.rs:16-69):
 * removes spent utxos into den, adds created into num */
int ok_muhash_add_tx(const uint8_t *blob, size_t blob_len, uint32_t tx_index,
                     uint64_t block_daa_score, uint64_t num[OK_U3072_LIMBS],
                     uint64_t den[OK_U3072_LIMBS]);

#ifdef __cplusplus
}
#endif

#endif /* OK_ORACLE_H */

/* ---------------- Merkle root + body-in-isolation (oracle) ---------------- */
#define OK_BODY_DUP_TX 10
#define OK_BODY_DOUBLE_SPEND 11
#define OK_BODY_CHAINED 12
/* calc_merkle_root (crypto/merkle/src/lib.rs:13-52) over 32B leaves */
void ok_merkle_root(const uint8_t *hashes, size_t n, uint8_t out32[32]);
/* calc_hash_merkle_root over a blob's txs (consensus/core/src/merkle.rs:5) */
int ok_blob_merkle_root(const uint8_t *blob, size_t blob_len, uint8_t out32[32]);
/* duplicate-tx / in-block double-spend / chained-tx checks
 * (body_validation_in_isolation.rs:126-173); 0 ok or OK_BODY_* */
int ok_body_check(const uint8_t *blob, size_t blob_len);

/* mempool batch validation ⇔ validate_mempool_transaction_in_utxo_context
 * (utxo_validation.rs:418-457): SkipMassCheck + computed contextual mass +
 * optional feerate threshold (<=0 disables; fee/normalized_mass <= threshold
 * → KV_ERR_FEERATE_TOO_LOW). */
int ok_validate_mempool(const uint8_t *blob, size_t blob_len, uint64_t pov_daa_score,
                        double feerate_threshold, int threads,
                        int32_t *tx_codes_out, uint64_t *fees_out);

/* test-harness hook: configure the KIP-21 seq-commitment accessor mock
 * (OpChainblockSeqCommit). NULL block disables (opcode -> InvalidOpcode,
 * matching a None accessor). */
void ok_script_set_seq_commit_mock(const uint8_t block32[32],
                                   const uint8_t commit32[32]);

/* ---- direct mass-vector exports (tests/golden/mass.json) ----
 * utxo_plurality (mass/mod.rs:83-105) */
uint64_t ok_test_plurality(uint32_t spk_len, int has_cov);
/* calc_storage_mass (mass/mod.rs:439-514) over bare (amount, spk_len, has_cov)
 * cells; 0 ok (mass in *mass_out), -1 incomputable, -2 too many cells */
int ok_test_storage_mass(uint32_t n_ins, const uint64_t *in_amounts,
                         const uint32_t *in_spk_lens, const uint8_t *in_has_cov,
                         uint32_t n_outs, const uint64_t *out_amounts,
                         const uint32_t *out_spk_lens, const uint8_t *out_has_cov,
                         uint64_t *mass_out);
/* Mass::normalized_max + MassCofactors::new (mass/mod.rs:258-265,298-308) */
uint64_t ok_normalized_max_limits(uint64_t storage_mass, uint64_t compute_mass,
                                  uint64_t transient_mass, uint64_t limit_storage,
                                  uint64_t limit_compute, uint64_t limit_transient);

/* pubkey parse-only validity (XOnlyPublicKey/PublicKey::from_slice succeed) */
int ok_xonly_pubkey_valid(const uint8_t pk32[32]);
int ok_compressed_pubkey_valid(const uint8_t pk33[33]);
