/* ORACLE internal header: parsed tx views over the ABI blob + serialization buffer. */
#ifndef OK_TX_H
#define OK_TX_H

#include "oracle.h"

typedef struct {
  const uint8_t *prev_tx_id; /* 32 */
  uint32_t prev_index;
  uint64_t sequence;
  uint8_t commit_kind; /* 0 sigop_count, 1 compute_budget */
  uint16_t commit_value;
  const uint8_t *sig_script;
  uint32_t sig_script_len;
  uint64_t utxo_amount;
  uint64_t utxo_daa_score;
  uint8_t utxo_is_coinbase;
  uint16_t utxo_spk_version;
  const uint8_t *utxo_spk;
  uint32_t utxo_spk_len;
  const uint8_t *utxo_covenant_id; /* 32 or NULL */
} ok_input;

typedef struct {
  uint64_t value;
  uint16_t spk_version;
  const uint8_t *spk;
  uint32_t spk_len;
  int has_covenant;
  uint16_t cov_auth_input;
  const uint8_t *cov_id; /* 32 or NULL */
} ok_output;

typedef struct {
  uint16_t version;
  uint32_t n_inputs;
  uint32_t n_outputs;
  uint64_t lock_time;
  const uint8_t *subnetwork_id; /* 20 */
  uint64_t gas;
  const uint8_t *payload;
  uint32_t payload_len;
  uint64_t storage_mass;
  const uint8_t *tx_id; /* 32, carried in blob */
  ok_input *inputs;
  ok_output *outputs;
} ok_tx;

int ok_tx_parse(const uint8_t *blob, size_t blob_len, uint32_t tx_index, ok_tx *tx);
void ok_tx_free(ok_tx *tx);
int ok_tx_is_coinbase(const ok_tx *tx);
void ok_tx_compute_id(const ok_tx *tx, uint8_t out32[32]);
void ok_tx_compute_hash(const ok_tx *tx, uint8_t out32[32]);

/* SigHashReusedValues (sighash.rs:14-41) */
typedef struct {
  uint8_t prevouts[32], sequences[32], sigops[32], outputs[32], payload[32];
  int have_prevouts, have_sequences, have_sigops, have_outputs, have_payload;
} ok_sighash_reused;

void ok_reused_init(ok_sighash_reused *r);
void ok_tx_sighash_schnorr(const ok_tx *tx, uint32_t input_index, uint8_t hash_type,
                           ok_sighash_reused *r, uint8_t out32[32]);
void ok_tx_sighash_ecdsa(const ok_tx *tx, uint32_t input_index, uint8_t hash_type,
                         ok_sighash_reused *r, uint8_t out32[32]);
void ok_tx_muhash(const ok_tx *tx, uint64_t block_daa_score,
                  uint64_t num[OK_U3072_LIMBS], uint64_t den[OK_U3072_LIMBS]);

/* growable byte buffer */
typedef struct {
  uint8_t *p;
  size_t len, cap;
} wbuf;
void wb_init(wbuf *w);
void wb_free(wbuf *w);
void wb_bytes(wbuf *w, const void *d, size_t n);
void wb_u8(wbuf *w, uint8_t v);
void wb_u16(wbuf *w, uint16_t v);
void wb_u32(wbuf *w, uint32_t v);
void wb_u64(wbuf *w, uint64_t v);
void wb_varbytes(wbuf *w, const uint8_t *d, size_t n);

/* script engine (ok_script.c): returns 0 or KV_SCRIPT_* code */
int ok_script_check_input(const ok_tx *tx, uint32_t input_index, uint64_t mass_per_sig_op,
                          ok_sighash_reused *reused);

#endif
