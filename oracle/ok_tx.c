/* ORACLE (test infrastructure only).
 *
 * Transaction-blob parsing (format: include/kaspa_engine_abi.h), transaction id
 * (consensus/core/src/hashing/tx.rs), sighash (consensus/core/src/hashing/
 * sighash.rs:140-292) and per-transaction MuHash serialization
 * (consensus/core/src/muhash.rs:16-69).
 */
#include "ok_tx.h"
#include <stdlib.h>
#include <string.h>

/* ---------------- little-endian readers ---------------- */

static uint16_t rd16(const uint8_t *p) { uint16_t v; memcpy(&v, p, 2); return v; }
static uint32_t rd32(const uint8_t *p) { uint32_t v; memcpy(&v, p, 4); return v; }
static uint64_t rd64(const uint8_t *p) { uint64_t v; memcpy(&v, p, 8); return v; }

int ok_tx_parse(const uint8_t *blob, size_t blob_len, uint32_t tx_index, ok_tx *tx) {
  memset(tx, 0, sizeof(*tx));
  if (blob_len < 4) return -1;
  uint32_t n_txs = rd32(blob);
  if (tx_index >= n_txs) return -1;
  if (blob_len < 4 + 4ull * n_txs) return -1;
  size_t off = rd32(blob + 4 + 4ull * tx_index);
  const uint8_t *end = blob + blob_len;
#define NEED(nbytes)                                                           \
  do {                                                                         \
    if ((size_t)(end - p) < (size_t)(nbytes)) goto fail;                       \
  } while (0)
  const uint8_t *p = blob + off;
  if (off > blob_len) return -1;
  NEED(2 + 2 + 2 + 2 + 8 + 20 + 4 + 8 + 8 + 32);
  tx->version = rd16(p); p += 2;
  tx->n_inputs = rd16(p); p += 2;
  tx->n_outputs = rd16(p); p += 2;
  p += 2; /* pad */
  tx->lock_time = rd64(p); p += 8;
  tx->subnetwork_id = p; p += 20;
  tx->payload_len = rd32(p); p += 4;
  tx->gas = rd64(p); p += 8;
  tx->storage_mass = rd64(p); p += 8;
  tx->tx_id = p; p += 32;
  NEED(tx->payload_len);
  tx->payload = p; p += tx->payload_len;

  tx->inputs = calloc(tx->n_inputs ? tx->n_inputs : 1, sizeof(ok_input));
  tx->outputs = calloc(tx->n_outputs ? tx->n_outputs : 1, sizeof(ok_output));
  if (!tx->inputs || !tx->outputs) goto fail;

  for (uint32_t i = 0; i < tx->n_inputs; i++) {
    ok_input *in = &tx->inputs[i];
    NEED(32 + 4 + 8 + 1 + 1 + 2 + 4);
    in->prev_tx_id = p; p += 32;
    in->prev_index = rd32(p); p += 4;
    in->sequence = rd64(p); p += 8;
    in->commit_kind = *p++;
    p++; /* pad */
    in->commit_value = rd16(p); p += 2;
    in->sig_script_len = rd32(p); p += 4;
    NEED(in->sig_script_len);
    in->sig_script = p; p += in->sig_script_len;
    NEED(8 + 8 + 1 + 1 + 2 + 4);
    in->utxo_amount = rd64(p); p += 8;
    in->utxo_daa_score = rd64(p); p += 8;
    in->utxo_is_coinbase = *p++;
    uint8_t has_cov = *p++;
    in->utxo_spk_version = rd16(p); p += 2;
    in->utxo_spk_len = rd32(p); p += 4;
    NEED(in->utxo_spk_len);
    in->utxo_spk = p; p += in->utxo_spk_len;
    if (has_cov) {
      NEED(32);
      in->utxo_covenant_id = p; p += 32;
    }
  }
  for (uint32_t i = 0; i < tx->n_outputs; i++) {
    ok_output *o = &tx->outputs[i];
    NEED(8 + 2 + 2 + 4);
    o->value = rd64(p); p += 8;
    o->spk_version = rd16(p); p += 2;
    p += 2; /* pad */
    o->spk_len = rd32(p); p += 4;
    NEED(o->spk_len);
    o->spk = p; p += o->spk_len;
    NEED(1);
    o->has_covenant = *p++;
    if (o->has_covenant) {
      NEED(2 + 32);
      o->cov_auth_input = rd16(p); p += 2;
      o->cov_id = p; p += 32;
    }
  }
  return 0;
fail:
  ok_tx_free(tx);
  return -1;
#undef NEED
}

void ok_tx_free(ok_tx *tx) {
  free(tx->inputs);
  free(tx->outputs);
  tx->inputs = NULL;
  tx->outputs = NULL;
}

int ok_tx_is_coinbase(const ok_tx *tx) {
  /* SUBNETWORK_ID_COINBASE = from_byte(1) (consensus/core/src/subnets.rs) */
  if (tx->subnetwork_id[0] != 1) return 0;
  for (int i = 1; i < 20; i++)
    if (tx->subnetwork_id[i]) return 0;
  return 1;
}

static int subnetwork_is_native(const ok_tx *tx) {
  for (int i = 0; i < 20; i++)
    if (tx->subnetwork_id[i]) return 0;
  return 1;
}

/* ---------------- growable serialization buffer ---------------- */

void wb_init(wbuf *w) {
  w->cap = 512;
  w->len = 0;
  w->p = malloc(w->cap);
}
void wb_free(wbuf *w) { free(w->p); w->p = NULL; }
void wb_bytes(wbuf *w, const void *d, size_t n) {
  if (w->len + n > w->cap) {
    while (w->len + n > w->cap) w->cap *= 2;
    w->p = realloc(w->p, w->cap);
  }
  memcpy(w->p + w->len, d, n);
  w->len += n;
}
void wb_u8(wbuf *w, uint8_t v) { wb_bytes(w, &v, 1); }
void wb_u16(wbuf *w, uint16_t v) { wb_bytes(w, &v, 2); }
void wb_u32(wbuf *w, uint32_t v) { wb_bytes(w, &v, 4); }
void wb_u64(wbuf *w, uint64_t v) { wb_bytes(w, &v, 8); }
void wb_varbytes(wbuf *w, const uint8_t *d, size_t n) {
  wb_u64(w, (uint64_t)n);
  if (n) wb_bytes(w, d, n);
}

/* ---------------- transaction id / hash (hashing/tx.rs:51-136) -------------- */

#define TXF_EXCL_SIG 1
#define TXF_EXCL_MASS 2
#define TXF_EXCL_PAYLOAD 4

static void write_transaction(wbuf *w, const ok_tx *tx, int flags) {
  wb_u16(w, tx->version);
  wb_u64(w, tx->n_inputs);
  for (uint32_t i = 0; i < tx->n_inputs; i++) {
    const ok_input *in = &tx->inputs[i];
    wb_bytes(w, in->prev_tx_id, 32);
    wb_u32(w, in->prev_index);
    if (!(flags & TXF_EXCL_SIG)) {
      wb_varbytes(w, in->sig_script, in->sig_script_len);
      if (tx->version < 1) /* version_expects_sig_op_count_field */
        wb_u8(w, in->commit_kind == 0 ? (uint8_t)in->commit_value : 0);
    } else {
      wb_varbytes(w, NULL, 0);
    }
    wb_u64(w, in->sequence);
    if (!(flags & TXF_EXCL_MASS) && tx->version >= 1)
      wb_u16(w, in->commit_kind == 1 ? in->commit_value : 0);
  }
  wb_u64(w, tx->n_outputs);
  for (uint32_t i = 0; i < tx->n_outputs; i++) {
    const ok_output *o = &tx->outputs[i];
    wb_u64(w, o->value);
    wb_u16(w, o->spk_version);
    wb_varbytes(w, o->spk, o->spk_len);
    if (tx->version >= 1) {
      wb_u8(w, o->has_covenant ? 1 : 0);
      if (o->has_covenant) {
        wb_u16(w, o->cov_auth_input);
        wb_bytes(w, o->cov_id, 32);
      }
    }
  }
  wb_u64(w, tx->lock_time);
  wb_bytes(w, tx->subnetwork_id, 20);
  wb_u64(w, tx->gas);
  if (!(flags & TXF_EXCL_PAYLOAD))
    wb_varbytes(w, tx->payload, tx->payload_len);
  else
    wb_varbytes(w, NULL, 0);
  if (!(flags & TXF_EXCL_MASS)) {
    if (tx->version < 1) {
      if (tx->storage_mass > 0) wb_u64(w, tx->storage_mass);
    } else {
      wb_u64(w, tx->storage_mass);
    }
  }
}

static const uint8_t KEY_TXID[] = "TransactionID";
static const uint8_t KEY_TXHASH[] = "TransactionHash";
static const uint8_t KEY_SIGNING[] = "TransactionSigningHash";
static const uint8_t DOMAIN_SIGNING_ECDSA[] = "TransactionSigningHashECDSA";

static void b3_key_pad(const char *s, uint8_t key[32]) {
  memset(key, 0, 32);
  memcpy(key, s, strlen(s));
}

void ok_tx_compute_id(const ok_tx *tx, uint8_t out32[32]) {
  if (tx->version == 0) {
    wbuf w;
    wb_init(&w);
    write_transaction(&w, tx, TXF_EXCL_SIG | TXF_EXCL_MASS);
    ok_blake2b_keyed(KEY_TXID, sizeof(KEY_TXID) - 1, w.p, w.len, out32);
    wb_free(&w);
  } else {
    /* id_v1 (hashing/tx.rs:207-218): TransactionV1Id(payload_digest ‖ rest_digest) */
    uint8_t key[32], payload_digest[32], rest_digest[32];
    b3_key_pad("PayloadDigest", key);
    ok_blake3_keyed(key, tx->payload, tx->payload_len, payload_digest);
    wbuf w;
    wb_init(&w);
    write_transaction(&w, tx, TXF_EXCL_SIG | TXF_EXCL_MASS | TXF_EXCL_PAYLOAD);
    b3_key_pad("TransactionRest", key);
    ok_blake3_keyed(key, w.p, w.len, rest_digest);
    wb_free(&w);
    uint8_t both[64];
    memcpy(both, payload_digest, 32);
    memcpy(both + 32, rest_digest, 32);
    b3_key_pad("TransactionV1Id", key);
    ok_blake3_keyed(key, both, 64, out32);
  }
}

void ok_tx_compute_hash(const ok_tx *tx, uint8_t out32[32]) {
  wbuf w;
  wb_init(&w);
  write_transaction(&w, tx, 0);
  ok_blake2b_keyed(KEY_TXHASH, sizeof(KEY_TXHASH) - 1, w.p, w.len, out32);
  wb_free(&w);
}

/* ---------------- sighash (hashing/sighash.rs) ---------------- */

#define SIGHASH_ALL 0x01
#define SIGHASH_NONE 0x02
#define SIGHASH_SINGLE 0x04
#define SIGHASH_ACP 0x80
#define SIGHASH_MASK 0x07

static void signing_hash(const uint8_t *d, size_t n, uint8_t out[32]) {
  ok_blake2b_keyed(KEY_SIGNING, sizeof(KEY_SIGNING) - 1, d, n, out);
}

void ok_reused_init(ok_sighash_reused *r) { memset(r, 0, sizeof(*r)); }

static const uint8_t *prev_outputs_hash(const ok_tx *tx, uint8_t ht, ok_sighash_reused *r,
                                        uint8_t tmp[32]) {
  static const uint8_t zero[32] = {0};
  if (ht & SIGHASH_ACP) return zero;
  if (!r->have_prevouts) {
    wbuf w;
    wb_init(&w);
    for (uint32_t i = 0; i < tx->n_inputs; i++) {
      wb_bytes(&w, tx->inputs[i].prev_tx_id, 32);
      wb_u32(&w, tx->inputs[i].prev_index);
    }
    signing_hash(w.p, w.len, r->prevouts);
    wb_free(&w);
    r->have_prevouts = 1;
  }
  (void)tmp;
  return r->prevouts;
}

static const uint8_t *sequences_hash(const ok_tx *tx, uint8_t ht, ok_sighash_reused *r) {
  static const uint8_t zero[32] = {0};
  uint8_t m = ht & SIGHASH_MASK;
  if (m == SIGHASH_SINGLE || (ht & SIGHASH_ACP) || m == SIGHASH_NONE) return zero;
  if (!r->have_sequences) {
    wbuf w;
    wb_init(&w);
    for (uint32_t i = 0; i < tx->n_inputs; i++) wb_u64(&w, tx->inputs[i].sequence);
    signing_hash(w.p, w.len, r->sequences);
    wb_free(&w);
    r->have_sequences = 1;
  }
  return r->sequences;
}

static const uint8_t *sig_op_counts_hash(const ok_tx *tx, uint8_t ht, ok_sighash_reused *r) {
  static const uint8_t zero[32] = {0};
  if (ht & SIGHASH_ACP) return zero;
  if (!r->have_sigops) {
    wbuf w;
    wb_init(&w);
    for (uint32_t i = 0; i < tx->n_inputs; i++)
      wb_u8(&w, tx->inputs[i].commit_kind == 0 ? (uint8_t)tx->inputs[i].commit_value : 0);
    signing_hash(w.p, w.len, r->sigops);
    wb_free(&w);
    r->have_sigops = 1;
  }
  return r->sigops;
}

static void hash_output_into(wbuf *w, const ok_output *o, uint16_t version) {
  wb_u64(w, o->value);
  wb_u16(w, o->spk_version);
  wb_varbytes(w, o->spk, o->spk_len);
  if (version >= 1) {
    wb_u8(w, o->has_covenant ? 1 : 0);
    if (o->has_covenant) {
      wb_u16(w, o->cov_auth_input);
      wb_bytes(w, o->cov_id, 32);
    }
  }
}

static void outputs_hash(const ok_tx *tx, uint8_t ht, ok_sighash_reused *r,
                         uint32_t input_index, uint8_t out[32]) {
  static const uint8_t zero[32] = {0};
  uint8_t m = ht & SIGHASH_MASK;
  if (m == SIGHASH_NONE) {
    memcpy(out, zero, 32);
    return;
  }
  if (m == SIGHASH_SINGLE) {
    if (input_index >= tx->n_outputs) {
      memcpy(out, zero, 32);
      return;
    }
    wbuf w;
    wb_init(&w);
    hash_output_into(&w, &tx->outputs[input_index], tx->version);
    signing_hash(w.p, w.len, out);
    wb_free(&w);
    return;
  }
  if (!r->have_outputs) {
    wbuf w;
    wb_init(&w);
    for (uint32_t i = 0; i < tx->n_outputs; i++)
      hash_output_into(&w, &tx->outputs[i], tx->version);
    signing_hash(w.p, w.len, r->outputs);
    wb_free(&w);
    r->have_outputs = 1;
  }
  memcpy(out, r->outputs, 32);
}

static const uint8_t *payload_hash(const ok_tx *tx, ok_sighash_reused *r) {
  static const uint8_t zero[32] = {0};
  if (subnetwork_is_native(tx) && tx->payload_len == 0) return zero;
  if (!r->have_payload) {
    wbuf w;
    wb_init(&w);
    wb_varbytes(&w, tx->payload, tx->payload_len);
    signing_hash(w.p, w.len, r->payload);
    wb_free(&w);
    r->have_payload = 1;
  }
  return r->payload;
}

/* calc_schnorr_signature_hash (sighash.rs:245-280) */
void ok_tx_sighash_schnorr(const ok_tx *tx, uint32_t input_index, uint8_t hash_type,
                           ok_sighash_reused *r, uint8_t out32[32]) {
  const ok_input *in = &tx->inputs[input_index];
  uint8_t tmp[32], oh[32];
  wbuf w;
  wb_init(&w);
  wb_u16(&w, tx->version);
  wb_bytes(&w, prev_outputs_hash(tx, hash_type, r, tmp), 32);
  wb_bytes(&w, sequences_hash(tx, hash_type, r), 32);
  if (tx->version < 1) wb_bytes(&w, sig_op_counts_hash(tx, hash_type, r), 32);
  /* outpoint */
  wb_bytes(&w, in->prev_tx_id, 32);
  wb_u32(&w, in->prev_index);
  /* utxo script public key */
  wb_u16(&w, in->utxo_spk_version);
  wb_varbytes(&w, in->utxo_spk, in->utxo_spk_len);
  wb_u64(&w, in->utxo_amount);
  wb_u64(&w, in->sequence);
  if (tx->version < 1)
    wb_u8(&w, in->commit_kind == 0 ? (uint8_t)in->commit_value : 0);
  outputs_hash(tx, hash_type, r, input_index, oh);
  wb_bytes(&w, oh, 32);
  wb_u64(&w, tx->lock_time);
  wb_bytes(&w, tx->subnetwork_id, 20);
  wb_u64(&w, tx->gas);
  wb_bytes(&w, payload_hash(tx, r), 32);
  wb_u8(&w, hash_type);
  signing_hash(w.p, w.len, out32);
  wb_free(&w);
}

/* calc_ecdsa_signature_hash (sighash.rs:282-292) */
void ok_tx_sighash_ecdsa(const ok_tx *tx, uint32_t input_index, uint8_t hash_type,
                         ok_sighash_reused *r, uint8_t out32[32]) {
  uint8_t inner[32];
  ok_tx_sighash_schnorr(tx, input_index, hash_type, r, inner);
  ok_sha256_domain(DOMAIN_SIGNING_ECDSA, sizeof(DOMAIN_SIGNING_ECDSA) - 1, inner, 32,
                   out32);
}

/* ---------------- muhash of a tx (consensus/core/src/muhash.rs:16-69) -------- */

static void write_utxo(wbuf *w, const uint8_t outpoint_txid[32], uint32_t outpoint_index,
                       uint64_t daa, uint64_t amount, uint8_t is_coinbase,
                       uint16_t spk_version, const uint8_t *spk, uint32_t spk_len,
                       const uint8_t *covenant_id) {
  wb_bytes(w, outpoint_txid, 32);
  wb_u32(w, outpoint_index);
  wb_u64(w, daa);
  wb_u64(w, amount);
  wb_u8(w, is_coinbase ? 1 : 0);
  wb_u16(w, spk_version);
  wb_varbytes(w, spk, spk_len);
  if (covenant_id) wb_bytes(w, covenant_id, 32);
}

void ok_tx_muhash(const ok_tx *tx, uint64_t block_daa_score,
                  uint64_t num[OK_U3072_LIMBS], uint64_t den[OK_U3072_LIMBS]) {
  uint64_t elem[OK_U3072_LIMBS];
  int is_cb = ok_tx_is_coinbase(tx);
  /* spent utxos → denominator (remove_element_builder) */
  for (uint32_t i = 0; i < tx->n_inputs; i++) {
    const ok_input *in = &tx->inputs[i];
    wbuf w;
    wb_init(&w);
    write_utxo(&w, in->prev_tx_id, in->prev_index, in->utxo_daa_score, in->utxo_amount,
               in->utxo_is_coinbase, in->utxo_spk_version, in->utxo_spk, in->utxo_spk_len,
               in->utxo_covenant_id);
    ok_muhash_element(w.p, w.len, elem);
    wb_free(&w);
    ok_u3072_mul(den, elem);
  }
  /* created utxos → numerator */
  for (uint32_t i = 0; i < tx->n_outputs; i++) {
    const ok_output *o = &tx->outputs[i];
    wbuf w;
    wb_init(&w);
    write_utxo(&w, tx->tx_id, i, block_daa_score, o->value, is_cb, o->spk_version, o->spk,
               o->spk_len, o->has_covenant ? o->cov_id : NULL);
    ok_muhash_element(w.p, w.len, elem);
    wb_free(&w);
    ok_u3072_mul(num, elem);
  }
}

/* ---------------- public blob-level wrappers ---------------- */

int ok_sighash(const uint8_t *blob, size_t blob_len, uint32_t tx_index,
               uint32_t input_index, uint8_t hash_type, int ecdsa, uint8_t out32[32]) {
  ok_tx tx;
  if (ok_tx_parse(blob, blob_len, tx_index, &tx)) return -1;
  if (input_index >= tx.n_inputs) {
    ok_tx_free(&tx);
    return -1;
  }
  ok_sighash_reused r;
  ok_reused_init(&r);
  if (ecdsa)
    ok_tx_sighash_ecdsa(&tx, input_index, hash_type, &r, out32);
  else
    ok_tx_sighash_schnorr(&tx, input_index, hash_type, &r, out32);
  ok_tx_free(&tx);
  return 0;
}

int ok_tx_id(const uint8_t *blob, size_t blob_len, uint32_t tx_index, uint8_t out32[32]) {
  ok_tx tx;
  if (ok_tx_parse(blob, blob_len, tx_index, &tx)) return -1;
  ok_tx_compute_id(&tx, out32);
  ok_tx_free(&tx);
  return 0;
}

int ok_muhash_add_tx(const uint8_t *blob, size_t blob_len, uint32_t tx_index,
                     uint64_t block_daa_score, uint64_t num[OK_U3072_LIMBS],
                     uint64_t den[OK_U3072_LIMBS]) {
  ok_tx tx;
  if (ok_tx_parse(blob, blob_len, tx_index, &tx)) return -1;
  ok_tx_muhash(&tx, block_daa_score, num, den);
  ok_tx_free(&tx);
  return 0;
}

int ok_tx_hash_blob(const uint8_t *blob, size_t blob_len, uint32_t tx_index,
                    uint8_t out32[32]) {
  ok_tx tx;
  if (ok_tx_parse(blob, blob_len, tx_index, &tx)) return -1;
  ok_tx_compute_hash(&tx, out32);
  ok_tx_free(&tx);
  return 0;
}
