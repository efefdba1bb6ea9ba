/* Merkle root + block-body-in-isolation checks (oracle restatement).
 *
 * calc_merkle_root ⇔ crypto/merkle/src/lib.rs:13-52: pad the leaf count to the
 * next power of two; absent left child propagates None upward; absent right
 * child hashes against ZERO_HASH; node = keyed-blake2b-256("MerkleBranchHash",
 * left32 ‖ right32) (crypto/hashes/src/hashers.rs:28). A 1-leaf tree is the
 * leaf; 0 leaves → ZERO_HASH.
 *
 * Body checks ⇔ consensus/src/pipeline/body_processor/
 * body_validation_in_isolation.rs: check_duplicate_transactions (:152),
 * check_block_double_spends (:126), check_no_chained_transactions (:136).
 */
#include "oracle.h"
#include "ok_tx.h"
#include <stdlib.h>
#include <string.h>

static const uint8_t MERKLE_KEY[] = "MerkleBranchHash";

void ok_merkle_root(const uint8_t *hashes /* 32*n */, size_t n, uint8_t out32[32]) {
  if (n == 0) {
    memset(out32, 0, 32);
    return;
  }
  if (n == 1) {
    memcpy(out32, hashes, 32);
    return;
  }
  size_t pot = 1;
  while (pot < n) pot <<= 1;
  /* level buffer: present-flag + hash per slot */
  uint8_t *cur = (uint8_t *)malloc(pot * 32);
  uint8_t *pres = (uint8_t *)malloc(pot);
  memcpy(cur, hashes, n * 32);
  for (size_t i = 0; i < pot; i++) pres[i] = i < n;
  size_t width = pot;
  uint8_t zero[32] = {0};
  while (width > 1) {
    for (size_t i = 0; i < width; i += 2) {
      size_t o = i / 2;
      if (!pres[i]) {
        pres[o] = 0;
        continue;
      }
      uint8_t buf[64];
      memcpy(buf, cur + i * 32, 32);
      memcpy(buf + 32, pres[i + 1] ? cur + (i + 1) * 32 : zero, 32);
      ok_blake2b_keyed(MERKLE_KEY, sizeof(MERKLE_KEY) - 1, buf, 64, cur + o * 32);
      pres[o] = 1;
    }
    width /= 2;
  }
  memcpy(out32, cur, 32);
  free(cur);
  free(pres);
}

/* hash-merkle-root over a blob's tx hashes (calc_hash_merkle_root,
 * consensus/core/src/merkle.rs:5) */
int ok_blob_merkle_root(const uint8_t *blob, size_t blob_len, uint8_t out32[32]) {
  if (blob_len < 4) return -1;
  uint32_t n;
  memcpy(&n, blob, 4);
  uint8_t *hashes = (uint8_t *)malloc((size_t)n * 32);
  for (uint32_t t = 0; t < n; t++)
    if (ok_tx_hash_blob(blob, blob_len, t, hashes + (size_t)t * 32) != 0) {
      free(hashes);
      return -1;
    }
  ok_merkle_root(hashes, n, out32);
  free(hashes);
  return 0;
}

struct op36 {
  uint8_t b[36];
};

static int cmp36(const void *a, const void *b) { return memcmp(a, b, 36); }
static int cmp32(const void *a, const void *b) { return memcmp(a, b, 32); }

/* returns 0 ok, or the first-failing rule code:
 * duplicate tx → OK_BODY_DUP_TX, double spend → OK_BODY_DOUBLE_SPEND,
 * chained tx → OK_BODY_CHAINED — the check order of
 * validate_body_in_isolation (:19-27). */
int ok_body_check(const uint8_t *blob, size_t blob_len) {
  if (blob_len < 4) return -1;
  uint32_t n_txs;
  memcpy(&n_txs, blob, 4);
  int rc = 0;
  size_t n_in_total = 0, n_out_total = 0;
  uint8_t *ids = (uint8_t *)malloc((size_t)n_txs * 32);
  uint8_t *ids_sorted;
  for (uint32_t t = 0; t < n_txs; t++) {
    ok_tx tx;
    if (ok_tx_parse(blob, blob_len, t, &tx) != 0) {
      free(ids);
      return -1;
    }
    n_in_total += tx.n_inputs;
    n_out_total += tx.n_outputs;
    ok_tx_free(&tx);
    if (ok_tx_id(blob, blob_len, t, ids + (size_t)t * 32) != 0) {
      free(ids);
      return -1;
    }
  }
  /* duplicate transactions (by id) */
  ids_sorted = (uint8_t *)malloc((size_t)n_txs * 32);
  memcpy(ids_sorted, ids, (size_t)n_txs * 32);
  qsort(ids_sorted, n_txs, 32, cmp32);
  for (uint32_t t = 0; t + 1 < n_txs; t++)
    if (memcmp(ids_sorted + (size_t)t * 32, ids_sorted + (size_t)(t + 1) * 32,
               32) == 0) {
      rc = OK_BODY_DUP_TX;
      break;
    }
  free(ids_sorted);
  /* double spends: duplicate previous outpoints */
  if (!rc && n_in_total > 1) {
    struct op36 *ops = (struct op36 *)malloc(n_in_total * sizeof(struct op36));
    size_t k = 0;
    for (uint32_t t = 0; t < n_txs; t++) {
      ok_tx tx;
      ok_tx_parse(blob, blob_len, t, &tx);
      for (uint32_t i = 0; i < tx.n_inputs; i++, k++) {
        memcpy(ops[k].b, tx.inputs[i].prev_tx_id, 32);
        memcpy(ops[k].b + 32, &tx.inputs[i].prev_index, 4);
      }
      ok_tx_free(&tx);
    }
    qsort(ops, n_in_total, sizeof(struct op36), cmp36);
    for (size_t i = 0; i + 1 < n_in_total; i++)
      if (memcmp(ops[i].b, ops[i + 1].b, 36) == 0) {
        rc = OK_BODY_DOUBLE_SPEND;
        break;
      }
    free(ops);
  }
  /* chained: an input spending an outpoint created in this block */
  if (!rc && n_in_total && n_out_total) {
    struct op36 *created =
        (struct op36 *)malloc(n_out_total * sizeof(struct op36));
    size_t k = 0;
    for (uint32_t t = 0; t < n_txs; t++) {
      ok_tx tx;
      ok_tx_parse(blob, blob_len, t, &tx);
      for (uint32_t i = 0; i < tx.n_outputs; i++, k++) {
        memcpy(created[k].b, ids + (size_t)t * 32, 32);
        memcpy(created[k].b + 32, &i, 4);
      }
      ok_tx_free(&tx);
    }
    qsort(created, n_out_total, sizeof(struct op36), cmp36);
    for (uint32_t t = 0; t < n_txs && !rc; t++) {
      ok_tx tx;
      ok_tx_parse(blob, blob_len, t, &tx);
      for (uint32_t i = 0; i < tx.n_inputs; i++) {
        struct op36 key;
        memcpy(key.b, tx.inputs[i].prev_tx_id, 32);
        memcpy(key.b + 32, &tx.inputs[i].prev_index, 4);
        if (bsearch(&key, created, n_out_total, sizeof(struct op36), cmp36)) {
          rc = OK_BODY_CHAINED;
          break;
        }
      }
      ok_tx_free(&tx);
    }
    free(created);
  }
  free(ids);
  return rc;
}
