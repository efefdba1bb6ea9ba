/* ORACLE (test infrastructure only).
 *
 * txscript interpreter restating crypto/txscript/src/{lib.rs, opcodes/mod.rs,
 * data_stack.rs, runtime_resource_meter.rs}. Scope: every opcode except the
 * introspection (0xb2-0xc9, 0xcb-0xd6), ZK-precompile (0xa6) and seq-commit
 * families, which return KV_SCRIPT_UNSUPPORTED_OPCODE when EXECUTED — those are
 * out of the round-1 hot-path scope (covenants/zk/introspection),
 * and the HIP engine routes identically, so oracle↔engine parity holds.
 */
#include "kaspa_engine_abi.h"
#include "ok_tx.h"
#include <stdlib.h>
#include <string.h>

#define MAX_STACK_SIZE 244
#define MAX_SCRIPTS_SIZE 1000000
#define MAX_SCRIPT_ELEMENT_SIZE 1000000
#define MAX_OPS_PER_SCRIPT 1000000
#define MAX_PUB_KEYS 20
#define STANDARD_SPK_MAX_SIZE 35
#define NO_COST_OPCODE 0x60
#define SCRIPT_UNITS_PER_GRAM 100ULL
#define SCRIPT_UNITS_PER_SIGOP_COUNT_UNIT (1000ULL * 100ULL)
#define SCRIPT_UNITS_PER_COMPUTE_BUDGET_UNIT (100ULL * 100ULL)
#define FREE_SCRIPT_UNITS_PER_INPUT (SCRIPT_UNITS_PER_COMPUTE_BUDGET_UNIT - 1)
#define SEQUENCE_LOCK_TIME_DISABLED (1ULL << 63)
#define SEQUENCE_LOCK_TIME_MASK 0x00000000ffffffffULL
#define LOCK_TIME_THRESHOLD 500000000000ULL
#define MAX_TX_IN_SEQUENCE_NUM UINT64_MAX

/* stack entry */
typedef struct {
  uint8_t *d;
  uint32_t len;
} sent;

#define STACK_CAP 300
typedef struct {
  sent it[STACK_CAP];
  int n;
  uint64_t pushed_bytes;
} stk;

typedef struct {
  stk d, a;
  const ok_tx *tx;
  const ok_input *input;
  uint32_t idx;
  int is_p2sh;
  ok_sighash_reused *reused;
  uint8_t cond[4096];
  int cond_n;
  int32_t num_ops;
  uint64_t sigop_units;
  uint64_t limit_units;
  uint64_t remaining_units;
} eng;

enum { COND_FALSE = 0, COND_TRUE = 1, COND_SKIP = 2 };

static void stk_init(stk *s) { s->n = 0; s->pushed_bytes = 0; }

static void stk_clear(stk *s) {
  for (int i = 0; i < s->n; i++) free(s->it[i].d);
  s->n = 0;
}

static int stk_push_copy(stk *s, const uint8_t *d, uint32_t len, int metered) {
  if (len > MAX_SCRIPT_ELEMENT_SIZE) return KV_SCRIPT_ELEMENT_TOO_BIG;
  if (s->n >= STACK_CAP) return KV_SCRIPT_STACK_SIZE_EXCEEDED;
  uint8_t *cp = malloc(len ? len : 1);
  if (len) memcpy(cp, d, len);
  s->it[s->n].d = cp;
  s->it[s->n].len = len;
  s->n++;
  if (metered) s->pushed_bytes += len;
  return 0;
}

/* push taking ownership */
static int stk_push_own(stk *s, uint8_t *d, uint32_t len, int metered) {
  if (len > MAX_SCRIPT_ELEMENT_SIZE) { free(d); return KV_SCRIPT_ELEMENT_TOO_BIG; }
  if (s->n >= STACK_CAP) { free(d); return KV_SCRIPT_STACK_SIZE_EXCEEDED; }
  s->it[s->n].d = d;
  s->it[s->n].len = len;
  s->n++;
  if (metered) s->pushed_bytes += len;
  return 0;
}

/* ---------------- numbers (data_stack.rs:127-247) ---------------- */

static int num_deser(const sent *e, int64_t *out) { /* deserialize_i64 */
  if (e->len > 8) return KV_SCRIPT_NUMBER_TOO_BIG; /* SizedEncodeInt<8> path */
  if (e->len == 0) { *out = 0; return 0; }
  uint8_t msb = e->d[e->len - 1];
  int64_t sign = 1 - 2 * (int64_t)(msb >> 7);
  int64_t acc = msb & 0x7f;
  for (int i = (int)e->len - 2; i >= 0; i--) acc = (acc << 8) + e->d[i];
  *out = acc * sign;
  return 0;
}

static uint32_t num_ser(int64_t v, uint8_t out[9]) { /* serialize_i64, size=None */
  int neg = v < 0;
  uint64_t pos = neg ? (uint64_t)(-(v + 1)) + 1 : (uint64_t)v;
  uint32_t n = 0;
  int last_sat = 0;
  while (pos) {
    uint8_t b = pos & 0xff;
    last_sat = (b & 0x80) != 0;
    out[n++] = b;
    pos >>= 8;
  }
  if (n && last_sat) out[n++] = 0;
  if (neg && n) out[n - 1] |= 0x80;
  return n;
}

static int stk_push_num(stk *s, int64_t v, int metered) {
  /* SizedEncodeInt<8> serialization (data_stack.rs:127-190): i64::MIN needs a
   * 9th byte and fails — the reference surfaces it as Serialization error
   * (mapped like NumberTooBig to UNKNOWN_ERROR in its vector harness) */
  if (v == INT64_MIN) return KV_SCRIPT_NUMBER_TOO_BIG;
  uint8_t buf[9];
  uint32_t n = num_ser(v, buf);
  return stk_push_copy(s, buf, n, metered);
}

static int ent_bool(const sent *e) { /* OpcodeData<bool> */
  if (e->len == 0) return 0;
  if (e->d[e->len - 1] & 0x7f) return 1;
  for (uint32_t i = 0; i + 1 < e->len; i++)
    if (e->d[i]) return 1;
  return 0;
}

static int stk_pop_num(stk *s, int64_t *out) { /* pop_items::<1,i64> */
  if (s->n < 1) return KV_SCRIPT_INVALID_STACK_OPERATION;
  sent *e = &s->it[s->n - 1];
  int rc = num_deser(e, out);
  if (!rc) { free(e->d); s->n--; }
  return rc;
}

static int stk_pop_i32(stk *s, int32_t *out) {
  int64_t v;
  int rc = stk_pop_num(s, &v);
  if (rc) return rc;
  if (v > INT32_MAX || v < INT32_MIN) return KV_SCRIPT_NUMBER_TOO_BIG;
  *out = (int32_t)v;
  return 0;
}

static int stk_pop_bool(stk *s, int *out) {
  if (s->n < 1) return KV_SCRIPT_INVALID_STACK_OPERATION;
  sent *e = &s->it[--s->n];
  *out = ent_bool(e);
  free(e->d);
  return 0;
}

/* pop raw (ownership transferred to caller) */
static int stk_pop_raw(stk *s, int count, sent out[]) {
  if (s->n < count) return KV_SCRIPT_INVALID_STACK_OPERATION;
  for (int i = 0; i < count; i++) out[i] = s->it[s->n - count + i];
  s->n -= count;
  return 0;
}


/* ---- KIP-21 seq-commitment accessor (OpChainblockSeqCommit, mod.rs:1389).
 * Production queries DAG reachability; the oracle exposes the same shape the
 * reference's vector harness mocks (lib.rs:2482-2512): one known chain block
 * mapping to one commitment. Unset (all-zero) => opcode disabled =>
 * InvalidOpcode, matching a None accessor. */
static uint8_t g_seqc_block[32];
static uint8_t g_seqc_commit[32];
static int g_seqc_set = 0;

void ok_script_set_seq_commit_mock(const uint8_t block32[32],
                                   const uint8_t commit32[32]) {
  if (!block32) {
    g_seqc_set = 0;
    return;
  }
  memcpy(g_seqc_block, block32, 32);
  memcpy(g_seqc_commit, commit32, 32);
  g_seqc_set = 1;
}

/* ---- introspection helpers (opcodes/mod.rs:196-207 substring/i32_to_usize,
 * tx.rs ScriptPublicKey::to_bytes = version u16 LE + script) ---- */
static int intro_input(eng *E, const ok_input **out) {
  int32_t idx;
  int rc = stk_pop_i32(&E->d, &idx);
  if (rc) return rc;
  if (idx < 0) return KV_SCRIPT_INVALID_INDEX;
  if ((uint32_t)idx >= E->tx->n_inputs) return KV_SCRIPT_INVALID_INPUT_INDEX;
  *out = &E->tx->inputs[idx];
  return 0;
}

static int intro_output(eng *E, const ok_output **out) {
  int32_t idx;
  int rc = stk_pop_i32(&E->d, &idx);
  if (rc) return rc;
  if (idx < 0) return KV_SCRIPT_INVALID_INDEX;
  if ((uint32_t)idx >= E->tx->n_outputs) return KV_SCRIPT_INVALID_INPUT_INDEX;
  *out = &E->tx->outputs[idx];
  return 0;
}

static int intro_substr(eng *E, const uint8_t *data, uint32_t len,
                        int32_t start, int32_t end) {
  if (start < 0 || end < 0) return KV_SCRIPT_INVALID_INDEX;
  if (end < start) return KV_SCRIPT_INVALID_RANGE;
  if ((uint32_t)(end - start) > 1000000u) return KV_SCRIPT_ELEMENT_TOO_BIG;
  if ((uint32_t)end > len) return KV_SCRIPT_INVALID_SOURCE; /* OutOfBounds */
  return stk_push_copy(&E->d, data + start, (uint32_t)(end - start), 1);
}

static int intro_push_spk(eng *E, uint16_t version, const uint8_t *script,
                          uint32_t len) {
  uint8_t *buf = (uint8_t *)malloc(2 + (size_t)len);
  buf[0] = (uint8_t)(version & 0xff);
  buf[1] = (uint8_t)(version >> 8);
  memcpy(buf + 2, script, len);
  int rc = stk_push_copy(&E->d, buf, 2 + len, 1);
  free(buf);
  return rc;
}

static int intro_spk_substr(eng *E, uint16_t version, const uint8_t *script,
                            uint32_t len, int32_t start, int32_t end) {
  uint8_t *buf = (uint8_t *)malloc(2 + (size_t)len);
  buf[0] = (uint8_t)(version & 0xff);
  buf[1] = (uint8_t)(version >> 8);
  memcpy(buf + 2, script, len);
  int rc = intro_substr(E, buf, 2 + len, start, end);
  free(buf);
  return rc;
}

static void free_ents(sent *e, int n) {
  for (int i = 0; i < n; i++) free(e[i].d);
}

/* ---------------- meter (runtime_resource_meter.rs) ---------------- */

static int consume_units(eng *E, uint64_t units) {
  if (units > E->remaining_units) return KV_SCRIPT_EXCEEDED_SCRIPT_UNITS;
  E->remaining_units -= units;
  return 0;
}

static int consume_sigop(eng *E) {
  return consume_units(E, E->sigop_units);
}

static int charge_pushed(eng *E) {
  uint64_t pb = E->d.pushed_bytes + E->a.pushed_bytes;
  E->d.pushed_bytes = 0;
  E->a.pushed_bytes = 0;
  return consume_units(E, pb);
}

/* ---------------- sighash-type / signature checks ---------------- */

static int sighashtype_valid(uint8_t t) {
  return t == 0x01 || t == 0x02 || t == 0x04 || t == 0x81 || t == 0x82 || t == 0x84;
}

/* returns: 0/1 validity via *valid, or positive KV error */
static int check_schnorr(eng *E, uint8_t hash_type, const sent *key, const uint8_t *sig,
                         uint32_t sig_len, int from_stack, const uint8_t *stack_msg,
                         int *valid) {
  int rc = consume_sigop(E);
  if (rc) return rc;
  /* XOnlyPublicKey::from_slice parses FULLY (length AND curve) before the
   * signature is looked at (lib.rs:857-859) */
  if (key->len != 32) return KV_SCRIPT_INVALID_PUBKEY;
  if (!ok_xonly_pubkey_valid(key->d)) return KV_SCRIPT_INVALID_PUBKEY;
  if (sig_len != 64) return KV_SCRIPT_INVALID_SIGNATURE;
  uint8_t msg[32];
  if (from_stack) {
    memcpy(msg, stack_msg, 32);
  } else {
    ok_tx_sighash_schnorr(E->tx, E->idx, hash_type, E->reused, msg);
  }
  int r = ok_schnorr_verify(key->d, msg, sig);
  if (r < 0) return KV_SCRIPT_INVALID_PUBKEY; /* x not on curve / >= p */
  *valid = r;
  return 0;
}

static int check_ecdsa(eng *E, uint8_t hash_type, const sent *key, const uint8_t *sig,
                       uint32_t sig_len, int from_stack, const uint8_t *stack_msg,
                       int *valid) {
  int rc = consume_sigop(E);
  if (rc) return rc;
  /* check_pub_key_encoding_ecdsa (length), then PublicKey::from_slice
   * (prefix + curve) — both before the signature (lib.rs:888-890) */
  if (key->len != 33) return KV_SCRIPT_PUBKEY_FORMAT;
  if (!ok_compressed_pubkey_valid(key->d)) return KV_SCRIPT_INVALID_PUBKEY;
  if (sig_len != 64) return KV_SCRIPT_INVALID_SIGNATURE;
  uint8_t msg[32];
  if (from_stack) {
    memcpy(msg, stack_msg, 32);
  } else {
    ok_tx_sighash_ecdsa(E->tx, E->idx, hash_type, E->reused, msg);
  }
  int r = ok_ecdsa_verify(key->d, msg, sig);
  if (r == -1) return KV_SCRIPT_INVALID_PUBKEY;
  if (r == -2) return KV_SCRIPT_INVALID_SIGNATURE;
  *valid = r;
  return 0;
}

/* op_check_multisig_schnorr_or_ecdsa (lib.rs:759-843) */
static int op_multisig(eng *E, int ecdsa) {
  int rc;
  int32_t num_keys;
  if ((rc = stk_pop_i32(&E->d, &num_keys))) return rc;
  if (num_keys < 0 || num_keys > MAX_PUB_KEYS) return KV_SCRIPT_INVALID_PUBKEY_COUNT;
  E->num_ops += num_keys;
  if (E->num_ops > MAX_OPS_PER_SCRIPT) return KV_SCRIPT_TOO_MANY_OPERATIONS;
  sent keys[MAX_PUB_KEYS];
  if ((rc = stk_pop_raw(&E->d, num_keys, keys))) return rc;
  int32_t num_sigs;
  if ((rc = stk_pop_i32(&E->d, &num_sigs))) {
    free_ents(keys, num_keys);
    return rc;
  }
  if (num_sigs < 0 || num_sigs > num_keys) {
    free_ents(keys, num_keys);
    return KV_SCRIPT_INVALID_SIGNATURE_COUNT;
  }
  sent sigs[MAX_PUB_KEYS];
  if ((rc = stk_pop_raw(&E->d, num_sigs, sigs))) {
    free_ents(keys, num_keys);
    return rc;
  }

  int failed = 0;
  int key_pos = 0; /* consumed keys */
  for (int si = 0; si < num_sigs; si++) {
    if (sigs[si].len == 0) { failed = 1; break; }
    uint8_t typ = sigs[si].d[sigs[si].len - 1];
    uint32_t slen = sigs[si].len - 1;
    if (!sighashtype_valid(typ)) {
      free_ents(keys, num_keys);
      free_ents(sigs, num_sigs);
      return KV_SCRIPT_INVALID_SIG_HASH_TYPE;
    }
    int matched = 0;
    while (1) {
      if (num_keys - key_pos < num_sigs - si) { failed = 1; break; }
      const sent *key = &keys[key_pos++];
      int valid = 0;
      rc = ecdsa ? check_ecdsa(E, typ, key, sigs[si].d, slen, 0, NULL, &valid)
                 : check_schnorr(E, typ, key, sigs[si].d, slen, 0, NULL, &valid);
      if (rc) {
        free_ents(keys, num_keys);
        free_ents(sigs, num_sigs);
        return rc;
      }
      if (valid) { matched = 1; break; }
    }
    if (!matched) break; /* failed set inside */
  }

  int any_nonempty = 0;
  for (int si = 0; si < num_sigs; si++)
    if (sigs[si].len) any_nonempty = 1;
  free_ents(keys, num_keys);
  free_ents(sigs, num_sigs);
  if (failed && any_nonempty) return KV_SCRIPT_NULL_FAIL;
  /* push bool !failed */
  if (failed) return stk_push_copy(&E->d, NULL, 0, 1);
  uint8_t one = 1;
  return stk_push_copy(&E->d, &one, 1, 1);
}

static int op_checksig(eng *E, int ecdsa) {
  sent se[2];
  int rc = stk_pop_raw(&E->d, 2, se); /* [sig, key], key on top */
  if (rc) return rc;
  sent sig = se[0], key = se[1];
  if (sig.len == 0) {
    free(sig.d);
    free(key.d);
    return stk_push_copy(&E->d, NULL, 0, 1); /* false */
  }
  uint8_t typ = sig.d[sig.len - 1];
  if (!sighashtype_valid(typ)) {
    free(sig.d);
    free(key.d);
    return KV_SCRIPT_INVALID_SIG_HASH_TYPE;
  }
  int valid = 0;
  rc = ecdsa ? check_ecdsa(E, typ, &key, sig.d, sig.len - 1, 0, NULL, &valid)
             : check_schnorr(E, typ, &key, sig.d, sig.len - 1, 0, NULL, &valid);
  free(sig.d);
  free(key.d);
  if (rc) return rc;
  if (valid) {
    uint8_t one = 1;
    return stk_push_copy(&E->d, &one, 1, 1);
  }
  return stk_push_copy(&E->d, NULL, 0, 1);
}

static int op_checksig_from_stack(eng *E, int ecdsa) {
  sent se[3];
  int rc = stk_pop_raw(&E->d, 3, se); /* [signature, msg_hash, pubkey] */
  if (rc) return rc;
  sent sig = se[0], msg = se[1], key = se[2];
  if (msg.len != 32) {
    free_ents(se, 3);
    return KV_SCRIPT_INVALID_STATE;
  }
  int valid = 0;
  rc = ecdsa ? check_ecdsa(E, 0, &key, sig.d, sig.len, 1, msg.d, &valid)
             : check_schnorr(E, 0, &key, sig.d, sig.len, 1, msg.d, &valid);
  free_ents(se, 3);
  if (rc) return rc;
  if (valid) {
    uint8_t one = 1;
    return stk_push_copy(&E->d, &one, 1, 1);
  }
  return stk_push_copy(&E->d, NULL, 0, 1);
}

/* ---------------- helpers for stack manipulation ops ---------------- */

static int dup_items(eng *E, int size) {
  stk *s = &E->d;
  if (s->n < size) return KV_SCRIPT_INVALID_STACK_OPERATION;
  /* copy the top `size` entries (pre-push snapshot) */
  int base = s->n - size;
  for (int i = 0; i < size; i++) {
    int rc = stk_push_copy(s, s->it[base + i].d, s->it[base + i].len, 1);
    if (rc) return rc;
  }
  return 0;
}

static int over_items(eng *E, int size) {
  stk *s = &E->d;
  if (s->n < 2 * size) return KV_SCRIPT_INVALID_STACK_OPERATION;
  int base = s->n - 2 * size;
  for (int i = 0; i < size; i++) {
    int rc = stk_push_copy(s, s->it[base + i].d, s->it[base + i].len, 1);
    if (rc) return rc;
  }
  return 0;
}

static void rotate_left(stk *s, int start, int count, int by) {
  /* rotate s->it[start..start+count] left by `by` */
  sent tmp[64];
  for (int i = 0; i < by; i++) tmp[i] = s->it[start + i];
  memmove(&s->it[start], &s->it[start + by], (count - by) * sizeof(sent));
  for (int i = 0; i < by; i++) s->it[start + count - by + i] = tmp[i];
}

static int rot_items(eng *E, int size) {
  stk *s = &E->d;
  if (s->n < 3 * size) return KV_SCRIPT_INVALID_STACK_OPERATION;
  rotate_left(s, s->n - 3 * size, 3 * size, size);
  return 0;
}

static int swap_items(eng *E, int size) {
  stk *s = &E->d;
  if (s->n < 2 * size) return KV_SCRIPT_INVALID_STACK_OPERATION;
  rotate_left(s, s->n - 2 * size, 2 * size, size);
  return 0;
}

static int drop_items(eng *E, int size) {
  stk *s = &E->d;
  if (s->n < size) return KV_SCRIPT_INVALID_STACK_OPERATION;
  for (int i = 0; i < size; i++) free(s->it[--s->n].d);
  return 0;
}

/* binary numeric op */
#define POP2(a, b)                                                             \
  int64_t a, b;                                                                \
  do {                                                                         \
    if (E->d.n < 2) return KV_SCRIPT_INVALID_STACK_OPERATION;                  \
    int rc_;                                                                   \
    if ((rc_ = stk_pop_num(&E->d, &b))) return rc_;                            \
    if ((rc_ = stk_pop_num(&E->d, &a))) return rc_;                            \
  } while (0)

/* NOTE on POP2 order: reference pops as array [a, b] = split_off(len-2), so a is the
 * DEEPER element. We pop top (b) then next (a). But deserialization errors: the
 * reference deserializes in array order (a first) — order of error reporting can
 * differ only when both are malformed with different errors; both map to errors
 * anyway. Accepted deviation (same accept/reject). */

static int is_executing(const eng *E) {
  return E->cond_n == 0 || E->cond[E->cond_n - 1] == COND_TRUE;
}

/* ---------------- single opcode execution ----------------
 * `data`/`dlen` = push payload for opcodes ≤ 0x4e. Returns KV code. */
static int exec_opcode(eng *E, uint8_t op, const uint8_t *data, uint32_t dlen) {
  int rc;
  switch (op) {
    case 0x00:
      return stk_push_copy(&E->d, NULL, 0, 0); /* unmetered literal */
    case 0x4f:
      return stk_push_num(&E->d, -1, 0);
    case 0x50: /* OpReserved — push class, errors when executed */
      return KV_SCRIPT_OPCODE_RESERVED;
    case 0x61: /* nop */
      return 0;
    case 0x62:
      return KV_SCRIPT_OPCODE_RESERVED;
    case 0x63: { /* if */
      int cond = COND_SKIP;
      if (is_executing(E)) {
        int b;
        if ((rc = stk_pop_bool(&E->d, &b))) return rc;
        cond = b ? COND_TRUE : COND_FALSE;
      }
      if (E->cond_n >= 4096) return KV_SCRIPT_INVALID_STATE;
      E->cond[E->cond_n++] = (uint8_t)cond;
      return 0;
    }
    case 0x64: { /* notif */
      int cond = COND_SKIP;
      if (is_executing(E)) {
        int b;
        if ((rc = stk_pop_bool(&E->d, &b))) return rc;
        cond = b ? COND_FALSE : COND_TRUE;
      }
      if (E->cond_n >= 4096) return KV_SCRIPT_INVALID_STATE;
      E->cond[E->cond_n++] = (uint8_t)cond;
      return 0;
    }
    case 0x67: /* else */
      if (E->cond_n == 0) return KV_SCRIPT_INVALID_STATE;
      {
        uint8_t *c = &E->cond[E->cond_n - 1];
        *c = (*c == COND_TRUE) ? COND_FALSE : (*c == COND_FALSE ? COND_TRUE : COND_SKIP);
      }
      return 0;
    case 0x68: /* endif */
      if (E->cond_n == 0) return KV_SCRIPT_INVALID_STATE;
      E->cond_n--;
      return 0;
    case 0x69: { /* verify */
      int b;
      if ((rc = stk_pop_bool(&E->d, &b))) return rc;
      return b ? 0 : KV_SCRIPT_VERIFY_ERROR;
    }
    case 0x6a:
      return KV_SCRIPT_EARLY_RETURN;
    case 0x6b: { /* toaltstack */
      sent e[1];
      if ((rc = stk_pop_raw(&E->d, 1, e))) return rc;
      return stk_push_own(&E->a, e[0].d, e[0].len, 0);
    }
    case 0x6c: { /* fromaltstack */
      if (E->a.n < 1) return KV_SCRIPT_EMPTY_STACK;
      sent e = E->a.it[--E->a.n];
      return stk_push_own(&E->d, e.d, e.len, 0);
    }
    case 0x6d:
      return drop_items(E, 2);
    case 0x6e:
      return dup_items(E, 2);
    case 0x6f:
      return dup_items(E, 3);
    case 0x70:
      return over_items(E, 2);
    case 0x71:
      return rot_items(E, 2);
    case 0x72:
      return swap_items(E, 2);
    case 0x73: { /* ifdup */
      if (E->d.n < 1) return KV_SCRIPT_INVALID_STACK_OPERATION;
      sent *t = &E->d.it[E->d.n - 1];
      if (ent_bool(t)) return stk_push_copy(&E->d, t->d, t->len, 1);
      return 0;
    }
    case 0x74:
      return stk_push_num(&E->d, E->d.n, 1);
    case 0x75:
      return drop_items(E, 1);
    case 0x76:
      return dup_items(E, 1);
    case 0x77: /* nip */
      if (E->d.n < 2) return KV_SCRIPT_INVALID_STACK_OPERATION;
      free(E->d.it[E->d.n - 2].d);
      E->d.it[E->d.n - 2] = E->d.it[E->d.n - 1];
      E->d.n--;
      return 0;
    case 0x78:
      return over_items(E, 1);
    case 0x79: { /* pick */
      int32_t loc;
      if ((rc = stk_pop_i32(&E->d, &loc))) return rc;
      if (loc < 0 || loc >= E->d.n) return KV_SCRIPT_INVALID_STATE;
      sent *e = &E->d.it[E->d.n - loc - 1];
      return stk_push_copy(&E->d, e->d, e->len, 1);
    }
    case 0x7a: { /* roll */
      int32_t loc;
      if ((rc = stk_pop_i32(&E->d, &loc))) return rc;
      if (loc < 0 || loc >= E->d.n) return KV_SCRIPT_INVALID_STATE;
      if (loc == 0) return 0;
      rotate_left(&E->d, E->d.n - loc - 1, loc + 1, 1);
      return 0;
    }
    case 0x7b:
      return rot_items(E, 1);
    case 0x7c:
      return swap_items(E, 1);
    case 0x7d: { /* tuck: insert copy of top before top-1 */
      if (E->d.n < 2) return KV_SCRIPT_INVALID_STACK_OPERATION;
      sent *top = &E->d.it[E->d.n - 1];
      if (E->d.n >= STACK_CAP) return KV_SCRIPT_STACK_SIZE_EXCEEDED;
      uint8_t *cp = malloc(top->len ? top->len : 1);
      memcpy(cp, top->d, top->len);
      sent ins = {cp, top->len};
      /* insert at n-2 */
      memmove(&E->d.it[E->d.n - 1], &E->d.it[E->d.n - 2], 2 * sizeof(sent));
      E->d.it[E->d.n - 2] = ins;
      E->d.n++;
      E->d.pushed_bytes += ins.len;
      return 0;
    }
    case 0x7e: { /* cat */
      sent e[2];
      if (E->d.n < 2) return KV_SCRIPT_EMPTY_STACK; /* pop()? twice */
      if ((rc = stk_pop_raw(&E->d, 2, e))) return rc;
      uint32_t nlen = e[0].len + e[1].len;
      uint8_t *cp = malloc(nlen ? nlen : 1);
      memcpy(cp, e[0].d, e[0].len);
      memcpy(cp + e[0].len, e[1].d, e[1].len);
      free_ents(e, 2);
      return stk_push_own(&E->d, cp, nlen, 1);
    }
    case 0x7f: { /* substr */
      int64_t start64, end64;
      if (E->d.n < 2) return KV_SCRIPT_INVALID_STACK_OPERATION;
      if ((rc = stk_pop_num(&E->d, &end64))) return rc;
      if ((rc = stk_pop_num(&E->d, &start64))) return rc;
      if (start64 > INT32_MAX || start64 < INT32_MIN || end64 > INT32_MAX ||
          end64 < INT32_MIN)
        return KV_SCRIPT_NUMBER_TOO_BIG;
      sent e[1];
      if (E->d.n < 1) return KV_SCRIPT_EMPTY_STACK; /* pop()? */
      if ((rc = stk_pop_raw(&E->d, 1, e))) return rc;
      if (start64 < 0 || end64 < 0) { free(e[0].d); return KV_SCRIPT_INVALID_INDEX; }
      uint64_t start = (uint64_t)start64, end = (uint64_t)end64;
      if (end < start) { free(e[0].d); return KV_SCRIPT_INVALID_RANGE; }
      if (end - start > MAX_SCRIPT_ELEMENT_SIZE) {
        free(e[0].d);
        return KV_SCRIPT_ELEMENT_TOO_BIG;
      }
      if (end > e[0].len) { free(e[0].d); return KV_SCRIPT_INVALID_RANGE; }
      uint32_t nlen = (uint32_t)(end - start);
      uint8_t *cp = malloc(nlen ? nlen : 1);
      memcpy(cp, e[0].d + start, nlen);
      free(e[0].d);
      return stk_push_own(&E->d, cp, nlen, 1);
    }
    case 0x82: { /* size */
      if (E->d.n < 1) return KV_SCRIPT_INVALID_STACK_OPERATION;
      return stk_push_num(&E->d, E->d.it[E->d.n - 1].len, 1);
    }
    case 0x83: { /* invert */
      if (E->d.n < 1) return KV_SCRIPT_EMPTY_STACK;
      sent *t = &E->d.it[E->d.n - 1];
      for (uint32_t i = 0; i < t->len; i++) t->d[i] = ~t->d[i];
      /* pop+push metered in reference (push(r)) */
      E->d.pushed_bytes += t->len;
      return 0;
    }
    case 0x84:
    case 0x85:
    case 0x86: { /* and/or/xor */
      sent e[2];
      if (E->d.n < 1) return KV_SCRIPT_EMPTY_STACK;
      if (E->d.n < 2) {
        /* reference pops b then a separately; one element → EmptyStack on 2nd pop */
        return KV_SCRIPT_EMPTY_STACK;
      }
      if ((rc = stk_pop_raw(&E->d, 2, e))) return rc;
      if (e[0].len != e[1].len) { free_ents(e, 2); return KV_SCRIPT_INVALID_STATE; }
      for (uint32_t i = 0; i < e[0].len; i++) {
        if (op == 0x84) e[0].d[i] &= e[1].d[i];
        else if (op == 0x85) e[0].d[i] |= e[1].d[i];
        else e[0].d[i] ^= e[1].d[i];
      }
      free(e[1].d);
      return stk_push_own(&E->d, e[0].d, e[0].len, 1);
    }
    case 0x87: { /* equal */
      sent e[2];
      if (E->d.n < 2) return KV_SCRIPT_INVALID_STACK_OPERATION;
      if ((rc = stk_pop_raw(&E->d, 2, e))) return rc;
      int eq = e[0].len == e[1].len && !memcmp(e[0].d, e[1].d, e[0].len);
      free_ents(e, 2);
      if (eq) {
        uint8_t one = 1;
        return stk_push_copy(&E->d, &one, 1, 1);
      }
      return stk_push_copy(&E->d, NULL, 0, 1);
    }
    case 0x88: { /* equalverify */
      sent e[2];
      if (E->d.n < 2) return KV_SCRIPT_INVALID_STACK_OPERATION;
      if ((rc = stk_pop_raw(&E->d, 2, e))) return rc;
      int eq = e[0].len == e[1].len && !memcmp(e[0].d, e[1].d, e[0].len);
      free_ents(e, 2);
      return eq ? 0 : KV_SCRIPT_VERIFY_ERROR;
    }
    case 0x89:
    case 0x8a:
      return KV_SCRIPT_OPCODE_RESERVED;
    case 0x8b: { /* 1add */
      int64_t v;
      if ((rc = stk_pop_num(&E->d, &v))) return rc;
      if (v == INT64_MAX) return KV_SCRIPT_NUMBER_TOO_BIG;
      return stk_push_num(&E->d, v + 1, 1);
    }
    case 0x8c: {
      int64_t v;
      if ((rc = stk_pop_num(&E->d, &v))) return rc;
      if (v == INT64_MIN) return KV_SCRIPT_NUMBER_TOO_BIG;
      return stk_push_num(&E->d, v - 1, 1);
    }
    case 0x8f: {
      int64_t v;
      if ((rc = stk_pop_num(&E->d, &v))) return rc;
      if (v == INT64_MIN) return KV_SCRIPT_NUMBER_TOO_BIG;
      return stk_push_num(&E->d, -v, 1);
    }
    case 0x90: {
      int64_t v;
      if ((rc = stk_pop_num(&E->d, &v))) return rc;
      if (v == INT64_MIN) return KV_SCRIPT_NUMBER_TOO_BIG;
      return stk_push_num(&E->d, v < 0 ? -v : v, 1);
    }
    case 0x91: {
      int64_t v;
      if ((rc = stk_pop_num(&E->d, &v))) return rc;
      return stk_push_num(&E->d, v == 0, 1);
    }
    case 0x92: {
      int64_t v;
      if ((rc = stk_pop_num(&E->d, &v))) return rc;
      return stk_push_num(&E->d, v != 0, 1);
    }
    case 0x93: {
      POP2(a, b);
      if ((b > 0 && a > INT64_MAX - b) || (b < 0 && a < INT64_MIN - b))
        return KV_SCRIPT_NUMBER_TOO_BIG;
      return stk_push_num(&E->d, a + b, 1);
    }
    case 0x94: {
      POP2(a, b);
      if ((b < 0 && a > INT64_MAX + b) || (b > 0 && a < INT64_MIN + b))
        return KV_SCRIPT_NUMBER_TOO_BIG;
      return stk_push_num(&E->d, a - b, 1);
    }
    case 0x95: {
      POP2(a, b);
      if (a != 0 && b != 0) {
        int64_t r = (int64_t)((uint64_t)a * (uint64_t)b);
        if (r / b != a || (a == INT64_MIN && b == -1)) return KV_SCRIPT_NUMBER_TOO_BIG;
        return stk_push_num(&E->d, r, 1);
      }
      return stk_push_num(&E->d, 0, 1);
    }
    case 0x96: {
      POP2(a, b);
      if (b == 0 || (a == INT64_MIN && b == -1)) return KV_SCRIPT_INVALID_STATE;
      return stk_push_num(&E->d, a / b, 1);
    }
    case 0x97: {
      POP2(a, b);
      if (b == 0 || (a == INT64_MIN && b == -1)) return KV_SCRIPT_INVALID_STATE;
      return stk_push_num(&E->d, a % b, 1);
    }
    case 0x9a: {
      POP2(a, b);
      return stk_push_num(&E->d, a != 0 && b != 0, 1);
    }
    case 0x9b: {
      POP2(a, b);
      return stk_push_num(&E->d, a != 0 || b != 0, 1);
    }
    case 0x9c: {
      POP2(a, b);
      return stk_push_num(&E->d, a == b, 1);
    }
    case 0x9d: {
      POP2(a, b);
      return a == b ? 0 : KV_SCRIPT_VERIFY_ERROR;
    }
    case 0x9e: {
      POP2(a, b);
      return stk_push_num(&E->d, a != b, 1);
    }
    case 0x9f: {
      POP2(a, b);
      return stk_push_num(&E->d, a < b, 1);
    }
    case 0xa0: {
      POP2(a, b);
      return stk_push_num(&E->d, a > b, 1);
    }
    case 0xa1: {
      POP2(a, b);
      return stk_push_num(&E->d, a <= b, 1);
    }
    case 0xa2: {
      POP2(a, b);
      return stk_push_num(&E->d, a >= b, 1);
    }
    case 0xa3: {
      POP2(a, b);
      return stk_push_num(&E->d, a < b ? a : b, 1);
    }
    case 0xa4: {
      POP2(a, b);
      return stk_push_num(&E->d, a > b ? a : b, 1);
    }
    case 0xa5: { /* within: pop [x, l, u] */
      int64_t x, l, u;
      if (E->d.n < 3) return KV_SCRIPT_INVALID_STACK_OPERATION;
      if ((rc = stk_pop_num(&E->d, &u))) return rc;
      if ((rc = stk_pop_num(&E->d, &l))) return rc;
      if ((rc = stk_pop_num(&E->d, &x))) return rc;
      return stk_push_num(&E->d, x >= l && x < u, 1);
    }
    case 0xa7: { /* blake2b with key */
      sent e[2];
      if ((rc = stk_pop_raw(&E->d, 2, e))) return rc; /* [data, key] */
      if (e[1].len > 64) { free_ents(e, 2); return KV_SCRIPT_ELEMENT_TOO_BIG; }
      if ((rc = consume_units(E, (uint64_t)e[0].len * 2))) {
        free_ents(e, 2);
        return rc;
      }
      uint8_t h[32];
      ok_blake2b_keyed(e[1].d, e[1].len, e[0].d, e[0].len, h);
      free_ents(e, 2);
      return stk_push_copy(&E->d, h, 32, 1);
    }
    case 0xa8: { /* sha256 */
      sent e[1];
      if ((rc = stk_pop_raw(&E->d, 1, e))) return rc;
      if ((rc = consume_units(E, e[0].len))) { free(e[0].d); return rc; }
      uint8_t h[32];
      ok_sha256(e[0].d, e[0].len, h);
      free(e[0].d);
      return stk_push_copy(&E->d, h, 32, 1);
    }
    case 0xa9:
      return op_multisig(E, 1);
    case 0xaa: { /* blake2b */
      sent e[1];
      if ((rc = stk_pop_raw(&E->d, 1, e))) return rc;
      if ((rc = consume_units(E, (uint64_t)e[0].len * 2))) { free(e[0].d); return rc; }
      uint8_t h[32];
      ok_blake2b_keyed(NULL, 0, e[0].d, e[0].len, h);
      free(e[0].d);
      return stk_push_copy(&E->d, h, 32, 1);
    }
    case 0xab:
      return op_checksig(E, 1);
    case 0xac:
      return op_checksig(E, 0);
    case 0xad: { /* checksigverify */
      if ((rc = op_checksig(E, 0))) return rc;
      int b;
      if ((rc = stk_pop_bool(&E->d, &b))) return rc;
      return b ? 0 : KV_SCRIPT_VERIFY_ERROR;
    }
    case 0xae:
      return op_multisig(E, 0);
    case 0xaf: {
      if ((rc = op_multisig(E, 0))) return rc;
      int b;
      if ((rc = stk_pop_bool(&E->d, &b))) return rc;
      return b ? 0 : KV_SCRIPT_VERIFY_ERROR;
    }
    case 0xb0: { /* checklocktimeverify */
      sent e[1];
      if ((rc = stk_pop_raw(&E->d, 1, e))) return rc;
      if (e[0].len > 8) { free(e[0].d); return KV_SCRIPT_NUMBER_TOO_BIG; }
      uint64_t stack_lt = 0;
      for (uint32_t i = 0; i < e[0].len; i++) stack_lt |= (uint64_t)e[0].d[i] << (8 * i);
      free(e[0].d);
      uint64_t tx_lt = E->tx->lock_time;
      int tx_daa = tx_lt < LOCK_TIME_THRESHOLD, st_daa = stack_lt < LOCK_TIME_THRESHOLD;
      if (tx_daa != st_daa) return KV_SCRIPT_UNSATISFIED_LOCKTIME;
      if (stack_lt > tx_lt) return KV_SCRIPT_UNSATISFIED_LOCKTIME;
      if (E->input->sequence == MAX_TX_IN_SEQUENCE_NUM)
        return KV_SCRIPT_UNSATISFIED_LOCKTIME;
      return 0;
    }
    case 0xb1: { /* checksequenceverify */
      sent e[1];
      if ((rc = stk_pop_raw(&E->d, 1, e))) return rc;
      if (e[0].len > 8) { free(e[0].d); return KV_SCRIPT_NUMBER_TOO_BIG; }
      uint64_t st = 0;
      for (uint32_t i = 0; i < e[0].len; i++) st |= (uint64_t)e[0].d[i] << (8 * i);
      free(e[0].d);
      if (st & SEQUENCE_LOCK_TIME_DISABLED) return 0;
      if (E->input->sequence & SEQUENCE_LOCK_TIME_DISABLED)
        return KV_SCRIPT_UNSATISFIED_LOCKTIME;
      if ((st & SEQUENCE_LOCK_TIME_MASK) > (E->input->sequence & SEQUENCE_LOCK_TIME_MASK))
        return KV_SCRIPT_UNSATISFIED_LOCKTIME;
      return 0;
    }
    case 0xd7:
      return op_checksig_from_stack(E, 0);
    case 0xd8:
      return op_checksig_from_stack(E, 1);
    case 0xd9: { /* blake3 */
      sent e[1];
      if ((rc = stk_pop_raw(&E->d, 1, e))) return rc;
      if ((rc = consume_units(E, e[0].len))) { free(e[0].d); return rc; }
      uint8_t h[32];
      ok_blake3(e[0].d, e[0].len, h);
      free(e[0].d);
      return stk_push_copy(&E->d, h, 32, 1);
    }
    case 0xda: { /* blake3 with key (32B exact) */
      sent e[2];
      if ((rc = stk_pop_raw(&E->d, 2, e))) return rc; /* [data, key] */
      if (e[1].len != 32) { free_ents(e, 2); return KV_SCRIPT_MALFORMED_PUSH; }
      if ((rc = consume_units(E, e[0].len))) { free_ents(e, 2); return rc; }
      uint8_t h[32];
      ok_blake3_keyed(e[1].d, e[0].d, e[0].len, h);
      free_ents(e, 2);
      return stk_push_copy(&E->d, h, 32, 1);
    }
    case 0xcd: { /* OpNum2Bin (opcodes/mod.rs:1297-1308) */
      int32_t size;
      if ((rc = stk_pop_i32(&E->d, &size))) return rc;
      if (size < 0) return KV_SCRIPT_INVALID_INDEX; /* i32_to_usize */
      if (size > 8) return KV_SCRIPT_NOT_MINIMAL_DATA;
      int64_t v;
      if ((rc = stk_pop_num(&E->d, &v))) return rc;
      /* serialize_i64(v, Some(size)) (data_stack.rs:127-162): minimal
       * magnitude bytes (with saturation byte), error if longer than size,
       * zero-pad to size, then OR the sign bit into the LAST byte */
      uint8_t buf[9];
      uint32_t n = 0;
      uint64_t mag = v < 0 ? (uint64_t)(-(v + 1)) + 1 : (uint64_t)v;
      int sat = 0;
      while (mag) {
        buf[n] = (uint8_t)(mag & 0xff);
        sat = (buf[n] & 0x80) != 0;
        mag >>= 8;
        n++;
      }
      if (sat) buf[n++] = 0;
      if (n > (uint32_t)size) return KV_SCRIPT_NUMBER_TOO_BIG; /* Serialization */
      while (n < (uint32_t)size) buf[n++] = 0;
      if (v < 0) buf[n - 1] |= 0x80;
      return stk_push_copy(&E->d, buf, n, 1);
    }
    case 0xce: { /* OpBin2Num: pop as i64 (<=8B), re-push minimally encoded */
      int64_t v;
      if ((rc = stk_pop_num(&E->d, &v))) return rc;
      return stk_push_num(&E->d, v, 1);
    }
    /* ---- transaction introspection (KIP-10 family, opcodes/mod.rs:969-1445).
     * The oracle always executes in TxInput source context. Index pops follow
     * i32_to_usize (negative -> InvalidIndex); substring() maps end<start ->
     * InvalidRange, out-of-bounds -> OutOfBoundsSubstring (both surface as
     * UNKNOWN_ERROR in the reference's vector harness). */
    case 0xb2: /* OpTxVersion */
      return stk_push_num(&E->d, (int64_t)E->tx->version, 1);
    case 0xb3: /* OpTxInputCount */
      return stk_push_num(&E->d, (int64_t)E->tx->n_inputs, 1);
    case 0xb4: /* OpTxOutputCount */
      return stk_push_num(&E->d, (int64_t)E->tx->n_outputs, 1);
    case 0xb5: /* OpTxLockTime */
      return stk_push_num(&E->d, (int64_t)E->tx->lock_time, 1);
    case 0xb6: /* OpTxSubnetId */
      return stk_push_copy(&E->d, E->tx->subnetwork_id, 20, 1);
    case 0xb7: /* OpTxGas */
      return stk_push_num(&E->d, (int64_t)E->tx->gas, 1);
    case 0xb8: { /* OpTxPayloadSubstr */
      int32_t se[2];
      if ((rc = stk_pop_i32(&E->d, &se[1]))) return rc; /* end on top */
      if ((rc = stk_pop_i32(&E->d, &se[0]))) return rc;
      return intro_substr(E, E->tx->payload, E->tx->payload_len, se[0], se[1]);
    }
    case 0xb9: /* OpTxInputIndex */
      return stk_push_num(&E->d, (int64_t)E->idx, 1);
    case 0xba: { /* OpOutpointTxId */
      const ok_input *in2;
      if ((rc = intro_input(E, &in2))) return rc;
      return stk_push_copy(&E->d, in2->prev_tx_id, 32, 1);
    }
    case 0xbb: { /* OpOutpointIndex */
      const ok_input *in2;
      if ((rc = intro_input(E, &in2))) return rc;
      return stk_push_num(&E->d, (int64_t)in2->prev_index, 1);
    }
    case 0xbc: { /* OpTxInputScriptSigSubstr */
      int32_t se[2];
      if ((rc = stk_pop_i32(&E->d, &se[1]))) return rc;
      if ((rc = stk_pop_i32(&E->d, &se[0]))) return rc;
      const ok_input *in2;
      if ((rc = intro_input(E, &in2))) return rc;
      return intro_substr(E, in2->sig_script, in2->sig_script_len, se[0], se[1]);
    }
    case 0xbd: { /* OpTxInputSeq: raw 8B LE (bitflag field) */
      const ok_input *in2;
      if ((rc = intro_input(E, &in2))) return rc;
      uint8_t b[8];
      for (int i = 0; i < 8; i++) b[i] = (uint8_t)(in2->sequence >> (8 * i));
      return stk_push_copy(&E->d, b, 8, 1);
    }
    case 0xbe: { /* OpTxInputAmount */
      const ok_input *in2;
      if ((rc = intro_input(E, &in2))) return rc;
      if (in2->utxo_amount > (uint64_t)INT64_MAX) return KV_SCRIPT_NUMBER_TOO_BIG;
      return stk_push_num(&E->d, (int64_t)in2->utxo_amount, 1);
    }
    case 0xbf: { /* OpTxInputSpk: version u16 LE + script */
      const ok_input *in2;
      if ((rc = intro_input(E, &in2))) return rc;
      return intro_push_spk(E, in2->utxo_spk_version, in2->utxo_spk,
                            in2->utxo_spk_len);
    }
    case 0xc0: { /* OpTxInputDaaScore */
      const ok_input *in2;
      if ((rc = intro_input(E, &in2))) return rc;
      return stk_push_num(&E->d, (int64_t)in2->utxo_daa_score, 1);
    }
    case 0xc1: { /* OpTxInputIsCoinbase */
      const ok_input *in2;
      if ((rc = intro_input(E, &in2))) return rc;
      return stk_push_num(&E->d, in2->utxo_is_coinbase ? 1 : 0, 1);
    }
    case 0xc2: { /* OpTxOutputAmount */
      const ok_output *o2;
      if ((rc = intro_output(E, &o2))) return rc;
      if (o2->value > (uint64_t)INT64_MAX) return KV_SCRIPT_NUMBER_TOO_BIG;
      return stk_push_num(&E->d, (int64_t)o2->value, 1);
    }
    case 0xc3: { /* OpTxOutputSpk */
      const ok_output *o2;
      if ((rc = intro_output(E, &o2))) return rc;
      return intro_push_spk(E, o2->spk_version, o2->spk, o2->spk_len);
    }
    case 0xc4: /* OpTxPayloadLen */
      return stk_push_num(&E->d, (int64_t)E->tx->payload_len, 1);
    case 0xc5: { /* OpTxInputSpkLen (to_bytes len = 2 + script len) */
      const ok_input *in2;
      if ((rc = intro_input(E, &in2))) return rc;
      return stk_push_num(&E->d, 2 + (int64_t)in2->utxo_spk_len, 1);
    }
    case 0xc6: { /* OpTxInputSpkSubstr (over version-prefixed bytes) */
      int32_t se[2];
      if ((rc = stk_pop_i32(&E->d, &se[1]))) return rc;
      if ((rc = stk_pop_i32(&E->d, &se[0]))) return rc;
      const ok_input *in2;
      if ((rc = intro_input(E, &in2))) return rc;
      return intro_spk_substr(E, in2->utxo_spk_version, in2->utxo_spk,
                              in2->utxo_spk_len, se[0], se[1]);
    }
    case 0xc7: { /* OpTxOutputSpkLen */
      const ok_output *o2;
      if ((rc = intro_output(E, &o2))) return rc;
      return stk_push_num(&E->d, 2 + (int64_t)o2->spk_len, 1);
    }
    case 0xc8: { /* OpTxOutputSpkSubstr */
      int32_t se[2];
      if ((rc = stk_pop_i32(&E->d, &se[1]))) return rc;
      if ((rc = stk_pop_i32(&E->d, &se[0]))) return rc;
      const ok_output *o2;
      if ((rc = intro_output(E, &o2))) return rc;
      return intro_spk_substr(E, o2->spk_version, o2->spk, o2->spk_len,
                              se[0], se[1]);
    }
    case 0xc9: { /* OpTxInputScriptSigLen */
      const ok_input *in2;
      if ((rc = intro_input(E, &in2))) return rc;
      return stk_push_num(&E->d, (int64_t)in2->sig_script_len, 1);
    }
    case 0xcb: { /* OpAuthOutputCount (covenants.rs:74) */
      int32_t ii;
      if ((rc = stk_pop_i32(&E->d, &ii))) return rc;
      if (ii < 0) return KV_SCRIPT_INVALID_INDEX;
      if ((uint32_t)ii >= E->tx->n_inputs) return KV_SCRIPT_INVALID_INPUT_INDEX;
      int64_t cnt = 0;
      for (uint32_t o = 0; o < E->tx->n_outputs; o++)
        if (E->tx->outputs[o].has_covenant &&
            E->tx->outputs[o].cov_auth_input == (uint16_t)ii)
          cnt++;
      return stk_push_num(&E->d, cnt, 1);
    }
    case 0xcc: { /* OpAuthOutputIdx (covenants.rs:66) */
      int32_t k, ii;
      if ((rc = stk_pop_i32(&E->d, &k))) return rc;
      if ((rc = stk_pop_i32(&E->d, &ii))) return rc;
      if (ii < 0 || k < 0) return KV_SCRIPT_INVALID_INDEX;
      if ((uint32_t)ii >= E->tx->n_inputs) return KV_SCRIPT_INVALID_INPUT_INDEX;
      int64_t seen = 0;
      for (uint32_t o = 0; o < E->tx->n_outputs; o++)
        if (E->tx->outputs[o].has_covenant &&
            E->tx->outputs[o].cov_auth_input == (uint16_t)ii) {
          if (seen == k) return stk_push_num(&E->d, (int64_t)o, 1);
          seen++;
        }
      return KV_SCRIPT_INVALID_SOURCE; /* CovenantsError -> UNKNOWN_ERROR */
    }
    case 0xcf: { /* OpInputCovenantId (ZERO_HASH when absent) */
      const ok_input *in2;
      if ((rc = intro_input(E, &in2))) return rc;
      uint8_t zero[32] = {0};
      return stk_push_copy(
          &E->d, in2->utxo_covenant_id ? in2->utxo_covenant_id : zero, 32, 1);
    }
    case 0xd0: case 0xd1: case 0xd2: case 0xd3: { /* OpCov*{Count,Idx} */
      int32_t k = 0;
      if (op == 0xd1 || op == 0xd3) {
        /* stack: [k, covenant_id(top)] — covenant id popped first below */
      }
      sent ide;
      if ((rc = stk_pop_raw(&E->d, 1, &ide))) return rc;
      if (ide.len != 32) {
        free(ide.d);
        return KV_SCRIPT_INVALID_STATE;
      }
      if (op == 0xd1 || op == 0xd3) {
        if ((rc = stk_pop_i32(&E->d, &k))) {
          free(ide.d);
          return rc;
        }
        if (k < 0) {
          free(ide.d);
          return KV_SCRIPT_INVALID_INDEX;
        }
      }
      int64_t seen = 0;
      int want_inputs = (op == 0xd0 || op == 0xd1);
      int want_idx = (op == 0xd1 || op == 0xd3);
      int found = -1;
      if (want_inputs) {
        for (uint32_t i = 0; i < E->tx->n_inputs; i++) {
          const ok_input *in2 = &E->tx->inputs[i];
          if (in2->utxo_covenant_id && !memcmp(in2->utxo_covenant_id, ide.d, 32)) {
            if (want_idx && seen == k) found = (int)i;
            seen++;
          }
        }
      } else {
        for (uint32_t o = 0; o < E->tx->n_outputs; o++) {
          const ok_output *o2 = &E->tx->outputs[o];
          if (o2->has_covenant && !memcmp(o2->cov_id, ide.d, 32)) {
            if (want_idx && seen == k) found = (int)o;
            seen++;
          }
        }
      }
      free(ide.d);
      if (!want_idx) return stk_push_num(&E->d, seen, 1);
      if (found < 0) return KV_SCRIPT_INVALID_SOURCE; /* CovenantsError */
      return stk_push_num(&E->d, (int64_t)found, 1);
    }
    case 0xd4: { /* OpChainblockSeqCommit (KIP-21) */
      if (!g_seqc_set) return KV_SCRIPT_INVALID_OPCODE; /* accessor None */
      sent blk;
      if ((rc = stk_pop_raw(&E->d, 1, &blk))) return rc;
      if (blk.len != 32) {
        free(blk.d);
        return KV_SCRIPT_INVALID_STATE; /* Hash::try_from */
      }
      int known = memcmp(blk.d, g_seqc_block, 32) == 0;
      free(blk.d);
      /* mock: known block => ancestor with a commitment; anything else =>
       * is_chain_ancestor None => BlockAlreadyPruned (UNKNOWN_ERROR class) */
      if (!known) return KV_SCRIPT_INVALID_SOURCE;
      return stk_push_copy(&E->d, g_seqc_commit, 32, 1);
    }
    case 0xd5: { /* OpOutputCovenantId */
      const ok_output *o2;
      if ((rc = intro_output(E, &o2))) return rc;
      uint8_t zero[32] = {0};
      return stk_push_copy(&E->d, o2->has_covenant ? o2->cov_id : zero, 32, 1);
    }
    case 0xd6: { /* OpOutputAuthorizingInput (-1 when no covenant) */
      const ok_output *o2;
      if ((rc = intro_output(E, &o2))) return rc;
      return stk_push_num(
          &E->d, o2->has_covenant ? (int64_t)o2->cov_auth_input : -1, 1);
    }
    default:
      if (op <= 0x4e) { /* data pushes (literal, unmetered) */
        return stk_push_copy(&E->d, data, dlen, 0);
      }
      if (op >= 0x51 && op <= 0x60) /* Op1..Op16 (literal, unmetered) */
        return stk_push_num(&E->d, op - 0x50, 0);
      if (op == 0x80 || op == 0x81 || op == 0x8d || op == 0x8e || op == 0x98 ||
          op == 0x99)
        return KV_SCRIPT_OPCODE_DISABLED; /* unreachable: checked pre-exec */
      if (op == 0x65 || op == 0x66) return KV_SCRIPT_OPCODE_RESERVED;
      if (op == 0xa6 /* OpZkPrecompile */ || op == 0xd4 /* OpChainblockSeqCommit
             (needs the KIP-21 reachability accessor — out of scope) */)
        return KV_SCRIPT_UNSUPPORTED_OPCODE;
      return KV_SCRIPT_INVALID_OPCODE; /* 0xca, 0xdb..0xff */
  }
}

static int op_is_disabled(uint8_t op) {
  return op == 0x80 || op == 0x81 || op == 0x8d || op == 0x8e || op == 0x98 ||
         op == 0x99;
}
static int op_always_illegal(uint8_t op) { return op == 0x65 || op == 0x66; }
static int op_is_push(uint8_t op) { return op <= NO_COST_OPCODE; }
static int op_is_conditional(uint8_t op) { return op >= 0x63 && op <= 0x68; }

/* execute one script (lib.rs:617-651). verify_only_push for the sig script. */
static int execute_script(eng *E, const uint8_t *script, uint32_t slen,
                          int verify_only_push) {
  uint32_t pc = 0;
  int rc = 0;
  while (pc < slen) {
    uint8_t op = script[pc++];
    const uint8_t *data = NULL;
    uint32_t dlen = 0;
    if (op >= 0x01 && op <= 0x4b) {
      dlen = op;
      if (pc + dlen > slen) { rc = KV_SCRIPT_MALFORMED_PUSH; break; }
      data = script + pc;
      pc += dlen;
    } else if (op >= 0x4c && op <= 0x4e) {
      uint32_t szlen = op == 0x4c ? 1 : (op == 0x4d ? 2 : 4);
      if (pc + szlen > slen) { rc = KV_SCRIPT_MALFORMED_PUSH; break; }
      dlen = 0;
      for (uint32_t i = 0; i < szlen; i++) dlen |= (uint32_t)script[pc + i] << (8 * i);
      pc += szlen;
      if (pc + dlen > slen || dlen > slen) { rc = KV_SCRIPT_MALFORMED_PUSH; break; }
      data = script + pc;
      pc += dlen;
    }
    /* execute_script checks (lib.rs:618-631) */
    if (op_is_disabled(op)) { rc = KV_SCRIPT_OPCODE_DISABLED; break; }
    if (op_always_illegal(op)) { rc = KV_SCRIPT_OPCODE_RESERVED; break; }
    if (verify_only_push && !op_is_push(op)) {
      rc = KV_SCRIPT_NOT_PUSH_ONLY;
      break;
    }
    /* execute_opcode (lib.rs:576-598) */
    if (!op_is_push(op)) {
      E->num_ops += 1;
      if (E->num_ops > MAX_OPS_PER_SCRIPT) { rc = KV_SCRIPT_TOO_MANY_OPERATIONS; break; }
    } else if (dlen > MAX_SCRIPT_ELEMENT_SIZE) {
      rc = KV_SCRIPT_ELEMENT_TOO_BIG;
      break;
    }
    if (is_executing(E) || op_is_conditional(op)) {
      rc = exec_opcode(E, op, data, dlen);
      if (rc) break;
      rc = charge_pushed(E);
      if (rc) break;
    }
    if (E->d.n + E->a.n > MAX_STACK_SIZE) { rc = KV_SCRIPT_STACK_SIZE_EXCEEDED; break; }
  }
  if (rc == 0 && E->cond_n != 0) rc = KV_SCRIPT_UNBALANCED_CONDITIONAL;
  /* alt stack does not persist; num_ops is per script (lib.rs:646-648) */
  stk_clear(&E->a);
  E->a.pushed_bytes = 0;
  E->num_ops = 0;
  return rc;
}

static int is_p2sh_spk(const uint8_t *spk, uint32_t len) {
  /* OpBlake2b OpData32 <32> OpEqual (script_class.rs:77-82) */
  return len == 35 && spk[0] == 0xaa && spk[1] == 0x20 && spk[34] == 0x87;
}

/* final-stack check (lib.rs:734-748) */
static int check_error_condition(eng *E, int final_script) {
  if (final_script) {
    if (E->d.n > 1) return KV_SCRIPT_CLEAN_STACK;
    if (E->d.n == 0) return KV_SCRIPT_EMPTY_STACK;
  }
  int b;
  int rc = stk_pop_bool(&E->d, &b);
  if (rc) return rc == KV_SCRIPT_INVALID_STACK_OPERATION ? KV_SCRIPT_INVALID_STACK_OPERATION : rc;
  return b ? 0 : KV_SCRIPT_EVAL_FALSE;
}

/* TxScriptEngine::execute for one populated input (lib.rs:653-721) */
int ok_script_check_input(const ok_tx *tx, uint32_t input_index, uint64_t mass_per_sig_op,
                          ok_sighash_reused *reused) {
  const ok_input *in = &tx->inputs[input_index];
  eng E;
  memset(&E, 0, sizeof(E));
  stk_init(&E.d);
  stk_init(&E.a);
  E.tx = tx;
  E.input = in;
  E.idx = input_index;
  E.reused = reused;
  E.sigop_units = mass_per_sig_op * SCRIPT_UNITS_PER_GRAM;
  /* script_units_limit = compute_commit.allowed_script_units()
   * (tx.rs:100-103, mass/units.rs:22-24) */
  uint64_t committed = in->commit_kind == 0
                           ? (uint64_t)in->commit_value * SCRIPT_UNITS_PER_SIGOP_COUNT_UNIT
                           : (uint64_t)in->commit_value * SCRIPT_UNITS_PER_COMPUTE_BUDGET_UNIT;
  uint64_t limit = committed + FREE_SCRIPT_UNITS_PER_INPUT; /* saturating in u64 */
  if (limit < committed) limit = UINT64_MAX;
  E.limit_units = limit;
  E.remaining_units = limit;

  int rc;
  /* unknown spk version accepted without execution (lib.rs:655-659) */
  if (in->utxo_spk_version > 0) return 0;

  E.is_p2sh = is_p2sh_spk(in->utxo_spk, in->utxo_spk_len);

  /* charge oversized utxo spk (lib.rs:661-676): grams → units */
  uint64_t extra = in->utxo_spk_len > STANDARD_SPK_MAX_SIZE
                       ? (uint64_t)(in->utxo_spk_len - STANDARD_SPK_MAX_SIZE)
                       : 0;
  if ((rc = consume_units(&E, extra * SCRIPT_UNITS_PER_GRAM))) goto out;

  if (in->sig_script_len == 0 && in->utxo_spk_len == 0) {
    rc = KV_SCRIPT_EVAL_FALSE;
    goto out;
  }
  if (in->sig_script_len > MAX_SCRIPTS_SIZE || in->utxo_spk_len > MAX_SCRIPTS_SIZE) {
    rc = KV_SCRIPT_SCRIPT_SIZE;
    goto out;
  }

  {
    sent saved[STACK_CAP];
    int saved_n = -1;
    /* run [sig_script, spk] skipping empty ones (lib.rs:694-704) */
    int script_idx = 0;
    const uint8_t *scripts[2] = {in->sig_script, in->utxo_spk};
    uint32_t lens[2] = {in->sig_script_len, in->utxo_spk_len};
    for (int i = 0; i < 2; i++) {
      if (lens[i] == 0) { script_idx++; continue; }
      int verify_only_push = (i == 0); /* idx==0 && TxInput source */
      if (E.is_p2sh && i == 1) {
        /* save dstack (deep copy) */
        saved_n = E.d.n;
        for (int k = 0; k < saved_n; k++) {
          saved[k].len = E.d.it[k].len;
          saved[k].d = malloc(saved[k].len ? saved[k].len : 1);
          memcpy(saved[k].d, E.d.it[k].d, saved[k].len);
        }
      }
      rc = execute_script(&E, scripts[i], lens[i], verify_only_push);
      if (rc) {
        if (saved_n >= 0) free_ents(saved, saved_n);
        goto out;
      }
      script_idx++;
    }
    if (E.is_p2sh) {
      rc = check_error_condition(&E, 0);
      if (rc) {
        if (saved_n >= 0) free_ents(saved, saved_n);
        goto out;
      }
      /* restore saved stack */
      stk_clear(&E.d);
      if (saved_n < 0) { rc = KV_SCRIPT_EMPTY_STACK; goto out; }
      for (int k = 0; k < saved_n; k++) E.d.it[k] = saved[k];
      E.d.n = saved_n;
      /* pop redeem script */
      if (E.d.n == 0) { rc = KV_SCRIPT_EMPTY_STACK; goto out; }
      sent redeem = E.d.it[--E.d.n];
      rc = execute_script(&E, redeem.d, redeem.len, 0);
      free(redeem.d);
      if (rc) goto out;
    }
    rc = check_error_condition(&E, 1);
  }
out:
  stk_clear(&E.d);
  stk_clear(&E.a);
  return rc;
}
