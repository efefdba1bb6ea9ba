/* ORACLE (test infrastructure only).
 *
 * validate_populated_transaction_and_get_fee + check_scripts + muhash reduce,
 * restating consensus/src/processes/transaction_validator/
 * tx_validation_in_utxo_context.rs:37-218 and consensus/src/pipeline/
 * virtual_processor/utxo_validation.rs:297-348.
 *
 * Round-1 scope notes (documented in DESIGN.md):
 *  - KV_FLAGS_FULL's storage-mass commitment check (check_mass_commitment,
 *    :126-134) is not yet restated; callers must pass KV_FLAGS_SKIP_MASS_CHECK
 *    (the flag the reference's own validator tests use). Full returns
 *    KV_ERR_BAD_BLOB to fail loudly rather than silently skip.
 *  - Covenant context (check_covenant_info) is out of scope; blobs carrying
 *    covenant bindings are rejected the same way.
 */
#include "kaspa_engine_abi.h"
#include "ok_tx.h"
#include <math.h>
#include <string.h>
#ifdef _OPENMP
#include <omp.h>
#endif

#define MAX_SOMPI (29000000000ULL * 100000000ULL)
#define SEQ_DISABLED (1ULL << 63)
#define SEQ_MASK 0x00000000ffffffffULL

/* default consensus params (config/params.rs:582-650 mainnet) */
#define DEFAULT_COINBASE_MATURITY 1000
#define DEFAULT_MASS_PER_SIG_OP 1000


/* ---------------- storage mass (KIP-0009), restating
 * consensus/core/src/mass/mod.rs:385-514 + utxo_plurality (:86-105).
 * storage_mass_parameter = STORAGE_MASS_PARAMETER (constants.rs:26) = 1e12. */
#define STORM_PARAM (100000000ULL * 10000ULL)

static uint64_t plurality(uint32_t spk_len, int has_cov) {
  uint64_t sz = 63ull + spk_len + (has_cov ? 32 : 0);
  return (sz + 99) / 100;
}

/* returns 0 ok (mass in *out), -1 incomputable */
static int calc_storage_mass_tx(const ok_tx *tx, uint64_t *out) {
  if (ok_tx_is_coinbase(tx)) {
    *out = 0;
    return 0;
  }
  uint64_t outs_plurality = 0, harmonic_outs = 0;
  for (uint32_t i = 0; i < tx->n_outputs; i++) {
    const ok_output *o = &tx->outputs[i];
    uint64_t p = plurality(o->spk_len, o->cov_id != NULL);
    if (o->value == 0) return -1; /* zero-value outputs are rejected upstream
                                     (in-isolation checks); treat as incomputable */
    uint64_t cp, cpp;
    if (__builtin_mul_overflow(STORM_PARAM, p, &cp)) return -1;
    if (__builtin_mul_overflow(cp, p, &cpp)) return -1;
    uint64_t h;
    if (__builtin_add_overflow(harmonic_outs, cpp / o->value, &h)) return -1;
    harmonic_outs = h;
    outs_plurality += p;
  }
  /* relaxed path: |O| = 1, |I| = 1, or |O| = |I| = 2 */
  int relaxed;
  uint64_t ins_plurality_small = 0;
  if (outs_plurality == 1) {
    relaxed = 1;
  } else if (tx->n_inputs > 2) {
    relaxed = 0;
  } else {
    for (uint32_t i = 0; i < tx->n_inputs; i++)
      ins_plurality_small +=
          plurality(tx->inputs[i].utxo_spk_len, tx->inputs[i].utxo_covenant_id != NULL);
    relaxed = (ins_plurality_small == 1) ||
              (outs_plurality == 2 && ins_plurality_small == 2);
  }
  if (relaxed) {
    uint64_t harmonic_ins = 0;
    for (uint32_t i = 0; i < tx->n_inputs; i++) {
      const ok_input *in = &tx->inputs[i];
      uint64_t p = plurality(in->utxo_spk_len, in->utxo_covenant_id != NULL);
      if (in->utxo_amount == 0) return -1;
      uint64_t c = STORM_PARAM * p * p / in->utxo_amount; /* no overflow: plurality
                       bounded by max spk len (verify_utxo_plurality_limits) */
      uint64_t s = harmonic_ins + c;
      harmonic_ins = s < harmonic_ins ? UINT64_MAX : s; /* saturating_add */
    }
    *out = harmonic_outs > harmonic_ins ? harmonic_outs - harmonic_ins : 0;
    return 0;
  }
  uint64_t ins_plurality = 0, sum_ins = 0;
  for (uint32_t i = 0; i < tx->n_inputs; i++) {
    const ok_input *in = &tx->inputs[i];
    ins_plurality += plurality(in->utxo_spk_len, in->utxo_covenant_id != NULL);
    sum_ins += in->utxo_amount;
  }
  uint64_t mean_ins = sum_ins / ins_plurality;
  if (mean_ins == 0) mean_ins = 1;
  uint64_t arith;
  if (__builtin_mul_overflow(ins_plurality, STORM_PARAM / mean_ins, &arith))
    arith = UINT64_MAX; /* saturating_mul */
  *out = harmonic_outs > arith ? harmonic_outs - arith : 0;
  return 0;
}

static int validate_one_tx(const ok_tx *tx, uint64_t pov_daa, uint32_t flags,
                           uint64_t coinbase_maturity, uint64_t mass_per_sig_op,
                           uint64_t *fee_out) {
  /* 1. coinbase maturity (:81-97) */
  for (uint32_t i = 0; i < tx->n_inputs; i++) {
    const ok_input *in = &tx->inputs[i];
    if (in->utxo_is_coinbase && in->utxo_daa_score + coinbase_maturity > pov_daa)
      return KV_ERR_IMMATURE_COINBASE;
  }
  /* 2. input amounts (:99-114) */
  uint64_t total_in = 0;
  for (uint32_t i = 0; i < tx->n_inputs; i++) {
    uint64_t amt = tx->inputs[i].utxo_amount;
    if (total_in + amt < total_in) return KV_ERR_INPUT_AMOUNT_OVERFLOW;
    total_in += amt;
    if (total_in > MAX_SOMPI) return KV_ERR_INPUT_AMOUNT_TOO_HIGH;
  }
  /* 3. outputs (:116-124) */
  uint64_t total_out = 0;
  for (uint32_t i = 0; i < tx->n_outputs; i++) total_out += tx->outputs[i].value;
  if (total_in < total_out) return KV_ERR_SPEND_TOO_HIGH;
  uint64_t fee = total_in - total_out;
  /* 4. mass commitment (check_mass_commitment, :126-134) */
  if (flags != KV_FLAGS_SKIP_MASS_CHECK) {
    uint64_t calc = 0;
    if (calc_storage_mass_tx(tx, &calc)) return KV_ERR_MASS_INCOMPUTABLE;
    if (calc != tx->storage_mass) return KV_ERR_WRONG_MASS;
  }
  /* 5. sequence locks (:136-161) */
  for (uint32_t i = 0; i < tx->n_inputs; i++) {
    const ok_input *in = &tx->inputs[i];
    if ((in->sequence & SEQ_DISABLED) == SEQ_DISABLED) continue;
    int64_t relative_lock = (int64_t)(in->sequence & SEQ_MASK);
    int64_t lock_daa = (int64_t)in->utxo_daa_score + relative_lock - 1;
    if (lock_daa >= (int64_t)pov_daa) return KV_ERR_SEQUENCE_LOCK;
  }
  /* 6. covenants: reject blobs with covenant data (round-1 out of scope) */
  for (uint32_t i = 0; i < tx->n_outputs; i++)
    if (tx->outputs[i].has_covenant) return KV_ERR_BAD_BLOB;
  /* 7. scripts (:60-65,163-218) */
  if (flags != KV_FLAGS_SKIP_SCRIPT_CHECKS) {
    ok_sighash_reused reused;
    ok_reused_init(&reused);
    for (uint32_t i = 0; i < tx->n_inputs; i++) {
      int src = ok_script_check_input(tx, i, mass_per_sig_op, &reused);
      if (src) {
        /* map_script_err (:220-222) */
        return tx->inputs[i].sig_script_len == 0 ? KV_ERR_SIGNATURE_EMPTY_BASE + src
                                                 : KV_ERR_SIGNATURE_INVALID_BASE + src;
      }
    }
  }
  *fee_out = fee;
  return KV_OK;
}

static int validate_impl(const uint8_t *blob, size_t blob_len, uint64_t pov_daa_score,
                         uint64_t block_daa_score, uint32_t flags, int threads,
                         int32_t *tx_codes_out, uint64_t *fees_out,
                         uint8_t muhash_out[32]) {
  if (blob_len < 4) return -1;
  uint32_t n_txs;
  memcpy(&n_txs, blob, 4);

  uint64_t num[OK_U3072_LIMBS], den[OK_U3072_LIMBS];
  ok_u3072_one(num);
  ok_u3072_one(den);
  int bad = 0;

#ifdef _OPENMP
  if (threads > 1) {
    omp_set_num_threads(threads);
#pragma omp parallel
    {
      uint64_t lnum[OK_U3072_LIMBS], lden[OK_U3072_LIMBS];
      ok_u3072_one(lnum);
      ok_u3072_one(lden);
#pragma omp for schedule(dynamic, 4)
      for (uint32_t t = 0; t < n_txs; t++) {
        ok_tx tx;
        if (ok_tx_parse(blob, blob_len, t, &tx)) {
          tx_codes_out[t] = KV_ERR_BAD_BLOB;
          fees_out[t] = 0;
          __atomic_store_n(&bad, 1, __ATOMIC_RELAXED);
          continue;
        }
        uint64_t fee = 0;
        int code = ok_tx_is_coinbase(&tx)
                       ? KV_ERR_BAD_BLOB /* coinbase never enters this path */
                       : validate_one_tx(&tx, pov_daa_score, flags,
                                         DEFAULT_COINBASE_MATURITY,
                                         DEFAULT_MASS_PER_SIG_OP, &fee);
        tx_codes_out[t] = code;
        fees_out[t] = fee;
        if (code == KV_OK && muhash_out)
          ok_tx_muhash(&tx, block_daa_score, lnum, lden);
        ok_tx_free(&tx);
      }
      if (muhash_out) {
#pragma omp critical
        {
          ok_u3072_mul(num, lnum);
          ok_u3072_mul(den, lden);
        }
      }
    }
    if (muhash_out) ok_muhash_finalize(num, den, muhash_out);
    return bad ? -1 : 0;
  }
#endif
  (void)threads;
  for (uint32_t t = 0; t < n_txs; t++) {
    ok_tx tx;
    if (ok_tx_parse(blob, blob_len, t, &tx)) {
      tx_codes_out[t] = KV_ERR_BAD_BLOB;
      fees_out[t] = 0;
      bad = 1;
      continue;
    }
    uint64_t fee = 0;
    int code = ok_tx_is_coinbase(&tx)
                   ? KV_ERR_BAD_BLOB /* coinbase never enters this path */
                   : validate_one_tx(&tx, pov_daa_score, flags,
                                     DEFAULT_COINBASE_MATURITY,
                                     DEFAULT_MASS_PER_SIG_OP, &fee);
    tx_codes_out[t] = code;
    fees_out[t] = fee;
    if (code == KV_OK && muhash_out) ok_tx_muhash(&tx, block_daa_score, num, den);
    ok_tx_free(&tx);
  }
  if (muhash_out) ok_muhash_finalize(num, den, muhash_out);
  return bad ? -1 : 0;
}

int ok_validate_block(const uint8_t *blob, size_t blob_len, uint64_t pov_daa_score,
                      uint64_t block_daa_score, uint32_t flags, int32_t *tx_codes_out,
                      uint64_t *fees_out, uint8_t muhash_out[32]) {
  return validate_impl(blob, blob_len, pov_daa_score, block_daa_score, flags, 1,
                       tx_codes_out, fees_out, muhash_out);
}

int ok_validate_block_parallel(const uint8_t *blob, size_t blob_len,
                               uint64_t pov_daa_score, uint64_t block_daa_score,
                               uint32_t flags, int threads, int32_t *tx_codes_out,
                               uint64_t *fees_out, uint8_t muhash_out[32]) {
  return validate_impl(blob, blob_len, pov_daa_score, block_daa_score, flags, threads,
                       tx_codes_out, fees_out, muhash_out);
}

int ok_check_input_script(const uint8_t *blob, size_t blob_len, uint32_t tx_index,
                          uint32_t input_index) {
  ok_tx tx;
  if (ok_tx_parse(blob, blob_len, tx_index, &tx)) return -1;
  if (input_index >= tx.n_inputs) {
    ok_tx_free(&tx);
    return -1;
  }
  ok_sighash_reused reused;
  ok_reused_init(&reused);
  int rc = ok_script_check_input(&tx, input_index, DEFAULT_MASS_PER_SIG_OP, &reused);
  ok_tx_free(&tx);
  return rc;
}

/* ---------------- parallel schnorr-tuple verify (CPU baseline for config 2) ----
 * tuples: n × 128B = r‖s‖pk‖msg (the kv_verify_schnorr_batch layout). */
int ok_verify_schnorr_batch(const uint8_t *tuples, size_t n, int threads,
                            uint64_t *bitmap_out) {
  size_t words = (n + 63) / 64;
  memset(bitmap_out, 0, words * 8);
#ifdef _OPENMP
  omp_set_num_threads(threads > 0 ? threads : 1);
#pragma omp parallel for schedule(dynamic, 64)
#endif
  for (size_t i = 0; i < n; i++) {
    const uint8_t *t = tuples + i * 128;
    uint8_t sig[64];
    memcpy(sig, t, 64); /* r ‖ s */
    int v = ok_schnorr_verify(t + 64, t + 96, sig) == 1;
    if (v)
#ifdef _OPENMP
#pragma omp atomic
#endif
      bitmap_out[i / 64] |= (1ULL << (i % 64));
  }
  return 0;
}

/* ---------------- synthetic workload generation (bench/test harness only) ----
 * Deterministic (pk,msg,sig) tuples for BASELINE config 2: keys/messages derived
 * from a seed via keyed blake2b, signatures real BIP-340 (aux = zeros).
 * invalid_permille of tuples get one sig byte flipped. */
int ok_gen_schnorr_tuples(uint64_t seed, size_t n, uint32_t invalid_permille,
                          uint8_t *out /* n × 128: r‖s‖pk‖msg */, int threads) {
#ifdef _OPENMP
  omp_set_num_threads(threads > 0 ? threads : 1);
#pragma omp parallel for schedule(dynamic, 64)
#endif
  for (size_t i = 0; i < n; i++) {
    uint8_t buf[16], sk[32], pk[32], msg[32], sig[64];
    memcpy(buf, &seed, 8);
    uint64_t idx = i;
    memcpy(buf + 8, &idx, 8);
    for (int attempt = 0;; attempt++) {
      uint8_t tag = (uint8_t)attempt;
      uint8_t b2[17];
      memcpy(b2, buf, 16);
      b2[16] = tag;
      ok_blake2b_keyed((const uint8_t *)"kv-gen-key", 10, b2, 17, sk);
      if (ok_pubkey_xonly(sk, pk)) break;
    }
    ok_blake2b_keyed((const uint8_t *)"kv-gen-msg", 10, buf, 16, msg);
    ok_schnorr_sign(sk, msg, NULL, sig);
    if (invalid_permille && (i % 1000) < invalid_permille) sig[40] ^= 0x20;
    uint8_t *t = out + i * 128;
    memcpy(t, sig, 64);
    memcpy(t + 64, pk, 32);
    memcpy(t + 96, msg, 32);
  }
  return 0;
}

/* ECDSA tuples: n × 132: r‖s‖pk33‖msg‖pad3 */
int ok_gen_ecdsa_tuples(uint64_t seed, size_t n, uint32_t invalid_permille,
                        uint8_t *out, int threads) {
#ifdef _OPENMP
  omp_set_num_threads(threads > 0 ? threads : 1);
#pragma omp parallel for schedule(dynamic, 64)
#endif
  for (size_t i = 0; i < n; i++) {
    uint8_t buf[16], sk[32], pk[33], msg[32], sig[64];
    memcpy(buf, &seed, 8);
    uint64_t idx = i;
    memcpy(buf + 8, &idx, 8);
    for (int attempt = 0;; attempt++) {
      uint8_t b2[17];
      memcpy(b2, buf, 16);
      b2[16] = (uint8_t)attempt;
      ok_blake2b_keyed((const uint8_t *)"kv-gen-ekey", 11, b2, 17, sk);
      if (ok_pubkey_compressed(sk, pk)) break;
    }
    ok_blake2b_keyed((const uint8_t *)"kv-gen-emsg", 11, buf, 16, msg);
    ok_ecdsa_sign(sk, msg, sig);
    if (invalid_permille && (i % 1000) < invalid_permille) sig[40] ^= 0x20;
    uint8_t *t = out + i * 132;
    memcpy(t, sig, 64);
    memcpy(t + 64, pk, 33);
    memcpy(t + 97, msg, 32);
    t[129] = t[130] = t[131] = 0;
  }
  return 0;
}

/* ---- mempool validation ⇔ validate_mempool_transaction_in_utxo_context
 * (utxo_validation.rs:418-457): contextual storage mass is COMPUTED (the
 * carried commitment is ignored — the mempool sets it), validation runs with
 * SkipMassCheck, and the optional feerate threshold compares
 * fee / normalized_max(mass) <= threshold → FeerateTooLow
 * (tx_validation_in_utxo_context.rs:69-77). Normalization uses the mainnet
 * cofactors (mass/mod.rs:177-190,298-308; params.rs:623-626: limits
 * {compute 500k, storage 500k, transient 1M} → storage cofactor 1.0,
 * transient 0.5), mass_per_tx_byte 1, mass_per_script_pub_key_byte 10,
 * GRAMS_PER_SIGOP_COUNT_UNIT 1000, GRAMS_PER_COMPUTE_BUDGET_UNIT 100,
 * TRANSIENT_BYTE_TO_MASS_FACTOR 4 (constants.rs:31, mass/units.rs:4-5). */

static uint64_t mp_estimated_size(const ok_tx *tx) {
  /* transaction_estimated_serialized_size (mass/mod.rs:21-78) */
  uint64_t size = 2 + 8 + 8 + 8 + 20 + 8 + 32 + 8 + tx->payload_len;
  for (uint32_t i = 0; i < tx->n_inputs; i++) {
    size += 32 + 4 + 8 + tx->inputs[i].sig_script_len + 8;
    if (tx->version >= 1) size += 2;
  }
  for (uint32_t i = 0; i < tx->n_outputs; i++) {
    size += 8 + 2 + 8 + tx->outputs[i].spk_len;
    if (tx->outputs[i].has_covenant) size += 2 + 32;
  }
  return size;
}

/* Generic normalized_max over per-dimension limits ⇔ Mass::normalized_max +
 * MassCofactors::new (mass/mod.rs:258-265,298-308): cofactor_i = Lc/L_i (f64),
 * normalized = ceil(m_i · cofactor_i), result = max over dimensions. */
uint64_t ok_normalized_max_limits(uint64_t storage_mass, uint64_t compute_mass,
                                  uint64_t transient_mass, uint64_t limit_storage,
                                  uint64_t limit_compute, uint64_t limit_transient) {
  double cs = (double)limit_compute / (double)limit_storage;
  double ct = (double)limit_compute / (double)limit_transient;
  uint64_t sn = (uint64_t)ceil((double)storage_mass * cs);
  uint64_t tn = (uint64_t)ceil((double)transient_mass * ct);
  uint64_t m = sn;
  if (compute_mass > m) m = compute_mass;
  if (tn > m) m = tn;
  return m;
}

static uint64_t mp_normalized_mass(const ok_tx *tx, uint64_t storage_mass) {
  if (ok_tx_is_coinbase(tx)) return storage_mass; /* non-contextual = 0 */
  uint64_t size = mp_estimated_size(tx);
  uint64_t spk_bytes = 0;
  for (uint32_t i = 0; i < tx->n_outputs; i++)
    spk_bytes += 2 + tx->outputs[i].spk_len;
  uint64_t script_mass = 0;
  if (tx->version >= 1) {
    for (uint32_t i = 0; i < tx->n_inputs; i++)
      script_mass += 100ull * (tx->inputs[i].commit_kind == 1
                                   ? tx->inputs[i].commit_value
                                   : 0);
  } else {
    for (uint32_t i = 0; i < tx->n_inputs; i++)
      script_mass += 1000ull * (tx->inputs[i].commit_kind == 0
                                    ? tx->inputs[i].commit_value
                                    : 0);
  }
  uint64_t compute = size * 1 + spk_bytes * 10 + script_mass;
  uint64_t transient = size * 4;
  /* mainnet limits {storage 500k, compute 500k, transient 1M}, params.rs:626 */
  return ok_normalized_max_limits(storage_mass, compute, transient, 500000,
                                  500000, 1000000);
}

int ok_validate_mempool(const uint8_t *blob, size_t blob_len, uint64_t pov_daa_score,
                        double feerate_threshold, int threads,
                        int32_t *tx_codes_out, uint64_t *fees_out) {
  uint8_t mh[32];
  int rc = ok_validate_block_parallel(blob, blob_len, pov_daa_score, pov_daa_score,
                                      KV_FLAGS_SKIP_MASS_CHECK, threads,
                                      tx_codes_out, fees_out, mh);
  if (rc != 0) return rc;
  uint32_t n_txs;
  memcpy(&n_txs, blob, 4);
  for (uint32_t t = 0; t < n_txs; t++) {
    ok_tx tx;
    if (ok_tx_parse(blob, blob_len, t, &tx) != 0) return -1;
    uint64_t storage = 0;
    if (calc_storage_mass_tx(&tx, &storage)) {
      /* contextual mass computed BEFORE validation in the reference —
       * MassIncomputable wins over any validation error */
      tx_codes_out[t] = KV_ERR_MASS_INCOMPUTABLE;
      fees_out[t] = 0;
    } else if (tx_codes_out[t] == 0 && feerate_threshold > 0) {
      uint64_t m = mp_normalized_mass(&tx, storage);
      if (m > 0 && (double)fees_out[t] / (double)m <= feerate_threshold) {
        tx_codes_out[t] = KV_ERR_FEERATE_TOO_LOW;
        fees_out[t] = 0;
      }
    }
    ok_tx_free(&tx);
  }
  return 0;
}

/* ---- direct test exports for the golden mass vectors (tests/golden/mass.json,
 * extracted from consensus/core/src/mass/mod.rs:531-953). Test infrastructure:
 * they expose the same static helpers the validation path uses. ---- */

uint64_t ok_test_plurality(uint32_t spk_len, int has_cov) {
  return plurality(spk_len, has_cov);
}

/* Build a minimal non-coinbase tx view carrying only the fields
 * calc_storage_mass_tx reads. Returns 0 ok (mass in *out), -1 incomputable. */
int ok_test_storage_mass(uint32_t n_ins, const uint64_t *in_amounts,
                         const uint32_t *in_spk_lens, const uint8_t *in_has_cov,
                         uint32_t n_outs, const uint64_t *out_amounts,
                         const uint32_t *out_spk_lens, const uint8_t *out_has_cov,
                         uint64_t *mass_out) {
  static const uint8_t zero_subnet[20] = {0};
  static const uint8_t cov_stub[32] = {0};
  ok_input ins[64];
  ok_output outs[64];
  if (n_ins > 64 || n_outs > 64) return -2;
  memset(ins, 0, sizeof(ins[0]) * n_ins);
  memset(outs, 0, sizeof(outs[0]) * n_outs);
  for (uint32_t i = 0; i < n_ins; i++) {
    ins[i].utxo_amount = in_amounts[i];
    ins[i].utxo_spk_len = in_spk_lens ? in_spk_lens[i] : 0;
    ins[i].utxo_covenant_id = (in_has_cov && in_has_cov[i]) ? cov_stub : NULL;
  }
  for (uint32_t i = 0; i < n_outs; i++) {
    outs[i].value = out_amounts[i];
    outs[i].spk_len = out_spk_lens ? out_spk_lens[i] : 0;
    outs[i].cov_id = (out_has_cov && out_has_cov[i]) ? cov_stub : NULL;
  }
  ok_tx tx;
  memset(&tx, 0, sizeof(tx));
  tx.n_inputs = n_ins;
  tx.n_outputs = n_outs;
  tx.subnetwork_id = zero_subnet;
  tx.inputs = ins;
  tx.outputs = outs;
  return calc_storage_mass_tx(&tx, mass_out);
}
