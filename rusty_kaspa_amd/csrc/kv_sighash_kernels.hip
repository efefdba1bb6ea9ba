/* MI355X-native sighash + MuHash kernels (gfx950, PRODUCT code).
 *
 * kv_tx_subhash_kernel  ⇔ the reused per-tx subhashes (SigHashReusedValues,
 *   consensus/core/src/hashing/sighash.rs:14-41,140-221): one tx per lane,
 *   keyed-BLAKE2b over the blob's input/output records. This is the
 *   quadratic-hashing avoidance the reference gets from SigHashReusedValues —
 *   subhashes computed once per tx and shared by all of its inputs.
 * kv_sighash_assemble_kernel ⇔ calc_schnorr/ecdsa_signature_hash
 *   (sighash.rs:245-292): one (input, hash_type) job per lane; emits ready
 *   128B/132B verify tuples for the EC kernels.
 * kv_muhash_element_kernel ⇔ MuHashElementBuilder::finalize
 *   (crypto/muhash/src/lib.rs:148-169) over write_utxo serialization
 *   (consensus/core/src/muhash.rs:55-69): keyed-BLAKE2b → ChaCha20 384B.
 * kv_u3072_reduce_kernel ⇔ the muhash monoid reduce
 *   (utxo_validation.rs:319-348): strided modular-multiply chains.
 */
#include "kv_blob_device.h"
#include "kv_hash_device.h"
#include "kv_u3072.h"

namespace kv {

__device__ __constant__ static const uint8_t KEY_SIGNING[22] = {
    'T', 'r', 'a', 'n', 's', 'a', 'c', 't', 'i', 'o', 'n',
    'S', 'i', 'g', 'n', 'i', 'n', 'g', 'H', 'a', 's', 'h'};
/* SHA256("TransactionSigningHashECDSA") — the new_with_prefix domain hash */
__device__ __constant__ static const uint8_t ECDSA_DOMAIN_HASH[32] = {
    0xa4, 0xf2, 0xec, 0xe4, 0x5a, 0x28, 0x6c, 0xb1, 0xec, 0x0a, 0x4e,
    0x4d, 0x38, 0x34, 0x68, 0xd0, 0x00, 0xf7, 0x17, 0x57, 0x05, 0x2b,
    0x15, 0x04, 0xaa, 0x34, 0x95, 0x32, 0x8d, 0xf5, 0xf4, 0xea};
__device__ __constant__ static const uint8_t KEY_MUHASH_ELEM[13] = {
    'M', 'u', 'H', 'a', 's', 'h', 'E', 'l', 'e', 'm', 'e', 'n', 't'};

#define KV_SIGHASH_ALL 0x01
#define KV_SIGHASH_NONE 0x02
#define KV_SIGHASH_SINGLE 0x04
#define KV_SIGHASH_ACP 0x80
#define KV_SIGHASH_MASK 0x07

/* per-tx subhash block: prevouts ‖ sequences ‖ sigops ‖ outputs ‖ payload */
#define SUBHASH_STRIDE 160

__device__ inline void hash_output_fields(b2b_state &S, const blob_output &o,
                                          uint16_t version) {
  b2b_update_u64(S, o.value);
  b2b_update_u16(S, o.spk_version);
  b2b_update_u64(S, o.spk_len);
  b2b_update(S, o.spk, o.spk_len);
  if (version >= 1) {
    uint8_t has = o.has_covenant ? 1 : 0;
    b2b_update(S, &has, 1);
    if (o.has_covenant) {
      b2b_update_u16(S, o.cov_auth_input);
      b2b_update(S, o.cov_id, 32);
    }
  }
}

extern "C" __global__ void kv_tx_subhash_kernel(const uint8_t *__restrict__ blob,
                                                uint32_t n_txs,
                                                uint8_t *__restrict__ subhashes) {
  uint32_t t = blockIdx.x * blockDim.x + threadIdx.x;
  if (t >= n_txs) return;
  blob_tx tx;
  blob_tx_at(blob, t, tx);
  uint8_t *out = subhashes + (size_t)t * SUBHASH_STRIDE;

  /* ONE live blake2b state at a time: three interleaved states kept ~500B of
   * per-lane scratch hot through the whole walk (the r02 PMC measured the
   * kernel 14x over its algorithmic write traffic). Re-walking the input
   * records costs L2-resident re-reads, far cheaper than the spills. */
  const uint8_t *outs_start;
  {
    b2b_state S;
    b2b_init_keyed(S, KEY_SIGNING, 22);
    const uint8_t *p = tx.inputs0;
    for (uint32_t i = 0; i < tx.n_inputs; i++) {
      blob_input in;
      blob_input_at(p, in);
      b2b_update(S, in.prev_tx_id, 32);
      b2b_update_u32(S, in.prev_index);
      p = in.end;
    }
    outs_start = p;
    b2b_final(S, out);
  }
  {
    b2b_state S;
    b2b_init_keyed(S, KEY_SIGNING, 22);
    const uint8_t *p = tx.inputs0;
    for (uint32_t i = 0; i < tx.n_inputs; i++) {
      blob_input in;
      blob_input_at(p, in);
      b2b_update_u64(S, in.sequence);
      p = in.end;
    }
    b2b_final(S, out + 32);
  }
  {
    b2b_state S;
    b2b_init_keyed(S, KEY_SIGNING, 22);
    const uint8_t *p = tx.inputs0;
    for (uint32_t i = 0; i < tx.n_inputs; i++) {
      blob_input in;
      blob_input_at(p, in);
      uint8_t sop = in.commit_kind == 0 ? (uint8_t)in.commit_value : 0;
      b2b_update(S, &sop, 1);
      p = in.end;
    }
    b2b_final(S, out + 64);
  }
  const uint8_t *p = outs_start;
  {
    b2b_state outS;
    b2b_init_keyed(outS, KEY_SIGNING, 22);
    for (uint32_t i = 0; i < tx.n_outputs; i++) {
      blob_output o;
      blob_output_at(p, o);
      hash_output_fields(outS, o, tx.version);
      p = o.end;
    }
    b2b_final(outS, out + 96);
  }

  /* payload hash: ZERO when native subnetwork and empty payload */
  int native = 1;
  for (int i = 0; i < 20; i++) native &= (tx.subnetwork_id[i] == 0);
  if (native && tx.payload_len == 0) {
    for (int i = 0; i < 32; i++) out[128 + i] = 0;
  } else {
    b2b_state pS;
    b2b_init_keyed(pS, KEY_SIGNING, 22);
    b2b_update_u64(pS, tx.payload_len);
    b2b_update(pS, tx.payload, tx.payload_len);
    b2b_final(pS, out + 128);
  }
}

/* one sighash/verify-tuple assembly job */
struct kv_job {
  uint32_t tx_index;
  uint32_t input_off;   /* byte offset of the input record in the blob */
  uint32_t input_index;
  uint32_t sig_off;     /* byte offset of the 64B signature */
  uint32_t pk_off;      /* byte offset of the pubkey (32B xonly / 33B) */
  uint8_t hash_type;
  uint8_t ecdsa;
  uint16_t _pad;
};

/* Compute the 32B signing hash for one job (the shared core of the assemble
 * and msg-only kernels) ⇔ calc_schnorr/ecdsa_signature_hash
 * (sighash.rs:245-292). */
__device__ __forceinline__ static void kv_compute_sighash_msg(
    const uint8_t *__restrict__ blob, const uint8_t *__restrict__ subhashes,
    const kv_job job, uint8_t msg[32]) {
  blob_tx tx;
  blob_tx_at(blob, job.tx_index, tx);
  blob_input in;
  blob_input_at(blob + job.input_off, in);
  const uint8_t *sub = subhashes + (size_t)job.tx_index * SUBHASH_STRIDE;
  uint8_t ht = job.hash_type;
  uint8_t m = ht & KV_SIGHASH_MASK;
  int acp = (ht & KV_SIGHASH_ACP) != 0;
  static const uint8_t ZERO32[32] = {0};

  b2b_state S;
  b2b_init_keyed(S, KEY_SIGNING, 22);
  b2b_update_u16(S, tx.version);
  b2b_update(S, acp ? ZERO32 : sub, 32); /* prevouts */
  b2b_update(S, (m == KV_SIGHASH_SINGLE || acp || m == KV_SIGHASH_NONE) ? ZERO32
                                                                        : sub + 32,
             32); /* sequences */
  if (tx.version < 1) b2b_update(S, acp ? ZERO32 : sub + 64, 32); /* sigops */
  /* outpoint */
  b2b_update(S, in.prev_tx_id, 32);
  b2b_update_u32(S, in.prev_index);
  /* utxo spk */
  b2b_update_u16(S, in.utxo_spk_version);
  b2b_update_u64(S, in.utxo_spk_len);
  b2b_update(S, in.utxo_spk, in.utxo_spk_len);
  b2b_update_u64(S, in.utxo_amount);
  b2b_update_u64(S, in.sequence);
  if (tx.version < 1) {
    uint8_t sop = in.commit_kind == 0 ? (uint8_t)in.commit_value : 0;
    b2b_update(S, &sop, 1);
  }
  /* outputs hash by type */
  if (m == KV_SIGHASH_NONE) {
    b2b_update(S, ZERO32, 32);
  } else if (m == KV_SIGHASH_SINGLE) {
    if (job.input_index >= tx.n_outputs) {
      b2b_update(S, ZERO32, 32);
    } else {
      /* walk: inputs then outputs to the target index */
      const uint8_t *p = tx.inputs0;
      for (uint32_t i = 0; i < tx.n_inputs; i++) {
        blob_input w;
        blob_input_at(p, w);
        p = w.end;
      }
      blob_output o;
      for (uint32_t i = 0;; i++) {
        blob_output_at(p, o);
        if (i == job.input_index) break;
        p = o.end;
      }
      b2b_state oS;
      b2b_init_keyed(oS, KEY_SIGNING, 22);
      hash_output_fields(oS, o, tx.version);
      uint8_t oh[32];
      b2b_final(oS, oh);
      b2b_update(S, oh, 32);
    }
  } else {
    b2b_update(S, sub + 96, 32);
  }
  b2b_update_u64(S, tx.lock_time);
  b2b_update(S, tx.subnetwork_id, 20);
  b2b_update_u64(S, tx.gas);
  b2b_update(S, sub + 128, 32); /* payload hash */
  b2b_update(S, &ht, 1);
  b2b_final(S, msg);

  if (job.ecdsa) {
    /* calc_ecdsa_signature_hash: SHA256(domain_hash ‖ schnorr_hash) */
    uint32_t h[8] = {0x6a09e667, 0xbb67ae85, 0x3c6ef372, 0xa54ff53a,
                     0x510e527f, 0x9b05688c, 0x1f83d9ab, 0x5be0cd19};
    uint32_t w[16];
#pragma unroll
    for (int i = 0; i < 8; i++) w[i] = be32(ECDSA_DOMAIN_HASH + 4 * i);
#pragma unroll
    for (int i = 0; i < 8; i++) w[8 + i] = be32(msg + 4 * i);
    sha256_compress(h, w);
    w[0] = 0x80000000u;
#pragma unroll
    for (int i = 1; i < 15; i++) w[i] = 0;
    w[15] = 64 * 8;
    sha256_compress(h, w);
#pragma unroll
    for (int i = 0; i < 8; i++) {
      msg[4 * i] = (uint8_t)(h[i] >> 24);
      msg[4 * i + 1] = (uint8_t)(h[i] >> 16);
      msg[4 * i + 2] = (uint8_t)(h[i] >> 8);
      msg[4 * i + 3] = (uint8_t)h[i];
    }
  }
}

extern "C" __global__ void kv_sighash_assemble_kernel(
    const uint8_t *__restrict__ blob, const uint8_t *__restrict__ subhashes,
    const kv_job *__restrict__ jobs, uint32_t n_jobs,
    uint8_t *__restrict__ schnorr_tuples, uint8_t *__restrict__ ecdsa_tuples) {
  uint32_t ji = blockIdx.x * blockDim.x + threadIdx.x;
  if (ji >= n_jobs) return;
  kv_job job = jobs[ji];
  uint8_t msg[32];
  kv_compute_sighash_msg(blob, subhashes, job, msg);
  if (job.ecdsa) {
    uint8_t *t = ecdsa_tuples + (size_t)ji * 132;
    for (int i = 0; i < 64; i++) t[i] = blob[job.sig_off + i];
    for (int i = 0; i < 33; i++) t[64 + i] = blob[job.pk_off + i];
    for (int i = 0; i < 32; i++) t[97 + i] = msg[i];
    t[129] = t[130] = t[131] = 0;
  } else {
    uint8_t *t = schnorr_tuples + (size_t)ji * 128;
    for (int i = 0; i < 64; i++) t[i] = blob[job.sig_off + i];
    for (int i = 0; i < 32; i++) t[64 + i] = blob[job.pk_off + i];
    for (int i = 0; i < 32; i++) t[96 + i] = msg[i];
  }
}

/* msg-only variant for interpreter-collected verify requests: sig/pk come
 * from the SCRIPT STACK (host-prefilled in the tuple buffer), only the
 * signing hash is computed here. job.sig_off is repurposed as the TUPLE SLOT:
 * the hash lands at out + sig_off*stride + msg_off (not every tuple needs a
 * sighash — literal-msg requests have none, so slots are sparse in jobs). */
extern "C" __global__ void kv_sighash_msg_kernel(
    const uint8_t *__restrict__ blob, const uint8_t *__restrict__ subhashes,
    const kv_job *__restrict__ jobs, uint32_t n_jobs, uint8_t *__restrict__ out,
    uint32_t stride, uint32_t msg_off) {
  uint32_t ji = blockIdx.x * blockDim.x + threadIdx.x;
  if (ji >= n_jobs) return;
  kv_job job = jobs[ji];
  uint8_t msg[32];
  kv_compute_sighash_msg(blob, subhashes, job, msg);
  uint8_t *t = out + (size_t)job.sig_off * stride + msg_off;
  for (int i = 0; i < 32; i++) t[i] = msg[i];
}

/* ---------------- MuHash ---------------- */

/* one element job: serialize a utxo, hash, expand to a U3072 element */
struct kv_elem_job {
  uint32_t tx_index;
  uint32_t rec_off;     /* input record offset (spend) or output record (create) */
  uint32_t out_index;   /* for create: outpoint index */
  uint8_t is_create;    /* 1 = created utxo (numerator), 0 = spent (denominator) */
  uint8_t is_coinbase;  /* tx is_coinbase (create entries) */
  uint16_t _pad;
  uint64_t block_daa_score;
};

extern "C" __global__ void kv_muhash_element_kernel(const uint8_t *__restrict__ blob,
                                                    const kv_elem_job *__restrict__ jobs,
                                                    uint32_t n_jobs,
                                                    uint64_t *__restrict__ elements) {
  uint32_t ji = blockIdx.x * blockDim.x + threadIdx.x;
  if (ji >= n_jobs) return;
  kv_elem_job job = jobs[ji];
  blob_tx tx;
  blob_tx_at(blob, job.tx_index, tx);
  b2b_state S;
  b2b_init_keyed(S, KEY_MUHASH_ELEM, 13);
  if (job.is_create) {
    blob_output o;
    blob_output_at(blob + job.rec_off, o);
    b2b_update(S, tx.tx_id, 32);
    b2b_update_u32(S, job.out_index);
    b2b_update_u64(S, job.block_daa_score);
    b2b_update_u64(S, o.value);
    uint8_t cb = job.is_coinbase ? 1 : 0;
    b2b_update(S, &cb, 1);
    b2b_update_u16(S, o.spk_version);
    b2b_update_u64(S, o.spk_len);
    b2b_update(S, o.spk, o.spk_len);
    if (o.has_covenant) b2b_update(S, o.cov_id, 32);
  } else {
    blob_input in;
    blob_input_at(blob + job.rec_off, in);
    b2b_update(S, in.prev_tx_id, 32);
    b2b_update_u32(S, in.prev_index);
    b2b_update_u64(S, in.utxo_daa_score);
    b2b_update_u64(S, in.utxo_amount);
    uint8_t cb = in.utxo_is_coinbase ? 1 : 0;
    b2b_update(S, &cb, 1);
    b2b_update_u16(S, in.utxo_spk_version);
    b2b_update_u64(S, in.utxo_spk_len);
    b2b_update(S, in.utxo_spk, in.utxo_spk_len);
    if (in.utxo_has_cov) b2b_update(S, in.utxo_cov_id, 32);
  }
  uint8_t hash[32];
  b2b_final(S, hash);
  uint32_t key[8];
#pragma unroll
  for (int i = 0; i < 8; i++)
    key[i] = (uint32_t)hash[4 * i] | ((uint32_t)hash[4 * i + 1] << 8) |
             ((uint32_t)hash[4 * i + 2] << 16) | ((uint32_t)hash[4 * i + 3] << 24);
  uint8_t stream[64];
  uint64_t *out = elements + (size_t)ji * KVU_LIMBS;
  for (int blk = 0; blk < 6; blk++) {
    chacha20_block(key, blk, stream);
#pragma unroll
    for (int i = 0; i < 8; i++) {
      uint64_t w = 0;
#pragma unroll
      for (int j = 0; j < 8; j++) w |= (uint64_t)stream[8 * i + j] << (8 * j);
      out[blk * 8 + i] = w;
    }
  }
}

/* strided modular-multiply chains: thread t multiplies elements t, t+stride, …
 * into an accumulator, writes partials[t]. Host iterates until one remains. */
extern "C" __global__ void kv_u3072_reduce_kernel(const uint64_t *__restrict__ elements,
                                                  uint32_t n, uint32_t stride,
                                                  uint64_t *__restrict__ partials) {
  uint32_t t = blockIdx.x * blockDim.x + threadIdx.x;
  if (t >= stride) return;
  u3072 acc;
  u3072_one(acc);
  int any = 0;
  for (uint32_t i = t; i < n; i += stride) {
    u3072 e;
    for (int k = 0; k < KVU_LIMBS; k++) e.l[k] = elements[(size_t)i * KVU_LIMBS + k];
    if (!any) {
      acc = e; /* the one-is-identity shortcut (u3072.rs:91-99) */
      any = 1;
    } else {
      u3072 r;
      u3072_mulmod(r, acc, e);
      acc = r;
    }
  }
  for (int k = 0; k < KVU_LIMBS; k++) partials[(size_t)t * KVU_LIMBS + k] = acc.l[k];
}


/* hashing::tx::hash ⇔ consensus/core/src/hashing/tx.rs:20-24 (crescendo):
 * keyed blake2b-256("TransactionHash") over write_transaction with signature
 * scripts, per-version commit fields, payload var-bytes and the mass rule
 * (v0: mass hashed only when > 0; v1+: always) — one tx per lane. */
extern "C" __global__ void kv_tx_hash_kernel(const uint8_t *__restrict__ blob,
                                             uint32_t n_txs,
                                             uint8_t *__restrict__ hashes_out) {
  uint32_t t = blockIdx.x * blockDim.x + threadIdx.x;
  if (t >= n_txs) return;
  blob_tx tx;
  blob_tx_at(blob, t, tx);
  uint64_t storage_mass = bl_rd64(tx.base + 48);
  static const uint8_t KEY[] = "TransactionHash";
  b2b_state S;
  b2b_init_keyed(S, KEY, sizeof(KEY) - 1);
  b2b_update_u16(S, tx.version);
  b2b_update_u64(S, tx.n_inputs);
  const uint8_t *p = tx.inputs0;
  for (uint32_t i = 0; i < tx.n_inputs; i++) {
    blob_input in;
    blob_input_at(p, in);
    b2b_update(S, in.prev_tx_id, 32);
    b2b_update_u32(S, in.prev_index);
    b2b_update_u64(S, in.sig_script_len);
    b2b_update(S, in.sig_script, in.sig_script_len);
    if (tx.version < 1) {
      uint8_t soc = in.commit_kind == 0 ? (uint8_t)in.commit_value : 0;
      b2b_update(S, &soc, 1);
    }
    b2b_update_u64(S, in.sequence);
    if (tx.version >= 1)
      b2b_update_u16(S, in.commit_kind == 1 ? in.commit_value : 0);
    p = in.end;
  }
  b2b_update_u64(S, tx.n_outputs);
  for (uint32_t i = 0; i < tx.n_outputs; i++) {
    blob_output o;
    blob_output_at(p, o);
    b2b_update_u64(S, o.value);
    b2b_update_u16(S, o.spk_version);
    b2b_update_u64(S, o.spk_len);
    b2b_update(S, o.spk, o.spk_len);
    if (tx.version >= 1) {
      uint8_t hc = o.has_covenant ? 1 : 0;
      b2b_update(S, &hc, 1);
      if (o.has_covenant) {
        b2b_update_u16(S, o.cov_auth_input);
        b2b_update(S, o.cov_id, 32);
      }
    }
    p = o.end;
  }
  b2b_update_u64(S, tx.lock_time);
  b2b_update(S, tx.subnetwork_id, 20);
  b2b_update_u64(S, tx.gas);
  b2b_update_u64(S, tx.payload_len);
  b2b_update(S, tx.payload, tx.payload_len);
  if (tx.version < 1) {
    if (storage_mass > 0) b2b_update_u64(S, storage_mass);
  } else {
    b2b_update_u64(S, storage_mass);
  }
  b2b_final(S, hashes_out + (size_t)t * 32);
}

} // namespace kv

/* ---------------- wave-cooperative U3072 mulmod ----------------
 * One 3072-bit value per WAVE: lane l (l < 48) holds limb l in a register —
 * no scratch at all (the per-thread mulmod above is scratch-latency bound).
 * Column k of a·b mod (2^3072 − C):
 *   S_k = Σ_i a_i · b_{(k−i) mod 48} · (C if i > k else 1)
 * accumulated per lane in a 192-bit (lo,mid,hi) register triple via cross-lane
 * shfl broadcasts, then redistributed (mid→k+1, hi→k+2, wrapping ×C) and
 * carry-propagated lane-to-lane until quiescent. CDNA4-native: wave64 shfl is
 * ds_bpermute, wholly in-register. */

__device__ inline uint64_t u3072_wave_mulmod(uint64_t a_limb, uint64_t b_limb) {
  const uint64_t C = KVU_PRIME_DIFF;
  int lane = threadIdx.x & 63;
  int k = lane < 48 ? lane : 0;
  uint64_t lo = 0, mid = 0, hi = 0;
#pragma unroll 1
  for (int i = 0; i < 48; i++) {
    uint64_t ai = __shfl((long long)a_limb, i);
    int j = k - i;
    uint64_t w = 1;
    if (j < 0) {
      j += 48;
      w = C;
    }
    uint64_t bj = __shfl((long long)b_limb, j);
    unsigned __int128 p = (unsigned __int128)ai * bj;
    uint64_t plo = (uint64_t)p, phi = (uint64_t)(p >> 64);
    unsigned __int128 t = (unsigned __int128)plo * w;   /* ≤ 2^85 when w=C */
    unsigned __int128 t2 = (unsigned __int128)phi * w;
    unsigned __int128 c = (unsigned __int128)lo + (uint64_t)t;
    lo = (uint64_t)c;
    c = (c >> 64) + (uint64_t)(t >> 64) + (unsigned __int128)mid + (uint64_t)t2;
    mid = (uint64_t)c;
    hi += (uint64_t)(c >> 64) + (uint64_t)(t2 >> 64);
  }
  /* redistribute mid→column k+1, hi→column k+2 (wrap ×C past column 47) */
  uint64_t mid_in = __shfl((long long)mid, (k + 47) % 48);
  uint64_t hi_in = __shfl((long long)hi, (k + 46) % 48);
  uint64_t w1 = (k == 0) ? C : 1;
  uint64_t w2 = (k <= 1) ? C : 1;
  unsigned __int128 s = (unsigned __int128)lo + (unsigned __int128)mid_in * w1 +
                        (unsigned __int128)hi_in * w2;
  uint64_t limb = (uint64_t)s;
  uint64_t carry = (uint64_t)(s >> 64);
  /* lane-to-lane carry propagation until quiescent (values shrink fast) */
  while (__any(lane < 48 && carry != 0)) {
    uint64_t cin = __shfl((long long)carry, (k + 47) % 48);
    if (k == 0) {
      unsigned __int128 cc = (unsigned __int128)cin * C;
      s = (unsigned __int128)limb + (uint64_t)cc;
      carry = (uint64_t)(s >> 64) + (uint64_t)(cc >> 64);
    } else {
      s = (unsigned __int128)limb + cin;
      carry = (uint64_t)(s >> 64);
    }
    limb = (uint64_t)s;
  }
  return limb;
}

/* one WAVE per partial: wave w chains elements w, w+stride, … */
extern "C" __global__ void kv_u3072_reduce_wave_kernel(
    const uint64_t *__restrict__ elements, uint32_t n, uint32_t stride,
    uint64_t *__restrict__ partials) {
  uint32_t wave = (blockIdx.x * blockDim.x + threadIdx.x) / 64;
  int lane = threadIdx.x & 63;
  if (wave >= stride) return;
  uint64_t acc = 0;
  int any = 0;
  for (uint32_t i = wave; i < n; i += stride) {
    uint64_t e = lane < 48 ? elements[(size_t)i * KVU_LIMBS + lane] : 0;
    if (!any) {
      acc = e;
      any = 1;
    } else {
      acc = u3072_wave_mulmod(acc, e);
    }
  }
  if (!any) acc = (lane == 0) ? 1 : 0; /* identity */
  if (lane < 48) partials[(size_t)wave * KVU_LIMBS + lane] = acc;
}
