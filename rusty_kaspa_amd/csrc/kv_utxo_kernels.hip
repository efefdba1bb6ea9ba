/* MI355X-native GPU-resident UTXO set (gfx950, PRODUCT code).
 *
 * ⇔ the virtual UTXO set consulted by the populate step
 * (consensus/src/pipeline/virtual_processor/utxo_validation.rs:351-390 over
 * UtxoCollection = HashMap<TransactionOutpoint, UtxoEntry>,
 * consensus/core/src/utxo/utxo_collection.rs:5) and the diff application
 * (utxo_diff.rs:224 add_transaction: remove spent, add created).
 *
 * Open-addressing table in HBM, linear probing. Slot layout (SoA-of-AoS,
 * 112B per slot, 16B-aligned):
 *   state u32 (0 empty / 2 ready / 3 tombstone) + pad
 *   key   36B outpoint (tx_id 32 ‖ index u32) padded to 40
 *   value 64B: amount u64, daa_score u64, flags u16 (bit0 coinbase),
 *              spk_version u16, spk_len u32, spk[36] (inline; standard SPKs
 *              are ≤35B — longer scripts are rejected at the host API and
 *              documented as the round-2 arena extension)
 *
 * Outpoint tx-ids are keyed-BLAKE2b outputs (uniform), so the hash is simply
 * the first key word mixed with the output index. The probe path is the
 * HBM-random-access-bound row of SURVEY §8d: one ~112B line per probe.
 */
#include <hip/hip_runtime.h>
#include <stdint.h>

namespace kv {

struct utxo_slot {
  uint32_t state;
  uint32_t _pad;
  uint8_t key[40];   /* 36 used */
  uint8_t value[64];
};

#define KV_SLOT_EMPTY 0u
#define KV_SLOT_CLAIMED 1u
#define KV_SLOT_READY 2u
#define KV_SLOT_TOMB 3u

__device__ __forceinline__ uint64_t op_hash(const uint8_t *op36) {
  uint64_t h;
  /* outpoint tx_id is itself a keyed hash — first 8 bytes are uniform */
  memcpy(&h, op36, 8);
  uint32_t idx;
  memcpy(&idx, op36 + 32, 4);
  return h ^ (0x9e3779b97f4a7c15ULL * (idx + 1));
}

__device__ __forceinline__ int key_eq(const uint8_t *a, const uint8_t *b) {
  uint64_t x = 0;
#pragma unroll
  for (int i = 0; i < 4; i++) {
    uint64_t wa, wb;
    memcpy(&wa, a + 8 * i, 8);
    memcpy(&wb, b + 8 * i, 8);
    x |= wa ^ wb;
  }
  uint32_t ia, ib;
  memcpy(&ia, a + 32, 4);
  memcpy(&ib, b + 32, 4);
  return x == 0 && ia == ib;
}

/* one outpoint per lane; values are prepacked 64B records.
 *
 * Tombstone discipline: an upsert must probe its FULL chain (until an EMPTY
 * slot or a key match) before claiming anything — a tombstone left early in a
 * live key's chain by a prior remove must not be claimed until the key is
 * known absent, or the table would hold two READY slots for one outpoint and a
 * later remove would resurrect the stale one. We remember the first tombstone
 * seen and claim it only after reaching the chain's end without a match. */
extern "C" __global__ void kv_utxo_upsert_kernel(utxo_slot *table, uint64_t cap_mask,
                                                 const uint8_t *__restrict__ outpoints,
                                                 const uint8_t *__restrict__ values,
                                                 unsigned long long n,
                                                 int *__restrict__ fail_flag) {
  unsigned long long i = (unsigned long long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  const uint8_t *key = outpoints + i * 36;
  const uint8_t *val = values + i * 64;
  const uint64_t home = op_hash(key) & cap_mask;
  for (;;) { /* restart only on a lost claim race (rare) */
    uint64_t cand = ~0ULL; /* first tombstone seen this scan */
    uint64_t slot = home;
    int restart = 0;
    uint64_t probe = 0;
    for (; probe <= cap_mask; probe++, slot = (slot + 1) & cap_mask) {
      utxo_slot *s = &table[slot];
      uint32_t st = __hip_atomic_load(&s->state, __ATOMIC_ACQUIRE,
                                      __HIP_MEMORY_SCOPE_AGENT);
      /* wait for a concurrent claimer of this slot to publish its key */
      while (st == KV_SLOT_CLAIMED)
        st = __hip_atomic_load(&s->state, __ATOMIC_ACQUIRE, __HIP_MEMORY_SCOPE_AGENT);
      if (st == KV_SLOT_READY) {
        if (key_eq(s->key, key)) { /* overwrite value (upsert semantics) */
          for (int k = 0; k < 64; k++) s->value[k] = val[k];
          return;
        }
        continue;
      }
      if (st == KV_SLOT_TOMB) {
        if (cand == ~0ULL) cand = slot;
        continue; /* keep probing: the key may live further down the chain */
      }
      /* EMPTY: end of this key's chain — the key is absent. Claim the
       * remembered tombstone if any, else this empty slot. */
      {
        uint64_t tgt = (cand != ~0ULL) ? cand : slot;
        uint32_t expected = (cand != ~0ULL) ? KV_SLOT_TOMB : KV_SLOT_EMPTY;
        utxo_slot *t = &table[tgt];
        if (__hip_atomic_compare_exchange_strong(&t->state, &expected, KV_SLOT_CLAIMED,
                                                 __ATOMIC_ACQ_REL, __ATOMIC_ACQUIRE,
                                                 __HIP_MEMORY_SCOPE_AGENT)) {
          for (int k = 0; k < 36; k++) t->key[k] = key[k];
          for (int k = 0; k < 64; k++) t->value[k] = val[k];
          __hip_atomic_store(&t->state, KV_SLOT_READY, __ATOMIC_RELEASE,
                             __HIP_MEMORY_SCOPE_AGENT);
          return;
        }
        restart = 1; /* another lane took the slot: rescan from home */
        break;
      }
    }
    if (restart) continue;
    if (fail_flag) *fail_flag = 1; /* table full */
    return;
  }
}

extern "C" __global__ void kv_utxo_remove_kernel(utxo_slot *table, uint64_t cap_mask,
                                                 const uint8_t *__restrict__ outpoints,
                                                 unsigned long long n) {
  unsigned long long i = (unsigned long long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  const uint8_t *key = outpoints + i * 36;
  uint64_t slot = op_hash(key) & cap_mask;
  for (uint64_t probe = 0; probe <= cap_mask; probe++, slot = (slot + 1) & cap_mask) {
    utxo_slot *s = &table[slot];
    uint32_t st = s->state;
    if (st == KV_SLOT_EMPTY) return; /* not present */
    if (st == KV_SLOT_READY && key_eq(s->key, key)) {
      __hip_atomic_store(&s->state, KV_SLOT_TOMB, __ATOMIC_RELEASE,
                         __HIP_MEMORY_SCOPE_AGENT);
      return;
    }
  }
}

/* batched lookup: entry copied to out (64B) when found; found bitmap by wave
 * ballot. Launched after upserts complete (stream-ordered), so plain loads. */
extern "C" __global__ void kv_utxo_lookup_kernel(const utxo_slot *__restrict__ table,
                                                 uint64_t cap_mask,
                                                 const uint8_t *__restrict__ outpoints,
                                                 unsigned long long n,
                                                 uint8_t *__restrict__ entries_out,
                                                 unsigned long long *__restrict__ found) {
  unsigned long long i = (unsigned long long)blockIdx.x * blockDim.x + threadIdx.x;
  int hit = 0;
  if (i < n) {
    const uint8_t *key = outpoints + i * 36;
    uint64_t slot = op_hash(key) & cap_mask;
    for (uint64_t probe = 0; probe <= cap_mask; probe++, slot = (slot + 1) & cap_mask) {
      const utxo_slot *s = &table[slot];
      uint32_t st = s->state;
      if (st == KV_SLOT_EMPTY) break;
      if (st == KV_SLOT_READY && key_eq(s->key, key)) {
        uint8_t *out = entries_out + i * 64;
        for (int k = 0; k < 64; k++) out[k] = s->value[k];
        hit = 1;
        break;
      }
    }
  }
  unsigned long long mask = __ballot(hit);
  if ((threadIdx.x & 63) == 0 && i < n) found[i / 64] = mask;
}

/* gather out-of-line scripts from the arena: one job per BLOCK, 256 threads
 * striding the bytes (spans reach KV_UTXO_MAX_SPK = 10KB) */
struct arena_gather_job {
  uint32_t src_off; /* byte offset in the arena */
  uint32_t len;
  uint32_t dst_off; /* byte offset in out */
  uint32_t _pad;
};

extern "C" __global__ void kv_arena_gather_kernel(const uint8_t *__restrict__ arena,
                                                  const arena_gather_job *__restrict__ jobs,
                                                  uint32_t n_jobs,
                                                  uint8_t *__restrict__ out) {
  uint32_t j = blockIdx.x;
  if (j >= n_jobs) return;
  arena_gather_job job = jobs[j];
  for (uint32_t i = threadIdx.x; i < job.len; i += blockDim.x)
    out[job.dst_off + i] = arena[job.src_off + i];
}

} // namespace kv
