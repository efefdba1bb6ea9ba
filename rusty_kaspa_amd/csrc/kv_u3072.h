/* PRODUCT U3072 arithmetic mod 2^3072 - 1103717 (host + device).
 *
 * Engine-side implementation of the MuHash group operation
 * (⇔ crypto/muhash/src/u3072.rs). Structurally independent from the oracle's
 * reduce-as-you-go restatement: full 96-limb schoolbook product, then a
 * two-pass fold of the high half through 2^3072 ≡ PRIME_DIFF. Values are kept
 * in [0, 2^3072); canonicalization (mod p) happens at serialize/finalize.
 */
#ifndef KV_U3072_H
#define KV_U3072_H

#include <stdint.h>
#include <string.h>

#ifdef __HIPCC__
#include <hip/hip_runtime.h>
#define KV_HD __host__ __device__
#else
#define KV_HD
#endif

namespace kv {

#define KVU_LIMBS 48
#define KVU_PRIME_DIFF 1103717ULL

struct u3072 {
  uint64_t l[KVU_LIMBS];
};

KV_HD inline void u3072_one(u3072 &a) {
  for (int i = 0; i < KVU_LIMBS; i++) a.l[i] = 0;
  a.l[0] = 1;
}

KV_HD inline uint64_t kvu_addc(uint64_t a, uint64_t b, uint64_t &carry) {
  uint64_t s = a + b;
  uint64_t c1 = s < a;
  uint64_t s2 = s + carry;
  carry = c1 + (s2 < s);
  return s2;
}

#ifdef __HIPCC__
KV_HD inline uint64_t kvu_umulhi(uint64_t a, uint64_t b) {
#ifdef __HIP_DEVICE_COMPILE__
  return __umul64hi(a, b);
#else
  return (uint64_t)(((unsigned __int128)a * b) >> 64);
#endif
}
#else
inline uint64_t kvu_umulhi(uint64_t a, uint64_t b) {
  return (uint64_t)(((unsigned __int128)a * b) >> 64);
}
#endif

/* t[96] = a * b, row-wise schoolbook */
KV_HD inline void u3072_mul_full(uint64_t t[2 * KVU_LIMBS], const u3072 &a,
                                 const u3072 &b) {
  for (int i = 0; i < 2 * KVU_LIMBS; i++) t[i] = 0;
  for (int i = 0; i < KVU_LIMBS; i++) {
    uint64_t carry = 0;
    uint64_t ai = a.l[i];
    for (int j = 0; j < KVU_LIMBS; j++) {
      uint64_t lo = ai * b.l[j];
      uint64_t hi = kvu_umulhi(ai, b.l[j]);
      uint64_t c = 0;
      t[i + j] = kvu_addc(t[i + j], lo, c);
      uint64_t c2 = 0;
      t[i + j] = kvu_addc(t[i + j], carry, c2);
      carry = hi + c + c2;
    }
    t[i + KVU_LIMBS] = carry;
  }
}

/* r = t mod-ish: fold hi*C twice; result < 2^3072 (not canonical) */
KV_HD inline void u3072_fold(u3072 &r, const uint64_t t[2 * KVU_LIMBS]) {
  /* pass 1: s[49] = t_lo + C * t_hi */
  uint64_t s[KVU_LIMBS + 1];
  uint64_t cA = 0, cB = 0;
  uint64_t prev_hi = 0;
  for (int i = 0; i < KVU_LIMBS; i++) {
    uint64_t hi_limb = t[KVU_LIMBS + i];
    uint64_t lo = hi_limb * KVU_PRIME_DIFF;
    uint64_t hi = kvu_umulhi(hi_limb, KVU_PRIME_DIFF);
    uint64_t v = kvu_addc(t[i], lo, cA);
    s[i] = kvu_addc(v, prev_hi, cB);
    prev_hi = hi;
  }
  s[KVU_LIMBS] = cA + cB + prev_hi; /* < 2^22 (C is 21 bits) */
  /* pass 2: r = s[0..47] + C * s[48] */
  uint64_t c = 0;
  uint64_t add = s[KVU_LIMBS] * KVU_PRIME_DIFF; /* < 2^43 */
  r.l[0] = kvu_addc(s[0], add, c);
  for (int i = 1; i < KVU_LIMBS; i++) r.l[i] = kvu_addc(s[i], 0, c);
  /* carry out → wrapped 2^3072 once more; value tiny */
  if (c) {
    uint64_t c2 = 0;
    r.l[0] = kvu_addc(r.l[0], KVU_PRIME_DIFF, c2);
    for (int i = 1; i < KVU_LIMBS && c2; i++) r.l[i] = kvu_addc(r.l[i], 0, c2);
  }
}

KV_HD inline void u3072_mulmod(u3072 &r, const u3072 &a, const u3072 &b) {
  uint64_t t[2 * KVU_LIMBS];
  u3072_mul_full(t, a, b);
  u3072_fold(r, t);
}

/* canonicalize into [0, p) */
KV_HD inline void u3072_canon(u3072 &a) {
  /* a >= p ⇔ a >= 2^3072 - PRIME_DIFF ⇔ l[0] > max-PRIME_DIFF && rest max */
  int ge = a.l[0] > (~0ULL - KVU_PRIME_DIFF);
  for (int i = 1; i < KVU_LIMBS && ge; i++) ge = (a.l[i] == ~0ULL);
  if (ge) {
    /* a -= p  ==  a += PRIME_DIFF (mod 2^3072) */
    uint64_t c = 0;
    a.l[0] = kvu_addc(a.l[0], KVU_PRIME_DIFF, c);
    for (int i = 1; i < KVU_LIMBS; i++) a.l[i] = kvu_addc(a.l[i], 0, c);
  }
}

} // namespace kv

#endif /* KV_U3072_H */
