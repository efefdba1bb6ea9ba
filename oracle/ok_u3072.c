/* ORACLE (test infrastructure only).
 *
 * U3072 arithmetic mod 2^3072 - 1103717, restating crypto/muhash/src/u3072.rs
 * (48 × u64 limbs, reduce-as-you-go multiply; the algorithm originates in
 * Bitcoin Core's MuHash3072). Inverse restates u3072.rs:157-183 +
 * math/src/uint.rs mod_inverse (extended binary GCD).
 */
#include "oracle.h"
#include <string.h>

#define LIMBS OK_U3072_LIMBS
#define PRIME_DIFF 1103717ULL

typedef unsigned __int128 u128;

void ok_u3072_one(uint64_t a[LIMBS]) {
  memset(a, 0, LIMBS * 8);
  a[0] = 1;
}

int ok_u3072_is_overflow(const uint64_t a[LIMBS]) {
  /* u3072.rs:48-56 */
  if (a[0] <= UINT64_MAX - PRIME_DIFF) return 0;
  for (int i = 1; i < LIMBS; i++)
    if (a[i] != UINT64_MAX) return 0;
  return 1;
}

void ok_u3072_full_reduce(uint64_t a[LIMBS]) {
  /* u3072.rs:77-88: add PRIME_DIFF with carry propagation (wrapping) */
  uint64_t low = PRIME_DIFF, high = 0;
  for (int i = 0; i < LIMBS; i++) {
    uint64_t limb = a[i];
    uint64_t nlow = low + limb;
    int ov1 = nlow < limb;
    uint64_t nhigh = high + (uint64_t)ov1;
    int ov2 = nhigh < high;
    a[i] = nlow;
    low = nhigh;
    high = (uint64_t)ov2;
  }
}

/* helpers restating u3072.rs:240-287 */
static inline void mul_wide(uint64_t a, uint64_t b, uint64_t *lo, uint64_t *hi) {
  u128 t = (u128)a * b;
  *lo = (uint64_t)t;
  *hi = (uint64_t)(t >> 64);
}

/* [c0,c1,c2] = [c0,c1] + n * [d0,d1,d2] (c2 starts 0) — u3072.rs:251-262 */
static inline void mulnadd3(uint64_t *c0, uint64_t *c1, uint64_t *c2, uint64_t d0,
                            uint64_t d1, uint64_t d2, uint64_t n) {
  u128 t = (u128)d0 * n + *c0;
  uint64_t r0 = (uint64_t)t;
  t >>= 64;
  t += (u128)d1 * n + *c1;
  uint64_t r1 = (uint64_t)t;
  t >>= 64;
  uint64_t r2 = (uint64_t)t + d2 * n;
  *c0 = r0;
  *c1 = r1;
  *c2 = r2;
}

/* [low,high,carry] += a*b — u3072.rs:267-274 */
static inline void muladd3(uint64_t a, uint64_t b, uint64_t *low, uint64_t *high,
                           uint64_t *carry) {
  uint64_t tl, th;
  mul_wide(a, b, &tl, &th);
  uint64_t nlow = *low + tl;
  th += (nlow < tl);
  uint64_t nhigh = *high + th;
  *carry += (nhigh < th);
  *low = nlow;
  *high = nhigh;
}

/* [low,high] *= n — u3072.rs:279-287 */
static inline void muln2(uint64_t *low, uint64_t *high, uint64_t n) {
  u128 t = (u128)(*low) * n;
  uint64_t lo = (uint64_t)t;
  t >>= 64;
  t += (u128)(*high) * n;
  *low = lo;
  *high = (uint64_t)t;
}

void ok_u3072_mul(uint64_t a[LIMBS], const uint64_t b[LIMBS]) {
  /* u3072.rs:90-154, including the one-is-identity shortcut (:97-100) */
  int is_one = (a[0] == 1);
  for (int i = 1; is_one && i < LIMBS; i++)
    if (a[i] != 0) is_one = 0;
  if (is_one) {
    memcpy(a, b, LIMBS * 8);
    return;
  }

  uint64_t carry_low = 0, carry_high = 0, carry_highest = 0;
  uint64_t tmp[LIMBS];

  for (int j = 0; j < LIMBS - 1; j++) {
    uint64_t low, high, carry = 0;
    mul_wide(a[j + 1], b[LIMBS - 1], &low, &high);
    for (int i = 2 + j; i < LIMBS; i++)
      muladd3(a[i], b[LIMBS + j - i], &low, &high, &carry);
    mulnadd3(&carry_low, &carry_high, &carry_highest, low, high, carry, PRIME_DIFF);
    for (int i = 0; i <= j; i++)
      muladd3(a[i], b[j - i], &carry_low, &carry_high, &carry_highest);
    tmp[j] = carry_low;
    carry_low = carry_high;
    carry_high = carry_highest;
    carry_highest = 0;
  }

  for (int i = 0; i < LIMBS; i++)
    muladd3(a[i], b[LIMBS - 1 - i], &carry_low, &carry_high, &carry_highest);

  tmp[LIMBS - 1] = carry_low;
  carry_low = carry_high;
  carry_high = carry_highest;

  muln2(&carry_low, &carry_high, PRIME_DIFF);
  for (int i = 0; i < LIMBS; i++) {
    uint64_t nlow = carry_low + tmp[i];
    int ov1 = nlow < tmp[i];
    uint64_t nhigh = carry_high + (uint64_t)ov1;
    int ov2 = nhigh < carry_high;
    a[i] = nlow;
    carry_low = nhigh;
    carry_high = (uint64_t)ov2;
  }

  if (ok_u3072_is_overflow(a)) ok_u3072_full_reduce(a);
  if (carry_low != 0) ok_u3072_full_reduce(a);
}

/* ---- 3072-bit helpers for the inverse (binary extended GCD, math/src/uint.rs:347
 * mod_inverse semantics: a^-1 mod (2^3072 - 1103717)) ---- */

static int big_is_zero(const uint64_t *a) {
  for (int i = 0; i < LIMBS; i++)
    if (a[i]) return 0;
  return 1;
}

static int big_cmp(const uint64_t *a, const uint64_t *b) {
  for (int i = LIMBS - 1; i >= 0; i--) {
    if (a[i] < b[i]) return -1;
    if (a[i] > b[i]) return 1;
  }
  return 0;
}

static void big_sub(uint64_t *a, const uint64_t *b) { /* a -= b (a >= b) */
  uint64_t borrow = 0;
  for (int i = 0; i < LIMBS; i++) {
    uint64_t bi = b[i] + borrow;
    uint64_t nb = (bi < borrow) || (a[i] < bi);
    a[i] = a[i] - bi;
    borrow = nb;
  }
}

static void big_shr1(uint64_t *a) {
  for (int i = 0; i < LIMBS; i++) {
    uint64_t hi = (i + 1 < LIMBS) ? (a[i + 1] & 1) : 0;
    a[i] = (a[i] >> 1) | (hi << 63);
  }
}

/* a = (a + b) mod m, where a,b < m */
static void big_addmod(uint64_t *a, const uint64_t *b, const uint64_t *m) {
  uint64_t carry = 0;
  for (int i = 0; i < LIMBS; i++) {
    uint64_t s = a[i] + carry;
    uint64_t c1 = s < carry;
    uint64_t s2 = s + b[i];
    uint64_t c2 = s2 < b[i];
    a[i] = s2;
    carry = c1 | c2;
  }
  if (carry || big_cmp(a, m) >= 0) big_sub(a, m);
}

static void big_prime(uint64_t *p) {
  for (int i = 0; i < LIMBS; i++) p[i] = UINT64_MAX;
  p[0] -= PRIME_DIFF - 1;
}

/* half of x mod m (m odd): x/2 if even else (x+m)/2 */
static void big_halfmod(uint64_t *x, const uint64_t *m) {
  if (x[0] & 1) {
    uint64_t carry = 0;
    for (int i = 0; i < LIMBS; i++) {
      uint64_t s = x[i] + carry;
      uint64_t c1 = s < carry;
      uint64_t s2 = s + m[i];
      uint64_t c2 = s2 < m[i];
      x[i] = s2;
      carry = c1 | c2;
    }
    big_shr1(x);
    if (carry) x[LIMBS - 1] |= 1ULL << 63; /* the add overflowed into bit 3072 */
  } else {
    big_shr1(x);
  }
}

static void u3072_inverse(const uint64_t in[LIMBS], uint64_t out[LIMBS]) {
  /* u3072.rs:157-173: reduce first; inverse(0) = 0 */
  uint64_t a[LIMBS];
  memcpy(a, in, LIMBS * 8);
  if (ok_u3072_is_overflow(a)) ok_u3072_full_reduce(a);
  if (big_is_zero(a)) {
    memset(out, 0, LIMBS * 8);
    return;
  }
  /* binary extended gcd: invariants u*x ≡ a, v*x ≡ b (mod p) won't hold exactly in
   * that form; we use the standard odd-modulus algorithm:
   *   u=a, v=p, x1=1, x2=0
   *   while u!=1 and v!=1:
   *     while u even: u/=2; x1 = half_mod(x1)
   *     while v even: v/=2; x2 = half_mod(x2)
   *     if u>=v: u-=v; x1=(x1-x2) mod p
   *     else:    v-=u; x2=(x2-x1) mod p
   *   result = (u==1) ? x1 : x2 */
  uint64_t p[LIMBS], u[LIMBS], v[LIMBS], x1[LIMBS], x2[LIMBS], tmp[LIMBS];
  big_prime(p);
  memcpy(u, a, LIMBS * 8);
  memcpy(v, p, LIMBS * 8);
  memset(x1, 0, LIMBS * 8);
  x1[0] = 1;
  memset(x2, 0, LIMBS * 8);

  uint64_t one[LIMBS];
  memset(one, 0, LIMBS * 8);
  one[0] = 1;

  while (big_cmp(u, one) != 0 && big_cmp(v, one) != 0) {
    while (!(u[0] & 1)) {
      big_shr1(u);
      big_halfmod(x1, p);
    }
    while (!(v[0] & 1)) {
      big_shr1(v);
      big_halfmod(x2, p);
    }
    if (big_cmp(u, v) >= 0) {
      big_sub(u, v);
      /* x1 = (x1 - x2) mod p */
      if (big_cmp(x1, x2) >= 0) {
        big_sub(x1, x2);
      } else {
        memcpy(tmp, p, LIMBS * 8);
        big_sub(tmp, x2);
        big_addmod(x1, tmp, p); /* x1 + (p - x2) */
      }
    } else {
      big_sub(v, u);
      if (big_cmp(x2, x1) >= 0) {
        big_sub(x2, x1);
      } else {
        memcpy(tmp, p, LIMBS * 8);
        big_sub(tmp, x1);
        big_addmod(x2, tmp, p);
      }
    }
  }
  memcpy(out, (big_cmp(u, one) == 0) ? x1 : x2, LIMBS * 8);
}

void ok_u3072_div(uint64_t a[LIMBS], const uint64_t b[LIMBS]) {
  /* u3072.rs:175-191 */
  uint64_t binv[LIMBS];
  uint64_t bb[LIMBS];
  memcpy(bb, b, LIMBS * 8);
  if (ok_u3072_is_overflow(bb)) ok_u3072_full_reduce(bb);
  u3072_inverse(bb, binv);
  if (ok_u3072_is_overflow(a)) ok_u3072_full_reduce(a);
  ok_u3072_mul(a, binv);
  if (ok_u3072_is_overflow(a)) ok_u3072_full_reduce(a);
}

/* ---------------- MuHash element & finalize (crypto/muhash/src/lib.rs) -------- */

static const uint8_t MUHASH_ELEMENT_KEY[] = "MuHashElement";
static const uint8_t MUHASH_FINALIZE_KEY[] = "MuHashFinalize";

void ok_muhash_element(const uint8_t *data, size_t len, uint64_t out[LIMBS]) {
  uint8_t hash[32], stream[384];
  ok_blake2b_keyed(MUHASH_ELEMENT_KEY, sizeof(MUHASH_ELEMENT_KEY) - 1, data, len, hash);
  ok_chacha20_block384(hash, stream);
  for (int i = 0; i < LIMBS; i++) {
    uint64_t w = 0;
    for (int j = 0; j < 8; j++) w |= (uint64_t)stream[8 * i + j] << (8 * j);
    out[i] = w;
  }
}

void ok_muhash_finalize(uint64_t num[LIMBS], uint64_t den[LIMBS], uint8_t out32[32]) {
  /* normalize (lib.rs:99-102) then serialize LE (lib.rs:105-108) then keyed hash */
  ok_u3072_div(num, den);
  ok_u3072_one(den);
  uint8_t ser[384];
  for (int i = 0; i < LIMBS; i++)
    for (int j = 0; j < 8; j++) ser[8 * i + j] = (uint8_t)(num[i] >> (8 * j));
  ok_blake2b_keyed(MUHASH_FINALIZE_KEY, sizeof(MUHASH_FINALIZE_KEY) - 1, ser, 384, out32);
}
