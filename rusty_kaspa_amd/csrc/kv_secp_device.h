/* MI355X-native secp256k1 device arithmetic (gfx950).
 *
 * PRODUCT code — the compute path of the engine. Written for the CDNA4
 * execution model: one signature per lane, 4×u64 limb field elements held in
 * VGPRs, carry chains expressed through __umul64hi / add-with-carry patterns
 * the AMDGPU backend lowers to v_mad_u64_u32 / v_add*_co chains, branch-free
 * table selection (no per-lane divergent memory indexing into scratch).
 *
 * Replaces the verification math of the vendored libsecp256k1 used by
 * crypto/txscript/src/lib.rs:869 (schnorr) and :899 (ecdsa). Independent
 * implementation from oracle/ok_secp.c (different structure: branchless,
 * fused reduction, Jacobian-only checks without affine conversion where
 * possible); parity is established by tests against the oracle and the
 * reference's mainnet-signature vectors.
 */
#ifndef KV_SECP_DEVICE_H
#define KV_SECP_DEVICE_H

#ifndef KV_HOST_TEST
#include <hip/hip_runtime.h>
#endif
#include <stdint.h>

namespace kv {

/* Build-time tuning knobs (perf experiments; default = the proven-safe config):
 *  -DKV_GROUP_INLINE   : inline the group ops instead of __noinline__ calls
 *                        (removes ABI spills; loops stay rolled via unroll 1)
 *  -DKV_DOUBLE_INLINE  : inline ONLY gej_double (half the ladder's calls)
 * Full inlining reproducibly hangs gfx950 (long-branch relaxation on the
 * giant body); gej_double alone is the bounded experiment. */
#ifdef KV_GROUP_INLINE
#define KV_GROUP_ATTR inline
#else
#define KV_GROUP_ATTR __noinline__
#endif
#if defined(KV_GROUP_INLINE) || defined(KV_DOUBLE_INLINE)
#define KV_DOUBLE_ATTR inline
#else
#define KV_DOUBLE_ATTR __noinline__
#endif

typedef uint64_t u64;
typedef uint32_t u32;
typedef unsigned __int128 u128; /* 128-bit accumulators: LLVM lowers their
  adds to native v_add_co/v_addc carry chains — the hand-rolled (s<a) carry
  extraction compiled to v_cmp+v_cndmask chains, ~2× the instructions */

/* ---------- 256-bit field element mod p = 2^256 - 0x1000003D1 ---------- */

struct fe {
  u64 n[4];
};

#define KV_P0 0xFFFFFFFEFFFFFC2FULL
#define KV_P1 0xFFFFFFFFFFFFFFFFULL
#define KV_PC 0x1000003D1ULL /* 2^256 - p */

__device__ __forceinline__ u64 addc(u64 a, u64 b, u64 &carry) {
  u128 t = (u128)a + b + carry;
  carry = (u64)(t >> 64);
  return (u64)t;
}

__device__ __forceinline__ u64 subb(u64 a, u64 b, u64 &borrow) {
  u128 t = (u128)a - b - borrow;
  borrow = (u64)(t >> 64) & 1;
  return (u64)t;
}

__device__ __forceinline__ void fe_norm_once(fe &a) {
  /* conditional subtract p, branchless */
  u64 ge = (a.n[3] == KV_P1) & (a.n[2] == KV_P1) & (a.n[1] == KV_P1) &
           (a.n[0] >= KV_P0);
  u64 mask = 0 - ge;
  u64 borrow = 0;
  a.n[0] = subb(a.n[0], KV_P0 & mask, borrow);
  a.n[1] = subb(a.n[1], KV_P1 & mask, borrow);
  a.n[2] = subb(a.n[2], KV_P1 & mask, borrow);
  a.n[3] = subb(a.n[3], KV_P1 & mask, borrow);
}

/* full 4x4 multiply (row-wise carry chain) + two-fold reduction via
 * 2^256 ≡ PC (mod p). Straight-line, branch-free. */
__device__ __forceinline__ void fe_mul_inner(u64 t[8], const u64 *an, const u64 *bn) {
#pragma unroll
  for (int i = 0; i < 8; i++) t[i] = 0;
#pragma unroll
  for (int i = 0; i < 4; i++) {
    u64 carry = 0;
#pragma unroll
    for (int j = 0; j < 4; j++) {
      u128 cur = (u128)an[i] * bn[j] + t[i + j] + carry;
      t[i + j] = (u64)cur;
      carry = (u64)(cur >> 64);
    }
    t[i + 4] = carry;
  }
}

__device__ __forceinline__ void fe_reduce8(fe &r, const u64 t[8]) {
  /* fold 1: s[0..4] = t[0..3] + PC * t[4..7] (PC is 33 bits) */
  u128 c = (u128)t[0] + (u128)t[4] * KV_PC;
  u64 s0 = (u64)c;
  c = (c >> 64) + t[1] + (u128)t[5] * KV_PC;
  u64 s1 = (u64)c;
  c = (c >> 64) + t[2] + (u128)t[6] * KV_PC;
  u64 s2 = (u64)c;
  c = (c >> 64) + t[3] + (u128)t[7] * KV_PC;
  u64 s3 = (u64)c;
  u64 s4 = (u64)(c >> 64); /* < 2^34 */
  /* fold 2: r = s[0..3] + PC * s4 (product < 2^67) */
  c = (u128)s0 + (u128)s4 * KV_PC;
  r.n[0] = (u64)c;
  c = (c >> 64) + s1;
  r.n[1] = (u64)c;
  c = (c >> 64) + s2;
  r.n[2] = (u64)c;
  c = (c >> 64) + s3;
  r.n[3] = (u64)c;
  /* carry out → wrapped 2^256 once more; value tiny (< 2^68) */
  if ((u64)(c >> 64)) {
    c = (u128)r.n[0] + KV_PC;
    r.n[0] = (u64)c;
    u64 cc = (u64)(c >> 64);
    r.n[1] += cc;
    /* (r.n[1] was tiny: no further carry) */
  }
  fe_norm_once(r);
}

__device__ __forceinline__ void fe_mul(fe &r, const fe &a, const fe &b) {
  u64 t[8];
  fe_mul_inner(t, a.n, b.n);
  fe_reduce8(r, t);
}

/* ================= 10x26 field representation (fe26) =================
 * (limb convention: l0..l8 < m*2^26, l9 < m*2^22 for magnitude m)
 *
 * The 4xu64 representation above lowers every limb product into
 * v_mad_u64_u32 + v_add_co/v_addc VCC-carry chains; measured disassembly of
 * the group ops shows 24% s_nop (VCC read-after-write padding) + 30% carry
 * adds + 16% movs — only ~10% of issue slots multiply. The 10x26 form
 * (limbs < 2^26, base 2^26, 19 column sums < 2^62 for operand magnitude <= 8)
 * accumulates whole columns in plain 64-bit v_mad_u64_u32 chains with NO
 * carry flags at all, eliminating the hazard padding. Derived and fuzzed
 * against exact bigint arithmetic (20k cases, every intermediate < 2^64)
 * before transcription.
 *
 * Magnitude discipline (limb_i < m * 2^26, top < m * 2^26):
 *   - fe26_mul / fe26_sqr outputs: m = 1 (top limb may graze 2^26: m <= 2)
 *   - inputs to mul/sqr MUST have m <= 8
 *   - fe26_add: m = ma + mb;  fe26_neg(m): result m+1;  adds/subs may run to
 *     m <= 31 (u32 limb bound) before a fe26_norm_weak (-> m = 1)
 * Group ops below annotate magnitudes; the host-test build asserts them. */

#define KV26_M ((u32)0x3FFFFFF)
#define KV26_R0 ((u64)0x3D10) /* 2^260 mod p = R0 + 2^36 (C<<4) */

struct fe26 {
#ifdef KV_FE26_ALIGN
  alignas(16) u32 l[10];
#else
  u32 l[10];
#endif
};

/* mul/sqr safety bound: limbs <= 2*8*2^26 = 2^30 ("magnitude 8" with the
 * negate convention limb(m) <= 2m*2^26): 10 column products of (2^30)^2 sum
 * to < 1.4*2^63 < 2^64. The host-test build asserts it on every call. */
#ifdef KV_HOST_TEST
#include <assert.h>
#define KV26_CHECK_MUL_IN(a)                                                   \
  do {                                                                         \
    for (int _i = 0; _i < 10; _i++) assert((a).l[_i] <= (1u << 30));           \
  } while (0)
#else
#define KV26_CHECK_MUL_IN(a)
#endif

__device__ __forceinline__ void fe26_reduce19(fe26 &r, const u64 t[19]) {
  /* fold columns 10..18 into 0..12 via 2^260 == R0 + 2^36 (mod p)·2^(26k) */
  u64 s[13];
#pragma unroll
  for (int k = 0; k < 10; k++) s[k] = t[k];
  s[10] = s[11] = s[12] = 0;
#pragma unroll
  for (int k = 0; k < 9; k++) {
    u64 lo = t[k + 10] & KV26_M;
    u64 hi = t[k + 10] >> 26;
    s[k] += lo * KV26_R0;
    s[k + 1] += (lo << 10) + hi * KV26_R0;
    s[k + 2] += hi << 10;
  }
  u64 out[13], c = 0;
#pragma unroll
  for (int k = 0; k < 13; k++) {
    u64 d = s[k] + c;
    out[k] = d & KV26_M;
    c = d >> 26;
  }
  /* second fold: tiny spill in out[10..12] */
  u64 extra = out[10] | (out[11] << 26) | (out[12] << 52);
  u64 lo = extra & KV26_M, hi = extra >> 26;
  out[0] += lo * KV26_R0;
  out[1] += (lo << 10) + hi * KV26_R0;
  out[2] += hi << 10;
  c = 0;
#pragma unroll
  for (int k = 0; k < 10; k++) {
    u64 d = out[k] + c;
    r.l[k] = (u32)(d & KV26_M);
    c = d >> 26;
  }
  /* residual carry (tiny): 2^260 fold once more */
  r.l[0] += (u32)(c * KV26_R0);
  r.l[1] += (u32)(c << 10);
  /* keep the top limb 22-bit (the magnitude convention scales l9 by 2^22):
   * x*2^256 == x*(2^32 + 977) */
  u32 x = r.l[9] >> 22;
  r.l[9] &= 0x3FFFFF;
  r.l[0] += x * 977u;
  r.l[1] += x << 6;
}

__device__ __forceinline__ void fe26_mul_inner(u64 t[19], const fe26 &a,
                                               const fe26 &b) {
  KV26_CHECK_MUL_IN(a);
  KV26_CHECK_MUL_IN(b);
#pragma unroll
  for (int k = 0; k < 19; k++) t[k] = 0;
#pragma unroll
  for (int i = 0; i < 10; i++)
#pragma unroll
    for (int j = 0; j < 10; j++) t[i + j] += (u64)a.l[i] * b.l[j];

}

__device__ __forceinline__ void fe26_sqr_inner(u64 t[19], const fe26 &a) {
  KV26_CHECK_MUL_IN(a);
#pragma unroll
  for (int k = 0; k < 19; k++) t[k] = 0;
#pragma unroll
  for (int i = 0; i < 10; i++) {
#pragma unroll
    for (int j = i + 1; j < 10; j++) t[i + j] += (u64)a.l[i] * a.l[j];
  }
#pragma unroll
  for (int k = 0; k < 19; k++) t[k] <<= 1;
#pragma unroll
  for (int i = 0; i < 10; i++) t[2 * i] += (u64)a.l[i] * a.l[i];
}

__device__ __forceinline__ void fe26_mul(fe26 &r, const fe26 &a, const fe26 &b) {
  u64 t[19];
  fe26_mul_inner(t, a, b);
  fe26_reduce19(r, t);
}

__device__ __forceinline__ void fe26_sqr(fe26 &r, const fe26 &a) {
  u64 t[19];
  fe26_sqr_inner(t, a);
  fe26_reduce19(r, t);
}

__device__ __forceinline__ void fe26_add(fe26 &r, const fe26 &a, const fe26 &b) {
#pragma unroll
  for (int i = 0; i < 10; i++) r.l[i] = a.l[i] + b.l[i];
}

/* r = -a for input magnitude <= m (result magnitude m+1) */
__device__ __forceinline__ void fe26_neg(fe26 &r, const fe26 &a, u32 m) {
  r.l[0] = (u32)(0x3FFFC2FUL * 2 * (m + 1)) - a.l[0];
  r.l[1] = (u32)(0x3FFFFBFUL * 2 * (m + 1)) - a.l[1];
#pragma unroll
  for (int i = 2; i < 9; i++) r.l[i] = (u32)(0x3FFFFFFUL * 2 * (m + 1)) - a.l[i];
  r.l[9] = (u32)(0x3FFFFFUL * 2 * (m + 1)) - a.l[9];
}

__device__ __forceinline__ void fe26_mul_int(fe26 &r, u32 k) {
#pragma unroll
  for (int i = 0; i < 10; i++) r.l[i] *= k;
}

__device__ __forceinline__ void fe26_cmov(fe26 &r, const fe26 &a, u32 cond) {
  u32 mask = 0 - cond;
#pragma unroll
  for (int i = 0; i < 10; i++) r.l[i] = (r.l[i] & ~mask) | (a.l[i] & mask);
}

/* one carry sweep + 2^260 fold: magnitude -> 1 (l0/l1 may graze 2^26+2^20,
 * i.e. "magnitude 2" for safety accounting; value stays < 2^260) */
__device__ __forceinline__ void fe26_norm_weak(fe26 &a) {
  u32 c = 0;
#pragma unroll
  for (int i = 0; i < 10; i++) {
    u32 d = a.l[i] + c;
    a.l[i] = d & KV26_M;
    c = d >> 26;
  }
  /* c = value >> 260 (<= input magnitude); 2^260 == R0 + 2^36 (mod p) */
  a.l[0] += c * (u32)KV26_R0;
  a.l[1] += c << 10;
  u32 x = a.l[9] >> 22;
  a.l[9] &= 0x3FFFFF;
  a.l[0] += x * 977u;
  a.l[1] += x << 6;
}

/* full (canonical) normalize — used on the rare paths only (equality, zero
 * tests, serialization). Deterministic bounded sequence: sweeps + top folds
 * until value < 2^256, then a branchless conditional subtract of p via the
 * "value + C overflows bit 256" test. */
__device__ __forceinline__ void fe26_normalize(fe26 &a) {
  fe26_norm_weak(a);
  /* 3 sweep+fold passes: if a pass folds x>0 the value drops below 2^37-ish,
   * so the next pass has x=0 and leaves canonical limbs */
#pragma unroll 1
  for (int pass = 0; pass < 3; pass++) {
    u32 c = 0;
#pragma unroll
    for (int i = 0; i < 10; i++) {
      u32 d = a.l[i] + c;
      a.l[i] = d & KV26_M;
      c = d >> 26;
    }
    /* fold 2^256 overflow of the top limb: x*2^256 == x*(2^32 + 977) */
    u32 x = (a.l[9] >> 22) | (c << 4);
    a.l[9] &= 0x3FFFFF;
    a.l[0] += x * 977u;
    a.l[1] += x << 6;
  }
  /* value < 2^256 now (limbs canonical after one more sweep inside the +C
   * pass below). value >= p  <=>  value + C >= 2^256 */
  u32 t[10];
  u64 d = (u64)a.l[0] + 977u;
  t[0] = (u32)(d & KV26_M);
  d = (d >> 26) + a.l[1] + (1u << 6);
  t[1] = (u32)(d & KV26_M);
#pragma unroll
  for (int i = 2; i < 10; i++) {
    d = (d >> 26) + a.l[i];
    t[i] = (u32)(d & KV26_M);
  }
  u32 ge = (u32)(t[9] >> 22) & 1; /* bit 256 set -> value >= p */
  t[9] &= 0x3FFFFF;
  u32 mask = 0 - ge;
#pragma unroll
  for (int i = 0; i < 10; i++) a.l[i] = (a.l[i] & ~mask) | (t[i] & mask);
}

/* does a weakly-reduced value equal 0 mod p? (normalize a copy fully) */
__device__ __forceinline__ int fe26_is_zero(const fe26 &a) {
  fe26 t = a;
  fe26_normalize(t);
  u32 z = 0;
#pragma unroll
  for (int i = 0; i < 10; i++) z |= t.l[i];
  return z == 0;
}

__device__ __forceinline__ int fe26_eq(const fe26 &a, const fe26 &b) {
  fe26 ta = a, tb = b;
  fe26_normalize(ta);
  fe26_normalize(tb);
  u32 d = 0;
#pragma unroll
  for (int i = 0; i < 10; i++) d |= ta.l[i] ^ tb.l[i];
  return d == 0;
}

/* conversions at the byte/u64 boundary */
__device__ __forceinline__ void fe26_from_fe(fe26 &r, const fe &a) {
#pragma unroll
  for (int i = 0; i < 10; i++) {
    int bit = 26 * i;
    int w = bit >> 6, sh = bit & 63;
    u64 v = a.n[w] >> sh;
    if (sh > 38 && w < 3) v |= a.n[w + 1] << (64 - sh);
    r.l[i] = (u32)(v & KV26_M);
  }
  r.l[9] &= 0x3FFFFF;
}

/* input must be fully normalized */
__device__ __forceinline__ void fe26_to_fe(fe &r, const fe26 &a) {
  r.n[0] = (u64)a.l[0] | ((u64)a.l[1] << 26) | ((u64)a.l[2] << 52);
  r.n[1] = ((u64)a.l[2] >> 12) | ((u64)a.l[3] << 14) | ((u64)a.l[4] << 40);
  r.n[2] = ((u64)a.l[4] >> 24) | ((u64)a.l[5] << 2) | ((u64)a.l[6] << 28) |
           ((u64)a.l[7] << 54);
  r.n[3] = ((u64)a.l[7] >> 10) | ((u64)a.l[8] << 16) | ((u64)a.l[9] << 42);
}

/* ---------- scalar mod n ---------- */

struct sc {
  u64 d[4];
};

#define KV_N0 0xBFD25E8CD0364141ULL
#define KV_N1 0xBAAEDCE6AF48A03BULL
#define KV_N2 0xFFFFFFFFFFFFFFFEULL
#define KV_N3 0xFFFFFFFFFFFFFFFFULL
/* 2^256 - n (129 bits, limbs) */
#define KV_NC0 0x402DA1732FC9BEBFULL
#define KV_NC1 0x4551231950B75FC4ULL
#define KV_NC2 1ULL

__device__ __forceinline__ int sc_gte_n(const sc &a) {
  if (a.d[3] > KV_N3) return 1;
  if (a.d[3] < KV_N3) return 0;
  if (a.d[2] > KV_N2) return 1;
  if (a.d[2] < KV_N2) return 0;
  if (a.d[1] > KV_N1) return 1;
  if (a.d[1] < KV_N1) return 0;
  return a.d[0] >= KV_N0;
}

__device__ __forceinline__ void sc_sub_n(sc &a) {
  u64 borrow = 0;
  a.d[0] = subb(a.d[0], KV_N0, borrow);
  a.d[1] = subb(a.d[1], KV_N1, borrow);
  a.d[2] = subb(a.d[2], KV_N2, borrow);
  a.d[3] = subb(a.d[3], KV_N3, borrow);
}

__device__ __forceinline__ int sc_is_zero(const sc &a) {
  return (a.d[0] | a.d[1] | a.d[2] | a.d[3]) == 0;
}

/* from big-endian bytes; returns 1 on overflow (input >= n) */
__device__ __forceinline__ int sc_from_be(sc &r, const uint8_t b[32]) {
#pragma unroll
  for (int i = 0; i < 4; i++) {
    u64 w = 0;
#pragma unroll
    for (int j = 0; j < 8; j++) w = (w << 8) | b[8 * (3 - i) + j];
    r.d[i] = w;
  }
  int ov = sc_gte_n(r);
  if (ov) sc_sub_n(r);
  return ov;
}

__device__ __forceinline__ void sc_neg(sc &r, const sc &a) {
  u64 is_zero = sc_is_zero(a);
  u64 borrow = 0;
  r.d[0] = subb(KV_N0, a.d[0], borrow);
  r.d[1] = subb(KV_N1, a.d[1], borrow);
  r.d[2] = subb(KV_N2, a.d[2], borrow);
  r.d[3] = subb(KV_N3, a.d[3], borrow);
  u64 mask = 0 - is_zero;
  r.d[0] &= ~mask;
  r.d[1] &= ~mask;
  r.d[2] &= ~mask;
  r.d[3] &= ~mask;
}

/* 512-bit → mod n: three bounded folds with NC = 2^256 - n (129 bits, 3 limbs).
 * Each fold: value = lo(4 limbs) + NC × hi(k limbs); widths shrink 4→3→1→ε. */
__device__ inline void sc_reduce8(sc &r, const u64 tin[8]) {
  const u64 NC[3] = {KV_NC0, KV_NC1, KV_NC2};
  /* fold 1: s[7] = t[0..3] + NC × t[4..7] */
  u64 s[7] = {tin[0], tin[1], tin[2], tin[3], 0, 0, 0};
  for (int i = 0; i < 4; i++) {
    u64 carry = 0;
    for (int j = 0; j < 3; j++) {
      u64 lo = tin[4 + i] * NC[j];
      u64 hi = __umul64hi(tin[4 + i], NC[j]);
      u64 c = 0;
      s[i + j] = addc(s[i + j], lo, c);
      u64 c2 = 0;
      s[i + j] = addc(s[i + j], carry, c2);
      carry = hi + c + c2;
    }
    /* deposit final carry at s[i+3] and ripple */
    u64 c = 0;
    s[i + 3] = addc(s[i + 3], carry, c);
    for (int k = i + 4; k < 7 && c; k++) s[k] = addc(s[k], 0, c);
  }
  /* fold 2: u[5] = s[0..3] + NC × s[4..6] */
  u64 u[5] = {s[0], s[1], s[2], s[3], 0};
  for (int i = 0; i < 3; i++) {
    u64 carry = 0;
    for (int j = 0; j < 3; j++) {
      int pos = i + j;
      u64 lo = s[4 + i] * NC[j];
      u64 hi = __umul64hi(s[4 + i], NC[j]);
      u64 c = 0;
      u[pos] = addc(u[pos], lo, c);
      u64 c2 = 0;
      u[pos] = addc(u[pos], carry, c2);
      carry = hi + c + c2;
    }
    /* deposit final carry; for i==2 the carry is provably 0 (value < 2^259) */
    if (i + 3 < 5) {
      u64 c = 0;
      u[i + 3] = addc(u[i + 3], carry, c);
      for (int k = i + 4; k < 5 && c; k++) u[k] = addc(u[k], 0, c);
    }
  }
  /* fold 3: w[4] + carry = u[0..3] + NC × u[4]  (u[4] small) */
  u64 w[4] = {u[0], u[1], u[2], u[3]};
  {
    u64 carry = 0;
    for (int j = 0; j < 3; j++) {
      u64 lo = u[4] * NC[j];
      u64 hi = __umul64hi(u[4], NC[j]);
      u64 c = 0;
      w[j] = addc(w[j], lo, c);
      u64 c2 = 0;
      w[j] = addc(w[j], carry, c2);
      carry = hi + c + c2;
    }
    u64 c = 0;
    w[3] = addc(w[3], carry, c);
    /* fold 4: carry bit ⇒ += NC once more (cannot carry again: value < 2^130+ε) */
    if (c) {
      u64 cc = 0;
      w[0] = addc(w[0], NC[0], cc);
      w[1] = addc(w[1], NC[1], cc);
      w[2] = addc(w[2], NC[2], cc);
      w[3] = addc(w[3], 0, cc);
    }
  }
  r.d[0] = w[0]; r.d[1] = w[1]; r.d[2] = w[2]; r.d[3] = w[3];
  while (sc_gte_n(r)) sc_sub_n(r);
}

__device__ inline void sc_mul(sc &r, const sc &a, const sc &b) {
  u64 t[8];
  fe_mul_inner(t, a.d, b.d);
  sc_reduce8(r, t);
}

__device__ KV_GROUP_ATTR void sc_sqrn(sc &r, int n) {
#pragma unroll 1
  for (int i = 0; i < n; i++) sc_mul(r, r, r);
}

/* scalar inverse mod n via Fermat (ECDSA only): n-2 = (2^124-1)·2^132 + tail.
 * All-ones head by addition chain (124 sqr + 10 mul), 132-bit tail by 4-bit
 * fixed windows of the constant exponent (132 sqr + 33 + 14 table muls)
 * ≈ 313 sc_mul vs ~512 for the naive 256-step square-and-multiply. The window
 * digits and the head split are derived and checked symbolically in python
 * before transcription. */
__device__ KV_GROUP_ATTR void sc_inv(sc &r, const sc &a) {
  static const uint8_t WIN[33] = {14, 11, 10, 10, 14, 13, 12, 14, 6,  10, 15,
                                  4,  8,  10, 0,  3,  11, 11, 15, 13, 2,  5,
                                  14, 8,  12, 13, 0,  3,  6,  4,  1,  3,  15};
  sc tab[16];
  tab[0] = {{1, 0, 0, 0}};
  tab[1] = a;
#pragma unroll 1
  for (int i = 2; i < 16; i++) sc_mul(tab[i], tab[i - 1], a);
  /* head: x124 = a^(2^124-1); xk here means a^(2^k-1), so x3 = a^7 and
   * x4 = a^15 — both already in the window table */
  sc x6, x12, x24, t;
  const sc &x3 = tab[7], &x4 = tab[15];
  t = x3;
  sc_sqrn(t, 3);
  sc_mul(x6, t, x3);
  t = x6;
  sc_sqrn(t, 6);
  sc_mul(x12, t, x6);
  t = x12;
  sc_sqrn(t, 12);
  sc_mul(x24, t, x12);
  t = x24;
  sc_sqrn(t, 24);
  sc_mul(t, t, x24); /* x48 */
  sc x48 = t;
  sc_sqrn(t, 48);
  sc_mul(t, t, x48); /* x96 */
  sc_sqrn(t, 24);
  sc_mul(t, t, x24); /* x120 */
  sc_sqrn(t, 4);
  sc_mul(t, t, x4); /* x124 */
  /* tail: 33 4-bit windows */
#pragma unroll 1
  for (int i = 0; i < 33; i++) {
    sc_sqrn(t, 4);
    sc m;
    sc_mul(m, t, tab[WIN[i]]);
    t = m;
  }
  r = t;
}

/* ---------- fe26 chains: Fermat powers on the 10x26 form ---------- */

__device__ KV_GROUP_ATTR void fe26_sqrn(fe26 &r, int n) {
#pragma unroll 1
  for (int i = 0; i < n; i++) fe26_sqr(r, r);
}

/* addition-chain blocks a^(2^k-1); same chain as the 4xu64 version above
 * (verified symbolically), restated on fe26 */
__device__ KV_GROUP_ATTR void fe26_chain223(fe26 &x223, fe26 &x22, fe26 &x2,
                                            fe26 &x3, const fe26 &a) {
  fe26 t, x11, x44, x88;
  fe26_sqr(t, a);
  fe26_mul(x2, t, a);
  fe26_sqr(t, x2);
  fe26_mul(x3, t, a);
  t = x3;
  fe26_sqrn(t, 3);
  fe26_mul(t, t, x3);
  fe26_sqrn(t, 3);
  fe26_mul(t, t, x3);
  fe26_sqrn(t, 2);
  fe26_mul(x11, t, x2);
  t = x11;
  fe26_sqrn(t, 11);
  fe26_mul(x22, t, x11);
  t = x22;
  fe26_sqrn(t, 22);
  fe26_mul(x44, t, x22);
  t = x44;
  fe26_sqrn(t, 44);
  fe26_mul(x88, t, x44);
  t = x88;
  fe26_sqrn(t, 88);
  fe26_mul(t, t, x88);
  fe26_sqrn(t, 44);
  fe26_mul(t, t, x44);
  fe26_sqrn(t, 3);
  fe26_mul(x223, t, x3);
}

__device__ inline void fe26_inv(fe26 &r, const fe26 &a) {
  fe26 x223, x22, x2, x3, t;
  fe26_chain223(x223, x22, x2, x3, a);
  t = x223;
  fe26_sqrn(t, 23);
  fe26_mul(t, t, x22);
  fe26_sqrn(t, 5);
  fe26_mul(t, t, a);
  fe26_sqrn(t, 3);
  fe26_mul(t, t, x2);
  fe26_sqrn(t, 2);
  fe26_mul(r, t, a);
}

/* sqrt via a^((p+1)/4); returns 1 if r*r == a */
__device__ inline int fe26_sqrt(fe26 &r, const fe26 &a) {
  fe26 x223, x22, x2, x3, t, chk;
  fe26_chain223(x223, x22, x2, x3, a);
  t = x223;
  fe26_sqrn(t, 23);
  fe26_mul(t, t, x22);
  fe26_sqrn(t, 6);
  fe26_mul(t, t, x2);
  fe26_sqrn(t, 2);
  fe26_sqr(chk, t);
  r = t;
  fe26 aa = a;
  return fe26_eq(chk, aa);
}

/* ---------- group: Jacobian points, a=0 b=7 curve, fe26 coordinates ----------
 * Infinity is represented as Z == 0 (no flag field, no early returns).
 * Magnitude discipline: every group-op OUTPUT coordinate has magnitude <= 2
 * (fe26_norm_weak where sums exceed it); every mul/sqr INPUT stays <= 8 —
 * the per-line annotations below track the worst case. */

struct ge {
  fe26 x, y; /* affine */
};

struct gej {
  fe26 x, y, z; /* z == 0 ⇔ infinity */
};

__device__ __forceinline__ void fe26_set_int(fe26 &r, u32 v) {
  r.l[0] = v;
#pragma unroll
  for (int i = 1; i < 10; i++) r.l[i] = 0;
}

__device__ __forceinline__ void gej_set_infinity(gej &r) {
  fe26_set_int(r.x, 1);
  fe26_set_int(r.y, 1);
  fe26_set_int(r.z, 0);
}

__device__ __forceinline__ int gej_is_infinity(const gej &a) {
  return fe26_is_zero(a.z);
}

/* doubling: straight-line, valid for z==0 (result keeps z==0).
 * y == 0 cannot occur on secp256k1 (no 2-torsion). Inputs magnitude <= 2. */
__device__ __forceinline__ void gej_double_impl(gej &r, const gej &a) {
  fe26 A, B, C, D, E, F, t, zz;
  fe26_sqr(A, a.x);        /* 1 */
  fe26_sqr(B, a.y);        /* 1 */
  fe26_sqr(C, B);          /* 1 */
  fe26_mul(zz, a.y, a.z);  /* 1 */
  fe26_add(t, a.x, B);     /* 2+2=4 */
  fe26_sqr(t, t);          /* 1 */
  E = A;
  fe26_mul_int(E, 3);      /* 6 */
  fe26_sqr(F, E);          /* 1 */
  fe26 nA, nC;
  fe26_neg(nA, A, 2);      /* 3 */
  fe26_neg(nC, C, 2);      /* 3 */
  fe26_add(t, t, nA);
  fe26_add(D, t, nC);      /* 1+3+3 = 8 (pre-double) */
  fe26_add(D, D, D);       /* 16 */
  fe26_norm_weak(D);       /* -> 2 */
  fe26 nx, ny, nz, nD;
  fe26_neg(nD, D, 2);      /* 3 */
  fe26_add(nx, F, nD);
  fe26_add(nx, nx, nD);    /* 2+3+3 = 8 */
  fe26_norm_weak(nx);      /* -> 2 */
  fe26 nnx;
  fe26_neg(nnx, nx, 2);    /* 3 */
  fe26_add(t, D, nnx);     /* 2+3 = 5: mul input ok */
  fe26_mul(t, E, t);       /* E 6, t 5 -> 1 */
  fe26 C8;
  C8 = C;
  fe26_mul_int(C8, 8);     /* 16 */
  fe26 nC8;
  fe26_neg(nC8, C8, 16);   /* 17 */
  fe26_add(ny, t, nC8);    /* 19 <= 31 */
  fe26_norm_weak(ny);      /* -> 2 */
  fe26_add(nz, zz, zz);    /* 4 */
  fe26_norm_weak(nz);      /* -> 2 */
  r.x = nx;
  r.y = ny;
  r.z = nz;
}

__device__ KV_DOUBLE_ATTR void gej_double(gej &r, const gej &a) {
  gej_double_impl(r, a);
}

/* four doublings in one call frame (accumulator crosses the noinline ABI once
 * per window instead of four times) */
__device__ KV_GROUP_ATTR void gej_double4(gej &r, const gej &a) {
  gej t;
  gej_double_impl(t, a);
  gej_double_impl(r, t);
  gej_double_impl(t, r);
  gej_double_impl(r, t);
}

__device__ __forceinline__ void gej_cmov(gej &r, const gej &a, u64 cond) {
  u32 c = (u32)(cond & 1);
  fe26_cmov(r.x, a.x, c);
  fe26_cmov(r.y, a.y, c);
  fe26_cmov(r.z, a.z, c);
}

/* mixed add (b affine, magnitude <= 3 after phi/negation); a coords mag <= 2 */
__device__ __forceinline__ void gej_add_ge_impl(gej &r, const gej &a, const ge &b) {
  u64 a_inf = (u64)fe26_is_zero(a.z);
  fe26 z1z1, u2, s2, h, hh, i, j, rr, v, t;
  fe26_sqr(z1z1, a.z);       /* 1 */
  fe26_mul(s2, b.y, a.z);    /* 1 */
  fe26_mul(u2, b.x, z1z1);   /* 1 */
  fe26_mul(s2, s2, z1z1);    /* 1 */
  fe26 nX, nY;
  fe26_neg(nX, a.x, 2);      /* 3 */
  fe26_neg(nY, a.y, 2);      /* 3 */
  fe26_add(h, u2, nX);       /* 5 */
  fe26_add(rr, s2, nY);      /* 5 */
  if (!a_inf && fe26_is_zero(h)) {
    /* rare: same x. rr==0 -> doubling; else opposite points -> infinity */
    if (fe26_is_zero(rr)) {
      gej t2;
      gej ain = a;
      fe26_norm_weak(ain.x);
      fe26_norm_weak(ain.y);
      fe26_norm_weak(ain.z);
      gej_double_impl(t2, ain);
      r = t2;
    } else {
      gej_set_infinity(r);
    }
    return;
  }
  fe26 zz;
  fe26_add(zz, a.z, h);      /* 7 */
  fe26_sqr(zz, zz);          /* 1 */
  fe26_sqr(hh, h);           /* 1 */
  i = hh;
  fe26_mul_int(i, 4);        /* 8 */
  fe26_mul(j, h, i);         /* h 5, i 8 -> 1 */
  fe26_mul(v, a.x, i);       /* 1 */
  fe26 nx, ny, nz;
  fe26_sqr(nx, rr);          /* rr 5 -> 1 */
  fe26_mul_int(nx, 4);       /* 8: (2r)^2 */
  fe26 nj, nv;
  fe26_neg(nj, j, 2);        /* 3 */
  fe26_neg(nv, v, 2);        /* 3 */
  fe26_add(nx, nx, nj);
  fe26_add(nx, nx, nv);
  fe26_add(nx, nx, nv);      /* 8+3+3+3 = 17 */
  fe26_norm_weak(nx);        /* -> 2 */
  fe26 nnx;
  fe26_neg(nnx, nx, 2);      /* 3 */
  fe26_add(t, v, nnx);       /* 5 */
  fe26_mul(t, rr, t);        /* rr 5, t 5 -> 1 */
  fe26_add(t, t, t);         /* 4: 2*rr*(v-nx) */
  fe26 y1j;
  fe26_mul(y1j, a.y, j);     /* 1 */
  fe26_add(y1j, y1j, y1j);   /* 4 */
  fe26 ny1j;
  fe26_neg(ny1j, y1j, 4);    /* 5 */
  fe26_add(ny, t, ny1j);     /* 9 */
  fe26_norm_weak(ny);        /* -> 2 */
  fe26 nzz, nhh;
  fe26_neg(nzz, z1z1, 2);    /* 3 */
  fe26_neg(nhh, hh, 2);      /* 3 */
  fe26_add(nz, zz, nzz);
  fe26_add(nz, nz, nhh);     /* 2+3+3 = 8 */
  fe26_norm_weak(nz);        /* -> 2 */
  /* a was infinity -> result is b (z = 1) */
  fe26 one;
  fe26_set_int(one, 1);
  u32 ci = (u32)(a_inf & 1);
  fe26_cmov(nx, b.x, ci);
  fe26_cmov(ny, b.y, ci);
  fe26_cmov(nz, one, ci);
  r.x = nx;
  r.y = ny;
  r.z = nz;
}

/* full add; a and b coords magnitude <= 3 */
__device__ KV_GROUP_ATTR void gej_add_ge(gej &r, const gej &a, const ge &b) {
  gej_add_ge_impl(r, a, b);
}

__device__ KV_GROUP_ATTR void gej_add(gej &r, const gej &a, const gej &b) {
  u64 a_inf = (u64)fe26_is_zero(a.z);
  fe26 z1z1, z2z2, u1, u2, s1, s2, h, i, j, rr, v, t;
  fe26_sqr(z1z1, a.z);      /* 1 */
  fe26_sqr(z2z2, b.z);      /* 1 */
  fe26_mul(u1, a.x, z2z2);  /* 1 */
  fe26_mul(u2, b.x, z1z1);  /* 1 */
  fe26_mul(s1, a.y, b.z);   /* 1 */
  fe26_mul(s1, s1, z2z2);   /* 1 */
  fe26_mul(s2, b.y, a.z);   /* 1 */
  fe26_mul(s2, s2, z1z1);   /* 1 */
  fe26 nu1, ns1;
  fe26_neg(nu1, u1, 2);     /* 3 */
  fe26_neg(ns1, s1, 2);     /* 3 */
  fe26_add(h, u2, nu1);     /* 5 */
  fe26_add(rr, s2, ns1);    /* 5 */
  if (!a_inf && !fe26_is_zero(b.z) && fe26_is_zero(h)) {
    if (fe26_is_zero(rr)) {
      gej t2;
      gej ain = a;
      fe26_norm_weak(ain.x);
      fe26_norm_weak(ain.y);
      fe26_norm_weak(ain.z);
      gej_double_impl(t2, ain);
      r = t2;
    } else {
      gej_set_infinity(r);
    }
    return;
  }
  fe26 zz;
  fe26_add(zz, a.z, b.z);   /* 2+3 = 5 */
  fe26_sqr(zz, zz);         /* 1 */
  fe26_sqr(i, h);           /* 1 */
  fe26_mul_int(i, 4);       /* 8 */
  fe26_mul(j, h, i);        /* 1 */
  fe26_mul(v, u1, i);       /* 1 */
  fe26 nx, ny, nz;
  fe26_sqr(nx, rr);         /* 1 */
  fe26_mul_int(nx, 4);      /* 8 */
  fe26 nj, nv;
  fe26_neg(nj, j, 2);
  fe26_neg(nv, v, 2);
  fe26_add(nx, nx, nj);
  fe26_add(nx, nx, nv);
  fe26_add(nx, nx, nv);     /* 17 */
  fe26_norm_weak(nx);       /* -> 2 */
  fe26 nnx;
  fe26_neg(nnx, nx, 2);
  fe26_add(t, v, nnx);      /* 5 */
  fe26_mul(t, rr, t);       /* 1 */
  fe26_add(t, t, t);        /* 4 */
  fe26 s1j;
  fe26_mul(s1j, s1, j);     /* 1 */
  fe26_add(s1j, s1j, s1j);  /* 4 */
  fe26 ns1j;
  fe26_neg(ns1j, s1j, 4);   /* 5 */
  fe26_add(ny, t, ns1j);    /* 9 */
  fe26_norm_weak(ny);       /* -> 2 */
  fe26 nzz1, nzz2;
  fe26_neg(nzz1, z1z1, 2);
  fe26_neg(nzz2, z2z2, 2);
  fe26_add(zz, zz, nzz1);
  fe26_add(zz, zz, nzz2);   /* 1+3+3 = 7 */
  fe26_mul(nz, zz, h);      /* zz 7, h 5 -> 1 */
  /* a infinity -> result = b */
  u32 ci = (u32)(a_inf & 1);
  fe26_cmov(nx, b.x, ci);
  fe26_cmov(ny, b.y, ci);
  fe26_cmov(nz, b.z, ci);
  r.x = nx;
  r.y = ny;
  r.z = nz;
}

} // namespace kv

#endif /* KV_SECP_DEVICE_H */
