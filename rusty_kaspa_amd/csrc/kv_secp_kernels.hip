/* MI355X-native batched signature verification kernels (gfx950, PRODUCT code).
 *
 * kv_schnorr_verify_kernel ⇔ secp256k1::schnorr::Signature::verify at
 *   crypto/txscript/src/lib.rs:869 (BIP-340), one signature per lane.
 * kv_ecdsa_verify_kernel   ⇔ secp256k1::ecdsa::Signature::verify at
 *   crypto/txscript/src/lib.rs:899 (low-S enforced like libsecp256k1).
 *
 * Layout: tuples are 128B (Schnorr: r‖s‖pk_x‖msg) / 129B-padded-to-132
 * (ECDSA: r‖s‖pk33‖msg, see engine) records in HBM; loads are a handful of
 * dwordx4 per lane against ~5k 256-bit field multiplies of ALU work — this
 * kernel is integer-ALU bound by ~300 ops/byte, so layout
 * matters far less than VGPR pressure and carry-chain depth.
 *
 * Verdicts are wave-ballot-compressed: each 64-lane wave writes one u64 of the
 * bitmap with a single store (no atomics).
 */
#include "kv_hash_device.h"
#include "kv_secp_device.h"

namespace kv {

/* SHA-256 midstate after compressing tag_hash‖tag_hash for
 * tag = "BIP0340/challenge" (fixed by BIP-340; precomputed, verified against a
 * full SHA-256 at build time in tests). */
__device__ __constant__ static const uint32_t BIP340_CHALLENGE_MID[8] = {
    0x9cecba11u, 0x23925381u, 0x11679112u, 0xd1627e0fu,
    0x97c87550u, 0x003cc765u, 0x90f61164u, 0x33e9b66au};

/* status codes (engine-internal, align with KV_SCRIPT_* mapping in host) */
enum : uint8_t {
  KVS_VALID = 0,
  KVS_INVALID = 1,
  KVS_BAD_PUBKEY = 2,
  KVS_BAD_SIG = 3,
};

__device__ __forceinline__ void fe_from_be(fe &r, const uint8_t b[32]) {
#pragma unroll
  for (int i = 0; i < 4; i++) {
    u64 w = 0;
#pragma unroll
    for (int j = 0; j < 8; j++) w = (w << 8) | b[8 * (3 - i) + j];
    r.n[i] = w;
  }
}

__device__ __forceinline__ int fe_gte_p_full(const fe &a) {
  return (a.n[3] == KV_P1) & (a.n[2] == KV_P1) & (a.n[1] == KV_P1) &
         (a.n[0] >= KV_P0);
}

/* lift x (BE bytes) to an affine point with chosen y parity; 0 on failure.
 * The >= p overflow check runs on the exact 4xu64 form; the curve equation
 * and sqrt run on fe26. */
__device__ inline int lift_x_parity(ge &P, const uint8_t xb[32], u32 want_odd) {
  fe xu;
  fe_from_be(xu, xb);
  if (fe_gte_p_full(xu)) return 0;
  fe26 x, x3, y2, y;
  fe26_from_fe(x, xu);
  fe26_sqr(x3, x);
  fe26_mul(x3, x3, x);
  fe26 seven;
  fe26_set_int(seven, 7);
  fe26_add(y2, x3, seven);
  if (!fe26_sqrt(y, y2)) return 0;
  fe26_normalize(y);
  fe26 ny;
  fe26_neg(ny, y, 1);
  fe26_cmov(y, ny, (y.l[0] & 1) ^ want_odd);
  P.x = x;
  P.y = y; /* magnitude <= 2 */
  return 1;
}

__device__ inline int lift_x_even(ge &P, const uint8_t xb[32]) {
  return lift_x_parity(P, xb, 0);
}

/* parse 33-byte compressed pubkey; 0 on failure */
__device__ inline int parse_compressed(ge &P, const uint8_t pk[33]) {
  if (pk[0] != 0x02 && pk[0] != 0x03) return 0;
  return lift_x_parity(P, pk + 1, pk[0] == 0x03);
}

/* affine G (4xu64 limbs; converted to fe26 once in the table-init kernel) */
__device__ __constant__ static const u64 GE_G_X[4] = {
    0x59F2815B16F81798ULL, 0x029BFCDB2DCE28D9ULL, 0x55A06295CE870B07ULL,
    0x79BE667EF9DCBBACULL};
__device__ __constant__ static const u64 GE_G_Y[4] = {
    0x9C47D08FFB10D4B8ULL, 0xFD17B448A6855419ULL, 0x5DA4FBFC0E1108A8ULL,
    0x483ADA7726A3C465ULL};

/* 8-bit fixed-base table: KV_G_TABLE8[d] = d·G, d = 1..255 (entry 0 = G,
 * discarded by the digit-0 cmov). G is FIXED, so the G/φG ladder streams can
 * afford one shared 20KB L2-resident table and add every SECOND 4-bit window
 * (17 digits per 129-bit half-scalar instead of 33) — 32 fewer mixed adds
 * per verify. */
__device__ ge KV_G_TABLE8[256];
__device__ static fe26 KV_G8_Z[256];    /* init-time scratch */
__device__ static fe26 KV_G8_PREF[256]; /* init-time scratch */

extern "C" __global__ void kv_ec_table_init_kernel() {
  if (threadIdx.x != 0 || blockIdx.x != 0) return;
  ge G;
  {
    fe gx, gy;
#pragma unroll
    for (int i = 0; i < 4; i++) {
      gx.n[i] = GE_G_X[i];
      gy.n[i] = GE_G_Y[i];
    }
    fe26_from_fe(G.x, gx);
    fe26_from_fe(G.y, gy);
  }
  /* one Montgomery batch inversion over all 255 Zs instead of 255 Fermat
   * chains (the serial per-entry form cost ~35ms of every kv_create) */
  gej acc;
  acc.x = G.x;
  acc.y = G.y;
  fe26_set_int(acc.z, 1);
  KV_G_TABLE8[0] = G;
  for (int k = 1; k <= 255; k++) {
    KV_G_TABLE8[k].x = acc.x;
    KV_G_TABLE8[k].y = acc.y;
    KV_G8_Z[k] = acc.z;
    gej t;
    gej_add_ge(t, acc, G);
    acc = t;
  }
  KV_G8_PREF[1] = KV_G8_Z[1];
  for (int k = 2; k <= 255; k++)
    fe26_mul(KV_G8_PREF[k], KV_G8_PREF[k - 1], KV_G8_Z[k]);
  fe26 inv;
  fe26_inv(inv, KV_G8_PREF[255]);
  for (int k = 255; k >= 1; k--) {
    fe26 zi;
    if (k > 1) {
      fe26_mul(zi, inv, KV_G8_PREF[k - 1]);
      fe26_mul(inv, inv, KV_G8_Z[k]);
    } else {
      zi = inv;
    }
    fe26 zi2, zi3;
    fe26_sqr(zi2, zi);
    fe26_mul(zi3, zi2, zi);
    fe26_mul(KV_G_TABLE8[k].x, KV_G_TABLE8[k].x, zi2);
    fe26_mul(KV_G_TABLE8[k].y, KV_G_TABLE8[k].y, zi3);
    fe26_normalize(KV_G_TABLE8[k].x);
    fe26_normalize(KV_G_TABLE8[k].y);
  }
}

/* ---------------- GLV endomorphism scalar decomposition ----------------
 * secp256k1 has an efficient endomorphism φ(x,y) = (β·x, y) with φ(P) = λ·P.
 * Splitting each 256-bit scalar k into k1 + k2·λ with |k1|,|k2| ≤ 2^129 halves
 * the doubling ladder (33 4-bit windows instead of 64). All constants below
 * were DERIVED in-repo (cube roots of unity via Tonelli–Shanks, lattice basis
 * via extended Euclid) and verified against φ(G) = λ·G — see the derivation
 * script output quoted in DESIGN.md. Decomposition uses the shift method:
 * c_i = (g_i·k) >> 384, k1 = k − c1·a1 − c2·a2, k2 = −(c1·b1 + c2·b2),
 * computed in 256-bit two's complement. */

__device__ __constant__ static const u64 GLV_BETA[4] = {
    0xc1396c28719501eeULL, 0x9cf0497512f58995ULL, 0x6e64479eac3434e9ULL,
    0x7ae96a2b657c0710ULL};
__device__ __constant__ static const u64 GLV_G1[4] = {
    0xe893209a45dbb031ULL, 0x3daa8a1471e8ca7fULL, 0xe86c90e49284eb15ULL,
    0x3086d221a7d46bcdULL};
__device__ __constant__ static const u64 GLV_G2[4] = {
    0x1571b4ae8ac47f71ULL, 0x221208ac9df506c6ULL, 0x6f547fa90abfe4c4ULL,
    0xe4437ed6010e8828ULL};
__device__ __constant__ static const u64 GLV_A1[4] = {
    0xe86c90e49284eb15ULL, 0x3086d221a7d46bcdULL, 0x0000000000000000ULL,
    0x0000000000000000ULL};
__device__ __constant__ static const u64 GLV_B1[4] = {
    0x90ab8056f5401b3dULL, 0x1bbc8129fef177d7ULL, 0xffffffffffffffffULL,
    0xffffffffffffffffULL};
__device__ __constant__ static const u64 GLV_A2[4] = {
    0x57c1108d9d44cfd8ULL, 0x14ca50f7a8e2f3f6ULL, 0x0000000000000001ULL,
    0x0000000000000000ULL};
__device__ __constant__ static const u64 GLV_B2[4] = {
    0xe86c90e49284eb15ULL, 0x3086d221a7d46bcdULL, 0x0000000000000000ULL,
    0x0000000000000000ULL};

/* low-256 product r = (a*b) mod 2^256 (two's-complement arithmetic) */
__device__ __forceinline__ void mul_low256(u64 r[4], const u64 a[4], const u64 b[4]) {
  u64 t[8];
  fe_mul_inner(t, a, b);
#pragma unroll
  for (int i = 0; i < 4; i++) r[i] = t[i];
}

/* c = (g*k) >> 384, a 128-bit result */
__device__ __forceinline__ void mul_shift384(u64 c[2], const u64 g[4], const u64 k[4]) {
  u64 t[8];
  fe_mul_inner(t, g, k);
  c[0] = t[6];
  c[1] = t[7];
}

struct glv_half {
  u64 d[3];  /* |k_i|, ≤ 2^129 */
  u64 neg;   /* 1 if k_i < 0 */
};

__device__ inline void glv_split(const sc &k, glv_half &h1, glv_half &h2) {
  u64 c1[4] = {0, 0, 0, 0}, c2[4] = {0, 0, 0, 0};
  mul_shift384(c1, GLV_G1, k.d);
  mul_shift384(c2, GLV_G2, k.d);
  u64 t1[4], t2[4], k1[4], k2[4];
  /* k1 = k − c1·a1 − c2·a2 (mod 2^256) */
  mul_low256(t1, c1, GLV_A1);
  mul_low256(t2, c2, GLV_A2);
  {
    u64 borrow = 0;
#pragma unroll
    for (int i = 0; i < 4; i++) k1[i] = subb(k.d[i], t1[i], borrow);
    borrow = 0;
#pragma unroll
    for (int i = 0; i < 4; i++) k1[i] = subb(k1[i], t2[i], borrow);
  }
  /* k2 = −(c1·b1 + c2·b2), all mod 2^256 two's complement */
  mul_low256(t1, c1, GLV_B1);
  mul_low256(t2, c2, GLV_B2);
  {
    u64 carry = 0;
#pragma unroll
    for (int i = 0; i < 4; i++) t1[i] = addc(t1[i], t2[i], carry);
    u64 borrow = 0;
#pragma unroll
    for (int i = 0; i < 4; i++) k2[i] = subb(0, t1[i], borrow);
  }
  /* two's-complement abs */
  u64 n1 = k1[3] >> 63, n2 = k2[3] >> 63;
  u64 carry = 1;
#pragma unroll
  for (int i = 0; i < 4; i++) {
    u64 v = n1 ? ~k1[i] : k1[i];
    u128 s = (u128)v + (n1 ? carry : 0);
    if (n1) {
      k1[i] = (u64)s;
      carry = (u64)(s >> 64);
    } else {
      k1[i] = v;
    }
  }
  carry = 1;
#pragma unroll
  for (int i = 0; i < 4; i++) {
    u64 v = n2 ? ~k2[i] : k2[i];
    u128 s = (u128)v + (n2 ? carry : 0);
    if (n2) {
      k2[i] = (u64)s;
      carry = (u64)(s >> 64);
    } else {
      k2[i] = v;
    }
  }
  h1.d[0] = k1[0]; h1.d[1] = k1[1]; h1.d[2] = k1[2]; h1.neg = n1;
  h2.d[0] = k2[0]; h2.d[1] = k2[1]; h2.d[2] = k2[2]; h2.neg = n2;
}

__device__ __forceinline__ u64 glv_digit(const glv_half &h, int w) {
  int bit = w * 4;
  int limb = bit >> 6, sh = bit & 63;
  u64 v = h.d[limb] >> sh;
  if (sh > 60 && limb < 2) v |= h.d[limb + 1] << (64 - sh);
  return v & 15;
}

/* 8-bit digit at 4-bit-window position w (w even): bits [4w, 4w+8) */
__device__ __forceinline__ u64 glv_digit8(const glv_half &h, int w) {
  int bit = w * 4;
  int limb = bit >> 6, sh = bit & 63;
  u64 v = h.d[limb] >> sh;
  if (sh > 56 && limb < 2) v |= h.d[limb + 1] << (64 - sh);
  return v & 255;
}

/* Signed fixed-window P digits are the DEFAULT (measured 43.8M vs 39.1M
 * verifies/s: the 8-entry table halves the per-lane scratch table and its
 * build cost); -DKV_NO_SIGNED_PTAB restores unsigned 15-entry windows. */
#ifndef KV_NO_SIGNED_PTAB
#define KV_SIGNED_PTAB 1
#endif

#ifdef KV_SIGNED_PTAB
/* signed fixed 4-bit recode: 33 digits in [-8, 8] (LSB order). |h| < 2^130,
 * so the top digit (<= 4) absorbs the final carry without overflow. */
__device__ __forceinline__ void glv_recode_signed(const glv_half &h,
                                                  int8_t dig[33]) {
  u64 carry = 0;
#pragma unroll 1
  for (int w = 0; w < 33; w++) {
    u64 v = glv_digit(h, w) + carry;
    carry = v > 8;
    dig[w] = (int8_t)((long long)v - (long long)(carry << 4));
  }
}
#endif

/* One full ladder window in ONE call frame: 4 doublings + the 4 stream adds
 * (G, φG, P, φP). The group ops inline INSIDE this body, so the accumulator
 * crosses the noinline ABI once per window instead of five times — the r02
 * PMC showed the per-call scratch spills of R/temps were the kernel's
 * dominant memory traffic (~119KB/verify). Full kernel-wide inlining is not
 * an option: it reproducibly hangs gfx950 (measured again this round); this
 * bounded body (~5k instructions) stays under that cliff. */
__device__ KV_GROUP_ATTR void gej_window_step(gej &R, const ge *ptab,
                                              const fe26 &beta, u64 g_active,
                                              u64 dg1, u64 ng1,
                                              u64 dg2, u64 ng2, u64 dp1, u64 np1,
                                              u64 dp2, u64 np2) {
  gej t;
  gej_double_impl(t, R);
  gej_double_impl(R, t);
  gej_double_impl(t, R);
  gej_double_impl(R, t);
  if (g_active) { /* wave-uniform: G adds land on every second window */
  /* G stream (8-bit fixed-base digits) */
  {
    ge e = KV_G_TABLE8[dg1];
    fe26 ny;
    fe26_neg(ny, e.y, 2);
    fe26_cmov(e.y, ny, (u32)ng1);
    gej_add_ge_impl(t, R, e);
    gej_cmov(R, t, (u64)(dg1 != 0));
  }
  /* φG stream: (β·x, ±y) */
  {
    ge e = KV_G_TABLE8[dg2];
    fe26 bx;
    fe26_mul(bx, e.x, beta);
    e.x = bx;
    fe26 ny;
    fe26_neg(ny, e.y, 2);
    fe26_cmov(e.y, ny, (u32)ng2);
    gej_add_ge_impl(t, R, e);
    gej_cmov(R, t, (u64)(dg2 != 0));
  }
  } /* g_active */
  /* P stream (mixed add vs the affine per-lane table) */
  {
    ge e = ptab[dp1];
    fe26 ny;
    fe26_neg(ny, e.y, 2);
    fe26_cmov(e.y, ny, (u32)np1);
    gej_add_ge_impl(t, R, e);
    gej_cmov(R, t, (u64)(dp1 != 0));
  }
  /* φP stream: (β·x, ±y) */
  {
    ge e = ptab[dp2];
    fe26 bx;
    fe26_mul(bx, e.x, beta);
    e.x = bx;
    fe26 ny;
    fe26_neg(ny, e.y, 2);
    fe26_cmov(e.y, ny, (u32)np2);
    gej_add_ge_impl(t, R, e);
    gej_cmov(R, t, (u64)(dp2 != 0));
  }
}

/* Pair-window frames are the DEFAULT (measured 41.1M vs 39.6M verifies/s);
 * -DKV_NO_PAIR_WINDOWS restores one-window frames for experiments. */
#ifndef KV_NO_PAIR_WINDOWS
#define KV_PAIR_WINDOWS 1
#endif

#ifdef KV_PAIR_WINDOWS
/* Two ladder windows per call frame: 8 doublings + both windows' stream adds
 * (G digits are 8-bit and land only on the even window). Halves the
 * accumulator's ABI crossings (17 frames vs 33) at the cost of a ~2x body —
 * under the gfx950 long-body hang cliff (full inlining still hangs). */
__device__ __forceinline__ void gej_window_pair_impl(
    gej &R, const ge *ptab, const fe26 &beta, u64 dg1, u64 ng1, u64 dg2, u64 ng2,
    u64 hp1, u64 hn1, u64 hp2, u64 hn2, /* odd-window P digits + negs */
    u64 lp1, u64 ln1, u64 lp2, u64 ln2 /* even-window P digits + negs */) {
  gej t;
  gej_double_impl(t, R);
  gej_double_impl(R, t);
  gej_double_impl(t, R);
  gej_double_impl(R, t);
  { /* odd window: P streams only */
    ge e = ptab[hp1];
    fe26 ny;
    fe26_neg(ny, e.y, 2);
    fe26_cmov(e.y, ny, (u32)hn1);
    gej_add_ge_impl(t, R, e);
    gej_cmov(R, t, (u64)(hp1 != 0));
  }
  {
    ge e = ptab[hp2];
    fe26 bx;
    fe26_mul(bx, e.x, beta);
    e.x = bx;
    fe26 ny;
    fe26_neg(ny, e.y, 2);
    fe26_cmov(e.y, ny, (u32)hn2);
    gej_add_ge_impl(t, R, e);
    gej_cmov(R, t, (u64)(hp2 != 0));
  }
  gej_double_impl(t, R);
  gej_double_impl(R, t);
  gej_double_impl(t, R);
  gej_double_impl(R, t);
  { /* even window: G comb + P streams */
    ge e = KV_G_TABLE8[dg1];
    fe26 ny;
    fe26_neg(ny, e.y, 2);
    fe26_cmov(e.y, ny, (u32)ng1);
    gej_add_ge_impl(t, R, e);
    gej_cmov(R, t, (u64)(dg1 != 0));
  }
  {
    ge e = KV_G_TABLE8[dg2];
    fe26 bx;
    fe26_mul(bx, e.x, beta);
    e.x = bx;
    fe26 ny;
    fe26_neg(ny, e.y, 2);
    fe26_cmov(e.y, ny, (u32)ng2);
    gej_add_ge_impl(t, R, e);
    gej_cmov(R, t, (u64)(dg2 != 0));
  }
  {
    ge e = ptab[lp1];
    fe26 ny;
    fe26_neg(ny, e.y, 2);
    fe26_cmov(e.y, ny, (u32)ln1);
    gej_add_ge_impl(t, R, e);
    gej_cmov(R, t, (u64)(lp1 != 0));
  }
  {
    ge e = ptab[lp2];
    fe26 bx;
    fe26_mul(bx, e.x, beta);
    e.x = bx;
    fe26 ny;
    fe26_neg(ny, e.y, 2);
    fe26_cmov(e.y, ny, (u32)ln2);
    gej_add_ge_impl(t, R, e);
    gej_cmov(R, t, (u64)(lp2 != 0));
  }
}

__device__ KV_GROUP_ATTR void gej_window_step2(gej &R, const ge *ptab,
                                               const fe26 &beta, u64 dg1, u64 ng1,
                                               u64 dg2, u64 ng2, u64 hp1, u64 hn1,
                                               u64 hp2, u64 hn2, u64 lp1, u64 ln1,
                                               u64 lp2, u64 ln2) {
  gej_window_pair_impl(R, ptab, beta, dg1, ng1, dg2, ng2, hp1, hn1, hp2, hn2,
                       lp1, ln1, lp2, ln2);
}

#ifdef KV_QUAD_WINDOWS
/* four windows (two pairs) per frame: 9 frames per ladder */
__device__ KV_GROUP_ATTR void gej_window_step4(gej &R, const ge *ptab,
                                               const fe26 &beta,
                                               const u64 a[12], const u64 b[12]) {
  gej_window_pair_impl(R, ptab, beta, a[0], a[1], a[2], a[3], a[4], a[5], a[6],
                       a[7], a[8], a[9], a[10], a[11]);
  gej_window_pair_impl(R, ptab, beta, b[0], b[1], b[2], b[3], b[4], b[5], b[6],
                       b[7], b[8], b[9], b[10], b[11]);
}
#endif
#endif

/* R = gs·G + ps·P via GLV-split 4-bit windows: 33 window steps of 4 doublings
 * + 4 selected adds (G, φG, P, φP streams; φ applied at add time as one β·x
 * field multiply; negative half-scalars negate the added point's y). */
__device__ inline void ecmult_double(gej &R, const sc &gs, const sc &ps,
                                     const ge &P) {
  glv_half g1h, g2h, p1h, p2h;
  glv_split(gs, g1h, g2h);
  glv_split(ps, p1h, p2h);
  /* Per-lane P table in AFFINE form, batch-inverted with Montgomery's trick.
   * The round-1 Jacobian table kept ~252MB of in-flight per-lane tables
   * (16 entries x 120B x 131k resident lanes) — right at the 256MB MALL
   * capacity — and its dynamic-indexed scratch loads paid ~44KB of fetch
   * per verify (r01 PMC). Affine entries are 80B (working set ~168MB, fits
   * MALL), and the P streams become MIXED adds (11 vs 16 fe_muls); the one
   * extra field inversion (+~360 fe_muls with the back-substitution) is
   * paid back by the 66 cheaper ladder adds (-330). */
#ifdef KV_SIGNED_PTAB
#define KV_PTAB_MAX 8
#else
#define KV_PTAB_MAX 15
#endif
  ge ptab[KV_PTAB_MAX + 1];
  {
    fe26 ztab[KV_PTAB_MAX + 1], pref[KV_PTAB_MAX + 1];
    gej acc;
    acc.x = P.x;
    acc.y = P.y;
    fe26_set_int(acc.z, 1);
    ptab[1] = P;
#pragma unroll 1
    for (int k = 2; k <= KV_PTAB_MAX; k++) {
      gej t;
      gej_add_ge(t, acc, P);
      acc = t;
      ptab[k].x = acc.x;
      ptab[k].y = acc.y;
      ztab[k] = acc.z;
    }
    /* batch-invert z2..zmax: one fe26_inv + 3 muls per entry */
    pref[2] = ztab[2];
#pragma unroll 1
    for (int k = 3; k <= KV_PTAB_MAX; k++) fe26_mul(pref[k], pref[k - 1], ztab[k]);
    fe26 inv;
    fe26_inv(inv, pref[KV_PTAB_MAX]);
#pragma unroll 1
    for (int k = KV_PTAB_MAX; k >= 2; k--) {
      fe26 zi;
      if (k > 2) {
        fe26_mul(zi, inv, pref[k - 1]);
        fe26_mul(inv, inv, ztab[k]);
      } else {
        zi = inv;
      }
      fe26 zi2, zi3;
      fe26_sqr(zi2, zi);
      fe26_mul(zi3, zi2, zi);
      fe26_mul(ptab[k].x, ptab[k].x, zi2);
      fe26_mul(ptab[k].y, ptab[k].y, zi3);
    }
    ptab[0] = ptab[1]; /* digit 0: add computed then discarded by cmov — a
                          well-formed point keeps magnitudes in range */
  }
#ifdef KV_SIGNED_PTAB
  int8_t dp1[33], dp2[33];
  glv_recode_signed(p1h, dp1);
  glv_recode_signed(p2h, dp2);
#endif
  fe26 beta;
  {
    fe bu;
#pragma unroll
    for (int i = 0; i < 4; i++) bu.n[i] = GLV_BETA[i];
    fe26_from_fe(beta, bu);
  }
  gej_set_infinity(R);
  /* per-stream digit/neg accessors: unsigned fixed windows by default,
   * signed fixed windows (8-entry table) under KV_SIGNED_PTAB */
#ifdef KV_SIGNED_PTAB
#define KV_DP1(w) ((u64)(dp1[w] < 0 ? -dp1[w] : dp1[w]))
#define KV_DN1(w) ((u64)((dp1[w] < 0) ^ (int)p1h.neg))
#define KV_DP2(w) ((u64)(dp2[w] < 0 ? -dp2[w] : dp2[w]))
#define KV_DN2(w) ((u64)((dp2[w] < 0) ^ (int)p2h.neg))
#else
#define KV_DP1(w) glv_digit(p1h, w)
#define KV_DN1(w) (p1h.neg)
#define KV_DP2(w) glv_digit(p2h, w)
#define KV_DN2(w) (p2h.neg)
#endif
#if defined(KV_QUAD_WINDOWS)
  /* window 32 alone, then 8 quad frames (two pairs each) */
  gej_window_step(R, ptab, beta, 1, glv_digit8(g1h, 32), g1h.neg,
                  glv_digit8(g2h, 32), g2h.neg, KV_DP1(32), KV_DN1(32),
                  KV_DP2(32), KV_DN2(32));
#pragma unroll 1
  for (int w = 31; w >= 3; w -= 4) {
    u64 a[12] = {glv_digit8(g1h, w - 1), g1h.neg, glv_digit8(g2h, w - 1),
                 g2h.neg, KV_DP1(w), KV_DN1(w), KV_DP2(w), KV_DN2(w),
                 KV_DP1(w - 1), KV_DN1(w - 1), KV_DP2(w - 1), KV_DN2(w - 1)};
    u64 b[12] = {glv_digit8(g1h, w - 3), g1h.neg, glv_digit8(g2h, w - 3),
                 g2h.neg, KV_DP1(w - 2), KV_DN1(w - 2), KV_DP2(w - 2), KV_DN2(w - 2),
                 KV_DP1(w - 3), KV_DN1(w - 3), KV_DP2(w - 3), KV_DN2(w - 3)};
    gej_window_step4(R, ptab, beta, a, b);
  }
#elif defined(KV_PAIR_WINDOWS)
  /* window 32 alone (G digit at even position 32), then 16 window pairs */
  gej_window_step(R, ptab, beta, 1, glv_digit8(g1h, 32), g1h.neg,
                  glv_digit8(g2h, 32), g2h.neg, KV_DP1(32), KV_DN1(32),
                  KV_DP2(32), KV_DN2(32));
#pragma unroll 1
  for (int w = 31; w >= 1; w -= 2) {
    gej_window_step2(R, ptab, beta,
                     glv_digit8(g1h, w - 1), g1h.neg,
                     glv_digit8(g2h, w - 1), g2h.neg,
                     KV_DP1(w), KV_DN1(w), KV_DP2(w), KV_DN2(w),
                     KV_DP1(w - 1), KV_DN1(w - 1), KV_DP2(w - 1), KV_DN2(w - 1));
  }
#else
#pragma unroll 1
  for (int w = 32; w >= 0; w--) {
    u64 g_active = (w & 1) == 0; /* 8-bit G digits at even window positions */
    gej_window_step(R, ptab, beta, g_active,
                    g_active ? glv_digit8(g1h, w) : 0, g1h.neg,
                    g_active ? glv_digit8(g2h, w) : 0, g2h.neg,
                    KV_DP1(w), KV_DN1(w), KV_DP2(w), KV_DN2(w));
  }
#endif
#undef KV_DP1
#undef KV_DN1
#undef KV_DP2
#undef KV_DN2
#undef KV_PTAB_MAX
}

/* One BIP-340 verification; returns KVS_* */
__device__ inline uint8_t schnorr_verify_one(const uint8_t *rb, const uint8_t *sb,
                                             const uint8_t *pkb, const uint8_t *msg) {
  ge P;
  if (!lift_x_even(P, pkb)) return KVS_BAD_PUBKEY;
  fe rxu;
  fe_from_be(rxu, rb);
  if (fe_gte_p_full(rxu)) return KVS_INVALID;
  fe26 rx;
  fe26_from_fe(rx, rxu);
  sc s;
  if (sc_from_be(s, sb)) return KVS_INVALID;
  uint8_t eh[32];
  sha256_tagged96(BIP340_CHALLENGE_MID, rb, pkb, msg, eh);
  sc e, ne;
  sc_from_be(e, eh);
  sc_neg(ne, e);
  gej R;
  ecmult_double(R, s, ne, P);
  if (gej_is_infinity(R)) return KVS_INVALID;
  /* affine via one inversion: need x == r and even y */
  fe26 zi, zi2, zi3, xa, ya;
  fe26_inv(zi, R.z);
  fe26_sqr(zi2, zi);
  fe26_mul(zi3, zi2, zi);
  fe26_mul(xa, R.x, zi2);
  fe26_mul(ya, R.y, zi3);
  fe26_normalize(ya);
  if (ya.l[0] & 1) return KVS_INVALID;
  return fe26_eq(xa, rx) ? KVS_VALID : KVS_INVALID;
}

#ifndef KV_LB
/* default: 3 blocks/CU minimum. The window-fused ladder fits ~124 VGPRs, so
 * 3 waves/SIMD hide more of the residual stalls — measured 36.2M vs 34.7M
 * (2 blocks) and 32.1M (4 blocks) schnorr verifies/s at 1M tuples. */
#define KV_LB __launch_bounds__(256, 3)
#endif
extern "C" __global__ void KV_LB kv_schnorr_verify_kernel(const uint8_t *__restrict__ tuples,
                                                    unsigned long long n,
                                                    unsigned long long *__restrict__ bitmap,
                                                    uint8_t *__restrict__ status) {
  unsigned long long i = (unsigned long long)blockIdx.x * blockDim.x + threadIdx.x;
  int valid = 0;
  if (i < n) {
    const uint8_t *t = tuples + i * 128;
    uint8_t st = schnorr_verify_one(t, t + 32, t + 64, t + 96);
    if (status) status[i] = st;
    valid = (st == KVS_VALID);
  }
  unsigned long long mask = __ballot(valid);
  if ((threadIdx.x & 63) == 0) {
    unsigned long long word = i / 64; /* lane0's index is 64-aligned */
    if (i < n) bitmap[word] = mask;
  }
}

/* ---------------- ECDSA ---------------- */

__device__ inline uint8_t ecdsa_verify_one(const uint8_t *rb, const uint8_t *sb,
                                           const uint8_t *pk33, const uint8_t *msg) {
  ge P;
  if (!parse_compressed(P, pk33)) return KVS_BAD_PUBKEY;
  sc r, s;
  if (sc_from_be(r, rb)) return KVS_BAD_SIG;  /* parse_compact overflow */
  if (sc_from_be(s, sb)) return KVS_BAD_SIG;
  if (sc_is_zero(r) || sc_is_zero(s)) return KVS_INVALID;
  /* libsecp256k1 secp256k1_ecdsa_verify rejects high-S */
  static const u64 NHALF[4] = {0xDFE92F46681B20A0ULL, 0x5D576E7357A4501DULL,
                               0xFFFFFFFFFFFFFFFFULL, 0x7FFFFFFFFFFFFFFFULL};
  int high = 0;
  for (int i = 3; i >= 0; i--) {
    if (s.d[i] > NHALF[i]) { high = 1; break; }
    if (s.d[i] < NHALF[i]) break;
  }
  if (high) return KVS_INVALID;
  sc z;
  sc_from_be(z, msg); /* reduced mod n like secp256k1_scalar_set_b32 */
  sc w, u1, u2;
  sc_inv(w, s);
  sc_mul(u1, z, w);
  sc_mul(u2, r, w);
  gej R;
  ecmult_double(R, u1, u2, P);
  if (gej_is_infinity(R)) return KVS_INVALID;
  /* x(R) ≡ r (mod n): X == (r + k·n)·Z² for k ∈ {0,1} with r+n < p */
  fe26 z2;
  fe26_sqr(z2, R.z);
  fe rf;
  rf.n[0] = r.d[0];
  rf.n[1] = r.d[1];
  rf.n[2] = r.d[2];
  rf.n[3] = r.d[3];
  fe26 rf26, t;
  fe26_from_fe(rf26, rf);
  fe26_mul(t, rf26, z2);
  if (fe26_eq(t, R.x)) return KVS_VALID;
  /* r + n */
  u64 carry = 0;
  fe rn;
  rn.n[0] = addc(rf.n[0], KV_N0, carry);
  rn.n[1] = addc(rf.n[1], KV_N1, carry);
  rn.n[2] = addc(rf.n[2], KV_N2, carry);
  rn.n[3] = addc(rf.n[3], KV_N3, carry);
  if (!carry && !fe_gte_p_full(rn)) {
    fe26 rn26;
    fe26_from_fe(rn26, rn);
    fe26_mul(t, rn26, z2);
    if (fe26_eq(t, R.x)) return KVS_VALID;
  }
  return KVS_INVALID;
}

extern "C" __global__ void KV_LB kv_ecdsa_verify_kernel(const uint8_t *__restrict__ tuples,
                                                  unsigned long long n,
                                                  unsigned long long *__restrict__ bitmap,
                                                  uint8_t *__restrict__ status) {
  unsigned long long i = (unsigned long long)blockIdx.x * blockDim.x + threadIdx.x;
  int valid = 0;
  if (i < n) {
    const uint8_t *t = tuples + i * 132; /* r32‖s32‖pk33‖msg32‖pad3 */
    uint8_t st = ecdsa_verify_one(t, t + 32, t + 64, t + 97);
    if (status) status[i] = st;
    valid = (st == KVS_VALID);
  }
  unsigned long long mask = __ballot(valid);
  if ((threadIdx.x & 63) == 0 && i < n) bitmap[i / 64] = mask;
}

} // namespace kv
