/* ORACLE (test infrastructure only).
 *
 * secp256k1 from scratch: field/scalar arithmetic, Jacobian point ops, BIP-340
 * Schnorr verify/sign and ECDSA verify/sign.
 *
 * The reference's implementation is the vendored libsecp256k1 C library inside
 * the secp256k1-sys 0.10.1 crate (NOT present under /root/reference). Parity at
 * this boundary is pinned by the reference's own embedded mainnet-signature
 * transactions (tests/golden/mainnet_txs.json ← consensus/src/processes/
 * transaction_validator/tx_validation_in_utxo_context.rs:489-1041) and by
 * sign→verify round trips. Behavioral notes restated from libsecp256k1 0.10.x:
 *   - xonly pubkey parse fails if x >= p or x not on curve     (→ InvalidPubkey)
 *   - compressed pubkey parse (02/03) fails likewise           (→ InvalidPubkey)
 *   - ecdsa compact parse fails if r >= n or s >= n            (→ InvalidSignature)
 *   - ecdsa verify rejects high-S (non-normalized) signatures  (→ bool false)
 *   - schnorr verify: r >= p or s >= n → bool false (no parse error)
 */
#include "oracle.h"
#include <pthread.h>
#include <string.h>

typedef unsigned __int128 u128;
typedef uint64_t u64;

/* ---------------- 256-bit field mod p = 2^256 - 0x1000003D1 ---------------- */

typedef struct { u64 n[4]; } fe;

static const u64 P_LIMB[4] = {0xFFFFFFFEFFFFFC2FULL, 0xFFFFFFFFFFFFFFFFULL,
                              0xFFFFFFFFFFFFFFFFULL, 0xFFFFFFFFFFFFFFFFULL};
#define P_C 0x1000003D1ULL /* 2^256 - p */

static int fe_cmp_p(const fe *a) {
  for (int i = 3; i >= 0; i--) {
    if (a->n[i] < P_LIMB[i]) return -1;
    if (a->n[i] > P_LIMB[i]) return 1;
  }
  return 0;
}

static void fe_sub_p(fe *a) {
  u64 borrow = 0;
  for (int i = 0; i < 4; i++) {
    u64 bi = P_LIMB[i] + borrow;
    u64 nb = (bi < borrow) || (a->n[i] < bi);
    a->n[i] -= bi;
    borrow = nb;
  }
}

static void fe_from_bytes(fe *r, const uint8_t b[32]) { /* big-endian */
  for (int i = 0; i < 4; i++) {
    u64 w = 0;
    for (int j = 0; j < 8; j++) w = (w << 8) | b[8 * (3 - i) + j];
    r->n[i] = w;
  }
}

static void fe_to_bytes(uint8_t b[32], const fe *a) {
  for (int i = 0; i < 4; i++)
    for (int j = 0; j < 8; j++) b[8 * (3 - i) + j] = (uint8_t)(a->n[i] >> (56 - 8 * j));
}

static int fe_is_zero(const fe *a) { return !(a->n[0] | a->n[1] | a->n[2] | a->n[3]); }
static int fe_eq(const fe *a, const fe *b) {
  return a->n[0] == b->n[0] && a->n[1] == b->n[1] && a->n[2] == b->n[2] && a->n[3] == b->n[3];
}
static int fe_is_odd(const fe *a) { return (int)(a->n[0] & 1); }

static void fe_add(fe *r, const fe *a, const fe *b) {
  u128 c = 0;
  for (int i = 0; i < 4; i++) {
    c += (u128)a->n[i] + b->n[i];
    r->n[i] = (u64)c;
    c >>= 64;
  }
  if (c) { /* wrapped 2^256: += P_C */
    u128 c2 = P_C;
    for (int i = 0; i < 4 && c2; i++) {
      c2 += r->n[i];
      r->n[i] = (u64)c2;
      c2 >>= 64;
    }
  }
  if (fe_cmp_p(r) >= 0) fe_sub_p(r);
}

static void fe_neg(fe *r, const fe *a) {
  if (fe_is_zero(a)) { *r = *a; return; }
  u64 borrow = 0;
  for (int i = 0; i < 4; i++) {
    u64 ai = a->n[i] + borrow;
    u64 nb = (ai < borrow) || (P_LIMB[i] < ai);
    r->n[i] = P_LIMB[i] - ai;
    borrow = nb;
  }
  (void)borrow;
}

static void fe_sub(fe *r, const fe *a, const fe *b) {
  fe nb;
  fe_neg(&nb, b);
  fe_add(r, a, &nb);
}

/* 512-bit → mod p reduction: r = lo + hi * P_C, iterate */
static void fe_reduce8(fe *r, const u64 t[8]) {
  u64 lo[5];
  /* first fold */
  u128 c = 0;
  for (int i = 0; i < 4; i++) {
    c += (u128)t[i] + (u128)t[4 + i] * P_C;
    lo[i] = (u64)c;
    c >>= 64;
  }
  lo[4] = (u64)c; /* <= ~2^33+ */
  /* second fold: lo4 * P_C into lo0.. */
  c = (u128)lo[4] * P_C;
  for (int i = 0; i < 4 && c; i++) {
    c += lo[i];
    lo[i] = (u64)c;
    c >>= 64;
  }
  /* c can only be nonzero if the add overflowed past limb 3 — then value wrapped
   * 2^256 exactly once more */
  if (c) {
    u128 c2 = P_C;
    for (int i = 0; i < 4 && c2; i++) {
      c2 += lo[i];
      lo[i] = (u64)c2;
      c2 >>= 64;
    }
  }
  memcpy(r->n, lo, 32);
  if (fe_cmp_p(r) >= 0) fe_sub_p(r);
}

static void fe_mul(fe *r, const fe *a, const fe *b) {
  u64 t[8] = {0};
  for (int i = 0; i < 4; i++) {
    u64 carry = 0;
    for (int j = 0; j < 4; j++) {
      u128 cur = (u128)a->n[i] * b->n[j] + t[i + j] + carry;
      t[i + j] = (u64)cur;
      carry = (u64)(cur >> 64);
    }
    t[i + 4] = carry;
  }
  fe_reduce8(r, t);
}

static void fe_sqr(fe *r, const fe *a) { fe_mul(r, a, a); }

static void fe_mul_int(fe *r, const fe *a, u64 k) {
  u64 t[8] = {0};
  u64 carry = 0;
  for (int i = 0; i < 4; i++) {
    u128 cur = (u128)a->n[i] * k + carry;
    t[i] = (u64)cur;
    carry = (u64)(cur >> 64);
  }
  t[4] = carry;
  fe_reduce8(r, t);
}

/* a^e mod p via square-and-multiply over big-endian exponent bytes */
static void fe_pow(fe *r, const fe *a, const uint8_t e[32]) {
  fe result = {{1, 0, 0, 0}}, base = *a;
  for (int i = 255; i >= 0; i--) {
    fe_sqr(&result, &result);
    if ((e[31 - i / 8] >> (i % 8)) & 1) fe_mul(&result, &result, &base);
  }
  /* note: loop above squares even before first set bit — harmless (1^2=1) */
  *r = result;
}

static void fe_inv(fe *r, const fe *a) {
  /* p - 2 */
  static const uint8_t pm2[32] = {0xFF, 0xFF, 0xFF, 0xFF, 0xFF, 0xFF, 0xFF, 0xFF,
                                  0xFF, 0xFF, 0xFF, 0xFF, 0xFF, 0xFF, 0xFF, 0xFF,
                                  0xFF, 0xFF, 0xFF, 0xFF, 0xFF, 0xFF, 0xFF, 0xFF,
                                  0xFF, 0xFF, 0xFF, 0xFE, 0xFF, 0xFF, 0xFC, 0x2D};
  fe_pow(r, a, pm2);
}

/* sqrt via a^((p+1)/4); returns 1 if square (r^2 == a) */
static int fe_sqrt(fe *r, const fe *a) {
  static const uint8_t e[32] = {0x3F, 0xFF, 0xFF, 0xFF, 0xFF, 0xFF, 0xFF, 0xFF,
                                0xFF, 0xFF, 0xFF, 0xFF, 0xFF, 0xFF, 0xFF, 0xFF,
                                0xFF, 0xFF, 0xFF, 0xFF, 0xFF, 0xFF, 0xFF, 0xFF,
                                0xFF, 0xFF, 0xFF, 0xFF, 0xBF, 0xFF, 0xFF, 0x0C};
  fe cand;
  fe_pow(&cand, a, e);
  fe sq;
  fe_sqr(&sq, &cand);
  if (!fe_eq(&sq, a)) return 0;
  *r = cand;
  return 1;
}

/* ---------------- scalar mod n ---------------- */

typedef struct { u64 d[4]; } sc;

static const u64 N_LIMB[4] = {0xBFD25E8CD0364141ULL, 0xBAAEDCE6AF48A03BULL,
                              0xFFFFFFFFFFFFFFFEULL, 0xFFFFFFFFFFFFFFFFULL};
/* 2^256 - n */
static const u64 NC_LIMB[3] = {0x402DA1732FC9BEBFULL, 0x4551231950B75FC4ULL, 1ULL};
/* n/2 (floor) */
static const u64 NHALF[4] = {0xDFE92F46681B20A0ULL, 0x5D576E7357A4501DULL,
                             0xFFFFFFFFFFFFFFFFULL, 0x7FFFFFFFFFFFFFFFULL};

static int sc_cmp_n(const sc *a) {
  for (int i = 3; i >= 0; i--) {
    if (a->d[i] < N_LIMB[i]) return -1;
    if (a->d[i] > N_LIMB[i]) return 1;
  }
  return 0;
}

static void sc_sub_n(sc *a) {
  u64 borrow = 0;
  for (int i = 0; i < 4; i++) {
    u64 bi = N_LIMB[i] + borrow;
    u64 nb = (bi < borrow) || (a->d[i] < bi);
    a->d[i] -= bi;
    borrow = nb;
  }
}

static int sc_is_zero(const sc *a) { return !(a->d[0] | a->d[1] | a->d[2] | a->d[3]); }

static int sc_is_high(const sc *a) { /* a > n/2 */
  for (int i = 3; i >= 0; i--) {
    if (a->d[i] > NHALF[i]) return 1;
    if (a->d[i] < NHALF[i]) return 0;
  }
  return 0; /* equal → not high */
}

/* returns overflow flag (input >= n) */
static int sc_from_bytes(sc *r, const uint8_t b[32]) {
  for (int i = 0; i < 4; i++) {
    u64 w = 0;
    for (int j = 0; j < 8; j++) w = (w << 8) | b[8 * (3 - i) + j];
    r->d[i] = w;
  }
  if (sc_cmp_n(r) >= 0) {
    sc_sub_n(r);
    return 1;
  }
  return 0;
}

static void sc_to_bytes(uint8_t b[32], const sc *a) {
  for (int i = 0; i < 4; i++)
    for (int j = 0; j < 8; j++) b[8 * (3 - i) + j] = (uint8_t)(a->d[i] >> (56 - 8 * j));
}

static void sc_add(sc *r, const sc *a, const sc *b) {
  u128 c = 0;
  u64 t[5];
  for (int i = 0; i < 4; i++) {
    c += (u128)a->d[i] + b->d[i];
    t[i] = (u64)c;
    c >>= 64;
  }
  t[4] = (u64)c;
  memcpy(r->d, t, 32);
  if (t[4] || sc_cmp_n(r) >= 0) sc_sub_n(r); /* a,b < n → sum < 2n: one subtract */
}

static void sc_neg(sc *r, const sc *a) {
  if (sc_is_zero(a)) { *r = *a; return; }
  u64 borrow = 0;
  for (int i = 0; i < 4; i++) {
    u64 ai = a->d[i] + borrow;
    u64 nb = (ai < borrow) || (N_LIMB[i] < ai);
    r->d[i] = N_LIMB[i] - ai;
    borrow = nb;
  }
}

/* 512-bit → mod n: fold hi*NC (129-bit) repeatedly */
static void sc_reduce8(sc *r, const u64 t_in[8]) {
  u64 v[9];
  memcpy(v, t_in, 64);
  v[8] = 0;
  int top = 8; /* number of limbs potentially nonzero */
  while (top > 4) {
    /* v = v_lo(4) + v_hi * NC, v_hi has (top-4) limbs */
    int hi_len = top - 4;
    u64 hi[5];
    memcpy(hi, v + 4, hi_len * 8);
    u64 acc[9] = {0};
    memcpy(acc, v, 32); /* lo 4 limbs */
    /* acc += hi * NC (NC has 3 limbs) */
    for (int i = 0; i < hi_len; i++) {
      u64 carry = 0;
      for (int j = 0; j < 3; j++) {
        u128 cur = (u128)hi[i] * NC_LIMB[j] + acc[i + j] + carry;
        acc[i + j] = (u64)cur;
        carry = (u64)(cur >> 64);
      }
      int k = i + 3;
      while (carry) {
        u128 cur = (u128)acc[k] + carry;
        acc[k] = (u64)cur;
        carry = (u64)(cur >> 64);
        k++;
      }
    }
    memcpy(v, acc, 72);
    /* new top: hi_len + 3 is the max limb index+1 of the folded part */
    int nt = hi_len + 3;
    if (nt < 4) nt = 4;
    /* trim leading zeros */
    top = nt > 4 ? nt : 4;
    while (top > 4 && v[top] == 0 && top >= 4) {
      if (v[top] == 0 && top + 1 <= 8) {}
      break;
    }
    /* recompute actual top */
    int actual = 4;
    for (int i = 8; i >= 4; i--)
      if (v[i] != 0) { actual = i + 1; break; }
    top = actual;
  }
  memcpy(r->d, v, 32);
  while (sc_cmp_n(r) >= 0) sc_sub_n(r);
}

static void sc_mul(sc *r, const sc *a, const sc *b) {
  u64 t[8] = {0};
  for (int i = 0; i < 4; i++) {
    u64 carry = 0;
    for (int j = 0; j < 4; j++) {
      u128 cur = (u128)a->d[i] * b->d[j] + t[i + j] + carry;
      t[i + j] = (u64)cur;
      carry = (u64)(cur >> 64);
    }
    t[i + 4] = carry;
  }
  sc_reduce8(r, t);
}

static void sc_inv(sc *r, const sc *a) { /* Fermat: a^(n-2) */
  static const uint8_t nm2[32] = {0xFF, 0xFF, 0xFF, 0xFF, 0xFF, 0xFF, 0xFF, 0xFF,
                                  0xFF, 0xFF, 0xFF, 0xFF, 0xFF, 0xFF, 0xFF, 0xFE,
                                  0xBA, 0xAE, 0xDC, 0xE6, 0xAF, 0x48, 0xA0, 0x3B,
                                  0xBF, 0xD2, 0x5E, 0x8C, 0xD0, 0x36, 0x41, 0x3F};
  sc result = {{1, 0, 0, 0}}, base = *a;
  for (int i = 255; i >= 0; i--) {
    sc_mul(&result, &result, &result);
    if ((nm2[31 - i / 8] >> (i % 8)) & 1) sc_mul(&result, &result, &base);
  }
  *r = result;
}

/* ---------------- group (Jacobian) ---------------- */

typedef struct { fe x, y; int infinity; } ge;
typedef struct { fe x, y, z; int infinity; } gej;

static const fe GE_GX = {{0x59F2815B16F81798ULL, 0x029BFCDB2DCE28D9ULL,
                          0x55A06295CE870B07ULL, 0x79BE667EF9DCBBACULL}};
static const fe GE_GY = {{0x9C47D08FFB10D4B8ULL, 0xFD17B448A6855419ULL,
                          0x5DA4FBFC0E1108A8ULL, 0x483ADA7726A3C465ULL}};

static void gej_set_ge(gej *r, const ge *a) {
  r->x = a->x;
  r->y = a->y;
  r->z = (fe){{1, 0, 0, 0}};
  r->infinity = a->infinity;
}

static void gej_set_infinity(gej *r) {
  memset(r, 0, sizeof(*r));
  r->infinity = 1;
}

static void gej_double(gej *r, const gej *a) {
  if (a->infinity) { *r = *a; return; }
  /* y = 0 cannot happen on secp256k1 (no 2-torsion), skip check */
  fe A, B, C, D, E, F, t;
  fe_sqr(&A, &a->x);          /* A = X^2 */
  fe_sqr(&B, &a->y);          /* B = Y^2 */
  fe_sqr(&C, &B);             /* C = B^2 */
  fe_add(&t, &a->x, &B);
  fe_sqr(&t, &t);
  fe_sub(&t, &t, &A);
  fe_sub(&t, &t, &C);
  fe_add(&D, &t, &t);         /* D = 2((X+B)^2 - A - C) */
  fe_mul_int(&E, &A, 3);      /* E = 3A */
  fe_sqr(&F, &E);             /* F = E^2 */
  fe_sub(&r->x, &F, &D);
  fe_sub(&r->x, &r->x, &D);   /* X' = F - 2D */
  fe_sub(&t, &D, &r->x);
  fe_mul(&t, &E, &t);
  fe C8;
  fe_mul_int(&C8, &C, 8);
  fe ny;
  fe_sub(&ny, &t, &C8);       /* Y' = E(D - X') - 8C */
  fe_mul(&t, &a->y, &a->z);
  fe_add(&r->z, &t, &t);      /* Z' = 2YZ */
  r->y = ny;
  r->infinity = 0;
}

static void gej_add(gej *r, const gej *a, const gej *b) {
  if (a->infinity) { *r = *b; return; }
  if (b->infinity) { *r = *a; return; }
  fe z1z1, z2z2, u1, u2, s1, s2, h, i, j, rr, v, t;
  fe_sqr(&z1z1, &a->z);
  fe_sqr(&z2z2, &b->z);
  fe_mul(&u1, &a->x, &z2z2);
  fe_mul(&u2, &b->x, &z1z1);
  fe_mul(&s1, &a->y, &b->z);
  fe_mul(&s1, &s1, &z2z2);
  fe_mul(&s2, &b->y, &a->z);
  fe_mul(&s2, &s2, &z1z1);
  fe_sub(&h, &u2, &u1);
  fe_sub(&rr, &s2, &s1);
  if (fe_is_zero(&h)) {
    if (fe_is_zero(&rr)) { gej_double(r, a); return; }
    gej_set_infinity(r);
    return;
  }
  fe_add(&rr, &rr, &rr); /* r = 2(S2-S1) */
  fe_add(&i, &h, &h);
  fe_sqr(&i, &i);        /* I = (2H)^2 */
  fe_mul(&j, &h, &i);    /* J = H*I */
  fe_mul(&v, &u1, &i);   /* V = U1*I */
  fe_sqr(&r->x, &rr);
  fe_sub(&r->x, &r->x, &j);
  fe_sub(&r->x, &r->x, &v);
  fe_sub(&r->x, &r->x, &v); /* X3 = r^2 - J - 2V */
  fe_sub(&t, &v, &r->x);
  fe_mul(&t, &rr, &t);
  fe s1j;
  fe_mul(&s1j, &s1, &j);
  fe_add(&s1j, &s1j, &s1j);
  fe ny;
  fe_sub(&ny, &t, &s1j);    /* Y3 = r(V-X3) - 2 S1 J */
  fe zz;
  fe_add(&zz, &a->z, &b->z);
  fe_sqr(&zz, &zz);
  fe_sub(&zz, &zz, &z1z1);
  fe_sub(&zz, &zz, &z2z2);
  fe_mul(&r->z, &zz, &h);   /* Z3 = ((Z1+Z2)^2 - Z1Z1 - Z2Z2) * H */
  r->y = ny;
  r->infinity = 0;
}

static void gej_add_ge(gej *r, const gej *a, const ge *b) {
  gej bj;
  gej_set_ge(&bj, b);
  gej_add(r, a, &bj); /* oracle favors simplicity over the mixed-add formula */
}

static void gej_neg(gej *r, const gej *a) {
  *r = *a;
  fe_neg(&r->y, &a->y);
}

static void ge_neg(ge *r, const ge *a) {
  *r = *a;
  fe_neg(&r->y, &a->y);
}

/* Jacobian → affine (one inversion) */
static void ge_set_gej(ge *r, const gej *a) {
  if (a->infinity) { memset(r, 0, sizeof(*r)); r->infinity = 1; return; }
  fe zi, zi2, zi3;
  fe_inv(&zi, &a->z);
  fe_sqr(&zi2, &zi);
  fe_mul(&zi3, &zi2, &zi);
  fe_mul(&r->x, &a->x, &zi2);
  fe_mul(&r->y, &a->y, &zi3);
  r->infinity = 0;
}

/* ---- wNAF ---- */

static int wnaf(int8_t out[257], const sc *a, int w) {
  /* standard: digits odd in [-(2^(w-1)-1), 2^(w-1)-1], returns bit length */
  u64 k[5] = {a->d[0], a->d[1], a->d[2], a->d[3], 0};
  int len = 0;
  int i = 0;
  memset(out, 0, 257);
  while (k[0] | k[1] | k[2] | k[3] | k[4]) {
    if (k[0] & 1) {
      int word = (int)(k[0] & ((1u << w) - 1));
      if (word > (1 << (w - 1))) word -= (1 << w);
      out[i] = (int8_t)word;
      /* k -= word */
      if (word >= 0) {
        u64 borrow = (u64)word;
        for (int j = 0; j < 5 && borrow; j++) {
          u64 nb = k[j] < borrow;
          k[j] -= borrow;
          borrow = nb;
        }
      } else {
        u64 carry = (u64)(-word);
        for (int j = 0; j < 5 && carry; j++) {
          k[j] += carry;
          carry = k[j] < carry;
        }
      }
      len = i + 1;
    }
    /* shift right 1 */
    for (int j = 0; j < 4; j++) k[j] = (k[j] >> 1) | (k[j + 1] << 63);
    k[4] >>= 1;
    i++;
    if (i > 256) break;
  }
  return len;
}

/* ---- fixed-base G tables (built lazily, thread-safe enough for tests via
 * call-before-threads init) ---- */

#define G_WINDOW 8
#define G_TABLE_SIZE (1 << (G_WINDOW - 1)) /* 128 odd multiples: G,3G,...,255G */
static ge g_table[G_TABLE_SIZE];
static int g_table_ready = 0;
static pthread_mutex_t g_table_mu = PTHREAD_MUTEX_INITIALIZER;

static void ensure_g_table(void) {
  if (__atomic_load_n(&g_table_ready, __ATOMIC_ACQUIRE)) return;
  pthread_mutex_lock(&g_table_mu);
  if (__atomic_load_n(&g_table_ready, __ATOMIC_ACQUIRE)) {
    pthread_mutex_unlock(&g_table_mu);
    return;
  }
  ge g = {GE_GX, GE_GY, 0};
  gej gj, g2;
  gej_set_ge(&gj, &g);
  gej_double(&g2, &gj); /* 2G */
  gej cur = gj;
  for (int i = 0; i < G_TABLE_SIZE; i++) {
    ge_set_gej(&g_table[i], &cur);
    gej next;
    gej_add(&next, &cur, &g2);
    cur = next;
  }
  __atomic_store_n(&g_table_ready, 1, __ATOMIC_RELEASE);
  pthread_mutex_unlock(&g_table_mu);
}

/* r = gs*G + ps*P (either may be NULL) */
static void ecmult(gej *r, const sc *gs, const sc *ps, const ge *P) {
  ensure_g_table();
  int8_t ng[257], np[257];
  int lg = 0, lp = 0;
  if (gs) lg = wnaf(ng, gs, G_WINDOW);
  if (ps) lp = wnaf(np, ps, 5);
  /* P odd multiples table (Jacobian): P, 3P, ..., 15P */
  gej ptab[8];
  if (ps) {
    gej pj, p2;
    gej_set_ge(&pj, P);
    gej_double(&p2, &pj);
    ptab[0] = pj;
    for (int i = 1; i < 8; i++) gej_add(&ptab[i], &ptab[i - 1], &p2);
  }
  int bits = lg > lp ? lg : lp;
  gej_set_infinity(r);
  for (int i = bits - 1; i >= 0; i--) {
    gej t;
    gej_double(&t, r);
    *r = t;
    if (i < lg && ng[i]) {
      int d = ng[i];
      ge e = g_table[(d > 0 ? d : -d) / 2];
      if (d < 0) ge_neg(&e, &e);
      gej_add_ge(&t, r, &e);
      *r = t;
    }
    if (i < lp && np[i]) {
      int d = np[i];
      gej e = ptab[(d > 0 ? d : -d) / 2];
      if (d < 0) { gej tmp; gej_neg(&tmp, &e); e = tmp; }
      gej_add(&t, r, &e);
      *r = t;
    }
  }
}

/* lift x to point with even y; returns 0 if x >= p or not on curve */
static int ge_lift_x_even(ge *r, const uint8_t x32[32]) {
  /* check x < p */
  fe x;
  fe_from_bytes(&x, x32);
  /* detect overflow: re-serialize and compare */
  uint8_t chk[32];
  fe_to_bytes(chk, &x);
  if (memcmp(chk, x32, 32) != 0) return 0; /* x >= p wrapped */
  /* actually fe_from_bytes doesn't reduce; compare against p directly */
  if (fe_cmp_p(&x) >= 0) return 0;
  fe x3, y2, seven = {{7, 0, 0, 0}};
  fe_sqr(&x3, &x);
  fe_mul(&x3, &x3, &x);
  fe_add(&y2, &x3, &seven);
  fe y;
  if (!fe_sqrt(&y, &y2)) return 0;
  if (fe_is_odd(&y)) fe_neg(&y, &y);
  r->x = x;
  r->y = y;
  r->infinity = 0;
  return 1;
}

static int ge_parse_compressed(ge *r, const uint8_t pk33[33]) {
  if (pk33[0] != 0x02 && pk33[0] != 0x03) return 0;
  fe x;
  fe_from_bytes(&x, pk33 + 1);
  if (fe_cmp_p(&x) >= 0) return 0;
  fe x3, y2, seven = {{7, 0, 0, 0}};
  fe_sqr(&x3, &x);
  fe_mul(&x3, &x3, &x);
  fe_add(&y2, &x3, &seven);
  fe y;
  if (!fe_sqrt(&y, &y2)) return 0;
  if (fe_is_odd(&y) != (pk33[0] == 0x03)) fe_neg(&y, &y);
  r->x = x;
  r->y = y;
  r->infinity = 0;
  return 1;
}

/* ---------------- BIP-340 Schnorr ---------------- */

static void tagged_hash(const char *tag, const uint8_t *d1, size_t l1,
                        const uint8_t *d2, size_t l2, const uint8_t *d3, size_t l3,
                        uint8_t out32[32]) {
  uint8_t th[32];
  ok_sha256((const uint8_t *)tag, strlen(tag), th);
  ok_sha256_state S;
  ok_sha256_init(&S);
  ok_sha256_update(&S, th, 32);
  ok_sha256_update(&S, th, 32);
  if (d1) ok_sha256_update(&S, d1, l1);
  if (d2) ok_sha256_update(&S, d2, l2);
  if (d3) ok_sha256_update(&S, d3, l3);
  ok_sha256_final(&S, out32);
}


/* pubkey parse-only checks (XOnlyPublicKey::from_slice /
 * PublicKey::from_slice succeed): 1 = parseable, 0 = not on curve / bad
 * prefix / x >= p. Used by the script engine to order InvalidPubkey before
 * signature-length errors, matching the reference's parse order. */
int ok_xonly_pubkey_valid(const uint8_t pk32[32]) {
  ge P;
  return ge_lift_x_even(&P, pk32) ? 1 : 0;
}

int ok_compressed_pubkey_valid(const uint8_t pk33[33]) {
  ge P;
  return ge_parse_compressed(&P, pk33) ? 1 : 0;
}

int ok_schnorr_verify(const uint8_t pk32[32], const uint8_t msg32[32],
                      const uint8_t sig64[64]) {
  ge P;
  if (!ge_lift_x_even(&P, pk32)) return -1; /* pubkey parse error */
  /* r < p check */
  fe rx;
  fe_from_bytes(&rx, sig64);
  if (fe_cmp_p(&rx) >= 0) return 0;
  /* s < n check */
  sc s;
  if (sc_from_bytes(&s, sig64 + 32)) return 0;
  /* e = tagged_hash(challenge, r||pk||m) mod n */
  uint8_t eh[32];
  tagged_hash("BIP0340/challenge", sig64, 32, pk32, 32, msg32, 32, eh);
  sc e;
  sc_from_bytes(&e, eh);
  sc ne;
  sc_neg(&ne, &e);
  gej R;
  ecmult(&R, &s, &ne, &P); /* R = s*G - e*P */
  if (R.infinity) return 0;
  ge Ra;
  ge_set_gej(&Ra, &R);
  if (fe_is_odd(&Ra.y)) return 0;
  return fe_eq(&Ra.x, &rx) ? 1 : 0;
}

int ok_pubkey_xonly(const uint8_t seckey32[32], uint8_t xonly_out[32]) {
  sc d;
  if (sc_from_bytes(&d, seckey32)) return 0;
  if (sc_is_zero(&d)) return 0;
  gej Pj;
  ecmult(&Pj, &d, NULL, NULL);
  ge P;
  ge_set_gej(&P, &Pj);
  fe_to_bytes(xonly_out, &P.x);
  return 1;
}

int ok_pubkey_compressed(const uint8_t seckey32[32], uint8_t pk33_out[33]) {
  sc d;
  if (sc_from_bytes(&d, seckey32)) return 0;
  if (sc_is_zero(&d)) return 0;
  gej Pj;
  ecmult(&Pj, &d, NULL, NULL);
  ge P;
  ge_set_gej(&P, &Pj);
  pk33_out[0] = fe_is_odd(&P.y) ? 0x03 : 0x02;
  fe_to_bytes(pk33_out + 1, &P.x);
  return 1;
}

int ok_schnorr_sign(const uint8_t seckey32[32], const uint8_t msg32[32],
                    const uint8_t *aux32, uint8_t sig_out[64]) {
  static const uint8_t zeros[32] = {0};
  if (!aux32) aux32 = zeros;
  sc d;
  if (sc_from_bytes(&d, seckey32) || sc_is_zero(&d)) return 0;
  gej Pj;
  ecmult(&Pj, &d, NULL, NULL);
  ge P;
  ge_set_gej(&P, &Pj);
  if (fe_is_odd(&P.y)) { sc t; sc_neg(&t, &d); d = t; }
  uint8_t pkb[32];
  fe_to_bytes(pkb, &P.x);
  uint8_t auxh[32], t[32], db[32];
  tagged_hash("BIP0340/aux", aux32, 32, NULL, 0, NULL, 0, auxh);
  sc_to_bytes(db, &d);
  for (int i = 0; i < 32; i++) t[i] = db[i] ^ auxh[i];
  uint8_t kh[32];
  tagged_hash("BIP0340/nonce", t, 32, pkb, 32, msg32, 32, kh);
  sc k;
  sc_from_bytes(&k, kh);
  if (sc_is_zero(&k)) return 0;
  gej Rj;
  ecmult(&Rj, &k, NULL, NULL);
  ge R;
  ge_set_gej(&R, &Rj);
  if (fe_is_odd(&R.y)) { sc t2; sc_neg(&t2, &k); k = t2; }
  uint8_t rb[32];
  fe_to_bytes(rb, &R.x);
  uint8_t eh[32];
  tagged_hash("BIP0340/challenge", rb, 32, pkb, 32, msg32, 32, eh);
  sc e;
  sc_from_bytes(&e, eh);
  sc ed, s;
  sc_mul(&ed, &e, &d);
  sc_add(&s, &k, &ed);
  memcpy(sig_out, rb, 32);
  sc_to_bytes(sig_out + 32, &s);
  return 1;
}

/* ---------------- ECDSA ---------------- */

int ok_ecdsa_verify(const uint8_t pk33[33], const uint8_t msg32[32],
                    const uint8_t sig64[64]) {
  ge P;
  if (!ge_parse_compressed(&P, pk33)) return -1;
  sc r, s;
  if (sc_from_bytes(&r, sig64)) return -2;      /* r >= n: parse overflow */
  if (sc_from_bytes(&s, sig64 + 32)) return -2; /* s >= n: parse overflow */
  if (sc_is_zero(&r) || sc_is_zero(&s)) return 0;
  if (sc_is_high(&s)) return 0; /* libsecp verify rejects non-low-S */
  sc z;
  sc_from_bytes(&z, msg32);
  sc w, u1, u2;
  sc_inv(&w, &s);
  sc_mul(&u1, &z, &w);
  sc_mul(&u2, &r, &w);
  gej R;
  ecmult(&R, &u1, &u2, &P);
  if (R.infinity) return 0;
  /* check x(R) ≡ r (mod n): X == (r + k*n) * Z^2 for k in {0,1} with r+n < p */
  fe z2;
  fe_sqr(&z2, &R.z);
  uint8_t rb[32];
  sc_to_bytes(rb, &r);
  fe rf;
  fe_from_bytes(&rf, rb);
  fe t;
  fe_mul(&t, &rf, &z2);
  if (fe_eq(&t, &R.x)) return 1;
  /* r + n (fits in field iff r + n < p) */
  u64 carry = 0;
  fe rn = rf;
  for (int i = 0; i < 4; i++) {
    u128 cur = (u128)rn.n[i] + N_LIMB[i] + carry;
    rn.n[i] = (u64)cur;
    carry = (u64)(cur >> 64);
  }
  if (!carry && fe_cmp_p(&rn) < 0) {
    fe_mul(&t, &rn, &z2);
    if (fe_eq(&t, &R.x)) return 1;
  }
  return 0;
}

int ok_ecdsa_sign(const uint8_t seckey32[32], const uint8_t msg32[32],
                  uint8_t sig_out[64]) {
  sc d;
  if (sc_from_bytes(&d, seckey32) || sc_is_zero(&d)) return 0;
  /* deterministic nonce (oracle-only; not a consensus surface) */
  uint8_t kh[32];
  tagged_hash("kaspa-oracle/ecdsa-nonce", seckey32, 32, msg32, 32, NULL, 0, kh);
  sc k;
  sc_from_bytes(&k, kh);
  for (int attempt = 0; attempt < 64; attempt++) {
    if (!sc_is_zero(&k)) {
      gej Rj;
      ecmult(&Rj, &k, NULL, NULL);
      ge R;
      ge_set_gej(&R, &Rj);
      uint8_t xb[32];
      fe_to_bytes(xb, &R.x);
      sc r;
      sc_from_bytes(&r, xb); /* r = x mod n */
      if (!sc_is_zero(&r)) {
        sc z, kinv, rd, sum, s;
        sc_from_bytes(&z, msg32);
        sc_inv(&kinv, &k);
        sc_mul(&rd, &r, &d);
        sc_add(&sum, &z, &rd);
        sc_mul(&s, &kinv, &sum);
        if (!sc_is_zero(&s)) {
          if (sc_is_high(&s)) { sc t; sc_neg(&t, &s); s = t; }
          sc_to_bytes(sig_out, &r);
          sc_to_bytes(sig_out + 32, &s);
          return 1;
        }
      }
    }
    /* retry with k+1 */
    sc one = {{1, 0, 0, 0}};
    sc t;
    sc_add(&t, &k, &one);
    k = t;
  }
  return 0;
}
