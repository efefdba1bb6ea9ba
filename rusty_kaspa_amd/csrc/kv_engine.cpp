/* kv_engine.cpp — host side of the MI355X-native transaction-validation engine
 * (PRODUCT code). Implements the C-ABI in include/kaspa_engine_abi.h.
 *
 * The host is deliberately thin: device memory and stream management, batch
 * staging, and the MuHash finalization (one 3072-bit modular inverse per
 * block batch — host work by design, mirroring how the reference computes the
 * final commitment once per block: crypto/muhash/src/u3072.rs:157-183).
 * ALL per-signature and per-input compute runs in HIP kernels; there is no
 * CPU fallback — calls fail loudly when no GPU is present.
 */
#include "kaspa_engine_abi.h"
#include "kv_u3072.h"

#include <hip/hip_runtime.h>
#include <mutex>
#include <stdio.h>
#include <stdlib.h>
#include <string.h>
#include <string>

/* kernels compiled into this TU (single translation unit keeps the build to one
 * hipcc invocation, no -fgpu-rdc) */
#include "kv_secp_kernels.hip"

static thread_local std::string g_last_error;

static void set_error(const char *msg) { g_last_error = msg ? msg : ""; }

extern "C" const char *kv_last_error(void) { return g_last_error.c_str(); }

#define HIP_CHECK(expr)                                                        \
  do {                                                                         \
    hipError_t err_ = (expr);                                                  \
    if (err_ != hipSuccess) {                                                  \
      set_error(hipGetErrorString(err_));                                      \
      return -2;                                                               \
    }                                                                          \
  } while (0)

struct kv_ctx {
  kv_params params;
  hipStream_t stream;
  std::mutex mu;
  /* grow-on-demand device scratch */
  uint8_t *d_in = nullptr;
  size_t d_in_cap = 0;
  uint64_t *d_bitmap = nullptr;
  size_t d_bitmap_cap = 0;
  uint8_t *d_status = nullptr;
  size_t d_status_cap = 0;
  uint64_t cache_hits = 0, cache_misses = 0, cache_insertions = 0;
};

static int ensure_cap(void **ptr, size_t *cap, size_t need) {
  if (*cap >= need) return 0;
  if (*ptr) (void)hipFree(*ptr);
  size_t newcap = need + need / 2;
  if (hipMalloc(ptr, newcap) != hipSuccess) {
    *ptr = nullptr;
    *cap = 0;
    set_error("hipMalloc failed");
    return -2;
  }
  *cap = newcap;
  return 0;
}

extern "C" kv_ctx *kv_create(const kv_params *params) {
  int count = 0;
  if (hipGetDeviceCount(&count) != hipSuccess || count == 0) {
    set_error("kv_create: no HIP device available (the engine has no CPU fallback)");
    return nullptr;
  }
  int dev = params && params->device >= 0 ? params->device : 0;
  if (hipSetDevice(dev) != hipSuccess) {
    set_error("kv_create: hipSetDevice failed");
    return nullptr;
  }
  kv_ctx *ctx = new kv_ctx();
  if (params) ctx->params = *params;
  else
    ctx->params = kv_params{1000, 1000, 10000, 0};
  if (hipStreamCreate(&ctx->stream) != hipSuccess) {
    set_error("kv_create: hipStreamCreate failed");
    delete ctx;
    return nullptr;
  }
  return ctx;
}

extern "C" void kv_destroy(kv_ctx *ctx) {
  if (!ctx) return;
  if (ctx->d_in) (void)hipFree(ctx->d_in);
  if (ctx->d_bitmap) (void)hipFree(ctx->d_bitmap);
  if (ctx->d_status) (void)hipFree(ctx->d_status);
  (void)hipStreamDestroy(ctx->stream);
  delete ctx;
}

/* ---------------- batched verify entry points ---------------- */

static int verify_batch(kv_ctx *ctx, const uint8_t *tuples, size_t n, size_t rec_size,
                        int ecdsa, uint64_t *bitmap_out, uint8_t *status_out) {
  if (n == 0) return 0;
  std::lock_guard<std::mutex> lk(ctx->mu);
  size_t words = (n + 63) / 64;
  if (ensure_cap((void **)&ctx->d_in, &ctx->d_in_cap, n * rec_size)) return -2;
  if (ensure_cap((void **)&ctx->d_bitmap, &ctx->d_bitmap_cap, words * 8)) return -2;
  if (ensure_cap((void **)&ctx->d_status, &ctx->d_status_cap, n)) return -2;
  HIP_CHECK(hipMemcpyAsync(ctx->d_in, tuples, n * rec_size, hipMemcpyHostToDevice,
                           ctx->stream));
  HIP_CHECK(hipMemsetAsync(ctx->d_bitmap, 0, words * 8, ctx->stream));
  int block = 256;
  unsigned long long grid = (n + block - 1) / block;
  if (ecdsa)
    hipLaunchKernelGGL(kv::kv_ecdsa_verify_kernel, dim3(grid), dim3(block), 0, ctx->stream,
                       ctx->d_in, (unsigned long long)n, (unsigned long long *)ctx->d_bitmap,
                       status_out ? ctx->d_status : nullptr);
  else
    hipLaunchKernelGGL(kv::kv_schnorr_verify_kernel, dim3(grid), dim3(block), 0, ctx->stream,
                       ctx->d_in, (unsigned long long)n, (unsigned long long *)ctx->d_bitmap,
                       status_out ? ctx->d_status : nullptr);
  HIP_CHECK(hipGetLastError());
  HIP_CHECK(hipMemcpyAsync(bitmap_out, ctx->d_bitmap, words * 8, hipMemcpyDeviceToHost,
                           ctx->stream));
  if (status_out)
    HIP_CHECK(hipMemcpyAsync(status_out, ctx->d_status, n, hipMemcpyDeviceToHost,
                             ctx->stream));
  HIP_CHECK(hipStreamSynchronize(ctx->stream));
  return 0;
}

extern "C" int kv_verify_schnorr_batch(kv_ctx *ctx, const uint8_t *tuples, size_t n,
                                       uint64_t *bitmap_out) {
  return verify_batch(ctx, tuples, n, 128, 0, bitmap_out, nullptr);
}

/* extended variant exposing per-signature status (valid/invalid/bad-pubkey/bad-sig)
 * — what the validate path consumes to distinguish script errors from false */
extern "C" int kv_verify_schnorr_batch_status(kv_ctx *ctx, const uint8_t *tuples,
                                              size_t n, uint64_t *bitmap_out,
                                              uint8_t *status_out) {
  return verify_batch(ctx, tuples, n, 128, 0, bitmap_out, status_out);
}

extern "C" int kv_verify_ecdsa_batch(kv_ctx *ctx, const uint8_t *tuples, size_t n,
                                     uint64_t *bitmap_out) {
  return verify_batch(ctx, tuples, n, 132, 1, bitmap_out, nullptr);
}

extern "C" int kv_verify_ecdsa_batch_status(kv_ctx *ctx, const uint8_t *tuples, size_t n,
                                            uint64_t *bitmap_out, uint8_t *status_out) {
  return verify_batch(ctx, tuples, n, 132, 1, bitmap_out, status_out);
}

/* ---------------- MuHash host finalization ---------------- */

/* host keyed blake2b-256 (RFC 7693) — product host code */
namespace {
static const uint64_t H_IV[8] = {0x6a09e667f3bcc908ULL, 0xbb67ae8584caa73bULL,
                                 0x3c6ef372fe94f82bULL, 0xa54ff53a5f1d36f1ULL,
                                 0x510e527fade682d1ULL, 0x9b05688c2b3e6c1fULL,
                                 0x1f83d9abfb41bd6bULL, 0x5be0cd19137e2179ULL};
static const uint8_t H_SIG[12][16] = {
    {0, 1, 2, 3, 4, 5, 6, 7, 8, 9, 10, 11, 12, 13, 14, 15},
    {14, 10, 4, 8, 9, 15, 13, 6, 1, 12, 0, 2, 11, 7, 5, 3},
    {11, 8, 12, 0, 5, 2, 15, 13, 10, 14, 3, 6, 7, 1, 9, 4},
    {7, 9, 3, 1, 13, 12, 11, 14, 2, 6, 5, 10, 4, 0, 15, 8},
    {9, 0, 5, 7, 2, 4, 10, 15, 14, 1, 11, 12, 6, 8, 3, 13},
    {2, 12, 6, 10, 0, 11, 8, 3, 4, 13, 7, 5, 15, 14, 1, 9},
    {12, 5, 1, 15, 14, 13, 4, 10, 0, 7, 6, 3, 9, 2, 8, 11},
    {13, 11, 7, 14, 12, 1, 3, 9, 5, 0, 15, 4, 8, 6, 2, 10},
    {6, 15, 14, 9, 11, 3, 0, 8, 12, 2, 13, 7, 1, 4, 10, 5},
    {10, 2, 8, 4, 7, 6, 1, 5, 15, 11, 9, 14, 3, 12, 13, 0},
    {0, 1, 2, 3, 4, 5, 6, 7, 8, 9, 10, 11, 12, 13, 14, 15},
    {14, 10, 4, 8, 9, 15, 13, 6, 1, 12, 0, 2, 11, 7, 5, 3}};

static inline uint64_t rot64(uint64_t x, int n) { return (x >> n) | (x << (64 - n)); }

static void h_b2b_compress(uint64_t h[8], const uint8_t blk[128], uint64_t t, int last) {
  uint64_t v[16], m[16];
  for (int i = 0; i < 16; i++) memcpy(&m[i], blk + 8 * i, 8);
  for (int i = 0; i < 8; i++) v[i] = h[i];
  for (int i = 0; i < 8; i++) v[i + 8] = H_IV[i];
  v[12] ^= t;
  if (last) v[14] = ~v[14];
#define HG(a, b, c, d, x, y)                                                   \
  v[a] += v[b] + (x); v[d] = rot64(v[d] ^ v[a], 32); v[c] += v[d];             \
  v[b] = rot64(v[b] ^ v[c], 24); v[a] += v[b] + (y);                           \
  v[d] = rot64(v[d] ^ v[a], 16); v[c] += v[d]; v[b] = rot64(v[b] ^ v[c], 63);
  for (int r = 0; r < 12; r++) {
    const uint8_t *s = H_SIG[r];
    HG(0, 4, 8, 12, m[s[0]], m[s[1]]); HG(1, 5, 9, 13, m[s[2]], m[s[3]]);
    HG(2, 6, 10, 14, m[s[4]], m[s[5]]); HG(3, 7, 11, 15, m[s[6]], m[s[7]]);
    HG(0, 5, 10, 15, m[s[8]], m[s[9]]); HG(1, 6, 11, 12, m[s[10]], m[s[11]]);
    HG(2, 7, 8, 13, m[s[12]], m[s[13]]); HG(3, 4, 9, 14, m[s[14]], m[s[15]]);
  }
#undef HG
  for (int i = 0; i < 8; i++) h[i] ^= v[i] ^ v[i + 8];
}

static void h_blake2b_keyed(const uint8_t *key, size_t keylen, const uint8_t *data,
                            size_t len, uint8_t out[32]) {
  uint64_t h[8];
  for (int i = 0; i < 8; i++) h[i] = H_IV[i];
  h[0] ^= 32ULL | ((uint64_t)keylen << 8) | (1ULL << 16) | (1ULL << 24);
  uint8_t blk[128];
  uint64_t t = 0;
  if (keylen) {
    memset(blk, 0, 128);
    memcpy(blk, key, keylen);
    if (len == 0) {
      h_b2b_compress(h, blk, 128, 1);
      goto fin;
    }
    t = 128;
    h_b2b_compress(h, blk, 128, 0);
  }
  while (len > 128) {
    t += 128;
    h_b2b_compress(h, data, t, 0);
    data += 128;
    len -= 128;
  }
  memset(blk, 0, 128);
  memcpy(blk, data, len);
  t += len;
  h_b2b_compress(h, blk, t, 1);
fin:
  for (int i = 0; i < 32; i++) out[i] = (uint8_t)(h[i / 8] >> (8 * (i % 8)));
}

/* ---- 3072-bit helpers for the modular inverse (binary extended gcd) ---- */

using kv::u3072;

static int big_cmp(const uint64_t *a, const uint64_t *b) {
  for (int i = KVU_LIMBS - 1; i >= 0; i--) {
    if (a[i] != b[i]) return a[i] < b[i] ? -1 : 1;
  }
  return 0;
}

static void big_sub(uint64_t *a, const uint64_t *b) {
  uint64_t borrow = 0;
  for (int i = 0; i < KVU_LIMBS; i++) {
    uint64_t bi = b[i] + borrow;
    uint64_t nb = (bi < borrow) || (a[i] < bi);
    a[i] -= bi;
    borrow = nb;
  }
}

static void big_shr1(uint64_t *a, uint64_t top_bit) {
  for (int i = 0; i < KVU_LIMBS; i++) {
    uint64_t hi = (i + 1 < KVU_LIMBS) ? (a[i + 1] & 1) : top_bit;
    a[i] = (a[i] >> 1) | (hi << 63);
  }
}

static void prime_limbs(uint64_t *p) {
  for (int i = 0; i < KVU_LIMBS; i++) p[i] = ~0ULL;
  p[0] -= KVU_PRIME_DIFF - 1;
}

static void mod_half(uint64_t *x, const uint64_t *p) {
  if (x[0] & 1) {
    uint64_t c = 0;
    for (int i = 0; i < KVU_LIMBS; i++) x[i] = kv::kvu_addc(x[i], p[i], c);
    big_shr1(x, c);
  } else {
    big_shr1(x, 0);
  }
}

static void mod_sub(uint64_t *x, const uint64_t *y, const uint64_t *p) {
  /* x = (x - y) mod p, both < p */
  if (big_cmp(x, y) >= 0) {
    big_sub(x, y);
  } else {
    uint64_t t[KVU_LIMBS];
    memcpy(t, p, sizeof(t));
    big_sub(t, y); /* p - y */
    uint64_t c = 0;
    for (int i = 0; i < KVU_LIMBS; i++) x[i] = kv::kvu_addc(x[i], t[i], c);
    /* x + (p-y): may exceed p → subtract once */
    uint64_t pl[KVU_LIMBS];
    memcpy(pl, p, sizeof(pl));
    if (c || big_cmp(x, pl) >= 0) big_sub(x, pl);
  }
}

static void u3072_inverse(const u3072 &in, u3072 &out) {
  u3072 a = in;
  kv::u3072_canon(a);
  int zero = 1;
  for (int i = 0; i < KVU_LIMBS; i++)
    if (a.l[i]) zero = 0;
  if (zero) {
    memset(out.l, 0, sizeof(out.l));
    return;
  }
  uint64_t p[KVU_LIMBS], u[KVU_LIMBS], v[KVU_LIMBS], x1[KVU_LIMBS], x2[KVU_LIMBS];
  prime_limbs(p);
  memcpy(u, a.l, sizeof(u));
  memcpy(v, p, sizeof(v));
  memset(x1, 0, sizeof(x1));
  x1[0] = 1;
  memset(x2, 0, sizeof(x2));
  uint64_t one[KVU_LIMBS];
  memset(one, 0, sizeof(one));
  one[0] = 1;
  while (big_cmp(u, one) != 0 && big_cmp(v, one) != 0) {
    while (!(u[0] & 1)) {
      big_shr1(u, 0);
      mod_half(x1, p);
    }
    while (!(v[0] & 1)) {
      big_shr1(v, 0);
      mod_half(x2, p);
    }
    if (big_cmp(u, v) >= 0) {
      big_sub(u, v);
      mod_sub(x1, x2, p);
    } else {
      big_sub(v, u);
      mod_sub(x2, x1, p);
    }
  }
  memcpy(out.l, big_cmp(u, one) == 0 ? x1 : x2, sizeof(out.l));
}

} // namespace

extern "C" int kv_muhash_combine(kv_ctx *ctx, uint8_t *acc_partial768,
                                 const uint8_t *other_partial768) {
  (void)ctx;
  u3072 an, ad, bn, bd, r;
  memcpy(an.l, acc_partial768, 384);
  memcpy(ad.l, acc_partial768 + 384, 384);
  memcpy(bn.l, other_partial768, 384);
  memcpy(bd.l, other_partial768 + 384, 384);
  kv::u3072_mulmod(r, an, bn);
  memcpy(acc_partial768, r.l, 384);
  kv::u3072_mulmod(r, ad, bd);
  memcpy(acc_partial768 + 384, r.l, 384);
  return 0;
}

extern "C" int kv_muhash_finalize(kv_ctx *ctx, const uint8_t *partial768,
                                  uint8_t *hash32_out) {
  (void)ctx;
  u3072 num, den, dinv, r;
  memcpy(num.l, partial768, 384);
  memcpy(den.l, partial768 + 384, 384);
  u3072_inverse(den, dinv);
  kv::u3072_mulmod(r, num, dinv);
  kv::u3072_canon(r);
  uint8_t ser[384];
  memcpy(ser, r.l, 384); /* limbs are LE on all supported hosts */
  static const uint8_t KEY[] = "MuHashFinalize";
  h_blake2b_keyed(KEY, sizeof(KEY) - 1, ser, 384, hash32_out);
  return 0;
}

extern "C" int kv_sig_cache_stats(kv_ctx *ctx, kv_cache_stats *out) {
  out->insertions = ctx->cache_insertions;
  out->hits = ctx->cache_hits;
  out->misses = ctx->cache_misses;
  return 0;
}

/* ---------------- staged bench mode ----------------
 * Stage a tuple batch into HBM once; time repeated verify launches with the
 * inputs already resident (bench.py's timed region starts after staging —
 * the PCIe-inclusive rate of kv_verify_schnorr_batch is reported separately
 * in DESIGN.md). Kernel duration is measured with hipEvents on the engine's
 * OWN stream (torch.cuda.Event would watch torch's stream, not ours). */

extern "C" int kv_stage_tuples(kv_ctx *ctx, const uint8_t *tuples, size_t n,
                               int ecdsa) {
  std::lock_guard<std::mutex> lk(ctx->mu);
  size_t rec = ecdsa ? 132 : 128;
  size_t words = (n + 63) / 64;
  if (ensure_cap((void **)&ctx->d_in, &ctx->d_in_cap, n * rec)) return -2;
  if (ensure_cap((void **)&ctx->d_bitmap, &ctx->d_bitmap_cap, words * 8)) return -2;
  HIP_CHECK(hipMemcpy(ctx->d_in, tuples, n * rec, hipMemcpyHostToDevice));
  return 0;
}

/* run one staged verify launch; writes kernel milliseconds to *kernel_ms.
 * Does NOT copy the bitmap back (kv_fetch_bitmap does). */
extern "C" int kv_verify_staged(kv_ctx *ctx, size_t n, int ecdsa, double *kernel_ms) {
  std::lock_guard<std::mutex> lk(ctx->mu);
  int block = 256;
  unsigned long long grid = (n + block - 1) / block;
  hipEvent_t t0, t1;
  HIP_CHECK(hipEventCreate(&t0));
  HIP_CHECK(hipEventCreate(&t1));
  HIP_CHECK(hipEventRecord(t0, ctx->stream));
  if (ecdsa)
    hipLaunchKernelGGL(kv::kv_ecdsa_verify_kernel, dim3(grid), dim3(block), 0,
                       ctx->stream, ctx->d_in, (unsigned long long)n,
                       (unsigned long long *)ctx->d_bitmap, nullptr);
  else
    hipLaunchKernelGGL(kv::kv_schnorr_verify_kernel, dim3(grid), dim3(block), 0,
                       ctx->stream, ctx->d_in, (unsigned long long)n,
                       (unsigned long long *)ctx->d_bitmap, nullptr);
  HIP_CHECK(hipGetLastError());
  HIP_CHECK(hipEventRecord(t1, ctx->stream));
  HIP_CHECK(hipEventSynchronize(t1));
  float ms = 0.f;
  HIP_CHECK(hipEventElapsedTime(&ms, t0, t1));
  if (kernel_ms) *kernel_ms = (double)ms;
  (void)hipEventDestroy(t0);
  (void)hipEventDestroy(t1);
  return 0;
}

extern "C" int kv_fetch_bitmap(kv_ctx *ctx, size_t n, uint64_t *bitmap_out) {
  std::lock_guard<std::mutex> lk(ctx->mu);
  size_t words = (n + 63) / 64;
  HIP_CHECK(hipMemcpy(bitmap_out, ctx->d_bitmap, words * 8, hipMemcpyDeviceToHost));
  return 0;
}

/* ---------------- full-block validate path ----------------
 * Implemented in stages this round; entry points fail loudly (never silently
 * skip) until their GPU pipeline is wired. */

extern "C" int kv_sighash_batch(kv_ctx *ctx, const uint8_t *blob, size_t blob_len,
                                const kv_sighash_job *jobs, size_t n,
                                uint8_t *hashes_out) {
  (void)ctx; (void)blob; (void)blob_len; (void)jobs; (void)n; (void)hashes_out;
  set_error("kv_sighash_batch: GPU sighash pipeline not wired yet (round 1 WIP)");
  return -3;
}

extern "C" int kv_validate_block(kv_ctx *ctx, const uint8_t *blob, size_t blob_len,
                                 uint64_t pov_daa_score, uint64_t block_daa_score,
                                 uint32_t flags, int32_t *tx_codes_out,
                                 uint64_t *fees_out, uint8_t *muhash_partial_out) {
  (void)ctx; (void)blob; (void)blob_len; (void)pov_daa_score; (void)block_daa_score;
  (void)flags; (void)tx_codes_out; (void)fees_out; (void)muhash_partial_out;
  set_error("kv_validate_block: GPU block pipeline not wired yet (round 1 WIP)");
  return -3;
}
