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
#include <array>
#include <chrono>
#include <unordered_map>

/* kernels compiled into this TU (single translation unit keeps the build to one
 * hipcc invocation, no -fgpu-rdc) */
#include "kv_secp_kernels.hip"
#include "kv_sighash_kernels.hip"
#include "kv_utxo_kernels.hip"
#include "kv_validate_host.inc"
#include "kv_script_host.inc"

static thread_local std::string g_last_error;

static void set_error(const char *msg) { g_last_error = msg ? msg : ""; }

static void set_error_loc(const char *msg, const char *file, int line,
                          const char *expr) {
  g_last_error = std::string(msg ? msg : "") + " at " + file + ":" +
                 std::to_string(line) + " (" + expr + ")";
}

extern "C" const char *kv_last_error(void) { return g_last_error.c_str(); }

#define HIP_CHECK(expr)                                                        \
  do {                                                                         \
    hipError_t err_ = (expr);                                                  \
    if (err_ != hipSuccess) {                                                  \
      set_error_loc(hipGetErrorString(err_), __FILE__, __LINE__, #expr);       \
      return -2;                                                               \
    }                                                                          \
  } while (0)

struct kv_sig_key {
  uint64_t w[4];
  bool operator==(const kv_sig_key &o) const {
    return ((w[0] ^ o.w[0]) | (w[1] ^ o.w[1]) | (w[2] ^ o.w[2]) |
            (w[3] ^ o.w[3])) == 0;
  }
};
struct kv_sig_key_hash {
  size_t operator()(const kv_sig_key &k) const { return (size_t)k.w[0]; }
};

struct kv_ctx {
  kv_params params;
  hipStream_t stream;
  hipStream_t stream2; /* ecdsa verify chain (overlaps the schnorr chain) */
  hipStream_t stream3; /* optimistic muhash chain (overlaps both) */
  hipEvent_t ev_blob;  /* blob upload complete (sync-only, no timing) */
  hipEvent_t ev_sub;   /* subhash kernel complete */
  std::mutex mu;
  /* KIP-21 seq-commitment accessor (kv_set_seq_commit_accessor) */
  kv_seq_commit_accessor_fn seqc_fn = nullptr;
  void *seqc_user = nullptr;
  /* per-context validate scratch (ValidateBufs; opaque here because the type
   * lives with the validate section). Per-ctx so two engine contexts in one
   * process never share device scratch. */
  void *vb = nullptr;
  /* per-kernel timing events of the last validate call (kv_get_validate_timings):
   * pairs (start,stop) for subhash, s-assemble, e-assemble, schnorr, ecdsa, muhash */
  hipEvent_t tev[12] = {};
  bool tev_init = false;
  kv_validate_timings last_timings = {};
  /* grow-on-demand device scratch */
  uint8_t *d_in = nullptr;
  size_t d_in_cap = 0;
  uint64_t *d_bitmap = nullptr;
  size_t d_bitmap_cap = 0;
  uint8_t *d_status = nullptr;
  size_t d_status_cap = 0;
  uint64_t cache_hits = 0, cache_misses = 0, cache_insertions = 0;
  /* sig cache ⇔ TransactionValidator sig_cache (crypto/txscript/src/caches.rs:
   * 57-82): verdicts of (txid, entries-digest, input, hash-type, sig, pk)
   * keyed checks survive across calls — the mempool→block revalidation dedup.
   * Key is a blake2b-256 so a collision is cryptographically excluded. */
  std::unordered_map<kv_sig_key, uint8_t, kv_sig_key_hash> sig_cache;
  /* reusable populate-path scratch (capacity persists across calls — a fresh
   * multi-MB vector per block showed up as page-fault spikes in the bench) */
  std::vector<uint8_t> pop_buf, ops_buf, ent_buf;
  /* PINNED host buffers for the verify-status readbacks: a pageable-dest
   * hipMemcpyAsync blocks the enqueue thread until the producing kernels
   * finish, which serialized the three streams (measured 7.7ms of 8.3ms in
   * the enqueue phase) */
  uint8_t *h_s_status = nullptr, *h_e_status = nullptr;
  size_t h_s_cap = 0, h_e_cap = 0;
  /* per-call PINNED upload staging arena: hipMemcpyAsync from pageable host
   * memory blocks the enqueue thread (synchronous staging inside HIP); all
   * per-call H2D uploads bounce through this bump arena instead. Regions
   * stay valid until the call's final stream syncs. */
  uint8_t *h_stage = nullptr;
  size_t h_stage_cap = 0, h_stage_off = 0;
  uint8_t *h_mu_partial = nullptr; /* pinned 768B muhash partial landing pad:
    a pageable-dest D2H of even 384B blocks the enqueue thread until the
    whole reduce chain drains (measured ~3.9ms) */
  std::vector<uint64_t> found_buf;
  /* GPU-resident UTXO set */
  kv::utxo_slot *d_utxo = nullptr;
  uint64_t utxo_cap = 0; /* power of two */
  /* out-of-line spk arena (entries with spk_len > 36; flags bit1) */
  uint8_t *d_arena = nullptr;
  uint64_t arena_cap = 0, arena_head = 0;
  void *d_gjobs = nullptr;
  size_t d_gjobs_cap = 0;
  void *d_gout = nullptr;
  size_t d_gout_cap = 0;
  int *d_utxo_fail = nullptr;
  uint8_t *d_op_in = nullptr;
  size_t d_op_cap = 0;
  uint8_t *d_val_in = nullptr;
  size_t d_val_cap = 0;
  uint8_t *d_ent_out = nullptr;
  size_t d_ent_cap = 0;
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
  if (hipStreamCreate(&ctx->stream) != hipSuccess ||
      hipStreamCreate(&ctx->stream2) != hipSuccess ||
      hipStreamCreate(&ctx->stream3) != hipSuccess ||
      hipEventCreateWithFlags(&ctx->ev_blob, hipEventDisableTiming) != hipSuccess ||
      hipEventCreateWithFlags(&ctx->ev_sub, hipEventDisableTiming) != hipSuccess) {
    set_error("kv_create: stream/event creation failed");
    delete ctx;
    return nullptr;
  }
  /* fill the shared affine G multiples table (idempotent, once per device) */
  hipLaunchKernelGGL(kv::kv_ec_table_init_kernel, dim3(1), dim3(64), 0, ctx->stream);
  if (hipStreamSynchronize(ctx->stream) != hipSuccess) {
    set_error("kv_create: G-table init failed");
    delete ctx;
    return nullptr;
  }
  return ctx;
}

namespace {
void kv_vb_release(void *vb); /* defined with ValidateBufs below */
}

extern "C" void kv_destroy(kv_ctx *ctx) {
  if (!ctx) return;
  if (ctx->vb) kv_vb_release(ctx->vb);
  if (ctx->d_in) (void)hipFree(ctx->d_in);
  if (ctx->d_bitmap) (void)hipFree(ctx->d_bitmap);
  if (ctx->d_status) (void)hipFree(ctx->d_status);
  if (ctx->h_s_status) (void)hipHostFree(ctx->h_s_status);
  if (ctx->h_e_status) (void)hipHostFree(ctx->h_e_status);
  if (ctx->h_stage) (void)hipHostFree(ctx->h_stage);
  if (ctx->h_mu_partial) (void)hipHostFree(ctx->h_mu_partial);
  if (ctx->d_arena) (void)hipFree(ctx->d_arena);
  if (ctx->d_gjobs) (void)hipFree(ctx->d_gjobs);
  if (ctx->d_gout) (void)hipFree(ctx->d_gout);
  (void)hipStreamDestroy(ctx->stream);
  (void)hipStreamDestroy(ctx->stream2);
  (void)hipStreamDestroy(ctx->stream3);
  (void)hipEventDestroy(ctx->ev_blob);
  (void)hipEventDestroy(ctx->ev_sub);
  if (ctx->tev_init)
    for (int i = 0; i < 12; i++) (void)hipEventDestroy(ctx->tev[i]);
  if (ctx->d_utxo) (void)hipFree(ctx->d_utxo);
  if (ctx->d_utxo_fail) (void)hipFree(ctx->d_utxo_fail);
  if (ctx->d_op_in) (void)hipFree(ctx->d_op_in);
  if (ctx->d_val_in) (void)hipFree(ctx->d_val_in);
  if (ctx->d_ent_out) (void)hipFree(ctx->d_ent_out);
  (void)hipGetLastError(); /* teardown calls are fire-and-forget; do not leave
      their errors sticky for the next engine's launch checks (a swallowed
      invalid-argument here surfaced as a spurious failure in the NEXT
      context's first kernel-launch check) */
  delete ctx;
}

/* ---------------- batched verify entry points ---------------- */

static int verify_batch(kv_ctx *ctx, const uint8_t *tuples, size_t n, size_t rec_size,
                        int ecdsa, uint64_t *bitmap_out, uint8_t *status_out) {
  if (n == 0) return 0;
  std::lock_guard<std::mutex> lk(ctx->mu);
  (void)hipGetLastError(); /* clear any stale per-thread error so the
      launch-config checks below only see THIS call's launches */
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
  kvhost::h_blake2b_keyed(KEY, sizeof(KEY) - 1, ser, 384, hash32_out);
  return 0;
}

extern "C" int kv_set_seq_commit_accessor(kv_ctx *ctx, kv_seq_commit_accessor_fn fn,
                                          void *user) {
  std::lock_guard<std::mutex> lk(ctx->mu);
  (void)hipGetLastError(); /* clear any stale per-thread error so the
      launch-config checks below only see THIS call's launches */
  ctx->seqc_fn = fn;
  ctx->seqc_user = user;
  return 0;
}

extern "C" int kv_get_validate_timings(kv_ctx *ctx, kv_validate_timings *out) {
  if (!ctx || !out) return -1;
  std::lock_guard<std::mutex> lk(ctx->mu);
  (void)hipGetLastError(); /* clear any stale per-thread error so the
      launch-config checks below only see THIS call's launches */
  *out = ctx->last_timings;
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
  (void)hipGetLastError(); /* clear any stale per-thread error so the
      launch-config checks below only see THIS call's launches */
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
  (void)hipGetLastError(); /* clear any stale per-thread error so the
      launch-config checks below only see THIS call's launches */
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
  (void)hipGetLastError(); /* clear any stale per-thread error so the
      launch-config checks below only see THIS call's launches */
  size_t words = (n + 63) / 64;
  HIP_CHECK(hipMemcpy(bitmap_out, ctx->d_bitmap, words * 8, hipMemcpyDeviceToHost));
  return 0;
}

/* ---------------- full-block validate path ----------------
 * kv_validate_block ⇔ validate_populated_transaction_and_get_fee fanned out
 * per tx + the muhash monoid reduce (tx_validation_in_utxo_context.rs:37-67,
 * utxo_validation.rs:319-348). Host does the cheap integer checks and template
 * flattening; EVERY hash and EC verify runs on the GPU. */

using namespace kvhost;

namespace {

struct DeviceBuf {
  void *p = nullptr;
  size_t cap = 0;
  int ensure(size_t need) {
    if (cap >= need) return 0;
    if (p) (void)hipFree(p);
    size_t nc = need + need / 2 + 256;
    if (hipMalloc(&p, nc) != hipSuccess) {
      p = nullptr;
      cap = 0;
      set_error("hipMalloc failed (validate)");
      return -2;
    }
    cap = nc;
    return 0;
  }
};

struct ValidateBufs {
  DeviceBuf blob, subhashes, s_jobs, e_jobs, s_tuples, e_tuples, s_bitmap, e_bitmap,
      s_status, e_status, elem_jobs, elements, partials_a, partials_b, tx_hashes;
  void release() {
    DeviceBuf *bufs[] = {&blob, &subhashes, &s_jobs, &e_jobs, &s_tuples,
                         &e_tuples, &s_bitmap, &e_bitmap, &s_status, &e_status,
                         &elem_jobs, &elements, &partials_a, &partials_b,
                         &tx_hashes};
    for (auto *b : bufs) {
      if (b->p) (void)hipFree(b->p);
      b->p = nullptr;
      b->cap = 0;
    }
  }
};

static ValidateBufs &vb_of(kv_ctx *ctx) {
  if (!ctx->vb) ctx->vb = new ValidateBufs();
  return *(ValidateBufs *)ctx->vb;
}

void kv_vb_release(void *p) {
  auto *vb = (ValidateBufs *)p;
  vb->release();
  delete vb;
}

/* classify one input; appends jobs. Returns plan. */
/* non-template scripts: run the host general interpreter (kv_script_host.inc)
 * in collect mode. Signature-free scripts — hash puzzles, timelocks,
 * introspection/covenant logic, anyone-can-spend — resolve right here; a
 * script whose execution reaches a signature site suspends with its verify
 * requests in plan.pending and resolves over the GPU batch in the
 * interpreter-rounds phase of validate_block_impl. Only zk-precompile scripts
 * defer to the reference CPU interpreter (KV_TX_DEFER_TO_CPU). */
static void interp_classify(kv_ctx *ctx, const HTx &tx, const HInput &in,
                            uint32_t input_index, uint64_t sigop_units,
                            InputPlan &pl) {
  kvhost::KvsRunCtx rctx;
  rctx.memo = &pl.memo;
  rctx.pending = &pl.pending;
  rctx.seqc_fn = ctx->seqc_fn;
  rctx.seqc_user = ctx->seqc_user;
  int rc = kvh_run_input_script(tx, in, input_index, sigop_units, &rctx);
  if (rc == kvhost::KVH_SCRIPT_SUSPEND) {
    pl.kind = PLAN_INTERP;
    return;
  }
  int base = in.sig_script_len == 0 ? KV_ERR_SIGNATURE_EMPTY_BASE
                                    : KV_ERR_SIGNATURE_INVALID_BASE;
  if (rc == kvhost::KVH_SCRIPT_DEFER)
    pl.pre_code = KV_TX_DEFER_TO_CPU;
  else if (rc != 0)
    pl.pre_code = base + rc;
  else
    pl.pre_code = 0;
}

static InputPlan classify_input(kv_ctx *ctx, const HTx &tx, const HInput &in,
                                uint64_t sigop_units,
                                std::vector<kv::kv_job> &sjobs,
                                std::vector<kv::kv_job> &ejobs,
                                uint32_t tx_index, uint32_t input_index) {
  InputPlan pl;
  pl.limit_units = committed_limit(in);
  if (in.utxo_spk_version > 0) return pl; /* unknown version: accepted (lib.rs:655) */
  const uint8_t *spk = in.utxo_spk;
  uint32_t spklen = in.utxo_spk_len;
  /* oversized utxo spk charge (lib.rs:661-676) */
  if (spklen > 35) {
    uint64_t extra = (uint64_t)(spklen - 35) * KVH_UNITS_PER_GRAM;
    if (extra > pl.limit_units) {
      pl.pre_code = KV_ERR_SIGNATURE_INVALID_BASE + KV_SCRIPT_EXCEEDED_SCRIPT_UNITS;
      if (in.sig_script_len == 0)
        pl.pre_code = KV_ERR_SIGNATURE_EMPTY_BASE + KV_SCRIPT_EXCEEDED_SCRIPT_UNITS;
      return pl;
    }
    pl.spent_units += extra;
  }
  if (spklen == 0 && in.sig_script_len == 0) {
    pl.pre_code = KV_ERR_SIGNATURE_EMPTY_BASE + KV_SCRIPT_EVAL_FALSE;
    return pl;
  }
  int is_p2pk = spklen == 34 && spk[0] == 0x20 && spk[33] == 0xac;
  int is_p2pk_ecdsa = spklen == 35 && spk[0] == 0x21 && spk[34] == 0xab;
  int is_p2sh = spklen == 35 && spk[0] == 0xaa && spk[1] == 0x20 && spk[34] == 0x87;

  if (is_p2pk || is_p2pk_ecdsa) {
    const uint32_t siglen = 65; /* sig64 + type (both schnorr and ecdsa) */
    if (in.sig_script_len == 0) {
      /* spk alone: checksig pops 2 from a 1-deep stack */
      pl.pre_code = KV_ERR_SIGNATURE_EMPTY_BASE + KV_SCRIPT_INVALID_STACK_OPERATION;
      return pl;
    }
    if (in.sig_script_len != 1 + siglen || in.sig_script[0] != siglen) {
      interp_classify(ctx, tx, in, input_index, sigop_units, pl);
      return pl;
    }
    uint8_t type = in.sig_script[siglen];
    if (!valid_sighash_type(type)) {
      pl.pre_code = KV_ERR_SIGNATURE_INVALID_BASE + KV_SCRIPT_INVALID_SIG_HASH_TYPE;
      return pl;
    }
    if (sigop_units > pl.limit_units - pl.spent_units) {
      pl.pre_code = KV_ERR_SIGNATURE_INVALID_BASE + KV_SCRIPT_EXCEEDED_SCRIPT_UNITS;
      return pl;
    }
    pl.kind = PLAN_P2PK;
    pl.ecdsa = is_p2pk_ecdsa;
    kv::kv_job job;
    job.tx_index = tx_index;
    job.input_off = in.rec_off;
    job.input_index = input_index;
    job.sig_off = in.sig_script_off + 1;
    job.pk_off = in.utxo_spk_off + 1;
    job.hash_type = type;
    job.ecdsa = is_p2pk_ecdsa ? 1 : 0;
    job._pad = 0;
    if (is_p2pk_ecdsa) {
      pl.job = (int32_t)ejobs.size();
      ejobs.push_back(job);
    } else {
      pl.job = (int32_t)sjobs.size();
      sjobs.push_back(job);
    }
    return pl;
  }

  if (is_p2sh) {
    if (in.sig_script_len == 0) {
      /* spk alone: OpBlake2b pops from empty stack */
      pl.pre_code = KV_ERR_SIGNATURE_EMPTY_BASE + KV_SCRIPT_INVALID_STACK_OPERATION;
      return pl;
    }
    std::vector<std::pair<uint32_t, uint32_t>> pushes;
    if (!parse_pushes(nullptr, in.sig_script, in.sig_script_len, in.sig_script_off,
                      pushes) ||
        pushes.empty()) {
      interp_classify(ctx, tx, in, input_index, sigop_units, pl);
      return pl;
    }
    /* last push = redeem candidate (need the actual bytes: offset into blob is
     * relative; caller gave us pointers via in.sig_script) */
    auto redeem_pp = pushes.back();
    const uint8_t *redeem = in.sig_script + (redeem_pp.first - in.sig_script_off);
    uint32_t rlen = redeem_pp.second;
    /* p2sh hash check (spk exec): blake2b(redeem) == h32 — budget first */
    uint64_t blake_cost = (uint64_t)rlen * 2 + 32; /* data cost + pushed hash */
    if (pl.spent_units + blake_cost > pl.limit_units) {
      pl.pre_code = KV_ERR_SIGNATURE_INVALID_BASE + KV_SCRIPT_EXCEEDED_SCRIPT_UNITS;
      return pl;
    }
    pl.spent_units += blake_cost;
    uint8_t h[32];
    h_blake2b_keyed(nullptr, 0, redeem, rlen, h);
    if (memcmp(h, spk + 2, 32) != 0) {
      /* OpEqual pushes empty (0 units); check_error(false) → EvalFalse */
      pl.pre_code = KV_ERR_SIGNATURE_INVALID_BASE + KV_SCRIPT_EVAL_FALSE;
      return pl;
    }
    pl.spent_units += 1; /* OpEqual pushed [1] */
    /* parse canonical multisig redeem: OP_m (0x20 pk)×n OP_n 0xae */
    if (rlen < 3 || redeem[0] < 0x51 || redeem[0] > 0x60) {
      interp_classify(ctx, tx, in, input_index, sigop_units, pl);
      return pl;
    }
    int m = redeem[0] - 0x50;
    uint32_t rp = 1;
    std::vector<uint32_t> key_offs;
    while (rp + 33 <= rlen && redeem[rp] == 0x20) {
      key_offs.push_back(redeem_pp.first + rp + 1);
      rp += 33;
    }
    int n = (int)key_offs.size();
    if (rp + 2 != rlen || n < 1 || n > 20 || redeem[rp] != (uint8_t)(0x50 + n) ||
        redeem[rp + 1] != 0xae || m > n || m < 1) {
      interp_classify(ctx, tx, in, input_index, sigop_units, pl);
      return pl;
    }
    int nsigs = (int)pushes.size() - 1;
    if (nsigs != m) {
      /* stack mismatch: fewer → InvalidStackOperation when multisig pops;
       * more → extra entries → CleanStack at the end. Handle the simple
       * common cases; refuse others. */
      if (nsigs < m) {
        pl.pre_code = KV_ERR_SIGNATURE_INVALID_BASE + KV_SCRIPT_INVALID_STACK_OPERATION;
        return pl;
      }
      pl.extra_stack = true; /* resolved after matching */
    }
    pl.kind = PLAN_MULTISIG;
    pl.msig_m = m;
    pl.msig_n = n;
    pl.key_offs = key_offs;
    /* sigs = the last m pushes before the redeem (stack top-down ordering:
     * multisig pops the TOP m entries = the last m pushes).
     * A sig whose length is neither 0 nor 65 needs the current KEY's parse
     * status before its own length error (check order: cost, pubkey,
     * signature — lib.rs:884-890), which the template fast path cannot
     * provide; such inputs run through the general interpreter instead. */
    for (int si = nsigs - m; si < nsigs; si++) {
      uint32_t sl = pushes[si].second;
      if (sl != 0 && sl != 65) {
        pl = InputPlan();
        pl.limit_units = committed_limit(in);
        interp_classify(ctx, tx, in, input_index, sigop_units, pl);
        return pl;
      }
    }
    for (int si = nsigs - m; si < nsigs; si++) {
      MsigSig ms;
      ms.off = pushes[si].first;
      ms.len = pushes[si].second;
      ms.type = ms.len ? in.sig_script[ms.off - in.sig_script_off + ms.len - 1] : 0;
      ms.job0 = -1;
      if (ms.len == 65 && valid_sighash_type(ms.type)) {
        ms.job0 = (int32_t)sjobs.size();
        for (int ki = 0; ki < n; ki++) {
          kv::kv_job job;
          job.tx_index = tx_index;
          job.input_off = in.rec_off;
          job.input_index = input_index;
          job.sig_off = ms.off;
          job.pk_off = key_offs[ki];
          job.hash_type = ms.type;
          job.ecdsa = 0;
          job._pad = 0;
          sjobs.push_back(job);
        }
      }
      pl.msig_sigs.push_back(ms);
    }
    return pl;
  }

  interp_classify(ctx, tx, in, input_index, sigop_units, pl);
  return pl;
}

/* resolve one input after GPU statuses are back; returns full KV code or 0 */
static int resolve_input(const InputPlan &pl, const HInput &in, uint64_t sigop_units,
                         const uint8_t *s_status, const uint8_t *e_status) {
  int empty_base = in.sig_script_len == 0 ? KV_ERR_SIGNATURE_EMPTY_BASE
                                          : KV_ERR_SIGNATURE_INVALID_BASE;
  if (pl.pre_code) return pl.pre_code;
  if (pl.kind == PLAN_NONE) return 0;
  uint64_t spent = pl.spent_units;
  if (pl.kind == PLAN_P2PK) {
    uint8_t st = pl.ecdsa ? e_status[pl.job] : s_status[pl.job];
    spent += sigop_units; /* pre-checked to fit */
    if (st == 2) return empty_base + KV_SCRIPT_INVALID_PUBKEY;
    if (st == 3) return empty_base + KV_SCRIPT_INVALID_SIGNATURE;
    if (st == 0) {
      if (spent + 1 > pl.limit_units)
        return empty_base + KV_SCRIPT_EXCEEDED_SCRIPT_UNITS;
      return 0;
    }
    return empty_base + KV_SCRIPT_EVAL_FALSE;
  }
  /* multisig greedy matching (lib.rs:759-843) */
  int kp = 0;
  bool failed = false;
  int m = pl.msig_m, n = pl.msig_n;
  for (int si = 0; si < m; si++) {
    const MsigSig &ms = pl.msig_sigs[si];
    if (ms.len == 0) {
      failed = true;
      break;
    }
    if (!valid_sighash_type(ms.type))
      return empty_base + KV_SCRIPT_INVALID_SIG_HASH_TYPE;
    bool matched = false;
    while (true) {
      if (n - kp < m - si) {
        failed = true;
        break;
      }
      /* consume one sigop for this try */
      if (spent + sigop_units > pl.limit_units)
        return empty_base + KV_SCRIPT_EXCEEDED_SCRIPT_UNITS;
      spent += sigop_units;
      int key = kp++;
      if (ms.len != 65) /* sig blob wrong size: InvalidSignature error */
        return empty_base + KV_SCRIPT_INVALID_SIGNATURE;
      uint8_t st = s_status[ms.job0 + key];
      if (st == 2) return empty_base + KV_SCRIPT_INVALID_PUBKEY;
      if (st == 0) {
        matched = true;
        break;
      }
    }
    if (!matched) break;
  }
  bool any_nonempty = false;
  for (const auto &ms : pl.msig_sigs)
    if (ms.len) any_nonempty = true;
  if (failed && any_nonempty) return empty_base + KV_SCRIPT_NULL_FAIL;
  if (failed) {
    /* push false (0 units) → EvalFalse (or CleanStack first if extra) */
    if (pl.extra_stack) return empty_base + KV_SCRIPT_CLEAN_STACK;
    return empty_base + KV_SCRIPT_EVAL_FALSE;
  }
  if (spent + 1 > pl.limit_units)
    return empty_base + KV_SCRIPT_EXCEEDED_SCRIPT_UNITS;
  if (pl.extra_stack) return empty_base + KV_SCRIPT_CLEAN_STACK;
  return 0;
}

} // namespace

extern "C" int kv_sighash_batch(kv_ctx *ctx, const uint8_t *blob, size_t blob_len,
                                const kv_sighash_job *jobs_in, size_t n,
                                uint8_t *hashes_out) {
  /* standalone sighash service over the blob (tests/parity). Jobs are split by
   * kind — the assemble kernel uses different tuple strides for schnorr (128B)
   * and ecdsa (132B). */
  std::lock_guard<std::mutex> lk(ctx->mu);
  ValidateBufs &vb = vb_of(ctx);
  (void)hipGetLastError(); /* clear any stale per-thread error so the
      launch-config checks below only see THIS call's launches */
  vector<HTx> txs;
  if (parse_blob_host(blob, blob_len, txs) < 0) {
    set_error("kv_sighash_batch: malformed blob");
    return -1;
  }
  std::vector<kv::kv_job> sjobs, ejobs;
  std::vector<std::pair<int, size_t>> slots(n); /* (kind, index in its array) */
  for (size_t i = 0; i < n; i++) {
    const kv_sighash_job &j = jobs_in[i];
    if (j.tx_index >= txs.size() || j.input_index >= txs[j.tx_index].inputs.size()) {
      set_error("kv_sighash_batch: bad job index");
      return -1;
    }
    const HInput &in = txs[j.tx_index].inputs[j.input_index];
    kv::kv_job job{j.tx_index, in.rec_off, j.input_index, 0, 0, j.hash_type, j.ecdsa, 0};
    if (j.ecdsa) {
      slots[i] = {1, ejobs.size()};
      ejobs.push_back(job);
    } else {
      slots[i] = {0, sjobs.size()};
      sjobs.push_back(job);
    }
  }
  uint32_t n_txs = (uint32_t)txs.size();
  if (vb.blob.ensure(blob_len) || vb.subhashes.ensure((size_t)n_txs * 160))
    return -2;
  HIP_CHECK(hipMemcpyAsync(vb.blob.p, blob, blob_len, hipMemcpyHostToDevice,
                           ctx->stream));
  hipLaunchKernelGGL(kv::kv_tx_subhash_kernel, dim3((n_txs + 255) / 256), dim3(256), 0,
                     ctx->stream, (const uint8_t *)vb.blob.p, n_txs,
                     (uint8_t *)vb.subhashes.p);
  std::vector<uint8_t> stuples(sjobs.size() * 128), etuples(ejobs.size() * 132);
  if (!sjobs.empty()) {
    if (vb.s_jobs.ensure(sjobs.size() * sizeof(kv::kv_job)) ||
        vb.s_tuples.ensure(sjobs.size() * 128))
      return -2;
    HIP_CHECK(hipMemcpyAsync(vb.s_jobs.p, sjobs.data(),
                             sjobs.size() * sizeof(kv::kv_job), hipMemcpyHostToDevice,
                             ctx->stream));
    hipLaunchKernelGGL(kv::kv_sighash_assemble_kernel,
                       dim3(((uint32_t)sjobs.size() + 255) / 256), dim3(256), 0,
                       ctx->stream, (const uint8_t *)vb.blob.p,
                       (const uint8_t *)vb.subhashes.p,
                       (const kv::kv_job *)vb.s_jobs.p, (uint32_t)sjobs.size(),
                       (uint8_t *)vb.s_tuples.p, (uint8_t *)vb.s_tuples.p);
    HIP_CHECK(hipMemcpyAsync(stuples.data(), vb.s_tuples.p, stuples.size(),
                             hipMemcpyDeviceToHost, ctx->stream));
  }
  if (!ejobs.empty()) {
    if (vb.e_jobs.ensure(ejobs.size() * sizeof(kv::kv_job)) ||
        vb.e_tuples.ensure(ejobs.size() * 132))
      return -2;
    HIP_CHECK(hipMemcpyAsync(vb.e_jobs.p, ejobs.data(),
                             ejobs.size() * sizeof(kv::kv_job), hipMemcpyHostToDevice,
                             ctx->stream));
    hipLaunchKernelGGL(kv::kv_sighash_assemble_kernel,
                       dim3(((uint32_t)ejobs.size() + 255) / 256), dim3(256), 0,
                       ctx->stream, (const uint8_t *)vb.blob.p,
                       (const uint8_t *)vb.subhashes.p,
                       (const kv::kv_job *)vb.e_jobs.p, (uint32_t)ejobs.size(),
                       (uint8_t *)vb.e_tuples.p, (uint8_t *)vb.e_tuples.p);
    HIP_CHECK(hipMemcpyAsync(etuples.data(), vb.e_tuples.p, etuples.size(),
                             hipMemcpyDeviceToHost, ctx->stream));
  }
  HIP_CHECK(hipGetLastError());
  HIP_CHECK(hipStreamSynchronize(ctx->stream));
  for (size_t i = 0; i < n; i++) {
    const uint8_t *msg = slots[i].first
                             ? etuples.data() + slots[i].second * 132 + 97
                             : stuples.data() + slots[i].second * 128 + 96;
    memcpy(hashes_out + 32 * i, msg, 32);
  }
  return 0;
}

/* timing-event helpers (kv_get_validate_timings) */
static void tev_ensure(kv_ctx *ctx) {
  if (ctx->tev_init) return;
  for (int i = 0; i < 12; i++) (void)hipEventCreate(&ctx->tev[i]);
  ctx->tev_init = true;
}
static inline void tev_rec(kv_ctx *ctx, int i, hipStream_t stream = nullptr) {
  (void)hipEventRecord(ctx->tev[i], stream ? stream : ctx->stream);
}
static double tev_ms(kv_ctx *ctx, int pair) {
  float ms = 0.f;
  if (hipEventElapsedTime(&ms, ctx->tev[2 * pair], ctx->tev[2 * pair + 1]) !=
      hipSuccess)
    return 0.0;
  return (double)ms;
}

/* reserve the per-call pinned staging arena (call while all streams are
 * idle — start of a validate call); returns -2 on allocation failure */
static int stage_reserve(kv_ctx *ctx, size_t need) {
  ctx->h_stage_off = 0;
  if (ctx->h_stage_cap >= need) return 0;
  if (ctx->h_stage) (void)hipHostFree(ctx->h_stage);
  if (ctx->h_mu_partial) (void)hipHostFree(ctx->h_mu_partial);
  size_t nc = need + need / 2 + 4096;
  if (hipHostMalloc(&ctx->h_stage, nc) != hipSuccess) {
    ctx->h_stage = nullptr;
    ctx->h_stage_cap = 0;
    set_error("hipHostMalloc failed (staging)");
    return -2;
  }
  ctx->h_stage_cap = nc;
  return 0;
}

/* copy src into the pinned arena and return the pinned pointer (NULL when the
 * arena is exhausted — callers then fall back to the pageable source) */
static const uint8_t *stage_push(kv_ctx *ctx, const void *src, size_t len) {
  if (!ctx->h_stage || ctx->h_stage_off + len > ctx->h_stage_cap) return nullptr;
  uint8_t *dst = ctx->h_stage + ctx->h_stage_off;
  memcpy(dst, src, len);
  ctx->h_stage_off += len;
  return dst;
}

/* stage-then-upload: pinned when the arena has room, pageable otherwise */
static inline int h2d_staged(kv_ctx *ctx, void *dst, const void *src, size_t len,
                             hipStream_t stream) {
  const uint8_t *p = stage_push(ctx, src, len);
  HIP_CHECK(hipMemcpyAsync(dst, p ? (const void *)p : src, len,
                           hipMemcpyHostToDevice, stream));
  return 0;
}

/* Enqueue the muhash element + reduce chain for txs with include[t] != 0 on
 * `stream`, fully async — the caller syncs the stream before reading outp.
 * The blob must already be resident in vb.blob. With no work the identity
 * partial is written synchronously and *launched stays false. */
static int enqueue_muhash(kv_ctx *ctx, const std::vector<HTx> &txs,
                          const uint8_t *include, uint64_t block_daa_score,
                          hipStream_t stream, bool *launched,
                          std::vector<kv::kv_elem_job> &jobs) {
  ValidateBufs &vb = vb_of(ctx);
  if (!ctx->h_mu_partial &&
      hipHostMalloc(&ctx->h_mu_partial, 768) != hipSuccess) {
    ctx->h_mu_partial = nullptr;
    set_error("hipHostMalloc failed (muhash partial)");
    return -2;
  }
  uint8_t *outp = ctx->h_mu_partial;
  jobs.clear();
  for (size_t t = 0; t < txs.size(); t++) {
    if (!include[t]) continue;
    const HTx &tx = txs[t];
    uint8_t cb = h_is_coinbase(tx) ? 1 : 0;
    for (uint32_t i = 0; i < tx.outputs.size(); i++)
      jobs.push_back(kv::kv_elem_job{(uint32_t)t, tx.output_offs[i], i, 1, cb, 0,
                                     block_daa_score});
  }
  size_t n_num = jobs.size();
  for (size_t t = 0; t < txs.size(); t++) {
    if (!include[t]) continue;
    const HTx &tx = txs[t];
    uint8_t cb = h_is_coinbase(tx) ? 1 : 0;
    for (auto &in : tx.inputs)
      jobs.push_back(kv::kv_elem_job{(uint32_t)t, in.rec_off, 0, 0, cb, 0,
                                     block_daa_score});
  }
  size_t n_all = jobs.size(), n_den = n_all - n_num;
  kv::u3072 one;
  kv::u3072_one(one);
  *launched = false;
  if (n_all == 0) {
    memcpy(outp, one.l, 384);
    memcpy(outp + 384, one.l, 384);
    return 0;
  }
  if (vb.elem_jobs.ensure(n_all * sizeof(kv::kv_elem_job)) ||
      vb.elements.ensure(n_all * KVU_LIMBS * 8) ||
      vb.partials_a.ensure(1024 * KVU_LIMBS * 8) ||
      vb.partials_b.ensure(1024 * KVU_LIMBS * 8))
    return -2;
  if (h2d_staged(ctx, vb.elem_jobs.p, jobs.data(),
                 n_all * sizeof(kv::kv_elem_job), stream))
    return -2;
  tev_rec(ctx, 10, stream);
  hipLaunchKernelGGL(kv::kv_muhash_element_kernel,
                     dim3(((uint32_t)n_all + 255) / 256), dim3(256), 0, stream,
                     (const uint8_t *)vb.blob.p,
                     (const kv::kv_elem_job *)vb.elem_jobs.p, (uint32_t)n_all,
                     (uint64_t *)vb.elements.p);
  /* reduce numerator then denominator halves down to one value each; the
   * stride schedule is pure host arithmetic, so the whole chain enqueues
   * without any device->host round trip */
  for (int half = 0; half < 2; half++) {
    size_t cnt = half == 0 ? n_num : n_den;
    uint64_t *src = (uint64_t *)vb.elements.p + (half == 0 ? 0 : n_num * KVU_LIMBS);
    uint64_t *pa = (uint64_t *)vb.partials_a.p;
    uint64_t *pb = (uint64_t *)vb.partials_b.p;
    if (cnt == 0) {
      memcpy(outp + half * 384, one.l, 384);
      continue;
    }
    uint32_t n_cur = (uint32_t)cnt;
    uint64_t *cur = src;
    while (n_cur > 1) {
      /* wave-cooperative mulmod (one value per wave, limbs in registers).
       * Keep every pass parallel: each wave chains <=64 values, so the final
       * pass never degenerates into one wave grinding 512 serial mulmods. */
      uint32_t stride = n_cur > 64 ? (n_cur + 63) / 64 : 1;
      if (stride > 1024) stride = 1024;
      uint32_t waves_per_block = 4; /* 256 threads */
      uint32_t blocks = (stride + waves_per_block - 1) / waves_per_block;
      hipLaunchKernelGGL(kv_u3072_reduce_wave_kernel, dim3(blocks), dim3(256), 0,
                         stream, cur, n_cur, stride, pa);
      n_cur = stride;
      cur = pa;
      std::swap(pa, pb);
    }
    HIP_CHECK(hipMemcpyAsync(outp + half * 384, cur, 384, hipMemcpyDeviceToHost,
                             stream));
  }
  tev_rec(ctx, 11, stream);
  HIP_CHECK(hipGetLastError());
  *launched = true;
  return 0;
}

/* caller holds ctx->mu; pre_codes (optional) pre-fails txs (e.g. missing
 * outpoints from the populate step) so they skip validation and muhash */
static int validate_block_impl(kv_ctx *ctx, const uint8_t *blob, size_t blob_len,
                               uint64_t pov_daa_score, uint64_t block_daa_score,
                               uint32_t flags, int32_t *tx_codes_out,
                               uint64_t *fees_out, uint8_t *muhash_partial_out,
                               const int32_t *pre_codes) {
  vector<HTx> txs;
  int n_txs = parse_blob_host(blob, blob_len, txs);
  if (n_txs < 0) {
    set_error("kv_validate_block: malformed blob");
    return -1;
  }
  uint64_t sigop_units = ctx->params.mass_per_sig_op * KVH_UNITS_PER_GRAM;
  ValidateBufs &vb = vb_of(ctx);
  const bool kv_timing = getenv("KV_TIMING") != nullptr;
  auto vt_now = []() { return std::chrono::steady_clock::now(); };
  auto vt_ms = [](std::chrono::steady_clock::time_point a,
                  std::chrono::steady_clock::time_point b) {
    return std::chrono::duration<double, std::milli>(b - a).count();
  };
  auto vt0 = vt_now();
  tev_ensure(ctx);
  bool tev_rec_pair[6] = {false, false, false, false, false, false};
  ctx->last_timings = kv_validate_timings{};
  {
    /* generous upper bound: blob + verify jobs + muhash jobs */
    size_t n_units = 0;
    for (auto &tx : txs) n_units += tx.inputs.size() + tx.outputs.size();
    if (stage_reserve(ctx, blob_len + n_units * (sizeof(kv::kv_job) +
                                                 sizeof(kv::kv_elem_job)) +
                               (size_t)n_txs * 64 + 65536))
      return -2;
  }

  /* phase 1: host integer checks + classification, fanned over the host cores
   * (⇔ the reference's rayon pool) with per-chunk job lists so the GPU job
   * order after concatenation equals the serial order */
  std::vector<kv::kv_job> sjobs, ejobs;
  std::vector<std::vector<InputPlan>> plans(n_txs);
  std::vector<int32_t> codes(n_txs, 0);
  std::vector<uint64_t> fees(n_txs, 0);
  if (pre_codes)
    for (int t = 0; t < n_txs; t++) codes[t] = pre_codes[t];
  unsigned P1T = kvh_threads((uint32_t)n_txs);
  uint32_t p1_chunk = ((uint32_t)n_txs + P1T - 1) / P1T;
  std::vector<std::vector<kv::kv_job>> tl_s(P1T), tl_e(P1T);
  auto phase1_tx = [&](uint32_t t, std::vector<kv::kv_job> &sjobs,
                       std::vector<kv::kv_job> &ejobs) {
    HTx &tx = txs[t];
    if (codes[t]) return;
    if (h_is_coinbase(tx)) { /* coinbase never enters this path (utxo_validation.rs:297) */
      codes[t] = KV_ERR_BAD_BLOB;
      return;
    }
    int code = 0;
    for (auto &in : tx.inputs)
      if (in.utxo_is_coinbase &&
          in.utxo_daa_score + ctx->params.coinbase_maturity > pov_daa_score) {
        code = KV_ERR_IMMATURE_COINBASE;
        break;
      }
    uint64_t total_in = 0, total_out = 0;
    if (!code) {
      for (auto &in : tx.inputs) {
        if (total_in + in.utxo_amount < total_in) {
          code = KV_ERR_INPUT_AMOUNT_OVERFLOW;
          break;
        }
        total_in += in.utxo_amount;
        if (total_in > KVH_MAX_SOMPI) {
          code = KV_ERR_INPUT_AMOUNT_TOO_HIGH;
          break;
        }
      }
    }
    if (!code) {
      for (auto &o : tx.outputs) total_out += o.value;
      if (total_in < total_out) code = KV_ERR_SPEND_TOO_HIGH;
    }
    if (!code && flags != KV_FLAGS_SKIP_MASS_CHECK) {
      /* check_mass_commitment (tx_validation_in_utxo_context.rs:126-134) */
      uint64_t calc = 0;
      if (kvh_storage_mass(tx, false, &calc))
        code = KV_ERR_MASS_INCOMPUTABLE;
      else if (calc != tx.storage_mass)
        code = KV_ERR_WRONG_MASS;
    }
    if (!code) {
      for (auto &in : tx.inputs) {
        if ((in.sequence & KVH_SEQ_DISABLED) == KVH_SEQ_DISABLED) continue;
        int64_t lock = (int64_t)in.utxo_daa_score + (int64_t)(in.sequence & KVH_SEQ_MASK) - 1;
        if (lock >= (int64_t)pov_daa_score) {
          code = KV_ERR_SEQUENCE_LOCK;
          break;
        }
      }
    }
    if (!code)
      for (auto &o : tx.outputs)
        if (o.has_covenant) {
          code = KV_ERR_BAD_BLOB; /* covenants out of round-1 scope */
          break;
        }
    codes[t] = code;
    fees[t] = total_in - total_out;
    if (code || flags == KV_FLAGS_SKIP_SCRIPT_CHECKS) return;
    plans[t].reserve(tx.inputs.size());
    for (uint32_t i = 0; i < tx.inputs.size(); i++)
      plans[t].push_back(
          classify_input(ctx, tx, tx.inputs[i], sigop_units, sjobs, ejobs, t, i));
  };
  {
    if (P1T == 1) {
      for (uint32_t t = 0; t < (uint32_t)n_txs; t++)
        phase1_tx(t, tl_s[0], tl_e[0]);
    } else {
      KvhPool::inst().run([&](unsigned k) {
        uint32_t lo = k * p1_chunk,
                 hi = std::min((uint32_t)n_txs, lo + p1_chunk);
        for (uint32_t t = lo; t < hi; t++) phase1_tx(t, tl_s[k], tl_e[k]);
      });
    }
    /* concatenate chunk job lists (serial order preserved) and rebase the
     * per-plan job indices by each chunk's offset */
    std::vector<size_t> s_base(P1T, 0), e_base(P1T, 0);
    size_t sa = 0, ea = 0;
    for (unsigned k = 0; k < P1T; k++) {
      s_base[k] = sa;
      e_base[k] = ea;
      sa += tl_s[k].size();
      ea += tl_e[k].size();
    }
    sjobs.reserve(sa);
    ejobs.reserve(ea);
    for (unsigned k = 0; k < P1T; k++) {
      sjobs.insert(sjobs.end(), tl_s[k].begin(), tl_s[k].end());
      ejobs.insert(ejobs.end(), tl_e[k].begin(), tl_e[k].end());
    }
    if (P1T > 1)
      for (int t = 0; t < n_txs; t++) {
        unsigned k = (uint32_t)t / p1_chunk;
        for (auto &pl : plans[t]) {
          if (pl.job >= 0) pl.job += (int32_t)(pl.ecdsa ? e_base[k] : s_base[k]);
          for (auto &ms : pl.msig_sigs)
            if (ms.job0 >= 0) ms.job0 += (int32_t)s_base[k];
        }
      }
  }

  auto vt1 = vt_now();
  auto vt1a = vt1, vt1b = vt1, vt1c = vt1;

  /* phase 1.5: sig cache ⇔ TransactionValidator sig_cache (caches.rs:57-82).
   * Key = blake2b-256(tx_id ‖ digest(all input entries) ‖ input_index ‖
   * hash_type ‖ ecdsa ‖ sig ‖ pk): tx_id pins everything else the sighash
   * commits to (it excludes sig scripts and utxo entries — those are keyed
   * explicitly). Hits skip the GPU entirely; the job lists are compacted to
   * the misses and statuses scattered back. */
  size_t ns_all = sjobs.size(), ne_all = ejobs.size();
  std::vector<uint8_t> s_status(ns_all), e_status(ne_all);
  std::vector<uint32_t> s_map, e_map;
  std::vector<kv_sig_key> s_keys, e_keys;
  bool use_cache = ctx->params.sig_cache_size > 0 && ns_all + ne_all > 0;
  if (use_cache) {
    std::vector<uint8_t> need(n_txs, 0);
    for (auto &j : sjobs) need[j.tx_index] = 1;
    for (auto &j : ejobs) need[j.tx_index] = 1;
    std::vector<std::array<uint8_t, 32>> ed(n_txs);
    kvh_parallel_for((uint32_t)n_txs, [&](uint32_t t) {
      if (!need[t]) return;
      std::vector<uint8_t> buf;
      buf.reserve(txs[t].inputs.size() * 60);
      for (auto &in : txs[t].inputs) {
        uint8_t tmp[22];
        memcpy(tmp, &in.utxo_amount, 8);
        memcpy(tmp + 8, &in.utxo_daa_score, 8);
        tmp[16] = in.utxo_is_coinbase;
        tmp[17] = in.utxo_has_cov;
        memcpy(tmp + 18, &in.utxo_spk_version, 2);
        uint16_t sl = (uint16_t)in.utxo_spk_len;
        memcpy(tmp + 20, &sl, 2);
        buf.insert(buf.end(), tmp, tmp + 22);
        buf.insert(buf.end(), in.utxo_spk, in.utxo_spk + in.utxo_spk_len);
      }
      h_blake2b_keyed(nullptr, 0, buf.data(), buf.size(), ed[t].data());
    });
    auto keygen = [&](const std::vector<kv::kv_job> &jobs,
                      std::vector<kv_sig_key> &keys) {
      keys.resize(jobs.size());
      kvh_parallel_for((uint32_t)jobs.size(), [&](uint32_t i) {
        const kv::kv_job &j = jobs[i];
        uint8_t buf[32 + 32 + 4 + 2 + 64 + 33];
        size_t o = 0;
        memcpy(buf + o, txs[j.tx_index].tx_id, 32);
        o += 32;
        memcpy(buf + o, ed[j.tx_index].data(), 32);
        o += 32;
        memcpy(buf + o, &j.input_index, 4);
        o += 4;
        buf[o++] = j.hash_type;
        buf[o++] = j.ecdsa;
        memcpy(buf + o, blob + j.sig_off, 64);
        o += 64;
        size_t pklen = j.ecdsa ? 33 : 32;
        memcpy(buf + o, blob + j.pk_off, pklen);
        o += pklen;
        uint8_t h[32];
        h_blake2b_keyed(nullptr, 0, buf, o, h);
        memcpy(keys[i].w, h, 32);
      });
    };
    keygen(sjobs, s_keys);
    keygen(ejobs, e_keys);
    auto probe = [&](std::vector<kv::kv_job> &jobs,
                     const std::vector<kv_sig_key> &keys,
                     std::vector<uint8_t> &status, std::vector<uint32_t> &map) {
      std::vector<kv::kv_job> miss;
      miss.reserve(jobs.size());
      for (size_t i = 0; i < jobs.size(); i++) {
        auto it = ctx->sig_cache.find(keys[i]);
        if (it != ctx->sig_cache.end()) {
          status[i] = it->second;
          ctx->cache_hits++;
        } else {
          ctx->cache_misses++;
          map.push_back((uint32_t)i);
          miss.push_back(jobs[i]);
        }
      }
      jobs.swap(miss);
    };
    probe(sjobs, s_keys, s_status, s_map);
    probe(ejobs, e_keys, e_status, e_map);
  }

  /* phase 2: GPU — subhashes, sighash+tuple assembly, EC verify (cache
   * misses only when the cache is on) */
  size_t ns = sjobs.size(), ne = ejobs.size();
  std::vector<uint8_t> s_gpu(ns), e_gpu(ne);
  bool blob_uploaded = false, subhash_done = false;
  if (ns + ne > 0) {
    if (vb.blob.ensure(blob_len) || vb.subhashes.ensure((size_t)n_txs * 160))
      return -2;
    if (h2d_staged(ctx, vb.blob.p, blob, blob_len, ctx->stream)) return -2;
    HIP_CHECK(hipEventRecord(ctx->ev_blob, ctx->stream));
    blob_uploaded = true;
    tev_rec(ctx, 0);
    hipLaunchKernelGGL(kv::kv_tx_subhash_kernel, dim3((n_txs + 255) / 256), dim3(256),
                       0, ctx->stream, (const uint8_t *)vb.blob.p, (uint32_t)n_txs,
                       (uint8_t *)vb.subhashes.p);
    tev_rec(ctx, 1);
    tev_rec_pair[0] = true;
    HIP_CHECK(hipEventRecord(ctx->ev_sub, ctx->stream));
    subhash_done = true;
    if (ns) {
      if (vb.s_jobs.ensure(ns * sizeof(kv::kv_job)) ||
          vb.s_tuples.ensure(ns * 128) || vb.s_bitmap.ensure((ns + 63) / 64 * 8) ||
          vb.s_status.ensure(ns))
        return -2;
      if (h2d_staged(ctx, vb.s_jobs.p, sjobs.data(), ns * sizeof(kv::kv_job),
                     ctx->stream))
        return -2;
      tev_rec(ctx, 2);
      hipLaunchKernelGGL(kv::kv_sighash_assemble_kernel,
                         dim3(((uint32_t)ns + 255) / 256), dim3(256), 0, ctx->stream,
                         (const uint8_t *)vb.blob.p, (const uint8_t *)vb.subhashes.p,
                         (const kv::kv_job *)vb.s_jobs.p, (uint32_t)ns,
                         (uint8_t *)vb.s_tuples.p, (uint8_t *)vb.s_tuples.p);
      tev_rec(ctx, 3);
      tev_rec_pair[1] = true;
      tev_rec(ctx, 6);
      hipLaunchKernelGGL(kv::kv_schnorr_verify_kernel, dim3(((uint32_t)ns + 255) / 256),
                         dim3(256), 0, ctx->stream, (const uint8_t *)vb.s_tuples.p,
                         (unsigned long long)ns, (unsigned long long *)vb.s_bitmap.p,
                         (uint8_t *)vb.s_status.p);
      tev_rec(ctx, 7);
      tev_rec_pair[3] = true;
      ctx->last_timings.n_schnorr += ns;
    }
    if (ne) {
      /* ecdsa chain on stream2: at block-path sizes both verify kernels are
       * tiny latency-bound dispatches (a few hundred waves on a 2048-wave
       * machine) — overlapping them hides the whole ecdsa chain behind the
       * schnorr one */
      if (vb.e_jobs.ensure(ne * sizeof(kv::kv_job)) ||
          vb.e_tuples.ensure(ne * 132) || vb.e_bitmap.ensure((ne + 63) / 64 * 8) ||
          vb.e_status.ensure(ne))
        return -2;
      HIP_CHECK(hipStreamWaitEvent(ctx->stream2, ctx->ev_sub, 0));
      if (h2d_staged(ctx, vb.e_jobs.p, ejobs.data(), ne * sizeof(kv::kv_job),
                     ctx->stream2))
        return -2;
      tev_rec(ctx, 4, ctx->stream2);
      hipLaunchKernelGGL(kv::kv_sighash_assemble_kernel,
                         dim3(((uint32_t)ne + 255) / 256), dim3(256), 0, ctx->stream2,
                         (const uint8_t *)vb.blob.p, (const uint8_t *)vb.subhashes.p,
                         (const kv::kv_job *)vb.e_jobs.p, (uint32_t)ne,
                         (uint8_t *)vb.e_tuples.p, (uint8_t *)vb.e_tuples.p);
      tev_rec(ctx, 5, ctx->stream2);
      tev_rec_pair[2] = true;
      tev_rec(ctx, 8, ctx->stream2);
      hipLaunchKernelGGL(kv::kv_ecdsa_verify_kernel, dim3(((uint32_t)ne + 255) / 256),
                         dim3(256), 0, ctx->stream2, (const uint8_t *)vb.e_tuples.p,
                         (unsigned long long)ne, (unsigned long long *)vb.e_bitmap.p,
                         (uint8_t *)vb.e_status.p);
      tev_rec(ctx, 9, ctx->stream2);
      tev_rec_pair[4] = true;
      ctx->last_timings.n_ecdsa += ne;
    }
    HIP_CHECK(hipGetLastError());
  }
  vt1a = vt_now(); /* phase-2 launches done */

  /* optimistic muhash on stream3, overlapped with the verify chains: include
   * every tx that passed the phase-1 integer checks. Phase 4 consumes the
   * result when the final accept set matches (the overwhelmingly common
   * case) and re-enqueues the exact set otherwise. */
  bool mu_opt_launched = false;
  std::vector<uint8_t> mu_inc;
  std::vector<kv::kv_elem_job> mu_jobs;
  if (muhash_partial_out) {
    mu_inc.resize(n_txs);
    for (int t = 0; t < n_txs; t++) mu_inc[t] = codes[t] == 0;
    if (!blob_uploaded) {
      if (vb.blob.ensure(blob_len)) return -2;
      if (h2d_staged(ctx, vb.blob.p, blob, blob_len, ctx->stream)) return -2;
      HIP_CHECK(hipEventRecord(ctx->ev_blob, ctx->stream));
      blob_uploaded = true;
    }
    HIP_CHECK(hipStreamWaitEvent(ctx->stream3, ctx->ev_blob, 0));
    int mrc = enqueue_muhash(ctx, txs, mu_inc.data(), block_daa_score, ctx->stream3,
                             &mu_opt_launched, mu_jobs);
    if (mrc) return mrc;
  }
  vt1b = vt_now(); /* muhash enqueue done */
  /* status readbacks last (pinned, truly async), then join the streams */
  if (ns) {
    if (ctx->h_s_cap < ns) {
      if (ctx->h_s_status) (void)hipHostFree(ctx->h_s_status);
      size_t nc = ns + ns / 2 + 256;
      if (hipHostMalloc(&ctx->h_s_status, nc) != hipSuccess) {
        ctx->h_s_status = nullptr;
        ctx->h_s_cap = 0;
        set_error("hipHostMalloc failed (status)");
        return -2;
      }
      ctx->h_s_cap = nc;
    }
    HIP_CHECK(hipMemcpyAsync(ctx->h_s_status, vb.s_status.p, ns,
                             hipMemcpyDeviceToHost, ctx->stream));
  }
  if (ne) {
    if (ctx->h_e_cap < ne) {
      if (ctx->h_e_status) (void)hipHostFree(ctx->h_e_status);
      size_t nc = ne + ne / 2 + 256;
      if (hipHostMalloc(&ctx->h_e_status, nc) != hipSuccess) {
        ctx->h_e_status = nullptr;
        ctx->h_e_cap = 0;
        set_error("hipHostMalloc failed (status)");
        return -2;
      }
      ctx->h_e_cap = nc;
    }
    HIP_CHECK(hipMemcpyAsync(ctx->h_e_status, vb.e_status.p, ne,
                             hipMemcpyDeviceToHost, ctx->stream2));
  }
  auto vt2 = vt_now();
  if (ns + ne > 0) {
    HIP_CHECK(hipStreamSynchronize(ctx->stream));
    HIP_CHECK(hipStreamSynchronize(ctx->stream2));
    if (ns) memcpy(s_gpu.data(), ctx->h_s_status, ns);
    if (ne) memcpy(e_gpu.data(), ctx->h_e_status, ne);
  }
  auto vt3 = vt_now();

  /* scatter GPU statuses back and remember fresh verdicts */
  if (use_cache) {
    size_t cap = (size_t)ctx->params.sig_cache_size;
    for (size_t k = 0; k < ns; k++) {
      s_status[s_map[k]] = s_gpu[k];
      if (ctx->sig_cache.size() >= cap)
        ctx->sig_cache.erase(ctx->sig_cache.begin());
      ctx->sig_cache.emplace(s_keys[s_map[k]], s_gpu[k]);
      ctx->cache_insertions++;
    }
    for (size_t k = 0; k < ne; k++) {
      e_status[e_map[k]] = e_gpu[k];
      if (ctx->sig_cache.size() >= cap)
        ctx->sig_cache.erase(ctx->sig_cache.begin());
      ctx->sig_cache.emplace(e_keys[e_map[k]], e_gpu[k]);
      ctx->cache_insertions++;
    }
  } else {
    s_status.swap(s_gpu);
    e_status.swap(e_gpu);
  }

  /* phase 2.5: interpreter rounds — resolve suspended general-interpreter
   * scripts over the GPU verify batch, one signature site per script per
   * round (collect/replay protocol, kv_script_host.inc). Rounds =
   * max signature-site depth over this block's non-template scripts (almost
   * always 1); each round is one GPU batch over ALL suspended scripts, so
   * the launch count does not scale with signature count. */
  if (flags != KV_FLAGS_SKIP_SCRIPT_CHECKS) {
    std::vector<std::pair<uint32_t, uint32_t>> islots;
    for (int t = 0; t < n_txs; t++) {
      if (codes[t]) continue;
      for (uint32_t i = 0; i < plans[t].size(); i++)
        if (plans[t][i].kind == PLAN_INTERP) islots.emplace_back((uint32_t)t, i);
    }
    std::vector<uint8_t> st_h, et_h; /* host-side tuple arrays per round */
    while (!islots.empty()) {
      st_h.clear();
      et_h.clear();
      std::vector<kv::kv_job> mjobs_s, mjobs_e; /* sighash-msg jobs */
      struct ReqRef {
        uint8_t ecdsa;
        uint32_t idx;
      };
      std::vector<std::vector<ReqRef>> refs(islots.size());
      for (size_t sl = 0; sl < islots.size(); sl++) {
        uint32_t t = islots[sl].first, i = islots[sl].second;
        InputPlan &pl = plans[t][i];
        const HInput &in = txs[t].inputs[i];
        for (const auto &rq : pl.pending) {
          if (!rq.ecdsa) {
            uint32_t idx = (uint32_t)(st_h.size() / 128);
            st_h.resize(st_h.size() + 128);
            uint8_t *tp = st_h.data() + (size_t)idx * 128;
            memcpy(tp, rq.sig, 64);
            memcpy(tp + 64, rq.pk, 32);
            if (rq.literal)
              memcpy(tp + 96, rq.msg, 32);
            else /* sig_off carries the tuple slot for the msg kernel */
              mjobs_s.push_back(
                  kv::kv_job{t, in.rec_off, i, idx, 0, rq.hash_type, 0, 0});
            refs[sl].push_back({0, idx});
          } else {
            uint32_t idx = (uint32_t)(et_h.size() / 132);
            et_h.resize(et_h.size() + 132);
            uint8_t *tp = et_h.data() + (size_t)idx * 132;
            memcpy(tp, rq.sig, 64);
            memcpy(tp + 64, rq.pk, 33);
            if (rq.literal)
              memcpy(tp + 97, rq.msg, 32);
            else
              mjobs_e.push_back(
                  kv::kv_job{t, in.rec_off, i, idx, 0, rq.hash_type, 1, 0});
            refs[sl].push_back({1, idx});
          }
        }
      }
      size_t ns_i = st_h.size() / 128, ne_i = et_h.size() / 132;
      std::vector<uint8_t> s_st(ns_i), e_st(ne_i);
      if (ns_i + ne_i > 0) {
        if (!blob_uploaded) {
          if (vb.blob.ensure(blob_len)) return -2;
          HIP_CHECK(hipMemcpyAsync(vb.blob.p, blob, blob_len,
                                   hipMemcpyHostToDevice, ctx->stream));
          blob_uploaded = true;
        }
        if (!subhash_done) {
          if (vb.subhashes.ensure((size_t)n_txs * 160)) return -2;
          hipLaunchKernelGGL(kv::kv_tx_subhash_kernel, dim3((n_txs + 255) / 256),
                             dim3(256), 0, ctx->stream,
                             (const uint8_t *)vb.blob.p, (uint32_t)n_txs,
                             (uint8_t *)vb.subhashes.p);
          subhash_done = true;
        }
        if (ns_i) {
          if (vb.s_tuples.ensure(st_h.size()) ||
              vb.s_bitmap.ensure((ns_i + 63) / 64 * 8) ||
              vb.s_status.ensure(ns_i) ||
              (!mjobs_s.empty() &&
               vb.s_jobs.ensure(mjobs_s.size() * sizeof(kv::kv_job))))
            return -2;
          HIP_CHECK(hipMemcpyAsync(vb.s_tuples.p, st_h.data(), st_h.size(),
                                   hipMemcpyHostToDevice, ctx->stream));
          if (!mjobs_s.empty()) {
            HIP_CHECK(hipMemcpyAsync(vb.s_jobs.p, mjobs_s.data(),
                                     mjobs_s.size() * sizeof(kv::kv_job),
                                     hipMemcpyHostToDevice, ctx->stream));
            hipLaunchKernelGGL(kv::kv_sighash_msg_kernel,
                               dim3(((uint32_t)mjobs_s.size() + 255) / 256),
                               dim3(256), 0, ctx->stream,
                               (const uint8_t *)vb.blob.p,
                               (const uint8_t *)vb.subhashes.p,
                               (const kv::kv_job *)vb.s_jobs.p,
                               (uint32_t)mjobs_s.size(),
                               (uint8_t *)vb.s_tuples.p, 128u, 96u);
          }
          hipLaunchKernelGGL(kv::kv_schnorr_verify_kernel,
                             dim3(((uint32_t)ns_i + 255) / 256), dim3(256), 0,
                             ctx->stream, (const uint8_t *)vb.s_tuples.p,
                             (unsigned long long)ns_i,
                             (unsigned long long *)vb.s_bitmap.p,
                             (uint8_t *)vb.s_status.p);
          HIP_CHECK(hipMemcpyAsync(s_st.data(), vb.s_status.p, ns_i,
                                   hipMemcpyDeviceToHost, ctx->stream));
        }
        if (ne_i) {
          if (vb.e_tuples.ensure(et_h.size()) ||
              vb.e_bitmap.ensure((ne_i + 63) / 64 * 8) ||
              vb.e_status.ensure(ne_i) ||
              (!mjobs_e.empty() &&
               vb.e_jobs.ensure(mjobs_e.size() * sizeof(kv::kv_job))))
            return -2;
          HIP_CHECK(hipMemcpyAsync(vb.e_tuples.p, et_h.data(), et_h.size(),
                                   hipMemcpyHostToDevice, ctx->stream));
          if (!mjobs_e.empty()) {
            HIP_CHECK(hipMemcpyAsync(vb.e_jobs.p, mjobs_e.data(),
                                     mjobs_e.size() * sizeof(kv::kv_job),
                                     hipMemcpyHostToDevice, ctx->stream));
            hipLaunchKernelGGL(kv::kv_sighash_msg_kernel,
                               dim3(((uint32_t)mjobs_e.size() + 255) / 256),
                               dim3(256), 0, ctx->stream,
                               (const uint8_t *)vb.blob.p,
                               (const uint8_t *)vb.subhashes.p,
                               (const kv::kv_job *)vb.e_jobs.p,
                               (uint32_t)mjobs_e.size(),
                               (uint8_t *)vb.e_tuples.p, 132u, 97u);
          }
          hipLaunchKernelGGL(kv::kv_ecdsa_verify_kernel,
                             dim3(((uint32_t)ne_i + 255) / 256), dim3(256), 0,
                             ctx->stream, (const uint8_t *)vb.e_tuples.p,
                             (unsigned long long)ne_i,
                             (unsigned long long *)vb.e_bitmap.p,
                             (uint8_t *)vb.e_status.p);
          HIP_CHECK(hipMemcpyAsync(e_st.data(), vb.e_status.p, ne_i,
                                   hipMemcpyDeviceToHost, ctx->stream));
        }
        HIP_CHECK(hipGetLastError());
        HIP_CHECK(hipStreamSynchronize(ctx->stream));
      }
      /* feed statuses back as this site's memo entry and replay */
      std::vector<uint8_t> survive(islots.size(), 0);
      kvh_parallel_for((uint32_t)islots.size(), [&](uint32_t sl) {
        uint32_t t = islots[sl].first, i = islots[sl].second;
        InputPlan &pl = plans[t][i];
        std::vector<uint8_t> entry(refs[sl].size());
        for (size_t r = 0; r < refs[sl].size(); r++)
          entry[r] = refs[sl][r].ecdsa ? e_st[refs[sl][r].idx]
                                       : s_st[refs[sl][r].idx];
        pl.memo.push_back(std::move(entry));
        pl.pending.clear();
        kvhost::KvsRunCtx rctx;
        rctx.memo = &pl.memo;
        rctx.pending = &pl.pending;
        rctx.seqc_fn = ctx->seqc_fn;
        rctx.seqc_user = ctx->seqc_user;
        const HInput &in = txs[t].inputs[i];
        int rc = kvh_run_input_script(txs[t], in, i, sigop_units, &rctx);
        if (rc == kvhost::KVH_SCRIPT_SUSPEND) {
          survive[sl] = 1;
          return;
        }
        pl.kind = PLAN_NONE;
        if (rc == kvhost::KVH_SCRIPT_DEFER)
          pl.pre_code = KV_TX_DEFER_TO_CPU;
        else if (rc != 0)
          pl.pre_code = (in.sig_script_len == 0 ? KV_ERR_SIGNATURE_EMPTY_BASE
                                                : KV_ERR_SIGNATURE_INVALID_BASE) +
                        rc;
        else
          pl.pre_code = 0;
      });
      std::vector<std::pair<uint32_t, uint32_t>> next;
      for (size_t sl = 0; sl < islots.size(); sl++)
        if (survive[sl]) next.push_back(islots[sl]);
      islots.swap(next);
    }
  }

  auto vt4 = vt_now();
  /* phase 3: resolution (first failing input wins, sequential semantics);
   * read-only over the GPU statuses → fans over the host cores */
  kvh_parallel_for((uint32_t)n_txs, [&](uint32_t t) {
    if (codes[t] || flags == KV_FLAGS_SKIP_SCRIPT_CHECKS) return;
    for (uint32_t i = 0; i < txs[t].inputs.size(); i++) {
      int c = resolve_input(plans[t][i], txs[t].inputs[i], sigop_units,
                            s_status.data(), e_status.data());
      if (c) {
        codes[t] = c;
        break;
      }
    }
  });

  auto vt5 = vt_now();
  /* phase 4: consume the optimistic muhash, or re-enqueue over the exact
   * accept set when some tx failed validation after the integer checks */
  if (muhash_partial_out) {
    bool match = true;
    for (int t = 0; t < n_txs; t++)
      if ((codes[t] == 0) != (mu_inc[t] != 0)) {
        match = false;
        break;
      }
    HIP_CHECK(hipStreamSynchronize(ctx->stream3)); /* optimistic writes done */
    if (!match) {
      for (int t = 0; t < n_txs; t++) mu_inc[t] = codes[t] == 0;
      bool relaunched = false;
      int mrc = enqueue_muhash(ctx, txs, mu_inc.data(), block_daa_score,
                               ctx->stream, &relaunched, mu_jobs);
      if (mrc) return mrc;
      HIP_CHECK(hipStreamSynchronize(ctx->stream));
      tev_rec_pair[5] = relaunched;
    } else {
      tev_rec_pair[5] = mu_opt_launched;
    }
    memcpy(muhash_partial_out, ctx->h_mu_partial, 768);
  }

  /* collect per-kernel timings (the stream is synchronized by now) */
  HIP_CHECK(hipStreamSynchronize(ctx->stream));
  if (tev_rec_pair[0]) ctx->last_timings.subhash_ms = tev_ms(ctx, 0);
  if (tev_rec_pair[1]) ctx->last_timings.s_assemble_ms = tev_ms(ctx, 1);
  if (tev_rec_pair[2]) ctx->last_timings.e_assemble_ms = tev_ms(ctx, 2);
  if (tev_rec_pair[3]) ctx->last_timings.schnorr_ms = tev_ms(ctx, 3);
  if (tev_rec_pair[4]) ctx->last_timings.ecdsa_ms = tev_ms(ctx, 4);
  if (tev_rec_pair[5]) ctx->last_timings.muhash_ms = tev_ms(ctx, 5);

  for (int t = 0; t < n_txs; t++)
    if (codes[t]) fees[t] = 0; /* fee defined only for accepted txs */
  memcpy(tx_codes_out, codes.data(), (size_t)n_txs * 4);
  memcpy(fees_out, fees.data(), (size_t)n_txs * 8);
  if (kv_timing)
    fprintf(stderr,
            "[kv_timing] validate: parse+p1 %.2f cache+enq %.2f (p2 %.2f mu %.2f "
            "rb %.2f) gpu-wait %.2f interp %.2f resolve %.2f muhash-wait %.2f "
            "total %.2f ms\n",
            vt_ms(vt0, vt1), vt_ms(vt1, vt2), vt_ms(vt1, vt1a), vt_ms(vt1a, vt1b),
            vt_ms(vt1b, vt2), vt_ms(vt2, vt3), vt_ms(vt3, vt4), vt_ms(vt4, vt5),
            vt_ms(vt5, vt_now()), vt_ms(vt0, vt_now()));
  return 0;
}

extern "C" int kv_validate_block(kv_ctx *ctx, const uint8_t *blob, size_t blob_len,
                                 uint64_t pov_daa_score, uint64_t block_daa_score,
                                 uint32_t flags, int32_t *tx_codes_out,
                                 uint64_t *fees_out, uint8_t *muhash_partial_out) {
  if (!ctx) {
    set_error("kv_validate_block: null ctx");
    return -1;
  }
  std::lock_guard<std::mutex> lk(ctx->mu);
  (void)hipGetLastError(); /* clear any stale per-thread error so the
      launch-config checks below only see THIS call's launches */
  return validate_block_impl(ctx, blob, blob_len, pov_daa_score, block_daa_score,
                             flags, tx_codes_out, fees_out, muhash_partial_out,
                             nullptr);
}


/* ---------------- GPU-resident UTXO set (kv_utxo_kernels.hip) ----------------
 * ⇔ utxo_collection.rs:5 HashMap + the populate/diff steps
 * (utxo_validation.rs:351-390, utxo_diff.rs:224). Entries are packed 64B
 * records: amount u64 ‖ daa u64 ‖ flags u16 (bit0 coinbase) ‖ spk_version u16 ‖
 * spk_len u32 ‖ spk[36] inline. */

extern "C" int kv_utxo_reset(kv_ctx *ctx, uint64_t capacity) {
  std::lock_guard<std::mutex> lk(ctx->mu);
  (void)hipGetLastError(); /* clear any stale per-thread error so the
      launch-config checks below only see THIS call's launches */
  uint64_t cap = 64;
  while (cap < capacity * 2) cap <<= 1; /* ≤50% load factor */
  if (ctx->d_utxo) (void)hipFree(ctx->d_utxo);
  if (hipMalloc(&ctx->d_utxo, cap * sizeof(kv::utxo_slot)) != hipSuccess) {
    ctx->d_utxo = nullptr;
    ctx->utxo_cap = 0;
    set_error("kv_utxo_reset: hipMalloc failed");
    return -2;
  }
  HIP_CHECK(hipMemsetAsync(ctx->d_utxo, 0, cap * sizeof(kv::utxo_slot), ctx->stream));
  HIP_CHECK(hipStreamSynchronize(ctx->stream));
  ctx->utxo_cap = cap;
  ctx->arena_head = 0; /* the spk arena compacts on reset (allocation kept) */
  return 0;
}

static int utxo_stage(kv_ctx *ctx, const uint8_t *outpoints, const uint8_t *values,
                      size_t n) {
  if (ensure_cap((void **)&ctx->d_op_in, &ctx->d_op_cap, n * 36)) return -2;
  HIP_CHECK(hipMemcpyAsync(ctx->d_op_in, outpoints, n * 36, hipMemcpyHostToDevice,
                           ctx->stream));
  if (values) {
    if (ensure_cap((void **)&ctx->d_val_in, &ctx->d_val_cap, n * 64)) return -2;
    HIP_CHECK(hipMemcpyAsync(ctx->d_val_in, values, n * 64, hipMemcpyHostToDevice,
                             ctx->stream));
  }
  return 0;
}

static int utxo_upsert_nolock(kv_ctx *ctx, const uint8_t *outpoints,
                              const uint8_t *entries64, size_t n) {
  if (!ctx->d_utxo) {
    set_error("kv_utxo_upsert: call kv_utxo_reset first");
    return -1;
  }
  int rc = utxo_stage(ctx, outpoints, entries64, n);
  if (rc) return rc;
  int *d_fail = ctx->d_utxo_fail; /* per-ctx: a shared flag would race when
                                     two contexts upsert concurrently */
  if (!d_fail) {
    HIP_CHECK(hipMalloc(&d_fail, 4));
    ctx->d_utxo_fail = d_fail;
  }
  HIP_CHECK(hipMemsetAsync(d_fail, 0, 4, ctx->stream));
  hipLaunchKernelGGL(kv::kv_utxo_upsert_kernel, dim3(((uint32_t)n + 255) / 256),
                     dim3(256), 0, ctx->stream, ctx->d_utxo, ctx->utxo_cap - 1,
                     ctx->d_op_in, ctx->d_val_in, (unsigned long long)n, d_fail);
  int fail = 0;
  HIP_CHECK(hipMemcpyAsync(&fail, d_fail, 4, hipMemcpyDeviceToHost, ctx->stream));
  HIP_CHECK(hipStreamSynchronize(ctx->stream));
  if (fail) {
    set_error("kv_utxo_upsert: table full");
    return -3;
  }
  return 0;
}

extern "C" int kv_utxo_upsert(kv_ctx *ctx, const uint8_t *outpoints,
                              const uint8_t *entries64, size_t n) {
  std::lock_guard<std::mutex> lk(ctx->mu);
  (void)hipGetLastError(); /* clear any stale per-thread error so the
      launch-config checks below only see THIS call's launches */
  return utxo_upsert_nolock(ctx, outpoints, entries64, n);
}

/* grow the spk arena to fit `need` more bytes; preserves contents */
static int arena_reserve(kv_ctx *ctx, uint64_t need) {
  uint64_t want = ctx->arena_head + need;
  if (want <= ctx->arena_cap) return 0;
  uint64_t nc = ctx->arena_cap ? ctx->arena_cap : (1u << 20);
  while (nc < want) nc *= 2;
  uint8_t *na = nullptr;
  if (hipMalloc(&na, nc) != hipSuccess) {
    set_error("kv_utxo arena: hipMalloc failed");
    return -2;
  }
  if (ctx->d_arena && ctx->arena_head)
    HIP_CHECK(hipMemcpyAsync(na, ctx->d_arena, ctx->arena_head,
                             hipMemcpyDeviceToDevice, ctx->stream));
  HIP_CHECK(hipStreamSynchronize(ctx->stream));
  if (ctx->h_s_status) (void)hipHostFree(ctx->h_s_status);
  if (ctx->h_e_status) (void)hipHostFree(ctx->h_e_status);
  if (ctx->h_stage) (void)hipHostFree(ctx->h_stage);
  if (ctx->h_mu_partial) (void)hipHostFree(ctx->h_mu_partial);
  if (ctx->d_arena) (void)hipFree(ctx->d_arena);
  ctx->d_arena = na;
  ctx->arena_cap = nc;
  return 0;
}

/* general upsert: spk_len > 36 entries spill their scripts to the arena
 * (bump-allocated; spans leak on remove until kv_utxo_reset — documented) */
static int utxo_upsert_spk_nolock(kv_ctx *ctx, const uint8_t *outpoints,
                                  const uint8_t *entries64, const uint8_t *spk_blob,
                                  size_t spk_blob_len, size_t n) {
  if (!ctx->d_utxo) {
    set_error("kv_utxo_upsert_spk: call kv_utxo_reset first");
    return -1;
  }
  uint64_t long_total = 0;
  for (size_t i = 0; i < n; i++) {
    uint32_t spk_len;
    memcpy(&spk_len, entries64 + i * 64 + 20, 4);
    if (spk_len > 36) {
      if (spk_len > KV_UTXO_MAX_SPK) {
        set_error("kv_utxo_upsert_spk: spk exceeds KV_UTXO_MAX_SPK");
        return -1;
      }
      long_total += spk_len;
    }
  }
  if (long_total > spk_blob_len) {
    set_error("kv_utxo_upsert_spk: spk_blob shorter than the long entries");
    return -1;
  }
  if (long_total == 0)
    return utxo_upsert_nolock(ctx, outpoints, entries64, n);
  int rc = arena_reserve(ctx, long_total);
  if (rc) return rc;
  uint64_t base = ctx->arena_head;
  HIP_CHECK(hipMemcpyAsync(ctx->d_arena + base, spk_blob, long_total,
                           hipMemcpyHostToDevice, ctx->stream));
  ctx->arena_head += long_total;
  /* rewrite the long entries: flags |= ARENA, spk[0..4) = arena offset */
  std::vector<uint8_t> ents(entries64, entries64 + n * 64);
  uint64_t off = base;
  for (size_t i = 0; i < n; i++) {
    uint8_t *e = ents.data() + i * 64;
    uint32_t spk_len;
    memcpy(&spk_len, e + 20, 4);
    if (spk_len <= 36) continue;
    uint16_t flags;
    memcpy(&flags, e + 16, 2);
    flags |= KV_UTXO_F_SPK_ARENA;
    memcpy(e + 16, &flags, 2);
    uint32_t o32 = (uint32_t)off;
    memcpy(e + 24, &o32, 4);
    memset(e + 28, 0, 32);
    off += spk_len;
  }
  return utxo_upsert_nolock(ctx, outpoints, ents.data(), n);
}

extern "C" int kv_utxo_upsert_spk(kv_ctx *ctx, const uint8_t *outpoints,
                                  const uint8_t *entries64, const uint8_t *spk_blob,
                                  size_t spk_blob_len, size_t n) {
  std::lock_guard<std::mutex> lk(ctx->mu);
  (void)hipGetLastError(); /* clear any stale per-thread error so the
      launch-config checks below only see THIS call's launches */
  return utxo_upsert_spk_nolock(ctx, outpoints, entries64, spk_blob,
                                spk_blob_len, n);
}

static int utxo_remove_nolock(kv_ctx *ctx, const uint8_t *outpoints, size_t n) {
  if (!ctx->d_utxo) {
    set_error("kv_utxo_remove: call kv_utxo_reset first");
    return -1;
  }
  int rc = utxo_stage(ctx, outpoints, nullptr, n);
  if (rc) return rc;
  hipLaunchKernelGGL(kv::kv_utxo_remove_kernel, dim3(((uint32_t)n + 255) / 256),
                     dim3(256), 0, ctx->stream, ctx->d_utxo, ctx->utxo_cap - 1,
                     ctx->d_op_in, (unsigned long long)n);
  HIP_CHECK(hipGetLastError());
  HIP_CHECK(hipStreamSynchronize(ctx->stream));
  return 0;
}

extern "C" int kv_utxo_remove(kv_ctx *ctx, const uint8_t *outpoints, size_t n) {
  std::lock_guard<std::mutex> lk(ctx->mu);
  (void)hipGetLastError(); /* clear any stale per-thread error so the
      launch-config checks below only see THIS call's launches */
  return utxo_remove_nolock(ctx, outpoints, n);
}

static int utxo_lookup_nolock(kv_ctx *ctx, const uint8_t *outpoints, size_t n,
                              uint8_t *entries_out, uint64_t *found_bitmap,
                              double *kernel_ms);

/* gather the arena spans of entries flagged KV_UTXO_F_SPK_ARENA (in entry
 * order) into host memory, rewriting each entry's spk field to its offset in
 * the gathered buffer. `take(i)` -> entry pointer or NULL to skip. */
template <class TakeFn>
static int arena_gather_nolock(kv_ctx *ctx, size_t n, TakeFn take,
                               std::vector<uint8_t> &out) {
  std::vector<kv::arena_gather_job> jobs;
  uint32_t dst = 0;
  for (size_t i = 0; i < n; i++) {
    uint8_t *e = take(i);
    if (!e) continue;
    uint16_t flags;
    memcpy(&flags, e + 16, 2);
    if (!(flags & KV_UTXO_F_SPK_ARENA)) continue;
    uint32_t spk_len, src;
    memcpy(&spk_len, e + 20, 4);
    memcpy(&src, e + 24, 4);
    jobs.push_back(kv::arena_gather_job{src, spk_len, dst, 0});
    memcpy(e + 24, &dst, 4); /* rewrite to the gathered-buffer offset */
    dst += spk_len;
  }
  out.resize(dst);
  if (jobs.empty()) return 0;
  if (ensure_cap(&ctx->d_gjobs, &ctx->d_gjobs_cap,
                 jobs.size() * sizeof(kv::arena_gather_job)) ||
      ensure_cap(&ctx->d_gout, &ctx->d_gout_cap, dst))
    return -2;
  HIP_CHECK(hipMemcpyAsync(ctx->d_gjobs, jobs.data(),
                           jobs.size() * sizeof(kv::arena_gather_job),
                           hipMemcpyHostToDevice, ctx->stream));
  hipLaunchKernelGGL(kv::kv_arena_gather_kernel, dim3((uint32_t)jobs.size()),
                     dim3(256), 0, ctx->stream, ctx->d_arena,
                     (const kv::arena_gather_job *)ctx->d_gjobs,
                     (uint32_t)jobs.size(), (uint8_t *)ctx->d_gout);
  HIP_CHECK(hipGetLastError());
  HIP_CHECK(hipMemcpyAsync(out.data(), ctx->d_gout, dst, hipMemcpyDeviceToHost,
                           ctx->stream));
  HIP_CHECK(hipStreamSynchronize(ctx->stream));
  return 0;
}

extern "C" int kv_utxo_lookup_spk(kv_ctx *ctx, const uint8_t *outpoints, size_t n,
                                  uint8_t *entries_out, uint64_t *found_bitmap,
                                  uint8_t *spk_out, size_t spk_cap,
                                  size_t *spk_used) {
  std::lock_guard<std::mutex> lk(ctx->mu);
  (void)hipGetLastError(); /* clear any stale per-thread error so the
      launch-config checks below only see THIS call's launches */
  int rc = utxo_lookup_nolock(ctx, outpoints, n, entries_out, found_bitmap,
                              nullptr);
  if (rc) return rc;
  std::vector<uint8_t> gathered;
  rc = arena_gather_nolock(
      ctx, n,
      [&](size_t i) -> uint8_t * {
        int hit = (found_bitmap[i / 64] >> (i % 64)) & 1;
        return hit ? entries_out + i * 64 : nullptr;
      },
      gathered);
  if (rc) return rc;
  if (spk_used) *spk_used = gathered.size();
  if (gathered.empty()) return 0;
  if (gathered.size() > spk_cap) {
    set_error("kv_utxo_lookup_spk: spk_out too small");
    return -3;
  }
  memcpy(spk_out, gathered.data(), gathered.size());
  return 0;
}

static int utxo_lookup_nolock(kv_ctx *ctx, const uint8_t *outpoints, size_t n,
                              uint8_t *entries_out, uint64_t *found_bitmap,
                              double *kernel_ms) {
  if (!ctx->d_utxo) {
    set_error("kv_utxo_lookup: call kv_utxo_reset first");
    return -1;
  }
  const bool kvt = getenv("KV_TIMING") != nullptr;
  auto tn = []() { return std::chrono::steady_clock::now(); };
  auto t0_ = tn();
  int rc = utxo_stage(ctx, outpoints, nullptr, n);
  if (rc) return rc;
  auto t1_ = tn();
  size_t words = (n + 63) / 64;
  if (ensure_cap((void **)&ctx->d_ent_out, &ctx->d_ent_cap, n * 64)) return -2;
  if (ensure_cap((void **)&ctx->d_bitmap, &ctx->d_bitmap_cap, words * 8)) return -2;
  hipEvent_t t0, t1;
  HIP_CHECK(hipEventCreate(&t0));
  HIP_CHECK(hipEventCreate(&t1));
  HIP_CHECK(hipEventRecord(t0, ctx->stream));
  hipLaunchKernelGGL(kv::kv_utxo_lookup_kernel, dim3(((uint32_t)n + 255) / 256),
                     dim3(256), 0, ctx->stream, ctx->d_utxo, ctx->utxo_cap - 1,
                     ctx->d_op_in, (unsigned long long)n, ctx->d_ent_out,
                     (unsigned long long *)ctx->d_bitmap);
  HIP_CHECK(hipGetLastError());
  HIP_CHECK(hipEventRecord(t1, ctx->stream));
  if (entries_out)
    HIP_CHECK(hipMemcpyAsync(entries_out, ctx->d_ent_out, n * 64,
                             hipMemcpyDeviceToHost, ctx->stream));
  HIP_CHECK(hipMemcpyAsync(found_bitmap, ctx->d_bitmap, words * 8,
                           hipMemcpyDeviceToHost, ctx->stream));
  auto t2_ = tn();
  HIP_CHECK(hipStreamSynchronize(ctx->stream));
  if (kvt)
    fprintf(stderr, "[kv_timing] lookup inner: stage %.2f launch+copyq %.2f sync %.2f ms\n",
            std::chrono::duration<double, std::milli>(t1_ - t0_).count(),
            std::chrono::duration<double, std::milli>(t2_ - t1_).count(),
            std::chrono::duration<double, std::milli>(tn() - t2_).count());
  if (kernel_ms) {
    float ms = 0.f;
    HIP_CHECK(hipEventElapsedTime(&ms, t0, t1));
    *kernel_ms = ms;
  }
  (void)hipEventDestroy(t0);
  (void)hipEventDestroy(t1);
  return 0;
}

extern "C" int kv_utxo_lookup(kv_ctx *ctx, const uint8_t *outpoints, size_t n,
                              uint8_t *entries_out, uint64_t *found_bitmap,
                              double *kernel_ms) {
  std::lock_guard<std::mutex> lk(ctx->mu);
  (void)hipGetLastError(); /* clear any stale per-thread error so the
      launch-config checks below only see THIS call's launches */
  return utxo_lookup_nolock(ctx, outpoints, n, entries_out, found_bitmap,
                            kernel_ms);
}



static int populate_blob_nolock(kv_ctx *ctx, const uint8_t *blob, size_t blob_len,
                                std::vector<kvhost::HTx> &txs, int n_txs,
                                std::vector<int32_t> &pre_codes);

/* ---------------- mempool batch validation ----------------
 * ⇔ validate_mempool_transaction_in_utxo_context (utxo_validation.rs:418-457)
 * fanned over a batch: entries resolve inline from the blob (from_utxo_table
 * = 0, the caller populated) or from the GPU-resident table (= 1; a missing
 * outpoint fails that tx with KV_ERR_MISSING_OUTPOINT, mirroring
 * populate_mempool_transaction_in_utxo_context:392-415). The contextual
 * storage mass is COMPUTED per tx (the carried commitment is ignored — the
 * mempool sets it), validation runs with SkipMassCheck, and the optional
 * feerate threshold rejects fee / normalized_max(mass) <= threshold with
 * KV_ERR_FEERATE_TOO_LOW (tx_validation_in_utxo_context.rs:69-77). */
extern "C" int kv_validate_mempool(kv_ctx *ctx, const uint8_t *blob,
                                   size_t blob_len, uint64_t pov_daa_score,
                                   double feerate_threshold, int from_utxo_table,
                                   int32_t *tx_codes_out, uint64_t *fees_out) {
  if (!ctx) {
    set_error("kv_validate_mempool: null ctx");
    return -1;
  }
  std::lock_guard<std::mutex> lk(ctx->mu);
  (void)hipGetLastError(); /* clear any stale per-thread error so the
      launch-config checks below only see THIS call's launches */
  vector<HTx> txs;
  int n_txs = parse_blob_host(blob, blob_len, txs);
  if (n_txs < 0) {
    set_error("kv_validate_mempool: malformed blob");
    return -1;
  }
  std::vector<int32_t> pre_codes(n_txs, 0);
  const uint8_t *vblob = blob;
  size_t vlen = blob_len;
  vector<HTx> ptxs;
  const vector<HTx> *use_txs = &txs;
  if (from_utxo_table) {
    if (!ctx->d_utxo) {
      set_error("kv_validate_mempool: call kv_utxo_reset first");
      return -1;
    }
    int rc = populate_blob_nolock(ctx, blob, blob_len, txs, n_txs, pre_codes);
    if (rc) return rc;
    vblob = ctx->pop_buf.data();
    vlen = ctx->pop_buf.size();
    if (parse_blob_host(vblob, vlen, ptxs) != n_txs) {
      set_error("kv_validate_mempool: internal rebuild parse");
      return -1;
    }
    use_txs = &ptxs;
  }
  int rc = validate_block_impl(ctx, vblob, vlen, pov_daa_score, pov_daa_score,
                               KV_FLAGS_SKIP_MASS_CHECK, tx_codes_out, fees_out,
                               nullptr, pre_codes.data());
  if (rc) return rc;
  /* contextual mass + feerate post-pass (mass computed BEFORE validation in
   * the reference, so MassIncomputable wins over any validation error) */
  for (int t = 0; t < n_txs; t++) {
    const HTx &tx = (*use_txs)[t];
    if (pre_codes[t]) continue; /* populate failure stands */
    uint64_t storage = 0;
    if (kvh_storage_mass(tx, h_is_coinbase(tx), &storage)) {
      tx_codes_out[t] = KV_ERR_MASS_INCOMPUTABLE;
      fees_out[t] = 0;
    } else if (tx_codes_out[t] == 0 && feerate_threshold > 0) {
      uint64_t m = kvh_normalized_mass(tx, storage);
      if (m > 0 &&
          (double)fees_out[t] / (double)m <= feerate_threshold) {
        tx_codes_out[t] = KV_ERR_FEERATE_TOO_LOW;
        fees_out[t] = 0;
      }
    }
  }
  return 0;
}

/* ---------------- merkle root + body-in-isolation batch ----------------
 * ⇔ validate_body_in_isolation (consensus/src/pipeline/body_processor/
 * body_validation_in_isolation.rs): calc_hash_merkle_root (:38 via
 * consensus/core/src/merkle.rs:5 → crypto/merkle/src/lib.rs:31-52) over the
 * GPU-computed tx hashes, plus check_duplicate_transactions (:152),
 * check_block_double_spends (:126) and check_no_chained_transactions (:136).
 * The leaf hashes (one keyed blake2b per tx over the full serialization) run
 * on device; the log-depth fold and the set checks run on the host pool. */
extern "C" int kv_block_body_check(kv_ctx *ctx, const uint8_t *blob,
                                   size_t blob_len, uint8_t merkle_root_out[32],
                                   int32_t *rule_code_out) {
  if (!ctx) {
    set_error("kv_block_body_check: null ctx");
    return -1;
  }
  std::lock_guard<std::mutex> lk(ctx->mu);
  (void)hipGetLastError(); /* clear any stale per-thread error so the
      launch-config checks below only see THIS call's launches */
  ValidateBufs &vb = vb_of(ctx);
  vector<HTx> txs;
  int n_txs = parse_blob_host(blob, blob_len, txs);
  if (n_txs < 0) {
    set_error("kv_block_body_check: malformed blob");
    return -1;
  }
  /* leaf hashes on device */
  std::vector<uint8_t> hashes((size_t)n_txs * 32);
  if (n_txs > 0) {
    if (vb.blob.ensure(blob_len) || vb.tx_hashes.ensure((size_t)n_txs * 32))
      return -2;
    HIP_CHECK(hipMemcpyAsync(vb.blob.p, blob, blob_len, hipMemcpyHostToDevice,
                             ctx->stream));
    hipLaunchKernelGGL(kv::kv_tx_hash_kernel, dim3((n_txs + 255) / 256),
                       dim3(256), 0, ctx->stream, (const uint8_t *)vb.blob.p,
                       (uint32_t)n_txs, (uint8_t *)vb.tx_hashes.p);
    HIP_CHECK(hipGetLastError());
    HIP_CHECK(hipMemcpyAsync(hashes.data(), vb.tx_hashes.p,
                             (size_t)n_txs * 32, hipMemcpyDeviceToHost,
                             ctx->stream));
    HIP_CHECK(hipStreamSynchronize(ctx->stream));
  }
  /* merkle fold (crypto/merkle/src/lib.rs:31-52): pad to a power of two,
   * absent left → absent parent, absent right → ZERO_HASH */
  if (merkle_root_out) {
    if (n_txs == 0) {
      memset(merkle_root_out, 0, 32);
    } else if (n_txs == 1) {
      memcpy(merkle_root_out, hashes.data(), 32);
    } else {
      size_t pot = 1;
      while (pot < (size_t)n_txs) pot <<= 1;
      std::vector<uint8_t> cur(pot * 32, 0), pres(pot, 0);
      memcpy(cur.data(), hashes.data(), (size_t)n_txs * 32);
      for (int i = 0; i < n_txs; i++) pres[i] = 1;
      static const uint8_t ZERO[32] = {0};
      size_t width = pot;
      while (width > 1) {
        kvh_parallel_for((uint32_t)(width / 2), [&](uint32_t o) {
          size_t i = 2 * (size_t)o;
          if (!pres[i]) {
            pres[o] = 0;
            return;
          }
          uint8_t buf[64];
          memcpy(buf, cur.data() + i * 32, 32);
          memcpy(buf + 32, pres[i + 1] ? cur.data() + (i + 1) * 32 : ZERO, 32);
          uint8_t h[32];
          static const uint8_t MKEY[] = "MerkleBranchHash";
          h_blake2b_keyed(MKEY, sizeof(MKEY) - 1, buf, 64, h);
          memcpy(cur.data() + o * 32, h, 32);
          pres[o] = 1;
        });
        width /= 2;
      }
      memcpy(merkle_root_out, cur.data(), 32);
    }
  }
  /* body rule checks (first violation wins, reference order) */
  int32_t code = 0;
  {
    std::vector<std::array<uint8_t, 32>> ids(n_txs);
    for (int t = 0; t < n_txs; t++) memcpy(ids[t].data(), txs[t].tx_id, 32);
    std::sort(ids.begin(), ids.end());
    for (int t = 0; t + 1 < n_txs; t++)
      if (ids[t] == ids[t + 1]) {
        code = KV_ERR_BODY_DUP_TX;
        break;
      }
  }
  if (!code) {
    std::vector<std::array<uint8_t, 36>> ops;
    for (auto &tx : txs)
      for (auto &in : tx.inputs) {
        std::array<uint8_t, 36> o;
        memcpy(o.data(), in.prev_tx_id, 32);
        memcpy(o.data() + 32, &in.prev_index, 4);
        ops.push_back(o);
      }
    std::sort(ops.begin(), ops.end());
    for (size_t i = 0; i + 1 < ops.size(); i++)
      if (ops[i] == ops[i + 1]) {
        code = KV_ERR_BODY_DOUBLE_SPEND;
        break;
      }
    if (!code) {
      std::vector<std::array<uint8_t, 36>> created;
      for (auto &tx : txs)
        for (uint32_t i = 0; i < tx.outputs.size(); i++) {
          std::array<uint8_t, 36> o;
          memcpy(o.data(), tx.tx_id, 32);
          memcpy(o.data() + 32, &i, 4);
          created.push_back(o);
        }
      std::sort(created.begin(), created.end());
      for (auto &tx : txs) {
        for (auto &in : tx.inputs) {
          std::array<uint8_t, 36> key;
          memcpy(key.data(), in.prev_tx_id, 32);
          memcpy(key.data() + 32, &in.prev_index, 4);
          if (std::binary_search(created.begin(), created.end(), key)) {
            code = KV_ERR_BODY_CHAINED;
            break;
          }
        }
        if (code) break;
      }
    }
  }
  if (rule_code_out) *rule_code_out = code;
  return 0;
}

/* ---------------- populate + validate + diff-apply ----------------
 * ⇔ the virtual processor's populate step (utxo_validation.rs:351-390:
 * each input's UtxoEntry is resolved from the virtual UTXO set; a missing
 * outpoint fails the tx) followed by validation and utxo_diff application
 * (utxo_diff.rs:224 add_transaction: remove spent, add created).
 *
 * The incoming blob uses the same format but every input's UtxoEntry fields
 * are ignored (builders write zeros, utxo_spk_len = 0). Entries come from the
 * GPU-resident table (kv_utxo_reset/upsert); the engine rebuilds a populated
 * blob internally (one linear pass — the entry fields are the only part that
 * moves) and runs the standard pipeline on it. */
/* caller holds ctx->mu. Resolves every input's entry from the GPU table into
 * a rebuilt populated blob (ctx->pop_buf); missing outpoints pre-fail their tx
 * with KV_ERR_MISSING_OUTPOINT in pre_codes. */
static int populate_blob_nolock(kv_ctx *ctx, const uint8_t *blob, size_t blob_len,
                                vector<HTx> &txs, int n_txs,
                                std::vector<int32_t> &pre_codes) {

  /* populate: one GPU lookup over every input's outpoint. The gather and the
   * blob rebuild fan over the host pool (per-tx prefix-summed offsets). */
  std::vector<size_t> in_base(n_txs + 1, 0);
  for (int t = 0; t < n_txs; t++)
    in_base[t + 1] = in_base[t] + txs[t].inputs.size();
  size_t n_in_total = in_base[n_txs];
  std::vector<uint8_t> &ops = ctx->ops_buf;
  ops.resize(n_in_total * 36);
  kvh_parallel_for((uint32_t)n_txs, [&](uint32_t t) {
    uint8_t *dst = ops.data() + in_base[t] * 36;
    for (auto &in : txs[t].inputs) {
      memcpy(dst, in.prev_tx_id, 32);
      memcpy(dst + 32, &in.prev_index, 4);
      dst += 36;
    }
  });
  std::vector<uint8_t> &entries = ctx->ent_buf;
  entries.resize(n_in_total * 64);
  std::vector<uint64_t> &found = ctx->found_buf;
  found.assign((n_in_total + 63) / 64, 0);
  const bool kv_timing = getenv("KV_TIMING") != nullptr;
  auto tnow = []() { return std::chrono::steady_clock::now(); };
  auto tms = [](std::chrono::steady_clock::time_point a,
                std::chrono::steady_clock::time_point b) {
    return std::chrono::duration<double, std::milli>(b - a).count();
  };
  auto t_a = tnow();
  std::vector<uint8_t> gathered; /* out-of-line spk bytes, entry order */
  if (n_in_total) {
    int rc = utxo_lookup_nolock(ctx, ops.data(), n_in_total, entries.data(),
                                found.data(), nullptr);
    if (rc) return rc;
    rc = arena_gather_nolock(
        ctx, n_in_total,
        [&](size_t i) -> uint8_t * {
          int hit = (found[i / 64] >> (i % 64)) & 1;
          return hit ? entries.data() + i * 64 : nullptr;
        },
        gathered);
    if (rc) return rc;
  }
  auto t_b = tnow();

  /* rebuild the blob with populated entries; pre-fail txs with missing inputs */
  std::vector<size_t> new_off(n_txs + 1, 0);
  kvh_parallel_for((uint32_t)n_txs, [&](uint32_t t) {
    const HTx &tx = txs[t];
    uint32_t tx_end = ((int)t + 1 < n_txs) ? txs[t + 1].off : (uint32_t)blob_len;
    size_t sz;
    if (tx.inputs.empty()) {
      sz = tx_end - tx.off;
    } else {
      sz = tx.inputs[0].rec_off - tx.off; /* header + payload */
      size_t ii = in_base[t];
      for (auto &in : tx.inputs) {
        int hit = (found[ii / 64] >> (ii % 64)) & 1;
        uint32_t spk_len = 0;
        if (hit) memcpy(&spk_len, entries.data() + ii * 64 + 20, 4);
        else if (!pre_codes[t]) pre_codes[t] = KV_ERR_MISSING_OUTPOINT;
        sz += 52 + in.sig_script_len + 24 + spk_len;
        ii++;
      }
      uint32_t outs_start = tx.outputs.empty() ? tx_end : tx.output_offs[0];
      sz += tx_end - outs_start;
    }
    new_off[t + 1] = sz; /* size for now; prefixed below */
  });
  size_t hdr_len = 4 + 4ull * n_txs;
  new_off[0] = hdr_len;
  for (int t = 0; t < n_txs; t++) new_off[t + 1] += new_off[t];
  std::vector<uint8_t> &pop = ctx->pop_buf;
  pop.resize(new_off[n_txs]);
  memcpy(pop.data(), blob, 4);
  kvh_parallel_for((uint32_t)n_txs, [&](uint32_t t) {
    const HTx &tx = txs[t];
    uint32_t off32 = (uint32_t)new_off[t];
    memcpy(pop.data() + 4 + 4ull * t, &off32, 4);
    uint8_t *dst = pop.data() + new_off[t];
    uint32_t tx_end = ((int)t + 1 < n_txs) ? txs[t + 1].off : (uint32_t)blob_len;
    if (tx.inputs.empty()) {
      memcpy(dst, blob + tx.off, tx_end - tx.off);
      return;
    }
    size_t hp = tx.inputs[0].rec_off - tx.off;
    memcpy(dst, blob + tx.off, hp);
    dst += hp;
    size_t ii = in_base[t];
    for (auto &in : tx.inputs) {
      size_t pre = 52 + in.sig_script_len;
      memcpy(dst, blob + in.rec_off, pre);
      dst += pre;
      int hit = (found[ii / 64] >> (ii % 64)) & 1;
      const uint8_t *e = entries.data() + ii * 64;
      uint64_t amount = 0, daa = 0;
      uint16_t eflags = 0, spkv = 0;
      uint32_t spk_len = 0;
      if (hit) {
        memcpy(&amount, e, 8);
        memcpy(&daa, e + 8, 8);
        memcpy(&eflags, e + 16, 2);
        memcpy(&spkv, e + 18, 2);
        memcpy(&spk_len, e + 20, 4);
      }
      memcpy(dst, &amount, 8);
      memcpy(dst + 8, &daa, 8);
      dst[16] = (uint8_t)(eflags & 1);
      dst[17] = 0; /* covenants out of round-1 scope */
      memcpy(dst + 18, &spkv, 2);
      memcpy(dst + 20, &spk_len, 4);
      dst += 24;
      if (spk_len) {
        if (eflags & KV_UTXO_F_SPK_ARENA) {
          uint32_t go;
          memcpy(&go, e + 24, 4);
          memcpy(dst, gathered.data() + go, spk_len);
        } else {
          memcpy(dst, e + 24, spk_len);
        }
        dst += spk_len;
      }
      ii++;
    }
    uint32_t outs_start = tx.outputs.empty() ? tx_end : tx.output_offs[0];
    memcpy(dst, blob + outs_start, tx_end - outs_start);
  });

  if (kv_timing)
    fprintf(stderr, "[kv_timing] utxo populate: lookup %.2fms rebuild %.2fms\n",
            tms(t_a, t_b), tms(t_b, tnow()));
  return 0;
}

extern "C" int kv_validate_block_utxo(kv_ctx *ctx, const uint8_t *blob,
                                      size_t blob_len, uint64_t pov_daa_score,
                                      uint64_t block_daa_score, uint32_t flags,
                                      int apply_diff, int32_t *tx_codes_out,
                                      uint64_t *fees_out,
                                      uint8_t *muhash_partial_out) {
  if (!ctx) {
    set_error("kv_validate_block_utxo: null ctx");
    return -1;
  }
  std::lock_guard<std::mutex> lk(ctx->mu);
  (void)hipGetLastError(); /* clear any stale per-thread error so the
      launch-config checks below only see THIS call's launches */
  if (!ctx->d_utxo) {
    set_error("kv_validate_block_utxo: call kv_utxo_reset first");
    return -1;
  }
  vector<HTx> txs;
  int n_txs = parse_blob_host(blob, blob_len, txs);
  if (n_txs < 0) {
    set_error("kv_validate_block_utxo: malformed blob");
    return -1;
  }
  std::vector<int32_t> pre_codes(n_txs, 0);
  int rc = populate_blob_nolock(ctx, blob, blob_len, txs, n_txs, pre_codes);
  if (rc) return rc;
  std::vector<uint8_t> &pop = ctx->pop_buf;
  rc = validate_block_impl(ctx, pop.data(), pop.size(), pov_daa_score,
                           block_daa_score, flags, tx_codes_out, fees_out,
                           muhash_partial_out, pre_codes.data());
  if (rc || !apply_diff) return rc;

  /* diff apply for accepted txs: remove spent, upsert created */
  std::vector<uint8_t> del_ops, add_ops, add_ents, add_spk;
  for (int t = 0; t < n_txs; t++) {
    const HTx &tx = txs[t];
    if (tx_codes_out[t] != 0) continue;
    for (auto &in : tx.inputs) {
      uint8_t op[36];
      memcpy(op, in.prev_tx_id, 32);
      memcpy(op + 32, &in.prev_index, 4);
      del_ops.insert(del_ops.end(), op, op + 36);
    }
    for (uint32_t i = 0; i < tx.outputs.size(); i++) {
      const HOutput &o = tx.outputs[i];
      if (o.spk_len > KV_UTXO_MAX_SPK) {
        set_error("kv_validate_block_utxo: created spk exceeds KV_UTXO_MAX_SPK");
        return -4;
      }
      add_ops.insert(add_ops.end(), tx.tx_id, tx.tx_id + 32);
      const uint8_t *ix = (const uint8_t *)&i;
      add_ops.insert(add_ops.end(), ix, ix + 4);
      uint8_t e[64] = {0};
      memcpy(e, &o.value, 8);
      memcpy(e + 8, &block_daa_score, 8);
      /* flags: coinbase txs never enter this path → bit0 = 0 */
      memcpy(e + 18, &o.spk_version, 2);
      memcpy(e + 20, &o.spk_len, 4);
      if (o.spk_len <= 36)
        memcpy(e + 24, o.spk, o.spk_len);
      else /* out-of-line: bytes go to the arena via upsert_spk below */
        add_spk.insert(add_spk.end(), o.spk, o.spk + o.spk_len);
      add_ents.insert(add_ents.end(), e, e + 64);
    }
  }
  if (!del_ops.empty()) {
    rc = utxo_remove_nolock(ctx, del_ops.data(), del_ops.size() / 36);
    if (rc) return rc;
  }
  if (!add_ops.empty()) {
    rc = utxo_upsert_spk_nolock(ctx, add_ops.data(), add_ents.data(),
                                add_spk.data(), add_spk.size(),
                                add_ops.size() / 36);
    if (rc) return rc;
  }
  return 0;
}

/* ---- direct mass-vector test exports (tests/golden/mass.json, extracted from
 * consensus/core/src/mass/mod.rs:531-953). These expose the engine's OWN mass
 * helpers (kv_validate_host.inc) so the CPU suite can pin them against the
 * reference's vectors without a GPU. ---- */

extern "C" uint64_t kv_test_plurality(uint32_t spk_len, int has_cov) {
  return kvh_plurality(spk_len, has_cov != 0);
}

extern "C" uint64_t kv_test_normalized_max(uint64_t storage_mass, uint64_t compute_mass,
                                           uint64_t transient_mass, uint64_t limit_storage,
                                           uint64_t limit_compute,
                                           uint64_t limit_transient) {
  return kvh_normalized_max_limits(storage_mass, compute_mass, transient_mass,
                                   limit_storage, limit_compute, limit_transient);
}

extern "C" int kv_test_storage_mass(uint32_t n_ins, const uint64_t *in_amounts,
                                    const uint32_t *in_spk_lens, const uint8_t *in_has_cov,
                                    uint32_t n_outs, const uint64_t *out_amounts,
                                    const uint32_t *out_spk_lens,
                                    const uint8_t *out_has_cov, uint64_t *mass_out) {
  HTx tx;
  tx.version = 1;
  for (uint32_t i = 0; i < n_ins; i++) {
    HInput in{};
    in.utxo_amount = in_amounts[i];
    in.utxo_spk_len = in_spk_lens ? in_spk_lens[i] : 0;
    in.utxo_has_cov = (in_has_cov && in_has_cov[i]) ? 1 : 0;
    tx.inputs.push_back(in);
  }
  for (uint32_t i = 0; i < n_outs; i++) {
    HOutput o{};
    o.value = out_amounts[i];
    o.spk_len = out_spk_lens ? out_spk_lens[i] : 0;
    o.has_covenant = (out_has_cov && out_has_cov[i]) ? 1 : 0;
    tx.outputs.push_back(o);
  }
  return kvh_storage_mass(tx, /*is_coinbase=*/false, mass_out);
}
